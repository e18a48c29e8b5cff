// Cached worker-thread pool for chunk-level collective parallelism.
//
// Replaces the thread-per-chunk spawn in the session engine (round 1
// spawned up to 32 threads per >1 MiB collective, ~20 us each). Analog of
// Go's scheduler for the reference's goroutine-per-chunk model
// (session.go:292-317): tasks may BLOCK on socket waits, and tasks from
// different concurrent collectives may depend on each other's progress
// across ranks, so the pool must never queue a runnable task behind a
// blocked one — when no worker is idle, a new worker is spawned
// immediately (cached-pool semantics); idle workers expire after 30 s.
#pragma once

#include <condition_variable>
#include <deque>
#include <functional>
#include <mutex>
#include <thread>

namespace kf {

class CachedThreadPool {
  public:
    static CachedThreadPool &inst()
    {
        // leaked singleton: workers may outlive static destruction
        static CachedThreadPool *p = new CachedThreadPool();
        return *p;
    }

    void submit(std::function<void()> fn)
    {
        std::lock_guard<std::mutex> lk(mu_);
        q_.push_back(std::move(fn));
        if (idle_ == 0) {
            std::thread([this] { run(); }).detach();
        } else {
            cv_.notify_one();
        }
    }

  private:
    void run()
    {
        std::unique_lock<std::mutex> lk(mu_);
        for (;;) {
            if (q_.empty()) {
                ++idle_;
                const bool got = cv_.wait_for(
                    lk, std::chrono::seconds(30),
                    [this] { return !q_.empty(); });
                --idle_;
                if (!got) return;  // idle expiry
            }
            auto fn = std::move(q_.front());
            q_.pop_front();
            lk.unlock();
            fn();  // may block (socket waits); exceptions are the task's
            lk.lock();
        }
    }

    std::mutex mu_;
    std::condition_variable cv_;
    std::deque<std::function<void()>> q_;
    int idle_ = 0;
};

// Completion latch for fan-out/fan-in over the pool.
class Latch {
  public:
    explicit Latch(int n) : remaining_(n) {}
    void done()
    {
        std::lock_guard<std::mutex> lk(mu_);
        if (--remaining_ == 0) cv_.notify_all();
    }
    void wait()
    {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [this] { return remaining_ == 0; });
    }

  private:
    std::mutex mu_;
    std::condition_variable cv_;
    int remaining_;
};

}  // namespace kf
