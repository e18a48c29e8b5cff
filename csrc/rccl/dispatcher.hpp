// OrderedDispatcher: the deterministic GPU-collective serializer.
//
// Reference parity: srcs/cpp/src/nccl/scheduler.cpp (NCCLThread +
// LinearExecutor + NCCLScheduler). Every RCCL launch for one scope funnels
// through ONE worker thread, and tasks registered under a known name are
// RELEASED in an agreed cross-rank order regardless of per-rank arrival
// order — the property that makes concurrent collective sources (gradient
// buckets, hierarchical hops, gossip, noise-scale probes) safe on one
// communicator.
//
// MI355X-native divergence from the reference: rounds RECYCLE. The
// reference rebuilds a LinearExecutor every step (Reset per step, arrival
// order re-broadcast each time); here `reset(names, order)` installs a
// persistent round — when every named task of a round has been released,
// the round re-arms automatically, so the steady-state hot path pays no
// per-step reset and no control-plane traffic. Arrival order of the last
// completed round is still recorded (`last_arrival`) so callers can run
// the reference's order-agreement broadcast whenever they choose
// (scheduler.cpp:93-119 semantics, exposed as _rccl.scheduler_agree).
#pragma once

#include <condition_variable>
#include <cstdint>
#include <deque>
#include <functional>
#include <map>
#include <mutex>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace kf {

class OrderedDispatcher {
  public:
    using Task = std::function<void()>;

    OrderedDispatcher()
    {
        worker_ = std::thread([this] { this->run(); });
    }

    ~OrderedDispatcher()
    {
        {
            std::lock_guard<std::mutex> lk(mu_);
            stopping_ = true;
            qcv_.notify_all();
        }
        worker_.join();
    }

    // Install the named-task set and its release order for subsequent
    // rounds. order[i] = index into `names` of the i-th task to release.
    // Blocks until any in-flight round completes (all its names arrived).
    void reset(const std::vector<std::string> &names,
               const std::vector<int32_t> &order)
    {
        if (order.size() != names.size())
            throw std::invalid_argument("order/names size mismatch");
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [this] { return arrived_ == 0 || !have_round_; });
        slot_of_.clear();
        for (size_t i = 0; i < names.size(); ++i) {
            if (!slot_of_.emplace(names[i], (int32_t)i).second)
                throw std::invalid_argument("duplicate task name: " +
                                            names[i]);
        }
        release_order_.assign(order.begin(), order.end());
        n_ = (int32_t)names.size();
        have_round_ = n_ > 0;
        arm();
    }

    // Enqueue a task. Named tasks (name in the current round set) are
    // released in the agreed order; anonymous tasks (name == "" or not
    // part of a round) are released immediately in arrival order.
    void start(const std::string &name, Task task)
    {
        std::lock_guard<std::mutex> lk(mu_);
        if (!have_round_ || name.empty()) {
            push(std::move(task));
            return;
        }
        auto it = slot_of_.find(name);
        if (it == slot_of_.end()) {
            push(std::move(task));
            return;
        }
        const int32_t slot = it->second;
        if (pending_[slot])
            throw std::runtime_error("task '" + name +
                                     "' started twice in one round");
        pending_[slot] = true;
        tasks_[slot] = std::move(task);
        arrival_.push_back(slot);
        ++arrived_;
        // release the maximal prefix of the agreed order that has arrived
        while (next_release_ < n_ &&
               pending_[release_order_[next_release_]]) {
            const int32_t s = release_order_[next_release_];
            push(std::move(tasks_[s]));
            ++next_release_;
        }
        if (next_release_ == n_) {
            last_arrival_ = arrival_;
            arm();  // round complete: re-arm for the next one
            cv_.notify_all();
        }
    }

    // Run a task on the worker thread and wait for it (drains everything
    // queued before it).
    void run_sync(Task task)
    {
        std::mutex m;
        std::condition_variable c;
        bool done = false;
        {
            std::lock_guard<std::mutex> lk(mu_);
            push([&] {
                task();
                std::lock_guard<std::mutex> lk2(m);
                done = true;
                c.notify_all();
            });
        }
        std::unique_lock<std::mutex> lk(m);
        c.wait(lk, [&] { return done; });
    }

    // Arrival order (slot indices) of the last COMPLETED round.
    std::vector<int32_t> last_arrival() const
    {
        std::lock_guard<std::mutex> lk(mu_);
        return last_arrival_;
    }

    // Blocks until the in-flight round (if any) completes.
    void wait_round()
    {
        std::unique_lock<std::mutex> lk(mu_);
        cv_.wait(lk, [this] { return arrived_ == 0 || !have_round_; });
    }

    int round_size() const
    {
        std::lock_guard<std::mutex> lk(mu_);
        return have_round_ ? (int)n_ : 0;
    }

  private:
    void arm()
    {
        pending_.assign(n_, false);
        tasks_.assign(n_, Task());
        arrival_.clear();
        arrived_ = 0;
        next_release_ = 0;
    }

    void push(Task t)  // caller holds mu_
    {
        q_.push_back(std::move(t));
        qcv_.notify_one();
    }

    void run()
    {
        for (;;) {
            Task t;
            {
                std::unique_lock<std::mutex> lk(mu_);
                qcv_.wait(lk, [this] { return stopping_ || !q_.empty(); });
                if (q_.empty() && stopping_) return;
                t = std::move(q_.front());
                q_.pop_front();
            }
            t();  // tasks must not throw (RCCL errors are captured inside)
        }
    }

    mutable std::mutex mu_;
    std::condition_variable cv_;   // round completion
    std::condition_variable qcv_;  // queue
    std::deque<Task> q_;
    std::thread worker_;
    bool stopping_ = false;

    // round state
    bool have_round_ = false;
    int32_t n_ = 0;
    std::map<std::string, int32_t> slot_of_;
    std::vector<int32_t> release_order_;
    std::vector<bool> pending_;
    std::vector<Task> tasks_;
    std::vector<int32_t> arrival_;
    std::vector<int32_t> last_arrival_;
    int32_t arrived_ = 0;
    int32_t next_release_ = 0;
};

}  // namespace kf
