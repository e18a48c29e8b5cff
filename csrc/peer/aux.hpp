#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <functional>
#include <map>
#include <mutex>
#include <string>
#include <thread>

namespace kf {

// Bind the calling process to an even share of the host CPUs
// (KUNGFU_USE_AFFINITY). Returns CPUs bound, or -1.
int bind_cpu_affinity(int local_rank, int local_size);

// Watchdog that reports collective ops exceeding a time threshold
// (KUNGFU_CONFIG_ENABLE_STALL_DETECTION; reference utils/stalldetector.go).
class StallDetector {
  public:
    explicit StallDetector(double threshold_sec = 3.0);
    ~StallDetector();
    uint64_t enter(const std::string &what);
    void leave(uint64_t id);

  private:
    void watch();
    struct Op {
        std::string what;
        std::chrono::steady_clock::time_point start;
        bool reported;
    };
    double threshold_;
    std::mutex mu_;
    std::condition_variable cv_;
    std::map<uint64_t, Op> ops_;
    uint64_t next_id_ = 0;
    bool stopping_ = false;
    std::thread watcher_;
};

// Minimal HTTP server for the Prometheus-style /metrics text endpoint
// (KUNGFU_CONFIG_ENABLE_MONITORING; reference monitor/server.go).
class MetricsServer {
  public:
    MetricsServer(uint16_t port, std::function<std::string()> render);
    ~MetricsServer();
    bool ok() const { return fd_ >= 0; }

  private:
    void serve();
    std::function<std::string()> render_;
    int fd_ = -1;
    std::atomic<bool> stopping_{false};
    std::thread thread_;
};

}  // namespace kf
