// Auxiliary subsystems: NUMA/CPU affinity, stall detection, metrics server.
//
// Reference parity:
//  - NUMA affinity: srcs/cpp/src/numa/{affinity,placement}.cpp — bind each
//    local rank to an even partition of the host CPUs (KUNGFU_USE_AFFINITY).
//    Implemented with sched_setaffinity over /proc-visible CPUs (no hwloc
//    dependency).
//  - Stall detection: srcs/go/utils/stalldetector.go — report collective
//    ops that run longer than a threshold
//    (KUNGFU_CONFIG_ENABLE_STALL_DETECTION).
//  - Metrics endpoint: srcs/go/monitor/server.go — Prometheus-style text
//    on peer port + 10000 (KUNGFU_CONFIG_ENABLE_MONITORING).
#include "aux.hpp"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <cctype>
#include <sched.h>
#include <sys/socket.h>
#include <unistd.h>

#include <chrono>
#include <cstdio>
#include <cstring>
#include <sstream>
#include <string>
#include <thread>
#include <vector>

namespace kf {

namespace {

// Parse a kernel cpulist string like "0-31,64-95" into cpu indices.
std::vector<int> parse_cpulist(const std::string &s)
{
    std::vector<int> cpus;
    size_t i = 0;
    while (i < s.size()) {
        size_t j = s.find(',', i);
        std::string part = s.substr(i, j == std::string::npos ? j : j - i);
        size_t dash = part.find('-');
        try {
            if (dash == std::string::npos) {
                if (!part.empty() && isdigit((unsigned char)part[0]))
                    cpus.push_back(std::stoi(part));
            } else {
                int a = std::stoi(part.substr(0, dash));
                int b = std::stoi(part.substr(dash + 1));
                for (int c = a; c <= b; ++c) cpus.push_back(c);
            }
        } catch (...) {
        }
        if (j == std::string::npos) break;
        i = j + 1;
    }
    return cpus;
}

// NUMA-domain CPU lists from sysfs, in node order (the reference uses
// hwloc PU ordering, srcs/cpp/src/numa/affinity.cpp:26-63; sysfs gives
// the same physical grouping without the library).
std::vector<std::vector<int>> numa_nodes()
{
    std::vector<std::vector<int>> nodes;
    for (int n = 0; n < 64; ++n) {
        std::string path = "/sys/devices/system/node/node" +
                           std::to_string(n) + "/cpulist";
        FILE *f = std::fopen(path.c_str(), "r");
        if (!f) break;
        char buf[4096];
        std::string s;
        if (std::fgets(buf, sizeof(buf), f)) s = buf;
        std::fclose(f);
        auto cpus = parse_cpulist(s);
        if (!cpus.empty()) nodes.push_back(std::move(cpus));
    }
    return nodes;
}

}  // namespace

int bind_cpu_affinity(int local_rank, int local_size)
{
    const long ncpu = sysconf(_SC_NPROCESSORS_ONLN);
    if (ncpu <= 0 || local_size <= 0) return -1;
    cpu_set_t set;
    CPU_ZERO(&set);
    // NUMA-aware placement: ranks are spread across NUMA domains and each
    // rank's CPU slice stays INSIDE one domain (an even global partition
    // interleaves wrongly across SMT/NUMA boundaries on a real MI355X
    // host). Falls back to the even partition when sysfs has no topology.
    const auto nodes = numa_nodes();
    int bound = 0;
    if (nodes.size() > 1) {
        const int nn = (int)nodes.size();
        const int node = local_rank % nn;
        // ranks sharing a node split its cpulist evenly
        const int ranks_here =
            local_size / nn + (local_rank % nn < local_size % nn ? 1 : 0);
        const int sub = local_rank / nn;  // index among ranks on this node
        const auto &cpus = nodes[node];
        const int per =
            ranks_here > 0 ? (int)cpus.size() / ranks_here : (int)cpus.size();
        if (per > 0) {
            for (int i = sub * per;
                 i < (sub + 1) * per && i < (int)cpus.size(); ++i) {
                CPU_SET(cpus[i], &set);
                ++bound;
            }
        }
    }
    if (bound == 0) {
        const long per = ncpu / local_size > 0 ? ncpu / local_size : 1;
        const long lo = (local_rank % local_size) * per;
        for (long c = lo; c < lo + per && c < ncpu; ++c) {
            CPU_SET((int)c, &set);
            ++bound;
        }
    }
    if (sched_setaffinity(0, sizeof(set), &set) != 0) return -1;
    return bound;
}

// ---------- StallDetector ----------

StallDetector::StallDetector(double threshold_sec)
    : threshold_(threshold_sec)
{
    watcher_ = std::thread([this] { watch(); });
}

StallDetector::~StallDetector()
{
    {
        std::lock_guard<std::mutex> lk(mu_);
        stopping_ = true;
        cv_.notify_all();
    }
    if (watcher_.joinable()) watcher_.join();
}

uint64_t StallDetector::enter(const std::string &what)
{
    std::lock_guard<std::mutex> lk(mu_);
    const uint64_t id = next_id_++;
    ops_[id] = {what, std::chrono::steady_clock::now(), false};
    return id;
}

void StallDetector::leave(uint64_t id)
{
    std::lock_guard<std::mutex> lk(mu_);
    auto it = ops_.find(id);
    if (it != ops_.end()) {
        if (it->second.reported) {
            const double sec =
                std::chrono::duration<double>(
                    std::chrono::steady_clock::now() - it->second.start)
                    .count();
            std::fprintf(stderr, "[kungfu] %s recovered after %.1fs\n",
                         it->second.what.c_str(), sec);
        }
        ops_.erase(it);
    }
}

void StallDetector::watch()
{
    std::unique_lock<std::mutex> lk(mu_);
    while (!stopping_) {
        cv_.wait_for(lk, std::chrono::seconds(1));
        if (stopping_) return;
        const auto now = std::chrono::steady_clock::now();
        for (auto &kv : ops_) {
            const double sec =
                std::chrono::duration<double>(now - kv.second.start)
                    .count();
            if (sec > threshold_ && !kv.second.reported) {
                kv.second.reported = true;
                std::fprintf(stderr,
                             "[kungfu] %s stalled for %.1fs\n",
                             kv.second.what.c_str(), sec);
            }
        }
    }
}

// ---------- MetricsServer ----------

MetricsServer::MetricsServer(uint16_t port,
                             std::function<std::string()> render)
    : render_(std::move(render))
{
    fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (fd_ < 0) return;
    int one = 1;
    ::setsockopt(fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    addr.sin_port = htons(port);
    if (::bind(fd_, (sockaddr *)&addr, sizeof(addr)) != 0 ||
        ::listen(fd_, 16) != 0) {
        ::close(fd_);
        fd_ = -1;
        return;
    }
    thread_ = std::thread([this] { serve(); });
}

MetricsServer::~MetricsServer()
{
    stopping_ = true;
    if (fd_ >= 0) {
        ::shutdown(fd_, SHUT_RDWR);
        ::close(fd_);
    }
    if (thread_.joinable()) thread_.join();
}

void MetricsServer::serve()
{
    while (!stopping_) {
        int c = ::accept(fd_, nullptr, nullptr);
        if (c < 0) {
            if (stopping_) return;
            continue;
        }
        char buf[2048];
        ::recv(c, buf, sizeof(buf), 0);  // drain the request line
        std::string body = render_ ? render_() : "";
        std::ostringstream resp;
        resp << "HTTP/1.0 200 OK\r\nContent-Type: text/plain\r\n"
             << "Content-Length: " << body.size() << "\r\n\r\n"
             << body;
        std::string r = resp.str();
        ::send(c, r.data(), r.size(), MSG_NOSIGNAL);
        ::close(c);
    }
}

}  // namespace kf
