// pybind11 bindings for the KungFu-AMD C++ runtime (`kungfu_amd._core`).
//
// Reference parity: the cgo bridge srcs/go/libkungfu-comm/ + the Python C API
// srcs/cpp/src/python/c_api.cpp (rank/size/barrier/resize/propose/save/
// request/collectives). Buffers are passed as raw data pointers
// (tensor.data_ptr()) for zero-copy in/out of the C++ engine; blocking calls
// release the GIL.
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <chrono>
#include <cstring>
#include <deque>
#include <future>
#include <map>
#include <memory>
#include <mutex>
#include <tuple>

#include "../core/control_api.h"
#include "../core/graph.hpp"
#include "../core/plan.hpp"
#include "../peer/peer.hpp"

namespace py = pybind11;
using namespace kf;

namespace {

std::unique_ptr<Peer> g_peer;
std::mutex g_mu;

// Async collective handles (reference: HandleManager int-handle wait map,
// include/kungfu/utils/handler_manager.hpp + all_reduce_cuda_async).
// Each async op runs on its own thread; wait() joins and rethrows.
struct AsyncOps {
    std::mutex mu;
    uint64_t next = 1;
    std::map<uint64_t, std::future<void>> ops;

    uint64_t launch(std::function<void()> fn)
    {
        std::lock_guard<std::mutex> lk(mu);
        const uint64_t h = next++;
        ops[h] = std::async(std::launch::async, std::move(fn));
        return h;
    }
    void wait(uint64_t h)
    {
        std::future<void> f;
        {
            std::lock_guard<std::mutex> lk(mu);
            auto it = ops.find(h);
            if (it == ops.end())
                throw std::runtime_error("unknown async handle");
            f = std::move(it->second);
            ops.erase(it);
        }
        f.get();  // rethrows the op's exception, if any
    }
};
AsyncOps g_async;

Peer &peer()
{
    if (!g_peer) throw std::runtime_error("kungfu not initialized");
    return *g_peer;
}

// ---- control-plane C API (consumed by kungfu_amd._rccl) ----
// The RCCL layer bootstraps its communicators over THIS control plane
// (uniqueId broadcast, scheduler order agreement) instead of a parallel
// TCPStore rendezvous; see csrc/core/control_api.h.
extern "C" {
static Peer *cap_peer(void *ctx) { return static_cast<Peer *>(ctx); }
static int cap_rank(void *c) { return cap_peer(c)->rank(); }
static int cap_size(void *c) { return cap_peer(c)->size(); }
static int cap_local_rank(void *c) { return cap_peer(c)->local_rank(); }
static int cap_local_size(void *c) { return cap_peer(c)->local_size(); }
static int cap_host_count(void *c) { return cap_peer(c)->host_count(); }
static int cap_host_rank(void *c) { return cap_peer(c)->host_rank(); }
static uint32_t cap_version(void *c) { return cap_peer(c)->version(); }
static void cap_broadcast(void *c, void *buf, size_t len, int root,
                          const char *name)
{
    Workspace w;
    w.send = buf;
    w.recv = buf;
    w.count = len;
    w.dt = DType::U8;
    w.name = name;
    cap_peer(c)->session().broadcast(w, root);
}
static void cap_local_broadcast(void *c, void *buf, size_t len,
                                const char *name)
{
    Workspace w;
    w.send = buf;
    w.recv = buf;
    w.count = len;
    w.dt = DType::U8;
    w.name = name;
    cap_peer(c)->session().local_broadcast(w);
}
static void cap_barrier(void *c) { cap_peer(c)->session().barrier(); }
}  // extern "C"

kf_control_api g_control_api;

// Runtime tracing (reference: stdtracer TRACE_SCOPE, KUNGFU_ENABLE_TRACE;
// include/kungfu/utils/trace.hpp): env-gated event ring readable from
// python as chrome-trace-able tuples.
struct Tracer {
    bool enabled = false;
    std::mutex mu;
    std::deque<std::tuple<std::string, double, double>> events;  // us
    std::chrono::steady_clock::time_point t0 =
        std::chrono::steady_clock::now();

    static Tracer &get()
    {
        static Tracer t;
        static bool init = [] {
            const char *e = std::getenv("KUNGFU_ENABLE_TRACE");
            t.enabled = e && *e && std::string(e) != "0";
            return true;
        }();
        (void)init;
        return t;
    }
    void record(const char *what, double start_us, double dur_us)
    {
        std::lock_guard<std::mutex> lk(mu);
        if (events.size() > 100000) events.pop_front();
        events.emplace_back(what, start_us, dur_us);
    }
};

// RAII guard around blocking collective calls: stall detection + tracing
struct StallGuard {
    StallDetector *d = nullptr;
    uint64_t id = 0;
    const char *what_;
    std::chrono::steady_clock::time_point start_;
    explicit StallGuard(const char *what) : what_(what)
    {
        if (g_peer && g_peer->stall_detector()) {
            d = g_peer->stall_detector();
            id = d->enter(what);
        }
        if (Tracer::get().enabled)
            start_ = std::chrono::steady_clock::now();
    }
    ~StallGuard()
    {
        if (d) d->leave(id);
        auto &t = Tracer::get();
        if (t.enabled) {
            auto now = std::chrono::steady_clock::now();
            t.record(
                what_,
                std::chrono::duration<double, std::micro>(start_ - t.t0)
                    .count(),
                std::chrono::duration<double, std::micro>(now - start_)
                    .count());
        }
    }
};

Workspace make_ws(uintptr_t send, uintptr_t recv, size_t count, int dtype,
                  int op, const std::string &name)
{
    Workspace w;
    w.send = (const void *)send;
    w.recv = (void *)recv;
    w.count = count;
    w.dt = (DType)dtype;
    w.op = (ReduceOp)op;
    w.name = name;
    return w;
}

// Prim MST over a dense symmetric weight matrix (reference:
// srcs/cpp/include/kungfu/mst.hpp:10-58). Returns the parent array
// (parent[root] = root) usable with Session::set_tree.
std::vector<int> prim_mst(const std::vector<double> &w, int n)
{
    if ((int)w.size() != n * n) throw std::runtime_error("bad matrix");
    std::vector<int> parent(n, 0);
    std::vector<double> dist(n, 1e300);
    std::vector<bool> in(n, false);
    parent[0] = 0;
    dist[0] = 0;
    for (int it = 0; it < n; ++it) {
        int u = -1;
        for (int i = 0; i < n; ++i) {
            if (!in[i] && (u < 0 || dist[i] < dist[u])) u = i;
        }
        in[u] = true;
        for (int v = 0; v < n; ++v) {
            if (!in[v] && w[u * n + v] < dist[v]) {
                dist[v] = w[u * n + v];
                parent[v] = u;
            }
        }
    }
    return parent;
}

// Control-plane server for the launcher's watch/elastic mode: queues
// ("update", stage-json) frames from workers (reference runner/handler.go).
class RunnerServer {
  public:
    RunnerServer(const std::string &self_spec, bool use_unix)
        : self_(PeerID::parse(self_spec)),
          server_(std::make_unique<Server>(self_, use_unix))
    {
        server_->start(
            [this](const Handshake &hs, Frame &f, Conn &) {
                std::lock_guard<std::mutex> lk(mu_);
                q_.emplace_back(f.name,
                                std::string((const char *)f.data.data(),
                                            f.data.size()));
                cv_.notify_all();
                (void)hs;
            },
            [](uint32_t) { return true; });
    }
    // returns (name, payload) or None after timeout
    py::object poll(int timeout_ms)
    {
        std::pair<std::string, std::string> item;
        bool found = false;
        {
            py::gil_scoped_release rel;  // wait without the GIL
            std::unique_lock<std::mutex> lk(mu_);
            if (cv_.wait_for(lk, std::chrono::milliseconds(timeout_ms),
                             [this] { return !q_.empty(); })) {
                item = std::move(q_.front());
                q_.pop_front();
                found = true;
            }
        }
        if (!found) return py::none();
        return py::make_tuple(item.first, py::bytes(item.second));
    }
    void stop() { server_->stop(); }

  private:
    PeerID self_;
    std::unique_ptr<Server> server_;
    std::mutex mu_;
    std::condition_variable cv_;
    std::deque<std::pair<std::string, std::string>> q_;
};

}  // namespace

PYBIND11_MODULE(_core, m)
{
    m.doc() = "KungFu-AMD native runtime (control plane + CPU collectives)";

    // ---- lifecycle ----
    m.def("init", [] {
        std::lock_guard<std::mutex> lk(g_mu);
        if (g_peer) return;
        auto cfg = parse_env_config();
        g_peer = std::make_unique<Peer>(cfg);
        py::gil_scoped_release rel;
        g_peer->start();
    });
    m.def("finalize", [] {
        std::lock_guard<std::mutex> lk(g_mu);
        if (!g_peer) return;
        py::gil_scoped_release rel;
        g_peer.reset();
    });
    m.def("initialized", [] { return (bool)g_peer; });
    // Capsule handing the control plane to the RCCL layer (_rccl) for
    // communicator-id rendezvous and order agreement. Valid until
    // finalize(); _rccl.finalize() must run first (kungfu_amd.finalize
    // orders this).
    m.def("control_api", [] {
        if (!g_peer) throw std::runtime_error("kungfu not initialized");
        g_control_api.api_version = KF_CONTROL_API_VERSION;
        g_control_api.ctx = g_peer.get();
        g_control_api.rank = cap_rank;
        g_control_api.size = cap_size;
        g_control_api.local_rank = cap_local_rank;
        g_control_api.local_size = cap_local_size;
        g_control_api.host_count = cap_host_count;
        g_control_api.host_rank = cap_host_rank;
        g_control_api.cluster_version = cap_version;
        g_control_api.broadcast = cap_broadcast;
        g_control_api.local_broadcast = cap_local_broadcast;
        g_control_api.barrier = cap_barrier;
        return py::capsule(&g_control_api, KF_CONTROL_API_CAPSULE);
    });

    // ---- metadata ----
    m.def("rank", [] { return peer().rank(); });
    m.def("size", [] { return peer().size(); });
    m.def("local_rank", [] { return peer().local_rank(); });
    m.def("local_size", [] { return peer().local_size(); });
    m.def("host_count", [] { return peer().host_count(); });
    m.def("uid", [] { return peer().uid(); });
    m.def("detached", [] { return peer().detached(); });
    m.def("cluster_version", [] { return peer().version(); });

    // ---- collectives (blocking; GIL released) ----
    m.def("barrier", [] {
        py::gil_scoped_release rel;
        StallGuard sg("barrier");
        peer().session().barrier();
    });
    m.def("all_reduce",
          [](uintptr_t s, uintptr_t r, size_t count, int dt, int op,
             const std::string &name) {
              auto w = make_ws(s, r, count, dt, op, name);
              py::gil_scoped_release rel;
              StallGuard sg("all_reduce");
              peer().session().all_reduce(w);
          });
    m.def("reduce",
          [](uintptr_t s, uintptr_t r, size_t count, int dt, int op,
             const std::string &name) {
              auto w = make_ws(s, r, count, dt, op, name);
              py::gil_scoped_release rel;
              peer().session().reduce(w);
          });
    m.def("broadcast",
          [](uintptr_t s, uintptr_t r, size_t count, int dt,
             const std::string &name, int root) {
              auto w = make_ws(s, r, count, dt, 0, name);
              py::gil_scoped_release rel;
              StallGuard sg("broadcast");
              peer().session().broadcast(w, root);
          });
    m.def("all_gather",
          [](uintptr_t s, uintptr_t r, size_t count, int dt,
             const std::string &name) {
              auto w = make_ws(s, r, count, dt, 0, name);
              py::gil_scoped_release rel;
              peer().session().all_gather(w);
          });
    m.def("gather",
          [](uintptr_t s, uintptr_t r, size_t count, int dt,
             const std::string &name) {
              auto w = make_ws(s, r, count, dt, 0, name);
              py::gil_scoped_release rel;
              peer().session().gather(w);
          });
    m.def("local_reduce",
          [](uintptr_t s, uintptr_t r, size_t count, int dt, int op,
             const std::string &name) {
              auto w = make_ws(s, r, count, dt, op, name);
              py::gil_scoped_release rel;
              peer().session().local_reduce(w);
          });
    m.def("local_broadcast",
          [](uintptr_t s, uintptr_t r, size_t count, int dt,
             const std::string &name) {
              auto w = make_ws(s, r, count, dt, 0, name);
              py::gil_scoped_release rel;
              peer().session().local_broadcast(w);
          });
    m.def("cross_all_reduce",
          [](uintptr_t s, uintptr_t r, size_t count, int dt, int op,
             const std::string &name) {
              auto w = make_ws(s, r, count, dt, op, name);
              py::gil_scoped_release rel;
              peer().session().cross_all_reduce(w);
          });
    // ---- async variants (returns a handle; wait(handle) joins) ----
    m.def("all_reduce_async",
          [](uintptr_t sp, uintptr_t r, size_t count, int dt, int op,
             const std::string &name) {
              auto w = make_ws(sp, r, count, dt, op, name);
              return g_async.launch([w] {
                  StallGuard sg("all_reduce_async");
                  peer().session().all_reduce(w);
              });
          });
    m.def("broadcast_async",
          [](uintptr_t sp, uintptr_t r, size_t count, int dt,
             const std::string &name, int root) {
              auto w = make_ws(sp, r, count, dt, 0, name);
              return g_async.launch(
                  [w, root] { peer().session().broadcast(w, root); });
          });
    m.def("all_gather_async",
          [](uintptr_t sp, uintptr_t r, size_t count, int dt,
             const std::string &name) {
              auto w = make_ws(sp, r, count, dt, 0, name);
              return g_async.launch(
                  [w] { peer().session().all_gather(w); });
          });
    m.def("wait_handle", [](uint64_t h) {
        py::gil_scoped_release rel;
        g_async.wait(h);
    });

    m.def("consensus", [](py::bytes data, const std::string &name) {
        std::string s = data;
        py::gil_scoped_release rel;
        return peer().session().consensus(s.data(), s.size(), name);
    });

    // ---- P2P model store ----
    m.def("save", [](const std::string &name, uintptr_t data, size_t len) {
        py::gil_scoped_release rel;
        peer().save(name, (const void *)data, len);
    });
    m.def("request",
          [](int target, const std::string &name, uintptr_t dst,
             size_t len) {
              py::gil_scoped_release rel;
              return peer().request(target, name, (void *)dst, len);
          });

    // ---- elastic ----
    m.def("propose_new_size", [](int n) {
        py::gil_scoped_release rel;
        return peer().propose_new_size(n);
    });
    m.def("resize_cluster_from_url", [] {
        py::gil_scoped_release rel;
        auto rr = peer().resize_cluster_from_url();
        return std::make_pair(rr.changed, rr.detached);
    });
    m.def("resize", [](int n) {
        py::gil_scoped_release rel;
        auto rr = peer().resize(n);
        return std::make_pair(rr.changed, rr.detached);
    });

    // ---- adaptation / monitoring ----
    m.def("all_reduce_with",
          [](const std::vector<int> &parent, uintptr_t sp, uintptr_t r,
             size_t count, int dt, int op, const std::string &name) {
              auto w = make_ws(sp, r, count, dt, op, name);
              py::gil_scoped_release rel;
              StallGuard sg("all_reduce_with");
              peer().session().all_reduce_with(parent, w);
          });
    m.def("set_tree", [](const std::vector<int> &parent) {
        py::gil_scoped_release rel;
        peer().session().set_tree(parent);
    });
    m.def("set_strategy", [](const std::string &name) {
        auto s = strategy_from_name(name);
        py::gil_scoped_release rel;
        peer().session().set_strategy(s);
    });
    m.def("get_strategy",
          [] { return strategy_name(peer().session().strategy()); });
    m.def("peer_latencies_us", [] {
        py::gil_scoped_release rel;
        return peer().peer_latencies_us();
    });
    m.def("prim_mst", &prim_mst);
    m.def("strategy_stats", [] {
        auto st = peer().session().stats();
        py::list out;
        for (auto &s : st) {
            py::dict d;
            d["ops"] = s.ops;
            d["bytes"] = s.bytes;
            d["seconds"] = s.seconds;
            d["throughput"] = s.throughput();
            out.append(d);
        }
        return out;
    });
    m.def("reset_strategy_stats",
          [] { peer().session().reset_stats(); });
    m.def("check_interference", [](double ratio) {
        return peer().session().check_interference(ratio);
    });
    m.def("trace_events", [] {
        auto &t = Tracer::get();
        std::lock_guard<std::mutex> lk(t.mu);
        py::list out;
        for (auto &e : t.events) {
            out.append(py::make_tuple(std::get<0>(e), std::get<1>(e),
                                      std::get<2>(e)));
        }
        return out;
    });
    m.def("trace_enabled", [] { return Tracer::get().enabled; });
    m.def("ingress_bytes", [] {
        std::map<std::string, uint64_t> out;
        for (auto &kv : peer().ingress_bytes()) {
            PeerID p;
            p.ipv4 = (uint32_t)(kv.first >> 16);
            p.port = (uint16_t)(kv.first & 0xffff);
            out[p.str()] = kv.second;
        }
        return out;
    });
    m.def("egress_bytes", [] {
        auto eg = peer().egress_bytes();
        py::dict d;
        for (auto &kv : eg) {
            PeerID p;
            p.ipv4 = (uint32_t)(kv.first >> 16);
            p.port = (uint16_t)(kv.first & 0xffff);
            d[py::str(p.str())] = kv.second;
        }
        return d;
    });

    // ---- launcher support ----
    py::class_<RunnerServer>(m, "RunnerServer")
        .def(py::init<const std::string &, bool>(), py::arg("self_spec"),
             py::arg("use_unix") = false)
        .def("poll", &RunnerServer::poll, py::arg("timeout_ms"))
        .def("stop", &RunnerServer::stop,
             py::call_guard<py::gil_scoped_release>());

    // ---- plan/topology helpers (unit tests + tools) ----
    m.def("topology_digest",
          [](const std::string &peers, const std::string &strategy) {
              auto pl = PeerList::parse(peers);
              auto gs = gen_strategies(pl, strategy_from_name(strategy));
              std::string out;
              for (auto &g : gs) {
                  out += g.reduce.digest() + "/" + g.bcast.digest() + "\n";
              }
              return out;
          });
    m.def("topology_edges", [](int n, const std::string &strategy,
                               const std::string &peers) {
        PeerList pl;
        if (!peers.empty()) {
            pl = PeerList::parse(peers);
        } else {
            for (int i = 0; i < n; ++i)
                pl.peers.push_back(
                    PeerID::parse("127.0.0.1:" + std::to_string(10000 + i)));
        }
        auto gs = gen_strategies(pl, strategy_from_name(strategy));
        py::list out;
        for (auto &g : gs) {
            py::list redges, bedges, roots;
            for (int i = 0; i < g.reduce.n; ++i) {
                for (int j : g.reduce.nexts[i])
                    redges.append(py::make_tuple(i, j));
                if (g.reduce.self_loop[i]) roots.append(i);
            }
            for (int i = 0; i < g.bcast.n; ++i) {
                for (int j : g.bcast.nexts[i])
                    bedges.append(py::make_tuple(i, j));
            }
            py::dict d;
            d["reduce"] = redges;
            d["bcast"] = bedges;
            d["roots"] = roots;
            out.append(d);
        }
        return out;
    });
    m.def("cluster_resize_json",
          [](const std::string &json, int new_size, int port_base) {
              auto c = Cluster::from_json(json);
              return c.resized(new_size, port_base).json();
          });
    m.def("gen_peer_list",
          [](const std::string &hosts, int np, int port_base) {
              auto hl = HostList::parse(hosts);
              return hl.gen_peer_list(np, port_base).str();
          });
    m.def("gen_runner_list", [](const std::string &hosts, int port) {
        auto hl = HostList::parse(hosts);
        return hl.gen_runner_list(port).str();
    });
}
