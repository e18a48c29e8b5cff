// pybind11 bindings for the native RCCL layer (`kungfu_amd._rccl`).
//
// Replaces the round-1 torch.distributed GPU path: communicators are
// bootstrapped over the framework's own control plane (capsule from
// kungfu_amd._core, reference gpu_collective.cpp:169-191), ops are
// stream-ordered on dedicated comm streams, and every launch funnels
// through a per-scope OrderedDispatcher (reference scheduler.cpp).
//
// Tensors cross this boundary as raw device pointers + dtype codes and HIP
// streams as integers (same convention as kungfu_amd._hip) — no torch
// headers, no hipify.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <array>
#include <memory>
#include <mutex>

#include "rccl_layer.hpp"

namespace py = pybind11;
using namespace kf;

namespace {

struct Layer {
    std::array<std::unique_ptr<Controller>, 3> scopes;
    const kf_control_api *api = nullptr;
    int device = -1;
    bool gpu = false;
    std::mutex mu;

    Controller &scope(int s)
    {
        if (s < 0 || s > 2 || !scopes[s])
            throw std::runtime_error("scope not initialized");
        return *scopes[s];
    }
};

Layer g;

const kf_control_api *cap_api(py::capsule cap)
{
    auto *api = static_cast<const kf_control_api *>(cap.get_pointer());
    if (!api || api->api_version != KF_CONTROL_API_VERSION)
        throw std::runtime_error("bad control_api capsule");
    return api;
}

// dtype/op codes follow kungfu_amd.utils.dtypes (same ints as _core)
DType dt(int d) { return (DType)d; }
ReduceOp rop(int o) { return (ReduceOp)o; }
hipStream_t strm(uintptr_t s) { return (hipStream_t)s; }

void init_scope(int s, bool use_gpu)
{
    if (!g.api) throw std::runtime_error("_rccl.init not called");
    if (!g.scopes[s]) g.scopes[s] = std::make_unique<Controller>();
    g.scopes[s]->init(g.api, (Scope)s, g.device, use_gpu);
}

}  // namespace

PYBIND11_MODULE(_rccl, m)
{
    m.doc() = "KungFu-AMD native RCCL collective layer (gfx950/xGMI)";

    m.def("init",
          [](py::capsule cap, int device) {
              std::lock_guard<std::mutex> lk(g.mu);
              g.api = cap_api(cap);
              g.device = device;
              g.gpu = true;
              py::gil_scoped_release rel;
              init_scope(0, true);
          },
          py::arg("control_api"), py::arg("device"),
          "Bootstrap the GLOBAL communicator (uniqueId over the control "
          "plane). local/cross scopes are created lazily via init_scope.");

    m.def("init_cpu",
          [](py::capsule cap) {
              std::lock_guard<std::mutex> lk(g.mu);
              g.api = cap_api(cap);
              g.device = -1;
              g.gpu = false;
              init_scope(0, false);
              init_scope(1, false);
              init_scope(2, false);
          },
          py::arg("control_api"),
          "Dispatcher-only mode (no GPU): ordering/agreement tests.");

    m.def("init_scope", [](int s) {
        std::lock_guard<std::mutex> lk(g.mu);
        py::gil_scoped_release rel;
        init_scope(s, g.gpu);
    });

    m.def("reinit", [] {
        // after an elastic resize: rebuild every live scope against the
        // new cluster (reference ResetNcclHelper, ops/gpu/scheduler.cpp)
        std::lock_guard<std::mutex> lk(g.mu);
        py::gil_scoped_release rel;
        for (int s = 0; s < 3; ++s) {
            if (g.scopes[s]) g.scopes[s]->init(g.api, (Scope)s, g.device,
                                               g.gpu);
        }
    });

    m.def("finalize", [] {
        std::lock_guard<std::mutex> lk(g.mu);
        py::gil_scoped_release rel;
        for (auto &c : g.scopes) {
            if (c) c->destroy();
            c.reset();
        }
        g.api = nullptr;
    });

    m.def("active", [] { return g.api != nullptr && g.gpu; });
    m.def("scope_rank", [](int s) { return g.scope(s).rank(); });
    m.def("scope_size", [](int s) { return g.scope(s).size(); });
    m.def("scope_member", [](int s) { return g.scope(s).member(); });
    m.def("scope_ready",
          [](int s) { return (bool)g.scopes[s] && g.scopes[s]->member(); });

    // ---- collectives (async; return handle ids) ----
    m.def("all_reduce",
          [](int s, const std::string &name, uintptr_t send, uintptr_t recv,
             size_t count, int dtype, int op, uintptr_t stream) {
              py::gil_scoped_release rel;
              return g.scope(s).all_reduce(name, (const void *)send,
                                           (void *)recv, count, dt(dtype),
                                           rop(op), strm(stream));
          });
    m.def("broadcast",
          [](int s, const std::string &name, uintptr_t send, uintptr_t recv,
             size_t count, int dtype, int root, uintptr_t stream) {
              py::gil_scoped_release rel;
              return g.scope(s).broadcast(name, (const void *)send,
                                          (void *)recv, count, dt(dtype),
                                          root, strm(stream));
          });
    m.def("reduce",
          [](int s, const std::string &name, uintptr_t send, uintptr_t recv,
             size_t count, int dtype, int op, int root, uintptr_t stream) {
              py::gil_scoped_release rel;
              return g.scope(s).reduce(name, (const void *)send,
                                       (void *)recv, count, dt(dtype),
                                       rop(op), root, strm(stream));
          });
    m.def("all_gather",
          [](int s, const std::string &name, uintptr_t send, uintptr_t recv,
             size_t count_per_rank, int dtype, uintptr_t stream) {
              py::gil_scoped_release rel;
              return g.scope(s).all_gather(name, (const void *)send,
                                           (void *)recv, count_per_rank,
                                           dt(dtype), strm(stream));
          });
    m.def("reduce_scatter",
          [](int s, const std::string &name, uintptr_t send, uintptr_t recv,
             size_t count_per_rank, int dtype, int op, uintptr_t stream) {
              py::gil_scoped_release rel;
              return g.scope(s).reduce_scatter(
                  name, (const void *)send, (void *)recv, count_per_rank,
                  dt(dtype), rop(op), strm(stream));
          });
    m.def("send_recv",
          [](int s, const std::string &name, uintptr_t send, uintptr_t recv,
             size_t count, int dtype, int peer, uintptr_t stream) {
              py::gil_scoped_release rel;
              return g.scope(s).send_recv(name, (const void *)send,
                                          (void *)recv, count, dt(dtype),
                                          peer, strm(stream));
          });

    // inline variants: launched on the caller's stream from the calling
    // thread — ONLY valid under hipGraph stream capture (program order =
    // agreement; no handles, completion is stream-ordered)
    m.def("all_reduce_inline",
          [](int s, uintptr_t send, uintptr_t recv, size_t count,
             int dtype, int op, uintptr_t stream) {
              py::gil_scoped_release rel;
              g.scope(s).all_reduce_inline((const void *)send,
                                           (void *)recv, count, dt(dtype),
                                           rop(op), strm(stream));
          });
    m.def("broadcast_inline",
          [](int s, uintptr_t send, uintptr_t recv, size_t count,
             int dtype, int root, uintptr_t stream) {
              py::gil_scoped_release rel;
              g.scope(s).broadcast_inline((const void *)send,
                                          (void *)recv, count, dt(dtype),
                                          root, strm(stream));
          });

    // ---- completion ----
    m.def("wait", [](uint64_t h, uintptr_t stream) {
        py::gil_scoped_release rel;
        handle_wait_stream(h, strm(stream));
    });
    m.def("wait_host", [](uint64_t h) {
        py::gil_scoped_release rel;
        handle_wait_host(h);
    });

    // ---- deterministic ordering (reference NCCLScheduler surface) ----
    m.def("scheduler_reset", [](int s, const std::vector<std::string> &n) {
        py::gil_scoped_release rel;
        g.scope(s).scheduler_reset(n);
    });
    m.def("scheduler_agree", [](int s) {
        py::gil_scoped_release rel;
        return g.scope(s).scheduler_agree(g.api);
    });
    m.def("last_arrival",
          [](int s) { return g.scope(s).last_arrival(); });
    m.def("drain", [](int s) {
        py::gil_scoped_release rel;
        g.scope(s).drain();
    });
    // test hook: run a python callable under the dispatcher's ordering
    m.def("start_task", [](int s, const std::string &name, py::function f) {
        // the callable's refcount must be dropped WITH the GIL, and the
        // drop happens on the dispatcher thread when the task dies
        std::shared_ptr<py::function> fp(
            new py::function(std::move(f)), [](py::function *p) {
                py::gil_scoped_acquire acq;
                delete p;
            });
        g.scope(s).start_task(name, [fp] {
            py::gil_scoped_acquire acq;
            (*fp)();
        });
    });
}
