// Native RCCL-over-xGMI collective layer for kungfu_amd (gfx950).
//
// Reference parity: srcs/cpp/src/nccl/{gpu_collective,controller,helper}.cpp
// — communicator bootstrap with the uniqueId broadcast over the framework's
// OWN control plane (gpu_collective.cpp:169-191), per-scope controllers
// (global / intra-host local / cross-host masters, controller.cpp:7-39),
// and one serializing dispatcher per scope (scheduler.cpp).
//
// MI355X-native divergences:
//   * ops are STREAM-ORDERED: a ready-event recorded on the caller's HIP
//     stream orders the dedicated comm stream after the producer kernels,
//     and wait(handle, stream) orders consumers after the collective with
//     hipStreamWaitEvent — no host sync, no spin-wait (the reference
//     spin-waits GPU events, ops/gpu/collective.cpp:11-34, and host-syncs
//     its stream after every collective);
//   * bf16/f64/i64/u8 supported (the reference stops at f16/f32/i32);
//   * point-to-point send/recv pairs (xGMI gossip) are first-class.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <memory>
#include <mutex>
#include <string>
#include <unordered_map>
#include <vector>

#include <hip/hip_runtime.h>
#include <rccl/rccl.h>

#include "../core/common.hpp"
#include "../core/control_api.h"
#include "dispatcher.hpp"

namespace kf {

ncclDataType_t to_rccl_dtype(DType d);
ncclRedOp_t to_rccl_op(ReduceOp op);

// A pending collective: launched on the dispatcher thread, completion
// visible through the done event (stream-ordered) or host wait.
struct GpuHandle {
    hipEvent_t ready = nullptr;  // caller-stream producer fence
    hipEvent_t done = nullptr;   // recorded on the comm stream post-launch
    std::mutex mu;
    std::condition_variable cv;
    bool launched = false;
    std::string error;  // non-empty => launch failed on dispatcher thread

    ~GpuHandle();  // frees any events not released by a wait (error paths)
    void mark_launched(std::string err);
    void wait_launched();  // throws on captured error
};

enum class Scope : int { GLOBAL = 0, LOCAL = 1, CROSS = 2 };

// One RCCL communicator + dedicated comm stream for one scope.
class GpuComm {
  public:
    GpuComm(const ncclUniqueId &id, int size, int rank);
    ~GpuComm();
    GpuComm(const GpuComm &) = delete;

    int rank() const { return rank_; }
    int size() const { return size_; }
    hipStream_t stream() const { return stream_; }
    ncclComm_t comm() const { return comm_; }

  private:
    ncclComm_t comm_ = nullptr;
    hipStream_t stream_ = nullptr;
    int rank_, size_;
};

// Per-scope controller: communicator + ordered dispatcher + the round
// bookkeeping for order agreement.
class Controller {
  public:
    // Bootstrap (or re-bootstrap after elastic resize): uniqueId generated
    // at the scope root and broadcast over the control plane. use_gpu=false
    // builds a dispatcher-only controller (CPU tests / non-member ranks).
    void init(const kf_control_api *api, Scope scope, int device,
              bool use_gpu);
    void destroy();  // drain dispatcher, free the communicator
    bool member() const { return member_; }
    bool has_comm() const { return (bool)comm_; }
    int rank() const;
    int size() const;

    // Submit ops (caller thread): records the ready event on
    // caller_stream, enqueues the launch under `name` ordering, returns a
    // handle id registered in the global registry.
    uint64_t all_reduce(const std::string &name, const void *send,
                        void *recv, size_t count, DType dt, ReduceOp op,
                        hipStream_t caller);
    uint64_t broadcast(const std::string &name, const void *send, void *recv,
                       size_t count, DType dt, int root,
                       hipStream_t caller);
    uint64_t reduce(const std::string &name, const void *send, void *recv,
                    size_t count, DType dt, ReduceOp op, int root,
                    hipStream_t caller);
    uint64_t all_gather(const std::string &name, const void *send,
                        void *recv, size_t count_per_rank, DType dt,
                        hipStream_t caller);
    uint64_t reduce_scatter(const std::string &name, const void *send,
                            void *recv, size_t count_per_rank, DType dt,
                            ReduceOp op, hipStream_t caller);
    // paired point-to-point exchange (AD-PSGD gossip over xGMI)
    uint64_t send_recv(const std::string &name, const void *send,
                       void *recv, size_t count, DType dt, int peer,
                       hipStream_t caller);

    // Inline launch on the CALLER's stream from the calling thread — for
    // hipGraph stream capture ONLY (capture is single-threaded, so
    // program order IS the cross-rank agreement; the dispatcher's
    // cross-thread events cannot be captured).
    void all_reduce_inline(const void *send, void *recv, size_t count,
                           DType dt, ReduceOp op, hipStream_t stream);
    void broadcast_inline(const void *send, void *recv, size_t count,
                          DType dt, int root, hipStream_t stream);

    // Ordered-release round management (reference scheduler semantics).
    void scheduler_reset(const std::vector<std::string> &names);
    // Broadcast rank-0's last arrival order over the control plane and
    // adopt it as the release order; returns the agreed order.
    std::vector<int32_t> scheduler_agree(const kf_control_api *api);
    std::vector<int32_t> last_arrival() const
    {
        return disp_.last_arrival();
    }

    // Test hook: enqueue an arbitrary task under ordering.
    void start_task(const std::string &name, std::function<void()> fn)
    {
        disp_.start(name, std::move(fn));
    }
    void drain() { disp_.run_sync([] {}); }

  private:
    uint64_t submit(const std::string &name, hipStream_t caller,
                    std::function<void(hipStream_t)> launch);
    GpuComm &comm();

    Scope scope_ = Scope::GLOBAL;
    bool member_ = true;
    int rank_ = 0, size_ = 1;
    std::unique_ptr<GpuComm> comm_;
    OrderedDispatcher disp_;
    std::vector<std::string> round_names_;
};

// Handle registry (module-global).
uint64_t register_handle(std::shared_ptr<GpuHandle> h);
std::shared_ptr<GpuHandle> take_handle(uint64_t id);

// wait helpers (destroy the handle's events after enqueuing the wait)
void handle_wait_stream(uint64_t id, hipStream_t stream);
void handle_wait_host(uint64_t id);

}  // namespace kf
