// Implementation of the native RCCL layer. See rccl_layer.hpp.
#include "rccl_layer.hpp"

#include <cstdio>
#include <cstring>
#include <stdexcept>

namespace kf {

namespace {

struct HipError : std::runtime_error {
    using std::runtime_error::runtime_error;
};

void hip_check(hipError_t e, const char *what)
{
    if (e != hipSuccess) {
        throw HipError(std::string(what) + ": " + hipGetErrorString(e));
    }
}

void rccl_check(ncclResult_t e, const char *what)
{
    if (e != ncclSuccess) {
        throw std::runtime_error(std::string(what) + ": " +
                                 ncclGetErrorString(e) + " (" +
                                 std::to_string((int)e) + ")");
    }
}

}  // namespace

ncclDataType_t to_rccl_dtype(DType d)
{
    switch (d) {
    case DType::U8: return ncclUint8;
    case DType::I8: return ncclInt8;
    case DType::I32: return ncclInt32;
    case DType::I64: return ncclInt64;
    case DType::U32: return ncclUint32;
    case DType::U64: return ncclUint64;
    case DType::F16: return ncclFloat16;
    case DType::BF16: return ncclBfloat16;
    case DType::F32: return ncclFloat32;
    case DType::F64: return ncclFloat64;
    default:
        throw std::invalid_argument("dtype unsupported by RCCL");
    }
}

ncclRedOp_t to_rccl_op(ReduceOp op)
{
    switch (op) {
    case ReduceOp::SUM: return ncclSum;
    case ReduceOp::MIN: return ncclMin;
    case ReduceOp::MAX: return ncclMax;
    case ReduceOp::PROD: return ncclProd;
    }
    throw std::invalid_argument("bad reduce op");
}

// ---- GpuHandle ----

GpuHandle::~GpuHandle()
{
    // normal waits destroy the events and null the fields; this covers
    // error paths where the handle is dropped without a wait
    if (ready) (void)hipEventDestroy(ready);
    if (done) (void)hipEventDestroy(done);
}

void GpuHandle::mark_launched(std::string err)
{
    {
        std::lock_guard<std::mutex> lk(mu);
        launched = true;
        error = std::move(err);
    }
    cv.notify_all();
}

void GpuHandle::wait_launched()
{
    std::unique_lock<std::mutex> lk(mu);
    cv.wait(lk, [this] { return launched; });
    if (!error.empty()) throw std::runtime_error("rccl launch: " + error);
}

// ---- handle registry ----

namespace {
std::mutex g_hmu;
uint64_t g_next_handle = 1;
std::unordered_map<uint64_t, std::shared_ptr<GpuHandle>> g_handles;

void destroy_events(GpuHandle &h)
{
    // hipEventDestroy after the waits are ENQUEUED is legal: the runtime
    // defers resource release until the event completes.
    if (h.ready) (void)hipEventDestroy(h.ready);
    if (h.done) (void)hipEventDestroy(h.done);
    h.ready = h.done = nullptr;
}
}  // namespace

uint64_t register_handle(std::shared_ptr<GpuHandle> h)
{
    std::lock_guard<std::mutex> lk(g_hmu);
    const uint64_t id = g_next_handle++;
    g_handles.emplace(id, std::move(h));
    return id;
}

std::shared_ptr<GpuHandle> take_handle(uint64_t id)
{
    std::lock_guard<std::mutex> lk(g_hmu);
    auto it = g_handles.find(id);
    if (it == g_handles.end())
        throw std::runtime_error("unknown rccl handle");
    auto h = std::move(it->second);
    g_handles.erase(it);
    return h;
}

void handle_wait_stream(uint64_t id, hipStream_t stream)
{
    auto h = take_handle(id);
    h->wait_launched();
    hip_check(hipStreamWaitEvent(stream, h->done, 0), "stream wait");
    destroy_events(*h);
}

void handle_wait_host(uint64_t id)
{
    auto h = take_handle(id);
    h->wait_launched();
    hip_check(hipEventSynchronize(h->done), "event sync");
    destroy_events(*h);
}

// ---- GpuComm ----

GpuComm::GpuComm(const ncclUniqueId &id, int size, int rank)
    : rank_(rank), size_(size)
{
    hip_check(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking),
              "comm stream create");
    rccl_check(ncclCommInitRank(&comm_, size, id, rank),
               "ncclCommInitRank");
}

GpuComm::~GpuComm()
{
    if (comm_) (void)ncclCommDestroy(comm_);
    if (stream_) (void)hipStreamDestroy(stream_);
}

// ---- Controller ----

GpuComm &Controller::comm()
{
    if (!comm_)
        throw std::runtime_error(
            member_ ? "rccl communicator not initialized (CPU mode?)"
                    : "rank is not a member of this scope");
    return *comm_;
}

int Controller::rank() const { return rank_; }
int Controller::size() const { return size_; }

void Controller::init(const kf_control_api *api, Scope scope, int device,
                      bool use_gpu)
{
    destroy();
    scope_ = scope;
    void *ctx = api->ctx;
    member_ = true;
    switch (scope) {
    case Scope::GLOBAL:
        rank_ = api->rank(ctx);
        size_ = api->size(ctx);
        break;
    case Scope::LOCAL:
        rank_ = api->local_rank(ctx);
        size_ = api->local_size(ctx);
        break;
    case Scope::CROSS:
        // members: the local master of each host (reference
        // session/strategy.go:188-210 cross strategies)
        rank_ = api->host_rank(ctx);
        size_ = api->host_count(ctx);
        member_ = api->local_rank(ctx) == 0;
        break;
    }
    if (!use_gpu) return;

    // uniqueId rendezvous over the control plane — the whole point of the
    // native layer: no TCPStore, no env:// beside the framework's own
    // transport (reference gpu_collective.cpp:169-191).
    ncclUniqueId id;
    std::memset(&id, 0, sizeof(id));
    char name[64];
    std::snprintf(name, sizeof(name), "|rccl-id/%d/v%u", (int)scope,
                  api->cluster_version(ctx));
    hip_check(hipSetDevice(device), "hipSetDevice");
    if (scope == Scope::LOCAL) {
        if (rank_ == 0) rccl_check(ncclGetUniqueId(&id), "ncclGetUniqueId");
        api->local_broadcast(ctx, &id, sizeof(id), name);
    } else {
        // GLOBAL: generated by global rank 0. CROSS: also generated by
        // global rank 0 (the first host's master == cross rank 0); ALL
        // peers take part in the broadcast (it is a global collective),
        // non-members simply discard the id.
        if (api->rank(ctx) == 0)
            rccl_check(ncclGetUniqueId(&id), "ncclGetUniqueId");
        api->broadcast(ctx, &id, sizeof(id), 0, name);
    }
    if (member_) comm_ = std::make_unique<GpuComm>(id, size_, rank_);
}

void Controller::destroy()
{
    disp_.run_sync([] {});  // drain pending launches
    comm_.reset();
}

uint64_t Controller::submit(const std::string &name, hipStream_t caller,
                            std::function<void(hipStream_t)> launch)
{
    GpuComm &c = comm();  // throws on non-member / CPU mode
    auto h = std::make_shared<GpuHandle>();
    hip_check(hipEventCreateWithFlags(&h->ready, hipEventDisableTiming),
              "event create");
    hip_check(hipEventCreateWithFlags(&h->done, hipEventDisableTiming),
              "event create");
    // producer fence: everything enqueued on the caller stream so far
    // (i.e. the kernels that produced send_buf) precedes the collective
    hip_check(hipEventRecord(h->ready, caller), "ready record");
    hipStream_t cs = c.stream();
    hipEvent_t ready = h->ready, done = h->done;
    disp_.start(name, [h, cs, ready, done,
                       launch = std::move(launch)]() mutable {
        std::string err;
        try {
            hip_check(hipStreamWaitEvent(cs, ready, 0), "comm wait ready");
            launch(cs);
            hip_check(hipEventRecord(done, cs), "done record");
        } catch (const std::exception &e) {
            err = e.what();
        }
        h->mark_launched(std::move(err));
    });
    return register_handle(std::move(h));
}

uint64_t Controller::all_reduce(const std::string &name, const void *send,
                                void *recv, size_t count, DType dt,
                                ReduceOp op, hipStream_t caller)
{
    ncclComm_t cm = comm().comm();
    return submit(name, caller, [=](hipStream_t s) {
        rccl_check(ncclAllReduce(send, recv, count, to_rccl_dtype(dt),
                                 to_rccl_op(op), cm, s),
                   "ncclAllReduce");
    });
}

uint64_t Controller::broadcast(const std::string &name, const void *send,
                               void *recv, size_t count, DType dt, int root,
                               hipStream_t caller)
{
    ncclComm_t cm = comm().comm();
    return submit(name, caller, [=](hipStream_t s) {
        rccl_check(ncclBroadcast(send, recv, count, to_rccl_dtype(dt), root,
                                 cm, s),
                   "ncclBroadcast");
    });
}

uint64_t Controller::reduce(const std::string &name, const void *send,
                            void *recv, size_t count, DType dt, ReduceOp op,
                            int root, hipStream_t caller)
{
    ncclComm_t cm = comm().comm();
    return submit(name, caller, [=](hipStream_t s) {
        rccl_check(ncclReduce(send, recv, count, to_rccl_dtype(dt),
                              to_rccl_op(op), root, cm, s),
                   "ncclReduce");
    });
}

uint64_t Controller::all_gather(const std::string &name, const void *send,
                                void *recv, size_t count_per_rank, DType dt,
                                hipStream_t caller)
{
    ncclComm_t cm = comm().comm();
    return submit(name, caller, [=](hipStream_t s) {
        rccl_check(ncclAllGather(send, recv, count_per_rank,
                                 to_rccl_dtype(dt), cm, s),
                   "ncclAllGather");
    });
}

uint64_t Controller::reduce_scatter(const std::string &name,
                                    const void *send, void *recv,
                                    size_t count_per_rank, DType dt,
                                    ReduceOp op, hipStream_t caller)
{
    ncclComm_t cm = comm().comm();
    return submit(name, caller, [=](hipStream_t s) {
        rccl_check(ncclReduceScatter(send, recv, count_per_rank,
                                     to_rccl_dtype(dt), to_rccl_op(op), cm,
                                     s),
                   "ncclReduceScatter");
    });
}

uint64_t Controller::send_recv(const std::string &name, const void *send,
                               void *recv, size_t count, DType dt, int peer,
                               hipStream_t caller)
{
    ncclComm_t cm = comm().comm();
    return submit(name, caller, [=](hipStream_t s) {
        rccl_check(ncclGroupStart(), "group start");
        rccl_check(ncclSend(send, count, to_rccl_dtype(dt), peer, cm, s),
                   "ncclSend");
        rccl_check(ncclRecv(recv, count, to_rccl_dtype(dt), peer, cm, s),
                   "ncclRecv");
        rccl_check(ncclGroupEnd(), "group end");
    });
}

void Controller::all_reduce_inline(const void *send, void *recv,
                                   size_t count, DType dt, ReduceOp op,
                                   hipStream_t stream)
{
    rccl_check(ncclAllReduce(send, recv, count, to_rccl_dtype(dt),
                             to_rccl_op(op), comm().comm(), stream),
               "ncclAllReduce(inline)");
}

void Controller::broadcast_inline(const void *send, void *recv,
                                  size_t count, DType dt, int root,
                                  hipStream_t stream)
{
    rccl_check(ncclBroadcast(send, recv, count, to_rccl_dtype(dt), root,
                             comm().comm(), stream),
               "ncclBroadcast(inline)");
}

void Controller::scheduler_reset(const std::vector<std::string> &names)
{
    round_names_ = names;
    std::vector<int32_t> order(names.size());
    for (size_t i = 0; i < names.size(); ++i) order[i] = (int32_t)i;
    disp_.reset(names, order);
}

std::vector<int32_t> Controller::scheduler_agree(const kf_control_api *api)
{
    // Reference scheduler.cpp:93-119: rank 0 broadcasts its observed
    // arrival order; every rank adopts it as the release order.
    disp_.wait_round();
    std::vector<int32_t> order = disp_.last_arrival();
    const size_t n = round_names_.size();
    if (order.size() != n) {
        order.resize(n);
        for (size_t i = 0; i < n; ++i) order[i] = (int32_t)i;
    }
    char name[64];
    std::snprintf(name, sizeof(name), "|rccl-order/%d", (int)scope_);
    if (scope_ == Scope::LOCAL) {
        api->local_broadcast(api->ctx, order.data(),
                             n * sizeof(int32_t), name);
    } else {
        api->broadcast(api->ctx, order.data(), n * sizeof(int32_t), 0,
                       name);
    }
    disp_.reset(round_names_, order);
    return order;
}

}  // namespace kf
