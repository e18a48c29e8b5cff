#include "session.hpp"

#include <chrono>
#include <cstring>
#include <thread>

#include "../core/workers.hpp"

namespace kf {

namespace {
// FNV-1a for strategy sharding (reference shard.go hash-by-name/index)
uint64_t fnv1a(const std::string &s)
{
    uint64_t h = 1469598103934665603ull;
    for (char c : s) {
        h ^= (uint8_t)c;
        h *= 1099511628211ull;
    }
    return h;
}
}  // namespace

Session::Session(const PeerList &peers, int rank, Client &client,
                 CollectiveEndpoint &collective, Strategy strategy)
    : peers_(peers),
      rank_(rank),
      client_(client),
      collective_(collective),
      strategy_kind_(strategy)
{
    global_ = gen_strategies(peers_, strategy);
    local_ = gen_local_strategies(peers_);
    cross_ = gen_cross_strategies(peers_);
    stats_.resize(global_.size());
}

void Session::set_tree(const std::vector<int> &parent)
{
    if ((int)parent.size() != peers_.size())
        throw std::runtime_error("set_tree: bad forest size");
    global_ = {gen_from_forest(parent)};
    reduce_only_.reset();
    std::lock_guard<std::mutex> lk(stats_mu_);
    stats_.assign(1, {});
}

void Session::all_reduce_with(const std::vector<int> &parent,
                              const Workspace &w)
{
    if ((int)parent.size() != peers_.size())
        throw std::runtime_error("all_reduce_with: bad forest size");
    std::shared_ptr<std::vector<GraphPair>> g;
    {
        std::lock_guard<std::mutex> lk(stats_mu_);
        auto it = forest_cache_.find(parent);
        if (it == forest_cache_.end()) {
            it = forest_cache_
                     .emplace(parent,
                              std::make_shared<std::vector<GraphPair>>(
                                  std::vector<GraphPair>{
                                      gen_from_forest(parent)}))
                     .first;
        }
        g = it->second;
    }
    run_strategies(w, *g, true);
}

void Session::set_strategy(Strategy s)
{
    strategy_kind_ = s;
    global_ = gen_strategies(peers_, s);
    reduce_only_.reset();
    std::lock_guard<std::mutex> lk(stats_mu_);
    stats_.assign(global_.size(), {});
}

// One (reduce,bcast) graph-pair execution over one chunk.
// Reference: session.go runGraphs (222-290).
void Session::run_graphs(const Workspace &w, const GraphPair &g,
                         const std::string &suffix)
{
    const size_t bytes = w.count * dtype_size(w.dt);
    const auto &rg = g.reduce;
    const auto &bg = g.bcast;
    const bool isolated = rg.prevs[rank_].empty() &&
                          rg.nexts[rank_].empty() && !rg.self_loop[rank_] &&
                          bg.prevs[rank_].empty() && bg.nexts[rank_].empty();
    if (w.recv != w.send && w.send != nullptr) {
        std::memcpy(w.recv, w.send, bytes);
    }
    if (isolated) return;

    uint8_t *acc = (uint8_t *)w.recv;
    const std::string rname = w.name + suffix + "|r";
    const std::string bname = w.name + suffix + "|b";

    // reduce phase: aggregate partials from children, forward to parent
    for (int p : rg.prevs[rank_]) {
        auto buf = collective_.recv(peer(p), rname);
        if (buf.size() != bytes)
            throw std::runtime_error("reduce size mismatch on " + rname);
        reduce_inplace(acc, buf.data(), w.count, w.dt, w.op);
        collective_.recycle(std::move(buf));
    }
    for (int nx : rg.nexts[rank_]) {
        client_.send(peer(nx), ConnType::Collective, rname, 0, acc, bytes);
    }

    // bcast phase: receive the final value, pipeline onward
    if (!bg.prevs[rank_].empty()) {
        collective_.recv_into(peer(bg.prevs[rank_][0]), bname, acc, bytes);
    }
    for (int nx : bg.nexts[rank_]) {
        client_.send(peer(nx), ConnType::Collective, bname, 0, acc, bytes);
    }
}

// Bcast-graph-only execution (for broadcast).
void Session::run_bcast_graph(const Workspace &w, const Graph &bg,
                              const std::string &suffix)
{
    const size_t bytes = w.count * dtype_size(w.dt);
    const std::string bname = w.name + suffix + "|b";
    uint8_t *acc = (uint8_t *)w.recv;
    if (!bg.prevs[rank_].empty()) {
        collective_.recv_into(peer(bg.prevs[rank_][0]), bname, acc, bytes);
    } else if (w.recv != w.send && w.send != nullptr) {
        std::memcpy(w.recv, w.send, bytes);
    }
    for (int nx : bg.nexts[rank_]) {
        client_.send(peer(nx), ConnType::Collective, bname, 0, acc, bytes);
    }
}

// Chunked multi-strategy dispatch (reference session.go:292-317 + shard.go).
void Session::run_strategies(const Workspace &w,
                             const std::vector<GraphPair> &strategies,
                             bool monitored)
{
    const size_t elem = dtype_size(w.dt);
    const size_t bytes = w.count * elem;
    // Messages are keyed by (src, name) with FIFO queues, so repeated ops
    // on the same name (one per training step) stay correctly matched even
    // when ranks run ahead; no sequence number is needed (and one would
    // break concurrent ops issued in different orders across ranks).
    const std::string sfx_base = "";

    size_t nchunks = 1;
    if (bytes > kChunkBytes && strategies.size() > 0) {
        nchunks = (bytes + kChunkBytes - 1) / kChunkBytes;
        if (nchunks > kMaxChunks) nchunks = kMaxChunks;
    }
    auto t0 = std::chrono::steady_clock::now();
    if (nchunks == 1) {
        const size_t sidx = fnv1a(w.name) % strategies.size();
        run_graphs(w, strategies[sidx], sfx_base + "@" +
                                            std::to_string(sidx));
        if (monitored) {
            auto t1 = std::chrono::steady_clock::now();
            std::lock_guard<std::mutex> lk(stats_mu_);
            if (sidx < stats_.size()) {
                auto &st = stats_[sidx];
                st.ops++;
                st.bytes += bytes;
                st.seconds +=
                    std::chrono::duration<double>(t1 - t0).count();
            }
        }
        return;
    }

    const size_t per = (w.count + nchunks - 1) / nchunks;
    std::vector<std::string> errors(nchunks);
    size_t launched = 0;
    for (size_t c = 0; c < nchunks; ++c) {
        const size_t begin = c * per;
        if (std::min(per, w.count - begin) == 0 || begin >= w.count) break;
        ++launched;
    }
    Latch latch((int)launched);
    auto &pool = CachedThreadPool::inst();
    // chunks run on the cached worker pool (no thread spawn per chunk);
    // chunk 0 runs inline on the caller thread
    for (size_t c = 1; c < launched; ++c) {
        const size_t begin = c * per;
        const size_t cnt = std::min(per, w.count - begin);
        Workspace cw = w;
        cw.send = w.send ? (const uint8_t *)w.send + begin * elem : nullptr;
        cw.recv = (uint8_t *)w.recv + begin * elem;
        cw.count = cnt;
        const size_t sidx =
            (fnv1a(w.name) + c) % strategies.size();  // spread chunks
        std::string sfx = sfx_base + "." + std::to_string(c) + "@" +
                          std::to_string(sidx);
        pool.submit([this, cw, &strategies, sidx, sfx, &errors, c,
                     &latch] {
            try {
                run_graphs(cw, strategies[sidx], sfx);
            } catch (const std::exception &e) {
                errors[c] = e.what();
            }
            latch.done();
        });
    }
    if (launched > 0) {
        Workspace cw = w;
        cw.count = std::min(per, w.count);
        const size_t sidx = fnv1a(w.name) % strategies.size();
        try {
            run_graphs(cw, strategies[sidx],
                       sfx_base + ".0@" + std::to_string(sidx));
        } catch (const std::exception &e) {
            errors[0] = e.what();
        }
        latch.done();
    }
    latch.wait();
    for (auto &e : errors) {
        if (!e.empty()) throw std::runtime_error("chunk failed: " + e);
    }
    if (monitored) {
        auto t1 = std::chrono::steady_clock::now();
        std::lock_guard<std::mutex> lk(stats_mu_);
        const size_t sidx = fnv1a(w.name) % stats_.size();
        auto &st = stats_[sidx];
        st.ops++;
        st.bytes += bytes;
        st.seconds += std::chrono::duration<double>(t1 - t0).count();
    }
}

void Session::all_reduce(const Workspace &w)
{
    run_strategies(w, global_, true);
}

void Session::reduce(const Workspace &w)
{
    // reduce-only strategy list (reduce graphs + empty bcast), built once
    // from ALL global strategies so chunked reduces rotate across
    // topologies exactly like all-reduce (round-1 gap: always used
    // global_[0] and ran unchunked)
    std::shared_ptr<std::vector<GraphPair>> ro;
    {
        std::lock_guard<std::mutex> lk(stats_mu_);
        if (!reduce_only_ || reduce_only_->size() != global_.size()) {
            auto fresh = std::make_shared<std::vector<GraphPair>>();
            for (const auto &g : global_) {
                GraphPair p;
                p.reduce = g.reduce;
                p.bcast = Graph(p.reduce.n);  // result stays at root
                fresh->push_back(std::move(p));
            }
            reduce_only_ = std::move(fresh);
        }
        ro = reduce_only_;
    }
    run_strategies(w, *ro, false);
}

void Session::broadcast(const Workspace &w, int root)
{
    const std::string sfx = "@bc" + std::to_string(root);
    if (root == 0) {
        run_bcast_graph(w, global_[0].bcast, sfx);
    } else {
        // cache per-root star graphs (round-1 gap: regenerated per call)
        std::unique_lock<std::mutex> lk(stats_mu_);
        auto it = root_bcast_.find(root);
        if (it == root_bcast_.end()) {
            it = root_bcast_
                     .emplace(root, gen_star(peers_.size(), root))
                     .first;
        }
        GraphPair pair = it->second;
        lk.unlock();
        run_bcast_graph(w, pair.bcast, sfx);
    }
}

void Session::all_gather(const Workspace &w)
{
    const size_t elem = dtype_size(w.dt);
    const size_t bytes = w.count * elem;
    const std::string name = w.name + "@ag";
    const int n = peers_.size();
    uint8_t *out = (uint8_t *)w.recv;
    std::memcpy(out + (size_t)rank_ * bytes, w.send, bytes);
    // full-mesh exchange (reference allgather.go:17-45); sends fan out on
    // the cached worker pool
    Latch latch(n - 1 > 0 ? n - 1 : 0);
    auto &pool = CachedThreadPool::inst();
    for (int r = 0; r < n; ++r) {
        if (r == rank_) continue;
        pool.submit([this, r, name, bytes, &w, &latch] {
            try {
                client_.send(peer(r), ConnType::Collective, name, 0,
                             w.send, bytes);
            } catch (...) {
            }
            latch.done();
        });
    }
    for (int r = 0; r < n; ++r) {
        if (r == rank_) continue;
        collective_.recv_into(peer(r), name, out + (size_t)r * bytes,
                              bytes);
    }
    latch.wait();
}

void Session::gather(const Workspace &w)
{
    const size_t elem = dtype_size(w.dt);
    const size_t bytes = w.count * elem;
    const std::string name = w.name + "@g";
    if (rank_ == 0) {
        uint8_t *out = (uint8_t *)w.recv;
        std::memcpy(out, w.send, bytes);
        for (int r = 1; r < peers_.size(); ++r) {
            collective_.recv_into(peer(r), name, out + (size_t)r * bytes,
                                  bytes);
        }
    } else {
        client_.send(peer(0), ConnType::Collective, name, 0, w.send, bytes);
    }
}

void Session::barrier()
{
    uint8_t b = 0, out = 0;
    Workspace w;
    w.send = &b;
    w.recv = &out;
    w.count = 1;
    w.dt = DType::U8;
    w.op = ReduceOp::SUM;
    w.name = "|barrier";
    run_strategies(w, global_, false);
}

bool Session::consensus(const void *data, size_t len,
                        const std::string &name)
{
    // agree on length first, then min/max over the bytes
    uint64_t lmin = len, lmax = len;
    uint64_t local_len = len;
    Workspace wl;
    wl.count = 1;
    wl.dt = DType::U64;
    wl.name = name + "|len";
    wl.send = &local_len;
    wl.recv = &lmin;
    wl.op = ReduceOp::MIN;
    run_strategies(wl, global_, false);
    wl.recv = &lmax;
    wl.op = ReduceOp::MAX;
    run_strategies(wl, global_, false);
    if (lmin != lmax) return false;
    if (len == 0) return true;
    std::vector<uint8_t> mn(len), mx(len);
    Workspace w;
    w.send = data;
    w.count = len;
    w.dt = DType::U8;
    w.name = name + "|bytes";
    w.recv = mn.data();
    w.op = ReduceOp::MIN;
    run_strategies(w, global_, false);
    w.recv = mx.data();
    w.op = ReduceOp::MAX;
    run_strategies(w, global_, false);
    return std::memcmp(mn.data(), mx.data(), len) == 0 &&
           std::memcmp(mn.data(), data, len) == 0;
}

void Session::local_reduce(const Workspace &w)
{
    run_strategies(w, local_, false);
}

void Session::local_broadcast(const Workspace &w)
{
    run_bcast_graph(w, local_[0].bcast, "@lb");
}

void Session::cross_all_reduce(const Workspace &w)
{
    run_strategies(w, cross_, false);
}

std::vector<StrategyStat> Session::stats() const
{
    std::lock_guard<std::mutex> lk(stats_mu_);
    return stats_;
}

void Session::reset_stats()
{
    std::lock_guard<std::mutex> lk(stats_mu_);
    for (auto &s : stats_) s = {};
}

bool Session::check_interference(double ratio)
{
    std::lock_guard<std::mutex> lk(stats_mu_);
    double cur = 0;
    uint64_t total_ops = 0;
    for (auto &s : stats_) {
        cur += s.throughput();
        total_ops += s.ops;
    }
    if (total_ops == 0) return false;
    if (cur > best_throughput_) best_throughput_ = cur;
    return best_throughput_ > 0 && cur < ratio * best_throughput_;
}

}  // namespace kf
