// The CPU/TCP collective engine.
//
// Reference parity: srcs/go/kungfu/session/ — runGraphs partial-aggregation
// tree execution (session.go:222-290), >1MiB chunking with strategy hashing
// (session.go:292-317, shard.go), three strategy lists (global/local/cross,
// session.go:51-63), consensus via min/max all-reduce (session.go:124-155),
// allgather (allgather.go), gather-to-root (session.go:189-211), and
// per-strategy throughput stats (strategy.go:46-56, monitoring.go).
//
// MI355X-native divergence: the GPU hot path bypasses this engine entirely
// (RCCL over xGMI, see kungfu_amd/parallel/); this engine carries control
// traffic, CPU-plumbing mode, consensus/elastic agreement and the
// cross-host hop of hierarchical all-reduce.
#pragma once

#include <functional>
#include <map>
#include <memory>
#include <string>
#include <vector>

#include "../core/common.hpp"
#include "../core/plan.hpp"
#include "../net/endpoints.hpp"
#include "../net/transport.hpp"

namespace kf {

struct Workspace {
    const void *send = nullptr;
    void *recv = nullptr;
    size_t count = 0;
    DType dt = DType::F32;
    ReduceOp op = ReduceOp::SUM;
    std::string name;
};

struct StrategyStat {
    uint64_t ops = 0;
    uint64_t bytes = 0;
    double seconds = 0;
    double throughput() const { return seconds > 0 ? bytes / seconds : 0; }
};

class Session {
  public:
    Session(const PeerList &peers, int rank, Client &client,
            CollectiveEndpoint &collective, Strategy strategy);

    int rank() const { return rank_; }
    int size() const { return peers_.size(); }
    const PeerList &peers() const { return peers_; }

    void all_reduce(const Workspace &w);
    void reduce(const Workspace &w);     // result valid at root (rank 0)
    void broadcast(const Workspace &w, int root = 0);
    void all_gather(const Workspace &w);  // recv holds count*size elems
    void gather(const Workspace &w);      // to rank 0
    void barrier();
    // all peers agree on these bytes? (min/max all-reduce compare)
    bool consensus(const void *data, size_t len, const std::string &name);

    // hierarchical pieces (reference session/strategy.go:176-210)
    void local_reduce(const Workspace &w);
    void local_broadcast(const Workspace &w);
    void cross_all_reduce(const Workspace &w);

    // adaptive topology: install a forest (parent array) as the only global
    // strategy (reference adapt.go:45-52 SetTree)
    void set_tree(const std::vector<int> &parent);
    // one all-reduce over a caller-supplied forest (reference
    // allreduce.go:18-35 AllReduceWith); the built graph pair is cached
    // keyed by the parent array
    void all_reduce_with(const std::vector<int> &parent,
                         const Workspace &w);
    void set_strategy(Strategy s);
    Strategy strategy() const { return strategy_kind_; }

    // monitoring
    std::vector<StrategyStat> stats() const;
    void reset_stats();
    // true if current (last-window) throughput of the monitored strategy
    // dropped below ratio * best observed (interference vote input;
    // reference adaptiveStrategies.go:61-121)
    bool check_interference(double ratio);

  private:
    void run_strategies(const Workspace &w,
                        const std::vector<GraphPair> &strategies,
                        bool monitored);
    void run_graphs(const Workspace &w, const GraphPair &g,
                    const std::string &suffix);
    void run_bcast_graph(const Workspace &w, const Graph &bcast,
                         const std::string &suffix);
    const PeerID &peer(int r) const { return peers_.peers[r]; }

    PeerList peers_;
    int rank_;
    Client &client_;
    CollectiveEndpoint &collective_;
    Strategy strategy_kind_;
    std::vector<GraphPair> global_, local_, cross_;
    // lazily built graph caches; guarded by stats_mu_ (async collective
    // handles may run Session methods concurrently)
    std::shared_ptr<std::vector<GraphPair>> reduce_only_;
    std::map<int, GraphPair> root_bcast_;     // cached non-zero-root stars
    std::map<std::vector<int>,
             std::shared_ptr<std::vector<GraphPair>>> forest_cache_;
    mutable std::mutex stats_mu_;
    std::vector<StrategyStat> stats_;
    double best_throughput_ = 0;

    static constexpr size_t kChunkBytes = 1 << 20;  // 1 MiB
    static constexpr size_t kMaxChunks = 32;
};

}  // namespace kf
