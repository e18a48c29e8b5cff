// Collective topology graphs.
//
// Reference parity: srcs/go/plan/graph/graph.go (DAG with Prevs/Nexts/SelfLoop)
// and srcs/go/plan/topology.go (star / binary-tree / tree-by-host /
// binary-tree-star / multi-star / multi-BTS / circular-ring generators).
// Re-designed in C++: a Graph is adjacency lists over ranks; an all-reduce
// strategy is a (reduce, bcast) pair where data flows along reduce edges
// toward the root(s) with partial aggregation, then back along bcast edges.
#pragma once

#include <cstdint>
#include <string>
#include <vector>

namespace kf {

struct Graph {
    int n = 0;
    std::vector<std::vector<int>> prevs;  // incoming edges per rank
    std::vector<std::vector<int>> nexts;  // outgoing edges per rank
    std::vector<uint8_t> self_loop;      // rank aggregates in place (root)

    explicit Graph(int n_ = 0) : n(n_), prevs(n_), nexts(n_), self_loop(n_, 0)
    {
    }

    void add_edge(int from, int to)
    {
        if (from == to) {
            self_loop[from] = 1;
            return;
        }
        nexts[from].push_back(to);
        prevs[to].push_back(from);
    }

    // Reverse edges (reduce graph -> bcast graph), keeping self-loops.
    Graph reversed() const;

    // Stable byte digest for cluster-wide consensus on topology
    // (reference: graph.go:131-147).
    std::string digest() const;
};

struct GraphPair {
    Graph reduce;
    Graph bcast;
};

// --- generators (all return a GraphPair rooted as described) ---

// Every rank sends to `center`; bcast is the reverse star.
GraphPair gen_star(int n, int center);

// Heap-shaped binary tree rooted at 0 over the given rank ordering.
GraphPair gen_binary_tree(int n);

// Binary tree over an explicit rank permutation (order[0] is the root).
GraphPair gen_binary_tree_order(const std::vector<int> &order);

// Circular (ring) pair rotated by r: reduce chain r -> r+1 -> ... -> r+n-1,
// bcast chain from the root back around. k rotations pipeline k chunks.
GraphPair gen_circular(int n, int r);

// Hierarchical: intra-host stars into each host's local master, then a
// binary tree (or star) over the masters. `host_of[rank]` assigns hosts;
// `masters` lists the local master of each host in host order.
GraphPair gen_tree_star(const std::vector<int> &host_of,
                        const std::vector<int> &masters, bool binary_cross,
                        int root_host = 0);

// Build a (reduce,bcast) pair from an explicit forest: parent[i] is the
// parent rank of i, or i (or -1) for a root. Reference: FromForestArray
// (graph.go:46-70), used by the adaptive set_tree / MST topology path.
GraphPair gen_from_forest(const std::vector<int> &parent);

}  // namespace kf
