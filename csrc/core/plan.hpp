// Cluster plan: peer identities, peer lists, host specs, strategies.
//
// Reference parity: srcs/go/plan/{peerlist,cluster,hostspec,addr,id}.go and
// srcs/go/kungfu/base/strategy.go. Re-designed in C++ for the MI355X runtime:
// a peer is one process bound to one GPU (or CPU slot), identified by
// (ipv4, port); hosts are grouped by IP for hierarchical topologies.
#pragma once

#include <cstdint>
#include <optional>
#include <string>
#include <vector>

#include "graph.hpp"

namespace kf {

struct PeerID {
    uint32_t ipv4 = 0;  // host byte order
    uint16_t port = 0;

    bool operator==(const PeerID &o) const
    {
        return ipv4 == o.ipv4 && port == o.port;
    }
    bool operator!=(const PeerID &o) const { return !(*this == o); }
    bool operator<(const PeerID &o) const
    {
        return ipv4 != o.ipv4 ? ipv4 < o.ipv4 : port < o.port;
    }

    std::string str() const;                       // "a.b.c.d:port"
    static PeerID parse(const std::string &spec);  // "a.b.c.d:port"
    static uint32_t parse_ipv4(const std::string &s);
    static std::string ipv4_str(uint32_t ip);
    uint64_t key() const { return ((uint64_t)ipv4 << 16) | port; }
};

struct PeerList {
    std::vector<PeerID> peers;

    int size() const { return (int)peers.size(); }
    int rank_of(const PeerID &p) const;       // -1 if absent
    int local_rank_of(const PeerID &p) const; // rank among same-IP peers
    int local_size_of(const PeerID &p) const;
    std::vector<uint32_t> hosts() const;      // distinct IPs in first-seen order
    int host_count() const { return (int)hosts().size(); }
    // host index (into hosts()) per peer
    std::vector<int> host_of() const;
    // local master (lowest-rank peer) per host, in hosts() order
    std::vector<int> masters() const;
    bool operator==(const PeerList &o) const { return peers == o.peers; }

    std::string str() const;                          // comma-joined specs
    static PeerList parse(const std::string &specs);  // "ip:port,ip:port"
    // set difference: peers in *this but not in o (order preserved)
    std::vector<PeerID> sub(const PeerList &o) const;
    bool disjoint(const PeerList &o) const;
    PeerList select(const std::vector<int> &ranks) const;
};

// One host's capacity: "ip:slots[:public_ip]".
struct HostSpec {
    uint32_t ipv4 = 0;
    int slots = 1;
    uint32_t public_ipv4 = 0;

    static HostSpec parse(const std::string &spec);
};

struct HostList {
    std::vector<HostSpec> hosts;

    static HostList parse(const std::string &specs);  // comma-separated
    int cap() const;
    // Allocate np peers round-robin-by-slot with ports from port_base
    // (reference: hostspec.go GenPeerList).
    PeerList gen_peer_list(int np, int port_base) const;
    // One runner per host at runner_port.
    PeerList gen_runner_list(int runner_port) const;
};

// A cluster = runner list + worker list (reference: plan/cluster.go).
struct Cluster {
    PeerList runners;
    PeerList workers;

    bool operator==(const Cluster &o) const
    {
        return runners == o.runners && workers == o.workers;
    }
    // Grow/shrink the worker list to new_size, placing new workers on the
    // least-loaded host and allocating fresh ports (cluster.go:75-120).
    Cluster resized(int new_size, int port_base) const;
    std::string json() const;
    static Cluster from_json(const std::string &s);
};

enum class Strategy : uint8_t {
    STAR = 0,
    MULTI_STAR,
    RING,
    CLIQUE,
    TREE,
    BINARY_TREE,
    BINARY_TREE_STAR,
    MULTI_BINARY_TREE_STAR,
    AUTO,
};

Strategy strategy_from_name(const std::string &name);
std::string strategy_name(Strategy s);

// Build the global strategy list for a peer list
// (reference: session/strategy.go + topology.go generators; AUTO picks STAR
// for one host and BINARY_TREE_STAR for multi-host, strategy.go:165-174).
std::vector<GraphPair> gen_strategies(const PeerList &pl, Strategy s);

// Intra-host star rooted at the local master (for LocalReduce/LocalBcast).
std::vector<GraphPair> gen_local_strategies(const PeerList &pl);

// Ring over local masters only; non-master ranks are isolated
// (for CrossAllReduce; reference strategy.go:188-210).
std::vector<GraphPair> gen_cross_strategies(const PeerList &pl);

}  // namespace kf
