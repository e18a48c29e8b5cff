#include "plan.hpp"

#include <algorithm>
#include <map>
#include <set>
#include <sstream>
#include <stdexcept>

namespace kf {

// ---------- PeerID ----------

uint32_t PeerID::parse_ipv4(const std::string &s)
{
    uint32_t parts[4] = {0, 0, 0, 0};
    int idx = 0;
    uint32_t cur = 0;
    bool any = false;
    for (char c : s) {
        if (c == '.') {
            if (!any || idx >= 3) throw std::runtime_error("bad ipv4: " + s);
            parts[idx++] = cur;
            cur = 0;
            any = false;
        } else if (c >= '0' && c <= '9') {
            cur = cur * 10 + (uint32_t)(c - '0');
            if (cur > 255) throw std::runtime_error("bad ipv4: " + s);
            any = true;
        } else {
            throw std::runtime_error("bad ipv4: " + s);
        }
    }
    if (!any || idx != 3) throw std::runtime_error("bad ipv4: " + s);
    parts[3] = cur;
    return (parts[0] << 24) | (parts[1] << 16) | (parts[2] << 8) | parts[3];
}

std::string PeerID::ipv4_str(uint32_t ip)
{
    std::ostringstream o;
    o << ((ip >> 24) & 0xff) << '.' << ((ip >> 16) & 0xff) << '.'
      << ((ip >> 8) & 0xff) << '.' << (ip & 0xff);
    return o.str();
}

std::string PeerID::str() const
{
    return ipv4_str(ipv4) + ":" + std::to_string(port);
}

PeerID PeerID::parse(const std::string &spec)
{
    auto pos = spec.rfind(':');
    if (pos == std::string::npos)
        throw std::runtime_error("bad peer spec: " + spec);
    PeerID p;
    p.ipv4 = parse_ipv4(spec.substr(0, pos));
    int port = std::stoi(spec.substr(pos + 1));
    if (port <= 0 || port > 65535)
        throw std::runtime_error("bad port: " + spec);
    p.port = (uint16_t)port;
    return p;
}

// ---------- PeerList ----------

int PeerList::rank_of(const PeerID &p) const
{
    for (int i = 0; i < (int)peers.size(); ++i) {
        if (peers[i] == p) return i;
    }
    return -1;
}

int PeerList::local_rank_of(const PeerID &p) const
{
    int r = 0;
    for (const auto &q : peers) {
        if (q == p) return r;
        if (q.ipv4 == p.ipv4) ++r;
    }
    return -1;
}

int PeerList::local_size_of(const PeerID &p) const
{
    int c = 0;
    for (const auto &q : peers) {
        if (q.ipv4 == p.ipv4) ++c;
    }
    return c;
}

std::vector<uint32_t> PeerList::hosts() const
{
    std::vector<uint32_t> hs;
    for (const auto &p : peers) {
        if (std::find(hs.begin(), hs.end(), p.ipv4) == hs.end())
            hs.push_back(p.ipv4);
    }
    return hs;
}

std::vector<int> PeerList::host_of() const
{
    auto hs = hosts();
    std::vector<int> out(peers.size());
    for (size_t i = 0; i < peers.size(); ++i) {
        out[i] = (int)(std::find(hs.begin(), hs.end(), peers[i].ipv4) -
                       hs.begin());
    }
    return out;
}

std::vector<int> PeerList::masters() const
{
    auto hs = hosts();
    std::vector<int> ms(hs.size(), -1);
    auto ho = host_of();
    for (int i = 0; i < (int)peers.size(); ++i) {
        if (ms[ho[i]] < 0) ms[ho[i]] = i;
    }
    return ms;
}

std::string PeerList::str() const
{
    std::string s;
    for (size_t i = 0; i < peers.size(); ++i) {
        if (i) s += ',';
        s += peers[i].str();
    }
    return s;
}

PeerList PeerList::parse(const std::string &specs)
{
    PeerList pl;
    std::stringstream ss(specs);
    std::string item;
    while (std::getline(ss, item, ',')) {
        if (!item.empty()) pl.peers.push_back(PeerID::parse(item));
    }
    return pl;
}

std::vector<PeerID> PeerList::sub(const PeerList &o) const
{
    std::vector<PeerID> out;
    for (const auto &p : peers) {
        if (o.rank_of(p) < 0) out.push_back(p);
    }
    return out;
}

bool PeerList::disjoint(const PeerList &o) const
{
    for (const auto &p : peers) {
        if (o.rank_of(p) >= 0) return false;
    }
    return true;
}

PeerList PeerList::select(const std::vector<int> &ranks) const
{
    PeerList out;
    for (int r : ranks) out.peers.push_back(peers.at(r));
    return out;
}

// ---------- HostSpec / HostList ----------

HostSpec HostSpec::parse(const std::string &spec)
{
    HostSpec h;
    std::vector<std::string> parts;
    std::stringstream ss(spec);
    std::string item;
    while (std::getline(ss, item, ':')) parts.push_back(item);
    if (parts.empty() || parts.size() > 3)
        throw std::runtime_error("bad host spec: " + spec);
    h.ipv4 = PeerID::parse_ipv4(parts[0]);
    h.slots = parts.size() > 1 ? std::stoi(parts[1]) : 1;
    h.public_ipv4 = parts.size() > 2 ? PeerID::parse_ipv4(parts[2]) : h.ipv4;
    return h;
}

HostList HostList::parse(const std::string &specs)
{
    HostList hl;
    std::stringstream ss(specs);
    std::string item;
    while (std::getline(ss, item, ',')) {
        if (!item.empty()) hl.hosts.push_back(HostSpec::parse(item));
    }
    return hl;
}

int HostList::cap() const
{
    int c = 0;
    for (const auto &h : hosts) c += h.slots;
    return c;
}

PeerList HostList::gen_peer_list(int np, int port_base) const
{
    if (np > cap()) throw std::runtime_error("np exceeds host capacity");
    PeerList pl;
    std::vector<int> used(hosts.size(), 0);
    // fill host by host (slot-major), like the reference's GenPeerList
    for (size_t hi = 0; hi < hosts.size() && (int)pl.peers.size() < np; ++hi) {
        for (int s = 0; s < hosts[hi].slots && (int)pl.peers.size() < np;
             ++s) {
            PeerID p;
            p.ipv4 = hosts[hi].ipv4;
            p.port = (uint16_t)(port_base + s);
            pl.peers.push_back(p);
        }
    }
    return pl;
}

PeerList HostList::gen_runner_list(int runner_port) const
{
    PeerList pl;
    for (const auto &h : hosts) {
        PeerID p;
        p.ipv4 = h.ipv4;
        p.port = (uint16_t)runner_port;
        pl.peers.push_back(p);
    }
    return pl;
}

// ---------- Cluster ----------

Cluster Cluster::resized(int new_size, int port_base) const
{
    Cluster c = *this;
    int cur = (int)c.workers.peers.size();
    if (new_size < cur) {
        c.workers.peers.resize(new_size);
        return c;
    }
    // grow: least-loaded host first, fresh port on that host
    auto host_ips = runners.hosts();
    if (host_ips.empty()) host_ips = workers.hosts();
    for (int k = cur; k < new_size; ++k) {
        std::map<uint32_t, int> load;
        for (auto ip : host_ips) load[ip] = 0;
        std::map<uint32_t, std::set<uint16_t>> ports;
        for (const auto &w : c.workers.peers) {
            load[w.ipv4]++;
            ports[w.ipv4].insert(w.port);
        }
        uint32_t best = host_ips[0];
        for (auto ip : host_ips) {
            if (load[ip] < load[best]) best = ip;
        }
        uint16_t port = (uint16_t)port_base;
        while (ports[best].count(port)) ++port;
        PeerID p;
        p.ipv4 = best;
        p.port = port;
        c.workers.peers.push_back(p);
    }
    return c;
}

std::string Cluster::json() const
{
    auto arr = [](const PeerList &pl) {
        std::string s = "[";
        for (size_t i = 0; i < pl.peers.size(); ++i) {
            if (i) s += ',';
            s += '"' + pl.peers[i].str() + '"';
        }
        s += ']';
        return s;
    };
    return "{\"runners\":" + arr(runners) + ",\"workers\":" + arr(workers) +
           "}";
}

// Minimal parser for the fixed shape emitted by json() (tolerates
// whitespace). Not a general JSON parser.
Cluster Cluster::from_json(const std::string &s)
{
    auto grab = [&](const std::string &key) -> PeerList {
        auto kp = s.find("\"" + key + "\"");
        if (kp == std::string::npos)
            throw std::runtime_error("cluster json missing " + key);
        auto lb = s.find('[', kp);
        auto rb = s.find(']', lb);
        if (lb == std::string::npos || rb == std::string::npos)
            throw std::runtime_error("bad cluster json");
        PeerList pl;
        size_t i = lb;
        while (true) {
            auto q1 = s.find('"', i);
            if (q1 == std::string::npos || q1 > rb) break;
            auto q2 = s.find('"', q1 + 1);
            if (q2 == std::string::npos || q2 > rb) break;
            pl.peers.push_back(PeerID::parse(s.substr(q1 + 1, q2 - q1 - 1)));
            i = q2 + 1;
        }
        return pl;
    };
    Cluster c;
    c.runners = grab("runners");
    c.workers = grab("workers");
    return c;
}

// ---------- strategies ----------

Strategy strategy_from_name(const std::string &name)
{
    if (name == "STAR") return Strategy::STAR;
    if (name == "MULTI_STAR") return Strategy::MULTI_STAR;
    if (name == "RING") return Strategy::RING;
    if (name == "CLIQUE") return Strategy::CLIQUE;
    if (name == "TREE") return Strategy::TREE;
    if (name == "BINARY_TREE") return Strategy::BINARY_TREE;
    if (name == "BINARY_TREE_STAR") return Strategy::BINARY_TREE_STAR;
    if (name == "MULTI_BINARY_TREE_STAR")
        return Strategy::MULTI_BINARY_TREE_STAR;
    if (name == "AUTO") return Strategy::AUTO;
    throw std::runtime_error("unknown strategy: " + name);
}

std::string strategy_name(Strategy s)
{
    switch (s) {
    case Strategy::STAR: return "STAR";
    case Strategy::MULTI_STAR: return "MULTI_STAR";
    case Strategy::RING: return "RING";
    case Strategy::CLIQUE: return "CLIQUE";
    case Strategy::TREE: return "TREE";
    case Strategy::BINARY_TREE: return "BINARY_TREE";
    case Strategy::BINARY_TREE_STAR: return "BINARY_TREE_STAR";
    case Strategy::MULTI_BINARY_TREE_STAR: return "MULTI_BINARY_TREE_STAR";
    case Strategy::AUTO: return "AUTO";
    }
    return "?";
}

std::vector<GraphPair> gen_strategies(const PeerList &pl, Strategy s)
{
    const int n = pl.size();
    if (s == Strategy::AUTO) {
        s = pl.host_count() == 1 ? Strategy::STAR
                                 : Strategy::BINARY_TREE_STAR;
    }
    std::vector<GraphPair> out;
    switch (s) {
    case Strategy::STAR:
        out.push_back(gen_star(n, 0));
        break;
    case Strategy::MULTI_STAR: {
        // one star per host, centered at that host's local master — chunks
        // hash across them to spread ingress load over hosts
        for (int m : pl.masters()) out.push_back(gen_star(n, m));
        break;
    }
    case Strategy::RING:
        for (int r = 0; r < n; ++r) out.push_back(gen_circular(n, r));
        break;
    case Strategy::CLIQUE:
        // n rotated stars: with chunk hashing this spreads aggregation
        // across all ranks (reduce-scatter-like load balance)
        for (int c = 0; c < n; ++c) out.push_back(gen_star(n, c));
        break;
    case Strategy::TREE:
        out.push_back(gen_tree_star(pl.host_of(), pl.masters(),
                                    /*binary_cross=*/false));
        break;
    case Strategy::BINARY_TREE:
        out.push_back(gen_binary_tree(n));
        break;
    case Strategy::BINARY_TREE_STAR:
        out.push_back(gen_tree_star(pl.host_of(), pl.masters(),
                                    /*binary_cross=*/true));
        break;
    case Strategy::MULTI_BINARY_TREE_STAR: {
        const int h = pl.host_count();
        for (int r = 0; r < h; ++r) {
            out.push_back(gen_tree_star(pl.host_of(), pl.masters(),
                                        /*binary_cross=*/true, r));
        }
        break;
    }
    case Strategy::AUTO:
        break;  // unreachable
    }
    if (out.empty()) throw std::runtime_error("no strategies generated");
    return out;
}

std::vector<GraphPair> gen_local_strategies(const PeerList &pl)
{
    const int n = pl.size();
    Graph reduce(n);
    auto ho = pl.host_of();
    auto ms = pl.masters();
    for (int i = 0; i < n; ++i) {
        const int m = ms[ho[i]];
        if (i == m) {
            reduce.add_edge(i, i);
        } else {
            reduce.add_edge(i, m);
        }
    }
    return {GraphPair{reduce, reduce.reversed()}};
}

std::vector<GraphPair> gen_cross_strategies(const PeerList &pl)
{
    const int n = pl.size();
    auto ms = pl.masters();
    Graph reduce(n);
    Graph bcast(n);
    const int h = (int)ms.size();
    if (h == 1) {
        reduce.add_edge(ms[0], ms[0]);
        bcast.add_edge(ms[0], ms[0]);
    } else {
        // binary tree over masters rooted at the first
        reduce.add_edge(ms[0], ms[0]);
        for (int i = 1; i < h; ++i) reduce.add_edge(ms[i], ms[(i - 1) / 2]);
        bcast = reduce.reversed();
    }
    return {GraphPair{reduce, bcast}};
}

}  // namespace kf
