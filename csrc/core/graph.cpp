#include "graph.hpp"

#include <algorithm>
#include <functional>
#include <stdexcept>

namespace kf {

Graph Graph::reversed() const
{
    Graph g(n);
    g.self_loop = self_loop;
    for (int i = 0; i < n; ++i) {
        for (int j : nexts[i]) g.add_edge(j, i);
    }
    return g;
}

std::string Graph::digest() const
{
    std::string s;
    s.reserve(64);
    s += std::to_string(n);
    s += '[';
    for (int i = 0; i < n; ++i) {
        if (self_loop[i]) s += '*';
        for (int j : nexts[i]) {
            s += std::to_string(i);
            s += '>';
            s += std::to_string(j);
            s += ';';
        }
    }
    s += ']';
    return s;
}

GraphPair gen_star(int n, int center)
{
    Graph reduce(n);
    reduce.add_edge(center, center);
    for (int i = 0; i < n; ++i) {
        if (i != center) reduce.add_edge(i, center);
    }
    return {reduce, reduce.reversed()};
}

GraphPair gen_binary_tree_order(const std::vector<int> &order)
{
    const int n = (int)order.size();
    Graph reduce(n);
    reduce.add_edge(order[0], order[0]);
    for (int i = 1; i < n; ++i) {
        reduce.add_edge(order[i], order[(i - 1) / 2]);  // child -> parent
    }
    return {reduce, reduce.reversed()};
}

GraphPair gen_binary_tree(int n)
{
    std::vector<int> order(n);
    for (int i = 0; i < n; ++i) order[i] = i;
    return gen_binary_tree_order(order);
}

GraphPair gen_circular(int n, int r)
{
    // Reduce chain (r+1) -> (r+2) -> ... -> r, so rotation r is rooted at r
    // (rotation 0 keeps the conventional root 0). Bcast pipelines from the
    // root back around the ring.
    Graph reduce(n);
    reduce.add_edge(r, r);
    for (int i = 0; i + 1 < n; ++i) {
        reduce.add_edge((r + 1 + i) % n, (r + 2 + i) % n);
    }
    Graph bcast(n);
    bcast.add_edge(r, r);
    for (int i = 0; i + 1 < n; ++i) {
        bcast.add_edge((r + i) % n, (r + i + 1) % n);
    }
    return {reduce, bcast};
}

GraphPair gen_tree_star(const std::vector<int> &host_of,
                        const std::vector<int> &masters, bool binary_cross,
                        int root_host)
{
    const int n = (int)host_of.size();
    const int h = (int)masters.size();
    if (root_host < 0 || root_host >= h)
        throw std::runtime_error("bad root host");
    Graph reduce(n);
    // Intra-host: star into the local master.
    for (int i = 0; i < n; ++i) {
        const int m = masters[host_of[i]];
        if (i != m) reduce.add_edge(i, m);
    }
    // Cross-host over masters, rooted at masters[root_host].
    std::vector<int> order;
    order.push_back(masters[root_host]);
    for (int j = 0; j < h; ++j) {
        if (j != root_host) order.push_back(masters[j]);
    }
    if (binary_cross) {
        for (int i = 1; i < h; ++i) {
            reduce.add_edge(order[i], order[(i - 1) / 2]);
        }
    } else {
        for (int i = 1; i < h; ++i) reduce.add_edge(order[i], order[0]);
    }
    reduce.add_edge(order[0], order[0]);
    return {reduce, reduce.reversed()};
}

GraphPair gen_from_forest(const std::vector<int> &parent)
{
    const int n = (int)parent.size();
    Graph reduce(n);
    for (int i = 0; i < n; ++i) {
        int p = parent[i];
        if (p < 0 || p == i) {
            reduce.add_edge(i, i);  // root
        } else {
            if (p >= n) throw std::runtime_error("bad forest parent");
            reduce.add_edge(i, p);
        }
    }
    // cycle check: every rank must reach a root in <= n hops
    for (int i = 0; i < n; ++i) {
        int v = i;
        for (int hops = 0; hops <= n; ++hops) {
            int p = parent[v];
            if (p < 0 || p == v) goto ok;
            v = p;
        }
        throw std::runtime_error("forest has a cycle");
    ok:;
    }
    return {reduce, reduce.reversed()};
}

}  // namespace kf
