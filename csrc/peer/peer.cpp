#include "peer.hpp"

#include <arpa/inet.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <chrono>
#include <cstdlib>
#include <cstring>
#include <sstream>
#include <thread>

namespace kf {

// ---------- env config ----------

static std::string getenv_str(const char *k)
{
    const char *v = std::getenv(k);
    return v ? std::string(v) : std::string();
}

PeerConfig parse_env_config()
{
    PeerConfig cfg;
    std::string self = getenv_str("KUNGFU_SELF_SPEC");
    if (self.empty()) {
        // single-process fallback (reference env/config.go:66-78)
        cfg.single = true;
        cfg.self = PeerID::parse("127.0.0.1:1");
        cfg.init_peers.peers = {cfg.self};
        return cfg;
    }
    cfg.self = PeerID::parse(self);
    cfg.init_peers = PeerList::parse(getenv_str("KUNGFU_INIT_PEERS"));
    std::string runners = getenv_str("KUNGFU_INIT_RUNNERS");
    if (!runners.empty()) cfg.init_runners = PeerList::parse(runners);
    std::string ver = getenv_str("KUNGFU_INIT_CLUSTER_VERSION");
    if (!ver.empty()) cfg.init_version = (uint32_t)std::stoul(ver);
    std::string strat = getenv_str("KUNGFU_ALLREDUCE_STRATEGY");
    if (!strat.empty()) cfg.strategy = strategy_from_name(strat);
    cfg.config_server = getenv_str("KUNGFU_CONFIG_SERVER");
    std::string nounix = getenv_str("KUNGFU_NO_UNIX_SOCK");
    if (!nounix.empty() && nounix != "0") cfg.use_unix = false;
    return cfg;
}

// ---------- tiny HTTP client ----------

static bool http_request(const std::string &hostport,
                         const std::string &method, const std::string &path,
                         const std::string &body, std::string &resp_body,
                         int timeout_ms)
{
    std::string hp = hostport;
    auto scheme = hp.find("://");
    if (scheme != std::string::npos) hp = hp.substr(scheme + 3);
    std::string extra_path;
    auto slash = hp.find('/');
    if (slash != std::string::npos) {
        extra_path = hp.substr(slash);
        hp = hp.substr(0, slash);
    }
    PeerID target;
    try {
        target = PeerID::parse(hp);
    } catch (...) {
        return false;
    }
    int fd = ::socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) return false;
    timeval tv{timeout_ms / 1000, (timeout_ms % 1000) * 1000};
    ::setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    ::setsockopt(fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(target.ipv4);
    addr.sin_port = htons(target.port);
    if (::connect(fd, (sockaddr *)&addr, sizeof(addr)) != 0) {
        ::close(fd);
        return false;
    }
    std::string full_path =
        extra_path.empty() ? path : extra_path + (path == "/" ? "" : path);
    std::ostringstream req;
    req << method << " " << full_path << " HTTP/1.1\r\n"
        << "Host: " << hp << "\r\n"
        << "Connection: close\r\n"
        << "Content-Length: " << body.size() << "\r\n"
        << "Content-Type: application/json\r\n\r\n"
        << body;
    std::string r = req.str();
    const char *p = r.data();
    size_t left = r.size();
    while (left > 0) {
        ssize_t n = ::send(fd, p, left, MSG_NOSIGNAL);
        if (n <= 0) {
            ::close(fd);
            return false;
        }
        p += n;
        left -= (size_t)n;
    }
    std::string resp;
    char buf[4096];
    while (true) {
        ssize_t n = ::recv(fd, buf, sizeof(buf), 0);
        if (n <= 0) break;
        resp.append(buf, (size_t)n);
    }
    ::close(fd);
    auto hdr_end = resp.find("\r\n\r\n");
    if (hdr_end == std::string::npos) return false;
    // status line: "HTTP/1.1 200 ..."
    auto sp = resp.find(' ');
    if (sp == std::string::npos) return false;
    int status = std::atoi(resp.c_str() + sp + 1);
    resp_body = resp.substr(hdr_end + 4);
    return status >= 200 && status < 300;
}

bool http_get(const std::string &hostport, const std::string &path,
              std::string &body_out, int timeout_ms)
{
    return http_request(hostport, "GET", path, "", body_out, timeout_ms);
}

bool http_put(const std::string &hostport, const std::string &path,
              const std::string &body, int timeout_ms)
{
    std::string out;
    return http_request(hostport, "PUT", path, body, out, timeout_ms);
}

bool http_post(const std::string &hostport, const std::string &path,
               const std::string &body, int timeout_ms)
{
    std::string out;
    return http_request(hostport, "POST", path, body, out, timeout_ms);
}

// ---------- Peer ----------

Peer::Peer(const PeerConfig &cfg) : cfg_(cfg), version_(cfg.init_version) {}

Peer::~Peer() { close(); }

void Peer::start()
{
    if (started_) return;
    started_ = true;
    workers_ = cfg_.init_peers;
    runners_ = cfg_.init_runners;
    // aux subsystems driven by env (reference config/config.go:25-70)
    if (!getenv_str("KUNGFU_USE_AFFINITY").empty()) {
        bind_cpu_affinity(workers_.local_rank_of(cfg_.self),
                          workers_.local_size_of(cfg_.self));
    }
    if (!getenv_str("KUNGFU_CONFIG_ENABLE_STALL_DETECTION").empty()) {
        stall_ = std::make_unique<StallDetector>(3.0);
    }
    client_ = std::make_unique<Client>(cfg_.self);
    client_->set_token(version_);
    store_.set_owner_port(cfg_.self.port);
    if (!cfg_.single) {
        server_ = std::make_unique<Server>(cfg_.self, cfg_.use_unix);
        p2p_ = std::make_unique<P2PEndpoint>(store_, *client_, cfg_.self);
        server_->start(
            [this](const Handshake &hs, Frame &f, Conn &conn) {
                (void)conn;
                switch (hs.type) {
                case ConnType::Collective:
                    collective_.on_frame(hs.src, f);
                    break;
                case ConnType::P2P:
                    p2p_->on_frame(hs.src, f);
                    break;
                case ConnType::Control:
                    // workers currently receive no control messages;
                    // runners use RunnerServer (bindings) instead
                    break;
                default:
                    break;
                }
            },
            [this](uint32_t token) {
                // fencing: accept current or newer tokens; reject stale
                // peers from before the last resize (connection.go:81-87)
                return token + 1 >= version_;
            },
            [this](const Handshake &hs, const FrameHeader &h, Conn &conn) {
                // zero-copy fast path: collective payloads land straight
                // in the registered destination (or a pooled buffer)
                if (hs.type != ConnType::Collective) return false;
                return collective_.on_header(hs.src, h, conn);
            });
    }
    int r = workers_.rank_of(cfg_.self);
    if (r < 0) throw std::runtime_error("self not in init peer list");
    session_ = std::make_unique<Session>(workers_, r, *client_, collective_,
                                         cfg_.strategy);
    if (!cfg_.single &&
        !getenv_str("KUNGFU_CONFIG_ENABLE_MONITORING").empty()) {
        // Prometheus-style counters at peer port + 10000
        metrics_ = std::make_unique<MetricsServer>(
            (uint16_t)(cfg_.self.port + 10000), [this] {
                std::string out;
                for (auto &kv : client_->egress_all()) {
                    PeerID p;
                    p.ipv4 = (uint32_t)(kv.first >> 16);
                    p.port = (uint16_t)(kv.first & 0xffff);
                    out += "kungfu_egress_bytes_total{peer=\"" + p.str() +
                           "\"} " + std::to_string(kv.second) + "\n";
                }
                for (auto &kv : ingress_bytes()) {
                    PeerID p;
                    p.ipv4 = (uint32_t)(kv.first >> 16);
                    p.port = (uint16_t)(kv.first & 0xffff);
                    out += "kungfu_ingress_bytes_total{peer=\"" +
                           p.str() + "\"} " + std::to_string(kv.second) +
                           "\n";
                }
                return out;
            });
    }
    // wait for all peers to be reachable, then an initial barrier
    if (!cfg_.single && workers_.size() > 1) {
        for (const auto &p : workers_.peers) {
            if (p == cfg_.self) continue;
            if (!client_->wait(p, 120000))
                throw std::runtime_error("peer " + p.str() +
                                         " unreachable at startup");
        }
        session_->barrier();
    }
}

void Peer::close()
{
    if (!started_) return;
    started_ = false;
    metrics_.reset();
    stall_.reset();
    collective_.shutdown();
    if (p2p_) p2p_->shutdown();
    if (server_) server_->stop();
    session_.reset();
    server_.reset();
    client_.reset();
}

int Peer::rank() const { return session_ ? session_->rank() : 0; }
int Peer::size() const { return session_ ? session_->size() : 1; }
int Peer::local_rank() const
{
    return workers_.local_rank_of(cfg_.self);
}
int Peer::local_size() const
{
    return workers_.local_size_of(cfg_.self);
}
int Peer::host_count() const { return workers_.host_count(); }
int Peer::host_rank() const
{
    const auto hs = workers_.hosts();
    for (int i = 0; i < (int)hs.size(); ++i) {
        if (hs[i] == cfg_.self.ipv4) return i;
    }
    return 0;
}

Session &Peer::session()
{
    if (!session_) throw std::runtime_error("peer not started");
    return *session_;
}

void Peer::save(const std::string &name, const void *data, size_t len)
{
    store_.save(name, data, len);
}

bool Peer::request(int target_rank, const std::string &name, void *dst,
                   size_t len)
{
    if (target_rank < 0 || target_rank >= workers_.size()) return false;
    return request_addr(workers_.peers[target_rank], name, dst, len);
}

bool Peer::request_addr(const PeerID &target, const std::string &name,
                        void *dst, size_t len)
{
    if (cfg_.single || !p2p_) {
        auto blob = store_.get(name);
        if (!blob || blob->size() != len) return false;
        std::memcpy(dst, blob->data(), len);
        return true;
    }
    if (target == cfg_.self) {
        auto blob = store_.get(name);
        if (!blob || blob->size() != len) return false;
        std::memcpy(dst, blob->data(), len);
        return true;
    }
    return p2p_->request(target, name, dst, len);
}

std::string Peer::fetch_cluster_config()
{
    std::string body;
    if (cfg_.config_server.empty()) return "";
    if (!http_get(cfg_.config_server, "/config", body)) return "";
    return body;
}

int Peer::propose_new_size(int new_size)
{
    if (cfg_.config_server.empty()) return -1;
    std::string body = fetch_cluster_config();
    if (body.empty()) return -1;
    Cluster cur;
    try {
        cur = Cluster::from_json(body);
    } catch (...) {
        return -1;
    }
    if (new_size < 1) return -1;
    uint16_t port_base = cfg_.self.port;
    for (const auto &w : cur.workers.peers) {
        if (w.port < port_base) port_base = w.port;
    }
    Cluster next = cur.resized(new_size, port_base);
    if (!http_put(cfg_.config_server, "/config", next.json())) return -1;
    return 0;
}

ResizeResult Peer::resize_cluster_from_url()
{
    ResizeResult rr;
    if (cfg_.config_server.empty() || cfg_.single) return rr;
    // consensus retry loop over the fetched config (peer.go:236-263)
    std::string body;
    const auto deadline =
        std::chrono::steady_clock::now() + std::chrono::seconds(120);
    while (true) {
        body = fetch_cluster_config();
        if (!body.empty() &&
            session_->consensus(body.data(), body.size(), "|resize"))
            break;
        if (std::chrono::steady_clock::now() > deadline)
            throw std::runtime_error("resize consensus timed out");
        std::this_thread::sleep_for(std::chrono::milliseconds(100));
    }
    Cluster next = Cluster::from_json(body);
    if (next.workers == workers_) return rr;  // no change
    rr.changed = true;
    const uint32_t next_version = version_ + 1;
    // stage message for the runners (watch mode spawns/kills procs)
    if (session_->rank() == 0) {
        std::string stage = "{\"version\":" + std::to_string(next_version) +
                            ",\"cluster\":" + next.json() + "}";
        notify_runners(stage);
    }
    version_ = next_version;
    if (next.workers.rank_of(cfg_.self) < 0) {
        detached_ = true;
        rr.detached = true;
        return rr;
    }
    update_to(next.workers,
              next.runners.size() ? next.runners : runners_);
    return rr;
}

ResizeResult Peer::resize(int new_size)
{
    if (session_ && session_->rank() == 0 && new_size != size()) {
        if (propose_new_size(new_size) != 0)
            throw std::runtime_error("propose_new_size failed");
    }
    return resize_cluster_from_url();
}

void Peer::update_to(const PeerList &workers, const PeerList &runners)
{
    workers_ = workers;
    runners_ = runners;
    std::vector<PeerID> keeps = workers.peers;
    for (const auto &r : runners.peers) keeps.push_back(r);
    client_->reset(keeps, version_);
    int r = workers_.rank_of(cfg_.self);
    session_ = std::make_unique<Session>(workers_, r, *client_, collective_,
                                         cfg_.strategy);
    for (const auto &p : workers_.peers) {
        if (p == cfg_.self) continue;
        if (!client_->wait(p, 120000))
            throw std::runtime_error("peer " + p.str() +
                                     " unreachable after resize");
    }
    session_->barrier();
}

void Peer::notify_runners(const std::string &stage_json)
{
    for (const auto &r : runners_.peers) {
        try {
            client_->send(r, ConnType::Control, "update", 0,
                          stage_json.data(), stage_json.size());
        } catch (const std::exception &e) {
            std::fprintf(stderr, "[kungfu] notify runner %s failed: %s\n",
                         r.str().c_str(), e.what());
        }
    }
}

std::vector<int64_t> Peer::peer_latencies_us()
{
    std::vector<int64_t> out(workers_.size(), 0);
    for (int i = 0; i < workers_.size(); ++i) {
        if (i == session_->rank()) continue;
        out[i] = client_->ping(workers_.peers[i]);
    }
    return out;
}

}  // namespace kf
