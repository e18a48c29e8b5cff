// Peer runtime: lifecycle, env-config protocol, elastic resize.
//
// Reference parity: srcs/go/kungfu/peer/{peer,legacy,p2p}.go and
// srcs/go/kungfu/env/. Every worker process embeds one Peer: it starts the
// transport server, builds a Session over the initial cluster, and supports
// runtime cluster resize driven by an HTTP config server, with byte-level
// consensus among current peers and control-plane notifications to runners
// (peer.go:177-263).
#pragma once

#include <memory>
#include <string>

#include "../net/endpoints.hpp"
#include "../net/transport.hpp"
#include "../session/session.hpp"
#include "aux.hpp"

namespace kf {

struct PeerConfig {
    PeerID self;
    PeerList init_peers;    // workers
    PeerList init_runners;  // one per host
    uint32_t init_version = 0;
    Strategy strategy = Strategy::AUTO;
    std::string config_server;  // "host:port/path" or empty
    bool use_unix = true;
    bool single = false;  // single-process fallback (no env present)
};

// Env protocol (set by the launcher; reference env/envs.go):
//   KUNGFU_SELF_SPEC, KUNGFU_INIT_PEERS, KUNGFU_INIT_RUNNERS,
//   KUNGFU_INIT_CLUSTER_VERSION, KUNGFU_ALLREDUCE_STRATEGY,
//   KUNGFU_CONFIG_SERVER
PeerConfig parse_env_config();

// Minimal HTTP/1.1 helpers for the config server (JSON bodies).
// url_hostport is "ip:port"; path starts with '/'.
bool http_get(const std::string &hostport, const std::string &path,
              std::string &body_out, int timeout_ms = 5000);
bool http_put(const std::string &hostport, const std::string &path,
              const std::string &body, int timeout_ms = 5000);
bool http_post(const std::string &hostport, const std::string &path,
               const std::string &body, int timeout_ms = 5000);

struct ResizeResult {
    bool changed = false;
    bool detached = false;
};

class Peer {
  public:
    explicit Peer(const PeerConfig &cfg);
    ~Peer();
    void start();
    void close();

    int rank() const;
    int size() const;
    int local_rank() const;
    int local_size() const;
    int host_count() const;
    int host_rank() const;  // index of this peer's host among hosts
    uint32_t version() const { return version_; }
    std::string uid() const { return cfg_.self.str(); }
    bool detached() const { return detached_; }
    Session &session();

    // P2P model store (AD-PSGD)
    void save(const std::string &name, const void *data, size_t len);
    bool request(int target_rank, const std::string &name, void *dst,
                 size_t len);
    bool request_addr(const PeerID &target, const std::string &name,
                      void *dst, size_t len);

    // Elastic
    int propose_new_size(int new_size);
    ResizeResult resize_cluster_from_url();
    ResizeResult resize(int new_size);  // propose + resize (rank0 proposes)

    // Monitoring
    std::vector<int64_t> peer_latencies_us();
    StallDetector *stall_detector() { return stall_.get(); }
    std::map<uint64_t, uint64_t> egress_bytes() const
    {
        return client_->egress_all();
    }
    std::map<uint64_t, uint64_t> ingress_bytes() const
    {
        return server_ ? server_->ingress_all()
                       : std::map<uint64_t, uint64_t>{};
    }

  private:
    void update_to(const PeerList &workers, const PeerList &runners);
    void notify_runners(const std::string &stage_json);
    std::string fetch_cluster_config();

    PeerConfig cfg_;
    uint32_t version_ = 0;
    bool detached_ = false;
    bool started_ = false;
    PeerList workers_;
    PeerList runners_;
    std::unique_ptr<Server> server_;
    std::unique_ptr<Client> client_;
    CollectiveEndpoint collective_;
    BlobStore store_;
    std::unique_ptr<P2PEndpoint> p2p_;
    std::unique_ptr<Session> session_;
    std::unique_ptr<StallDetector> stall_;
    std::unique_ptr<MetricsServer> metrics_;
};

}  // namespace kf
