// Point-to-point transport: framed messages over TCP / Unix sockets.
//
// Reference parity: srcs/go/rchannel/{connection,client,server}
// (message framing message.go:42-199, handshake with cluster-version token
// connection.go:28-87, conn pool client.go, TCP+Unix listeners server.go).
// Re-designed in C++: connections are simplex (client sends frames to the
// remote peer's server); the only duplex conn type is Ping (server echoes).
// Colocated peers (same IPv4) use Unix domain sockets.
#pragma once

#include <atomic>
#include <cstdint>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <thread>
#include <vector>

#include "../core/plan.hpp"

namespace kf {

enum class ConnType : uint8_t {
    Ping = 0,
    Control = 1,
    Collective = 2,
    P2P = 3,
};

namespace msgflag {
constexpr uint32_t IsResponse = 1u << 0;
constexpr uint32_t RequestFailed = 1u << 1;
constexpr uint32_t IsRequest = 1u << 2;
constexpr uint32_t ShmRef = 1u << 3;  // payload is a /dev/shm path, not data
}  // namespace msgflag

struct Frame {
    std::string name;
    uint32_t flags = 0;
    std::vector<uint8_t> data;
};

struct FrameHeader {
    std::string name;
    uint32_t flags = 0;
    uint64_t len = 0;
};

// Size-classed reusable buffer pool (reference byte_slice_pool.go): the
// receive path would otherwise allocate a fresh heap buffer per frame.
class BufPool {
  public:
    std::vector<uint8_t> get(size_t n);
    void put(std::vector<uint8_t> &&v);

  private:
    static constexpr size_t kMaxPerClass = 16;
    std::mutex mu_;
    std::map<size_t, std::vector<std::vector<uint8_t>>> classes_;
};

// Thread-safe framed socket wrapper.
class Conn {
  public:
    explicit Conn(int fd) : fd_(fd) {}
    ~Conn() { close_fd(); }
    Conn(const Conn &) = delete;

    bool send_frame(const std::string &name, uint32_t flags,
                    const void *data, size_t len);
    bool read_frame(Frame &f);  // blocking; false on EOF/error
    // split read: header first, then the body into caller-owned memory
    // (zero-copy receive: the socket read lands in the registered
    // destination buffer, reference handler/collective.go:34-41)
    bool read_header(FrameHeader &h);
    bool read_body(void *dst, size_t len);
    void close_fd();
    int fd() const { return fd_; }
    std::mutex &write_mutex() { return wmu_; }

  private:
    std::atomic<int> fd_;
    std::mutex wmu_;
};

struct Handshake {
    ConnType type = ConnType::Ping;
    PeerID src;
    uint32_t token = 0;  // cluster version; stale peers are fenced off
};

using FrameHandler =
    std::function<void(const Handshake &, Frame &, Conn &)>;
// Optional header-level handler: may consume the body straight off the
// socket (zero-copy); returns true when it did. Falling through (false)
// makes the server read the body into a Frame and call the FrameHandler.
using HeaderHandler = std::function<bool(const Handshake &,
                                         const FrameHeader &, Conn &)>;

// Listens on TCP (self.port) and, when enabled, on a Unix socket keyed by
// port. One handler thread per accepted connection.
class Server {
  public:
    Server(const PeerID &self, bool use_unix);
    ~Server();
    // token_ok(peer_token) decides whether to accept (elastic fencing)
    void start(FrameHandler handler,
               std::function<bool(uint32_t)> token_ok,
               HeaderHandler header_handler = nullptr);
    void stop();
    static std::string unix_sock_path(const PeerID &peer);
    // ingress byte accounting per source peer (reference
    // monitor/counters.go meters both directions)
    std::map<uint64_t, uint64_t> ingress_all() const;

  private:
    void accept_loop(int listen_fd);
    void handle_conn(int fd);

    PeerID self_;
    bool use_unix_;
    int tcp_fd_ = -1, unix_fd_ = -1;
    std::atomic<bool> stopping_{false};
    FrameHandler handler_;
    HeaderHandler header_handler_;
    std::function<bool(uint32_t)> token_ok_;
    std::vector<std::thread> threads_;
    mutable std::mutex mu_;
    std::vector<std::shared_ptr<Conn>> conns_;
    std::map<uint64_t, uint64_t> ingress_;
};

// Outgoing connection pool: one cached conn per (remote, type).
class Client {
  public:
    explicit Client(const PeerID &self) : self_(self)
    {
        const char *e = std::getenv("KUNGFU_SHM_COLLECTIVES");
        shm_collectives_ = e && *e && std::string(e) != "0";
    }
    ~Client() { reset({}, 0); }

    void set_token(uint32_t token) { token_ = token; }
    // Send one frame; establishes/caches the connection. Throws on failure.
    void send(const PeerID &remote, ConnType type, const std::string &name,
              uint32_t flags, const void *data, size_t len);
    // RTT ping; returns microseconds, or -1 on failure.
    int64_t ping(const PeerID &remote, int timeout_ms = 2000);
    // Wait until remote answers pings (poll every poll_ms).
    bool wait(const PeerID &remote, int timeout_ms, int poll_ms = 200);
    // Drop all conns except to `keeps`; adopt new token (elastic resize).
    void reset(const std::vector<PeerID> &keeps, uint32_t token);

    // egress byte accounting (monitoring)
    uint64_t egress_bytes(const PeerID &remote) const;
    std::map<uint64_t, uint64_t> egress_all() const;

  private:
    std::shared_ptr<Conn> get_conn(const PeerID &remote, ConnType type,
                                   int connect_timeout_ms = 20000);

    PeerID self_;
    bool shm_collectives_ = false;
    std::atomic<uint64_t> shm_seq_{0};
    std::atomic<uint32_t> token_{0};
    mutable std::mutex mu_;
    std::map<std::pair<uint64_t, uint8_t>, std::shared_ptr<Conn>> pool_;
    std::map<uint64_t, uint64_t> egress_;
};

}  // namespace kf
