// Message endpoints: collective rendezvous queues, P2P blob request/response,
// control messages.
//
// Reference parity: srcs/go/rchannel/handler/{collective,p2p}.go and
// srcs/go/store/. The collective endpoint supports zero-copy receive
// (recv_into pre-registers the destination buffer, collective.go:34-41);
// early arrivals are buffered so connection handler threads never block.
#pragma once

#include <condition_variable>
#include <deque>
#include <functional>
#include <memory>
#include <mutex>
#include <unordered_map>

#include "transport.hpp"

namespace kf {

class CollectiveEndpoint {
  public:
    void on_frame(const PeerID &src, Frame &f);
    // Zero-copy receive: consume the frame body straight off the socket —
    // into the pre-registered destination when one is waiting (no copy at
    // all), else into a pooled buffer. Always consumes; throws on socket
    // failure (the server then drops the connection).
    bool on_header(const PeerID &src, const FrameHeader &h, Conn &conn);
    // return a consumed buffer to the receive pool
    void recycle(std::vector<uint8_t> &&v) { pool_.put(std::move(v)); }
    // Copy (or zero-copy if pre-registered before arrival) one message from
    // (src, name) into dst. Throws on size mismatch or shutdown.
    void recv_into(const PeerID &src, const std::string &name, void *dst,
                   size_t len);
    // Take ownership of one message buffer (receive path that aggregates).
    std::vector<uint8_t> recv(const PeerID &src, const std::string &name);
    void clear();     // drop pending state (on resize)
    void shutdown();  // wake all waiters with failure

  private:
    struct Slot {
        std::mutex mu;
        std::condition_variable cv;
        std::deque<std::vector<uint8_t>> q;  // early arrivals
        uint8_t *dst = nullptr;              // registered destination
        size_t dst_len = 0;
        bool filled = false;
        bool filling = false;  // socket read into dst in progress
        bool dead = false;
    };
    std::shared_ptr<Slot> slot(const PeerID &src, const std::string &name);

    std::mutex mu_;
    std::unordered_map<std::string, std::shared_ptr<Slot>> slots_;
    BufPool pool_;
    bool dead_ = false;
};

// Blob store for P2P model exchange (reference: srcs/go/store/ versioned
// store with a sliding version window, versionedstore.go:36-60).
// save() replaces the in-memory blob atomically via shared_ptr swap AND
// mirrors it into a versioned /dev/shm file, so colocated peers (the 8
// workers of one MI355X node) pull models at memcpy speed instead of
// through the loopback socket; the two most recent versions are kept on
// disk so an in-flight reader never races a concurrent save.
class BlobStore {
  public:
    explicit BlobStore(uint16_t owner_port = 0) : owner_port_(owner_port) {}
    ~BlobStore();
    void set_owner_port(uint16_t port) { owner_port_ = port; }
    void save(const std::string &name, const void *data, size_t len);
    std::shared_ptr<const std::vector<uint8_t>> get(
        const std::string &name) const;
    uint64_t version(const std::string &name) const;
    // Path of the current shm mirror ("" when shm is unavailable).
    std::string shm_path(const std::string &name) const;

  private:
    std::string shm_file(const std::string &name, uint64_t ver) const;

    uint16_t owner_port_;
    mutable std::mutex mu_;
    std::unordered_map<std::string,
                       std::shared_ptr<const std::vector<uint8_t>>>
        blobs_;
    std::unordered_map<std::string, uint64_t> versions_;
    std::unordered_map<std::string, std::string> shm_paths_;
};

// Request/response model pulls (AD-PSGD PairAveraging).
class P2PEndpoint {
  public:
    P2PEndpoint(BlobStore &store, Client &client, const PeerID &self)
        : store_(store), client_(client), self_(self)
    {
    }
    void on_frame(const PeerID &src, Frame &f);
    // Pull blob `name` from target into dst; true on success.
    bool request(const PeerID &target, const std::string &name, void *dst,
                 size_t len, int timeout_ms = 30000);
    void shutdown();

  private:
    struct Waiter {
        std::mutex mu;
        std::condition_variable cv;
        std::vector<uint8_t> data;
        bool done = false, failed = false, shm_ref = false;
    };
    BlobStore &store_;
    Client &client_;
    PeerID self_;
    std::mutex mu_;
    std::unordered_map<std::string, std::shared_ptr<Waiter>> waiters_;
};

}  // namespace kf
