#include "transport.hpp"

#include <arpa/inet.h>
#include <errno.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <stdexcept>

namespace kf {

namespace {

constexpr uint32_t kMagic = 0x4b464131;  // "KFA1"

bool write_all(int fd, const void *buf, size_t len)
{
    const char *p = (const char *)buf;
    while (len > 0) {
        ssize_t n = ::send(fd, p, len, MSG_NOSIGNAL);
        if (n <= 0) {
            if (n < 0 && (errno == EINTR)) continue;
            return false;
        }
        p += n;
        len -= (size_t)n;
    }
    return true;
}

bool read_all(int fd, void *buf, size_t len)
{
    char *p = (char *)buf;
    while (len > 0) {
        ssize_t n = ::recv(fd, p, len, 0);
        if (n <= 0) {
            if (n < 0 && errno == EINTR) continue;
            return false;
        }
        p += n;
        len -= (size_t)n;
    }
    return true;
}

void set_nodelay(int fd)
{
    int one = 1;
    ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

void set_bufsizes(int fd)
{
    int sz = 4 << 20;  // deep buffers: the collective engine streams MiB
    ::setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sz, sizeof(sz));
    ::setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &sz, sizeof(sz));
}

}  // namespace

// ---------- Conn ----------

bool Conn::send_frame(const std::string &name, uint32_t flags,
                      const void *data, size_t len)
{
    std::lock_guard<std::mutex> lk(wmu_);
    int fd = fd_.load();
    if (fd < 0) return false;
    uint32_t name_len = (uint32_t)name.size();
    uint64_t data_len = (uint64_t)len;
    // header assembled into one buffer to avoid 4 tiny writes
    std::vector<uint8_t> hdr(4 + name_len + 4 + 8);
    size_t off = 0;
    std::memcpy(hdr.data() + off, &name_len, 4);
    off += 4;
    std::memcpy(hdr.data() + off, name.data(), name_len);
    off += name_len;
    std::memcpy(hdr.data() + off, &flags, 4);
    off += 4;
    std::memcpy(hdr.data() + off, &data_len, 8);
    off += 8;
    if (!write_all(fd, hdr.data(), hdr.size())) return false;
    if (len > 0 && !write_all(fd, data, len)) return false;
    return true;
}

std::vector<uint8_t> BufPool::get(size_t n)
{
    size_t cls = 64;
    while (cls < n) cls <<= 1;
    {
        std::lock_guard<std::mutex> lk(mu_);
        auto it = classes_.find(cls);
        if (it != classes_.end() && !it->second.empty()) {
            auto v = std::move(it->second.back());
            it->second.pop_back();
            v.resize(n);
            return v;
        }
    }
    std::vector<uint8_t> v;
    v.reserve(cls);
    v.resize(n);
    return v;
}

void BufPool::put(std::vector<uint8_t> &&v)
{
    if (v.capacity() < 64) return;
    size_t cls = 64;
    while (cls < v.capacity()) cls <<= 1;
    if (cls != v.capacity()) cls >>= 1;  // conservative: class it fits
    std::lock_guard<std::mutex> lk(mu_);
    auto &vec = classes_[cls];
    if (vec.size() < kMaxPerClass) vec.emplace_back(std::move(v));
}

bool Conn::read_header(FrameHeader &h)
{
    int fd = fd_.load();
    if (fd < 0) return false;
    uint32_t name_len;
    if (!read_all(fd, &name_len, 4)) return false;
    if (name_len > (1u << 16)) return false;  // sanity
    h.name.resize(name_len);
    if (name_len && !read_all(fd, h.name.data(), name_len)) return false;
    if (!read_all(fd, &h.flags, 4)) return false;
    if (!read_all(fd, &h.len, 8)) return false;
    if (h.len > (1ull << 33)) return false;  // 8 GiB sanity cap
    return true;
}

bool Conn::read_body(void *dst, size_t len)
{
    int fd = fd_.load();
    if (fd < 0) return false;
    return len == 0 || read_all(fd, dst, len);
}

bool Conn::read_frame(Frame &f)
{
    int fd = fd_.load();
    if (fd < 0) return false;
    uint32_t name_len;
    if (!read_all(fd, &name_len, 4)) return false;
    if (name_len > (1u << 16)) return false;  // sanity
    f.name.resize(name_len);
    if (name_len && !read_all(fd, f.name.data(), name_len)) return false;
    if (!read_all(fd, &f.flags, 4)) return false;
    uint64_t data_len;
    if (!read_all(fd, &data_len, 8)) return false;
    if (data_len > (1ull << 33)) return false;  // 8 GiB sanity cap
    f.data.resize(data_len);
    if (data_len && !read_all(fd, f.data.data(), data_len)) return false;
    return true;
}

void Conn::close_fd()
{
    int fd = fd_.exchange(-1);
    if (fd >= 0) {
        ::shutdown(fd, SHUT_RDWR);
        ::close(fd);
    }
}

// ---------- Server ----------

std::string Server::unix_sock_path(const PeerID &peer)
{
    // keyed by ip AND port: loopback aliases (127.0.0.x multi-host
    // simulation) may reuse port numbers on one machine
    return "/tmp/kungfu-amd-" + std::to_string(peer.ipv4) + "-" +
           std::to_string(peer.port) + ".sock";
}

Server::Server(const PeerID &self, bool use_unix)
    : self_(self), use_unix_(use_unix)
{
}

Server::~Server() { stop(); }

void Server::start(FrameHandler handler,
                   std::function<bool(uint32_t)> token_ok,
                   HeaderHandler header_handler)
{
    handler_ = std::move(handler);
    token_ok_ = std::move(token_ok);
    header_handler_ = std::move(header_handler);

    // TCP listener on 0.0.0.0:port
    tcp_fd_ = ::socket(AF_INET, SOCK_STREAM, 0);
    if (tcp_fd_ < 0) throw std::runtime_error("socket() failed");
    int one = 1;
    ::setsockopt(tcp_fd_, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    // bind the advertised IP so several logical hosts can coexist on one
    // machine (e.g. 127.0.0.1 vs 127.0.0.2 loopback aliases); fall back
    // to ANY when the address is not a local interface (NAT/public IPs)
    addr.sin_addr.s_addr = htonl(self_.ipv4);
    addr.sin_port = htons(self_.port);
    if (::bind(tcp_fd_, (sockaddr *)&addr, sizeof(addr)) != 0) {
        addr.sin_addr.s_addr = htonl(INADDR_ANY);
        if (::bind(tcp_fd_, (sockaddr *)&addr, sizeof(addr)) != 0)
            throw std::runtime_error("bind failed on port " +
                                     std::to_string(self_.port) + ": " +
                                     std::strerror(errno));
    }
    if (::listen(tcp_fd_, 128) != 0)
        throw std::runtime_error("listen failed");
    {
        // threads_ is shared with the accept loops (they add handler
        // threads under mu_): every mutation must hold the lock
        std::lock_guard<std::mutex> lk(mu_);
        threads_.emplace_back([this] { accept_loop(tcp_fd_); });
    }

    if (use_unix_) {
        unix_fd_ = ::socket(AF_UNIX, SOCK_STREAM, 0);
        if (unix_fd_ >= 0) {
            sockaddr_un ua{};
            ua.sun_family = AF_UNIX;
            std::string path = unix_sock_path(self_);
            ::unlink(path.c_str());
            std::snprintf(ua.sun_path, sizeof(ua.sun_path), "%s",
                          path.c_str());
            if (::bind(unix_fd_, (sockaddr *)&ua, sizeof(ua)) == 0 &&
                ::listen(unix_fd_, 128) == 0) {
                std::lock_guard<std::mutex> lk(mu_);
                threads_.emplace_back([this] { accept_loop(unix_fd_); });
            } else {
                ::close(unix_fd_);
                unix_fd_ = -1;
            }
        }
    }
}

void Server::accept_loop(int listen_fd)
{
    while (!stopping_.load()) {
        int fd = ::accept(listen_fd, nullptr, nullptr);
        if (fd < 0) {
            if (stopping_.load()) return;
            if (errno == EINTR) continue;
            return;
        }
        std::lock_guard<std::mutex> lk(mu_);
        if (stopping_.load()) {
            // stop() may already be joining the (moved-out) thread list;
            // mutating threads_ here would race its iteration
            ::close(fd);
            return;
        }
        threads_.emplace_back([this, fd] { handle_conn(fd); });
    }
}

void Server::handle_conn(int fd)
{
    set_nodelay(fd);
    set_bufsizes(fd);
    // handshake: magic, type, src ip, src port, token
    uint8_t hs_buf[4 + 1 + 4 + 2 + 4];
    if (!read_all(fd, hs_buf, sizeof(hs_buf))) {
        ::close(fd);
        return;
    }
    uint32_t magic;
    std::memcpy(&magic, hs_buf, 4);
    if (magic != kMagic) {
        ::close(fd);
        return;
    }
    Handshake hs;
    hs.type = (ConnType)hs_buf[4];
    std::memcpy(&hs.src.ipv4, hs_buf + 5, 4);
    std::memcpy(&hs.src.port, hs_buf + 9, 2);
    std::memcpy(&hs.token, hs_buf + 11, 4);
    // ack carries accept/reject (0xffffffff = rejected, stale token)
    uint32_t ack = (token_ok_ && !token_ok_(hs.token)) ? 0xffffffffu
                                                       : hs.token;
    if (!write_all(fd, &ack, 4) || ack == 0xffffffffu) {
        ::close(fd);
        return;
    }

    auto conn = std::make_shared<Conn>(fd);
    {
        std::lock_guard<std::mutex> lk(mu_);
        conns_.push_back(conn);
    }
    FrameHeader h;
    while (!stopping_.load() && conn->read_header(h)) {
        {
            std::lock_guard<std::mutex> lk(mu_);
            ingress_[hs.src.key()] += h.len + h.name.size() + 16;
        }
        // zero-copy fast path: the endpoint may consume the body straight
        // off the socket into a pre-registered destination
        if (header_handler_) {
            bool consumed = false;
            try {
                consumed = header_handler_(hs, h, *conn);
            } catch (const std::exception &e) {
                std::fprintf(stderr,
                             "[kungfu] header handler error on '%s': %s\n",
                             h.name.c_str(), e.what());
                break;  // socket state unknown: drop the conn
            }
            if (consumed) continue;
        }
        Frame f;
        f.name = std::move(h.name);
        f.flags = h.flags;
        f.data.resize(h.len);
        if (!conn->read_body(f.data.data(), f.data.size())) break;
        if (hs.type == ConnType::Ping) {
            conn->send_frame(f.name, f.flags | msgflag::IsResponse,
                            f.data.data(), f.data.size());
            continue;
        }
        try {
            handler_(hs, f, *conn);
        } catch (const std::exception &e) {
            std::fprintf(stderr, "[kungfu] handler error on '%s': %s\n",
                         f.name.c_str(), e.what());
        }
    }
    conn->close_fd();
}

void Server::stop()
{
    if (stopping_.exchange(true)) return;
    if (tcp_fd_ >= 0) {
        ::shutdown(tcp_fd_, SHUT_RDWR);
        ::close(tcp_fd_);
    }
    if (unix_fd_ >= 0) {
        ::shutdown(unix_fd_, SHUT_RDWR);
        ::close(unix_fd_);
        ::unlink(unix_sock_path(self_).c_str());
    }
    std::vector<std::thread> ts;
    {
        std::lock_guard<std::mutex> lk(mu_);
        for (auto &c : conns_) c->close_fd();
        ts = std::move(threads_);  // join outside the lock, accept_loop
        threads_.clear();          // can no longer add (stopping_ set)
    }
    for (auto &t : ts) {
        if (t.joinable()) t.join();
    }
}

// ---------- Client ----------

std::shared_ptr<Conn> Client::get_conn(const PeerID &remote, ConnType type,
                                       int connect_timeout_ms)
{
    auto key = std::make_pair(remote.key(), (uint8_t)type);
    {
        std::lock_guard<std::mutex> lk(mu_);
        auto it = pool_.find(key);
        if (it != pool_.end()) return it->second;
    }
    // connect with retry (reference: 500 x 200ms; we use shorter period)
    const auto deadline = std::chrono::steady_clock::now() +
                          std::chrono::milliseconds(connect_timeout_ms);
    int fd = -1;
    while (true) {
        // Prefer Unix socket for colocated peers.
        if (remote.ipv4 == self_.ipv4 ||
            remote.ipv4 == PeerID::parse_ipv4("127.0.0.1")) {
            fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
            if (fd >= 0) {
                sockaddr_un ua{};
                ua.sun_family = AF_UNIX;
                std::snprintf(ua.sun_path, sizeof(ua.sun_path), "%s",
                              Server::unix_sock_path(remote).c_str());
                if (::connect(fd, (sockaddr *)&ua, sizeof(ua)) == 0) {
                    set_bufsizes(fd);
                    break;
                }
                ::close(fd);
                fd = -1;
            }
        }
        fd = ::socket(AF_INET, SOCK_STREAM, 0);
        if (fd >= 0) {
            sockaddr_in addr{};
            addr.sin_family = AF_INET;
            addr.sin_addr.s_addr = htonl(remote.ipv4);
            addr.sin_port = htons(remote.port);
            if (::connect(fd, (sockaddr *)&addr, sizeof(addr)) == 0) {
                set_nodelay(fd);
                set_bufsizes(fd);
                break;
            }
            ::close(fd);
            fd = -1;
        }
        if (std::chrono::steady_clock::now() > deadline)
            throw std::runtime_error("connect to " + remote.str() +
                                     " timed out");
        std::this_thread::sleep_for(std::chrono::milliseconds(50));
    }
    // handshake
    uint8_t hs[4 + 1 + 4 + 2 + 4];
    std::memcpy(hs, &kMagic, 4);
    hs[4] = (uint8_t)type;
    std::memcpy(hs + 5, &self_.ipv4, 4);
    std::memcpy(hs + 9, &self_.port, 2);
    uint32_t tok = token_.load();
    std::memcpy(hs + 11, &tok, 4);
    uint32_t ack = 0;
    if (!write_all(fd, hs, sizeof(hs)) || !read_all(fd, &ack, 4) ||
        ack == 0xffffffffu) {
        ::close(fd);
        throw std::runtime_error("handshake with " + remote.str() +
                                 " rejected (stale token?)");
    }
    auto conn = std::make_shared<Conn>(fd);
    std::lock_guard<std::mutex> lk(mu_);
    auto it = pool_.find(key);
    if (it != pool_.end()) {
        // lost the race; use existing
        conn->close_fd();
        return it->second;
    }
    pool_[key] = conn;
    return conn;
}

void Client::send(const PeerID &remote, ConnType type,
                  const std::string &name, uint32_t flags, const void *data,
                  size_t len)
{
    // Optional /dev/shm data path for colocated collective payloads
    // (KUNGFU_SHM_COLLECTIVES=1): the socket carries only a file
    // reference; the write completes before the frame is sent, so the
    // receiver's read-after-frame ordering is safe. The receiver unlinks
    // after consuming (endpoints.cpp).
    std::string shm_payload;
    if (shm_collectives_ && type == ConnType::Collective &&
        len >= (64u << 10) && remote.ipv4 == self_.ipv4 &&
        (flags & msgflag::ShmRef) == 0) {
        shm_payload = "/dev/shm/kfc-" + std::to_string(self_.key()) + "-" +
                      std::to_string(shm_seq_.fetch_add(1));
        FILE *f = std::fopen(shm_payload.c_str(), "wb");
        bool ok = f != nullptr;
        if (f) {
            ok = std::fwrite(data, 1, len, f) == len;
            std::fclose(f);
        }
        if (ok) {
            data = shm_payload.data();
            len = shm_payload.size();
            flags |= msgflag::ShmRef;
        } else {
            if (f) ::unlink(shm_payload.c_str());
            shm_payload.clear();  // fall back to inline payload
        }
    }
    for (int attempt = 0; attempt < 2; ++attempt) {
        auto conn = get_conn(remote, type);
        if (conn->send_frame(name, flags, data, len)) {
            std::lock_guard<std::mutex> lk(mu_);
            egress_[remote.key()] += len + name.size() + 16;
            return;
        }
        // drop broken conn, retry once
        std::lock_guard<std::mutex> lk(mu_);
        auto key = std::make_pair(remote.key(), (uint8_t)type);
        auto it = pool_.find(key);
        if (it != pool_.end() && it->second == conn) pool_.erase(it);
    }
    throw std::runtime_error("send to " + remote.str() + " failed");
}

int64_t Client::ping(const PeerID &remote, int timeout_ms)
{
    try {
        auto conn = get_conn(remote, ConnType::Ping, timeout_ms);
        auto t0 = std::chrono::steady_clock::now();
        std::lock_guard<std::mutex> lk(conn->write_mutex());
        // raw frame write + read on this duplex conn
        uint32_t name_len = 4, flags = 0;
        uint64_t data_len = 0;
        uint8_t buf[4 + 4 + 4 + 8];
        std::memcpy(buf, &name_len, 4);
        std::memcpy(buf + 4, "ping", 4);
        std::memcpy(buf + 8, &flags, 4);
        std::memcpy(buf + 12, &data_len, 8);
        if (!write_all(conn->fd(), buf, sizeof(buf))) return -1;
        Frame f;
        if (!conn->read_frame(f)) return -1;
        auto t1 = std::chrono::steady_clock::now();
        return std::chrono::duration_cast<std::chrono::microseconds>(t1 - t0)
            .count();
    } catch (...) {
        return -1;
    }
}

bool Client::wait(const PeerID &remote, int timeout_ms, int poll_ms)
{
    auto deadline = std::chrono::steady_clock::now() +
                    std::chrono::milliseconds(timeout_ms);
    while (std::chrono::steady_clock::now() < deadline) {
        if (ping(remote, poll_ms) >= 0) return true;
        std::this_thread::sleep_for(std::chrono::milliseconds(poll_ms));
    }
    return false;
}

void Client::reset(const std::vector<PeerID> &keeps, uint32_t token)
{
    token_.store(token);
    std::lock_guard<std::mutex> lk(mu_);
    for (auto it = pool_.begin(); it != pool_.end();) {
        bool keep = false;
        for (const auto &k : keeps) {
            if (k.key() == it->first.first) {
                keep = true;
                break;
            }
        }
        if (!keep) {
            it->second->close_fd();
            it = pool_.erase(it);
        } else {
            ++it;
        }
    }
}

std::map<uint64_t, uint64_t> Server::ingress_all() const
{
    std::lock_guard<std::mutex> lk(mu_);
    return ingress_;
}

uint64_t Client::egress_bytes(const PeerID &remote) const
{
    std::lock_guard<std::mutex> lk(mu_);
    auto it = egress_.find(remote.key());
    return it == egress_.end() ? 0 : it->second;
}

std::map<uint64_t, uint64_t> Client::egress_all() const
{
    std::lock_guard<std::mutex> lk(mu_);
    return egress_;
}

}  // namespace kf
