#include "endpoints.hpp"

#include <unistd.h>

#include <cstdio>
#include <cstring>
#include <stdexcept>

namespace kf {

// ---------- CollectiveEndpoint ----------

std::shared_ptr<CollectiveEndpoint::Slot> CollectiveEndpoint::slot(
    const PeerID &src, const std::string &name)
{
    std::string key = std::to_string(src.key()) + "|" + name;
    std::lock_guard<std::mutex> lk(mu_);
    auto &s = slots_[key];
    if (!s) {
        s = std::make_shared<Slot>();
        if (dead_) s->dead = true;
    }
    return s;
}

bool CollectiveEndpoint::on_header(const PeerID &src,
                                   const FrameHeader &h, Conn &conn)
{
    if (h.flags & msgflag::ShmRef) {
        // colocated shm data path: body is a /dev/shm file reference;
        // read the file straight into the registered destination when one
        // is waiting
        std::string path(h.len, '\0');
        if (!conn.read_body(path.data(), h.len))
            throw std::runtime_error("short shm-ref read on " + h.name);
        FILE *fp = std::fopen(path.c_str(), "rb");
        if (!fp) throw std::runtime_error("shm chunk missing: " + path);
        std::fseek(fp, 0, SEEK_END);
        const size_t n = (size_t)std::ftell(fp);
        std::fseek(fp, 0, SEEK_SET);
        auto s = slot(src, h.name);
        std::unique_lock<std::mutex> lk(s->mu);
        if (s->dst && !s->filled && !s->filling && n == s->dst_len) {
            s->filling = true;
            lk.unlock();
            const size_t got = std::fread(s->dst, 1, n, fp);
            std::fclose(fp);
            ::unlink(path.c_str());
            lk.lock();
            s->filling = false;
            if (got != n) {
                s->dead = true;
                s->cv.notify_all();
                throw std::runtime_error("short shm chunk read: " + path);
            }
            s->filled = true;
            s->cv.notify_all();
            return true;
        }
        auto buf = pool_.get(n);
        lk.unlock();
        const size_t got = std::fread(buf.data(), 1, n, fp);
        std::fclose(fp);
        ::unlink(path.c_str());
        if (got != n)
            throw std::runtime_error("short shm chunk read: " + path);
        lk.lock();
        s->q.push_back(std::move(buf));
        s->cv.notify_all();
        return true;
    }
    auto s = slot(src, h.name);
    std::unique_lock<std::mutex> lk(s->mu);
    if (s->dst && !s->filled && !s->filling && h.len == s->dst_len) {
        // zero-copy: socket read lands in the destination buffer
        s->filling = true;
        lk.unlock();
        const bool ok = conn.read_body(s->dst, h.len);
        lk.lock();
        s->filling = false;
        if (!ok) {
            s->dead = true;
            s->cv.notify_all();
            throw std::runtime_error("short body read on " + h.name);
        }
        s->filled = true;
        s->cv.notify_all();
        return true;
    }
    auto buf = pool_.get(h.len);
    lk.unlock();
    if (!conn.read_body(buf.data(), h.len))
        throw std::runtime_error("short body read on " + h.name);
    lk.lock();
    s->q.push_back(std::move(buf));
    s->cv.notify_all();
    return true;
}

void CollectiveEndpoint::on_frame(const PeerID &src, Frame &f)
{
    if (f.flags & msgflag::ShmRef) {
        // colocated shm data path: payload is a /dev/shm file reference;
        // swap it for the file contents and unlink (single consumer)
        std::string path((const char *)f.data.data(), f.data.size());
        FILE *fp = std::fopen(path.c_str(), "rb");
        if (!fp)
            throw std::runtime_error("shm chunk missing: " + path);
        std::fseek(fp, 0, SEEK_END);
        long n = std::ftell(fp);
        std::fseek(fp, 0, SEEK_SET);
        std::vector<uint8_t> buf((size_t)n);
        size_t got = std::fread(buf.data(), 1, (size_t)n, fp);
        std::fclose(fp);
        ::unlink(path.c_str());
        if (got != (size_t)n)
            throw std::runtime_error("short shm chunk read: " + path);
        f.data = std::move(buf);
        f.flags &= ~msgflag::ShmRef;
    }
    auto s = slot(src, f.name);
    std::lock_guard<std::mutex> lk(s->mu);
    if (s->dst && !s->filled) {
        if (f.data.size() != s->dst_len)
            throw std::runtime_error("collective size mismatch on " +
                                     f.name);
        std::memcpy(s->dst, f.data.data(), f.data.size());
        s->filled = true;
        s->cv.notify_all();
        return;
    }
    s->q.push_back(std::move(f.data));
    s->cv.notify_all();
}

void CollectiveEndpoint::recv_into(const PeerID &src, const std::string &name,
                                   void *dst, size_t len)
{
    auto s = slot(src, name);
    std::unique_lock<std::mutex> lk(s->mu);
    if (!s->q.empty()) {
        auto buf = std::move(s->q.front());
        s->q.pop_front();
        if (buf.size() != len)
            throw std::runtime_error("collective size mismatch on " + name);
        std::memcpy(dst, buf.data(), len);
        pool_.put(std::move(buf));
        return;
    }
    if (s->dead) throw std::runtime_error("endpoint shut down");
    s->dst = (uint8_t *)dst;
    s->dst_len = len;
    s->filled = false;
    s->cv.wait(lk, [&] { return s->filled || !s->q.empty() || s->dead; });
    if (s->filled) {
        s->dst = nullptr;
        return;
    }
    if (!s->q.empty()) {
        s->dst = nullptr;
        auto buf = std::move(s->q.front());
        s->q.pop_front();
        if (buf.size() != len)
            throw std::runtime_error("collective size mismatch on " + name);
        std::memcpy(dst, buf.data(), len);
        pool_.put(std::move(buf));
        return;
    }
    // dead: an in-flight zero-copy body read may still target dst — wait
    // it out before the caller frees the buffer
    s->cv.wait(lk, [&] { return !s->filling; });
    s->dst = nullptr;
    if (s->filled) return;
    throw std::runtime_error("recv_into aborted (endpoint shut down) on " +
                             name);
}

std::vector<uint8_t> CollectiveEndpoint::recv(const PeerID &src,
                                              const std::string &name)
{
    auto s = slot(src, name);
    std::unique_lock<std::mutex> lk(s->mu);
    s->cv.wait(lk, [&] { return !s->q.empty() || s->dead; });
    if (s->q.empty())
        throw std::runtime_error("recv aborted (endpoint shut down) on " +
                                 name);
    auto buf = std::move(s->q.front());
    s->q.pop_front();
    return buf;
}

void CollectiveEndpoint::clear()
{
    std::lock_guard<std::mutex> lk(mu_);
    slots_.clear();
    dead_ = false;
}

void CollectiveEndpoint::shutdown()
{
    std::lock_guard<std::mutex> lk(mu_);
    dead_ = true;
    for (auto &kv : slots_) {
        std::lock_guard<std::mutex> slk(kv.second->mu);
        kv.second->dead = true;
        kv.second->cv.notify_all();
    }
}

// ---------- BlobStore ----------

BlobStore::~BlobStore()
{
    // best-effort cleanup of shm mirrors
    std::lock_guard<std::mutex> lk(mu_);
    for (auto &kv : versions_) {
        for (uint64_t v = kv.second; v > 0 && v + 2 > kv.second; --v) {
            ::unlink(shm_file(kv.first, v).c_str());
        }
    }
}

std::string BlobStore::shm_file(const std::string &name, uint64_t ver) const
{
    std::string safe = name;
    for (auto &c : safe) {
        if (c == '/' || c == '.') c = '_';
    }
    return "/dev/shm/kungfu-amd-" + std::to_string(owner_port_) + "-" +
           safe + ".v" + std::to_string(ver);
}

void BlobStore::save(const std::string &name, const void *data, size_t len)
{
    auto blob = std::make_shared<const std::vector<uint8_t>>(
        (const uint8_t *)data, (const uint8_t *)data + len);
    uint64_t ver;
    {
        std::lock_guard<std::mutex> lk(mu_);
        blobs_[name] = std::move(blob);
        ver = ++versions_[name];
    }
    // shm mirror: write-then-rename so readers only ever see whole blobs
    std::string path = shm_file(name, ver);
    std::string tmp = path + ".tmp";
    FILE *f = std::fopen(tmp.c_str(), "wb");
    if (f) {
        bool ok = std::fwrite(data, 1, len, f) == len;
        std::fclose(f);
        if (ok && std::rename(tmp.c_str(), path.c_str()) == 0) {
            std::lock_guard<std::mutex> lk(mu_);
            shm_paths_[name] = path;
        } else {
            ::unlink(tmp.c_str());
        }
    }
    if (ver > 2) ::unlink(shm_file(name, ver - 2).c_str());
}

std::string BlobStore::shm_path(const std::string &name) const
{
    std::lock_guard<std::mutex> lk(mu_);
    auto it = shm_paths_.find(name);
    return it == shm_paths_.end() ? std::string() : it->second;
}

std::shared_ptr<const std::vector<uint8_t>> BlobStore::get(
    const std::string &name) const
{
    std::lock_guard<std::mutex> lk(mu_);
    auto it = blobs_.find(name);
    return it == blobs_.end() ? nullptr : it->second;
}

uint64_t BlobStore::version(const std::string &name) const
{
    std::lock_guard<std::mutex> lk(mu_);
    auto it = versions_.find(name);
    return it == versions_.end() ? 0 : it->second;
}

// ---------- P2PEndpoint ----------

void P2PEndpoint::on_frame(const PeerID &src, Frame &f)
{
    if (f.flags & msgflag::IsRequest) {
        // Colocated peers get a shm reference (memcpy-speed pull on one
        // MI355X node); remote peers get the payload inline.
        const bool colocated = src.ipv4 == self_.ipv4;
        if (colocated) {
            std::string path = store_.shm_path(f.name);
            if (!path.empty()) {
                client_.send(src, ConnType::P2P, f.name,
                             msgflag::IsResponse | msgflag::ShmRef,
                             path.data(), path.size());
                return;
            }
        }
        auto blob = store_.get(f.name);
        if (blob) {
            client_.send(src, ConnType::P2P, f.name, msgflag::IsResponse,
                         blob->data(), blob->size());
        } else {
            client_.send(src, ConnType::P2P, f.name,
                         msgflag::IsResponse | msgflag::RequestFailed,
                         nullptr, 0);
        }
        return;
    }
    if (f.flags & msgflag::IsResponse) {
        std::shared_ptr<Waiter> w;
        {
            std::lock_guard<std::mutex> lk(mu_);
            std::string key = std::to_string(src.key()) + "|" + f.name;
            auto it = waiters_.find(key);
            if (it == waiters_.end()) return;  // timed-out waiter
            w = it->second;
            waiters_.erase(it);
        }
        std::lock_guard<std::mutex> lk(w->mu);
        w->failed = (f.flags & msgflag::RequestFailed) != 0;
        w->shm_ref = (f.flags & msgflag::ShmRef) != 0;
        w->data = std::move(f.data);
        w->done = true;
        w->cv.notify_all();
        return;
    }
    // plain save push (unused for now)
}

bool P2PEndpoint::request(const PeerID &target, const std::string &name,
                          void *dst, size_t len, int timeout_ms)
{
    auto w = std::make_shared<Waiter>();
    std::string key = std::to_string(target.key()) + "|" + name;
    {
        std::lock_guard<std::mutex> lk(mu_);
        waiters_[key] = w;
    }
    try {
        client_.send(target, ConnType::P2P, name, msgflag::IsRequest,
                     nullptr, 0);
    } catch (...) {
        std::lock_guard<std::mutex> lk(mu_);
        waiters_.erase(key);
        return false;
    }
    std::unique_lock<std::mutex> lk(w->mu);
    if (!w->cv.wait_for(lk, std::chrono::milliseconds(timeout_ms),
                        [&] { return w->done; })) {
        std::lock_guard<std::mutex> glk(mu_);
        waiters_.erase(key);
        return false;
    }
    if (w->failed) return false;
    if (w->shm_ref) {
        std::string path((const char *)w->data.data(), w->data.size());
        FILE *f = std::fopen(path.c_str(), "rb");
        if (!f) return false;
        size_t got = std::fread(dst, 1, len, f);
        // exactly len bytes and nothing more (size must match)
        bool ok = got == len && std::fgetc(f) == EOF;
        std::fclose(f);
        return ok;
    }
    if (w->data.size() != len) return false;
    std::memcpy(dst, w->data.data(), len);
    return true;
}

void P2PEndpoint::shutdown()
{
    std::lock_guard<std::mutex> lk(mu_);
    for (auto &kv : waiters_) {
        std::lock_guard<std::mutex> wlk(kv.second->mu);
        kv.second->failed = true;
        kv.second->done = true;
        kv.second->cv.notify_all();
    }
    waiters_.clear();
}

}  // namespace kf
