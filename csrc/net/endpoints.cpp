#include "endpoints.hpp"

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

void CollectiveEndpoint::on_frame(const PeerID &src, Frame &f)
{
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
    s->dst = nullptr;
    if (!s->q.empty()) {
        auto buf = std::move(s->q.front());
        s->q.pop_front();
        if (buf.size() != len)
            throw std::runtime_error("collective size mismatch on " + name);
        std::memcpy(dst, buf.data(), len);
        return;
    }
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

void BlobStore::save(const std::string &name, const void *data, size_t len)
{
    auto blob = std::make_shared<const std::vector<uint8_t>>(
        (const uint8_t *)data, (const uint8_t *)data + len);
    std::lock_guard<std::mutex> lk(mu_);
    blobs_[name] = std::move(blob);
    versions_[name]++;
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
        // reply with our stored blob over our own client conn to src
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
