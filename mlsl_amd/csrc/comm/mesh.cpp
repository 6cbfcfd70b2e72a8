#include "mesh.hpp"

#include <netinet/in.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cstring>
#include <thread>

#include "../core/config.hpp"
#include "../core/log.hpp"
#include "../core/types.hpp"
#include "bootstrap.hpp"

namespace mlsl {

Channel::Channel(int fd) : fd_(fd) {
    TcpSetNonBlocking(fd_, true);
    TcpSetNoDelay(fd_);
}

Channel::~Channel() {
    if (fd_ >= 0) ::close(fd_);
}

bool Channel::StartSend(uint64_t tag, const void* buf, size_t len, bool* done_flag) {
    if (send_active_) return false;
    send_active_ = true;
    std::memcpy(send_hdr_, &tag, 8);
    uint64_t l = len;
    std::memcpy(send_hdr_ + 8, &l, 8);
    send_hdr_sent_ = 0;
    send_payload_ = static_cast<const uint8_t*>(buf);
    send_len_ = len;
    send_sent_ = 0;
    send_done_flag_ = done_flag;
    ProgressSend();
    return true;
}

void Channel::ProgressSend() {
    if (!send_active_) return;
    while (send_hdr_sent_ < 16) {
        ssize_t n = ::send(fd_, send_hdr_ + send_hdr_sent_, 16 - send_hdr_sent_,
                           MSG_NOSIGNAL);
        if (n < 0) {
            if (errno == EAGAIN || errno == EWOULDBLOCK || errno == EINTR) return;
            MLSL_THROW(std::string("mesh send failed: ") + std::strerror(errno));
        }
        send_hdr_sent_ += static_cast<size_t>(n);
    }
    while (send_sent_ < send_len_) {
        ssize_t n = ::send(fd_, send_payload_ + send_sent_, send_len_ - send_sent_,
                           MSG_NOSIGNAL);
        if (n < 0) {
            if (errno == EAGAIN || errno == EWOULDBLOCK || errno == EINTR) return;
            MLSL_THROW(std::string("mesh send failed: ") + std::strerror(errno));
        }
        send_sent_ += static_cast<size_t>(n);
    }
    send_active_ = false;
    if (send_done_flag_) *send_done_flag_ = true;
    send_done_flag_ = nullptr;
}

void Channel::PostRecv(uint64_t tag, void* buf, size_t len, bool* done_flag) {
    // Early arrival?
    auto it = unexpected_.find(tag);
    if (it != unexpected_.end()) {
        UnexpectedMsg& um = it->second;
        MLSL_CHECK(um.data.size() == len, "posted recv size != arrived size");
        std::memcpy(buf, um.data.data(), um.got);
        if (um.complete) {
            if (done_flag) *done_flag = true;
            unexpected_.erase(it);
            return;
        }
        // Partially arrived: switch the in-flight target to the user buffer.
        PendingRecv pr{static_cast<uint8_t*>(buf), len, um.got, done_flag};
        auto res = posted_.emplace(tag, pr);
        MLSL_CHECK(res.second, "duplicate posted recv tag");
        if (recv_in_msg_ && recv_unexp_ == &um) {
            recv_unexp_ = nullptr;
            recv_cur_ = &res.first->second;
        }
        unexpected_.erase(it);
        return;
    }
    PendingRecv pr{static_cast<uint8_t*>(buf), len, 0, done_flag};
    MLSL_CHECK(posted_.emplace(tag, pr).second, "duplicate posted recv tag");
}

void Channel::ProgressRecv() {
    if (peer_closed_) return;
    for (;;) {
        if (!recv_in_msg_) {
            while (recv_hdr_got_ < 16) {
                ssize_t n = ::recv(fd_, recv_hdr_ + recv_hdr_got_, 16 - recv_hdr_got_, 0);
                if (n < 0) {
                    if (errno == EAGAIN || errno == EWOULDBLOCK || errno == EINTR) return;
                    MLSL_THROW(std::string("mesh recv failed: ") + std::strerror(errno));
                }
                if (n == 0) {
                    // Peer finalized. Benign unless we still expect data
                    // (Context::Finalize barriers before closing, so this
                    // firing means a request was abandoned or a peer died).
                    if (recv_hdr_got_ != 0 || !posted_.empty()) {
                        std::string tags;
                        for (const auto& kv : posted_)
                            tags += std::to_string(kv.first) + "(" +
                                    std::to_string(kv.second.got) + "/" +
                                    std::to_string(kv.second.len) + ") ";
                        MLSL_THROW(
                            "mesh peer closed with receives outstanding: "
                            "hdr_got=" + std::to_string(recv_hdr_got_) +
                            " posted=[" + tags + "] unexpected=" +
                            std::to_string(unexpected_.size()));
                    }
                    peer_closed_ = true;
                    return;
                }
                recv_hdr_got_ += static_cast<size_t>(n);
            }
            std::memcpy(&recv_tag_, recv_hdr_, 8);
            std::memcpy(&recv_len_, recv_hdr_ + 8, 8);
            recv_hdr_got_ = 0;
            recv_in_msg_ = true;
            recv_cur_ = nullptr;
            recv_unexp_ = nullptr;
            auto it = posted_.find(recv_tag_);
            if (it != posted_.end()) {
                MLSL_CHECK(it->second.len == recv_len_, "recv size mismatch");
                recv_cur_ = &it->second;
            } else {
                UnexpectedMsg um;
                um.data.resize(recv_len_);
                recv_unexp_ = &unexpected_.emplace(recv_tag_, std::move(um))->second;
            }
        }
        // Payload.
        for (;;) {
            uint8_t* dst;
            size_t got;
            if (recv_cur_) {
                dst = recv_cur_->buf;
                got = recv_cur_->got;
            } else {
                dst = recv_unexp_->data.data();
                got = recv_unexp_->got;
            }
            if (got == recv_len_) break;
            ssize_t n = ::recv(fd_, dst + got, recv_len_ - got, 0);
            if (n < 0) {
                if (errno == EAGAIN || errno == EWOULDBLOCK || errno == EINTR) return;
                MLSL_THROW(std::string("mesh recv failed: ") + std::strerror(errno));
            }
            MLSL_CHECK(n != 0, "mesh peer closed mid-message");
            if (recv_cur_) recv_cur_->got += static_cast<size_t>(n);
            else recv_unexp_->got += static_cast<size_t>(n);
        }
        // Message complete.
        if (recv_cur_) {
            if (recv_cur_->done_flag) *recv_cur_->done_flag = true;
            posted_.erase(recv_tag_);
        } else {
            recv_unexp_->complete = true;
        }
        recv_in_msg_ = false;
        recv_cur_ = nullptr;
        recv_unexp_ = nullptr;
    }
}

void Channel::Progress() {
    ProgressSend();
    ProgressRecv();
}

Mesh::Mesh(Bootstrap& boot) : rank_(boot.Rank()), size_(boot.Size()) {
    chans_.resize(static_cast<size_t>(size_));
    if (size_ == 1) return;

    int my_port = 0;
    listen_fd_ = TcpListen(nullptr, 0, size_, &my_port);

    // Exchange (port) — all ranks are on 127.0.0.1 or reachable via
    // MASTER-net; we publish the port and the bootstrap-known host of rank 0
    // is reused (single-node focus; multi-node would publish the interface
    // address here).
    struct Addr { uint32_t port; char host[60]; };
    Addr mine{};
    mine.port = static_cast<uint32_t>(my_port);
    std::snprintf(mine.host, sizeof(mine.host), "127.0.0.1");
    if (const char* e = std::getenv("MLSL_MESH_HOST"))
        std::snprintf(mine.host, sizeof(mine.host), "%s", e);
    std::vector<Addr> all(static_cast<size_t>(size_));
    boot.Allgather(&mine, sizeof(Addr), all.data());

    // Accept from higher ranks on a helper thread while we dial lower ranks.
    const int expect = size_ - 1 - rank_;
    std::vector<int> accepted;
    std::thread acceptor([&]() {
        for (int i = 0; i < expect; ++i) {
            int fd = ::accept(listen_fd_, nullptr, nullptr);
            MLSL_CHECK(fd >= 0, "mesh accept failed");
            accepted.push_back(fd);
        }
    });
    for (int j = 0; j < rank_; ++j) {
        int fd = TcpConnectRetry(all[j].host, static_cast<int>(all[j].port),
                                 GlobalConfig().timeout_sec);
        uint32_t me = static_cast<uint32_t>(rank_);
        TcpSendAll(fd, &me, sizeof(me));
        chans_[j] = std::make_unique<Channel>(fd);
    }
    acceptor.join();
    for (int fd : accepted) {
        uint32_t peer = 0;
        TcpRecvAll(fd, &peer, sizeof(peer));
        MLSL_CHECK(peer < static_cast<uint32_t>(size_) && !chans_[peer],
                   "bad mesh hello");
        chans_[peer] = std::make_unique<Channel>(fd);
    }
    boot.Barrier();
    MLSL_LOG(DEBUG, "mesh up: rank %d/%d", rank_, size_);
}

Mesh::~Mesh() {
    if (listen_fd_ >= 0) ::close(listen_fd_);
}

bool Mesh::StartSend(int peer, uint64_t tag, const void* buf, size_t len, bool* done) {
    MLSL_CHECK(peer != rank_, "self send not routed through mesh");
    return chans_[peer]->StartSend(tag, buf, len, done);
}

void Mesh::PostRecv(int peer, uint64_t tag, void* buf, size_t len, bool* done) {
    MLSL_CHECK(peer != rank_, "self recv not routed through mesh");
    chans_[peer]->PostRecv(tag, buf, len, done);
}

void Mesh::Progress() {
    for (auto& c : chans_)
        if (c) c->Progress();
}

}  // namespace mlsl
