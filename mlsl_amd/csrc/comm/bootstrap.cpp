#include "bootstrap.hpp"

#include <arpa/inet.h>
#include <fcntl.h>
#include <netdb.h>
#include <poll.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <sys/socket.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <cstdlib>
#include <cstring>
#include <thread>

#include "../core/config.hpp"
#include "../core/log.hpp"
#include "../core/types.hpp"

namespace mlsl {

static int EnvInt(const char* name, int dflt) {
    if (const char* e = std::getenv(name)) return std::atoi(e);
    return dflt;
}

int Bootstrap::EnvRank() {
    int r = EnvInt("MLSL_RANK", -1);
    if (r < 0) r = EnvInt("RANK", -1);
    return r < 0 ? 0 : r;
}

int Bootstrap::EnvSize() {
    int s = EnvInt("MLSL_SIZE", -1);
    if (s < 0) s = EnvInt("WORLD_SIZE", -1);
    return s < 1 ? 1 : s;
}

static std::string EnvMasterAddr() {
    if (const char* e = std::getenv("MLSL_MASTER_ADDR")) return e;
    if (const char* e = std::getenv("MASTER_ADDR")) return e;
    return "127.0.0.1";
}

static int EnvRendezvousPort() {
    int p = EnvInt("MLSL_PORT", -1);
    if (p > 0) return p;
    int mp = EnvInt("MASTER_PORT", -1);
    if (mp > 0) return mp + 1;  // coexist with a torch.distributed store
    return 28650;
}

int TcpListen(const char* host, int port, int backlog, int* bound_port) {
    int fd = ::socket(AF_INET, SOCK_STREAM, 0);
    MLSL_CHECK(fd >= 0, "socket() failed");
    int one = 1;
    ::setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_port = htons(static_cast<uint16_t>(port));
    addr.sin_addr.s_addr = host ? ::inet_addr(host) : INADDR_ANY;
    if (::bind(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) != 0) {
        ::close(fd);
        MLSL_THROW(std::string("bind failed on port ") + std::to_string(port) +
                   ": " + std::strerror(errno));
    }
    MLSL_CHECK(::listen(fd, backlog) == 0, "listen failed");
    if (bound_port) {
        sockaddr_in got{};
        socklen_t gl = sizeof(got);
        ::getsockname(fd, reinterpret_cast<sockaddr*>(&got), &gl);
        *bound_port = ntohs(got.sin_port);
    }
    return fd;
}

int TcpConnectRetry(const std::string& host, int port, int timeout_sec) {
    auto deadline = std::chrono::steady_clock::now() + std::chrono::seconds(timeout_sec);
    for (;;) {
        int fd = ::socket(AF_INET, SOCK_STREAM, 0);
        MLSL_CHECK(fd >= 0, "socket() failed");
        sockaddr_in addr{};
        addr.sin_family = AF_INET;
        addr.sin_port = htons(static_cast<uint16_t>(port));
        if (::inet_pton(AF_INET, host.c_str(), &addr.sin_addr) != 1) {
            hostent* he = ::gethostbyname(host.c_str());
            MLSL_CHECK(he && he->h_addr_list[0], "cannot resolve " + host);
            std::memcpy(&addr.sin_addr, he->h_addr_list[0], sizeof(addr.sin_addr));
        }
        if (::connect(fd, reinterpret_cast<sockaddr*>(&addr), sizeof(addr)) == 0) {
            TcpSetNoDelay(fd);
            return fd;
        }
        ::close(fd);
        if (std::chrono::steady_clock::now() > deadline)
            MLSL_THROW("connect timeout to " + host + ":" + std::to_string(port));
        std::this_thread::sleep_for(std::chrono::milliseconds(50));
    }
}

void TcpSendAll(int fd, const void* buf, size_t len) {
    const char* p = static_cast<const char*>(buf);
    while (len) {
        ssize_t n = ::send(fd, p, len, MSG_NOSIGNAL);
        if (n < 0 && (errno == EINTR || errno == EAGAIN)) continue;
        MLSL_CHECK(n > 0, std::string("send failed: ") + std::strerror(errno));
        p += n;
        len -= static_cast<size_t>(n);
    }
}

void TcpRecvAll(int fd, void* buf, size_t len) {
    char* p = static_cast<char*>(buf);
    while (len) {
        ssize_t n = ::recv(fd, p, len, 0);
        if (n < 0 && (errno == EINTR || errno == EAGAIN)) continue;
        MLSL_CHECK(n > 0, std::string("recv failed/closed: ") + std::strerror(errno));
        p += n;
        len -= static_cast<size_t>(n);
    }
}

void TcpSetNonBlocking(int fd, bool nb) {
    int fl = ::fcntl(fd, F_GETFL, 0);
    ::fcntl(fd, F_SETFL, nb ? (fl | O_NONBLOCK) : (fl & ~O_NONBLOCK));
}

void TcpSetNoDelay(int fd) {
    int one = 1;
    ::setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
    // Large socket buffers: the mesh streams multi-MB collective segments;
    // default 200KB-ish buffers stall the nonblocking sender.
    int buf = 4 << 20;
    ::setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &buf, sizeof(buf));
    ::setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &buf, sizeof(buf));
}

Bootstrap::Bootstrap(int rank, int size) {
    rank_ = rank >= 0 ? rank : EnvRank();
    size_ = size >= 1 ? size : EnvSize();
    if (size_ == 1) return;

    const std::string master = EnvMasterAddr();
    const int port = EnvRendezvousPort();
    const int timeout = GlobalConfig().timeout_sec;

    if (rank_ == 0) {
        root_listen_ = TcpListen(nullptr, port, size_, nullptr);
        socks_.assign(size_, -1);
        // Deadline on the accept side too (the connect side already
        // retries against MLSL_TIMEOUT): a rank that dies before dialing
        // in must fail Init loudly, not hang rank 0 forever.
        const auto deadline = std::chrono::steady_clock::now() +
                              std::chrono::seconds(timeout);
        for (int i = 1; i < size_; ++i) {
            pollfd pf{root_listen_, POLLIN, 0};
            const auto left = std::chrono::duration_cast<std::chrono::milliseconds>(
                deadline - std::chrono::steady_clock::now());
            int pr = ::poll(&pf, 1, static_cast<int>(std::max<long>(0, left.count())));
            MLSL_CHECK(pr > 0, "bootstrap timeout: only " + std::to_string(i) +
                                   "/" + std::to_string(size_) +
                                   " ranks arrived within MLSL_TIMEOUT");
            int fd = ::accept(root_listen_, nullptr, nullptr);
            MLSL_CHECK(fd >= 0, "accept failed");
            TcpSetNoDelay(fd);
            uint32_t peer = 0;
            TcpRecvAll(fd, &peer, sizeof(peer));
            MLSL_CHECK(peer > 0 && peer < static_cast<uint32_t>(size_) &&
                       socks_[peer] == -1, "bad hello rank");
            socks_[peer] = fd;
        }
    } else {
        int fd = TcpConnectRetry(master, port, timeout);
        uint32_t me = static_cast<uint32_t>(rank_);
        TcpSendAll(fd, &me, sizeof(me));
        socks_.assign(1, fd);
    }
    MLSL_LOG(DEBUG, "bootstrap up: rank %d/%d via %s:%d", rank_, size_,
             master.c_str(), port);
}

Bootstrap::~Bootstrap() {
    for (int fd : socks_)
        if (fd >= 0) ::close(fd);
    if (root_listen_ >= 0) ::close(root_listen_);
}

void Bootstrap::Allgather(const void* mine, size_t len, void* out) {
    char* o = static_cast<char*>(out);
    if (size_ == 1) {
        std::memcpy(o, mine, len);
        return;
    }
    if (rank_ == 0) {
        std::memcpy(o, mine, len);
        for (int i = 1; i < size_; ++i) TcpRecvAll(socks_[i], o + i * len, len);
        for (int i = 1; i < size_; ++i) TcpSendAll(socks_[i], o, size_ * len);
    } else {
        TcpSendAll(socks_[0], mine, len);
        TcpRecvAll(socks_[0], o, size_ * len);
    }
}

void Bootstrap::Barrier() {
    uint8_t b = 1;
    std::vector<uint8_t> all(static_cast<size_t>(size_));
    Allgather(&b, 1, all.data());
}

}  // namespace mlsl
