// Netbench: TCP request/response benchmark between service instances.
//
// Reference analogue: LocalWorker netbench engine
// (/root/reference/source/workers/LocalWorker.cpp:626-881 connection setup,
// :7789-8064 transfer loops; toolkits/net/BasicSocket). Semantics: server
// instances accept one connection per client worker; each client worker
// sends its per-thread byte budget in blockSize chunks and waits for a
// respSize reply per chunk (round-trip latency = IO latency). Independent
// implementation: the server uses one poll() loop per worker over its
// connection subset.

#pragma once

#include <arpa/inet.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <unistd.h>

#include <cerrno>
#include <cstring>
#include <string>
#include <vector>

#include "common.h"

namespace eb {

inline void setSockBufs(int fd, int sendBuf, int recvBuf)
{
    if (sendBuf) setsockopt(fd, SOL_SOCKET, SO_SNDBUF, &sendBuf, sizeof(sendBuf));
    if (recvBuf) setsockopt(fd, SOL_SOCKET, SO_RCVBUF, &recvBuf, sizeof(recvBuf));
    int one = 1;
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
}

inline int netListen(int port, int backlog)
{
    int fd = socket(AF_INET, SOCK_STREAM, 0);
    if (fd < 0) throw WorkerError("netbench: socket() failed");
    int one = 1;
    setsockopt(fd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));
    sockaddr_in addr{};
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = INADDR_ANY;
    addr.sin_port = htons((uint16_t)port);
    if (bind(fd, (sockaddr*)&addr, sizeof(addr))) {
        close(fd);
        throw WorkerError("netbench: bind to port " + std::to_string(port) +
                          " failed: " + strerror(errno));
    }
    if (listen(fd, backlog)) {
        close(fd);
        throw WorkerError("netbench: listen failed");
    }
    return fd;
}

// connect with retry (client side; reference retries while services come up);
// bindDev non-empty binds the outgoing socket to that device (--netdevs)
inline int netConnect(const std::string& host, int port, const std::string& bindDev,
                      int timeoutSecs, const std::atomic<bool>& interruptFlag)
{
    struct addrinfo hints{}, *res = nullptr;
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    std::string portStr = std::to_string(port);

    auto deadline = std::chrono::steady_clock::now() + std::chrono::seconds(timeoutSecs);
    for (;;) {
        if (interruptFlag.load(std::memory_order_relaxed))
            throw InterruptedError();
        if (getaddrinfo(host.c_str(), portStr.c_str(), &hints, &res) == 0) {
            int fd = socket(res->ai_family, res->ai_socktype, res->ai_protocol);
            if (fd >= 0 && !bindDev.empty())
                setsockopt(fd, SOL_SOCKET, SO_BINDTODEVICE, bindDev.c_str(),
                           (socklen_t)bindDev.size());
            if (fd >= 0 && connect(fd, res->ai_addr, res->ai_addrlen) == 0) {
                freeaddrinfo(res);
                return fd;
            }
            if (fd >= 0) close(fd);
            freeaddrinfo(res);
            res = nullptr;
        }
        if (std::chrono::steady_clock::now() > deadline)
            throw WorkerError("netbench: connect to " + host + ":" + portStr +
                              " failed: " + strerror(errno));
        usleep(100 * 1000);
    }
}

inline bool sendExact(int fd, const char* buf, uint64_t len)
{
    while (len) {
        ssize_t n = send(fd, buf, len, MSG_NOSIGNAL);
        if (n <= 0) {
            if (n < 0 && errno == EINTR) continue;
            return false;
        }
        buf += n;
        len -= (uint64_t)n;
    }
    return true;
}

// recv exactly len bytes; returns false on EOF/error
inline bool recvExact(int fd, char* buf, uint64_t len)
{
    while (len) {
        ssize_t n = recv(fd, buf, len, 0);
        if (n <= 0) {
            if (n < 0 && errno == EINTR) continue;
            return false;
        }
        buf += n;
        len -= (uint64_t)n;
    }
    return true;
}

// recvExact with a socket receive timeout (set SO_RCVTIMEO first): a dead
// peer cannot hang the worker past the timeout window — EAGAIN checks the
// interrupt flag and keeps waiting (reference BasicSocket::recvExactT)
inline bool recvExactInterruptible(int fd, char* buf, uint64_t len,
                                   const std::atomic<bool>& interruptFlag)
{
    while (len) {
        ssize_t n = recv(fd, buf, len, 0);
        if (n <= 0) {
            if (n < 0 && (errno == EINTR || errno == EAGAIN || errno == EWOULDBLOCK)) {
                if (interruptFlag.load(std::memory_order_relaxed))
                    throw InterruptedError();
                continue;
            }
            return false;
        }
        buf += n;
        len -= (uint64_t)n;
    }
    return true;
}

inline void setRecvTimeout(int fd, int secs)
{
    struct timeval tv{secs, 0};
    setsockopt(fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
}

} // namespace eb
