#include "faabricamd/transport.h"

#include <arpa/inet.h>
#include <cerrno>
#include <cstring>
#include <fcntl.h>
#include <netdb.h>
#include <netinet/in.h>
#include <netinet/tcp.h>
#include <poll.h>
#include <sys/socket.h>
#include <sys/uio.h>
#include <unistd.h>

namespace faabricamd {

static std::atomic<int> portOffset{ -1 };

int getPortOffset()
{
    int v = portOffset.load(std::memory_order_relaxed);
    if (v < 0) {
        v = getEnvVarInt("FAABRIC_PORT_OFFSET", 0);
        portOffset.store(v, std::memory_order_relaxed);
    }
    return v;
}

void setPortOffset(int offset)
{
    portOffset.store(offset, std::memory_order_relaxed);
}

// ------------------------- TcpConnection ------------------------------------

TcpConnection::~TcpConnection()
{
    close();
}

TcpConnection::TcpConnection(TcpConnection&& o) noexcept
{
    fd = o.fd;
    o.fd = -1;
}

TcpConnection& TcpConnection::operator=(TcpConnection&& o) noexcept
{
    if (this != &o) {
        close();
        fd = o.fd;
        o.fd = -1;
    }
    return *this;
}

void TcpConnection::close()
{
    if (fd >= 0) {
        ::shutdown(fd, SHUT_RDWR);
        ::close(fd);
        fd = -1;
    }
}

static void setCommonSockOpts(int fd)
{
    int one = 1;
    // Data-plane latency options mirroring the reference's tuned TCP
    // (src/transport/tcp/SocketOptions.cpp:10-156)
    setsockopt(fd, IPPROTO_TCP, TCP_NODELAY, &one, sizeof(one));
#ifdef TCP_QUICKACK
    setsockopt(fd, IPPROTO_TCP, TCP_QUICKACK, &one, sizeof(one));
#endif
}

static std::string resolveHost(const std::string& host)
{
    if (host == "localhost") {
        return "127.0.0.1";
    }
    // Try as dotted quad first
    struct in_addr addr;
    if (inet_pton(AF_INET, host.c_str(), &addr) == 1) {
        return host;
    }
    struct addrinfo hints;
    std::memset(&hints, 0, sizeof(hints));
    hints.ai_family = AF_INET;
    hints.ai_socktype = SOCK_STREAM;
    struct addrinfo* res = nullptr;
    if (getaddrinfo(host.c_str(), nullptr, &hints, &res) != 0 ||
        res == nullptr) {
        throw FaabricException("cannot resolve host: " + host);
    }
    char buf[INET_ADDRSTRLEN];
    auto* sa = reinterpret_cast<struct sockaddr_in*>(res->ai_addr);
    inet_ntop(AF_INET, &sa->sin_addr, buf, sizeof(buf));
    std::string out(buf);
    freeaddrinfo(res);
    return out;
}

TcpConnection TcpConnection::dial(const std::string& host,
                                  int port,
                                  int timeoutMs)
{
    std::string ip = resolveHost(host);

    // Retry dialling for the timeout window: servers may still be starting
    int64_t deadline = getGlobalClockEpochMillis() + timeoutMs;
    int lastErr = 0;
    while (true) {
        int fd = ::socket(AF_INET, SOCK_STREAM, 0);
        if (fd < 0) {
            throw FaabricException("socket() failed");
        }
        setCommonSockOpts(fd);

        struct sockaddr_in addr;
        std::memset(&addr, 0, sizeof(addr));
        addr.sin_family = AF_INET;
        addr.sin_port = htons((uint16_t)port);
        inet_pton(AF_INET, ip.c_str(), &addr.sin_addr);

        if (::connect(fd, (struct sockaddr*)&addr, sizeof(addr)) == 0) {
            return TcpConnection(fd);
        }
        lastErr = errno;
        ::close(fd);
        if (getGlobalClockEpochMillis() >= deadline) {
            break;
        }
        usleep(20 * 1000);
    }
    throw FaabricException("connect to " + ip + ":" + std::to_string(port) +
                           " failed: " + strerror(lastErr));
}

void TcpConnection::sendAll(const void* data, size_t len)
{
    const char* p = (const char*)data;
    while (len > 0) {
        ssize_t n = ::send(fd, p, len, MSG_NOSIGNAL);
        if (n < 0) {
            if (errno == EINTR) {
                continue;
            }
            throw SocketClosedException("send failed: " +
                                        std::string(strerror(errno)));
        }
        p += n;
        len -= (size_t)n;
    }
}

void TcpConnection::recvAll(void* data, size_t len)
{
    char* p = (char*)data;
    while (len > 0) {
        ssize_t n = ::recv(fd, p, len, 0);
        if (n == 0) {
            throw SocketClosedException("peer closed");
        }
        if (n < 0) {
            if (errno == EINTR) {
                continue;
            }
            throw SocketClosedException("recv failed: " +
                                        std::string(strerror(errno)));
        }
        p += n;
        len -= (size_t)n;
    }
}

void TcpConnection::sendFrame(uint8_t code,
                              const void* body,
                              size_t len,
                              uint32_t seq)
{
    sendFrame2(code, body, len, nullptr, 0, seq);
}

void TcpConnection::sendFrame2(uint8_t code,
                               const void* a,
                               size_t lenA,
                               const void* b,
                               size_t lenB,
                               uint32_t seq)
{
    WireHeader hdr;
    hdr.code = code;
    hdr.seq = seq;
    hdr.size = lenA + lenB;

    struct iovec iov[3];
    int iovCount = 1;
    iov[0].iov_base = &hdr;
    iov[0].iov_len = sizeof(hdr);
    if (lenA > 0) {
        iov[iovCount].iov_base = const_cast<void*>(a);
        iov[iovCount].iov_len = lenA;
        iovCount++;
    }
    if (lenB > 0) {
        iov[iovCount].iov_base = const_cast<void*>(b);
        iov[iovCount].iov_len = lenB;
        iovCount++;
    }

    size_t total = sizeof(hdr) + lenA + lenB;
    struct msghdr msg;
    std::memset(&msg, 0, sizeof(msg));
    msg.msg_iov = iov;
    msg.msg_iovlen = (size_t)iovCount;

    size_t sent = 0;
    while (sent < total) {
        ssize_t n = ::sendmsg(fd, &msg, MSG_NOSIGNAL);
        if (n < 0) {
            if (errno == EINTR) {
                continue;
            }
            throw SocketClosedException("sendmsg failed: " +
                                        std::string(strerror(errno)));
        }
        sent += (size_t)n;
        if (sent >= total) {
            break;
        }
        // Advance iovecs past what was sent
        size_t skip = (size_t)n;
        while (skip > 0 && msg.msg_iovlen > 0) {
            if (skip >= msg.msg_iov[0].iov_len) {
                skip -= msg.msg_iov[0].iov_len;
                msg.msg_iov++;
                msg.msg_iovlen--;
            } else {
                msg.msg_iov[0].iov_base =
                  (char*)msg.msg_iov[0].iov_base + skip;
                msg.msg_iov[0].iov_len -= skip;
                skip = 0;
            }
        }
    }
}

bool TcpConnection::recvFrame(WireHeader& hdr, std::string& body)
{
    // Read the header; an orderly close before any byte means "done"
    char* p = (char*)&hdr;
    size_t need = sizeof(hdr);
    while (need > 0) {
        ssize_t n = ::recv(fd, p, need, 0);
        if (n == 0) {
            if (need == sizeof(hdr)) {
                return false; // clean close at frame boundary
            }
            throw SocketClosedException("peer closed mid-header");
        }
        if (n < 0) {
            if (errno == EINTR) {
                continue;
            }
            if (errno == ECONNRESET && need == sizeof(hdr)) {
                return false;
            }
            throw SocketClosedException("recv failed: " +
                                        std::string(strerror(errno)));
        }
        p += n;
        need -= (size_t)n;
    }

    // Sanity-cap the frame before allocating: a corrupt or hostile
    // header must drop the connection, not OOM the process
    static const uint64_t maxFrame =
      (uint64_t)getEnvVarInt("FAABRIC_MAX_FRAME_MB", 1024) * 1024 * 1024;
    if (hdr.size > maxFrame) {
        throw SocketClosedException(
          "frame size " + std::to_string(hdr.size) + " exceeds cap");
    }
    body.resize(hdr.size);
    if (hdr.size > 0) {
        recvAll(body.data(), hdr.size);
    }
    return true;
}

// ------------------------- TcpListener --------------------------------------

TcpListener::~TcpListener()
{
    close();
}

void TcpListener::listen(int port, int backlog)
{
    int newFd = ::socket(AF_INET, SOCK_STREAM, 0);
    if (newFd < 0) {
        throw FaabricException("socket() failed");
    }
    int one = 1;
    setsockopt(newFd, SOL_SOCKET, SO_REUSEADDR, &one, sizeof(one));

    struct sockaddr_in addr;
    std::memset(&addr, 0, sizeof(addr));
    addr.sin_family = AF_INET;
    addr.sin_addr.s_addr = htonl(INADDR_ANY);
    addr.sin_port = htons((uint16_t)port);
    if (::bind(newFd, (struct sockaddr*)&addr, sizeof(addr)) != 0) {
        int err = errno;
        ::close(newFd);
        throw FaabricException("bind port " + std::to_string(port) +
                               " failed: " + strerror(err));
    }
    if (::listen(newFd, backlog) != 0) {
        int err = errno;
        ::close(newFd);
        throw FaabricException("listen failed: " +
                               std::string(strerror(err)));
    }
    fd.store(newFd);
    port_ = port;
}

std::optional<TcpConnection> TcpListener::accept()
{
    while (true) {
        int f = fd.load();
        if (f < 0) {
            return std::nullopt;
        }
        int cfd = ::accept(f, nullptr, nullptr);
        if (cfd >= 0) {
            setCommonSockOpts(cfd);
            return TcpConnection(cfd);
        }
        if (errno == EINTR) {
            continue;
        }
        // Listener closed or fatal error
        return std::nullopt;
    }
}

void TcpListener::close()
{
    int old = fd.exchange(-1);
    if (old >= 0) {
        ::shutdown(old, SHUT_RDWR);
        ::close(old);
    }
}

// ------------------------- MessageEndpointServer ----------------------------

MessageEndpointServer::MessageEndpointServer(int asyncPortIn,
                                             int syncPortIn,
                                             std::string nameIn)
  : asyncPort(asyncPortIn)
  , syncPort(syncPortIn)
  , name(std::move(nameIn))
{}

MessageEndpointServer::~MessageEndpointServer()
{
    stop();
}

void MessageEndpointServer::start()
{
    if (running.load()) {
        return;
    }
    int off = getPortOffset();
    asyncListener.listen(asyncPort + off);
    syncListener.listen(syncPort + off);
    running.store(true);

    asyncAcceptThread =
      std::thread([this] { acceptLoop(asyncListener, false); });
    syncAcceptThread = std::thread([this] { acceptLoop(syncListener, true); });
    FAM_DEBUG("%s server listening on %d/%d (+%d)",
              name.c_str(),
              asyncPort,
              syncPort,
              off);
}

void MessageEndpointServer::stop()
{
    if (!running.exchange(false)) {
        return;
    }
    asyncListener.close();
    syncListener.close();
    {
        // Unblock handler threads parked in recv on live connections
        std::lock_guard<std::mutex> lock(connThreadsMx);
        for (int fd : activeConnFds) {
            ::shutdown(fd, SHUT_RDWR);
        }
    }
    if (asyncAcceptThread.joinable()) {
        asyncAcceptThread.join();
    }
    if (syncAcceptThread.joinable()) {
        syncAcceptThread.join();
    }
    std::vector<std::thread> toJoin;
    {
        std::lock_guard<std::mutex> lock(connThreadsMx);
        toJoin.swap(connThreads);
    }
    for (auto& t : toJoin) {
        if (t.joinable()) {
            t.join();
        }
    }
}

void MessageEndpointServer::acceptLoop(TcpListener& listener, bool isSync)
{
    while (running.load()) {
        auto conn = listener.accept();
        if (!conn.has_value()) {
            break;
        }
        std::lock_guard<std::mutex> lock(connThreadsMx);
        if (!running.load()) {
            break;
        }
        connThreads.emplace_back(
          [this, c = std::move(*conn), isSync]() mutable {
              connectionLoop(std::move(c), isSync);
          });
    }
}

void MessageEndpointServer::connectionLoop(TcpConnection conn, bool isSync)
{
    {
        std::lock_guard<std::mutex> lock(connThreadsMx);
        activeConnFds.insert(conn.rawFd());
    }
    int thisFd = conn.rawFd();
    WireHeader hdr;
    std::string body;
    while (running.load()) {
        try {
            if (!conn.recvFrame(hdr, body)) {
                break;
            }
        } catch (const SocketClosedException&) {
            break;
        }
        try {
            if (isSync) {
                std::string resp = doSyncRecv(hdr.code, body);
                conn.sendFrame(0, resp.data(), resp.size(), hdr.seq);
            } else {
                doAsyncRecv(hdr.code, body, hdr.seq);
            }
        } catch (const std::exception& e) {
            FAM_ERROR("%s server handler error (code %d): %s",
                      name.c_str(),
                      (int)hdr.code,
                      e.what());
            if (isSync) {
                // Error marker response: code 0xff
                try {
                    std::string err = e.what();
                    conn.sendFrame(0xff, err.data(), err.size(), hdr.seq);
                } catch (const SocketClosedException&) {
                    break;
                }
            }
        }
    }
    std::lock_guard<std::mutex> lock(connThreadsMx);
    activeConnFds.erase(thisFd);
}

// ------------------------- MessageEndpointClient ----------------------------

MessageEndpointClient::MessageEndpointClient(std::string hostIn,
                                             int asyncPortIn,
                                             int syncPortIn)
  : host(std::move(hostIn))
  , asyncPort(asyncPortIn)
  , syncPort(syncPortIn)
{}

void parseHostIdentity(const std::string& identity,
                       std::string& ipOut,
                       int& offsetOut)
{
    auto at = identity.find('@');
    if (at == std::string::npos) {
        ipOut = identity;
        offsetOut = 0;
    } else {
        ipOut = identity.substr(0, at);
        offsetOut = atoi(identity.c_str() + at + 1);
    }
}

TcpConnection& MessageEndpointClient::ensure(bool sync)
{
    TcpConnection& conn = sync ? syncConn : asyncConn;
    if (!conn.isOpen()) {
        // Host identities may carry a port offset ("ip@offset") so several
        // single-GPU worker processes can share one node's IP
        std::string ip;
        int offset = 0;
        parseHostIdentity(host, ip, offset);
        int port = (sync ? syncPort : asyncPort) + offset;
        conn = TcpConnection::dial(ip, port);
    }
    return conn;
}

void MessageEndpointClient::asyncSend(uint8_t code, const std::string& body)
{
    asyncSend(code, body.data(), body.size());
}

void MessageEndpointClient::asyncSend(uint8_t code,
                                      const void* body,
                                      size_t len)
{
    asyncSendSeq(code, body, len, 0);
}

void MessageEndpointClient::asyncSendSeq(uint8_t code,
                                         const void* body,
                                         size_t len,
                                         uint32_t seq)
{
    std::lock_guard<std::mutex> lock(asyncMx);
    try {
        ensure(false).sendFrame(code, body, len, seq);
    } catch (const SocketClosedException&) {
        // One reconnect attempt
        asyncConn.close();
        ensure(false).sendFrame(code, body, len, seq);
    }
}

std::string MessageEndpointClient::syncSend(uint8_t code,
                                            const std::string& body)
{
    std::lock_guard<std::mutex> lock(syncMx);
    WireHeader hdr;
    std::string resp;
    try {
        TcpConnection& conn = ensure(true);
        conn.sendFrame(code, body.data(), body.size());
        if (!conn.recvFrame(hdr, resp)) {
            throw SocketClosedException("server closed during sync call");
        }
    } catch (const SocketClosedException&) {
        syncConn.close();
        TcpConnection& conn = ensure(true);
        conn.sendFrame(code, body.data(), body.size());
        if (!conn.recvFrame(hdr, resp)) {
            throw SocketClosedException("server closed during sync retry");
        }
    }
    if (hdr.code == 0xff) {
        throw FaabricException("remote error from " + host + ": " + resp);
    }
    return resp;
}

} // namespace faabricamd
