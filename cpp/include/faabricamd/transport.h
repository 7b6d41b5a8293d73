// Transport: framed TCP message fabric.
//
// MI355X-native replacement for the reference's nng-based endpoint layer
// (reference: src/transport/MessageEndpoint.cpp, MessageEndpointServer.h:18-95,
//  src/transport/tcp/SendSocket.cpp / RecvSocket.cpp, transport/common.h:8-29,
//  transport/Message.h:11-25). Re-designed: plain POSIX TCP with
// thread-per-connection servers — the control plane is latency- not
// bandwidth-bound on a single 8-GPU node, and the bulk data plane is
// RCCL/xGMI peer copies, not this layer.
//
// Wire framing keeps the reference's 16-byte header shape:
//   1B op code | 3B pad | 4B seq | 8B body size, followed by the body.
#pragma once

#include <atomic>
#include <cstdint>
#include <functional>
#include <map>
#include <memory>
#include <mutex>
#include <optional>
#include <set>
#include <string>
#include <thread>
#include <vector>

#include "faabricamd/util.h"

namespace faabricamd {

// Fixed port map (reference: transport/common.h:8-29)
inline constexpr int STATE_ASYNC_PORT = 8003;
inline constexpr int STATE_SYNC_PORT = 8004;
inline constexpr int FUNCTION_CALL_ASYNC_PORT = 8005;
inline constexpr int FUNCTION_CALL_SYNC_PORT = 8006;
inline constexpr int SNAPSHOT_ASYNC_PORT = 8007;
inline constexpr int SNAPSHOT_SYNC_PORT = 8008;
inline constexpr int POINT_TO_POINT_ASYNC_PORT = 8009;
inline constexpr int POINT_TO_POINT_SYNC_PORT = 8010;
inline constexpr int PLANNER_ASYNC_PORT = 8011;
inline constexpr int PLANNER_SYNC_PORT = 8012;
inline constexpr int DEFAULT_MPI_BASE_PORT = 8020;

// Offset added to all ports this process's SERVERS bind, so several
// single-GPU worker processes can share one node's IP. Set via
// FAABRIC_PORT_OFFSET or programmatically before servers start. A host
// advertises itself as "ip@offset"; clients parse that identity and add
// the offset when dialling (see parseHostIdentity).
int getPortOffset();
void setPortOffset(int offset);
void parseHostIdentity(const std::string& identity,
                       std::string& ipOut,
                       int& offsetOut);

struct WireHeader
{
    uint8_t code = 0;
    uint8_t pad[3] = { 0, 0, 0 };
    uint32_t seq = 0;
    uint64_t size = 0;
};
static_assert(sizeof(WireHeader) == 16, "wire header must be 16 bytes");

// ------------------------- raw sockets --------------------------------------

class SocketClosedException : public FaabricException
{
  public:
    using FaabricException::FaabricException;
};

// Blocking connected socket with framed send/recv. Thread-compatible:
// callers serialise with their own mutex.
class TcpConnection
{
  public:
    TcpConnection() = default;
    explicit TcpConnection(int fdIn)
      : fd(fdIn)
    {}
    ~TcpConnection();
    TcpConnection(const TcpConnection&) = delete;
    TcpConnection& operator=(const TcpConnection&) = delete;
    TcpConnection(TcpConnection&& o) noexcept;
    TcpConnection& operator=(TcpConnection&& o) noexcept;

    // Dial host:port; throws on failure
    static TcpConnection dial(const std::string& host,
                              int port,
                              int timeoutMs = 5000);

    bool isOpen() const { return fd >= 0; }
    void close();

    void sendFrame(uint8_t code,
                   const void* body,
                   size_t len,
                   uint32_t seq = 0);
    // Scatter form: header + two body segments without copying
    void sendFrame2(uint8_t code,
                    const void* a,
                    size_t lenA,
                    const void* b,
                    size_t lenB,
                    uint32_t seq = 0);
    // Returns false on orderly peer close at a frame boundary
    bool recvFrame(WireHeader& hdr, std::string& body);

    void sendAll(const void* data, size_t len);
    void recvAll(void* data, size_t len);

    int rawFd() const { return fd; }

  private:
    int fd = -1;
};

// Listening socket. fd is atomic: close() runs from the stopping thread
// while accept() blocks in the accept thread.
class TcpListener
{
  public:
    TcpListener() = default;
    ~TcpListener();
    TcpListener(const TcpListener&) = delete;

    void listen(int port, int backlog = 128);
    // Accept one connection; returns nullopt if the listener was closed
    std::optional<TcpConnection> accept();
    void close();
    bool isOpen() const { return fd.load() >= 0; }
    int boundPort() const { return port_; }

  private:
    std::atomic<int> fd{ -1 };
    int port_ = 0;
};

// ------------------------- server fabric ------------------------------------

// Each service exposes an async port (fire-and-forget) and a sync port
// (request/response), mirroring the reference server pattern
// (MessageEndpointServer.h:44-95). Connections are handled by one thread
// each; handler callbacks run on that thread.
class MessageEndpointServer
{
  public:
    MessageEndpointServer(int asyncPortIn,
                          int syncPortIn,
                          std::string nameIn);
    virtual ~MessageEndpointServer();

    void start();
    void stop();

    // Subclass API (reference: MessageEndpointServer.h:64-66); seq carries
    // the frame header's sequence number (used by ordered PTP delivery)
    virtual void doAsyncRecv(uint8_t code,
                             const std::string& body,
                             uint32_t seq) = 0;
    virtual std::string doSyncRecv(uint8_t code, const std::string& body) = 0;

  private:
    void acceptLoop(TcpListener& listener, bool isSync);
    void connectionLoop(TcpConnection conn, bool isSync);

    int asyncPort;
    int syncPort;
    std::string name;
    TcpListener asyncListener;
    TcpListener syncListener;
    std::thread asyncAcceptThread;
    std::thread syncAcceptThread;
    std::mutex connThreadsMx;
    std::vector<std::thread> connThreads;
    std::set<int> activeConnFds; // shut down on stop() to unblock recv
    std::atomic<bool> running{ false };
};

// ------------------------- client -------------------------------------------

// Client with one lazily-dialled connection per (async|sync) channel.
// syncSend serialises request/response on the sync connection.
class MessageEndpointClient
{
  public:
    MessageEndpointClient(std::string hostIn,
                          int asyncPortIn,
                          int syncPortIn);

    void asyncSend(uint8_t code, const std::string& body);
    void asyncSend(uint8_t code, const void* body, size_t len);
    void asyncSendSeq(uint8_t code, const void* body, size_t len, uint32_t seq);
    std::string syncSend(uint8_t code, const std::string& body);

    const std::string& getHost() const { return host; }

  private:
    TcpConnection& ensure(bool sync);

    std::string host;
    int asyncPort;
    int syncPort;
    std::mutex asyncMx;
    std::mutex syncMx;
    TcpConnection asyncConn;
    TcpConnection syncConn;
};

} // namespace faabricamd
