// Cross-process same-node device transport over HIP IPC.
//
// The deployment shape is one worker process per GPU on a node. Device
// payloads between two such workers must move over xGMI as peer copies,
// not D2H -> TCP loopback -> H2D. Each receiver exports a per-sender
// staging arena in its own HBM via hipIpcMemHandle_t; the sender opens
// the handle once, bump/ring-allocates segments, writes payloads with
// hipMemcpyAsync on a side stream (peer copy over xGMI when the GPUs
// differ), and sends a small control RPC naming the segment. The
// receiver copies the segment out (local D2D/D2H at HBM speed) and acks
// the segment back so the ring can recycle it.
//
// Reference analog: the local fast path of the PTP broker
// (/root/reference/src/transport/PointToPointBroker.cpp:637-764) routes
// same-host messages through in-process nng pairs; here "local" means
// same-node-different-process, and the fast path is HIP IPC + xGMI.
#pragma once

#include <atomic>
#include <condition_variable>
#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

namespace faabricamd {

// True when both identities ("ip@portOffset") are on the same IP but are
// different worker processes (different offsets)
bool isSameNodeDifferentWorker(const std::string& a, const std::string& b);

// Result of an IPC_ARENA sync RPC: everything the sender needs to open
// the receiver's staging arena
struct IpcArenaInfo
{
    std::vector<uint8_t> handle; // 1: hipIpcMemHandle_t bytes (empty = no IPC)
    uint64_t size = 0;           // 2
    int32_t deviceId = -1;       // 3
    std::string encode() const;
    static IpcArenaInfo decode(const std::string& buf);
};

// Control message for a shipped PTP payload (async MESSAGE_IPC call)
struct IpcPtpMessage
{
    int32_t appId = 0;      // 1
    int32_t groupId = 0;    // 2
    int32_t sendIdx = 0;    // 3
    int32_t recvIdx = 0;    // 4
    std::string senderHost; // 5: names the (receiver, sender) arena
    uint64_t offset = 0;    // 6
    uint64_t size = 0;      // 7
    std::string encode() const;
    static IpcPtpMessage decode(const std::string& buf);
};

// One shipped chunk of a larger value (state KV pull/push, device
// snapshot ship). Values larger than the arena stream as a sequence of
// chunks, each acked as it is consumed so the ring recycles.
struct IpcChunk
{
    std::string user;      // 1 (state namespace; empty for snapshots)
    std::string key;       // 2
    uint64_t valOffset = 0; // 3: offset in the destination value/snapshot
    uint64_t ipcOffset = 0; // 4: segment in the receiver's arena
    uint64_t len = 0;       // 5
    std::string srcHost;    // 6: names the arena (the shipping side)
    uint64_t totalSize = 0; // 7: full value size (create-on-first)
    std::string encode() const;
    static IpcChunk decode(const std::string& buf);
};

// Segment ack (async IPC_ACK call, receiver -> sender)
struct IpcAck
{
    std::string receiverHost; // 1: names the sender's ring for this peer
    uint64_t offset = 0;      // 2
    uint64_t size = 0;        // 3
    std::string encode() const;
    static IpcAck decode(const std::string& buf);
};

// ---------------------------------------------------------------------------
// Receiver side: staging arenas in this worker's HBM, one per sender host
// ---------------------------------------------------------------------------
class IpcReceiver
{
  public:
    static IpcReceiver& get();

    // Serve an IPC_ARENA request: allocate (once) and export the arena
    // for senderHost. Returns an info with an empty handle when IPC is
    // unavailable (no GPU / export failed) so the sender falls back.
    IpcArenaInfo arenaFor(const std::string& senderHost);

    // Copy a shipped segment out of the arena. The caller acks afterwards.
    void copyToDevice(const std::string& senderHost,
                      uint64_t offset,
                      void* dstDev,
                      size_t size);
    void copyToHost(const std::string& senderHost,
                    uint64_t offset,
                    void* dstHost,
                    size_t size);

    void clear();
    ~IpcReceiver();

  private:
    struct Arena;
    std::mutex mx;
    std::map<std::string, std::shared_ptr<Arena>> arenas;
    std::shared_ptr<Arena> find(const std::string& senderHost);
};

// ---------------------------------------------------------------------------
// Sender side: opened peer arenas with ring allocation + ack-driven reuse
// ---------------------------------------------------------------------------
class IpcSender
{
  public:
    static IpcSender& get();

    // True when a peer arena to targetHost is (or can be) open. First call
    // does the IPC_ARENA sync RPC + hipIpcOpenMemHandle; the result is
    // cached, including failures (so CPU-only runs probe once).
    bool available(const std::string& targetHost);

    // Ship a device buffer into the target's arena; returns the segment
    // offset for the control message. Blocks while the ring is full until
    // acks free space (throws after a timeout).
    uint64_t ship(const std::string& targetHost,
                  const void* devPtr,
                  size_t size);
    // Same, from pageable/pinned host memory (H2D straight into the peer)
    uint64_t shipFromHost(const std::string& targetHost,
                          const void* hostPtr,
                          size_t size);

    void onAck(const std::string& targetHost,
               uint64_t offset,
               uint64_t size);

    // Segments/bytes shipped so far (tests assert the IPC path ran)
    uint64_t shippedSegments() const;
    uint64_t shippedBytes() const;

    // Arena capacity at targetHost (0 = unavailable); bulk transfers
    // pick their chunk size from this
    uint64_t peerCapacity(const std::string& targetHost);

    void clear();
    ~IpcSender();

  private:
    struct Peer;
    std::mutex mx;
    std::map<std::string, std::shared_ptr<Peer>> peers;
    std::atomic<uint64_t> nShipped{ 0 };
    std::atomic<uint64_t> bytesShipped{ 0 };
    std::shared_ptr<Peer> ensurePeer(const std::string& targetHost);
    uint64_t shipImpl(const std::string& targetHost,
                      const void* ptr,
                      size_t size,
                      bool fromHost);
};

} // namespace faabricamd
