// Point-to-point broker and groups: (groupId, groupIdx) addressed messaging,
// distributed locks, barriers and notifications.
//
// MI355X-native equivalent of the reference's PTP layer (reference:
// include/faabric/transport/PointToPointBroker.h:26-182,
// src/transport/PointToPointBroker.cpp:416-926) — re-designed: local
// delivery is an in-process queue (the bulk data plane for GPU payloads is
// RCCL/peer-copies in the MPI layer, not PTP), remote delivery via the PTP
// RPC pair, ordering by per-sender sequence numbers with an out-of-order
// buffer.
#pragma once

#include <condition_variable>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <unordered_map>
#include <vector>

#include "faabricamd/messages.h"
#include "faabricamd/queue.h"
#include "faabricamd/scheduling.h"
#include "faabricamd/transport.h"

namespace faabricamd {

// RPC call codes (reference: transport/PointToPointCall.h:5-14)
enum class PointToPointCall : uint8_t
{
    MAPPING = 0,
    MESSAGE = 1,
    LOCK_GROUP = 2,
    LOCK_GROUP_RECURSIVE = 3,
    UNLOCK_GROUP = 4,
    UNLOCK_GROUP_RECURSIVE = 5,
    // HIP-IPC peer transport between same-node workers (hipipc.h)
    IPC_ARENA = 6,   // sync: body = sender host, reply = IpcArenaInfo
    MESSAGE_IPC = 7, // async: IpcPtpMessage (payload already in arena)
    IPC_ACK = 8,     // async: IpcAck (receiver freed a segment)
    // Group lifecycle: the planner broadcasts this to involved hosts
    // when an app completes, so mappings/seq state don't accumulate
    GROUP_CLEAR = 9, // async: PointToPointMessage (groupId only)
};

inline constexpr int32_t POINT_TO_POINT_MAIN_IDX = 0;

// Dedicated sendIdx namespace for migration control messages so they never
// collide with app/MPI channels (see ptp.cpp SEND_OFF_* constants)
inline constexpr int32_t PTP_MIGRATION_CHANNEL_OFFSET = 3 * 16384 + 8192;

class PointToPointBroker;
PointToPointBroker& getPointToPointBroker();

// Group coordination: lock, barrier, notify
// (reference: transport/PointToPointBroker.h:26-100)
class PointToPointGroup
{
  public:
    static std::shared_ptr<PointToPointGroup> getGroup(int32_t groupId);
    static std::shared_ptr<PointToPointGroup> getOrAwaitGroup(int32_t groupId);
    static bool groupExists(int32_t groupId);
    static void addGroup(int32_t appId, int32_t groupId, int32_t groupSize);
    static void clearGroup(int32_t groupId);
    static void clear();

    PointToPointGroup(int32_t appIdIn, int32_t groupIdIn, int32_t sizeIn);

    void lock(int32_t groupIdx, bool recursive = false);
    void unlock(int32_t groupIdx, bool recursive = false);
    void localLock();
    void localUnlock();

    void barrier(int32_t groupIdx);
    void notify(int32_t groupIdx); // all non-main notify main; main waits

    int32_t getAppId() const { return appId; }
    int32_t getSize() const { return groupSize; }

    // Main-host side of the distributed lock
    void handleLockRequest(const std::string& fromHost,
                           int32_t groupIdx,
                           bool recursive);
    void handleUnlockRequest(const std::string& fromHost,
                             int32_t groupIdx,
                             bool recursive);

  private:
    int32_t appId;
    int32_t groupId;
    int32_t groupSize;

    std::mutex internalMx;        // protects the lock queue
    std::timed_mutex localMx;     // local fast-path lock
    std::recursive_timed_mutex localRecursiveMx;
    int recursiveDepth = 0;
    std::vector<std::pair<std::string, int32_t>> lockWaiters;
    bool lockHeld = false;
    int32_t lockHolderIdx = -1;
};

class PointToPointBroker
{
  public:
    PointToPointBroker();

    // --- mappings (control plane) ---
    std::set<std::string> setUpLocalMappingsFromSchedulingDecision(
      const SchedulingDecision& decision);
    void setAndSendMappingsFromSchedulingDecision(
      const SchedulingDecision& decision);
    void sendMappingsFromSchedulingDecision(
      const SchedulingDecision& decision,
      const std::set<std::string>& hostList);
    void waitForMappingsOnThisHost(int32_t groupId);

    std::string getHostForReceiver(int32_t groupId, int32_t recvIdx);
    int32_t getMpiPortForReceiver(int32_t groupId, int32_t recvIdx);
    std::set<int32_t> getIdxsRegisteredForGroup(int32_t groupId);

    // --- messaging (app plane) ---
    void sendMessage(int32_t appId,
                     int32_t groupId,
                     int32_t sendIdx,
                     int32_t recvIdx,
                     const uint8_t* data,
                     size_t size,
                     bool mustOrderMsgs = false);
    std::vector<uint8_t> recvMessage(int32_t groupId,
                                     int32_t sendIdx,
                                     int32_t recvIdx,
                                     bool mustOrderMsgs = false,
                                     int timeoutMs = DEFAULT_QUEUE_TIMEOUT_MS);

    // Device (HBM) payloads: same-process receivers get a D2D copy on the
    // broker's side stream (never staged through the host); cross-host
    // receivers are staged D2H and ride the normal RPC plane.
    void sendMessageDevice(int32_t appId,
                           int32_t groupId,
                           int32_t sendIdx,
                           int32_t recvIdx,
                           const void* devPtr,
                           size_t size,
                           bool mustOrderMsgs = false);
    // Receive into a device buffer (H2D-copies host-plane payloads)
    size_t recvMessageDevice(int32_t groupId,
                             int32_t sendIdx,
                             int32_t recvIdx,
                             void* devPtr,
                             size_t capacity,
                             bool mustOrderMsgs = false,
                             int timeoutMs = DEFAULT_QUEUE_TIMEOUT_MS);

    // Deliver a message that arrived over the network
    void deliverRemoteMessage(const PointToPointMessage& msg, uint32_t seq);

    // Deliver a MESSAGE_IPC control message (payload already in our arena)
    void deliverIpcMessage(int32_t groupId,
                           int32_t sendIdx,
                           int32_t recvIdx,
                           const std::string& senderHost,
                           uint64_t offset,
                           uint64_t size,
                           uint32_t seq);

    // Ack a consumed arena segment back to its sender
    void sendIpcAck(const std::string& senderHost,
                    uint64_t offset,
                    uint64_t size);

    // Group cleanup after an app finishes / migrates
    void clearGroup(int32_t groupId);

    // Clear locally, or tell `host` to clear (used by the planner when
    // an app completes)
    void sendGroupClear(const std::string& host, int32_t groupId);

    size_t debugMappingsCount();
    size_t debugChannelsCount();
    size_t debugSendSeqsCount();
    void clear();

    void updateHostForIdx(int32_t groupId,
                          int32_t groupIdx,
                          const std::string& newHost);

    // Drop cached outbound connections (hosts restarted)
    void clearClients();

    // Post-migration: barrier + clear send-seq state
    // (reference: src/transport/PointToPointBroker.cpp:910-926)
    void postMigrationHook(int32_t groupId, int32_t groupIdx);

  private:
    struct PtpPayload
    {
        std::vector<uint8_t> host;
        void* dev = nullptr; // HBM staging copy (same-process delivery)
        size_t devSize = 0;
        // HIP-IPC segment in OUR arena (cross-process same-node sender);
        // consumed by copying out of the arena then acking the segment
        bool isArena = false;
        std::string arenaHost; // sender host (names the arena)
        uint64_t arenaOff = 0;
        uint64_t arenaSize = 0;
    };

    struct Channel
    {
        std::mutex mx;
        std::condition_variable cv;
        std::map<uint32_t, PtpPayload> bufferedMsgs; // seq → payload
        std::deque<PtpPayload> unorderedMsgs;
        uint32_t nextRecvSeq = 0;
    };

    void deliverPayload(int32_t groupId,
                        int32_t sendIdx,
                        int32_t recvIdx,
                        PtpPayload payload,
                        uint32_t seq);
    PtpPayload recvPayload(int32_t groupId,
                           int32_t sendIdx,
                           int32_t recvIdx,
                           bool mustOrderMsgs,
                           int timeoutMs);
    void* sideStream(); // lazily-created HIP stream for D2D staging

    std::shared_ptr<Channel> getChannel(int32_t groupId,
                                        int32_t sendIdx,
                                        int32_t recvIdx);

    std::mutex brokerMx;
    // (groupId, groupIdx) → host; (groupId, groupIdx) → mpi port
    std::map<int64_t, std::string> mappings;
    std::map<int64_t, int32_t> mpiPorts;
    std::map<int32_t, std::set<int32_t>> groupIdxs;
    std::map<int32_t, std::shared_ptr<FlagWaiter>> groupFlags;

    // Sharded: every local MPI message hits this map, one global
    // mutex here was the old hot-path serialization point
    ConcurrentMap<int64_t, std::shared_ptr<Channel>> channels;

    std::mutex sendSeqMx;
    std::map<int64_t, uint32_t> sendSeqs;

    std::mutex clientsMx;
    std::map<std::string, std::shared_ptr<MessageEndpointClient>> clients;
    std::mutex streamMx;
    void* sideStream_ = nullptr;

    std::shared_ptr<MessageEndpointClient> getClient(const std::string& host);

    std::shared_ptr<FlagWaiter> getFlag(int32_t groupId);
};

// PTP RPC server (reference: src/transport/PointToPointServer.cpp:22-128)
class PointToPointServer : public MessageEndpointServer
{
  public:
    PointToPointServer();
    void doAsyncRecv(uint8_t code,
                     const std::string& body,
                     uint32_t seq) override;
    std::string doSyncRecv(uint8_t code, const std::string& body) override;
};

// PTP RPC client
class PointToPointClient : public MessageEndpointClient
{
  public:
    explicit PointToPointClient(const std::string& host);
    void sendMappings(const PointToPointMappings& mappings);
    void sendMessage(const PointToPointMessage& msg, uint32_t seq);
    void groupLock(int32_t appId,
                   int32_t groupId,
                   int32_t groupIdx,
                   bool recursive);
    void groupUnlock(int32_t appId,
                     int32_t groupId,
                     int32_t groupIdx,
                     bool recursive);
};

} // namespace faabricamd
