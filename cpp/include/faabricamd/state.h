// Distributed state KV.
//
// MI355X-native re-design of the reference state layer (reference:
// include/faabric/state/State.h:23-58, StateKeyValue.h:19-166,
// InMemoryStateKeyValue.cpp, src/state/StateServer.cpp:24-164,
// src/state/StateClient.cpp). Master-per-key becomes owner-host (one GPU
// per worker process); values can be HOST-resident or DEVICE-resident in
// HBM3E (device path wired through ops.cpp); chunked lazy pull and
// dirty-mask partial push are kept.
#pragma once

#include <atomic>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include "faabricamd/hipipc.h"
#include "faabricamd/messages.h"
#include "faabricamd/transport.h"

namespace faabricamd {

// Streaming chunk size (reference: state/StateKeyValue.h:19 — 64 KiB)
inline constexpr size_t STATE_STREAM_CHUNK_SIZE = 64 * 1024;

// (reference: state/State.h:11-21)
enum class StateCalls : uint8_t
{
    Pull = 1,
    Push = 2,
    Size = 3,
    Append = 4,
    ClearAppended = 5,
    PullAppended = 6,
    Delete = 7,
    // HIP-IPC bulk paths for device-resident values between same-node
    // workers (hipipc.h): payload rides the arena, RPC carries segments
    PullIpc = 8,
    PushIpc = 9,
    // Scripted-lock parity with the reference's Redis backend
    // (reference: redis/Redis.h:154-168)
    Lock = 10,
    Unlock = 11,
};

class StateKeyValue
{
  public:
    StateKeyValue(std::string userIn,
                  std::string keyIn,
                  size_t sizeIn,
                  std::string masterHostIn,
                  bool onDeviceIn = false,
                  int deviceIn = 0);
    ~StateKeyValue();

    const std::string& getUser() const { return user; }
    const std::string& getKey() const { return key; }
    size_t size() const { return valueSize; }
    bool isMaster() const;
    const std::string& getMasterHost() const { return masterHost; }

    // --- whole-value ---
    void get(uint8_t* buffer);
    std::vector<uint8_t> get();
    void set(const uint8_t* buffer, size_t n);
    void set(const std::vector<uint8_t>& data);

    // --- chunks ---
    void getChunk(uint64_t offset, uint8_t* buffer, size_t len);
    void setChunk(uint64_t offset, const uint8_t* buffer, size_t len);

    // --- lazy pull / push against the master ---
    void pull();     // fetch whole value (chunked)
    void pushFull(); // push whole value
    void flagDirty();
    void flagChunkDirty(uint64_t offset, size_t len);
    void pushPartial(); // push only dirty chunks

    // --- append channel ---
    void append(const uint8_t* data, size_t len);
    std::vector<std::vector<uint8_t>> getAppended(size_t nValues);
    void clearAppended();

    // Direct access for zero-copy users: host pointer, or the HBM
    // pointer when the value is device-resident (drains pending mirror
    // writes and invalidates the mirror — the caller may write HBM)
    uint8_t* getDataPtr();
    bool isOnDevice() const { return onDevice; }
    int getDevice() const { return device; }

    // Master-side servicing
    std::vector<uint8_t> serviceChunk(uint64_t offset, size_t len);
    void serviceSet(uint64_t offset, const uint8_t* data, size_t len);
    // IPC variants: ship a range into dstHost's arena / absorb a shipped
    // segment from our arena (master side of PullIpc/PushIpc)
    uint64_t serviceChunkIpc(const std::string& dstHost,
                             uint64_t offset,
                             size_t len);
    void serviceSetIpc(const std::string& srcHost,
                       uint64_t ipcOffset,
                       uint64_t valOffset,
                       size_t len);
    void serviceAppend(const uint8_t* data, size_t len);
    std::vector<std::vector<uint8_t>> serviceGetAppended(size_t n);
    void serviceClearAppended();

  private:
    std::string user;
    std::string key;
    size_t valueSize;
    std::string masterHost;

    std::mutex kvMx;
    std::vector<uint8_t> value; // host mode
    bool onDevice = false;      // HBM mode: value lives in devPtr
    int device = 0;
    uint8_t* devPtr = nullptr;

    // Pinned host mirror for device values (group-commit write-through):
    // chunk writes memcpy into the mirror and enqueue an async H2D on a
    // stripe stream with NO per-op sync — the ~12 us host-visible
    // latency of a synchronous 4 KiB HIP copy was the measured floor of
    // the batch path (BASELINE.md config-5 history). Reads are served
    // from the mirror (filled D2H on first touch). sync() is the
    // durability point; direct devPtr uses flush/invalidate around it.
    //
    // The mirror is STRIPED: the value is split into 64 KiB blocks
    // round-robined over KV_STRIPES (mutex, HIP stream) pairs so
    // concurrent executors touching different ranges never serialise on
    // one lock or one stream (128 concurrent kvtouch functions measured
    // ~228 us of thread-time each on the single-mutex design). Per-range
    // ordering holds because every byte maps to exactly one stripe.
  public:
    static constexpr int KV_STRIPES = 16;

  private:
    static constexpr uint64_t KV_STRIPE_BLOCK = 64 * 1024;
    // Stripes (mutex + HIP stream) live in one PROCESS-GLOBAL pool
    // shared by every KV: per-KV stream sets measured ~1 MiB+ of HBM
    // settle per stream. Ordering still holds — every (kv, byte) maps
    // to exactly one stripe.
    bool stripesReady = false;
    std::mutex mirrorMx; // guards mirror bring-up only
    uint8_t* mirror = nullptr;
    std::vector<char> mirrorValid; // per 4 KiB page (owned by its stripe)
    // Write-back dirty pages: writes memcpy into the mirror and mark
    // here; sync() coalesces dirty runs into few H2D enqueues (per-op
    // write-through enqueues convoyed on the HIP driver lock under
    // 128-thread batches)
    std::vector<char> mirrorDirty;
    std::atomic<bool> mirrorAnyDirty{ false };
    bool mirrorFailed = false;
    bool mirrorUsable();
    // Runtime-capped stripe count (FAABRIC_KV_STRIPES, default 16)
    int stripeOf(uint64_t offset) const;
    // Split [offset, offset+len) at stripe-block boundaries and run fn
    // per subrange with that stripe's lock held
    template<typename Fn>
    void forEachStripeRange(uint64_t offset, size_t len, Fn&& fn);
    void mirrorFillLocked(int stripe, uint64_t offset, size_t len);
    void mirrorInvalidate(uint64_t offset, size_t len);

  public:
    // Drain pending device writes (group-commit durability point)
    void sync();

  private:

    void readLocal(uint64_t offset, uint8_t* out, size_t len);
    void writeLocal(uint64_t offset, const uint8_t* data, size_t len);
    // True when bulk pull/push to the master should ride HIP IPC
    bool useIpcToMaster();
    void pullRangeIpc(uint64_t offset, size_t len);
    void pushRangeIpc(uint64_t offset, size_t len);
    std::vector<char> dirtyChunks; // one flag per STATE_STREAM_CHUNK_SIZE
    // Chunks already fetched from the master (reference: pulledMask) —
    // a chunk is pulled at most once; local writes count as fresh
    std::vector<char> pulledChunks;
    bool fullyPulled = false;
    std::vector<std::vector<uint8_t>> appendedValues;
};

class State
{
  public:
    static State& get();

    std::shared_ptr<StateKeyValue> getKV(const std::string& user,
                                         const std::string& key,
                                         size_t size);
    // HBM-resident value on this host's GPU (north star: distributed
    // state lives in the 288 GB HBM3E per GPU)
    // device -1 = this worker's configured GPU (FAABRIC_GPU_DEVICE)
    std::shared_ptr<StateKeyValue> getKVDevice(const std::string& user,
                                               const std::string& key,
                                               size_t size,
                                               int device = -1);
    std::shared_ptr<StateKeyValue> getKV(const std::string& user,
                                         const std::string& key);
    size_t getStateSize(const std::string& user, const std::string& key);
    void deleteKV(const std::string& user, const std::string& key);
    void deleteKVLocally(const std::string& user, const std::string& key);
    size_t getKVCount();
    void forceClearAll(bool global);
    // Drain every KV's pending device writes (group-commit point)
    void syncAll();

    // Distributed scripted locks (reference: Redis::acquireLock /
    // releaseLock with delete-if-token-matches, redis/Redis.h:154-168,
    // src/redis/Redis.cpp). The lock lives on the key's owner host (the
    // planner in "planner" state mode). acquire returns 0 on contention.
    uint64_t acquireLock(const std::string& user,
                         const std::string& key,
                         int expiryMs);
    void releaseLock(const std::string& user,
                     const std::string& key,
                     uint64_t token);
    uint64_t acquireLockLocal(const std::string& user,
                              const std::string& key,
                              int expiryMs);
    bool releaseLockLocal(const std::string& user,
                          const std::string& key,
                          uint64_t token);

    // Owner directory: where a key lives. Defaults to this host on first
    // getKV; setMasterHost lets the deployment pin owner GPUs.
    void setMasterHost(const std::string& user,
                       const std::string& key,
                       const std::string& host);
    std::string getMasterHost(const std::string& user,
                              const std::string& key);

  private:
    std::mutex mx;
    std::map<std::string, std::shared_ptr<StateKeyValue>> kvMap;
    std::map<std::string, std::string> masterMap;

    std::mutex locksMx;
    // key → (token, expiry deadline ms)
    std::map<std::string, std::pair<uint64_t, int64_t>> locks;
};

class StateServer : public MessageEndpointServer
{
  public:
    StateServer();
    void doAsyncRecv(uint8_t code,
                     const std::string& body,
                     uint32_t seq) override;
    std::string doSyncRecv(uint8_t code, const std::string& body) override;
};

class StateClient : public MessageEndpointClient
{
  public:
    explicit StateClient(const std::string& host);
    std::vector<uint8_t> pullChunk(const std::string& user,
                                   const std::string& key,
                                   uint64_t offset,
                                   size_t len);
    void pushChunk(const std::string& user,
                   const std::string& key,
                   uint64_t offset,
                   const uint8_t* data,
                   size_t len,
                   size_t totalSize = 0);
    // IPC bulk paths (payload in the arena; see StateCalls::PullIpc)
    IpcChunk pullChunkIpc(const std::string& user,
                          const std::string& key,
                          uint64_t offset,
                          size_t len,
                          const std::string& requesterHost);
    void pushChunkIpc(const IpcChunk& chunk);
    uint64_t acquireLock(const std::string& user,
                         const std::string& key,
                         int expiryMs);
    bool releaseLock(const std::string& user,
                     const std::string& key,
                     uint64_t token);
    size_t stateSize(const std::string& user, const std::string& key);
    void append(const std::string& user,
                const std::string& key,
                const uint8_t* data,
                size_t len);
    std::vector<std::vector<uint8_t>> pullAppended(const std::string& user,
                                                   const std::string& key,
                                                   size_t nValues);
    void clearAppended(const std::string& user, const std::string& key);
    void deleteKV(const std::string& user, const std::string& key);
};

std::shared_ptr<StateClient> getStateClient(const std::string& host);
void clearStateClients();

} // namespace faabricamd
