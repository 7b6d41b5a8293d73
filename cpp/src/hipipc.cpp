// HIP-IPC same-node cross-process device transport (see hipipc.h).
//
// Receiver arenas are plain hipMalloc allocations (mempool/async
// allocations are not IPC-exportable); the export handle travels over
// the PTP sync plane (IPC_ARENA call, ptp.cpp). Segment allocation on
// the sender is a first-fit free-gap scan over the outstanding-segment
// map — outstanding counts are small (messages in flight), acks arrive
// out of order, and this avoids ring wrap/pad bookkeeping entirely.
#include "faabricamd/hipipc.h"
#include "faabricamd/transport.h"
#include "faabricamd/util.h"
#include "faabricamd/wire.h"

#include <hip/hip_runtime.h>

#include <chrono>
#include <cstring>

namespace faabricamd {

// Implemented in ptp.cpp (rides the PTP sync plane)
IpcArenaInfo fetchIpcArenaFromHost(const std::string& host);

static constexpr uint64_t IPC_ALIGN = 256;

static uint64_t alignUp(uint64_t v)
{
    return (v + IPC_ALIGN - 1) & ~(IPC_ALIGN - 1);
}

static uint64_t arenaBytes()
{
    static const uint64_t bytes = []() {
        int mb = getEnvVarInt("FAABRIC_IPC_ARENA_MB", 128);
        if (mb < 1) {
            mb = 1;
        }
        return (uint64_t)mb * 1024 * 1024;
    }();
    return bytes;
}

static bool ipcDisabled()
{
    static const bool disabled =
      getEnvVarInt("FAABRIC_IPC_DISABLE", 0) != 0;
    return disabled;
}

bool isSameNodeDifferentWorker(const std::string& a, const std::string& b)
{
    std::string ipA;
    std::string ipB;
    int offA = 0;
    int offB = 0;
    parseHostIdentity(a, ipA, offA);
    parseHostIdentity(b, ipB, offB);
    return ipA == ipB && offA != offB;
}

// ------------------------- wire formats -------------------------------------

std::string IpcArenaInfo::encode() const
{
    PbWriter w;
    w.putBytes(1, handle);
    w.putUInt64(2, size);
    w.putInt32(3, deviceId);
    return w.take();
}

IpcArenaInfo IpcArenaInfo::decode(const std::string& buf)
{
    IpcArenaInfo m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.handle = r.asBytes(); break;
            case 2: m.size = r.asUInt64(); break;
            case 3: m.deviceId = r.asInt32(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string IpcPtpMessage::encode() const
{
    PbWriter w;
    w.putInt32(1, appId);
    w.putInt32(2, groupId);
    w.putInt32(3, sendIdx);
    w.putInt32(4, recvIdx);
    w.putString(5, senderHost);
    w.putUInt64(6, offset);
    w.putUInt64(7, size);
    return w.take();
}

IpcPtpMessage IpcPtpMessage::decode(const std::string& buf)
{
    IpcPtpMessage m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.appId = r.asInt32(); break;
            case 2: m.groupId = r.asInt32(); break;
            case 3: m.sendIdx = r.asInt32(); break;
            case 4: m.recvIdx = r.asInt32(); break;
            case 5: m.senderHost = r.asString(); break;
            case 6: m.offset = r.asUInt64(); break;
            case 7: m.size = r.asUInt64(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string IpcChunk::encode() const
{
    PbWriter w;
    w.putString(1, user);
    w.putString(2, key);
    w.putUInt64(3, valOffset);
    w.putUInt64(4, ipcOffset);
    w.putUInt64(5, len);
    w.putString(6, srcHost);
    w.putUInt64(7, totalSize);
    return w.take();
}

IpcChunk IpcChunk::decode(const std::string& buf)
{
    IpcChunk m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.user = r.asString(); break;
            case 2: m.key = r.asString(); break;
            case 3: m.valOffset = r.asUInt64(); break;
            case 4: m.ipcOffset = r.asUInt64(); break;
            case 5: m.len = r.asUInt64(); break;
            case 6: m.srcHost = r.asString(); break;
            case 7: m.totalSize = r.asUInt64(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string IpcAck::encode() const
{
    PbWriter w;
    w.putString(1, receiverHost);
    w.putUInt64(2, offset);
    w.putUInt64(3, size);
    return w.take();
}

IpcAck IpcAck::decode(const std::string& buf)
{
    IpcAck m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.receiverHost = r.asString(); break;
            case 2: m.offset = r.asUInt64(); break;
            case 3: m.size = r.asUInt64(); break;
            default: r.skip(t);
        }
    }
    return m;
}

// ------------------------- receiver ------------------------------------------

struct IpcReceiver::Arena
{
    void* base = nullptr;
    uint64_t size = 0;
    int deviceId = -1;
    hipStream_t stream = nullptr;
    IpcArenaInfo info; // cached export (empty handle = failed)
    std::mutex copyMx; // serialise copy-outs on the arena stream

    ~Arena()
    {
        if (stream != nullptr) {
            (void)hipStreamDestroy(stream);
        }
        if (base != nullptr) {
            (void)hipFree(base);
        }
    }
};

IpcReceiver& IpcReceiver::get()
{
    static IpcReceiver instance;
    return instance;
}

IpcArenaInfo IpcReceiver::arenaFor(const std::string& senderHost)
{
    std::lock_guard<std::mutex> lock(mx);
    auto it = arenas.find(senderHost);
    if (it != arenas.end()) {
        return it->second->info;
    }

    auto arena = std::make_shared<Arena>();
    arena->info.size = 0;
    arenas[senderHost] = arena; // cache failures too

    if (ipcDisabled()) {
        return arena->info;
    }
    int devices = 0;
    if (hipGetDeviceCount(&devices) != hipSuccess || devices == 0) {
        return arena->info;
    }
    uint64_t bytes = arenaBytes();
    (void)hipSetDevice(getEnvVarInt("FAABRIC_GPU_DEVICE", 0));
    if (hipMalloc(&arena->base, bytes) != hipSuccess) {
        arena->base = nullptr;
        return arena->info;
    }
    hipIpcMemHandle_t handle;
    if (hipIpcGetMemHandle(&handle, arena->base) != hipSuccess) {
        (void)hipFree(arena->base);
        arena->base = nullptr;
        return arena->info;
    }
    if (hipStreamCreateWithFlags(&arena->stream, hipStreamNonBlocking) !=
        hipSuccess) {
        (void)hipFree(arena->base);
        arena->base = nullptr;
        return arena->info;
    }
    (void)hipGetDevice(&arena->deviceId);
    arena->size = bytes;
    arena->info.size = bytes;
    arena->info.deviceId = arena->deviceId;
    arena->info.handle.assign((uint8_t*)&handle,
                              (uint8_t*)&handle + sizeof(handle));
    FAM_INFO("ipc: exported %lu MiB arena for sender %s",
             (unsigned long)(bytes >> 20),
             senderHost.c_str());
    return arena->info;
}

std::shared_ptr<IpcReceiver::Arena> IpcReceiver::find(
  const std::string& senderHost)
{
    std::lock_guard<std::mutex> lock(mx);
    auto it = arenas.find(senderHost);
    if (it == arenas.end() || it->second->base == nullptr) {
        throw FaabricException("ipc: no arena for sender " + senderHost);
    }
    return it->second;
}

void IpcReceiver::copyToDevice(const std::string& senderHost,
                               uint64_t offset,
                               void* dstDev,
                               size_t size)
{
    auto arena = find(senderHost);
    // Offsets arrive off the wire; guard the sum against wrap too
    if (offset > arena->size || size > arena->size ||
        offset + size > arena->size) {
        throw FaabricException("ipc: segment out of arena bounds");
    }
    std::lock_guard<std::mutex> lock(arena->copyMx);
    if (hipMemcpyAsync(dstDev,
                       (uint8_t*)arena->base + offset,
                       size,
                       hipMemcpyDeviceToDevice,
                       arena->stream) != hipSuccess ||
        hipStreamSynchronize(arena->stream) != hipSuccess) {
        throw FaabricException("ipc: arena D2D copy-out failed");
    }
}

void IpcReceiver::copyToHost(const std::string& senderHost,
                             uint64_t offset,
                             void* dstHost,
                             size_t size)
{
    auto arena = find(senderHost);
    // Offsets arrive off the wire; guard the sum against wrap too
    if (offset > arena->size || size > arena->size ||
        offset + size > arena->size) {
        throw FaabricException("ipc: segment out of arena bounds");
    }
    std::lock_guard<std::mutex> lock(arena->copyMx);
    if (hipMemcpyAsync(dstHost,
                       (uint8_t*)arena->base + offset,
                       size,
                       hipMemcpyDeviceToHost,
                       arena->stream) != hipSuccess ||
        hipStreamSynchronize(arena->stream) != hipSuccess) {
        throw FaabricException("ipc: arena D2H copy-out failed");
    }
}

void IpcReceiver::clear()
{
    std::lock_guard<std::mutex> lock(mx);
    arenas.clear();
}

IpcReceiver::~IpcReceiver()
{
    // Process teardown: leave HIP resources to the runtime (destroying
    // after hip has shut down faults)
    std::lock_guard<std::mutex> lock(mx);
    for (auto& [host, arena] : arenas) {
        arena->base = nullptr;
        arena->stream = nullptr;
    }
    arenas.clear();
}

// ------------------------- sender --------------------------------------------

struct IpcSender::Peer
{
    bool ok = false;
    void* base = nullptr;
    uint64_t cap = 0;
    int deviceId = -1;
    hipStream_t stream = nullptr;

    std::mutex mx;
    std::condition_variable cv;
    std::map<uint64_t, uint64_t> inUse; // offset -> len, sorted

    ~Peer()
    {
        if (stream != nullptr) {
            (void)hipStreamDestroy(stream);
        }
        if (base != nullptr) {
            (void)hipIpcCloseMemHandle(base);
        }
    }

    // First-fit gap scan; caller holds mx
    bool tryAlloc(uint64_t n, uint64_t& off)
    {
        uint64_t pos = 0;
        for (auto& [o, len] : inUse) {
            if (o - pos >= n) {
                off = pos;
                return true;
            }
            pos = o + len;
        }
        if (cap >= pos && cap - pos >= n) {
            off = pos;
            return true;
        }
        return false;
    }
};

IpcSender& IpcSender::get()
{
    static IpcSender instance;
    return instance;
}

std::shared_ptr<IpcSender::Peer> IpcSender::ensurePeer(
  const std::string& targetHost)
{
    {
        std::lock_guard<std::mutex> lock(mx);
        auto it = peers.find(targetHost);
        if (it != peers.end()) {
            return it->second;
        }
    }

    // Probe outside the map lock (sync RPC + IPC open take a while)
    auto peer = std::make_shared<Peer>();
    if (!ipcDisabled()) {
        int devices = 0;
        if (hipGetDeviceCount(&devices) == hipSuccess && devices > 0) {
            try {
                IpcArenaInfo info = fetchIpcArenaFromHost(targetHost);
                if (info.handle.size() == sizeof(hipIpcMemHandle_t) &&
                    info.size > 0) {
                    hipIpcMemHandle_t handle;
                    std::memcpy(&handle, info.handle.data(),
                                sizeof(handle));
                    void* base = nullptr;
                    if (hipIpcOpenMemHandle(
                          &base, handle,
                          hipIpcMemLazyEnablePeerAccess) == hipSuccess) {
                        hipStream_t s = nullptr;
                        if (hipStreamCreateWithFlags(
                              &s, hipStreamNonBlocking) == hipSuccess) {
                            peer->base = base;
                            peer->cap = info.size;
                            peer->deviceId = info.deviceId;
                            peer->stream = s;
                            peer->ok = true;
                            FAM_INFO(
                              "ipc: opened %lu MiB peer arena on %s",
                              (unsigned long)(info.size >> 20),
                              targetHost.c_str());
                        } else {
                            (void)hipIpcCloseMemHandle(base);
                        }
                    }
                }
            } catch (const std::exception& e) {
                FAM_WARN("ipc: arena fetch from %s failed: %s",
                         targetHost.c_str(),
                         e.what());
            }
        }
    }

    std::lock_guard<std::mutex> lock(mx);
    auto [it, inserted] = peers.emplace(targetHost, peer);
    return it->second; // keep the winner if two threads raced
}

bool IpcSender::available(const std::string& targetHost)
{
    return ensurePeer(targetHost)->ok;
}

uint64_t IpcSender::shipImpl(const std::string& targetHost,
                             const void* ptr,
                             size_t size,
                             bool fromHost)
{
    auto peer = ensurePeer(targetHost);
    if (!peer->ok) {
        throw FaabricException("ipc: no peer arena to " + targetHost);
    }
    uint64_t n = alignUp(size);
    if (n > peer->cap) {
        throw FaabricException("ipc: payload larger than peer arena");
    }

    uint64_t off = 0;
    {
        std::unique_lock<std::mutex> lock(peer->mx);
        auto deadline = std::chrono::steady_clock::now() +
                        std::chrono::seconds(60);
        while (!peer->tryAlloc(n, off)) {
            if (peer->cv.wait_until(lock, deadline) ==
                std::cv_status::timeout) {
                throw FaabricException(
                  "ipc: timed out waiting for arena space");
            }
        }
        peer->inUse[off] = n;
    }

    // Peer copy over xGMI (or same-GPU D2D) on our side stream; sync so
    // the control message never races the data
    hipError_t rc = hipMemcpyAsync(
      (uint8_t*)peer->base + off,
      ptr,
      size,
      fromHost ? hipMemcpyHostToDevice : hipMemcpyDeviceToDevice,
      peer->stream);
    if (rc == hipSuccess) {
        rc = hipStreamSynchronize(peer->stream);
    }
    if (rc != hipSuccess) {
        std::lock_guard<std::mutex> lock(peer->mx);
        peer->inUse.erase(off);
        peer->cv.notify_all();
        throw FaabricException("ipc: ship copy failed");
    }
    nShipped.fetch_add(1, std::memory_order_relaxed);
    bytesShipped.fetch_add(size, std::memory_order_relaxed);
    return off;
}

uint64_t IpcSender::shippedSegments() const
{
    return nShipped.load(std::memory_order_relaxed);
}

uint64_t IpcSender::shippedBytes() const
{
    return bytesShipped.load(std::memory_order_relaxed);
}

uint64_t IpcSender::peerCapacity(const std::string& targetHost)
{
    auto peer = ensurePeer(targetHost);
    return peer->ok ? peer->cap : 0;
}

uint64_t IpcSender::ship(const std::string& targetHost,
                         const void* devPtr,
                         size_t size)
{
    return shipImpl(targetHost, devPtr, size, /*fromHost=*/false);
}

uint64_t IpcSender::shipFromHost(const std::string& targetHost,
                                 const void* hostPtr,
                                 size_t size)
{
    return shipImpl(targetHost, hostPtr, size, /*fromHost=*/true);
}

void IpcSender::onAck(const std::string& targetHost,
                      uint64_t offset,
                      uint64_t size)
{
    std::shared_ptr<Peer> peer;
    {
        std::lock_guard<std::mutex> lock(mx);
        auto it = peers.find(targetHost);
        if (it == peers.end()) {
            return; // peer already cleared
        }
        peer = it->second;
    }
    std::lock_guard<std::mutex> lock(peer->mx);
    auto it = peer->inUse.find(offset);
    if (it != peer->inUse.end() && it->second == alignUp(size)) {
        peer->inUse.erase(it);
        peer->cv.notify_all();
    } else {
        FAM_WARN("ipc: stray ack off=%lu size=%lu from %s",
                 (unsigned long)offset,
                 (unsigned long)size,
                 targetHost.c_str());
    }
}

void IpcSender::clear()
{
    std::lock_guard<std::mutex> lock(mx);
    peers.clear();
}

IpcSender::~IpcSender()
{
    std::lock_guard<std::mutex> lock(mx);
    for (auto& [host, peer] : peers) {
        peer->base = nullptr;
        peer->stream = nullptr;
    }
    peers.clear();
}

} // namespace faabricamd
