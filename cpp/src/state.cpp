// State KV implementation (reference behavior: src/state/StateKeyValue.cpp,
// InMemoryStateKeyValue.cpp:1-187, StateServer.cpp:24-164,
// StateClient.cpp). See state.h for the re-design notes.
#include "faabricamd/state.h"
#include "faabricamd/ops.h"
#include "faabricamd/ptp.h"
#include "faabricamd/util.h"

#include <hip/hip_runtime.h>

#include <algorithm>
#include <cstring>

namespace faabricamd {

static std::string kvKeyOf(const std::string& user, const std::string& key)
{
    return user + "/" + key;
}

struct MirrorStripe
{
    std::mutex mx;
    void* stream = nullptr;
};

// Process-global stripe pool shared by all device KVs
static MirrorStripe gStripes[StateKeyValue::KV_STRIPES];
static std::mutex gStripesMx;
static bool gStripesReady = false;

static bool ensureGlobalStripes(int device)
{
    std::lock_guard<std::mutex> lock(gStripesMx);
    if (gStripesReady) {
        return true;
    }
    (void)hipSetDevice(device);
    for (int i = 0; i < StateKeyValue::KV_STRIPES; i++) {
        if (gStripes[i].stream == nullptr) {
            hipStream_t st = nullptr;
            if (hipStreamCreateWithFlags(&st, hipStreamNonBlocking) !=
                hipSuccess) {
                return false;
            }
            gStripes[i].stream = (void*)st;
        }
    }
    gStripesReady = true;
    return true;
}

StateKeyValue::StateKeyValue(std::string userIn,
                             std::string keyIn,
                             size_t sizeIn,
                             std::string masterHostIn,
                             bool onDeviceIn,
                             int deviceIn)
  : user(std::move(userIn))
  , key(std::move(keyIn))
  , valueSize(sizeIn)
  , masterHost(std::move(masterHostIn))
  , onDevice(onDeviceIn)
  , device(deviceIn)
{
    if (onDevice && device < 0) {
        device = getSystemConfig().gpuDevice;
    }
    if (onDevice) {
        if (!gpuAvailable()) {
            throw FaabricException("device state KV requires a GPU");
        }
        if (hipSetDevice(device) != hipSuccess ||
            hipMalloc(&devPtr, valueSize) != hipSuccess) {
            throw FaabricException("HBM alloc for state KV failed");
        }
        hipMemset(devPtr, 0, valueSize);
    } else {
        value.resize(valueSize, 0);
    }
    size_t nChunks =
      (valueSize + STATE_STREAM_CHUNK_SIZE - 1) / STATE_STREAM_CHUNK_SIZE;
    dirtyChunks.resize(std::max<size_t>(nChunks, 1), 0);
}

StateKeyValue::~StateKeyValue()
{
    if (stripesReady) {
        // Streams are process-global; just drain writes targeting our
        // HBM before it is freed
        for (int s = 0; s < KV_STRIPES; s++) {
            std::lock_guard<std::mutex> lock(gStripes[s].mx);
            (void)hipStreamSynchronize((hipStream_t)gStripes[s].stream);
        }
    }
    if (mirror != nullptr) {
        (void)hipHostFree(mirror);
    }
    if (devPtr != nullptr) {
        hipFree(devPtr);
    }
}

// ------------------------- pinned mirror ------------------------------------

static constexpr size_t MIRROR_PAGE = 4096;

static size_t mirrorMaxBytes()
{
    static const size_t v =
      (size_t)getEnvVarInt("FAABRIC_KV_MIRROR_MAX_MB", 64) * 1024 * 1024;
    return v;
}

bool StateKeyValue::mirrorUsable()
{
    if (!onDevice || mirrorFailed || valueSize > mirrorMaxBytes() ||
        valueSize == 0) {
        return false;
    }
    if (mirror != nullptr) {
        return true;
    }
    std::lock_guard<std::mutex> lock(mirrorMx);
    if (mirror != nullptr) {
        return true;
    }
    if (mirrorFailed) {
        return false;
    }
    void* pinned = nullptr;
    if (hipSetDevice(device) != hipSuccess ||
        hipHostMalloc(&pinned, valueSize, hipHostMallocDefault) !=
          hipSuccess) {
        mirrorFailed = true;
        return false;
    }
    if (!ensureGlobalStripes(device)) {
        (void)hipHostFree(pinned);
        mirrorFailed = true;
        return false;
    }
    mirrorValid.assign((valueSize + MIRROR_PAGE - 1) / MIRROR_PAGE, 0);
    mirrorDirty.assign(mirrorValid.size(), 0);
    stripesReady = true;
    mirror = (uint8_t*)pinned;
    return true;
}

template<typename Fn>
void StateKeyValue::forEachStripeRange(uint64_t offset, size_t len, Fn&& fn)
{
    uint64_t pos = offset;
    uint64_t end = offset + len;
    while (pos < end) {
        uint64_t blockEnd =
          (pos / KV_STRIPE_BLOCK + 1) * KV_STRIPE_BLOCK;
        uint64_t n = std::min(end, blockEnd) - pos;
        int s = stripeOf(pos);
        std::lock_guard<std::mutex> lock(gStripes[s].mx);
        fn(s, pos, (size_t)n);
        pos += n;
    }
}

// Runtime stripe count (1..KV_STRIPES) for A/B measurement
static int kvStripeCount()
{
    static const int v = []() {
        int n = getEnvVarInt("FAABRIC_KV_STRIPES", StateKeyValue::KV_STRIPES);
        if (n < 1) {
            n = 1;
        }
        if (n > StateKeyValue::KV_STRIPES) {
            n = StateKeyValue::KV_STRIPES;
        }
        return n;
    }();
    return v;
}

int StateKeyValue::stripeOf(uint64_t offset) const
{
    // Salt by KV identity so concurrent KVs do not all contend on the
    // same stripes of the global pool
    uint64_t salt = (uint64_t)(uintptr_t)this >> 6;
    return (int)((salt + offset / KV_STRIPE_BLOCK) % kvStripeCount());
}

void StateKeyValue::mirrorFillLocked(int stripe,
                                     uint64_t offset,
                                     size_t len)
{
    // Caller holds the stripe's mutex; [offset, +len) lies inside this
    // stripe's block. Fill whole stale pages. The D2H rides the stripe
    // stream, so it is ordered after that range's pending H2Ds.
    size_t firstPage = offset / MIRROR_PAGE;
    size_t lastPage = (offset + len - 1) / MIRROR_PAGE;
    size_t run = 0;
    bool filled = false;
    for (size_t p = firstPage; p <= lastPage + 1; p++) {
        bool stale = p <= lastPage && mirrorValid[p] == 0;
        if (stale) {
            run++;
            continue;
        }
        if (run > 0) {
            size_t start = (p - run) * MIRROR_PAGE;
            size_t n = std::min(run * MIRROR_PAGE, valueSize - start);
            (void)hipMemcpyAsync(mirror + start, devPtr + start, n,
                                 hipMemcpyDeviceToHost,
                                 (hipStream_t)gStripes[stripe].stream);
            for (size_t q = p - run; q < p; q++) {
                mirrorValid[q] = 1;
            }
            filled = true;
            run = 0;
        }
    }
    if (filled) {
        (void)hipStreamSynchronize((hipStream_t)gStripes[stripe].stream);
    }
}

void StateKeyValue::mirrorInvalidate(uint64_t offset, size_t len)
{
    if (mirror == nullptr || len == 0) {
        return;
    }
    forEachStripeRange(offset, len, [&](int s, uint64_t o, size_t n) {
        (void)s;
        size_t firstPage = o / MIRROR_PAGE;
        size_t lastPage = (o + n - 1) / MIRROR_PAGE;
        for (size_t p = firstPage;
             p <= lastPage && p < mirrorValid.size(); p++) {
            mirrorValid[p] = 0;
            // Direct-HBM landings are ordered after sync(); any dirty
            // bit left here belongs to a write the landing overwrote
            mirrorDirty[p] = 0;
        }
    });
}

void StateKeyValue::sync()
{
    if (mirror == nullptr) {
        return;
    }
    if (mirrorAnyDirty.exchange(false, std::memory_order_acq_rel)) {
        (void)hipSetDevice(device);
        // Coalesce dirty pages into maximal runs; each run is enqueued
        // per stripe block under that stripe's lock (mutual exclusion
        // with writers keeps the page content consistent). Pages
        // dirtied concurrently re-set the flag and are committed by
        // their writer's own sync.
        size_t nPages = mirrorDirty.size();
        size_t p = 0;
        while (p < nPages) {
            if (mirrorDirty[p] == 0) {
                p++;
                continue;
            }
            size_t q = p;
            while (q < nPages && mirrorDirty[q] != 0) {
                mirrorDirty[q] = 0;
                q++;
            }
            uint64_t start = (uint64_t)p * MIRROR_PAGE;
            size_t len =
              std::min<uint64_t>((uint64_t)q * MIRROR_PAGE, valueSize) -
              start;
            forEachStripeRange(start, len, [&](int s, uint64_t o,
                                               size_t n) {
                (void)hipMemcpyAsync(devPtr + o, mirror + o, n,
                                     hipMemcpyHostToDevice,
                                     (hipStream_t)gStripes[s].stream);
            });
            p = q;
        }
    }
    for (int s = 0; s < KV_STRIPES; s++) {
        std::lock_guard<std::mutex> lock(gStripes[s].mx);
        (void)hipStreamSynchronize((hipStream_t)gStripes[s].stream);
    }
}

// Per-thread HIP stream so concurrent chunk ops from different executors
// pipeline through the copy engines instead of serialising on the null
// stream
static hipStream_t threadCopyStream()
{
    thread_local hipStream_t stream = []() {
        hipStream_t s = nullptr;
        if (hipStreamCreateWithFlags(&s, hipStreamNonBlocking) !=
            hipSuccess) {
            s = nullptr;
        }
        return s;
    }();
    return stream;
}

// Write policy, measured same-box (profiles/kv_writeback_ab.json):
// write-through (default) enqueues one H2D per op from the worker
// thread that did the write — the enqueues overlap batch execution.
// Write-back (=1) marks pages dirty and coalesces H2Ds at sync(), but
// that serializes the whole flush on the sync caller's critical path:
// 23% slower at 128-function batches. Kept as a knob because the
// trade flips for workloads with many tiny writes and rare syncs.
static bool kvWriteBack()
{
    static const bool v =
      getEnvVarInt("FAABRIC_KV_WRITEBACK", 0) != 0;
    return v;
}

static bool kvAsyncCopy()
{
    static bool v = getEnvVarInt("FAABRIC_KV_ASYNC_COPY", 0) != 0;
    return v;
}

void StateKeyValue::readLocal(uint64_t offset, uint8_t* out, size_t len)
{
    if (onDevice) {
        if (mirrorUsable()) {
            (void)hipSetDevice(device);
            forEachStripeRange(
              offset, len, [&](int s, uint64_t o, size_t n) {
                  mirrorFillLocked(s, o, n); // no-op when pages valid
                  std::memcpy(out + (o - offset), mirror + o, n);
              });
            return;
        }
        (void)hipSetDevice(device);
        if (kvAsyncCopy()) {
            hipStream_t s = threadCopyStream();
            (void)hipMemcpyAsync(out, devPtr + offset, len,
                                 hipMemcpyDeviceToHost, s);
            (void)hipStreamSynchronize(s);
        } else {
            (void)hipMemcpy(out, devPtr + offset, len,
                            hipMemcpyDeviceToHost);
        }
    } else {
        std::memcpy(out, value.data() + offset, len);
    }
}

void StateKeyValue::writeLocal(uint64_t offset,
                               const uint8_t* data,
                               size_t len)
{
    if (onDevice) {
        if (mirrorUsable()) {
            // Group-commit write-back: memcpy into the pinned mirror and
            // mark the pages dirty; sync() (the durability point)
            // coalesces dirty runs into few H2D enqueues. Reads serve
            // from the mirror, so they always see the newest bytes.
            (void)hipSetDevice(device);
            forEachStripeRange(
              offset, len, [&](int s, uint64_t o, size_t n) {
                  // A partially-written invalid boundary page must be
                  // filled from HBM first, or its untouched bytes would
                  // read stale
                  size_t firstPage = o / MIRROR_PAGE;
                  size_t lastPage = (o + n - 1) / MIRROR_PAGE;
                  auto coversPage = [&](size_t p) {
                      uint64_t start = p * MIRROR_PAGE;
                      uint64_t end = std::min<uint64_t>(
                        start + MIRROR_PAGE, valueSize);
                      return o <= start && o + n >= end;
                  };
                  if (mirrorValid[firstPage] == 0 &&
                      !coversPage(firstPage)) {
                      mirrorFillLocked(s, firstPage * MIRROR_PAGE, 1);
                  }
                  if (lastPage != firstPage &&
                      mirrorValid[lastPage] == 0 &&
                      !coversPage(lastPage)) {
                      mirrorFillLocked(s, lastPage * MIRROR_PAGE, 1);
                  }
                  std::memcpy(mirror + o, data + (o - offset), n);
                  if (kvWriteBack()) {
                      for (size_t p = firstPage; p <= lastPage; p++) {
                          mirrorValid[p] = 1;
                          mirrorDirty[p] = 1;
                      }
                  } else {
                      for (size_t p = firstPage; p <= lastPage; p++) {
                          mirrorValid[p] = 1;
                      }
                      (void)hipMemcpyAsync(
                        devPtr + o, mirror + o, n,
                        hipMemcpyHostToDevice,
                        (hipStream_t)gStripes[s].stream);
                  }
              });
            if (kvWriteBack()) {
                mirrorAnyDirty.store(true, std::memory_order_release);
            }
            return;
        }
        (void)hipSetDevice(device);
        if (kvAsyncCopy()) {
            hipStream_t s = threadCopyStream();
            (void)hipMemcpyAsync(devPtr + offset, data, len,
                                 hipMemcpyHostToDevice, s);
            (void)hipStreamSynchronize(s);
        } else {
            (void)hipMemcpy(devPtr + offset, data, len,
                            hipMemcpyHostToDevice);
        }
    } else {
        std::memcpy(value.data() + offset, data, len);
    }
}

uint8_t* StateKeyValue::getDataPtr()
{
    if (!onDevice) {
        return value.data();
    }
    if (mirror != nullptr) {
        sync();
        mirrorInvalidate(0, valueSize);
    }
    return devPtr;
}

bool StateKeyValue::isMaster() const
{
    return masterHost == getSystemConfig().endpointHost;
}

bool StateKeyValue::useIpcToMaster()
{
    // Bulk device-value traffic to a same-node worker rides HIP IPC;
    // host values and cross-node masters keep the RPC plane
    if (!onDevice || isMaster()) {
        return false;
    }
    const std::string& thisHost = getSystemConfig().endpointHost;
    return isSameNodeDifferentWorker(thisHost, masterHost) &&
           IpcSender::get().available(masterHost);
}

void StateKeyValue::pullRangeIpc(uint64_t offset, size_t len)
{
    // Master ships straight out of its value into OUR arena; we land it
    // in HBM with a local D2D and ack the segment. Chunk size adapts to
    // whatever the master could ship (clamped to its view of our arena).
    const std::string& thisHost = getSystemConfig().endpointHost;
    auto cli = getStateClient(masterHost);
    uint64_t off = offset;
    uint64_t end = offset + len;
    while (off < end) {
        IpcChunk got = cli->pullChunkIpc(user, key, off, end - off,
                                         thisHost);
        if (got.len == 0 || off + got.len > end) {
            throw FaabricException("state ipc pull returned bad chunk");
        }
        (void)hipSetDevice(device);
        sync(); // pending mirror writes must not land on top of the pull
        IpcReceiver::get().copyToDevice(got.srcHost, got.ipcOffset,
                                        devPtr + off, got.len);
        mirrorInvalidate(off, got.len);
        getPointToPointBroker().sendIpcAck(got.srcHost, got.ipcOffset,
                                           got.len);
        off += got.len;
    }
}

void StateKeyValue::pushRangeIpc(uint64_t offset, size_t len)
{
    // Ship from our HBM into the master's arena, then a small RPC tells
    // it where; the master lands + acks. Chunked to half the arena so
    // consecutive chunks can overlap ship and land.
    const std::string& thisHost = getSystemConfig().endpointHost;
    uint64_t cap = IpcSender::get().peerCapacity(masterHost);
    if (cap == 0) {
        throw FaabricException("state ipc push without peer arena");
    }
    uint64_t chunk = std::max<uint64_t>(cap / 2, DEVICE_PAGE);
    auto cli = getStateClient(masterHost);
    uint64_t off = offset;
    uint64_t end = offset + len;
    sync(); // HBM must be current before shipping out of it
    while (off < end) {
        uint64_t n = std::min(chunk, end - off);
        (void)hipSetDevice(device);
        uint64_t ipcOff =
          IpcSender::get().ship(masterHost, devPtr + off, n);
        IpcChunk c;
        c.user = user;
        c.key = key;
        c.valOffset = off;
        c.ipcOffset = ipcOff;
        c.len = n;
        c.srcHost = thisHost;
        c.totalSize = valueSize;
        cli->pushChunkIpc(c);
        off += n;
    }
}

void StateKeyValue::get(uint8_t* buffer)
{
    pull();
    std::lock_guard<std::mutex> lock(kvMx);
    readLocal(0, buffer, valueSize);
}

std::vector<uint8_t> StateKeyValue::get()
{
    std::vector<uint8_t> out(valueSize);
    get(out.data());
    return out;
}

void StateKeyValue::set(const uint8_t* buffer, size_t n)
{
    {
        std::lock_guard<std::mutex> lock(kvMx);
        if (n > valueSize) {
            throw FaabricException("state set exceeds value size");
        }
        writeLocal(0, buffer, n);
        fullyPulled = true;
    }
    flagDirty();
    if (!isMaster()) {
        pushFull();
    }
}

void StateKeyValue::set(const std::vector<uint8_t>& data)
{
    set(data.data(), data.size());
}

void StateKeyValue::getChunk(uint64_t offset, uint8_t* buffer, size_t len)
{
    if (offset + len > valueSize) {
        throw FaabricException("state chunk read out of bounds");
    }
    // Lazy per-chunk pull: each streaming chunk is fetched from the
    // master at most once (reference: StateKeyValue pulledMask,
    // state/StateKeyValue.h:86-101); local writes keep chunks fresh
    // via flagChunkDirty/pulledChunks marking below
    if (!isMaster()) {
        bool needPull = false;
        size_t first = offset / STATE_STREAM_CHUNK_SIZE;
        size_t last = (offset + len - 1) / STATE_STREAM_CHUNK_SIZE;
        {
            std::lock_guard<std::mutex> lock(kvMx);
            if (!fullyPulled) {
                if (pulledChunks.size() != dirtyChunks.size()) {
                    pulledChunks.assign(dirtyChunks.size(), 0);
                }
                for (size_t c = first; c <= last; c++) {
                    if (pulledChunks[c] == 0 && dirtyChunks[c] == 0) {
                        needPull = true;
                        break;
                    }
                }
            }
        }
        // NOTE: chunks are marked pulled only after a SUCCESSFUL pull
        // below — marking first would turn a failed pull into silent
        // stale reads forever after
        auto markPulled = [&] {
            std::lock_guard<std::mutex> lock(kvMx);
            for (size_t c = first; c <= last; c++) {
                pulledChunks[c] = 1;
            }
        };
        if (!needPull) {
            if (onDevice) {
                readLocal(offset, buffer, len);
            } else {
                std::lock_guard<std::mutex> lock(kvMx);
                readLocal(offset, buffer, len);
            }
            return;
        }
        if (useIpcToMaster()) {
            try {
                pullRangeIpc(offset, len);
                readLocal(offset, buffer, len);
                markPulled();
                return;
            } catch (const std::exception& e) {
                FAM_WARN("state ipc chunk pull failed (%s); falling back",
                         e.what());
            }
        }
        // Lazy chunked pull of just this range
        auto cli = getStateClient(masterHost);
        auto data = cli->pullChunk(user, key, offset, len);
        if (onDevice) {
            writeLocal(offset, data.data(), data.size());
        } else {
            std::lock_guard<std::mutex> lock(kvMx);
            writeLocal(offset, data.data(), data.size());
        }
        std::memcpy(buffer, data.data(), len);
        markPulled();
        return;
    }
    if (onDevice) {
        readLocal(offset, buffer, len);
        return;
    }
    std::lock_guard<std::mutex> lock(kvMx);
    readLocal(offset, buffer, len);
}

void StateKeyValue::setChunk(uint64_t offset, const uint8_t* buffer,
                             size_t len)
{
    if (offset + len > valueSize) {
        throw FaabricException("state chunk write out of bounds");
    }
    if (onDevice) {
        writeLocal(offset, buffer, len);
    } else {
        std::lock_guard<std::mutex> lock(kvMx);
        writeLocal(offset, buffer, len);
    }
    flagChunkDirty(offset, len);
    {
        // Locally-written chunks are fresh: never re-pull them
        std::lock_guard<std::mutex> lock(kvMx);
        if (pulledChunks.size() != dirtyChunks.size()) {
            pulledChunks.assign(dirtyChunks.size(), 0);
        }
        for (size_t c = offset / STATE_STREAM_CHUNK_SIZE;
             c <= (offset + len - 1) / STATE_STREAM_CHUNK_SIZE; c++) {
            pulledChunks[c] = 1;
        }
    }
    if (!isMaster()) {
        if (useIpcToMaster()) {
            try {
                pushRangeIpc(offset, len);
                return;
            } catch (const std::exception& e) {
                FAM_WARN("state ipc chunk push failed (%s); falling back",
                         e.what());
            }
        }
        auto cli = getStateClient(masterHost);
        cli->pushChunk(user, key, offset, buffer, len, valueSize);
    }
}

void StateKeyValue::pull()
{
    if (isMaster()) {
        return;
    }
    {
        std::lock_guard<std::mutex> lock(kvMx);
        if (fullyPulled) {
            return;
        }
    }
    if (useIpcToMaster()) {
        try {
            pullRangeIpc(0, valueSize);
            std::lock_guard<std::mutex> lock(kvMx);
            fullyPulled = true;
            return;
        } catch (const std::exception& e) {
            FAM_WARN("state ipc pull failed (%s); falling back",
                     e.what());
        }
    }
    auto cli = getStateClient(masterHost);
    for (uint64_t off = 0; off < valueSize;
         off += STATE_STREAM_CHUNK_SIZE) {
        size_t len = std::min(STATE_STREAM_CHUNK_SIZE,
                              (size_t)(valueSize - off));
        auto data = cli->pullChunk(user, key, off, len);
        std::lock_guard<std::mutex> lock(kvMx);
        writeLocal(off, data.data(), data.size());
    }
    std::lock_guard<std::mutex> lock(kvMx);
    fullyPulled = true;
}

void StateKeyValue::pushFull()
{
    if (isMaster()) {
        return;
    }
    if (useIpcToMaster()) {
        try {
            pushRangeIpc(0, valueSize);
            std::lock_guard<std::mutex> lock(kvMx);
            std::fill(dirtyChunks.begin(), dirtyChunks.end(), 0);
            return;
        } catch (const std::exception& e) {
            FAM_WARN("state ipc push failed (%s); falling back",
                     e.what());
        }
    }
    auto cli = getStateClient(masterHost);
    std::lock_guard<std::mutex> lock(kvMx);
    std::vector<uint8_t> staging(STATE_STREAM_CHUNK_SIZE);
    for (uint64_t off = 0; off < valueSize;
         off += STATE_STREAM_CHUNK_SIZE) {
        size_t len = std::min(STATE_STREAM_CHUNK_SIZE,
                              (size_t)(valueSize - off));
        readLocal(off, staging.data(), len);
        cli->pushChunk(user, key, off, staging.data(), len,
                       valueSize);
    }
    std::fill(dirtyChunks.begin(), dirtyChunks.end(), 0);
}

void StateKeyValue::flagDirty()
{
    std::lock_guard<std::mutex> lock(kvMx);
    std::fill(dirtyChunks.begin(), dirtyChunks.end(), 1);
}

void StateKeyValue::flagChunkDirty(uint64_t offset, size_t len)
{
    std::lock_guard<std::mutex> lock(kvMx);
    size_t first = offset / STATE_STREAM_CHUNK_SIZE;
    size_t last = (offset + len - 1) / STATE_STREAM_CHUNK_SIZE;
    for (size_t i = first; i <= last && i < dirtyChunks.size(); i++) {
        dirtyChunks[i] = 1;
    }
}

void StateKeyValue::pushPartial()
{
    if (isMaster()) {
        std::lock_guard<std::mutex> lock(kvMx);
        std::fill(dirtyChunks.begin(), dirtyChunks.end(), 0);
        return;
    }
    if (useIpcToMaster()) {
        try {
            // Coalesce contiguous dirty 64 KiB chunks into single ships
            std::lock_guard<std::mutex> lock(kvMx);
            size_t i = 0;
            while (i < dirtyChunks.size()) {
                if (dirtyChunks[i] == 0) {
                    i++;
                    continue;
                }
                size_t j = i;
                while (j < dirtyChunks.size() && dirtyChunks[j] != 0) {
                    j++;
                }
                uint64_t off = i * STATE_STREAM_CHUNK_SIZE;
                size_t len = std::min((j - i) * STATE_STREAM_CHUNK_SIZE,
                                      (size_t)(valueSize - off));
                pushRangeIpc(off, len);
                std::fill(dirtyChunks.begin() + i,
                          dirtyChunks.begin() + j, 0);
                i = j;
            }
            return;
        } catch (const std::exception& e) {
            FAM_WARN("state ipc partial push failed (%s); falling back",
                     e.what());
        }
    }
    auto cli = getStateClient(masterHost);
    std::lock_guard<std::mutex> lock(kvMx);
    for (size_t i = 0; i < dirtyChunks.size(); i++) {
        if (dirtyChunks[i] == 0) {
            continue;
        }
        uint64_t off = i * STATE_STREAM_CHUNK_SIZE;
        size_t len = std::min(STATE_STREAM_CHUNK_SIZE,
                              (size_t)(valueSize - off));
        std::vector<uint8_t> staging(len);
        readLocal(off, staging.data(), len);
        cli->pushChunk(user, key, off, staging.data(), len,
                       valueSize);
        dirtyChunks[i] = 0;
    }
}

void StateKeyValue::append(const uint8_t* data, size_t len)
{
    if (isMaster()) {
        serviceAppend(data, len);
        return;
    }
    auto cli = getStateClient(masterHost);
    cli->append(user, key, data, len);
}

std::vector<std::vector<uint8_t>> StateKeyValue::getAppended(size_t nValues)
{
    if (isMaster()) {
        return serviceGetAppended(nValues);
    }
    auto cli = getStateClient(masterHost);
    return cli->pullAppended(user, key, nValues);
}

void StateKeyValue::clearAppended()
{
    if (isMaster()) {
        serviceClearAppended();
        return;
    }
    auto cli = getStateClient(masterHost);
    cli->clearAppended(user, key);
}

std::vector<uint8_t> StateKeyValue::serviceChunk(uint64_t offset, size_t len)
{
    if (offset + len > valueSize) {
        throw FaabricException("state chunk service out of bounds");
    }
    std::vector<uint8_t> out(len);
    if (onDevice) {
        readLocal(offset, out.data(), len);
        return out;
    }
    std::lock_guard<std::mutex> lock(kvMx);
    readLocal(offset, out.data(), len);
    return out;
}

void StateKeyValue::serviceSet(uint64_t offset,
                               const uint8_t* data,
                               size_t len)
{
    std::lock_guard<std::mutex> lock(kvMx);
    if (offset + len > valueSize) {
        throw FaabricException("state chunk service-set out of bounds");
    }
    writeLocal(offset, data, len);
}

uint64_t StateKeyValue::serviceChunkIpc(const std::string& dstHost,
                                        uint64_t offset,
                                        size_t len)
{
    if (offset > valueSize || len > valueSize ||
        offset + len > valueSize) {
        throw FaabricException("state ipc chunk out of bounds");
    }
    if (onDevice) {
        (void)hipSetDevice(device);
        sync(); // HBM must be current before shipping out of it
        return IpcSender::get().ship(dstHost, devPtr + offset, len);
    }
    std::lock_guard<std::mutex> lock(kvMx);
    return IpcSender::get().shipFromHost(dstHost, value.data() + offset,
                                         len);
}

void StateKeyValue::serviceSetIpc(const std::string& srcHost,
                                  uint64_t ipcOffset,
                                  uint64_t valOffset,
                                  size_t len)
{
    if (valOffset > valueSize || len > valueSize ||
        valOffset + len > valueSize) {
        throw FaabricException("state ipc set out of bounds");
    }
    if (onDevice) {
        (void)hipSetDevice(device);
        sync(); // order the landing write after pending mirror writes
        IpcReceiver::get().copyToDevice(srcHost, ipcOffset,
                                        devPtr + valOffset, len);
        mirrorInvalidate(valOffset, len);
    } else {
        std::lock_guard<std::mutex> lock(kvMx);
        IpcReceiver::get().copyToHost(srcHost, ipcOffset,
                                      value.data() + valOffset, len);
    }
    getPointToPointBroker().sendIpcAck(srcHost, ipcOffset, len);
}

void StateKeyValue::serviceAppend(const uint8_t* data, size_t len)
{
    std::lock_guard<std::mutex> lock(kvMx);
    appendedValues.emplace_back(data, data + len);
}

std::vector<std::vector<uint8_t>> StateKeyValue::serviceGetAppended(size_t n)
{
    std::lock_guard<std::mutex> lock(kvMx);
    if (n > appendedValues.size()) {
        throw FaabricException("not enough appended values");
    }
    return { appendedValues.begin(), appendedValues.begin() + n };
}

void StateKeyValue::serviceClearAppended()
{
    std::lock_guard<std::mutex> lock(kvMx);
    appendedValues.clear();
}

// ------------------------- State --------------------------------------------

State& State::get()
{
    static State st;
    return st;
}

// Default owner for a key with no pinned master. "inmemory" mode is the
// reference's master-per-key backend (first-toucher owns); "planner"
// mode is the second backend — the planner process hosts a global
// StateServer and owns every key, playing the role of the reference's
// Redis service (src/state/RedisStateKeyValue.cpp, redis/Redis.h).
static std::string defaultMasterHost()
{
    auto& conf = getSystemConfig();
    if (conf.stateMode == "planner") {
        return conf.plannerHost;
    }
    return conf.endpointHost;
}

std::shared_ptr<StateKeyValue> State::getKV(const std::string& user,
                                            const std::string& key,
                                            size_t size)
{
    std::lock_guard<std::mutex> lock(mx);
    std::string k = kvKeyOf(user, key);
    auto it = kvMap.find(k);
    if (it != kvMap.end()) {
        return it->second;
    }
    std::string master;
    auto mIt = masterMap.find(k);
    if (mIt != masterMap.end()) {
        master = mIt->second;
    } else {
        master = defaultMasterHost();
        masterMap[k] = master;
    }
    auto kv = std::make_shared<StateKeyValue>(user, key, size, master);
    kvMap[k] = kv;
    return kv;
}

std::shared_ptr<StateKeyValue> State::getKVDevice(const std::string& user,
                                                  const std::string& key,
                                                  size_t size,
                                                  int device)
{
    std::lock_guard<std::mutex> lock(mx);
    std::string k = kvKeyOf(user, key);
    auto it = kvMap.find(k);
    if (it != kvMap.end()) {
        return it->second;
    }
    std::string master;
    auto mIt = masterMap.find(k);
    if (mIt != masterMap.end()) {
        master = mIt->second;
    } else {
        master = defaultMasterHost();
        masterMap[k] = master;
    }
    auto kv = std::make_shared<StateKeyValue>(
      user, key, size, master, /*onDevice=*/true, device);
    kvMap[k] = kv;
    return kv;
}

std::shared_ptr<StateKeyValue> State::getKV(const std::string& user,
                                            const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    auto it = kvMap.find(kvKeyOf(user, key));
    if (it == kvMap.end()) {
        throw FaabricException("state key not found: " +
                               kvKeyOf(user, key));
    }
    return it->second;
}

size_t State::getStateSize(const std::string& user, const std::string& key)
{
    {
        std::lock_guard<std::mutex> lock(mx);
        auto it = kvMap.find(kvKeyOf(user, key));
        if (it != kvMap.end()) {
            return it->second->size();
        }
    }
    // Ask the owner if we know one
    std::string master = getMasterHost(user, key);
    if (!master.empty() && master != getSystemConfig().endpointHost) {
        return getStateClient(master)->stateSize(user, key);
    }
    return 0;
}

void State::deleteKV(const std::string& user, const std::string& key)
{
    std::string master;
    {
        std::lock_guard<std::mutex> lock(mx);
        auto mIt = masterMap.find(kvKeyOf(user, key));
        master = mIt == masterMap.end() ? "" : mIt->second;
    }
    if (!master.empty() && master != getSystemConfig().endpointHost) {
        getStateClient(master)->deleteKV(user, key);
    }
    deleteKVLocally(user, key);
}

void State::deleteKVLocally(const std::string& user, const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    kvMap.erase(kvKeyOf(user, key));
    masterMap.erase(kvKeyOf(user, key));
}

size_t State::getKVCount()
{
    std::lock_guard<std::mutex> lock(mx);
    return kvMap.size();
}

void State::syncAll()
{
    std::vector<std::shared_ptr<StateKeyValue>> kvs;
    {
        std::lock_guard<std::mutex> lock(mx);
        kvs.reserve(kvMap.size());
        for (auto& [k, kv] : kvMap) {
            kvs.push_back(kv);
        }
    }
    for (auto& kv : kvs) {
        kv->sync();
    }
}

void State::forceClearAll(bool global)
{
    (void)global;
    std::lock_guard<std::mutex> lock(mx);
    kvMap.clear();
    masterMap.clear();
}

uint64_t State::acquireLockLocal(const std::string& user,
                                 const std::string& key,
                                 int expiryMs)
{
    std::lock_guard<std::mutex> lock(locksMx);
    std::string k = kvKeyOf(user, key);
    int64_t now = getEpochMicros() / 1000;
    auto it = locks.find(k);
    if (it != locks.end() && it->second.second > now) {
        return 0; // held and not expired
    }
    uint64_t token = ((uint64_t)generateGid() << 32) | generateGid();
    if (token == 0) {
        token = 1;
    }
    locks[k] = { token, now + expiryMs };
    return token;
}

bool State::releaseLockLocal(const std::string& user,
                             const std::string& key,
                             uint64_t token)
{
    // Delete-if-token-matches, like the reference's scripted Redis
    // release (a stale holder cannot free a re-acquired lock)
    std::lock_guard<std::mutex> lock(locksMx);
    std::string k = kvKeyOf(user, key);
    auto it = locks.find(k);
    if (it == locks.end() || it->second.first != token) {
        return false;
    }
    locks.erase(it);
    return true;
}

uint64_t State::acquireLock(const std::string& user,
                            const std::string& key,
                            int expiryMs)
{
    std::string owner = getMasterHost(user, key);
    if (owner.empty()) {
        owner = defaultMasterHost();
    }
    if (owner == getSystemConfig().endpointHost) {
        return acquireLockLocal(user, key, expiryMs);
    }
    return getStateClient(owner)->acquireLock(user, key, expiryMs);
}

void State::releaseLock(const std::string& user,
                        const std::string& key,
                        uint64_t token)
{
    std::string owner = getMasterHost(user, key);
    if (owner.empty()) {
        owner = defaultMasterHost();
    }
    if (owner == getSystemConfig().endpointHost) {
        releaseLockLocal(user, key, token);
        return;
    }
    getStateClient(owner)->releaseLock(user, key, token);
}

void State::setMasterHost(const std::string& user,
                          const std::string& key,
                          const std::string& host)
{
    std::lock_guard<std::mutex> lock(mx);
    masterMap[kvKeyOf(user, key)] = host;
}

std::string State::getMasterHost(const std::string& user,
                                 const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    auto it = masterMap.find(kvKeyOf(user, key));
    return it == masterMap.end() ? "" : it->second;
}

// ------------------------- server -------------------------------------------

StateServer::StateServer()
  : MessageEndpointServer(STATE_ASYNC_PORT, STATE_SYNC_PORT, "state")
{}

void StateServer::doAsyncRecv(uint8_t code,
                              const std::string& body,
                              uint32_t seq)
{
    (void)code;
    (void)body;
    (void)seq;
    FAM_ERROR("state server has no async calls");
}

std::string StateServer::doSyncRecv(uint8_t code, const std::string& body)
{
    auto& st = State::get();
    switch ((StateCalls)code) {
        case StateCalls::Pull: {
            auto req = StateChunkRequest::decode(body);
            auto kv = st.getKV(req.user, req.key);
            StatePart part;
            part.user = req.user;
            part.key = req.key;
            part.offset = req.offset;
            part.data = kv->serviceChunk(req.offset, req.chunkSize);
            return part.encode();
        }
        case StateCalls::Push: {
            auto part = StatePart::decode(body);
            // Create on demand so a push can establish the value (sized
            // by the pusher's full value, not the first chunk)
            auto kv = st.getKV(
              part.user,
              part.key,
              std::max<size_t>(part.totalSize,
                               part.offset + part.data.size()));
            if (part.offset + part.data.size() > kv->size()) {
                throw FaabricException("push beyond registered size");
            }
            kv->serviceSet(part.offset, part.data.data(), part.data.size());
            return {};
        }
        case StateCalls::Size: {
            auto req = StateRequest::decode(body);
            StateSizeResponse resp;
            resp.user = req.user;
            resp.key = req.key;
            resp.stateSize = st.getStateSize(req.user, req.key);
            return resp.encode();
        }
        case StateCalls::PullIpc: {
            // Requester asks us (the master) to ship a range into ITS
            // arena; reply names the shipped segment. Clamp to the
            // requester's arena so a big pull streams in pieces.
            auto req = IpcChunk::decode(body);
            auto kv = st.getKV(req.user, req.key);
            uint64_t cap = IpcSender::get().peerCapacity(req.srcHost);
            if (cap == 0) {
                throw FaabricException("no ipc arena at requester");
            }
            uint64_t len =
              std::min(req.len, std::max<uint64_t>(cap / 2, DEVICE_PAGE));
            IpcChunk out;
            out.user = req.user;
            out.key = req.key;
            out.valOffset = req.valOffset;
            out.ipcOffset =
              kv->serviceChunkIpc(req.srcHost, req.valOffset, len);
            out.len = len;
            out.srcHost = getSystemConfig().endpointHost;
            return out.encode();
        }
        case StateCalls::PushIpc: {
            auto req = IpcChunk::decode(body);
            auto kv = st.getKV(req.user, req.key, req.totalSize);
            kv->serviceSetIpc(req.srcHost, req.ipcOffset, req.valOffset,
                              req.len);
            return {};
        }
        case StateCalls::Lock: {
            auto req = StateRequest::decode(body);
            int32_t expiryMs = 10000;
            if (req.data.size() >= sizeof(expiryMs)) {
                std::memcpy(&expiryMs, req.data.data(), sizeof(expiryMs));
            }
            uint64_t token =
              st.acquireLockLocal(req.user, req.key, expiryMs);
            std::string out(sizeof(token), '\0');
            std::memcpy(out.data(), &token, sizeof(token));
            return out;
        }
        case StateCalls::Unlock: {
            auto req = StateRequest::decode(body);
            uint64_t token = 0;
            if (req.data.size() >= sizeof(token)) {
                std::memcpy(&token, req.data.data(), sizeof(token));
            }
            bool ok = st.releaseLockLocal(req.user, req.key, token);
            return std::string(1, ok ? 1 : 0);
        }
        case StateCalls::Append: {
            auto req = StateRequest::decode(body);
            auto kv = st.getKV(req.user, req.key, 1);
            kv->serviceAppend(req.data.data(), req.data.size());
            return {};
        }
        case StateCalls::PullAppended: {
            auto req = StateAppendedRequest::decode(body);
            auto kv = st.getKV(req.user, req.key);
            StateAppendedResponse resp;
            resp.user = req.user;
            resp.key = req.key;
            resp.values = kv->serviceGetAppended(req.nValues);
            return resp.encode();
        }
        case StateCalls::ClearAppended: {
            auto req = StateRequest::decode(body);
            auto kv = st.getKV(req.user, req.key);
            kv->serviceClearAppended();
            return {};
        }
        case StateCalls::Delete: {
            auto req = StateRequest::decode(body);
            st.deleteKVLocally(req.user, req.key);
            return {};
        }
        default:
            throw FaabricException("state server: bad sync code " +
                                   std::to_string(code));
    }
}

// ------------------------- client -------------------------------------------

StateClient::StateClient(const std::string& host)
  : MessageEndpointClient(host, STATE_ASYNC_PORT, STATE_SYNC_PORT)
{}

std::vector<uint8_t> StateClient::pullChunk(const std::string& user,
                                            const std::string& key,
                                            uint64_t offset,
                                            size_t len)
{
    StateChunkRequest req;
    req.user = user;
    req.key = key;
    req.offset = offset;
    req.chunkSize = len;
    std::string resp = syncSend((uint8_t)StateCalls::Pull, req.encode());
    return StatePart::decode(resp).data;
}

IpcChunk StateClient::pullChunkIpc(const std::string& user,
                                   const std::string& key,
                                   uint64_t offset,
                                   size_t len,
                                   const std::string& requesterHost)
{
    IpcChunk req;
    req.user = user;
    req.key = key;
    req.valOffset = offset;
    req.len = len;
    req.srcHost = requesterHost; // who the master should ship to
    std::string resp =
      syncSend((uint8_t)StateCalls::PullIpc, req.encode());
    return IpcChunk::decode(resp);
}

void StateClient::pushChunkIpc(const IpcChunk& chunk)
{
    syncSend((uint8_t)StateCalls::PushIpc, chunk.encode());
}

void StateClient::pushChunk(const std::string& user,
                            const std::string& key,
                            uint64_t offset,
                            const uint8_t* data,
                            size_t len,
                            size_t totalSize)
{
    StatePart part;
    part.user = user;
    part.key = key;
    part.offset = offset;
    part.data.assign(data, data + len);
    part.totalSize = totalSize;
    syncSend((uint8_t)StateCalls::Push, part.encode());
}

size_t StateClient::stateSize(const std::string& user, const std::string& key)
{
    StateRequest req;
    req.user = user;
    req.key = key;
    std::string resp = syncSend((uint8_t)StateCalls::Size, req.encode());
    return StateSizeResponse::decode(resp).stateSize;
}

uint64_t StateClient::acquireLock(const std::string& user,
                                  const std::string& key,
                                  int expiryMs)
{
    StateRequest req;
    req.user = user;
    req.key = key;
    req.data.resize(sizeof(int32_t));
    int32_t e = expiryMs;
    std::memcpy(req.data.data(), &e, sizeof(e));
    std::string resp = syncSend((uint8_t)StateCalls::Lock, req.encode());
    uint64_t token = 0;
    if (resp.size() >= sizeof(token)) {
        std::memcpy(&token, resp.data(), sizeof(token));
    }
    return token;
}

bool StateClient::releaseLock(const std::string& user,
                              const std::string& key,
                              uint64_t token)
{
    StateRequest req;
    req.user = user;
    req.key = key;
    req.data.resize(sizeof(token));
    std::memcpy(req.data.data(), &token, sizeof(token));
    std::string resp =
      syncSend((uint8_t)StateCalls::Unlock, req.encode());
    return !resp.empty() && resp[0] == 1;
}

void StateClient::append(const std::string& user,
                         const std::string& key,
                         const uint8_t* data,
                         size_t len)
{
    StateRequest req;
    req.user = user;
    req.key = key;
    req.data.assign(data, data + len);
    syncSend((uint8_t)StateCalls::Append, req.encode());
}

std::vector<std::vector<uint8_t>> StateClient::pullAppended(
  const std::string& user,
  const std::string& key,
  size_t nValues)
{
    StateAppendedRequest req;
    req.user = user;
    req.key = key;
    req.nValues = (uint32_t)nValues;
    std::string resp =
      syncSend((uint8_t)StateCalls::PullAppended, req.encode());
    return StateAppendedResponse::decode(resp).values;
}

void StateClient::clearAppended(const std::string& user,
                                const std::string& key)
{
    StateRequest req;
    req.user = user;
    req.key = key;
    syncSend((uint8_t)StateCalls::ClearAppended, req.encode());
}

void StateClient::deleteKV(const std::string& user, const std::string& key)
{
    StateRequest req;
    req.user = user;
    req.key = key;
    syncSend((uint8_t)StateCalls::Delete, req.encode());
}

static std::mutex stClientsMx;
static std::map<std::string, std::shared_ptr<StateClient>> stClients;

std::shared_ptr<StateClient> getStateClient(const std::string& host)
{
    std::lock_guard<std::mutex> lock(stClientsMx);
    auto& cli = stClients[host];
    if (!cli) {
        cli = std::make_shared<StateClient>(host);
    }
    return cli;
}

void clearStateClients()
{
    std::lock_guard<std::mutex> lock(stClientsMx);
    stClients.clear();
}

} // namespace faabricamd
