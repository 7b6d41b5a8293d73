// Device ops: HBM-resident snapshot engine on the gfx950 kernels in
// cpp/hip/snapshot_kernels.hip (dirty-page tracking, compacted XOR diff,
// merge application, elementwise reductions).
//
// This is the MI355X-native replacement for the reference's fault-driven
// dirty tracking + CPU diff loops (reference: src/util/dirty.cpp,
// src/util/snapshot.cpp:30-652): there is no mprotect on HBM, so dirty
// pages come from a compare kernel against the snapshot baseline, and the
// XOR page diff IS the shippable delta (DIFFING_MODE=xor made native).
#pragma once

#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <vector>

#include <hip/hip_runtime.h>

namespace faabricamd {

bool gpuAvailable();
int gpuCount();

// Raw kernel entry points (extern "C" in the .hip TU)
extern "C" {
hipError_t famDirtyPages(const void* snap,
                         const void* cur,
                         uint64_t bytes,
                         uint32_t* flagsDev,
                         hipStream_t stream);
// NT device-to-device copy (beats hipMemcpy D2D for large HBM buffers)
hipError_t famCopyBuffer(const void* src,
                         void* dst,
                         uint64_t bytes,
                         hipStream_t stream);
// Deterministic pseudo-random fill / scattered page mutation (bench
// init for the random-byte snapshot shapes, BASELINE.json config 4)
hipError_t famFillRandom(void* buf,
                         uint64_t bytes,
                         uint64_t seed,
                         hipStream_t stream);
hipError_t famTouchPages(void* buf,
                         const uint32_t* pagesDev,
                         uint32_t nPages,
                         uint64_t seed,
                         hipStream_t stream);

hipError_t famXorBuffer(const void* a,
                        const void* b,
                        void* out,
                        uint64_t bytes,
                        hipStream_t stream);
hipError_t famDiffXorPages(const void* snap,
                           const void* cur,
                           uint64_t bytes,
                           uint32_t* ticketDev,
                           uint32_t* pageIdxDev,
                           void* payloadDev,
                           uint32_t* bitmapDev,
                           hipStream_t stream);
hipError_t famApplyXorPages(void* snap,
                            const uint32_t* pageIdxDev,
                            const void* payloadDev,
                            uint32_t nDirty,
                            hipStream_t stream);
hipError_t famApplyXorPagesEx(void* snap,
                              const uint32_t* pageIdxDev,
                              const void* payloadDev,
                              uint32_t nDirty,
                              int compact,
                              hipStream_t stream);
hipError_t famGatherPages(const void* payloadDev,
                          const uint32_t* pageIdxDev,
                          void* outDev,
                          uint32_t nDirty,
                          hipStream_t stream);
hipError_t famElementwiseOp(void* inout,
                            const void* in,
                            uint64_t count,
                            int dtype,
                            int op,
                            hipStream_t stream);
}

inline constexpr size_t DEVICE_PAGE = 4096;

// An HBM-resident snapshot of a device memory region, with the
// diff/merge pipeline. All methods are synchronous on the owned stream.
class DeviceSnapshot
{
  public:
    // device -1 = this worker's configured GPU (FAABRIC_GPU_DEVICE)
    DeviceSnapshot(size_t bytes, int device = -1);
    ~DeviceSnapshot();
    DeviceSnapshot(const DeviceSnapshot&) = delete;

    size_t size() const { return bytes_; }
    int device() const { return device_; }
    void* data() { return snap_; }

    // Baseline management
    void copyInHost(const void* hostBuf, size_t n, size_t offset = 0);
    void copyOutHost(void* hostBuf, size_t n, size_t offset = 0) const;
    void captureFromDevice(const void* devPtr); // snapshot := memory view

    // Dirty-page bitmap of devPtr vs the snapshot (one u32 flag per 4 KiB
    // page, copied back to the host)
    std::vector<uint32_t> dirtyPages(const void* devPtr);

    // Compacted XOR page diff of devPtr vs the snapshot. Returns the
    // number of dirty pages; the page indices + payload stay on-device
    // (diffPageIdx()/diffPayload()) ready to ship or apply.
    uint32_t diffXor(const void* devPtr);

    // Merge: snapshot ^= last diff (applies the compacted pages)
    void applyLastDiff();
    // Merge a diff produced elsewhere (e.g. shipped from a peer GPU)
    void applyDiffPages(const uint32_t* pageIdxDev,
                        const void* payloadDev,
                        uint32_t nDirty);

    // Ship/receive the compact wire form: {page indices, slot-compacted
    // 4 KiB payloads}
    void gatherLastDiffToHost(std::vector<uint32_t>& pagesOut,
                              std::vector<uint8_t>& payloadOut);
    void applyCompactDiffFromHost(const std::vector<uint32_t>& pages,
                                  const uint8_t* payload,
                                  size_t payloadBytes);

    // Queued packed diffs (wire form: [u32 n][u32 pages[n]][n*4KiB]) from
    // remote/other-thread merges, applied at join time
    void queuePackedDiff(std::vector<uint8_t> packed);
    int applyQueuedPackedDiffs();

    const uint32_t* diffPageIdx() const { return pageIdx_; }
    const void* diffPayload() const { return payload_; }
    uint32_t lastDiffPages() const { return lastDirty_; }
    hipStream_t stream() const { return stream_; }

  private:
    void ensureDiffBuffers();

    size_t bytes_ = 0;
    int device_ = 0;
    uint8_t* snap_ = nullptr;
    hipStream_t stream_ = nullptr;

    // Diff scratch (lazily allocated): worst case every page dirty.
    // The payload buffer is SPARSE (indexed by page); pageIdx_ is the
    // compacted dirty list.
    uint32_t* ticket_ = nullptr;
    uint32_t* pageIdx_ = nullptr;
    uint8_t* payload_ = nullptr;
    uint32_t* bitmap_ = nullptr;
    uint32_t lastDirty_ = 0;

    std::mutex queueMx_;
    std::vector<std::vector<uint8_t>> queuedPacked_;
};

// Registry of HBM-resident snapshots (device analog of SnapshotRegistry)
class DeviceSnapshotRegistry
{
  public:
    static DeviceSnapshotRegistry& get();
    std::shared_ptr<DeviceSnapshot> getSnapshot(const std::string& key);
    bool snapshotExists(const std::string& key);
    void registerSnapshot(const std::string& key,
                          std::shared_ptr<DeviceSnapshot> snap);
    void deleteSnapshot(const std::string& key);
    void clear();

  private:
    std::mutex mx;
    std::map<std::string, std::shared_ptr<DeviceSnapshot>> snapshots;
};

// Elementwise op on device buffers: inout = op(inout, in)
// dtype matches MpiDataType; op matches MpiOp (+4=sub, 5=xor)
void deviceElementwiseOp(void* inout,
                         const void* in,
                         uint64_t count,
                         int dtype,
                         int op,
                         hipStream_t stream = nullptr);

} // namespace faabricamd
