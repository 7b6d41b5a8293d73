// Device snapshot engine host-side wrappers (kernels:
// cpp/hip/snapshot_kernels.hip). Fails loudly when no GPU is present —
// the CPU path lives in snapshot.cpp, never silently substituted here.
#include "faabricamd/ops.h"
#include "faabricamd/util.h"

#include <cstring>

namespace faabricamd {

#define OPS_HIP_CHECK(call)                                                    \
    do {                                                                       \
        hipError_t err_ = (call);                                              \
        if (err_ != hipSuccess) {                                              \
            throw FaabricException(std::string("HIP error in ops: ") +         \
                                   hipGetErrorString(err_));                   \
        }                                                                      \
    } while (0)

bool gpuAvailable()
{
    return gpuCount() > 0;
}

int gpuCount()
{
    static int n = []() {
        int count = 0;
        if (hipGetDeviceCount(&count) != hipSuccess) {
            return 0;
        }
        return count;
    }();
    return n;
}

DeviceSnapshot::DeviceSnapshot(size_t bytes, int device)
  : bytes_(bytes)
  , device_(device)
{
    if (device_ < 0) {
        device_ = getSystemConfig().gpuDevice;
    }
    if ((bytes % DEVICE_PAGE) != 0) {
        throw FaabricException("device snapshot size must be page-aligned");
    }
    if (!gpuAvailable()) {
        throw FaabricException(
          "DeviceSnapshot requires an MI355X GPU (none visible)");
    }
    OPS_HIP_CHECK(hipSetDevice(device_));
    OPS_HIP_CHECK(hipMalloc(&snap_, bytes_));
    OPS_HIP_CHECK(hipStreamCreateWithFlags(&stream_, hipStreamNonBlocking));
}

DeviceSnapshot::~DeviceSnapshot()
{
    if (snap_ != nullptr) {
        hipFree(snap_);
    }
    if (ticket_ != nullptr) {
        hipFree(ticket_);
    }
    if (pageIdx_ != nullptr) {
        hipFree(pageIdx_);
    }
    if (payload_ != nullptr) {
        hipFree(payload_);
    }
    if (bitmap_ != nullptr) {
        hipFree(bitmap_);
    }
    if (stream_ != nullptr) {
        hipStreamDestroy(stream_);
    }
}

void DeviceSnapshot::copyInHost(const void* hostBuf, size_t n, size_t offset)
{
    OPS_HIP_CHECK(hipSetDevice(device_));
    OPS_HIP_CHECK(hipMemcpyAsync(
      snap_ + offset, hostBuf, n, hipMemcpyHostToDevice, stream_));
    OPS_HIP_CHECK(hipStreamSynchronize(stream_));
}

void DeviceSnapshot::copyOutHost(void* hostBuf, size_t n, size_t offset) const
{
    OPS_HIP_CHECK(hipSetDevice(device_));
    OPS_HIP_CHECK(hipMemcpyAsync(
      hostBuf, snap_ + offset, n, hipMemcpyDeviceToHost, stream_));
    OPS_HIP_CHECK(hipStreamSynchronize(stream_));
}

void DeviceSnapshot::captureFromDevice(const void* devPtr)
{
    OPS_HIP_CHECK(hipSetDevice(device_));
    OPS_HIP_CHECK(hipMemcpyAsync(
      snap_, devPtr, bytes_, hipMemcpyDeviceToDevice, stream_));
    OPS_HIP_CHECK(hipStreamSynchronize(stream_));
}

void DeviceSnapshot::ensureDiffBuffers()
{
    if (ticket_ != nullptr) {
        return;
    }
    size_t nPages = bytes_ / DEVICE_PAGE;
    OPS_HIP_CHECK(hipMalloc(&ticket_, sizeof(uint32_t)));
    OPS_HIP_CHECK(hipMalloc(&pageIdx_, nPages * sizeof(uint32_t)));
    OPS_HIP_CHECK(hipMalloc(&payload_, bytes_));
    OPS_HIP_CHECK(hipMalloc(&bitmap_, ((nPages + 31) / 32) * sizeof(uint32_t)));
}

std::vector<uint32_t> DeviceSnapshot::dirtyPages(const void* devPtr)
{
    OPS_HIP_CHECK(hipSetDevice(device_));
    size_t nPages = bytes_ / DEVICE_PAGE;
    uint32_t* flagsDev = nullptr;
    OPS_HIP_CHECK(hipMalloc(&flagsDev, nPages * sizeof(uint32_t)));
    OPS_HIP_CHECK(
      hipMemsetAsync(flagsDev, 0, nPages * sizeof(uint32_t), stream_));
    OPS_HIP_CHECK(famDirtyPages(snap_, devPtr, bytes_, flagsDev, stream_));
    std::vector<uint32_t> flags(nPages);
    OPS_HIP_CHECK(hipMemcpyAsync(flags.data(),
                                 flagsDev,
                                 nPages * sizeof(uint32_t),
                                 hipMemcpyDeviceToHost,
                                 stream_));
    OPS_HIP_CHECK(hipStreamSynchronize(stream_));
    hipFree(flagsDev);
    return flags;
}

uint32_t DeviceSnapshot::diffXor(const void* devPtr)
{
    OPS_HIP_CHECK(hipSetDevice(device_));
    ensureDiffBuffers();
    size_t nWords = (bytes_ / DEVICE_PAGE + 31) / 32;
    OPS_HIP_CHECK(hipMemsetAsync(ticket_, 0, sizeof(uint32_t), stream_));
    OPS_HIP_CHECK(
      hipMemsetAsync(bitmap_, 0, nWords * sizeof(uint32_t), stream_));
    OPS_HIP_CHECK(famDiffXorPages(snap_, devPtr, bytes_, ticket_, pageIdx_,
                                  payload_, bitmap_, stream_));
    uint32_t nDirty = 0;
    OPS_HIP_CHECK(hipMemcpyAsync(&nDirty,
                                 ticket_,
                                 sizeof(uint32_t),
                                 hipMemcpyDeviceToHost,
                                 stream_));
    OPS_HIP_CHECK(hipStreamSynchronize(stream_));
    lastDirty_ = nDirty;
    return nDirty;
}

void DeviceSnapshot::applyLastDiff()
{
    applyDiffPages(pageIdx_, payload_, lastDirty_);
}

void DeviceSnapshot::applyDiffPages(const uint32_t* pageIdxDev,
                                    const void* payloadDev,
                                    uint32_t nDirty)
{
    if (nDirty == 0) {
        return;
    }
    OPS_HIP_CHECK(hipSetDevice(device_));
    OPS_HIP_CHECK(
      famApplyXorPages(snap_, pageIdxDev, payloadDev, nDirty, stream_));
    OPS_HIP_CHECK(hipStreamSynchronize(stream_));
}

void DeviceSnapshot::gatherLastDiffToHost(std::vector<uint32_t>& pagesOut,
                                          std::vector<uint8_t>& payloadOut)
{
    OPS_HIP_CHECK(hipSetDevice(device_));
    pagesOut.resize(lastDirty_);
    payloadOut.resize((size_t)lastDirty_ * DEVICE_PAGE);
    if (lastDirty_ == 0) {
        return;
    }
    // Gather on-device, then one contiguous D2H copy
    uint8_t* compactDev = nullptr;
    OPS_HIP_CHECK(
      hipMalloc(&compactDev, (size_t)lastDirty_ * DEVICE_PAGE));
    OPS_HIP_CHECK(
      famGatherPages(payload_, pageIdx_, compactDev, lastDirty_, stream_));
    OPS_HIP_CHECK(hipMemcpyAsync(pagesOut.data(),
                                 pageIdx_,
                                 lastDirty_ * sizeof(uint32_t),
                                 hipMemcpyDeviceToHost,
                                 stream_));
    OPS_HIP_CHECK(hipMemcpyAsync(payloadOut.data(),
                                 compactDev,
                                 (size_t)lastDirty_ * DEVICE_PAGE,
                                 hipMemcpyDeviceToHost,
                                 stream_));
    OPS_HIP_CHECK(hipStreamSynchronize(stream_));
    hipFree(compactDev);
}

void DeviceSnapshot::applyCompactDiffFromHost(
  const std::vector<uint32_t>& pages,
  const uint8_t* payload,
  size_t payloadBytes)
{
    if (pages.empty()) {
        return;
    }
    if (payloadBytes != pages.size() * DEVICE_PAGE) {
        throw FaabricException("compact diff size mismatch");
    }
    // Page indices come off the wire; an out-of-range index would make
    // the XOR kernel write past the snapshot allocation in HBM.
    const size_t nPages = (bytes_ + DEVICE_PAGE - 1) / DEVICE_PAGE;
    for (uint32_t p : pages) {
        if ((size_t)p >= nPages) {
            throw FaabricException("compact diff page out of range");
        }
    }
    OPS_HIP_CHECK(hipSetDevice(device_));
    uint32_t* pagesDev = nullptr;
    uint8_t* payloadDev = nullptr;
    OPS_HIP_CHECK(hipMalloc(&pagesDev, pages.size() * sizeof(uint32_t)));
    OPS_HIP_CHECK(hipMalloc(&payloadDev, payloadBytes));
    OPS_HIP_CHECK(hipMemcpyAsync(pagesDev,
                                 pages.data(),
                                 pages.size() * sizeof(uint32_t),
                                 hipMemcpyHostToDevice,
                                 stream_));
    OPS_HIP_CHECK(hipMemcpyAsync(payloadDev,
                                 payload,
                                 payloadBytes,
                                 hipMemcpyHostToDevice,
                                 stream_));
    OPS_HIP_CHECK(famApplyXorPagesEx(snap_,
                                     pagesDev,
                                     payloadDev,
                                     (uint32_t)pages.size(),
                                     /*compact=*/1,
                                     stream_));
    OPS_HIP_CHECK(hipStreamSynchronize(stream_));
    hipFree(pagesDev);
    hipFree(payloadDev);
}

void DeviceSnapshot::queuePackedDiff(std::vector<uint8_t> packed)
{
    std::lock_guard<std::mutex> lock(queueMx_);
    queuedPacked_.push_back(std::move(packed));
}

int DeviceSnapshot::applyQueuedPackedDiffs()
{
    std::vector<std::vector<uint8_t>> toApply;
    {
        std::lock_guard<std::mutex> lock(queueMx_);
        toApply.swap(queuedPacked_);
    }
    for (const auto& packed : toApply) {
        if (packed.size() < 4) {
            continue;
        }
        uint32_t n = 0;
        std::memcpy(&n, packed.data(), 4);
        size_t headerBytes = 4 + (size_t)n * 4;
        if (packed.size() < headerBytes + (size_t)n * DEVICE_PAGE) {
            throw FaabricException("bad packed diff");
        }
        std::vector<uint32_t> pages(n);
        std::memcpy(pages.data(), packed.data() + 4, (size_t)n * 4);
        applyCompactDiffFromHost(pages,
                                 packed.data() + headerBytes,
                                 (size_t)n * DEVICE_PAGE);
    }
    return (int)toApply.size();
}

DeviceSnapshotRegistry& DeviceSnapshotRegistry::get()
{
    static DeviceSnapshotRegistry reg;
    return reg;
}

std::shared_ptr<DeviceSnapshot> DeviceSnapshotRegistry::getSnapshot(
  const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    auto it = snapshots.find(key);
    if (it == snapshots.end()) {
        throw FaabricException("device snapshot not found: " + key);
    }
    return it->second;
}

bool DeviceSnapshotRegistry::snapshotExists(const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    return snapshots.count(key) > 0;
}

void DeviceSnapshotRegistry::registerSnapshot(
  const std::string& key,
  std::shared_ptr<DeviceSnapshot> snap)
{
    std::lock_guard<std::mutex> lock(mx);
    snapshots[key] = std::move(snap);
}

void DeviceSnapshotRegistry::deleteSnapshot(const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    snapshots.erase(key);
}

void DeviceSnapshotRegistry::clear()
{
    std::lock_guard<std::mutex> lock(mx);
    snapshots.clear();
}

void deviceElementwiseOp(void* inout,
                         const void* in,
                         uint64_t count,
                         int dtype,
                         int op,
                         hipStream_t stream)
{
    OPS_HIP_CHECK(famElementwiseOp(inout, in, count, dtype, op, stream));
    OPS_HIP_CHECK(hipStreamSynchronize(stream));
}

} // namespace faabricamd
