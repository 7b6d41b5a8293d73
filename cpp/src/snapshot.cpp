// Snapshot subsystem: host-path implementation. Typed merge semantics match
// the reference bit-for-bit (reference: util/snapshot.h:163-246
// calculateDiffValue/applyDiffValue, src/util/snapshot.cpp:30-652). The
// device (HBM) path dispatches the same operations to gfx950 HIP kernels —
// see cpp/hip/snapshot_kernels.hip and ops.cpp.
#include "faabricamd/snapshot.h"
#include "faabricamd/flat.h"
#include "faabricamd/hipipc.h"
#include "faabricamd/ops.h"
#include "faabricamd/planner.h"
#include "faabricamd/ptp.h"
#include "faabricamd/util.h"

#include <algorithm>
#include <cstring>

namespace faabricamd {

SnapshotDiffMsg SnapshotDiff::toMsg() const
{
    SnapshotDiffMsg msg;
    msg.offset = (int32_t)offset;
    msg.dataType = (int32_t)dataType;
    msg.mergeOp = (int32_t)operation;
    msg.data = dataCopy;
    return msg;
}

SnapshotDiff SnapshotDiff::fromMsg(const SnapshotDiffMsg& msg)
{
    SnapshotDiff d;
    d.offset = (uint32_t)msg.offset;
    d.dataType = (SnapshotDataType)msg.dataType;
    d.operation = (SnapshotMergeOperation)msg.mergeOp;
    d.dataCopy = msg.data;
    return d;
}

// ------------------------- byte-level diff ----------------------------------

void diffArrayRegions(std::vector<SnapshotDiff>& diffs,
                      uint32_t startOffset,
                      uint32_t endOffset,
                      const uint8_t* original,
                      const uint8_t* updated)
{
    // Two-pass: compare DIFF_CHUNK_SIZE chunks with memcmp, refine
    // byte-wise on mismatch, merging adjacent dirty bytes into one diff
    // (reference: src/util/snapshot.cpp:30-100)
    uint32_t diffStart = 0;
    bool inDiff = false;

    uint32_t pos = startOffset;
    while (pos < endOffset) {
        uint32_t chunkEnd =
          std::min<uint32_t>(pos + (uint32_t)DIFF_CHUNK_SIZE, endOffset);
        if (std::memcmp(original + pos, updated + pos, chunkEnd - pos) == 0) {
            if (inDiff) {
                diffs.emplace_back(SnapshotDataType::Raw,
                                   SnapshotMergeOperation::Bytewise,
                                   diffStart,
                                   updated + diffStart,
                                   pos - diffStart);
                inDiff = false;
            }
            pos = chunkEnd;
            continue;
        }
        // Byte-wise refinement inside the mismatching chunk
        for (uint32_t b = pos; b < chunkEnd; b++) {
            bool dirty = original[b] != updated[b];
            if (dirty && !inDiff) {
                diffStart = b;
                inDiff = true;
            } else if (!dirty && inDiff) {
                diffs.emplace_back(SnapshotDataType::Raw,
                                   SnapshotMergeOperation::Bytewise,
                                   diffStart,
                                   updated + diffStart,
                                   b - diffStart);
                inDiff = false;
            }
        }
        pos = chunkEnd;
    }
    if (inDiff) {
        diffs.emplace_back(SnapshotDataType::Raw,
                           SnapshotMergeOperation::Bytewise,
                           diffStart,
                           updated + diffStart,
                           endOffset - diffStart);
    }
}

// ------------------------- typed diff helpers -------------------------------

template<typename T>
static bool calcDiffValue(const uint8_t* originalPtr,
                          const uint8_t* updatedPtr,
                          SnapshotMergeOperation op,
                          T& out)
{
    // Sum/Subtract/Product transmit the DELTA, Max/Min the VALUE
    // (reference: util/snapshot.h:163-210)
    T original;
    T updated;
    std::memcpy(&original, originalPtr, sizeof(T));
    std::memcpy(&updated, updatedPtr, sizeof(T));
    if (original == updated) {
        return false;
    }
    switch (op) {
        case SnapshotMergeOperation::Sum:
            out = updated - original;
            break;
        case SnapshotMergeOperation::Subtract:
            out = original - updated;
            break;
        case SnapshotMergeOperation::Product:
            if (original == (T)0) {
                throw FaabricException("product diff with zero original");
            }
            out = updated / original;
            break;
        case SnapshotMergeOperation::Max:
        case SnapshotMergeOperation::Min:
            out = updated;
            break;
        default:
            throw FaabricException("unsupported typed diff operation");
    }
    return true;
}

template<typename T>
static void applyDiffValue(uint8_t* targetPtr,
                           const uint8_t* diffPtr,
                           SnapshotMergeOperation op)
{
    T target;
    T value;
    std::memcpy(&target, targetPtr, sizeof(T));
    std::memcpy(&value, diffPtr, sizeof(T));
    switch (op) {
        case SnapshotMergeOperation::Sum:
            target += value;
            break;
        case SnapshotMergeOperation::Subtract:
            target -= value;
            break;
        case SnapshotMergeOperation::Product:
            target *= value;
            break;
        case SnapshotMergeOperation::Max:
            target = std::max(target, value);
            break;
        case SnapshotMergeOperation::Min:
            target = std::min(target, value);
            break;
        default:
            throw FaabricException("unsupported typed apply operation");
    }
    std::memcpy(targetPtr, &target, sizeof(T));
}

template<typename T>
static void addTypedDiffs(std::vector<SnapshotDiff>& diffs,
                          SnapshotDataType dataType,
                          SnapshotMergeOperation op,
                          uint32_t regionStart,
                          uint32_t regionEnd,
                          const uint8_t* original,
                          const uint8_t* updated,
                          const std::vector<char>& dirtyPages)
{
    for (uint32_t off = regionStart; off + sizeof(T) <= regionEnd;
         off += sizeof(T)) {
        size_t page = off / SNAPSHOT_PAGE_SIZE;
        if (page < dirtyPages.size() && dirtyPages[page] == 0) {
            // Skip elements on clean pages (fast path; elements spanning a
            // page boundary are conservatively checked)
            size_t lastPage = (off + sizeof(T) - 1) / SNAPSHOT_PAGE_SIZE;
            if (lastPage < dirtyPages.size() && dirtyPages[lastPage] == 0) {
                continue;
            }
        }
        T value;
        if (calcDiffValue<T>(original + off, updated + off, op, value)) {
            diffs.emplace_back(dataType,
                               op,
                               off,
                               reinterpret_cast<const uint8_t*>(&value),
                               sizeof(T));
        }
    }
}

void SnapshotMergeRegion::addDiffs(std::vector<SnapshotDiff>& diffs,
                                   const uint8_t* original,
                                   size_t originalSize,
                                   const uint8_t* updated,
                                   size_t updatedSize,
                                   const std::vector<char>& dirtyPages) const
{
    uint32_t regionEnd =
      length == 0 ? (uint32_t)updatedSize
                  : std::min<uint32_t>(offset + (uint32_t)length,
                                       (uint32_t)updatedSize);
    if (offset >= regionEnd) {
        return;
    }
    uint32_t cmpEnd = std::min<uint32_t>(regionEnd, (uint32_t)originalSize);

    switch (operation) {
        case SnapshotMergeOperation::Bytewise: {
            // Per dirty page inside the region, chunked byte-diff
            uint32_t firstPage = offset / SNAPSHOT_PAGE_SIZE;
            uint32_t lastPage =
              (cmpEnd + SNAPSHOT_PAGE_SIZE - 1) / SNAPSHOT_PAGE_SIZE;
            for (uint32_t p = firstPage; p < lastPage; p++) {
                if (p < dirtyPages.size() && dirtyPages[p] == 0) {
                    continue;
                }
                uint32_t start =
                  std::max<uint32_t>(offset, p * SNAPSHOT_PAGE_SIZE);
                uint32_t end = std::min<uint32_t>(
                  cmpEnd, (p + 1) * SNAPSHOT_PAGE_SIZE);
                if (start < end) {
                    diffArrayRegions(diffs, start, end, original, updated);
                }
            }
            break;
        }
        case SnapshotMergeOperation::XOR: {
            // Whole dirty pages, payload = updated ^ original
            uint32_t firstPage = offset / SNAPSHOT_PAGE_SIZE;
            uint32_t lastPage =
              (cmpEnd + SNAPSHOT_PAGE_SIZE - 1) / SNAPSHOT_PAGE_SIZE;
            for (uint32_t p = firstPage; p < lastPage; p++) {
                if (p < dirtyPages.size() && dirtyPages[p] == 0) {
                    continue;
                }
                uint32_t start =
                  std::max<uint32_t>(offset, p * SNAPSHOT_PAGE_SIZE);
                uint32_t end = std::min<uint32_t>(
                  cmpEnd, (p + 1) * SNAPSHOT_PAGE_SIZE);
                if (start >= end) {
                    continue;
                }
                if (std::memcmp(original + start, updated + start,
                                end - start) == 0) {
                    continue;
                }
                std::vector<uint8_t> payload(end - start);
                for (uint32_t b = 0; b < end - start; b++) {
                    payload[b] = original[start + b] ^ updated[start + b];
                }
                diffs.emplace_back(SnapshotDataType::Raw,
                                   SnapshotMergeOperation::XOR,
                                   start,
                                   payload.data(),
                                   payload.size());
            }
            break;
        }
        default: {
            switch (dataType) {
                case SnapshotDataType::Int:
                    addTypedDiffs<int32_t>(diffs,
                                           dataType,
                                           operation,
                                           offset,
                                           cmpEnd,
                                           original,
                                           updated,
                                           dirtyPages);
                    break;
                case SnapshotDataType::Long:
                    addTypedDiffs<int64_t>(diffs,
                                           dataType,
                                           operation,
                                           offset,
                                           cmpEnd,
                                           original,
                                           updated,
                                           dirtyPages);
                    break;
                case SnapshotDataType::Float:
                    addTypedDiffs<float>(diffs,
                                         dataType,
                                         operation,
                                         offset,
                                         cmpEnd,
                                         original,
                                         updated,
                                         dirtyPages);
                    break;
                case SnapshotDataType::Double:
                    addTypedDiffs<double>(diffs,
                                          dataType,
                                          operation,
                                          offset,
                                          cmpEnd,
                                          original,
                                          updated,
                                          dirtyPages);
                    break;
                default:
                    throw FaabricException(
                      "unsupported merge data type for typed op");
            }
        }
    }
}

// ------------------------- SnapshotData -------------------------------------

SnapshotData::SnapshotData(size_t sizeIn)
  : SnapshotData(sizeIn, sizeIn)
{}

SnapshotData::SnapshotData(size_t sizeIn, size_t maxSizeIn)
  : size_(sizeIn)
  , maxSize_(std::max(sizeIn, maxSizeIn))
{
    data_.resize(maxSize_, 0);
}

SnapshotData::SnapshotData(const std::vector<uint8_t>& dataIn)
  : SnapshotData(dataIn, dataIn.size())
{}

SnapshotData::SnapshotData(const std::vector<uint8_t>& dataIn,
                           size_t maxSizeIn)
  : size_(dataIn.size())
  , maxSize_(std::max(dataIn.size(), maxSizeIn))
{
    data_.resize(maxSize_, 0);
    std::memcpy(data_.data(), dataIn.data(), dataIn.size());
}

SnapshotData::~SnapshotData() = default;

uint8_t* SnapshotData::getMutableDataPtr(size_t offset)
{
    return data_.data() + offset;
}

const uint8_t* SnapshotData::getDataPtr(size_t offset) const
{
    return data_.data() + offset;
}

std::vector<uint8_t> SnapshotData::getDataCopy() const
{
    return getDataCopy(0, size_);
}

std::vector<uint8_t> SnapshotData::getDataCopy(uint32_t offset,
                                               size_t size) const
{
    if (offset + size > size_) {
        throw FaabricException("snapshot data copy out of bounds");
    }
    return { data_.begin() + offset, data_.begin() + offset + size };
}

void SnapshotData::copyInData(const std::vector<uint8_t>& buffer,
                              uint32_t offset)
{
    copyInData(buffer.data(), buffer.size(), offset);
}

void SnapshotData::copyInData(const uint8_t* buffer,
                              size_t size,
                              uint32_t offset)
{
    if (offset + size > maxSize_) {
        throw FaabricException("copyInData exceeds snapshot max size");
    }
    std::memcpy(data_.data() + offset, buffer, size);
    size_ = std::max(size_, (size_t)offset + size);
}

void SnapshotData::mapToMemory(uint8_t* target, size_t targetSize) const
{
    size_t n = std::min(targetSize, size_);
    std::memcpy(target, data_.data(), n);
}

void SnapshotData::setSnapshotSize(size_t newSize)
{
    if (newSize > maxSize_) {
        maxSize_ = newSize;
        data_.resize(maxSize_, 0);
    }
    size_ = newSize;
}

void SnapshotData::addMergeRegion(uint32_t offset,
                                  size_t length,
                                  SnapshotDataType dataType,
                                  SnapshotMergeOperation operation)
{
    std::lock_guard<std::mutex> lock(snapMx);
    mergeRegions[offset] =
      SnapshotMergeRegion(offset, length, dataType, operation);
}

void SnapshotData::fillGapsWithBytewiseRegions()
{
    // Cover unannotated ranges (reference: src/util/snapshot.cpp:259)
    std::lock_guard<std::mutex> lock(snapMx);
    const auto& conf = getSystemConfig();
    SnapshotMergeOperation fillOp = conf.diffingMode == "xor"
                                      ? SnapshotMergeOperation::XOR
                                      : SnapshotMergeOperation::Bytewise;
    uint32_t pos = 0;
    std::vector<SnapshotMergeRegion> gaps;
    for (const auto& [off, region] : mergeRegions) {
        if (off > pos) {
            gaps.emplace_back(pos, off - pos, SnapshotDataType::Raw, fillOp);
        }
        pos = std::max<uint32_t>(
          pos,
          region.length == 0 ? (uint32_t)size_
                             : off + (uint32_t)region.length);
    }
    if (pos < size_) {
        gaps.emplace_back(pos, 0, SnapshotDataType::Raw, fillOp);
    }
    for (auto& g : gaps) {
        mergeRegions[g.offset] = g;
    }
}

void SnapshotData::clearMergeRegions()
{
    std::lock_guard<std::mutex> lock(snapMx);
    mergeRegions.clear();
}

std::map<uint32_t, SnapshotMergeRegion> SnapshotData::getMergeRegions()
{
    std::lock_guard<std::mutex> lock(snapMx);
    return mergeRegions;
}

std::vector<SnapshotDiff> SnapshotData::diffWithDirtyRegions(
  const uint8_t* updated,
  size_t updatedSize,
  const std::vector<char>& dirtyPages)
{
    // (reference: src/util/snapshot.cpp:524 diffWithDirtyRegions)
    std::vector<SnapshotDiff> diffs;

    // Extension: memory grown past the snapshot ships bytewise
    if (updatedSize > size_) {
        diffs.emplace_back(SnapshotDataType::Raw,
                           SnapshotMergeOperation::Bytewise,
                           (uint32_t)size_,
                           updated + size_,
                           updatedSize - size_);
    }

    bool anyDirty =
      std::any_of(dirtyPages.begin(), dirtyPages.end(), [](char c) {
          return c != 0;
      });
    if (!anyDirty) {
        return diffs;
    }

    std::lock_guard<std::mutex> lock(snapMx);
    for (const auto& [off, region] : mergeRegions) {
        region.addDiffs(
          diffs, data_.data(), size_, updated, updatedSize, dirtyPages);
    }
    return diffs;
}

std::vector<SnapshotDiff> SnapshotData::diffWithMemory(const uint8_t* updated,
                                                       size_t updatedSize)
{
    size_t nPages =
      (std::max(updatedSize, size_) + SNAPSHOT_PAGE_SIZE - 1) /
      SNAPSHOT_PAGE_SIZE;
    std::vector<char> allDirty(nPages, 1);
    return diffWithDirtyRegions(updated, updatedSize, allDirty);
}

void SnapshotData::applyDiff(const SnapshotDiff& diff)
{
    if (diff.operation != SnapshotMergeOperation::XorPages &&
        diff.offset + diff.size() > maxSize_) {
        throw FaabricException("diff out of snapshot bounds");
    }
    if (diff.operation != SnapshotMergeOperation::XorPages) {
        size_ = std::max(size_, (size_t)diff.offset + diff.size());
    }
    uint8_t* target = data_.data() + diff.offset;

    switch (diff.operation) {
        case SnapshotMergeOperation::Bytewise:
            std::memcpy(target, diff.getData(), diff.size());
            break;
        case SnapshotMergeOperation::XOR: {
            const uint8_t* src = diff.getData();
            for (size_t i = 0; i < diff.size(); i++) {
                target[i] ^= src[i];
            }
            break;
        }
        case SnapshotMergeOperation::XorPages: {
            // Packed page form: [u32 n][u32 pages[n]][n x 4 KiB]
            const uint8_t* src = diff.getData();
            if (diff.size() < 4) {
                break;
            }
            uint32_t n = 0;
            std::memcpy(&n, src, 4);
            size_t header = 4 + (size_t)n * 4;
            if (diff.size() < header + (size_t)n * SNAPSHOT_PAGE_SIZE) {
                throw FaabricException("bad packed page diff");
            }
            for (uint32_t i = 0; i < n; i++) {
                uint32_t page = 0;
                std::memcpy(&page, src + 4 + (size_t)i * 4, 4);
                size_t off = (size_t)page * SNAPSHOT_PAGE_SIZE;
                // Page indices arrive off the wire (ThreadResult RPC);
                // an index past the snapshot would make maxSize_ - off
                // underflow into a heap OOB write. Reject the diff.
                if (off >= maxSize_) {
                    throw FaabricException("packed page diff out of range");
                }
                const uint8_t* payload =
                  src + header + (size_t)i * SNAPSHOT_PAGE_SIZE;
                size_t len =
                  std::min(SNAPSHOT_PAGE_SIZE, maxSize_ - off);
                size_ = std::max(size_, off + len);
                for (size_t b = 0; b < len; b++) {
                    data_[off + b] ^= payload[b];
                }
            }
            break;
        }
        default: {
            switch (diff.dataType) {
                case SnapshotDataType::Int:
                    applyDiffValue<int32_t>(target,
                                            diff.getData(),
                                            diff.operation);
                    break;
                case SnapshotDataType::Long:
                    applyDiffValue<int64_t>(target,
                                            diff.getData(),
                                            diff.operation);
                    break;
                case SnapshotDataType::Float:
                    applyDiffValue<float>(target,
                                          diff.getData(),
                                          diff.operation);
                    break;
                case SnapshotDataType::Double:
                    applyDiffValue<double>(target,
                                           diff.getData(),
                                           diff.operation);
                    break;
                default:
                    throw FaabricException(
                      "unsupported apply data type");
            }
        }
    }
}

void SnapshotData::applyDiffs(const std::vector<SnapshotDiff>& diffs)
{
    for (const auto& d : diffs) {
        applyDiff(d);
    }
}

void SnapshotData::queueDiffs(const std::vector<SnapshotDiff>& diffs)
{
    std::lock_guard<std::mutex> lock(snapMx);
    queuedDiffs.insert(queuedDiffs.end(), diffs.begin(), diffs.end());
}

int SnapshotData::writeQueuedDiffs()
{
    std::vector<SnapshotDiff> toApply;
    {
        std::lock_guard<std::mutex> lock(snapMx);
        toApply.swap(queuedDiffs);
    }
    applyDiffs(toApply);
    return (int)toApply.size();
}

// ------------------------- registry -----------------------------------------

SnapshotRegistry& SnapshotRegistry::get()
{
    static SnapshotRegistry reg;
    return reg;
}

std::shared_ptr<SnapshotData> SnapshotRegistry::getSnapshot(
  const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    auto it = snapshots.find(key);
    if (it == snapshots.end()) {
        throw FaabricException("snapshot not found: " + key);
    }
    return it->second;
}

bool SnapshotRegistry::snapshotExists(const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    return snapshots.count(key) > 0;
}

void SnapshotRegistry::registerSnapshot(const std::string& key,
                                        std::shared_ptr<SnapshotData> data)
{
    std::lock_guard<std::mutex> lock(mx);
    snapshots[key] = std::move(data);
}

void SnapshotRegistry::registerSnapshotIfNotExists(
  const std::string& key,
  std::shared_ptr<SnapshotData> data)
{
    std::lock_guard<std::mutex> lock(mx);
    if (snapshots.count(key) == 0) {
        snapshots[key] = std::move(data);
    }
}

void SnapshotRegistry::deleteSnapshot(const std::string& key)
{
    std::lock_guard<std::mutex> lock(mx);
    snapshots.erase(key);
}

size_t SnapshotRegistry::getSnapshotCount()
{
    std::lock_guard<std::mutex> lock(mx);
    return snapshots.size();
}

void SnapshotRegistry::clear()
{
    std::lock_guard<std::mutex> lock(mx);
    snapshots.clear();
}

// ------------------------- RPC ----------------------------------------------

SnapshotServer::SnapshotServer()
  : MessageEndpointServer(SNAPSHOT_ASYNC_PORT, SNAPSHOT_SYNC_PORT, "snapshot")
{}

void SnapshotServer::doAsyncRecv(uint8_t code,
                                 const std::string& body,
                                 uint32_t seq)
{
    (void)seq;
    if ((SnapshotCalls)code == SnapshotCalls::DeleteSnapshot) {
        // FlatBuffers body (faabric.fbs SnapshotDeleteRequest)
        auto req = FlatSnapshotDelete::decode(body);
        SnapshotRegistry::get().deleteSnapshot(req.key);
        // Device snapshots shipped here land in the device registry
        DeviceSnapshotRegistry::get().deleteSnapshot(req.key);
        return;
    }
    FAM_ERROR("snapshot server: bad async code %d", (int)code);
}

std::string SnapshotServer::doSyncRecv(uint8_t code, const std::string& body)
{
    switch ((SnapshotCalls)code) {
        case SnapshotCalls::PushSnapshot: {
            // FlatBuffers body (reference: faabric.fbs SnapshotPushRequest)
            auto req = FlatSnapshotPush::decode(body);
            auto snap = std::make_shared<SnapshotData>(
              req.contents, std::max<size_t>(req.maxSize,
                                             req.contents.size()));
            for (const auto& r : req.mergeRegions) {
                snap->addMergeRegion((uint32_t)r.offset,
                                     r.length,
                                     (SnapshotDataType)r.dataType,
                                     (SnapshotMergeOperation)r.mergeOp);
            }
            SnapshotRegistry::get().registerSnapshot(req.key, snap);
            return {};
        }
        case SnapshotCalls::PushSnapshotDevice: {
            // HBM-resident snapshot as host bytes (protobuf body — our
            // extension, kept off the flatbuffers surface)
            auto req = SnapshotPushRequest::decode(body);
            if (!gpuAvailable()) {
                throw FaabricException(
                  "device snapshot pushed to GPU-less host");
            }
            size_t rounded = (req.contents.size() + DEVICE_PAGE - 1) /
                             DEVICE_PAGE * DEVICE_PAGE;
            auto dsnap = std::make_shared<DeviceSnapshot>(rounded);
            dsnap->copyInHost(req.contents.data(), req.contents.size());
            DeviceSnapshotRegistry::get().registerSnapshot(req.key,
                                                           dsnap);
            return {};
        }
        case SnapshotCalls::PushSnapshotIpc: {
            // Streamed device snapshot: land the arena segment straight
            // in HBM (create the snapshot on the first chunk)
            auto c = IpcChunk::decode(body);
            if (!gpuAvailable()) {
                throw FaabricException(
                  "device snapshot pushed to GPU-less host");
            }
            auto& reg = DeviceSnapshotRegistry::get();
            std::shared_ptr<DeviceSnapshot> dsnap;
            if (reg.snapshotExists(c.key)) {
                dsnap = reg.getSnapshot(c.key);
                if (dsnap->size() <
                    (c.totalSize + DEVICE_PAGE - 1) / DEVICE_PAGE *
                      DEVICE_PAGE) {
                    dsnap.reset();
                }
            }
            if (!dsnap) {
                size_t rounded = (c.totalSize + DEVICE_PAGE - 1) /
                                 DEVICE_PAGE * DEVICE_PAGE;
                dsnap = std::make_shared<DeviceSnapshot>(rounded);
                reg.registerSnapshot(c.key, dsnap);
            }
            if (c.valOffset > dsnap->size() || c.len > dsnap->size() ||
                c.valOffset + c.len > dsnap->size()) {
                throw FaabricException("snapshot ipc chunk out of range");
            }
            IpcReceiver::get().copyToDevice(
              c.srcHost, c.ipcOffset,
              (uint8_t*)dsnap->data() + c.valOffset, c.len);
            getPointToPointBroker().sendIpcAck(c.srcHost, c.ipcOffset,
                                               c.len);
            return {};
        }
        case SnapshotCalls::PushSnapshotUpdate: {
            // FlatBuffers body (faabric.fbs SnapshotUpdateRequest)
            auto req = FlatSnapshotUpdate::decode(body);
            auto snap = SnapshotRegistry::get().getSnapshot(req.key);
            std::vector<SnapshotDiff> diffs;
            for (const auto& d : req.diffs) {
                SnapshotDiffMsg m;
                m.offset = d.offset;
                m.dataType = d.dataType;
                m.mergeOp = d.mergeOp;
                m.data = d.data;
                diffs.push_back(SnapshotDiff::fromMsg(m));
            }
            snap->applyDiffs(diffs);
            for (const auto& r : req.mergeRegions) {
                snap->addMergeRegion((uint32_t)r.offset,
                                     r.length,
                                     (SnapshotDataType)r.dataType,
                                     (SnapshotMergeOperation)r.mergeOp);
            }
            return {};
        }
        case SnapshotCalls::ThreadResult: {
            // FlatBuffers body (faabric.fbs ThreadResultRequest)
            auto flat = FlatThreadResult::decode(body);
            ThreadResultRequest req;
            req.appId = flat.appId;
            req.messageId = flat.messageId;
            req.returnValue = flat.returnValue;
            req.key = flat.key;
            req.executedHost = flat.executedHost;
            for (const auto& d : flat.diffs) {
                SnapshotDiffMsg m;
                m.offset = d.offset;
                m.dataType = d.dataType;
                m.mergeOp = d.mergeOp;
                m.data = d.data;
                req.diffs.push_back(m);
            }
            // Queue the diffs onto the main-thread snapshot, then forward
            // the thread's result (reference: SnapshotServer.cpp:28-62)
            if (!req.key.empty() && !req.diffs.empty()) {
                bool isDevice =
                  DeviceSnapshotRegistry::get().snapshotExists(req.key);
                if (isDevice) {
                    auto dsnap =
                      DeviceSnapshotRegistry::get().getSnapshot(req.key);
                    for (const auto& d : req.diffs) {
                        if ((SnapshotMergeOperation)d.mergeOp ==
                            SnapshotMergeOperation::XorPages) {
                            dsnap->queuePackedDiff(d.data);
                        }
                    }
                } else {
                    auto snap =
                      SnapshotRegistry::get().getSnapshot(req.key);
                    std::vector<SnapshotDiff> diffs;
                    for (const auto& d : req.diffs) {
                        diffs.push_back(SnapshotDiff::fromMsg(d));
                    }
                    snap->queueDiffs(diffs);
                }
            }
            auto msg = std::make_shared<Message>();
            msg->appId = req.appId;
            msg->id = req.messageId;
            msg->returnValue = req.returnValue;
            msg->executedHost = req.executedHost.empty()
                                  ? getSystemConfig().endpointHost
                                  : req.executedHost;
            getPlannerClient().setMessageResult(msg);
            return {};
        }
        default:
            throw FaabricException("snapshot server: bad sync code " +
                                   std::to_string(code));
    }
}

// Mock recording
static std::mutex snapMockMx;
static std::vector<std::pair<std::string, std::string>> mockedPushes;
static std::vector<std::pair<std::string, ThreadResultRequest>>
  mockedThreadResults;

std::vector<std::pair<std::string, std::string>> getSnapshotPushesMock()
{
    std::lock_guard<std::mutex> lock(snapMockMx);
    return mockedPushes;
}

std::vector<std::pair<std::string, ThreadResultRequest>>
getThreadResultsMock()
{
    std::lock_guard<std::mutex> lock(snapMockMx);
    return mockedThreadResults;
}

void clearMockedSnapshotRequests()
{
    std::lock_guard<std::mutex> lock(snapMockMx);
    mockedPushes.clear();
    mockedThreadResults.clear();
}

SnapshotClient::SnapshotClient(const std::string& host)
  : MessageEndpointClient(host, SNAPSHOT_ASYNC_PORT, SNAPSHOT_SYNC_PORT)
{}

void SnapshotClient::pushSnapshot(const std::string& key, SnapshotData& data)
{
    if (isMockMode()) {
        std::lock_guard<std::mutex> lock(snapMockMx);
        mockedPushes.emplace_back(getHost(), key);
        return;
    }
    // FlatBuffers wire format (reference: faabric.fbs SnapshotPushRequest,
    // SnapshotClient::pushSnapshot)
    FlatSnapshotPush req;
    req.key = key;
    req.maxSize = data.getMaxSize();
    req.contents = data.getDataCopy();
    for (const auto& [off, r] : data.getMergeRegions()) {
        FlatMergeRegion m;
        m.offset = (int32_t)r.offset;
        m.length = r.length;
        m.dataType = (int32_t)r.dataType;
        m.mergeOp = (int32_t)r.operation;
        req.mergeRegions.push_back(m);
    }
    syncSend((uint8_t)SnapshotCalls::PushSnapshot, req.encode());
}

void SnapshotClient::pushDeviceSnapshot(const std::string& key,
                                        const void* hostCopy,
                                        size_t size)
{
    if (isMockMode()) {
        std::lock_guard<std::mutex> lock(snapMockMx);
        mockedPushes.emplace_back(getHost(), key);
        return;
    }
    SnapshotPushRequest req;
    req.key = key;
    req.maxSize = size;
    req.onDevice = true;
    req.contents.assign((const uint8_t*)hostCopy,
                        (const uint8_t*)hostCopy + size);
    syncSend((uint8_t)SnapshotCalls::PushSnapshotDevice, req.encode());
}

void SnapshotClient::pushDeviceSnapshotFromDevice(const std::string& key,
                                                  const void* devPtr,
                                                  size_t size)
{
    if (isMockMode()) {
        std::lock_guard<std::mutex> lock(snapMockMx);
        mockedPushes.emplace_back(getHost(), key);
        return;
    }
    const std::string& thisHost = getSystemConfig().endpointHost;
    if (isSameNodeDifferentWorker(thisHost, getHost()) &&
        IpcSender::get().available(getHost())) {
        try {
            uint64_t cap = IpcSender::get().peerCapacity(getHost());
            uint64_t chunk = std::max<uint64_t>(cap / 2, 4096);
            uint64_t off = 0;
            while (off < size) {
                uint64_t n = std::min(chunk, (uint64_t)size - off);
                uint64_t ipcOff = IpcSender::get().ship(
                  getHost(), (const uint8_t*)devPtr + off, n);
                IpcChunk c;
                c.key = key;
                c.valOffset = off;
                c.ipcOffset = ipcOff;
                c.len = n;
                c.srcHost = thisHost;
                c.totalSize = size;
                syncSend((uint8_t)SnapshotCalls::PushSnapshotIpc,
                         c.encode());
                off += n;
            }
            return;
        } catch (const std::exception& e) {
            FAM_WARN("snapshot ipc ship failed (%s); falling back",
                     e.what());
        }
    }
    // Fallback: stage through the host and ride the RPC plane
    std::vector<uint8_t> hostCopy(size);
    if (hipMemcpy(hostCopy.data(), devPtr, size, hipMemcpyDeviceToHost) !=
        hipSuccess) {
        throw FaabricException("device snapshot copy-out failed");
    }
    pushDeviceSnapshot(key, hostCopy.data(), size);
}

void SnapshotClient::pushSnapshotUpdate(
  const std::string& key,
  const std::vector<SnapshotDiff>& diffs,
  const std::vector<SnapshotMergeRegion>& regions)
{
    // FlatBuffers wire format (faabric.fbs SnapshotUpdateRequest)
    FlatSnapshotUpdate req;
    req.key = key;
    for (const auto& d : diffs) {
        auto m = d.toMsg();
        FlatSnapshotDiff fd;
        fd.offset = m.offset;
        fd.dataType = m.dataType;
        fd.mergeOp = m.mergeOp;
        fd.data = std::move(m.data);
        req.diffs.push_back(std::move(fd));
    }
    for (const auto& r : regions) {
        FlatMergeRegion m;
        m.offset = (int32_t)r.offset;
        m.length = r.length;
        m.dataType = (int32_t)r.dataType;
        m.mergeOp = (int32_t)r.operation;
        req.mergeRegions.push_back(m);
    }
    syncSend((uint8_t)SnapshotCalls::PushSnapshotUpdate, req.encode());
}

void SnapshotClient::pushThreadResult(int32_t appId,
                                      int32_t messageId,
                                      int32_t returnValue,
                                      const std::string& key,
                                      const std::vector<SnapshotDiff>& diffs)
{
    ThreadResultRequest req;
    req.appId = appId;
    req.messageId = messageId;
    req.returnValue = returnValue;
    req.key = key;
    req.executedHost = getSystemConfig().endpointHost;
    for (const auto& d : diffs) {
        req.diffs.push_back(d.toMsg());
    }
    if (isMockMode()) {
        std::lock_guard<std::mutex> lock(snapMockMx);
        mockedThreadResults.emplace_back(getHost(), req);
        return;
    }
    // FlatBuffers wire format (faabric.fbs ThreadResultRequest)
    FlatThreadResult flat;
    flat.appId = req.appId;
    flat.messageId = req.messageId;
    flat.returnValue = req.returnValue;
    flat.key = req.key;
    flat.executedHost = req.executedHost;
    for (const auto& m : req.diffs) {
        FlatSnapshotDiff fd;
        fd.offset = m.offset;
        fd.dataType = m.dataType;
        fd.mergeOp = m.mergeOp;
        fd.data = m.data;
        flat.diffs.push_back(std::move(fd));
    }
    syncSend((uint8_t)SnapshotCalls::ThreadResult, flat.encode());
}

void SnapshotClient::deleteSnapshot(const std::string& key)
{
    // FlatBuffers wire format (faabric.fbs SnapshotDeleteRequest)
    FlatSnapshotDelete req;
    req.key = key;
    asyncSend((uint8_t)SnapshotCalls::DeleteSnapshot, req.encode());
}

static std::mutex snapClientsMx;
static std::map<std::string, std::shared_ptr<SnapshotClient>> snapClients;

std::shared_ptr<SnapshotClient> getSnapshotClient(const std::string& host)
{
    std::lock_guard<std::mutex> lock(snapClientsMx);
    auto& cli = snapClients[host];
    if (!cli) {
        cli = std::make_shared<SnapshotClient>(host);
    }
    return cli;
}

void clearSnapshotClients()
{
    std::lock_guard<std::mutex> lock(snapClientsMx);
    snapClients.clear();
}

} // namespace faabricamd
