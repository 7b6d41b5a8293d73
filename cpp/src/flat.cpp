// Schema-specific FlatBuffers codecs for the snapshot RPCs (see flat.h;
// field ids follow /root/reference/src/flat/faabric.fbs declaration
// order). Children are created before their parents so every uoffset
// points forward, per the format.
#include "faabricamd/flat.h"

namespace faabricamd {

static uint32_t writeMergeRegion(FlatWriter& w, const FlatMergeRegion& r)
{
    w.startTable();
    if (r.offset != 0) {
        w.addScalarField(0, (uint32_t)r.offset, 4);
    }
    if (r.length != 0) {
        w.addScalarField(1, r.length, 8);
    }
    if (r.dataType != 0) {
        w.addScalarField(2, (uint32_t)r.dataType, 4);
    }
    if (r.mergeOp != 0) {
        w.addScalarField(3, (uint32_t)r.mergeOp, 4);
    }
    return w.endTable();
}

static FlatMergeRegion readMergeRegion(const FlatReader& r, uint32_t pos)
{
    FlatMergeRegion out;
    out.offset = r.scalar<int32_t>(pos, 0);
    out.length = r.scalar<uint64_t>(pos, 1);
    out.dataType = r.scalar<int32_t>(pos, 2);
    out.mergeOp = r.scalar<int32_t>(pos, 3);
    return out;
}

static uint32_t writeDiff(FlatWriter& w, const FlatSnapshotDiff& d)
{
    uint32_t dataOff =
      w.createByteVector(d.data.data(), d.data.size());
    w.startTable();
    if (d.offset != 0) {
        w.addScalarField(0, (uint32_t)d.offset, 4);
    }
    if (d.dataType != 0) {
        w.addScalarField(1, (uint32_t)d.dataType, 4);
    }
    if (d.mergeOp != 0) {
        w.addScalarField(2, (uint32_t)d.mergeOp, 4);
    }
    w.addOffsetField(3, dataOff);
    return w.endTable();
}

static FlatSnapshotDiff readDiff(const FlatReader& r, uint32_t pos)
{
    FlatSnapshotDiff out;
    out.offset = r.scalar<int32_t>(pos, 0);
    out.dataType = r.scalar<int32_t>(pos, 1);
    out.mergeOp = r.scalar<int32_t>(pos, 2);
    out.data = r.bytes(pos, 3);
    return out;
}

std::string FlatSnapshotPush::encode() const
{
    FlatWriter w;
    std::vector<uint32_t> regionOffs;
    for (const auto& r : mergeRegions) {
        regionOffs.push_back(writeMergeRegion(w, r));
    }
    uint32_t regionsVec =
      mergeRegions.empty() ? 0 : w.createOffsetVector(regionOffs);
    uint32_t contentsVec =
      w.createByteVector(contents.data(), contents.size());
    uint32_t keyOff = w.createString(key);
    w.startTable();
    w.addOffsetField(0, keyOff);
    if (maxSize != 0) {
        w.addScalarField(1, maxSize, 8);
    }
    w.addOffsetField(2, contentsVec);
    w.addOffsetField(3, regionsVec);
    return w.finish(w.endTable());
}

FlatSnapshotPush FlatSnapshotPush::decode(const std::string& buf)
{
    FlatSnapshotPush out;
    FlatReader r(buf);
    uint32_t root = r.root();
    out.key = r.str(root, 0);
    out.maxSize = r.scalar<uint64_t>(root, 1);
    out.contents = r.bytes(root, 2);
    for (uint32_t pos : r.tableVector(root, 3)) {
        out.mergeRegions.push_back(readMergeRegion(r, pos));
    }
    return out;
}

std::string FlatSnapshotDelete::encode() const
{
    FlatWriter w;
    uint32_t keyOff = w.createString(key);
    w.startTable();
    w.addOffsetField(0, keyOff);
    return w.finish(w.endTable());
}

FlatSnapshotDelete FlatSnapshotDelete::decode(const std::string& buf)
{
    FlatSnapshotDelete out;
    FlatReader r(buf);
    out.key = r.str(r.root(), 0);
    return out;
}

std::string FlatSnapshotUpdate::encode() const
{
    FlatWriter w;
    std::vector<uint32_t> diffOffs;
    for (const auto& d : diffs) {
        diffOffs.push_back(writeDiff(w, d));
    }
    uint32_t diffsVec = diffs.empty() ? 0 : w.createOffsetVector(diffOffs);
    std::vector<uint32_t> regionOffs;
    for (const auto& r : mergeRegions) {
        regionOffs.push_back(writeMergeRegion(w, r));
    }
    uint32_t regionsVec =
      mergeRegions.empty() ? 0 : w.createOffsetVector(regionOffs);
    uint32_t keyOff = w.createString(key);
    w.startTable();
    w.addOffsetField(0, keyOff);
    w.addOffsetField(1, regionsVec);
    w.addOffsetField(2, diffsVec);
    return w.finish(w.endTable());
}

FlatSnapshotUpdate FlatSnapshotUpdate::decode(const std::string& buf)
{
    FlatSnapshotUpdate out;
    FlatReader r(buf);
    uint32_t root = r.root();
    out.key = r.str(root, 0);
    for (uint32_t pos : r.tableVector(root, 1)) {
        out.mergeRegions.push_back(readMergeRegion(r, pos));
    }
    for (uint32_t pos : r.tableVector(root, 2)) {
        out.diffs.push_back(readDiff(r, pos));
    }
    return out;
}

std::string FlatThreadResult::encode() const
{
    FlatWriter w;
    std::vector<uint32_t> diffOffs;
    for (const auto& d : diffs) {
        diffOffs.push_back(writeDiff(w, d));
    }
    uint32_t diffsVec = diffs.empty() ? 0 : w.createOffsetVector(diffOffs);
    uint32_t hostOff =
      executedHost.empty() ? 0 : w.createString(executedHost);
    uint32_t keyOff = w.createString(key);
    w.startTable();
    if (appId != 0) {
        w.addScalarField(0, (uint32_t)appId, 4);
    }
    if (messageId != 0) {
        w.addScalarField(1, (uint32_t)messageId, 4);
    }
    if (returnValue != 0) {
        w.addScalarField(2, (uint32_t)returnValue, 4);
    }
    w.addOffsetField(3, keyOff);
    w.addOffsetField(4, diffsVec);
    w.addOffsetField(5, hostOff);
    return w.finish(w.endTable());
}

FlatThreadResult FlatThreadResult::decode(const std::string& buf)
{
    FlatThreadResult out;
    FlatReader r(buf);
    uint32_t root = r.root();
    out.appId = r.scalar<int32_t>(root, 0);
    out.messageId = r.scalar<int32_t>(root, 1);
    out.returnValue = r.scalar<int32_t>(root, 2);
    out.key = r.str(root, 3);
    for (uint32_t pos : r.tableVector(root, 4)) {
        out.diffs.push_back(readDiff(r, pos));
    }
    out.executedHost = r.str(root, 5);
    return out;
}

} // namespace faabricamd
