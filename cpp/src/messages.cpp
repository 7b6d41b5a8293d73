#include "faabricamd/messages.h"
#include "faabricamd/util.h"
#include "faabricamd/wire.h"

namespace faabricamd {

// --------------------------- Message ---------------------------------------

std::string Message::encode() const
{
    PbWriter w;
    w.putInt32(1, id);
    w.putInt32(2, appId);
    w.putInt32(3, appIdx);
    w.putString(4, mainHost);
    w.putInt32(5, (int32_t)type);
    w.putString(6, user);
    w.putString(7, function);
    w.putBytes(8, inputData);
    w.putString(9, outputData);
    w.putInt32(10, funcPtr);
    w.putInt32(11, returnValue);
    w.putString(12, snapshotKey);
    w.putInt64(14, startTimestamp);
    w.putString(15, resultKey);
    w.putBool(16, executesLocally);
    w.putString(17, statusKey);
    w.putString(18, executedHost);
    w.putInt64(19, finishTimestamp);
    w.putInt32(27, groupId);
    w.putInt32(28, groupIdx);
    w.putInt32(29, groupSize);
    w.putBool(30, isMpi);
    w.putInt32(31, mpiWorldId);
    w.putInt32(32, mpiRank);
    w.putInt32(33, mpiWorldSize);
    w.putString(34, cmdline);
    w.putBool(35, recordExecGraph);
    w.putPackedInt32(36, chainedMsgIds);
    for (const auto& [k, v] : intExecGraphDetails) {
        PbWriter kv;
        kv.putString(1, k);
        kv.putInt32(2, v);
        w.putMessage(37, kv.buffer());
    }
    for (const auto& [k, v] : execGraphDetails) {
        PbWriter kv;
        kv.putString(1, k);
        kv.putString(2, v);
        w.putMessage(38, kv.buffer());
    }
    return w.take();
}

Message Message::decode(const std::string& buf)
{
    Message m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.id = r.asInt32(); break;
            case 2: m.appId = r.asInt32(); break;
            case 3: m.appIdx = r.asInt32(); break;
            case 4: m.mainHost = r.asString(); break;
            case 5: m.type = (MessageType)r.asInt32(); break;
            case 6: m.user = r.asString(); break;
            case 7: m.function = r.asString(); break;
            case 8: m.inputData = r.asBytes(); break;
            case 9: m.outputData = r.asString(); break;
            case 10: m.funcPtr = r.asInt32(); break;
            case 11: m.returnValue = r.asInt32(); break;
            case 12: m.snapshotKey = r.asString(); break;
            case 14: m.startTimestamp = r.asInt64(); break;
            case 15: m.resultKey = r.asString(); break;
            case 16: m.executesLocally = r.asBool(); break;
            case 17: m.statusKey = r.asString(); break;
            case 18: m.executedHost = r.asString(); break;
            case 19: m.finishTimestamp = r.asInt64(); break;
            case 27: m.groupId = r.asInt32(); break;
            case 28: m.groupIdx = r.asInt32(); break;
            case 29: m.groupSize = r.asInt32(); break;
            case 30: m.isMpi = r.asBool(); break;
            case 31: m.mpiWorldId = r.asInt32(); break;
            case 32: m.mpiRank = r.asInt32(); break;
            case 33: m.mpiWorldSize = r.asInt32(); break;
            case 34: m.cmdline = r.asString(); break;
            case 35: m.recordExecGraph = r.asBool(); break;
            case 36:
                if (t == WireType::LengthDelimited) {
                    m.chainedMsgIds = r.asPackedInt32();
                } else {
                    m.chainedMsgIds.push_back(r.asInt32());
                }
                break;
            case 37: {
                PbReader kv = r.asSub();
                std::string k;
                int32_t v = 0;
                uint32_t kf;
                WireType kt;
                while (kv.next(kf, kt)) {
                    if (kf == 1) {
                        k = kv.asString();
                    } else if (kf == 2) {
                        v = kv.asInt32();
                    } else {
                        kv.skip(kt);
                    }
                }
                m.intExecGraphDetails[k] = v;
                break;
            }
            case 38: {
                PbReader kv = r.asSub();
                std::string k;
                std::string v;
                uint32_t kf;
                WireType kt;
                while (kv.next(kf, kt)) {
                    if (kf == 1) {
                        k = kv.asString();
                    } else if (kf == 2) {
                        v = kv.asString();
                    } else {
                        kv.skip(kt);
                    }
                }
                m.execGraphDetails[k] = v;
                break;
            }
            default:
                r.skip(t);
        }
    }
    return m;
}

bool Message::operator==(const Message& o) const
{
    return encode() == o.encode();
}

// --------------------------- BatchExecuteRequest ----------------------------

std::string BatchExecuteRequest::encode() const
{
    PbWriter w;
    w.putInt32(1, appId);
    w.putInt32(2, groupId);
    w.putString(3, user);
    w.putString(4, function);
    w.putInt32(5, (int32_t)type);
    w.putString(6, snapshotKey);
    for (const auto& m : messages) {
        w.putMessage(7, m.encode());
    }
    w.putInt32(8, subType);
    w.putBytes(9, contextData);
    w.putBool(10, singleHost);
    w.putBool(11, singleHostHint);
    w.putBool(12, elasticScaleHint);
    return w.take();
}

BatchExecuteRequest BatchExecuteRequest::decode(const std::string& buf)
{
    BatchExecuteRequest b;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: b.appId = r.asInt32(); break;
            case 2: b.groupId = r.asInt32(); break;
            case 3: b.user = r.asString(); break;
            case 4: b.function = r.asString(); break;
            case 5: b.type = (BatchExecuteType)r.asInt32(); break;
            case 6: b.snapshotKey = r.asString(); break;
            case 7: b.messages.push_back(Message::decode(r.asString())); break;
            case 8: b.subType = r.asInt32(); break;
            case 9: b.contextData = r.asBytes(); break;
            case 10: b.singleHost = r.asBool(); break;
            case 11: b.singleHostHint = r.asBool(); break;
            case 12: b.elasticScaleHint = r.asBool(); break;
            default: r.skip(t);
        }
    }
    return b;
}

// --------------------------- BatchExecuteRequestStatus ----------------------

std::string BatchExecuteRequestStatus::encode() const
{
    PbWriter w;
    w.putInt32(1, appId);
    w.putBool(2, finished);
    for (const auto& m : messageResults) {
        w.putMessage(3, m.encode());
    }
    w.putInt32(4, expectedNumMessages);
    return w.take();
}

BatchExecuteRequestStatus BatchExecuteRequestStatus::decode(
  const std::string& buf)
{
    BatchExecuteRequestStatus s;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: s.appId = r.asInt32(); break;
            case 2: s.finished = r.asBool(); break;
            case 3:
                s.messageResults.push_back(Message::decode(r.asString()));
                break;
            case 4: s.expectedNumMessages = r.asInt32(); break;
            default: r.skip(t);
        }
    }
    return s;
}

// --------------------------- HostResources ----------------------------------

std::string HostResources::encode() const
{
    PbWriter w;
    w.putInt32(1, slots);
    w.putInt32(2, usedSlots);
    return w.take();
}

HostResources HostResources::decode(const std::string& buf)
{
    HostResources h;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: h.slots = r.asInt32(); break;
            case 2: h.usedSlots = r.asInt32(); break;
            default: r.skip(t);
        }
    }
    return h;
}

// --------------------------- State messages ---------------------------------

std::string StateRequest::encode() const
{
    PbWriter w;
    w.putString(1, user);
    w.putString(2, key);
    w.putBytes(3, data);
    return w.take();
}

StateRequest StateRequest::decode(const std::string& buf)
{
    StateRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.user = r.asString(); break;
            case 2: m.key = r.asString(); break;
            case 3: m.data = r.asBytes(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string StateChunkRequest::encode() const
{
    PbWriter w;
    w.putString(1, user);
    w.putString(2, key);
    w.putUInt64(3, offset);
    w.putUInt64(4, chunkSize);
    return w.take();
}

StateChunkRequest StateChunkRequest::decode(const std::string& buf)
{
    StateChunkRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.user = r.asString(); break;
            case 2: m.key = r.asString(); break;
            case 3: m.offset = r.asUInt64(); break;
            case 4: m.chunkSize = r.asUInt64(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string StatePart::encode() const
{
    PbWriter w;
    w.putString(1, user);
    w.putString(2, key);
    w.putUInt64(3, offset);
    w.putBytes(4, data);
    if (totalSize != 0) {
        w.putUInt64(5, totalSize);
    }
    return w.take();
}

StatePart StatePart::decode(const std::string& buf)
{
    StatePart m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.user = r.asString(); break;
            case 2: m.key = r.asString(); break;
            case 3: m.offset = r.asUInt64(); break;
            case 4: m.data = r.asBytes(); break;
            case 5: m.totalSize = r.asUInt64(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string StateSizeResponse::encode() const
{
    PbWriter w;
    w.putString(1, user);
    w.putString(2, key);
    w.putUInt64(3, stateSize);
    return w.take();
}

StateSizeResponse StateSizeResponse::decode(const std::string& buf)
{
    StateSizeResponse m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.user = r.asString(); break;
            case 2: m.key = r.asString(); break;
            case 3: m.stateSize = r.asUInt64(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string StateAppendedRequest::encode() const
{
    PbWriter w;
    w.putString(1, user);
    w.putString(2, key);
    w.putUInt64(3, nValues);
    return w.take();
}

StateAppendedRequest StateAppendedRequest::decode(const std::string& buf)
{
    StateAppendedRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.user = r.asString(); break;
            case 2: m.key = r.asString(); break;
            case 3: m.nValues = (uint32_t)r.asUInt64(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string StateAppendedResponse::encode() const
{
    PbWriter w;
    w.putString(1, user);
    w.putString(2, key);
    for (const auto& v : values) {
        PbWriter inner;
        inner.putBytes(2, v);
        w.putMessage(3, inner.buffer());
    }
    return w.take();
}

StateAppendedResponse StateAppendedResponse::decode(const std::string& buf)
{
    StateAppendedResponse m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.user = r.asString(); break;
            case 2: m.key = r.asString(); break;
            case 3: {
                PbReader sub = r.asSub();
                uint32_t sf;
                WireType st;
                std::vector<uint8_t> data;
                while (sub.next(sf, st)) {
                    if (sf == 2) {
                        data = sub.asBytes();
                    } else {
                        sub.skip(st);
                    }
                }
                m.values.push_back(std::move(data));
                break;
            }
            default: r.skip(t);
        }
    }
    return m;
}

// --------------------------- Point-to-point ---------------------------------

std::string PointToPointMessage::encode() const
{
    PbWriter w;
    w.putInt32(1, appId);
    w.putInt32(2, groupId);
    w.putInt32(3, sendIdx);
    w.putInt32(4, recvIdx);
    w.putBytes(5, data);
    return w.take();
}

PointToPointMessage PointToPointMessage::decode(const std::string& buf)
{
    PointToPointMessage m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.appId = r.asInt32(); break;
            case 2: m.groupId = r.asInt32(); break;
            case 3: m.sendIdx = r.asInt32(); break;
            case 4: m.recvIdx = r.asInt32(); break;
            case 5: m.data = r.asBytes(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string PointToPointMappings::encode() const
{
    PbWriter w;
    w.putInt32(1, appId);
    w.putInt32(2, groupId);
    for (const auto& m : mappings) {
        PbWriter inner;
        inner.putString(1, m.host);
        inner.putInt32(2, m.messageId);
        inner.putInt32(3, m.appIdx);
        inner.putInt32(4, m.groupIdx);
        inner.putInt32(5, m.mpiPort);
        w.putMessage(3, inner.buffer());
    }
    return w.take();
}

PointToPointMappings PointToPointMappings::decode(const std::string& buf)
{
    PointToPointMappings m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.appId = r.asInt32(); break;
            case 2: m.groupId = r.asInt32(); break;
            case 3: {
                PbReader sub = r.asSub();
                PointToPointMapping mp;
                uint32_t sf;
                WireType st;
                while (sub.next(sf, st)) {
                    switch (sf) {
                        case 1: mp.host = sub.asString(); break;
                        case 2: mp.messageId = sub.asInt32(); break;
                        case 3: mp.appIdx = sub.asInt32(); break;
                        case 4: mp.groupIdx = sub.asInt32(); break;
                        case 5: mp.mpiPort = sub.asInt32(); break;
                        default: sub.skip(st);
                    }
                }
                m.mappings.push_back(std::move(mp));
                break;
            }
            default: r.skip(t);
        }
    }
    return m;
}

std::string PendingMigration::encode() const
{
    PbWriter w;
    w.putInt32(1, appId);
    w.putInt32(2, groupId);
    w.putInt32(3, groupIdx);
    w.putString(4, srcHost);
    w.putString(5, dstHost);
    return w.take();
}

PendingMigration PendingMigration::decode(const std::string& buf)
{
    PendingMigration m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.appId = r.asInt32(); break;
            case 2: m.groupId = r.asInt32(); break;
            case 3: m.groupIdx = r.asInt32(); break;
            case 4: m.srcHost = r.asString(); break;
            case 5: m.dstHost = r.asString(); break;
            default: r.skip(t);
        }
    }
    return m;
}

// --------------------------- Planner messages -------------------------------

std::string Host::encode() const
{
    PbWriter w;
    w.putString(1, ip);
    w.putInt32(2, slots);
    w.putInt32(3, usedSlots);
    if (registerTsEpochMs != 0) {
        PbWriter ts;
        ts.putInt64(1, registerTsEpochMs);
        w.putMessage(4, ts.buffer());
    }
    for (const auto& p : mpiPorts) {
        PbWriter inner;
        inner.putInt32(1, p.port);
        inner.putBool(2, p.used);
        w.putMessage(5, inner.buffer());
    }
    return w.take();
}

Host Host::decode(const std::string& buf)
{
    Host h;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: h.ip = r.asString(); break;
            case 2: h.slots = r.asInt32(); break;
            case 3: h.usedSlots = r.asInt32(); break;
            case 4: {
                PbReader sub = r.asSub();
                uint32_t sf;
                WireType st;
                while (sub.next(sf, st)) {
                    if (sf == 1) {
                        h.registerTsEpochMs = sub.asInt64();
                    } else {
                        sub.skip(st);
                    }
                }
                break;
            }
            case 5: {
                PbReader sub = r.asSub();
                MpiPortState p;
                uint32_t sf;
                WireType st;
                while (sub.next(sf, st)) {
                    if (sf == 1) {
                        p.port = sub.asInt32();
                    } else if (sf == 2) {
                        p.used = sub.asBool();
                    } else {
                        sub.skip(st);
                    }
                }
                h.mpiPorts.push_back(p);
                break;
            }
            default: r.skip(t);
        }
    }
    return h;
}

std::string PlannerConfig::encode() const
{
    PbWriter w;
    w.putString(1, ip);
    w.putInt32(2, hostTimeout);
    w.putInt32(3, numThreadsHttpServer);
    return w.take();
}

PlannerConfig PlannerConfig::decode(const std::string& buf)
{
    PlannerConfig c;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: c.ip = r.asString(); break;
            case 2: c.hostTimeout = r.asInt32(); break;
            case 3: c.numThreadsHttpServer = r.asInt32(); break;
            default: r.skip(t);
        }
    }
    return c;
}

std::string RegisterHostRequest::encode() const
{
    PbWriter w;
    w.putMessage(1, host.encode());
    w.putBool(2, overwrite);
    return w.take();
}

RegisterHostRequest RegisterHostRequest::decode(const std::string& buf)
{
    RegisterHostRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.host = Host::decode(r.asString()); break;
            case 2: m.overwrite = r.asBool(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string RegisterHostResponse::encode() const
{
    PbWriter w;
    if (status != 0) {
        PbWriter st;
        st.putInt32(1, status);
        w.putMessage(1, st.buffer());
    }
    w.putMessage(2, config.encode());
    w.putInt32(3, hostId);
    return w.take();
}

RegisterHostResponse RegisterHostResponse::decode(const std::string& buf)
{
    RegisterHostResponse m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: {
                PbReader sub = r.asSub();
                uint32_t sf;
                WireType st;
                while (sub.next(sf, st)) {
                    if (sf == 1) {
                        m.status = sub.asInt32();
                    } else {
                        sub.skip(st);
                    }
                }
                break;
            }
            case 2: m.config = PlannerConfig::decode(r.asString()); break;
            case 3: m.hostId = r.asInt32(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string AvailableHostsResponse::encode() const
{
    PbWriter w;
    for (const auto& h : hosts) {
        w.putMessage(1, h.encode());
    }
    return w.take();
}

AvailableHostsResponse AvailableHostsResponse::decode(const std::string& buf)
{
    AvailableHostsResponse m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        if (f == 1) {
            m.hosts.push_back(Host::decode(r.asString()));
        } else {
            r.skip(t);
        }
    }
    return m;
}

std::string SetEvictedVmIpsRequest::encode() const
{
    PbWriter w;
    for (const auto& ip : vmIps) {
        w.putString(1, ip);
    }
    return w.take();
}

SetEvictedVmIpsRequest SetEvictedVmIpsRequest::decode(const std::string& buf)
{
    SetEvictedVmIpsRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        if (f == 1) {
            m.vmIps.push_back(r.asString());
        } else {
            r.skip(t);
        }
    }
    return m;
}

static std::string encodeInFlightApp(const InFlightAppEntry& e)
{
    PbWriter w;
    w.putInt32(1, e.appId);
    w.putInt32(2, e.subType);
    w.putInt32(3, e.size);
    for (const auto& ip : e.hostIps) {
        w.putString(4, ip);
    }
    return w.take();
}

static InFlightAppEntry decodeInFlightApp(const std::string& buf)
{
    InFlightAppEntry e;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: e.appId = r.asInt32(); break;
            case 2: e.subType = r.asInt32(); break;
            case 3: e.size = r.asInt32(); break;
            case 4: e.hostIps.push_back(r.asString()); break;
            default: r.skip(t);
        }
    }
    return e;
}

std::string GetInFlightAppsResponse::encode() const
{
    PbWriter w;
    for (const auto& a : apps) {
        w.putMessage(1, encodeInFlightApp(a));
    }
    w.putInt32(2, numMigrations);
    for (const auto& ip : nextEvictedVmIps) {
        w.putString(3, ip);
    }
    for (const auto& a : frozenApps) {
        w.putMessage(4, encodeInFlightApp(a));
    }
    return w.take();
}

GetInFlightAppsResponse GetInFlightAppsResponse::decode(const std::string& buf)
{
    GetInFlightAppsResponse m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.apps.push_back(decodeInFlightApp(r.asString())); break;
            case 2: m.numMigrations = r.asInt32(); break;
            case 3: m.nextEvictedVmIps.push_back(r.asString()); break;
            case 4:
                m.frozenApps.push_back(decodeInFlightApp(r.asString()));
                break;
            default: r.skip(t);
        }
    }
    return m;
}

// --------------------------- Snapshot messages ------------------------------

static std::string encodeMergeRegion(const SnapshotMergeRegionMsg& m)
{
    PbWriter w;
    w.putInt32(1, m.offset);
    w.putUInt64(2, m.length);
    w.putInt32(3, m.dataType);
    w.putInt32(4, m.mergeOp);
    return w.take();
}

static SnapshotMergeRegionMsg decodeMergeRegion(const std::string& buf)
{
    SnapshotMergeRegionMsg m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.offset = r.asInt32(); break;
            case 2: m.length = r.asUInt64(); break;
            case 3: m.dataType = r.asInt32(); break;
            case 4: m.mergeOp = r.asInt32(); break;
            default: r.skip(t);
        }
    }
    return m;
}

static std::string encodeDiff(const SnapshotDiffMsg& m)
{
    PbWriter w;
    w.putInt32(1, m.offset);
    w.putInt32(2, m.dataType);
    w.putInt32(3, m.mergeOp);
    w.putBytes(4, m.data);
    return w.take();
}

static SnapshotDiffMsg decodeDiff(const std::string& buf)
{
    SnapshotDiffMsg m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.offset = r.asInt32(); break;
            case 2: m.dataType = r.asInt32(); break;
            case 3: m.mergeOp = r.asInt32(); break;
            case 4: m.data = r.asBytes(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string SnapshotPushRequest::encode() const
{
    PbWriter w;
    w.putString(1, key);
    w.putUInt64(2, maxSize);
    w.putBytes(3, contents);
    for (const auto& m : mergeRegions) {
        w.putMessage(4, encodeMergeRegion(m));
    }
    w.putBool(5, onDevice);
    return w.take();
}

SnapshotPushRequest SnapshotPushRequest::decode(const std::string& buf)
{
    SnapshotPushRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.key = r.asString(); break;
            case 2: m.maxSize = r.asUInt64(); break;
            case 3: m.contents = r.asBytes(); break;
            case 4:
                m.mergeRegions.push_back(decodeMergeRegion(r.asString()));
                break;
            case 5: m.onDevice = r.asBool(); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string SnapshotUpdateRequest::encode() const
{
    PbWriter w;
    w.putString(1, key);
    for (const auto& m : mergeRegions) {
        w.putMessage(2, encodeMergeRegion(m));
    }
    for (const auto& d : diffs) {
        w.putMessage(3, encodeDiff(d));
    }
    return w.take();
}

SnapshotUpdateRequest SnapshotUpdateRequest::decode(const std::string& buf)
{
    SnapshotUpdateRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.key = r.asString(); break;
            case 2:
                m.mergeRegions.push_back(decodeMergeRegion(r.asString()));
                break;
            case 3: m.diffs.push_back(decodeDiff(r.asString())); break;
            default: r.skip(t);
        }
    }
    return m;
}

std::string SnapshotDeleteRequest::encode() const
{
    PbWriter w;
    w.putString(1, key);
    return w.take();
}

SnapshotDeleteRequest SnapshotDeleteRequest::decode(const std::string& buf)
{
    SnapshotDeleteRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        if (f == 1) {
            m.key = r.asString();
        } else {
            r.skip(t);
        }
    }
    return m;
}

std::string ThreadResultRequest::encode() const
{
    PbWriter w;
    w.putInt32(1, appId);
    w.putInt32(2, messageId);
    w.putInt32(3, returnValue);
    w.putString(4, key);
    for (const auto& d : diffs) {
        w.putMessage(5, encodeDiff(d));
    }
    w.putString(6, executedHost);
    return w.take();
}

ThreadResultRequest ThreadResultRequest::decode(const std::string& buf)
{
    ThreadResultRequest m;
    PbReader r(buf);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        switch (f) {
            case 1: m.appId = r.asInt32(); break;
            case 2: m.messageId = r.asInt32(); break;
            case 3: m.returnValue = r.asInt32(); break;
            case 4: m.key = r.asString(); break;
            case 5: m.diffs.push_back(decodeDiff(r.asString())); break;
            case 6: m.executedHost = r.asString(); break;
            default: r.skip(t);
        }
    }
    return m;
}

// --------------------------- Factories --------------------------------------

Message messageFactory(const std::string& user, const std::string& function)
{
    Message m;
    m.id = generateGidInt32();
    m.appId = generateGidInt32();
    m.user = user;
    m.function = function;
    m.startTimestamp = getGlobalClockEpochMillis();
    m.mainHost = getSystemConfig().endpointHost;
    return m;
}

BatchExecuteRequest batchExecFactory(const std::string& user,
                                     const std::string& function,
                                     int count)
{
    BatchExecuteRequest ber;
    ber.appId = generateGidInt32();
    ber.user = user;
    ber.function = function;
    for (int i = 0; i < count; i++) {
        Message m = messageFactory(user, function);
        m.appId = ber.appId;
        m.appIdx = i;
        ber.messages.push_back(std::move(m));
    }
    return ber;
}

void updateBatchExecAppId(BatchExecuteRequest& ber, int32_t newAppId)
{
    ber.appId = newAppId;
    for (auto& m : ber.messages) {
        m.appId = newAppId;
    }
}

void updateBatchExecGroupId(BatchExecuteRequest& ber, int32_t newGroupId)
{
    ber.groupId = newGroupId;
    for (auto& m : ber.messages) {
        m.groupId = newGroupId;
    }
}

bool isBatchExecRequestValid(const BatchExecuteRequest& ber)
{
    if (ber.user.empty() || ber.function.empty() || ber.appId == 0) {
        return false;
    }
    for (const auto& m : ber.messages) {
        if (m.user != ber.user || m.function != ber.function ||
            m.appId != ber.appId) {
            return false;
        }
    }
    return true;
}

} // namespace faabricamd
