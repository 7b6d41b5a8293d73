// Data model: hand-written structs encoded with the protobuf wire format
// using the SAME field numbers as the reference schemas, so the bytes stay
// interoperable (reference: src/proto/faabric.proto, src/planner/planner.proto;
// snapshot RPCs re-modelled from src/flat/faabric.fbs as protobuf-style
// messages — the flatbuffers encoding itself is not reproduced).
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

namespace faabricamd {

// --------------------------------------------------------------------------
// Core messages (reference: src/proto/faabric.proto:89-156 "Message")
// --------------------------------------------------------------------------

enum class MessageType : int32_t
{
    CALL = 0,
    KILL = 1,
    EMPTY = 2,
    FLUSH = 3,
};

struct Message
{
    int32_t id = 0;           // field 1
    int32_t appId = 0;        // field 2
    int32_t appIdx = 0;       // field 3
    std::string mainHost;     // field 4
    MessageType type = MessageType::CALL; // field 5
    std::string user;         // field 6
    std::string function;     // field 7
    std::vector<uint8_t> inputData; // field 8
    std::string outputData;   // field 9
    int32_t funcPtr = 0;      // field 10
    int32_t returnValue = 0;  // field 11
    std::string snapshotKey;  // field 12
    int64_t startTimestamp = 0;  // field 14
    std::string resultKey;       // field 15
    bool executesLocally = false; // field 16
    std::string statusKey;        // field 17
    std::string executedHost;     // field 18
    int64_t finishTimestamp = 0;  // field 19
    int32_t groupId = 0;   // field 27
    int32_t groupIdx = 0;  // field 28
    int32_t groupSize = 0; // field 29
    bool isMpi = false;        // field 30
    int32_t mpiWorldId = 0;    // field 31
    int32_t mpiRank = 0;       // field 32
    int32_t mpiWorldSize = 0;  // field 33
    std::string cmdline;       // field 34
    bool recordExecGraph = false;            // field 35
    std::vector<int32_t> chainedMsgIds;      // field 36
    std::map<std::string, int32_t> intExecGraphDetails;   // field 37
    std::map<std::string, std::string> execGraphDetails;  // field 38

    std::string encode() const;
    static Message decode(const std::string& buf);
    bool operator==(const Message& o) const;
};

enum class BatchExecuteType : int32_t
{
    FUNCTIONS = 0,
    THREADS = 1,
    PROCESSES = 2,
    MIGRATION = 3,
};

// reference: src/proto/faabric.proto:21-60
struct BatchExecuteRequest
{
    int32_t appId = 0;    // field 1
    int32_t groupId = 0;  // field 2
    std::string user;     // field 3
    std::string function; // field 4
    BatchExecuteType type = BatchExecuteType::FUNCTIONS; // field 5
    std::string snapshotKey;        // field 6
    std::vector<Message> messages;  // field 7
    int32_t subType = 0;            // field 8
    std::vector<uint8_t> contextData; // field 9
    bool singleHost = false;     // field 10
    bool singleHostHint = false; // field 11
    bool elasticScaleHint = false; // field 12

    std::string encode() const;
    static BatchExecuteRequest decode(const std::string& buf);
};

// reference: src/proto/faabric.proto:62-74
struct BatchExecuteRequestStatus
{
    int32_t appId = 0;                   // field 1
    bool finished = false;               // field 2
    std::vector<Message> messageResults; // field 3
    int32_t expectedNumMessages = 0;     // field 4

    std::string encode() const;
    static BatchExecuteRequestStatus decode(const std::string& buf);
};

struct HostResources
{
    int32_t slots = 0;     // field 1
    int32_t usedSlots = 0; // field 2

    std::string encode() const;
    static HostResources decode(const std::string& buf);
};

// --------------------------------------------------------------------------
// State service messages (reference: src/proto/faabric.proto:158-203)
// --------------------------------------------------------------------------

struct StateRequest
{
    std::string user; // 1
    std::string key;  // 2
    std::vector<uint8_t> data; // 3
    std::string encode() const;
    static StateRequest decode(const std::string& buf);
};

struct StateChunkRequest
{
    std::string user;   // 1
    std::string key;    // 2
    uint64_t offset = 0;    // 3
    uint64_t chunkSize = 0; // 4
    std::string encode() const;
    static StateChunkRequest decode(const std::string& buf);
};

struct StatePart
{
    std::string user;    // 1
    std::string key;     // 2
    uint64_t offset = 0; // 3
    std::vector<uint8_t> data; // 4
    // Full value size so a push can establish the value at the right
    // size on a host that has never seen the key (planner store mode)
    uint64_t totalSize = 0; // 5
    std::string encode() const;
    static StatePart decode(const std::string& buf);
};

struct StateSizeResponse
{
    std::string user; // 1
    std::string key;  // 2
    uint64_t stateSize = 0; // 3
    std::string encode() const;
    static StateSizeResponse decode(const std::string& buf);
};

struct StateAppendedRequest
{
    std::string user; // 1
    std::string key;  // 2
    uint32_t nValues = 0; // 3
    std::string encode() const;
    static StateAppendedRequest decode(const std::string& buf);
};

struct StateAppendedResponse
{
    std::string user; // 1
    std::string key;  // 2
    std::vector<std::vector<uint8_t>> values; // 3 (nested, data = field 2)
    std::string encode() const;
    static StateAppendedResponse decode(const std::string& buf);
};

// --------------------------------------------------------------------------
// Point-to-point (reference: src/proto/faabric.proto:207-236)
// --------------------------------------------------------------------------

struct PointToPointMessage
{
    int32_t appId = 0;   // 1
    int32_t groupId = 0; // 2
    int32_t sendIdx = 0; // 3
    int32_t recvIdx = 0; // 4
    std::vector<uint8_t> data; // 5
    std::string encode() const;
    static PointToPointMessage decode(const std::string& buf);
};

struct PointToPointMapping
{
    std::string host;    // 1
    int32_t messageId = 0; // 2
    int32_t appIdx = 0;    // 3
    int32_t groupIdx = 0;  // 4
    int32_t mpiPort = 0;   // 5
};

struct PointToPointMappings
{
    int32_t appId = 0;   // 1
    int32_t groupId = 0; // 2
    std::vector<PointToPointMapping> mappings; // 3
    std::string encode() const;
    static PointToPointMappings decode(const std::string& buf);
};

struct PendingMigration
{
    int32_t appId = 0;   // 1
    int32_t groupId = 0; // 2
    int32_t groupIdx = 0; // 3
    std::string srcHost; // 4
    std::string dstHost; // 5
    std::string encode() const;
    static PendingMigration decode(const std::string& buf);
};

// --------------------------------------------------------------------------
// Planner messages (reference: src/planner/planner.proto)
// --------------------------------------------------------------------------

struct MpiPortState
{
    int32_t port = 0; // 1
    bool used = false; // 2
};

struct Host
{
    std::string ip;        // 1
    int32_t slots = 0;     // 2
    int32_t usedSlots = 0; // 3
    int64_t registerTsEpochMs = 0; // 4 (nested Timestamp.epochMs=1)
    std::vector<MpiPortState> mpiPorts; // 5

    std::string encode() const;
    static Host decode(const std::string& buf);
};

struct PlannerConfig
{
    std::string ip;      // 1
    int32_t hostTimeout = 0; // 2
    int32_t numThreadsHttpServer = 0; // 3
    std::string encode() const;
    static PlannerConfig decode(const std::string& buf);
};

struct RegisterHostRequest
{
    Host host;              // 1
    bool overwrite = false; // 2
    std::string encode() const;
    static RegisterHostRequest decode(const std::string& buf);
};

struct RegisterHostResponse
{
    int32_t status = 0;  // 1 (ResponseStatus.status=1; 0=OK)
    PlannerConfig config; // 2
    int32_t hostId = 0;  // 3
    std::string encode() const;
    static RegisterHostResponse decode(const std::string& buf);
};

struct AvailableHostsResponse
{
    std::vector<Host> hosts; // 1
    std::string encode() const;
    static AvailableHostsResponse decode(const std::string& buf);
};

struct SetEvictedVmIpsRequest
{
    std::vector<std::string> vmIps; // 1
    std::string encode() const;
    static SetEvictedVmIpsRequest decode(const std::string& buf);
};

enum class HttpMessageType : int32_t
{
    NO_TYPE = 0,
    RESET = 1,
    FLUSH_AVAILABLE_HOSTS = 2,
    FLUSH_EXECUTORS = 3,
    FLUSH_SCHEDULING_STATE = 4,
    GET_AVAILABLE_HOSTS = 5,
    GET_CONFIG = 6,
    GET_EXEC_GRAPH = 7,
    GET_IN_FLIGHT_APPS = 8,
    EXECUTE_BATCH = 10,
    EXECUTE_BATCH_STATUS = 11,
    PRELOAD_SCHEDULING_DECISION = 12,
    SET_POLICY = 13,
    GET_POLICY = 14,
    SET_NEXT_EVICTED_VM = 15,
    // Runtime table sizes (results, in-flight, PTP state) for
    // observability and leak monitoring — extension beyond the
    // reference's op set
    GET_RUNTIME_METRICS = 16,
};

struct InFlightAppEntry
{
    int32_t appId = 0;  // 1
    int32_t subType = 0; // 2
    int32_t size = 0;    // 3
    std::vector<std::string> hostIps; // 4
};

struct GetInFlightAppsResponse
{
    std::vector<InFlightAppEntry> apps; // 1
    int32_t numMigrations = 0;          // 2
    std::vector<std::string> nextEvictedVmIps; // 3
    std::vector<InFlightAppEntry> frozenApps;  // 4 (hostIps unused)
    std::string encode() const;
    static GetInFlightAppsResponse decode(const std::string& buf);
};

// --------------------------------------------------------------------------
// Snapshot RPC messages (structurally mirrors src/flat/faabric.fbs, encoded
// protobuf-style — field numbers are this project's own)
// --------------------------------------------------------------------------

struct SnapshotMergeRegionMsg
{
    int32_t offset = 0;   // 1
    uint64_t length = 0;  // 2
    int32_t dataType = 0; // 3
    int32_t mergeOp = 0;  // 4
};

struct SnapshotDiffMsg
{
    int32_t offset = 0;   // 1
    int32_t dataType = 0; // 2
    int32_t mergeOp = 0;  // 3
    std::vector<uint8_t> data; // 4
};

struct SnapshotPushRequest
{
    std::string key;       // 1
    uint64_t maxSize = 0;  // 2
    std::vector<uint8_t> contents; // 3
    std::vector<SnapshotMergeRegionMsg> mergeRegions; // 4
    bool onDevice = false; // 5 (HBM-resident snapshot)
    std::string encode() const;
    static SnapshotPushRequest decode(const std::string& buf);
};

struct SnapshotUpdateRequest
{
    std::string key; // 1
    std::vector<SnapshotMergeRegionMsg> mergeRegions; // 2
    std::vector<SnapshotDiffMsg> diffs;               // 3
    std::string encode() const;
    static SnapshotUpdateRequest decode(const std::string& buf);
};

struct SnapshotDeleteRequest
{
    std::string key; // 1
    std::string encode() const;
    static SnapshotDeleteRequest decode(const std::string& buf);
};

struct ThreadResultRequest
{
    int32_t appId = 0;       // 1
    int32_t messageId = 0;   // 2
    int32_t returnValue = 0; // 3
    std::string key;         // 4
    std::vector<SnapshotDiffMsg> diffs; // 5
    std::string executedHost; // 6 (slot accounting on the planner)
    std::string encode() const;
    static ThreadResultRequest decode(const std::string& buf);
};

// --------------------------------------------------------------------------
// Factories (reference: src/util/batch.cpp batchExecFactory,
// src/util/func.cpp messageFactory)
// --------------------------------------------------------------------------

Message messageFactory(const std::string& user, const std::string& function);

BatchExecuteRequest batchExecFactory(const std::string& user,
                                     const std::string& function,
                                     int count);

void updateBatchExecAppId(BatchExecuteRequest& ber, int32_t newAppId);
void updateBatchExecGroupId(BatchExecuteRequest& ber, int32_t newGroupId);
bool isBatchExecRequestValid(const BatchExecuteRequest& ber);

} // namespace faabricamd
