// faabric-mi355x util foundation: config, logging, gids, time, bytes, network.
//
// MI355X-native re-design of the reference's util layer
// (reference: include/faabric/util/config.h:12-70, src/util/config.cpp:19-87,
//  src/util/gids.cpp:16-28, src/util/logging.cpp, util/bytes.h,
//  util/network.h:8-10). Not a port: single header, env-driven config with
// GPU-aware additions (gpusPerHost, hbmArenaBytes), printf-style logger.
#pragma once

#include <atomic>
#include <cstdarg>
#include <cstdint>
#include <cstring>
#include <string>
#include <vector>

namespace faabricamd {

// ----------------------------- logging ------------------------------------

enum class LogLevel : int
{
    trace = 0,
    debug = 1,
    info = 2,
    warn = 3,
    error = 4,
    off = 5,
};

LogLevel getLogLevel();
void setLogLevel(LogLevel lvl);
void logMessage(LogLevel lvl, const char* fmt, ...)
  __attribute__((format(printf, 2, 3)));

#define FAM_TRACE(...)                                                         \
    ::faabricamd::logMessage(::faabricamd::LogLevel::trace, __VA_ARGS__)
#define FAM_DEBUG(...)                                                         \
    ::faabricamd::logMessage(::faabricamd::LogLevel::debug, __VA_ARGS__)
#define FAM_INFO(...)                                                          \
    ::faabricamd::logMessage(::faabricamd::LogLevel::info, __VA_ARGS__)
#define FAM_WARN(...)                                                          \
    ::faabricamd::logMessage(::faabricamd::LogLevel::warn, __VA_ARGS__)
#define FAM_ERROR(...)                                                         \
    ::faabricamd::logMessage(::faabricamd::LogLevel::error, __VA_ARGS__)

// ----------------------------- errors --------------------------------------

class FaabricException : public std::exception
{
  public:
    explicit FaabricException(std::string msgIn)
      : msg(std::move(msgIn))
    {}
    const char* what() const noexcept override { return msg.c_str(); }

  private:
    std::string msg;
};

// Thrown out of executeTask when a function must migrate / freeze
// (reference: include/faabric/util/func.h:8-9 sentinels).
class FunctionMigratedException : public FaabricException
{
  public:
    using FaabricException::FaabricException;
};
class FunctionFrozenException : public FaabricException
{
  public:
    using FaabricException::FaabricException;
};

inline constexpr int MIGRATED_FUNCTION_RETURN_VALUE = -99;
inline constexpr int FROZEN_FUNCTION_RETURN_VALUE = -98;

// ----------------------------- config --------------------------------------

std::string getEnvVar(const std::string& key, const std::string& deflt);
int getEnvVarInt(const std::string& key, int deflt);

// Usable slots on this host. For the MI355X build a "slot" is by default a
// GPU (one executor slot per GPU on an 8xMI355X node); with no GPUs we fall
// back to CPU cores (reference: getUsableCores,
// src/util/environment.cpp + src/scheduler/Scheduler.cpp:65).
int getUsableCores();

struct SystemConfig
{
    // Global
    std::string logLevel;
    std::string endpointHost;   // this host's IP as seen by peers
    std::string plannerHost;
    int plannerPort = 8011;

    // Scheduling
    std::string batchSchedulerMode; // bin-pack | compact | spot
    int overrideCpuCount = 0;
    int overrideGpuCount = -1; // -1 = probe HIP
    // Which HIP device this worker's HBM lives on. Default deployments
    // pin visibility per worker (HIP_VISIBLE_DEVICES=rank, device 0);
    // all-visible deployments set FAABRIC_GPU_DEVICE=rank instead so
    // same-node IPC maps peer GPUs over xGMI.
    int gpuDevice = 0;
    // True when FAABRIC_GPU_DEVICE was set explicitly: this worker is
    // pinned to one GPU and executors must not spread across devices
    bool gpuDevicePinned = false;
    bool useGpu = true;        // slots are GPUs when available

    // Timeouts (seconds, matching reference defaults
    // src/util/config.cpp:60-75)
    int globalMessageTimeout = 60000; // ms
    int boundTimeout = 30000;         // ms

    // Transport
    int functionServerThreads = 4;
    int stateServerThreads = 2;
    int snapshotServerThreads = 2;
    int pointToPointServerThreads = 4;
    int plannerServerThreads = 4;

    // Dirty tracking / diffing (reference: src/util/config.cpp:81-82).
    // MI355X modes: "none" (every page dirty), "compare" (XOR-against-
    // baseline HIP kernel), "segfault" (host memory mprotect parity mode).
    std::string dirtyTrackingMode;
    std::string diffingMode; // xor | bytewise

    // State
    std::string stateMode; // inmemory

    // MPI
    int defaultMpiWorldSize = 5;
    int mpiBasePort = 8020;

    void initialise();
    void print() const;
};

SystemConfig& getSystemConfig();

// ----------------------------- gids ----------------------------------------

// Globally-unique-ish positive int32 ids for messages/apps/groups
// (reference: src/util/gids.cpp:16-28 — random key + atomic counter).
uint32_t generateGid();
int32_t generateGidInt32();

// ----------------------------- time ----------------------------------------

int64_t getGlobalClockEpochMillis();
int64_t getEpochMicros();
double getSecondsSinceEpoch();

// ----------------------------- strings/bytes -------------------------------

bool startsWith(const std::string& value, const std::string& prefix);
bool endsWith(const std::string& value, const std::string& suffix);
std::string randomString(size_t len);

inline std::vector<uint8_t> stringToBytes(const std::string& s)
{
    return { s.begin(), s.end() };
}
inline std::string bytesToString(const std::vector<uint8_t>& b)
{
    return { b.begin(), b.end() };
}

template<typename T>
void appendBytesOf(std::vector<uint8_t>& out, const T& v)
{
    const auto* p = reinterpret_cast<const uint8_t*>(&v);
    out.insert(out.end(), p, p + sizeof(T));
}

template<typename T>
T readBytesAt(const uint8_t* data, size_t offset)
{
    T v;
    std::memcpy(&v, data + offset, sizeof(T));
    return v;
}

// ----------------------------- network -------------------------------------

// IP of this host on its primary interface (reference: util/network.h:8-10).
std::string getPrimaryIPForThisHost(const std::string& interface = "");

// ----------------------------- testing -------------------------------------

// Mock mode: RPC clients record instead of sending
// (reference: src/util/testing.cpp:7-25).
void setMockMode(bool value);
bool isMockMode();
void setTestMode(bool value);
bool isTestMode();

// ----------------------------- keys ----------------------------------------

std::string funcToString(const std::string& user,
                         const std::string& function,
                         int messageId);

// Main-thread snapshot key "<user>/<func>_<appId>"
// (reference: src/util/func.cpp:152-159).
std::string getMainThreadSnapshotKey(const std::string& user,
                                     const std::string& function,
                                     int appId);

} // namespace faabricamd
