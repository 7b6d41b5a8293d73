#include "faabricamd/util.h"

#include <arpa/inet.h>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <ifaddrs.h>
#include <mutex>
#include <net/if.h>
#include <random>
#include <sys/types.h>
#include <thread>
#include <unistd.h>

namespace faabricamd {

// ----------------------------- logging ------------------------------------

static std::atomic<int> logLevelValue{ -1 };

static LogLevel parseLogLevel(const std::string& s)
{
    if (s == "trace") {
        return LogLevel::trace;
    }
    if (s == "debug") {
        return LogLevel::debug;
    }
    if (s == "info") {
        return LogLevel::info;
    }
    if (s == "warn" || s == "warning") {
        return LogLevel::warn;
    }
    if (s == "error") {
        return LogLevel::error;
    }
    if (s == "off") {
        return LogLevel::off;
    }
    return LogLevel::info;
}

LogLevel getLogLevel()
{
    int v = logLevelValue.load(std::memory_order_relaxed);
    if (v < 0) {
        v = (int)parseLogLevel(getEnvVar("LOG_LEVEL", "info"));
        logLevelValue.store(v, std::memory_order_relaxed);
    }
    return (LogLevel)v;
}

void setLogLevel(LogLevel lvl)
{
    logLevelValue.store((int)lvl, std::memory_order_relaxed);
}

void logMessage(LogLevel lvl, const char* fmt, ...)
{
    if ((int)lvl < (int)getLogLevel()) {
        return;
    }
    static const char* names[] = { "TRACE", "DEBUG", "INFO",
                                   "WARN",  "ERROR", "OFF" };
    char body[2048];
    va_list args;
    va_start(args, fmt);
    vsnprintf(body, sizeof(body), fmt, args);
    va_end(args);

    auto now = std::chrono::system_clock::now();
    auto ms = std::chrono::duration_cast<std::chrono::milliseconds>(
                now.time_since_epoch())
                .count();
    time_t secs = (time_t)(ms / 1000);
    struct tm tmv;
    localtime_r(&secs, &tmv);
    char ts[32];
    strftime(ts, sizeof(ts), "%H:%M:%S", &tmv);

    // Single fprintf keeps lines atomic-enough across threads
    fprintf(stderr,
            "[%s.%03d] [%lu] [%s] %s\n",
            ts,
            (int)(ms % 1000),
            (unsigned long)(std::hash<std::thread::id>{}(
                              std::this_thread::get_id()) %
                            100000),
            names[(int)lvl],
            body);
}

// ----------------------------- config --------------------------------------

std::string getEnvVar(const std::string& key, const std::string& deflt)
{
    const char* v = ::getenv(key.c_str());
    if (v == nullptr || *v == '\0') {
        return deflt;
    }
    return std::string(v);
}

int getEnvVarInt(const std::string& key, int deflt)
{
    const char* v = ::getenv(key.c_str());
    if (v == nullptr || *v == '\0') {
        return deflt;
    }
    return atoi(v);
}

int getUsableCores()
{
    int override = getEnvVarInt("OVERRIDE_CPU_COUNT", 0);
    if (override > 0) {
        return override;
    }
    unsigned int n = std::thread::hardware_concurrency();
    return n == 0 ? 1 : (int)n;
}

void SystemConfig::initialise()
{
    logLevel = getEnvVar("LOG_LEVEL", "info");
    endpointHost = getEnvVar("ENDPOINT_HOST", "");
    if (endpointHost.empty()) {
        endpointHost = getPrimaryIPForThisHost();
    }
    plannerHost = getEnvVar("PLANNER_HOST", "localhost");
    plannerPort = getEnvVarInt("PLANNER_PORT", 8011);

    batchSchedulerMode = getEnvVar("BATCH_SCHEDULER_MODE", "bin-pack");
    overrideCpuCount = getEnvVarInt("OVERRIDE_CPU_COUNT", 0);
    overrideGpuCount = getEnvVarInt("OVERRIDE_GPU_COUNT", -1);
    gpuDevice = getEnvVarInt("FAABRIC_GPU_DEVICE", 0);
    gpuDevicePinned = !getEnvVar("FAABRIC_GPU_DEVICE", "").empty();
    useGpu = getEnvVarInt("FAABRIC_USE_GPU", 1) != 0;

    globalMessageTimeout = getEnvVarInt("GLOBAL_MESSAGE_TIMEOUT", 60000);
    boundTimeout = getEnvVarInt("BOUND_TIMEOUT", 30000);

    functionServerThreads = getEnvVarInt("FUNCTION_SERVER_THREADS", 4);
    stateServerThreads = getEnvVarInt("STATE_SERVER_THREADS", 2);
    snapshotServerThreads = getEnvVarInt("SNAPSHOT_SERVER_THREADS", 2);
    pointToPointServerThreads = getEnvVarInt("POINT_TO_POINT_SERVER_THREADS", 4);
    plannerServerThreads = getEnvVarInt("PLANNER_SERVER_THREADS", 4);

    dirtyTrackingMode = getEnvVar("DIRTY_TRACKING_MODE", "compare");
    diffingMode = getEnvVar("DIFFING_MODE", "xor");

    stateMode = getEnvVar("STATE_MODE", "inmemory");

    defaultMpiWorldSize = getEnvVarInt("DEFAULT_MPI_WORLD_SIZE", 5);
    mpiBasePort = getEnvVarInt("MPI_BASE_PORT", 8020);
}

void SystemConfig::print() const
{
    FAM_INFO("--- faabric-mi355x config ---");
    FAM_INFO("endpointHost       %s", endpointHost.c_str());
    FAM_INFO("plannerHost        %s:%d", plannerHost.c_str(), plannerPort);
    FAM_INFO("batchScheduler     %s", batchSchedulerMode.c_str());
    FAM_INFO("dirtyTrackingMode  %s", dirtyTrackingMode.c_str());
    FAM_INFO("diffingMode        %s", diffingMode.c_str());
}

SystemConfig& getSystemConfig()
{
    static SystemConfig conf;
    static std::once_flag flag;
    std::call_once(flag, [&]() { conf.initialise(); });
    return conf;
}

// ----------------------------- gids ----------------------------------------

// Random base + atomic counter, folded into a positive int32
// (reference scheme: src/util/gids.cpp:16-28).
static std::atomic<uint64_t> gidCounter{ 0 };

static uint64_t gidKey()
{
    static uint64_t key = []() {
        std::random_device rd;
        std::mt19937_64 gen(rd() ^ (uint64_t)::getpid());
        return gen();
    }();
    return key;
}

uint32_t generateGid()
{
    // Random per-process base + atomic counter: unique within a process
    // for 2^31 draws, distinct across processes with high probability
    // (reference scheme: src/util/gids.cpp:16-28)
    uint64_t c = gidCounter.fetch_add(1, std::memory_order_relaxed);
    uint32_t r = (uint32_t)((gidKey() + c) % (uint64_t)INT32_MAX);
    return r == 0 ? 1 : r;
}

int32_t generateGidInt32()
{
    return (int32_t)generateGid();
}

// ----------------------------- time ----------------------------------------

int64_t getGlobalClockEpochMillis()
{
    return std::chrono::duration_cast<std::chrono::milliseconds>(
             std::chrono::system_clock::now().time_since_epoch())
      .count();
}

int64_t getEpochMicros()
{
    return std::chrono::duration_cast<std::chrono::microseconds>(
             std::chrono::system_clock::now().time_since_epoch())
      .count();
}

double getSecondsSinceEpoch()
{
    return (double)getEpochMicros() / 1e6;
}

// ----------------------------- strings -------------------------------------

bool startsWith(const std::string& value, const std::string& prefix)
{
    return value.rfind(prefix, 0) == 0;
}

bool endsWith(const std::string& value, const std::string& suffix)
{
    if (suffix.size() > value.size()) {
        return false;
    }
    return value.compare(value.size() - suffix.size(),
                         suffix.size(),
                         suffix) == 0;
}

std::string randomString(size_t len)
{
    static const char chars[] =
      "abcdefghijklmnopqrstuvwxyzABCDEFGHIJKLMNOPQRSTUVWXYZ0123456789";
    thread_local std::mt19937 gen(std::random_device{}());
    std::uniform_int_distribution<int> dist(0, sizeof(chars) - 2);
    std::string out(len, ' ');
    for (auto& c : out) {
        c = chars[dist(gen)];
    }
    return out;
}

// ----------------------------- network -------------------------------------

std::string getPrimaryIPForThisHost(const std::string& interface)
{
    struct ifaddrs* allAddrs = nullptr;
    if (getifaddrs(&allAddrs) == -1) {
        return "127.0.0.1";
    }

    std::string result = "127.0.0.1";
    for (struct ifaddrs* ifa = allAddrs; ifa != nullptr;
         ifa = ifa->ifa_next) {
        if (ifa->ifa_addr == nullptr ||
            ifa->ifa_addr->sa_family != AF_INET) {
            continue;
        }
        if ((ifa->ifa_flags & IFF_LOOPBACK) != 0) {
            continue;
        }
        if (!interface.empty() && interface != ifa->ifa_name) {
            continue;
        }
        char buf[INET_ADDRSTRLEN];
        auto* sa = reinterpret_cast<struct sockaddr_in*>(ifa->ifa_addr);
        if (inet_ntop(AF_INET, &sa->sin_addr, buf, sizeof(buf)) != nullptr) {
            result = buf;
            break;
        }
    }
    freeifaddrs(allAddrs);
    return result;
}

// ----------------------------- testing -------------------------------------

static std::atomic<bool> mockMode{ false };
static std::atomic<bool> testMode{ false };

void setMockMode(bool value)
{
    mockMode.store(value);
}
bool isMockMode()
{
    return mockMode.load();
}
void setTestMode(bool value)
{
    testMode.store(value);
}
bool isTestMode()
{
    return testMode.load();
}

// ----------------------------- keys ----------------------------------------

std::string funcToString(const std::string& user,
                         const std::string& function,
                         int messageId)
{
    std::string s = user + "/" + function;
    if (messageId != 0) {
        s += ":" + std::to_string(messageId);
    }
    return s;
}

std::string getMainThreadSnapshotKey(const std::string& user,
                                     const std::string& function,
                                     int appId)
{
    // Format kept identical to the reference (src/util/func.cpp:152-159)
    return user + "/" + function + "_" + std::to_string(appId);
}

} // namespace faabricamd
