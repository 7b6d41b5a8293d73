// Execution layer: Executor / ExecutorFactory / ExecutorContext and the
// per-host function registry.
//
// MI355X-native re-design of the reference executor
// (reference: include/faabric/executor/Executor.h:21-81,
//  ExecutorFactory.h:7-24, ExecutorContext.h:22-63,
//  src/executor/Executor.cpp). Differences by design:
//  - an Executor binds a HIP device (one executor slot per GPU on an
//    8xMI355X node) instead of a WASM memory; getMemoryView() exposes the
//    executor's HBM-backed snapshot arena
//  - the default executor dispatches to a (user, function) registry of
//    native C++ callables (Python callables can also be registered through
//    the bindings); embedders may still subclass Executor like in the
//    reference
#pragma once

#include <atomic>
#include <functional>
#include <memory>
#include <string>
#include <thread>
#include <vector>

#include "faabricamd/messages.h"
#include "faabricamd/queue.h"
#include "faabricamd/dirty.h"
#include "faabricamd/snapshot.h"

namespace faabricamd {

class Executor;

// ----------------------------- context -------------------------------------

// Thread-local execution context (reference: executor/ExecutorContext.h:22-63)
class ExecutorContext
{
  public:
    static bool isSet();
    static void set(Executor* executor,
                    std::shared_ptr<BatchExecuteRequest> req,
                    int msgIdx);
    static void unset();
    static ExecutorContext& get();

    Executor* getExecutor() const { return executor; }
    std::shared_ptr<BatchExecuteRequest> getBatchRequest() const
    {
        return req;
    }
    Message& getMsg();
    int getMsgIdx() const { return msgIdx; }

  private:
    Executor* executor = nullptr;
    std::shared_ptr<BatchExecuteRequest> req;
    int msgIdx = 0;
};

// ----------------------------- registry ------------------------------------

using FaabricFunction = std::function<int32_t(Message&)>;

class FunctionRegistry
{
  public:
    static FunctionRegistry& get();
    void registerFunction(const std::string& user,
                          const std::string& function,
                          FaabricFunction fn);
    FaabricFunction* getFunction(const std::string& user,
                                 const std::string& function);
    void clear();

  private:
    std::mutex mx;
    std::map<std::string, FaabricFunction> functions;
};

// ----------------------------- executor ------------------------------------

struct ExecutorTask
{
    ExecutorTask() = default;
    ExecutorTask(int msgIdxIn, std::shared_ptr<BatchExecuteRequest> reqIn)
      : msgIdx(msgIdxIn)
      , req(std::move(reqIn))
    {}
    int msgIdx = 0;
    std::shared_ptr<BatchExecuteRequest> req;
    bool stop = false;
};

class Executor
{
  public:
    explicit Executor(Message& msg);
    virtual ~Executor();

    // Dispatch tasks to the pool (reference: src/executor/Executor.cpp:111)
    void executeTasks(std::vector<int> msgIdxs,
                      std::shared_ptr<BatchExecuteRequest> req);

    // Fork-join THREADS batch over a shared memory snapshot (reference:
    // Executor::executeThreads; call stack SURVEY §3.4). Must be called
    // from a running task (ExecutorContext set). Snapshots this
    // executor's memory, gang-schedules `req` as THREADS, waits for all
    // thread results, merges the typed diffs back and re-maps the merged
    // snapshot over this executor's memory. Returns (msgId, returnValue).
    std::vector<std::pair<int32_t, int32_t>> executeThreads(
      std::shared_ptr<BatchExecuteRequest> req,
      const std::vector<SnapshotMergeRegion>& mergeRegions);

    // User hook: run one task. Default implementation dispatches to the
    // FunctionRegistry (reference: Executor.h:42 pure virtual)
    virtual int32_t executeTask(int threadPoolIdx,
                                int msgIdx,
                                std::shared_ptr<BatchExecuteRequest> req);

    // Memory / snapshot hooks (reference: Executor.h:50-81). The base
    // implementation manages a host-visible arena AND an optional
    // HBM-resident arena; once the device arena is set, THREADS fork-join
    // snapshots/diffs run on the gfx950 kernels.
    virtual std::pair<uint8_t*, size_t> getMemoryView();
    virtual void setMemorySize(size_t newSize);
    void setDeviceMemorySize(size_t newSize); // rounds up to 4 KiB pages
    std::pair<uint8_t*, size_t> getDeviceMemoryView();
    bool hasDeviceArena() const { return deviceArena != nullptr; }
    virtual void restore(const std::string& snapshotKey);
    virtual void reset(Message& msg);
    virtual void flush();

    // Claim lifecycle (reference: src/executor/Executor.cpp:580-590)
    bool tryClaim();
    void claim();
    void releaseClaim();
    bool isClaimed() const { return claimed.load(); }

    long getMillisSinceLastExec() const;
    void shutdown();

    std::string id;
    Message boundMsg;
    // HIP device this executor is bound to (-1 = no GPU on this host)
    int gpuDevice = -1;

    // THREADS support: merge dirty regions and ship thread results
    // (wired in the snapshot phase)
    void setThreadResult(Message& msg,
                         int32_t returnValue,
                         const std::string& key,
                         const std::vector<SnapshotDiffMsg>& diffs);

  protected:
    void threadPoolThread(int poolIdx);
    void handleTaskResult(Message& msg,
                          int32_t returnValue,
                          std::shared_ptr<BatchExecuteRequest> req,
                          bool isLastInBatch);

    int threadPoolSize = 0;
    std::vector<std::shared_ptr<Queue<ExecutorTask>>> threadTaskQueues;
    std::vector<std::thread> threadPoolThreads;
    std::mutex threadsMx;

    std::atomic<bool> claimed{ false };
    std::atomic<int64_t> lastExecMs{ 0 };

    // Fork-join snapshot lifecycle: the main-thread snapshot (keyed
    // "<user>/<function>_<appId>") is kept for same-app repeat forks and
    // deleted — locally and on the hosts it was shipped to — when a fork
    // for a DIFFERENT app starts or the executor is destroyed
    std::string lastForkSnapshotKey;
    std::vector<std::string> lastForkRemoteHosts;
    void cleanupForkSnapshot();

    // Batch accounting: tasks remaining in the current batch
    std::shared_ptr<std::atomic<int>> batchCounter;

    // Executor-local arena for THREADS snapshots (host path);
    // page-aligned so the segfault dirty tracker can mprotect it
    PageAlignedBuffer dummyMemory;

    // HBM-resident arena (device THREADS path)
    uint8_t* deviceArena = nullptr;
    size_t deviceArenaSize = 0;
};

class ExecutorFactory
{
  public:
    virtual ~ExecutorFactory() = default;
    virtual std::shared_ptr<Executor> createExecutor(Message& msg)
    {
        return std::make_shared<Executor>(msg);
    }
    virtual void flushHost() {}
};

void setExecutorFactory(std::shared_ptr<ExecutorFactory> factory);
std::shared_ptr<ExecutorFactory> getExecutorFactory();

// Function chaining from inside a running task (reference:
// Executor::addChainedMessage + faasm chaining host interface)
int32_t chainFunction(const std::string& user,
                      const std::string& function,
                      const std::vector<uint8_t>& input);
Message awaitChainedCall(int32_t msgId, int timeoutMs = 60000);

// Migration / freeze point for long-running (gang) functions. Returns 0
// to continue, or the MIGRATED/FROZEN sentinel the function must return
// immediately (cpp/src/migration.cpp)
int32_t migrationPoint(const std::vector<uint8_t>& reentryInput);

} // namespace faabricamd
