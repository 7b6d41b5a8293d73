// Per-host scheduler: executor pool + function-call RPC + reaper
// (reference: include/faabric/scheduler/Scheduler.h:41,
//  src/scheduler/Scheduler.cpp:48-448,
//  src/scheduler/FunctionCallServer.cpp:21-95).
#pragma once

#include <memory>
#include <mutex>
#include <shared_mutex>
#include <unordered_map>
#include <vector>

#include "faabricamd/executor.h"
#include "faabricamd/messages.h"
#include "faabricamd/queue.h"
#include "faabricamd/transport.h"

namespace faabricamd {

// RPC call codes (reference: scheduler/FunctionCallApi.h:4-10)
enum class FunctionCalls : uint8_t
{
    ExecuteFunctions = 1,
    Flush = 2,
    SetMessageResult = 3,
    BatchDone = 4,
};

class Scheduler
{
  public:
    static Scheduler& get();

    // Execute a batch scheduled to this host
    // (reference: src/scheduler/Scheduler.cpp:250)
    void executeBatch(std::shared_ptr<BatchExecuteRequest> req);

    // Resources advertised to the planner
    HostResources getThisHostResources();
    void setThisHostResources(const HostResources& res);

    // Periodic reaping of idle warm executors
    // (reference: src/scheduler/Scheduler.cpp:166-241)
    int reapStaleExecutors();
    void startReaper();
    void stopReaper();

    // Migration check entry point called periodically by long-running apps
    // (reference: src/scheduler/Scheduler.cpp:448)
    std::shared_ptr<PendingMigration> checkForMigrationOpportunities(
      Message& msg,
      int32_t overwriteNewGroupId = 0);

    void flushLocally();
    void shutdown();
    void reset();

    // Test-mode recording (reference: src/scheduler/Scheduler.cpp:263-268)
    std::vector<Message> getRecordedMessages();
    void clearRecordedMessages();

    size_t getExecutorCount();

  private:
    Scheduler();

    std::shared_ptr<Executor> claimExecutor(Message& msg);

    // Claims take the lock SHARED (concurrent tryClaim CAS scans from
    // the dispatcher threads — a plain mutex here measured ~41 us of
    // wall per claim from contention at 8 dispatchers x 128 messages);
    // executor creation and reaping take it unique.
    std::shared_mutex schedMx;
    struct WarmPool
    {
        std::vector<std::shared_ptr<Executor>> list;
        // Rotating claim hint: first-fit from index 0 costs O(N^2)
        // probes per N-message batch; rotating makes it ~O(1)
        std::atomic<size_t> hint{ 0 };
    };
    std::unordered_map<std::string, std::unique_ptr<WarmPool>> executors;
    HostResources overriddenResources;
    bool resourcesOverridden = false;

    std::vector<Message> recordedMessages;

    class ReaperThread;
    std::shared_ptr<ReaperThread> reaper;
};

// True when this process runs a FunctionCallServer (i.e. BATCH_DONE
// pushes from the planner can actually reach it)
bool functionCallServerRunning();

class FunctionCallServer : public MessageEndpointServer
{
  public:
    FunctionCallServer();
    ~FunctionCallServer() override;
    void doAsyncRecv(uint8_t code,
                     const std::string& body,
                     uint32_t seq) override;
    std::string doSyncRecv(uint8_t code, const std::string& body) override;
};

class FunctionCallClient : public MessageEndpointClient
{
  public:
    explicit FunctionCallClient(const std::string& host);
    void executeFunctions(const BatchExecuteRequest& req);
    void setMessageResult(const Message& msg);
    void sendFlush();
    void batchDone(int32_t appId);
};

std::shared_ptr<FunctionCallClient> getFunctionCallClient(
  const std::string& host);

// Local BATCH_DONE flag registry (planner pushes, PlannerClient::
// waitBatchDone sleeps). prepare() before registering with the planner
// so a push can never be missed; discard() cleans up after the wait.
std::shared_ptr<FlagWaiter> batchDoneWaiterPrepare(int32_t appId);
void batchDoneWaiterDiscard(int32_t appId);
void signalBatchDone(int32_t appId);
void clearFunctionCallClients();

// Mock-mode recording (reference: src/scheduler/FunctionCallClient.cpp:14-99)
std::vector<std::pair<std::string, BatchExecuteRequest>>
getBatchRequestsSentMock();
void clearMockedFunctionCalls();

} // namespace faabricamd
