// Control plane: the planner owns cluster membership and every scheduling
// decision (reference: include/faabric/planner/Planner.h:98-133,
// PlannerState.h:14-57, src/planner/Planner.cpp). Re-designed around the
// MI355X deployment: a "host" is one GPU-owning worker process (identified
// by "ip" or "ip@portOffset" so 8 workers can share one node's IP), and a
// slot is one GPU.
#pragma once

#include <deque>
#include <atomic>
#include <map>
#include <memory>
#include <set>
#include <shared_mutex>
#include <thread>
#include <vector>

#include "faabricamd/messages.h"
#include "faabricamd/queue.h"
#include "faabricamd/scheduling.h"
#include "faabricamd/transport.h"

namespace faabricamd {

// RPC call codes (reference: planner/PlannerApi.h:4-21)
enum class PlannerCalls : uint8_t
{
    Ping = 1,
    GetAvailableHosts = 2,
    RegisterHost = 3,
    RemoveHost = 4,
    SetMessageResult = 8,
    // Natural batching of result reports: when many executors finish in
    // a burst, their results coalesce into one RPC while the previous
    // one is in flight (group commit; single results flush immediately)
    SetMessageResultBatch = 15,
    GetMessageResult = 9,
    GetBatchResults = 10,
    GetSchedulingDecision = 11,
    GetNumMigrations = 12,
    CallBatch = 13,
    PreloadSchedulingDecision = 14,
    // Event-driven batch completion: the caller registers once and the
    // planner pushes BATCH_DONE to its function-call server when the
    // last result lands — replaces status polling (which contended with
    // result ingestion on the planner lock)
    WaitBatchDone = 16,
};

// Number of MPI data-plane ports in each host's pool
// (reference: src/planner/Planner.cpp:91-121)
inline constexpr int NUM_MPI_PORTS_PER_HOST = 64;

struct PlannerHost
{
    Host info;
    std::vector<bool> mpiPortUsed =
      std::vector<bool>(NUM_MPI_PORTS_PER_HOST, false);
};

struct PlannerState
{
    std::map<std::string, std::shared_ptr<PlannerHost>> hostMap;
    InFlightReqs inFlightReqs;
    std::map<int32_t, std::shared_ptr<SchedulingDecision>>
      preloadedSchedulingDecisions;
    // appId → msgId → result
    std::map<int32_t, std::map<int32_t, std::shared_ptr<Message>>> appResults;
    // Completed apps in completion order (epoch ms, appId): appResults
    // entries are purged after FAABRIC_RESULT_TTL_MS or beyond
    // FAABRIC_MAX_DONE_APPS, whichever hits first — without this a
    // long-running planner leaks every result ever produced
    std::deque<std::pair<int64_t, int32_t>> doneApps;
    // msgId → hosts waiting for the result push
    std::map<int32_t, std::vector<std::string>> appResultWaiters;
    std::map<int32_t, std::shared_ptr<BatchExecuteRequest>> evictedRequests;
    // appId → hosts to push BATCH_DONE to when the app's last result lands
    std::map<int32_t, std::vector<std::string>> batchDoneWaiters;
    // appId → (groupId, hosts) of placements superseded by migrations;
    // cleared (with their PTP state) when the app completes — mid-flight
    // clearing could destroy unconsumed verdict messages
    std::map<int32_t,
             std::vector<std::pair<int32_t, std::vector<std::string>>>>
      supersededGroups;
    std::set<std::string> nextEvictedHostIps;
    int numMigrations = 0;
};

class Planner
{
  public:
    static Planner& get();

    PlannerConfig getConfig();
    void printConfig();

    // --- membership ---
    bool registerHost(const Host& hostIn, bool overwrite);
    void removeHost(const Host& hostIn);
    std::vector<Host> getAvailableHosts();

    size_t debugAppResultsCount();
    size_t debugDoneAppsCount();
    size_t debugInFlightCount();

    // Returns true if the app is already complete (or unknown); else
    // records host for a BATCH_DONE push on completion
    bool registerBatchDoneWaiter(int32_t appId, const std::string& host);

    // --- scheduling ---
    std::shared_ptr<SchedulingDecision> callBatch(
      std::shared_ptr<BatchExecuteRequest> req);
    void dispatchSchedulingDecision(
      std::shared_ptr<BatchExecuteRequest> req,
      std::shared_ptr<SchedulingDecision> decision);
    void preloadSchedulingDecision(
      int32_t appId,
      std::shared_ptr<SchedulingDecision> decision);
    std::shared_ptr<SchedulingDecision> getSchedulingDecision(int32_t appId);

    // --- results ---
    void setMessageResult(std::shared_ptr<Message> msg);
    // Burst ingestion: one planner-lock acquisition for many results
    void setMessageResults(std::vector<std::shared_ptr<Message>>& msgs);
    // nullptr when not (yet) available; registers msg.mainHost as a waiter
    std::shared_ptr<Message> getMessageResult(const Message& msg);
    // nullptr when the app is unknown
    std::shared_ptr<BatchExecuteRequestStatus> getBatchResults(int32_t appId);

    // --- ops / introspection ---
    int getNumMigrations();
    GetInFlightAppsResponse getInFlightApps();
    void setNextEvictedVm(const std::set<std::string>& vmIps);
    void setPolicy(const std::string& policy);
    std::string getPolicy();

    // --- lifecycle ---
    void reset();          // clear everything incl. hosts
    void flushExecutors(); // tell every host to flush
    void flushSchedulingState();

    int hostTimeoutMs = 5000;

  private:
    void purgeOldResultsLocked();
    void setMessageResultLocked(
      const std::shared_ptr<Message>& msg,
      std::vector<std::pair<std::string, std::shared_ptr<Message>>>& waiters,
      std::vector<std::pair<std::string, int32_t>>& batchWaiters,
      std::vector<std::pair<std::string, int32_t>>& groupClears);

    Planner();

    std::shared_mutex plannerMx;
    PlannerState state;

    bool isHostExpired(const PlannerHost& host, int64_t nowMs) const;
    void flushHosts();
    std::shared_ptr<SchedulingDecision> getPreloadedSchedulingDecision(
      int32_t appId,
      const BatchExecuteRequest& req);
    int32_t claimHostMpiPort(std::shared_ptr<PlannerHost>& host);
    void releaseHostMpiPort(std::shared_ptr<PlannerHost>& host, int32_t port);
};

// ------------------------- server / client ----------------------------------

class PlannerServer : public MessageEndpointServer
{
  public:
    PlannerServer();
    ~PlannerServer() override;
    void doAsyncRecv(uint8_t code,
                     const std::string& body,
                     uint32_t seq) override;
    std::string doSyncRecv(uint8_t code, const std::string& body) override;

  private:
    // Result-ingestion pool: SetMessageResult frames from one worker
    // connection would otherwise decode+apply serially (~17 us each, the
    // wall-clock floor of a large batch). Results of distinct messages
    // commute, so a few workers drain them concurrently.
    void resultWorkerLoop();
    Queue<std::string> resultQueue;
    std::vector<std::thread> resultWorkers;
    std::atomic<bool> resultWorkersStop{ false };
};

// Per-process client with local result cache; results are pushed to waiting
// hosts by the planner through the worker's FunctionCallServer
// (reference: planner/PlannerClient.h:21-120)
class PlannerClient
{
  public:
    PlannerClient();
    ~PlannerClient();

    void ping();
    std::vector<Host> getAvailableHosts();
    int32_t registerHost(const Host& host, bool overwrite);
    void removeHost(const Host& host);

    // Entry point every caller uses to run a batch
    std::shared_ptr<SchedulingDecision> callFunctions(
      std::shared_ptr<BatchExecuteRequest> req);

    void setMessageResult(std::shared_ptr<Message> msg);
    // Burst ingestion: one planner-lock acquisition for many results
    void setMessageResults(std::vector<std::shared_ptr<Message>>& msgs);
    // One RPC carrying many results (see PlannerCalls::SetMessageResultBatch)
    void setMessageResultsBatch(
      const std::vector<std::shared_ptr<Message>>& msgs);
    // Called by the worker's FunctionCallServer when the planner pushes a
    // result to this host
    void setMessageResultLocally(std::shared_ptr<Message> msg);

    Message getMessageResult(const Message& msg, int timeoutMs);
    Message getMessageResult(int32_t appId, int32_t msgId, int timeoutMs);
    BatchExecuteRequestStatus getBatchResults(int32_t appId);
    // Lightweight poll: finished flag + result count only
    std::pair<bool, int> getBatchStatusCounts(int32_t appId);

    // Block until appId's batch fully completes. Event-driven: registers
    // with the planner for a BATCH_DONE push and sleeps on a local flag;
    // falls back to a coarse status poll so a lost push cannot hang the
    // caller. Returns false on timeout.
    bool waitBatchDone(int32_t appId, int timeoutMs);

    SchedulingDecision getSchedulingDecision(int32_t appId);
    void preloadSchedulingDecision(
      int32_t appId,
      const SchedulingDecision& decision);
    int getNumMigrations();

    void startKeepAlive();
    void stopKeepAlive();
    void clearCache();

  private:
    MessageEndpointClient rpc;

    std::mutex resultsMx;
    std::condition_variable resultsCv;
    std::map<int32_t, std::shared_ptr<Message>> localResults;

    class KeepAliveThread;
    std::shared_ptr<KeepAliveThread> keepAlive;
};

PlannerClient& getPlannerClient();
void resetPlannerClient();

} // namespace faabricamd
