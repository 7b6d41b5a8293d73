// Batch scheduling: decision objects and pluggable placement policies.
//
// Behavioral parity with the reference's batch-scheduler layer
// (reference: include/faabric/batch-scheduler/BatchScheduler.h:8-131,
//  SchedulingDecision.h:58-118, src/batch-scheduler/BinPackScheduler.cpp,
//  CompactScheduler.cpp, SpotScheduler.cpp) — re-implemented for the
// MI355X deployment shape where a "slot" is one GPU and a "host" is one
// GPU-owning worker process.
#pragma once

#include <map>
#include <mutex>
#include <memory>
#include <set>
#include <string>
#include <vector>

#include "faabricamd/messages.h"

namespace faabricamd {

// Sentinels (reference: batch-scheduler/BatchScheduler.h:8-19)
inline constexpr int32_t DO_NOT_MIGRATE = -98;
inline constexpr int32_t NOT_ENOUGH_SLOTS = -99;
inline constexpr int32_t MUST_FREEZE = -97;
inline const std::string MUST_EVICT_IP = "E.VI.CT.ME";

// Preloaded-gang-decision magic groupId (reference: src/planner/Planner.cpp:20-22)
inline constexpr int32_t FIXED_SIZE_PRELOADED_DECISION_GROUPID = -99;

enum class DecisionType
{
    NO_DECISION_TYPE = 0,
    NEW = 1,
    DIST_CHANGE = 2,
    SCALE_CHANGE = 3,
};

class SchedulingDecision
{
  public:
    SchedulingDecision() = default;
    SchedulingDecision(int32_t appIdIn, int32_t groupIdIn)
      : appId(appIdIn)
      , groupId(groupIdIn)
    {}

    static SchedulingDecision fromPointToPointMappings(
      const PointToPointMappings& mappings);
    PointToPointMappings toPointToPointMappings() const;

    bool operator==(const SchedulingDecision& rhs) const
    {
        return appId == rhs.appId && groupId == rhs.groupId &&
               hosts == rhs.hosts && messageIds == rhs.messageIds;
    }

    int32_t appId = 0;
    int32_t groupId = 0;
    int32_t nFunctions = 0;
    std::vector<std::string> hosts;
    std::vector<int32_t> messageIds;
    std::vector<int32_t> appIdxs;
    std::vector<int32_t> groupIdxs;
    std::vector<int32_t> mpiPorts;
    std::string returnHost;

    bool isSingleHost() const;
    void addMessage(const std::string& host, const Message& msg);
    void addMessage(const std::string& host,
                    int32_t messageId,
                    int32_t appIdx,
                    int32_t groupIdx);
    void addMessageInPosition(int32_t pos,
                              const std::string& host,
                              int32_t messageId,
                              int32_t appIdx,
                              int32_t groupIdx,
                              int32_t mpiPort);
    // Returns the vacated MPI port
    int32_t removeMessage(int32_t messageId);
    std::set<std::string> uniqueHosts() const;
    void print() const;
};

// Host state used during a scheduling pass
struct HostState
{
    HostState(std::string ipIn, int slotsIn, int usedSlotsIn)
      : ip(std::move(ipIn))
      , slots(slotsIn)
      , usedSlots(usedSlotsIn)
    {}
    std::string ip;
    int slots;
    int usedSlots;
};

using HostMap = std::map<std::string, std::shared_ptr<HostState>>;
using InFlightPair = std::pair<std::shared_ptr<BatchExecuteRequest>,
                               std::shared_ptr<SchedulingDecision>>;
using InFlightReqs = std::map<int32_t, InFlightPair>;

class BatchScheduler
{
  public:
    virtual ~BatchScheduler() = default;

    static DecisionType getDecisionType(
      const InFlightReqs& inFlightReqs,
      const BatchExecuteRequest& req);

    virtual std::shared_ptr<SchedulingDecision> makeSchedulingDecision(
      HostMap& hostMap,
      const InFlightReqs& inFlightReqs,
      const BatchExecuteRequest& req) = 0;

  protected:
    static int numSlotsAvailable(const std::shared_ptr<HostState>& h)
    {
        return std::max(0, h->slots - h->usedSlots);
    }
};

class BinPackScheduler : public BatchScheduler
{
  public:
    std::shared_ptr<SchedulingDecision> makeSchedulingDecision(
      HostMap& hostMap,
      const InFlightReqs& inFlightReqs,
      const BatchExecuteRequest& req) override;
};

// Compact: pack into as few hosts as possible, preferring already-occupied
// hosts (reference: src/batch-scheduler/CompactScheduler.cpp:339)
class CompactScheduler : public BatchScheduler
{
  public:
    std::shared_ptr<SchedulingDecision> makeSchedulingDecision(
      HostMap& hostMap,
      const InFlightReqs& inFlightReqs,
      const BatchExecuteRequest& req) override;
};

// Spot: like compact, but never schedules onto the next-evicted VM and
// returns MUST_FREEZE when an in-flight app cannot leave it
// (reference: src/batch-scheduler/SpotScheduler.cpp:258)
class SpotScheduler : public BatchScheduler
{
  public:
    std::shared_ptr<SchedulingDecision> makeSchedulingDecision(
      HostMap& hostMap,
      const InFlightReqs& inFlightReqs,
      const BatchExecuteRequest& req) override;
};

// Cache of past placements keyed (user, function, size), reused when the
// caller opts in (reference: src/batch-scheduler/DecisionCache.cpp —
// the CACHED topology hint)
class DecisionCache
{
  public:
    static DecisionCache& get();
    std::shared_ptr<SchedulingDecision> getCachedDecision(
      const BatchExecuteRequest& req);
    void addCachedDecision(const BatchExecuteRequest& req,
                           const SchedulingDecision& decision);
    void clear();
    size_t size();

  private:
    std::string keyOf(const BatchExecuteRequest& req);
    std::mutex mx;
    std::map<std::string, std::shared_ptr<SchedulingDecision>> cache;
};

std::shared_ptr<BatchScheduler> getBatchScheduler();
void resetBatchScheduler();
void resetBatchScheduler(const std::string& newMode);
std::string getBatchSchedulerMode();

// Shared helper: keep as many (host, message) pairs from the old decision
// as the new histogram allows (reference:
// src/batch-scheduler/BinPackScheduler.cpp:26-92)
std::shared_ptr<SchedulingDecision> minimiseNumOfMigrations(
  std::shared_ptr<SchedulingDecision> newDecision,
  std::shared_ptr<SchedulingDecision> oldDecision);

} // namespace faabricamd
