// Planner implementation (reference behavior: src/planner/Planner.cpp —
// callBatch :808-1290, dispatch :1293-1390, setMessageResult :394-540,
// getMessageResult :545-590, preload :592-650, getBatchResults :650-700,
// membership :166-360; src/planner/PlannerServer.cpp:22-182;
// src/planner/PlannerClient.cpp). Fresh implementation, see planner.h.
#include "faabricamd/planner.h"
#include "faabricamd/utilextras.h"
#include "faabricamd/ptp.h"
#include "faabricamd/scheduler.h"
#include "faabricamd/ops.h"
#include "faabricamd/snapshot.h"
#include "faabricamd/state.h"
#include "faabricamd/util.h"
#include "faabricamd/wire.h"

#include <string_view>
#include <algorithm>
#include <cassert>

namespace faabricamd {

Planner::Planner()
{
    hostTimeoutMs = getEnvVarInt("PLANNER_HOST_KEEPALIVE_MS", 5000);
}

Planner& Planner::get()
{
    static Planner planner;
    return planner;
}

PlannerConfig Planner::getConfig()
{
    PlannerConfig conf;
    conf.ip = getSystemConfig().endpointHost;
    conf.hostTimeout = hostTimeoutMs / 1000;
    conf.numThreadsHttpServer = 4;
    return conf;
}

void Planner::printConfig()
{
    FAM_INFO("planner config: hostTimeoutMs=%d", hostTimeoutMs);
}

// ----------------------------- membership ----------------------------------

bool Planner::isHostExpired(const PlannerHost& host, int64_t nowMs) const
{
    return (nowMs - host.info.registerTsEpochMs) > hostTimeoutMs;
}

bool Planner::registerHost(const Host& hostIn, bool overwrite)
{
    if (hostIn.ip.empty() || hostIn.slots < 0) {
        return false;
    }
    if (overwrite) {
        // A host (re-)registering from scratch may have restarted: cached
        // outbound connections to it are silently stale (async frames into
        // a dead socket do not error), so drop them all
        clearFunctionCallClients();
        clearSnapshotClients();
        clearStateClients();
        getPointToPointBroker().clearClients();
    }
    std::unique_lock lock(plannerMx);
    auto it = state.hostMap.find(hostIn.ip);
    if (it == state.hostMap.end() || overwrite) {
        auto h = std::make_shared<PlannerHost>();
        h->info = hostIn;
        h->info.usedSlots = 0;
        h->info.registerTsEpochMs = getGlobalClockEpochMillis();
        state.hostMap[hostIn.ip] = h;
        FAM_DEBUG("planner registered host %s (slots=%d)",
                  hostIn.ip.c_str(),
                  hostIn.slots);
    } else {
        // Keep-alive: refresh the timestamp (and slot count, not usage)
        it->second->info.registerTsEpochMs = getGlobalClockEpochMillis();
        it->second->info.slots = hostIn.slots;
    }
    return true;
}

void Planner::removeHost(const Host& hostIn)
{
    std::unique_lock lock(plannerMx);
    state.hostMap.erase(hostIn.ip);
}

std::vector<Host> Planner::getAvailableHosts()
{
    std::unique_lock lock(plannerMx);
    int64_t nowMs = getGlobalClockEpochMillis();
    std::vector<Host> out;
    for (auto it = state.hostMap.begin(); it != state.hostMap.end();) {
        if (isHostExpired(*it->second, nowMs)) {
            FAM_WARN("planner expiring host %s", it->first.c_str());
            it = state.hostMap.erase(it);
        } else {
            out.push_back(it->second->info);
            ++it;
        }
    }
    return out;
}

// ----------------------------- ports ---------------------------------------

int32_t Planner::claimHostMpiPort(std::shared_ptr<PlannerHost>& host)
{
    for (int i = 0; i < NUM_MPI_PORTS_PER_HOST; i++) {
        if (!host->mpiPortUsed[i]) {
            host->mpiPortUsed[i] = true;
            return getSystemConfig().mpiBasePort + i;
        }
    }
    throw FaabricException("no free MPI ports on host " + host->info.ip);
}

void Planner::releaseHostMpiPort(std::shared_ptr<PlannerHost>& host,
                                 int32_t port)
{
    int idx = port - getSystemConfig().mpiBasePort;
    if (idx >= 0 && idx < NUM_MPI_PORTS_PER_HOST) {
        host->mpiPortUsed[idx] = false;
    }
}

// ----------------------------- scheduling ----------------------------------

std::shared_ptr<SchedulingDecision> Planner::getPreloadedSchedulingDecision(
  int32_t appId,
  const BatchExecuteRequest& req)
{
    // Filter to the group idxs present in this BER, keeping the BER's
    // message ids (reference: src/planner/Planner.cpp:610-650)
    auto full = state.preloadedSchedulingDecisions.at(appId);
    auto filtered =
      std::make_shared<SchedulingDecision>(full->appId, full->groupId);
    for (const auto& msg : req.messages) {
        auto it = std::find(
          full->groupIdxs.begin(), full->groupIdxs.end(), msg.groupIdx);
        if (it == full->groupIdxs.end()) {
            throw FaabricException("preloaded decision missing group idx " +
                                   std::to_string(msg.groupIdx));
        }
        size_t i = (size_t)std::distance(full->groupIdxs.begin(), it);
        filtered->addMessage(
          full->hosts[i], msg.id, full->appIdxs[i], full->groupIdxs[i]);
        filtered->mpiPorts[filtered->nFunctions - 1] = full->mpiPorts[i];
    }
    return filtered;
}

std::shared_ptr<SchedulingDecision> Planner::callBatch(
  std::shared_ptr<BatchExecuteRequest> req)
{
    int32_t appId = req->appId;
    std::unique_lock lock(plannerMx);

    auto scheduler = getBatchScheduler();
    auto decisionType =
      BatchScheduler::getDecisionType(state.inFlightReqs, *req);
    bool isNew = decisionType == DecisionType::NEW;
    bool isScaleChange = decisionType == DecisionType::SCALE_CHANGE;
    bool isDistChange = decisionType == DecisionType::DIST_CHANGE;
    bool isMpi = !req->messages.empty() && req->messages[0].isMpi;
    bool existsPreloadedDec =
      state.preloadedSchedulingDecisions.count(appId) > 0;

    // Copy of the host map for the policy to scribble on; doomed VMs get
    // tagged for the SPOT policy
    HostMap hostMapCopy;
    for (const auto& [ip, h] : state.hostMap) {
        hostMapCopy[ip] = std::make_shared<HostState>(
          state.nextEvictedHostIps.count(ip) > 0 ? MUST_EVICT_IP : ip,
          h->info.slots,
          h->info.usedSlots);
    }

    // Elastic scale-up of a THREADS fork to all free slots on the main host
    // (reference: src/planner/Planner.cpp:832-891)
    if (isScaleChange && req->elasticScaleHint && !existsPreloadedDec) {
        auto oldDec = state.inFlightReqs.at(appId).second;
        const std::string& mainHost = oldDec->hosts.at(0);
        int numAvail = std::max(
          0,
          hostMapCopy.at(mainHost)->slots -
            hostMapCopy.at(mainHost)->usedSlots);
        int numRequested = (int)req->messages.size();
        int lastIdx =
          numRequested == 0 ? 0 : req->messages.back().groupIdx;
        for (int i = 0; i < numAvail - numRequested; i++) {
            int newIdx = lastIdx + i + 1;
            Message newMsg;
            if (numRequested == 0) {
                newMsg = state.inFlightReqs.at(appId).first->messages.at(0);
                newMsg.mainHost = mainHost;
                newMsg.funcPtr = req->groupId;
            } else {
                newMsg = req->messages.back();
            }
            newMsg.appIdx = newIdx;
            newMsg.groupIdx = newIdx;
            newMsg.id = generateGidInt32();
            req->messages.push_back(std::move(newMsg));
        }
    }

    // A migration re-schedules the same in-flight messages
    if (isDistChange) {
        auto oldReq = state.inFlightReqs.at(appId).first;
        req->subType = oldReq->subType;
        req->messages = oldReq->messages;
    }

    // NEW MPI requests gang-schedule the whole world up-front: schedule a
    // known-size request now, hand back only the first message, preload the
    // rest (reference: src/planner/Planner.cpp:960-1000)
    std::shared_ptr<SchedulingDecision> decision;
    std::shared_ptr<BatchExecuteRequest> knownSizeReq;
    if (!isDistChange && existsPreloadedDec) {
        decision = getPreloadedSchedulingDecision(appId, *req);
        if (isScaleChange) {
            state.preloadedSchedulingDecisions.erase(appId);
        }
    } else if (isNew && isMpi) {
        knownSizeReq = std::make_shared<BatchExecuteRequest>(*req);
        int worldSize = req->messages[0].mpiWorldSize;
        for (int i = (int)req->messages.size(); i < worldSize; i++) {
            Message m;
            m.appId = appId;
            m.groupIdx = i;
            knownSizeReq->messages.push_back(std::move(m));
        }
        decision = scheduler->makeSchedulingDecision(
          hostMapCopy, state.inFlightReqs, *knownSizeReq);
    } else {
        decision = scheduler->makeSchedulingDecision(
          hostMapCopy, state.inFlightReqs, *req);
    }

    // Failure sentinels propagate to the caller
    if (decision->appId == NOT_ENOUGH_SLOTS) {
        FAM_WARN("planner: not enough slots for app %d (requested %d)",
                 appId,
                 (int)req->messages.size());
        return decision;
    }
    if (decision->appId == DO_NOT_MIGRATE) {
        return decision;
    }
    if (decision->appId == MUST_FREEZE) {
        FAM_INFO("planner: freezing app %d", appId);
        state.evictedRequests[appId] = std::make_shared<BatchExecuteRequest>(
          *state.inFlightReqs.at(appId).first);
        return decision;
    }

    bool isSingleHost = false;
    {
        // Single-host when every message landed on one host
        auto uniq = decision->uniqueHosts();
        isSingleHost = uniq.size() == 1;
    }
    if (!isSingleHost && req->singleHostHint) {
        return std::make_shared<SchedulingDecision>(NOT_ENOUGH_SLOTS,
                                                    NOT_ENOUGH_SLOTS);
    }

    // Un-freeze bookkeeping (reference: src/planner/Planner.cpp:1038-1081)
    bool isUnfreezeDispatch = false;
    if (state.evictedRequests.count(appId) > 0) {
        if (isNew && !isMpi) {
            // Non-MPI app: all messages re-dispatch at once, bookkeeping
            // is complete as soon as the new decision exists
            state.evictedRequests.erase(appId);
        } else if (isNew && isMpi) {
            // Un-freeze of an MPI world: every rank is already known
            // (the frozen BER carries all messages with their freeze
            // snapshots + reentry input), so the whole gang dispatches
            // through the known-size path at once and every rank —
            // including 0 — re-enters by JOINING the re-built world.
            // (Departure from the reference's two-step re-create,
            // src/planner/Planner.cpp:1042-1081: dispatching directly
            // avoids rank 0 re-creating a world that its own freeze
            // snapshot marks as already existing.)
            isUnfreezeDispatch = true;
            state.evictedRequests.erase(appId);
        } else if (isMpi && !isDistChange) {
            auto evictedBer = state.evictedRequests.at(appId);
            for (auto& m : req->messages) {
                for (size_t j = 1; j < evictedBer->messages.size(); j++) {
                    const auto& old = evictedBer->messages[j];
                    if (m.groupIdx == old.groupIdx) {
                        m.id = old.id;
                        m.funcPtr = old.funcPtr;
                        m.inputData = old.inputData;
                        m.snapshotKey = old.snapshotKey;
                        break;
                    }
                }
            }
            state.evictedRequests.erase(appId);
        }
    }

    bool skipClaim =
      decision->groupId == FIXED_SIZE_PRELOADED_DECISION_GROUPID;

    int32_t newGroupId = generateGidInt32();
    decision->groupId = newGroupId;
    updateBatchExecGroupId(*req, newGroupId);

    auto& broker = getPointToPointBroker();
    switch (decisionType) {
        case DecisionType::NEW: {
            for (int i = 0; i < decision->nFunctions; i++) {
                auto host = state.hostMap.at(decision->hosts[i]);
                host->info.usedSlots++;
                // Data-plane ports only exist for MPI ranks (the RCCL
                // bootstrap rides the PTP plane, so the pool is small)
                if (isMpi) {
                    try {
                        decision->mpiPorts[i] = claimHostMpiPort(host);
                    } catch (const std::exception& e) {
                        FAM_ERROR("mpi port claim failed for app %d: %s",
                                  appId,
                                  e.what());
                    }
                }
            }

            if (isMpi && knownSizeReq != nullptr && !isUnfreezeDispatch) {
                auto preload = std::make_shared<SchedulingDecision>(*decision);
                preload->groupId = FIXED_SIZE_PRELOADED_DECISION_GROUPID;
                state.preloadedSchedulingDecisions[appId] = preload;
                // Hand back only the first message's slice now
                for (size_t i = 1; i < preload->messageIds.size(); i++) {
                    decision->removeMessage(preload->messageIds[i]);
                }
            }

            state.inFlightReqs[appId] = { req, decision };
            DecisionCache::get().addCachedDecision(*req, *decision);
            broker.setAndSendMappingsFromSchedulingDecision(*decision);
            break;
        }
        case DecisionType::SCALE_CHANGE: {
            if (!skipClaim) {
                for (int i = 0; i < decision->nFunctions; i++) {
                    state.hostMap.at(decision->hosts[i])->info.usedSlots++;
                }
            }
            auto oldReq = state.inFlightReqs.at(appId).first;
            auto oldDec = state.inFlightReqs.at(appId).second;
            updateBatchExecGroupId(*oldReq, newGroupId);
            oldDec->groupId = newGroupId;

            for (size_t i = 0; i < req->messages.size(); i++) {
                oldReq->messages.push_back(req->messages[i]);
                oldDec->addMessage(decision->hosts[i], req->messages[i]);
                if (!skipClaim && isMpi) {
                    oldDec->mpiPorts[oldDec->nFunctions - 1] =
                      claimHostMpiPort(
                        state.hostMap.at(decision->hosts[i]));
                } else if (skipClaim) {
                    oldDec->mpiPorts[oldDec->nFunctions - 1] =
                      decision->mpiPorts[i];
                }
            }
            broker.setAndSendMappingsFromSchedulingDecision(*oldDec);
            break;
        }
        case DecisionType::DIST_CHANGE: {
            auto oldReq = state.inFlightReqs.at(appId).first;
            auto oldDec = state.inFlightReqs.at(appId).second;

            // Hosts only in the old decision still need the new mappings
            std::set<std::string> evictedHosts;
            {
                auto newHosts = decision->uniqueHosts();
                for (const auto& h : oldDec->uniqueHosts()) {
                    if (newHosts.count(h) == 0) {
                        evictedHosts.insert(h);
                    }
                }
            }

            // Move slot + port accounting from old to new placement
            for (int i = 0; i < oldDec->nFunctions; i++) {
                if (decision->hosts[i] != oldDec->hosts[i]) {
                    auto oldHost = state.hostMap.at(oldDec->hosts[i]);
                    oldHost->info.usedSlots =
                      std::max(0, oldHost->info.usedSlots - 1);
                    releaseHostMpiPort(oldHost, oldDec->mpiPorts[i]);
                }
            }
            for (int i = 0; i < decision->nFunctions; i++) {
                if (decision->hosts[i] != oldDec->hosts[i]) {
                    auto newHost = state.hostMap.at(decision->hosts[i]);
                    newHost->info.usedSlots++;
                    try {
                        decision->mpiPorts[i] = claimHostMpiPort(newHost);
                    } catch (const std::exception& e) {
                        FAM_ERROR("mpi port claim failed: %s", e.what());
                    }
                }
            }

            state.numMigrations += 1;
            {
                auto oldHosts = oldDec->uniqueHosts();
                state.supersededGroups[appId].emplace_back(
                  oldDec->groupId,
                  std::vector<std::string>(oldHosts.begin(),
                                           oldHosts.end()));
            }
            updateBatchExecGroupId(*oldReq, newGroupId);
            state.inFlightReqs[appId] = { oldReq, decision };
            broker.setAndSendMappingsFromSchedulingDecision(*decision);
            broker.sendMappingsFromSchedulingDecision(*decision,
                                                      evictedHosts);
            break;
        }
        default:
            throw FaabricException("unrecognised decision type");
    }

    if (decisionType != DecisionType::DIST_CHANGE) {
        dispatchSchedulingDecision(req, decision);
    }
    return decision;
}

void Planner::dispatchSchedulingDecision(
  std::shared_ptr<BatchExecuteRequest> req,
  std::shared_ptr<SchedulingDecision> decision)
{
    // Split the BER per host preserving message order
    // (reference: src/planner/Planner.cpp:1293-1390)
    std::map<std::string, std::shared_ptr<BatchExecuteRequest>> hostReqs;
    bool isSingleHost = decision->uniqueHosts().size() == 1;

    for (size_t i = 0; i < req->messages.size(); i++) {
        const std::string& host = decision->hosts.at(i);
        auto& hr = hostReqs[host];
        if (!hr) {
            hr = std::make_shared<BatchExecuteRequest>();
            hr->appId = decision->appId;
            hr->groupId = decision->groupId;
            hr->user = req->user;
            hr->function = req->function;
            hr->snapshotKey = req->snapshotKey;
            hr->type = req->type;
            hr->subType = req->subType;
            hr->contextData = req->contextData;
            hr->singleHost = isSingleHost;
            hr->singleHostHint = req->singleHostHint;
            hr->elasticScaleHint = req->elasticScaleHint;
        }
        hr->messages.push_back(req->messages[i]);
    }

    bool isThreads = req->type == BatchExecuteType::THREADS;
    for (auto& [host, hr] : hostReqs) {
        // Snapshots travel ahead of THREADS forks and un-freezes
        if (isThreads && !isSingleHost && !hr->messages.empty()) {
            const auto& m0 = hr->messages[0];
            std::string key =
              getMainThreadSnapshotKey(m0.user, m0.function, m0.appId);
            if (host != req->messages[0].mainHost) {
                try {
                    auto snap = SnapshotRegistry::get().getSnapshot(key);
                    getSnapshotClient(host)->pushSnapshot(key, *snap);
                } catch (const std::exception& e) {
                    FAM_ERROR("push of snapshot %s to %s failed: %s",
                              key.c_str(),
                              host.c_str(),
                              e.what());
                }
            }
        }
        if (!isThreads && !hr->messages.empty() &&
            !hr->messages[0].snapshotKey.empty()) {
            for (const auto& m : hr->messages) {
                try {
                    auto snap =
                      SnapshotRegistry::get().getSnapshot(m.snapshotKey);
                    getSnapshotClient(host)->pushSnapshot(m.snapshotKey,
                                                          *snap);
                } catch (const std::exception& e) {
                    FAM_ERROR("push of snapshot %s to %s failed: %s",
                              m.snapshotKey.c_str(),
                              host.c_str(),
                              e.what());
                }
            }
        }
        getFunctionCallClient(host)->executeFunctions(*hr);
    }
}

void Planner::preloadSchedulingDecision(
  int32_t appId,
  std::shared_ptr<SchedulingDecision> decision)
{
    std::unique_lock lock(plannerMx);
    if (state.preloadedSchedulingDecisions.count(appId) > 0) {
        FAM_ERROR("preloaded decision already exists for app %d", appId);
        return;
    }
    auto preload = std::make_shared<SchedulingDecision>(*decision);
    preload->groupId = FIXED_SIZE_PRELOADED_DECISION_GROUPID;
    state.preloadedSchedulingDecisions[appId] = preload;
}

std::shared_ptr<SchedulingDecision> Planner::getSchedulingDecision(
  int32_t appId)
{
    std::shared_lock lock(plannerMx);
    auto it = state.inFlightReqs.find(appId);
    if (it == state.inFlightReqs.end()) {
        return nullptr;
    }
    return std::make_shared<SchedulingDecision>(*it->second.second);
}

// ----------------------------- results -------------------------------------

void Planner::setMessageResult(std::shared_ptr<Message> msg)
{
    std::vector<std::shared_ptr<Message>> one;
    one.push_back(std::move(msg));
    setMessageResults(one);
}

// Ingest a burst of results under ONE planner-lock acquisition (the
// result batcher ships ~15 per RPC; taking the unique lock per result
// was the dominant serialized cost of a completing batch)
void Planner::setMessageResults(
  std::vector<std::shared_ptr<Message>>& msgs)
{
    // (host, msg) result pushes and (host, appId) batch-done pushes
    // collected under the lock, sent after
    std::vector<std::pair<std::string, std::shared_ptr<Message>>> waiters;
    std::vector<std::pair<std::string, int32_t>> batchWaiters;
    std::vector<std::pair<std::string, int32_t>> groupClears;
    {
        std::unique_lock lock(plannerMx);
        for (auto& msg : msgs) {
            setMessageResultLocked(msg, waiters, batchWaiters, groupClears);
        }
    }

    // Completed apps: retire their PTP group on every involved host
    // (otherwise mappings/seq state accumulate forever)
    auto& broker = getPointToPointBroker();
    for (const auto& [host, groupId] : groupClears) {
        broker.sendGroupClear(host, groupId);
    }

    // A completed (non-frozen) message consumed its freeze/migration
    // snapshot: drop the planner's staged copy (one arena per freeze
    // otherwise). Frozen results keep theirs for the thaw.
    for (const auto& msg : msgs) {
        if (msg->returnValue != FROZEN_FUNCTION_RETURN_VALUE &&
            startsWith(msg->snapshotKey, "migration_")) {
            SnapshotRegistry::get().deleteSnapshot(msg->snapshotKey);
            DeviceSnapshotRegistry::get().deleteSnapshot(msg->snapshotKey);
        }
    }

    // Batch fully complete: push BATCH_DONE to registered hosts
    for (const auto& [host, appId] : batchWaiters) {
        try {
            getFunctionCallClient(host)->batchDone(appId);
        } catch (const std::exception& e) {
            FAM_ERROR(
              "batch-done push to %s failed: %s", host.c_str(), e.what());
        }
    }

    // Push results to hosts waiting on them
    for (const auto& [host, msg] : waiters) {
        try {
            getFunctionCallClient(host)->setMessageResult(*msg);
        } catch (const std::exception& e) {
            FAM_ERROR("result push to %s failed: %s", host.c_str(), e.what());
        }
    }
}

// Caller holds plannerMx. Results of completed apps are kept for
// late getMessageResult/getBatchResults fetches, then purged by age
// and count so a long-lived planner's memory stays bounded (the
// reference only clears results on an explicit flush)
void Planner::purgeOldResultsLocked()
{
    static const int64_t ttlMs =
      getEnvVarInt("FAABRIC_RESULT_TTL_MS", 5 * 60 * 1000);
    static const size_t maxDone =
      (size_t)getEnvVarInt("FAABRIC_MAX_DONE_APPS", 512);
    int64_t now = getGlobalClockEpochMillis();
    while (!state.doneApps.empty() &&
           (state.doneApps.size() > maxDone ||
            now - state.doneApps.front().first > ttlMs)) {
        int32_t appId = state.doneApps.front().second;
        state.doneApps.pop_front();
        // An app id re-entering flight (un-freeze) keeps its results
        if (state.inFlightReqs.count(appId) == 0) {
            state.appResults.erase(appId);
        }
    }
}

void Planner::setMessageResultLocked(
  const std::shared_ptr<Message>& msg,
  std::vector<std::pair<std::string, std::shared_ptr<Message>>>& waiters,
  std::vector<std::pair<std::string, int32_t>>& batchWaiters,
  std::vector<std::pair<std::string, int32_t>>& groupClears)
{
    int32_t appId = msg->appId;
    int32_t msgId = msg->id;

    // Migrated messages re-run with the same id; ignore this result
    // (reference: src/planner/Planner.cpp:404-408)
    if (msg->returnValue == MIGRATED_FUNCTION_RETURN_VALUE) {
        return;
    }
    {

        bool isFrozenMsg = msg->returnValue == FROZEN_FUNCTION_RETURN_VALUE;
        if (isFrozenMsg) {
            auto it = state.evictedRequests.find(appId);
            if (it == state.evictedRequests.end()) {
                FAM_ERROR("frozen message %d has no evicted app %d",
                          msgId,
                          appId);
            } else {
                for (auto& m : it->second->messages) {
                    if (m.id == msgId) {
                        m.funcPtr = msg->funcPtr;
                        m.inputData = msg->inputData;
                        m.snapshotKey = msg->snapshotKey;
                        m.returnValue = msg->returnValue;
                        break;
                    }
                }
            }
        }

        // Release the slot exactly once per message
        auto hostIt = state.hostMap.find(msg->executedHost);
        bool alreadySet = state.appResults[appId].count(msgId) > 0;
        if (hostIt != state.hostMap.end() && (!alreadySet || isFrozenMsg)) {
            hostIt->second->info.usedSlots =
              std::max(0, hostIt->second->info.usedSlots - 1);
        }

        if (!isFrozenMsg) {
            state.appResults[appId][msgId] = msg;
        }

        // Remove from in-flight accounting
        auto ifIt = state.inFlightReqs.find(appId);
        if (ifIt != state.inFlightReqs.end()) {
            auto req = ifIt->second.first;
            auto decision = ifIt->second.second;
            auto msgIt = std::find_if(
              req->messages.begin(),
              req->messages.end(),
              [&](const Message& m) { return m.id == msgId; });
            if (msgIt != req->messages.end()) {
                // Swap-and-pop (Message is fat; a mid-vector erase moves
                // every later message — O(n^2) per completing batch).
                // Alignment with the decision is preserved: its
                // removeMessage swap-and-pops the same position.
                *msgIt = std::move(req->messages.back());
                req->messages.pop_back();
                int32_t freedPort = decision->removeMessage(msgId);
                if (hostIt != state.hostMap.end()) {
                    releaseHostMpiPort(hostIt->second, freedPort);
                }
                if (req->messages.empty()) {
                    state.inFlightReqs.erase(appId);
                    state.preloadedSchedulingDecisions.erase(appId);
                    auto bwIt = state.batchDoneWaiters.find(appId);
                    if (bwIt != state.batchDoneWaiters.end()) {
                        for (auto& h : bwIt->second) {
                            batchWaiters.emplace_back(std::move(h), appId);
                        }
                        state.batchDoneWaiters.erase(bwIt);
                    }
                    state.doneApps.emplace_back(
                      getGlobalClockEpochMillis(), appId);
                    purgeOldResultsLocked();
                    if (decision->groupId != 0) {
                        std::set<std::string> hosts(
                          decision->hosts.begin(), decision->hosts.end());
                        hosts.insert(getSystemConfig().endpointHost);
                        for (const auto& h : hosts) {
                            groupClears.emplace_back(h,
                                                     decision->groupId);
                        }
                    }
                    // Placements superseded by migrations are quiescent
                    // once the app is done: retire them too
                    auto sgIt = state.supersededGroups.find(appId);
                    if (sgIt != state.supersededGroups.end()) {
                        for (auto& [gid, hosts] : sgIt->second) {
                            for (auto& h : hosts) {
                                groupClears.emplace_back(std::move(h),
                                                         gid);
                            }
                            groupClears.emplace_back(
                              getSystemConfig().endpointHost, gid);
                        }
                        state.supersededGroups.erase(sgIt);
                    }
                }
            }
        }

        if (isFrozenMsg) {
            return;
        }

        auto wIt = state.appResultWaiters.find(msgId);
        if (wIt != state.appResultWaiters.end()) {
            for (auto& h : wIt->second) {
                waiters.emplace_back(std::move(h), msg);
            }
            state.appResultWaiters.erase(wIt);
        }
    }
}

bool Planner::registerBatchDoneWaiter(int32_t appId,
                                      const std::string& host)
{
    std::unique_lock lock(plannerMx);
    if (state.inFlightReqs.count(appId) == 0) {
        // Completed (results in appResults) or unknown — either way the
        // caller should check status, not wait for a push
        return true;
    }
    state.batchDoneWaiters[appId].push_back(host);
    return false;
}

size_t Planner::debugAppResultsCount()
{
    std::shared_lock lock(plannerMx);
    size_t n = 0;
    for (const auto& [a, m] : state.appResults) {
        n += m.size();
    }
    return n;
}
size_t Planner::debugDoneAppsCount()
{
    std::shared_lock lock(plannerMx);
    return state.doneApps.size();
}
size_t Planner::debugInFlightCount()
{
    std::shared_lock lock(plannerMx);
    return state.inFlightReqs.size();
}

std::shared_ptr<Message> Planner::getMessageResult(const Message& msg)
{
    int32_t appId = msg.appId;
    int32_t msgId = msg.id;
    {
        std::shared_lock lock(plannerMx);
        auto appIt = state.appResults.find(appId);
        if (appIt != state.appResults.end()) {
            auto msgIt = appIt->second.find(msgId);
            if (msgIt != appIt->second.end()) {
                return msgIt->second;
            }
        }
    }
    if (!msg.mainHost.empty()) {
        std::unique_lock lock(plannerMx);
        auto appIt = state.appResults.find(appId);
        if (appIt != state.appResults.end() &&
            appIt->second.count(msgId) > 0) {
            return appIt->second.at(msgId);
        }
        state.appResultWaiters[msgId].push_back(msg.mainHost);
    }
    return nullptr;
}

std::shared_ptr<BatchExecuteRequestStatus> Planner::getBatchResults(
  int32_t appId)
{
    auto status = std::make_shared<BatchExecuteRequestStatus>();
    status->appId = appId;

    // Result polling doubles as the un-freeze trigger
    // (reference: src/planner/Planner.cpp:700-726)
    std::shared_ptr<BatchExecuteRequest> frozenBer;
    {
        std::shared_lock lock(plannerMx);
        auto evIt = state.evictedRequests.find(appId);
        if (evIt != state.evictedRequests.end()) {
            bool fullyFrozen = std::all_of(
              evIt->second->messages.begin(),
              evIt->second->messages.end(),
              [](const Message& m) {
                  return m.returnValue == FROZEN_FUNCTION_RETURN_VALUE;
              });
            if (fullyFrozen && state.inFlightReqs.count(appId) == 0) {
                frozenBer = std::make_shared<BatchExecuteRequest>(
                  *evIt->second);
            }
            if (!frozenBer) {
                status->finished = false;
                return status;
            }
        }
    }
    if (frozenBer) {
        // Re-schedule as NEW; the messages keep their freeze snapshots
        // and re-entry input
        for (auto& m : frozenBer->messages) {
            m.returnValue = 0;
            m.executedHost.clear();
            m.finishTimestamp = 0;
        }
        auto decision = callBatch(frozenBer);
        if (decision->appId == NOT_ENOUGH_SLOTS) {
            FAM_DEBUG("cannot un-freeze app %d yet: no slots", appId);
        }
        status->finished = false;
        return status;
    }

    std::shared_lock lock(plannerMx);
    auto it = state.appResults.find(appId);
    if (it == state.appResults.end()) {
        return nullptr;
    }
    for (const auto& [msgId, msg] : it->second) {
        status->messageResults.push_back(*msg);
    }
    status->finished = state.inFlightReqs.count(appId) == 0;
    return status;
}

// ----------------------------- ops ------------------------------------------

int Planner::getNumMigrations()
{
    std::shared_lock lock(plannerMx);
    return state.numMigrations;
}

GetInFlightAppsResponse Planner::getInFlightApps()
{
    std::shared_lock lock(plannerMx);
    GetInFlightAppsResponse resp;
    for (const auto& [appId, pair] : state.inFlightReqs) {
        InFlightAppEntry e;
        e.appId = appId;
        e.subType = pair.first->subType;
        e.size = pair.second->nFunctions;
        e.hostIps = pair.second->hosts;
        resp.apps.push_back(std::move(e));
    }
    resp.numMigrations = state.numMigrations;
    for (const auto& ip : state.nextEvictedHostIps) {
        resp.nextEvictedVmIps.push_back(ip);
    }
    for (const auto& [appId, ber] : state.evictedRequests) {
        InFlightAppEntry e;
        e.appId = appId;
        e.subType = ber->subType;
        e.size = (int)ber->messages.size();
        resp.frozenApps.push_back(std::move(e));
    }
    return resp;
}

void Planner::setNextEvictedVm(const std::set<std::string>& vmIps)
{
    std::unique_lock lock(plannerMx);
    state.nextEvictedHostIps = vmIps;
}

void Planner::setPolicy(const std::string& policy)
{
    resetBatchScheduler(policy);
}

std::string Planner::getPolicy()
{
    return getBatchSchedulerMode();
}

// ----------------------------- lifecycle ------------------------------------

void Planner::reset()
{
    std::unique_lock lock(plannerMx);
    state = PlannerState();
}

void Planner::flushSchedulingState()
{
    std::unique_lock lock(plannerMx);
    state.inFlightReqs.clear();
    state.appResults.clear();
    state.appResultWaiters.clear();
    state.preloadedSchedulingDecisions.clear();
    state.evictedRequests.clear();
    state.numMigrations = 0;
    for (auto& [ip, h] : state.hostMap) {
        h->info.usedSlots = 0;
        std::fill(h->mpiPortUsed.begin(), h->mpiPortUsed.end(), false);
    }
}

void Planner::flushExecutors()
{
    std::vector<std::string> hosts;
    {
        std::shared_lock lock(plannerMx);
        for (const auto& [ip, h] : state.hostMap) {
            hosts.push_back(ip);
        }
    }
    for (const auto& host : hosts) {
        try {
            getFunctionCallClient(host)->sendFlush();
        } catch (const std::exception& e) {
            FAM_ERROR("flush of %s failed: %s", host.c_str(), e.what());
        }
    }
}

void Planner::flushHosts()
{
    std::unique_lock lock(plannerMx);
    state.hostMap.clear();
}

// ----------------------------- server ---------------------------------------

PlannerServer::PlannerServer()
  : MessageEndpointServer(PLANNER_ASYNC_PORT, PLANNER_SYNC_PORT, "planner")
{
    const int nResultWorkers = getEnvVarInt("FAABRIC_RESULT_WORKERS", 4);
    for (int i = 0; i < nResultWorkers; i++) {
        resultWorkers.emplace_back([this] { resultWorkerLoop(); });
    }
}

PlannerServer::~PlannerServer()
{
    resultWorkersStop.store(true);
    for (size_t i = 0; i < resultWorkers.size(); i++) {
        resultQueue.enqueue(std::string());
    }
    for (auto& t : resultWorkers) {
        if (t.joinable()) {
            t.join();
        }
    }
}

void PlannerServer::resultWorkerLoop()
{
    while (true) {
        std::string body;
        try {
            body = resultQueue.dequeue(0);
        } catch (...) {
            continue;
        }
        if (resultWorkersStop.load()) {
            return;
        }
        if (body.empty()) {
            continue;
        }
        try {
            // Tag byte: 'S' single result, 'B' batch of results (one
            // lock acquisition for the whole burst)
            char tag = body[0];
            std::string_view payload(body.data() + 1, body.size() - 1);
            if (tag == 'B') {
                PROF_START(result_decode)
                std::vector<std::shared_ptr<Message>> msgs;
                std::string payloadStr(payload);
                PbReader r(payloadStr);
                uint32_t f;
                WireType t;
                while (r.next(f, t)) {
                    if (f != 1) {
                        r.skip(t);
                        continue;
                    }
                    auto sub = r.asString();
                    msgs.push_back(
                      std::make_shared<Message>(Message::decode(sub)));
                }
                PROF_END(result_decode)
                PROF_START(result_core)
                Planner::get().setMessageResults(msgs);
                PROF_END(result_core)
            } else {
                PROF_START(result_decode)
                auto msg = std::make_shared<Message>(
                  Message::decode(std::string(payload)));
                PROF_END(result_decode)
                PROF_START(result_core)
                Planner::get().setMessageResult(msg);
                PROF_END(result_core)
            }
        } catch (const std::exception& e) {
            FAM_ERROR("result ingestion failed: %s", e.what());
        }
    }
}

void PlannerServer::doAsyncRecv(uint8_t code,
                                const std::string& body,
                                uint32_t seq)
{
    (void)seq;
    if ((PlannerCalls)code == PlannerCalls::SetMessageResult) {
        if (resultWorkers.empty()) {
            auto msg = std::make_shared<Message>(Message::decode(body));
            Planner::get().setMessageResult(msg);
        } else {
            resultQueue.enqueue("S" + body);
        }
        return;
    }
    if ((PlannerCalls)code == PlannerCalls::SetMessageResultBatch) {
        // One RPC, many results: hand the whole burst to one worker so
        // it ingests under a single planner-lock acquisition
        if (resultWorkers.empty()) {
            std::vector<std::shared_ptr<Message>> msgs;
            PbReader r(body);
            uint32_t f;
            WireType t;
            while (r.next(f, t)) {
                if (f != 1) {
                    r.skip(t);
                    continue;
                }
                msgs.push_back(std::make_shared<Message>(
                  Message::decode(r.asString())));
            }
            Planner::get().setMessageResults(msgs);
        } else {
            resultQueue.enqueue("B" + body);
        }
        return;
    }
    FAM_ERROR("planner server: bad async code %d", (int)code);
}

std::string PlannerServer::doSyncRecv(uint8_t code, const std::string& body)
{
    auto& planner = Planner::get();
    switch ((PlannerCalls)code) {
        case PlannerCalls::Ping: {
            PbWriter w;
            w.putMessage(1, planner.getConfig().encode());
            return w.take();
        }
        case PlannerCalls::GetAvailableHosts: {
            AvailableHostsResponse resp;
            resp.hosts = planner.getAvailableHosts();
            return resp.encode();
        }
        case PlannerCalls::RegisterHost: {
            auto req = RegisterHostRequest::decode(body);
            bool ok = planner.registerHost(req.host, req.overwrite);
            RegisterHostResponse resp;
            resp.status = ok ? 0 : 1;
            resp.config = planner.getConfig();
            return resp.encode();
        }
        case PlannerCalls::RemoveHost: {
            auto req = RegisterHostRequest::decode(body);
            planner.removeHost(req.host);
            return {};
        }
        case PlannerCalls::WaitBatchDone: {
            PbReader r(body);
            int32_t appId = 0;
            std::string host;
            uint32_t f;
            WireType t;
            while (r.next(f, t)) {
                if (f == 1) {
                    appId = r.asInt32();
                } else if (f == 2) {
                    host = r.asString();
                } else {
                    r.skip(t);
                }
            }
            bool done = planner.registerBatchDoneWaiter(appId, host);
            PbWriter w;
            w.putBool(1, done);
            return w.take();
        }
        case PlannerCalls::SetMessageResult: {
            auto msg = std::make_shared<Message>(Message::decode(body));
            planner.setMessageResult(msg);
            return {};
        }
        case PlannerCalls::GetMessageResult: {
            Message query = Message::decode(body);
            auto result = planner.getMessageResult(query);
            if (!result) {
                Message empty;
                empty.type = MessageType::EMPTY;
                return empty.encode();
            }
            return result->encode();
        }
        case PlannerCalls::GetBatchResults: {
            BatchExecuteRequest req = BatchExecuteRequest::decode(body);
            auto status = planner.getBatchResults(req.appId);
            if (!status) {
                BatchExecuteRequestStatus empty;
                empty.appId = req.appId;
                empty.expectedNumMessages = -1; // marker: unknown app
                return empty.encode();
            }
            if (req.subType == 1) {
                // Count-only poll: skip re-serialising every result
                BatchExecuteRequestStatus counts;
                counts.appId = status->appId;
                counts.finished = status->finished;
                counts.expectedNumMessages =
                  (int32_t)status->messageResults.size();
                return counts.encode();
            }
            return status->encode();
        }
        case PlannerCalls::GetSchedulingDecision: {
            BatchExecuteRequest req = BatchExecuteRequest::decode(body);
            auto decision = planner.getSchedulingDecision(req.appId);
            if (!decision) {
                return PointToPointMappings{}.encode();
            }
            return decision->toPointToPointMappings().encode();
        }
        case PlannerCalls::GetNumMigrations: {
            PbWriter w;
            w.putInt32(1, planner.getNumMigrations());
            return w.take();
        }
        case PlannerCalls::CallBatch: {
            auto req = std::make_shared<BatchExecuteRequest>(
              BatchExecuteRequest::decode(body));
            auto decision = planner.callBatch(req);
            PointToPointMappings out = decision->toPointToPointMappings();
            out.appId = decision->appId;
            out.groupId = decision->groupId;
            return out.encode();
        }
        case PlannerCalls::PreloadSchedulingDecision: {
            auto mappings = PointToPointMappings::decode(body);
            auto decision = std::make_shared<SchedulingDecision>(
              SchedulingDecision::fromPointToPointMappings(mappings));
            planner.preloadSchedulingDecision(mappings.appId, decision);
            return {};
        }
        default:
            throw FaabricException("planner server: bad sync code " +
                                   std::to_string(code));
    }
}

// ----------------------------- client ---------------------------------------

class PlannerClient::KeepAliveThread : public PeriodicBackgroundThread
{
  public:
    void doWork() override
    {
        Host host;
        host.ip = getSystemConfig().endpointHost;
        host.slots = Scheduler::get().getThisHostResources().slots;
        getPlannerClient().registerHost(host, false);
    }
};

PlannerClient::PlannerClient()
  : rpc(getSystemConfig().plannerHost, PLANNER_ASYNC_PORT, PLANNER_SYNC_PORT)
{}

PlannerClient::~PlannerClient()
{
    stopKeepAlive();
}

void PlannerClient::ping()
{
    rpc.syncSend((uint8_t)PlannerCalls::Ping, "");
}

std::vector<Host> PlannerClient::getAvailableHosts()
{
    std::string resp =
      rpc.syncSend((uint8_t)PlannerCalls::GetAvailableHosts, "");
    return AvailableHostsResponse::decode(resp).hosts;
}

int32_t PlannerClient::registerHost(const Host& host, bool overwrite)
{
    RegisterHostRequest req;
    req.host = host;
    req.overwrite = overwrite;
    std::string resp =
      rpc.syncSend((uint8_t)PlannerCalls::RegisterHost, req.encode());
    auto r = RegisterHostResponse::decode(resp);
    if (r.status != 0) {
        throw FaabricException("planner rejected host registration");
    }
    return r.hostId;
}

void PlannerClient::removeHost(const Host& host)
{
    RegisterHostRequest req;
    req.host = host;
    rpc.syncSend((uint8_t)PlannerCalls::RemoveHost, req.encode());
}

std::shared_ptr<SchedulingDecision> PlannerClient::callFunctions(
  std::shared_ptr<BatchExecuteRequest> req)
{
    std::string resp =
      rpc.syncSend((uint8_t)PlannerCalls::CallBatch, req->encode());
    auto mappings = PointToPointMappings::decode(resp);
    auto decision = std::make_shared<SchedulingDecision>(
      SchedulingDecision::fromPointToPointMappings(mappings));
    decision->appId = mappings.appId;
    decision->groupId = mappings.groupId;
    return decision;
}

void PlannerClient::setMessageResult(std::shared_ptr<Message> msg)
{
    if (isMockMode()) {
        return;
    }
    rpc.asyncSend((uint8_t)PlannerCalls::SetMessageResult, msg->encode());
}

void PlannerClient::setMessageResultsBatch(
  const std::vector<std::shared_ptr<Message>>& msgs)
{
    if (isMockMode() || msgs.empty()) {
        return;
    }
    if (msgs.size() == 1) {
        setMessageResult(msgs[0]);
        return;
    }
    PbWriter w;
    for (const auto& m : msgs) {
        w.putMessage(1, m->encode());
    }
    rpc.asyncSend((uint8_t)PlannerCalls::SetMessageResultBatch, w.take());
}

void PlannerClient::setMessageResultLocally(std::shared_ptr<Message> msg)
{
    {
        std::lock_guard<std::mutex> lock(resultsMx);
        localResults[msg->id] = msg;
    }
    resultsCv.notify_all();
}

Message PlannerClient::getMessageResult(const Message& msg, int timeoutMs)
{
    return getMessageResult(msg.appId, msg.id, timeoutMs);
}

Message PlannerClient::getMessageResult(int32_t appId,
                                        int32_t msgId,
                                        int timeoutMs)
{
    int64_t deadline = getGlobalClockEpochMillis() + timeoutMs;
    int64_t nextRpcAt = 0;
    while (true) {
        {
            std::unique_lock<std::mutex> lock(resultsMx);
            auto it = localResults.find(msgId);
            if (it != localResults.end()) {
                Message out = *it->second;
                localResults.erase(it);
                return out;
            }
        }

        int64_t now = getGlobalClockEpochMillis();
        if (now >= deadline) {
            throw QueueTimeoutException("timed out waiting for result of " +
                                        std::to_string(msgId));
        }

        // RPC (also registers this host as a waiter); re-poll periodically
        // in case the push is lost
        if (now >= nextRpcAt) {
            Message query;
            query.appId = appId;
            query.id = msgId;
            query.mainHost = getSystemConfig().endpointHost;
            std::string resp = rpc.syncSend(
              (uint8_t)PlannerCalls::GetMessageResult, query.encode());
            Message result = Message::decode(resp);
            if (!(result.type == MessageType::EMPTY && result.id == 0)) {
                return result;
            }
            nextRpcAt = now + 2000;
        }

        std::unique_lock<std::mutex> lock(resultsMx);
        resultsCv.wait_for(lock, std::chrono::milliseconds(50), [&] {
            return localResults.count(msgId) > 0;
        });
    }
}

BatchExecuteRequestStatus PlannerClient::getBatchResults(int32_t appId)
{
    BatchExecuteRequest req;
    req.appId = appId;
    std::string resp =
      rpc.syncSend((uint8_t)PlannerCalls::GetBatchResults, req.encode());
    return BatchExecuteRequestStatus::decode(resp);
}

std::pair<bool, int> PlannerClient::getBatchStatusCounts(int32_t appId)
{
    BatchExecuteRequest req;
    req.appId = appId;
    req.subType = 1;
    std::string resp =
      rpc.syncSend((uint8_t)PlannerCalls::GetBatchResults, req.encode());
    auto status = BatchExecuteRequestStatus::decode(resp);
    if (status.expectedNumMessages == -1) {
        return { false, -1 };
    }
    return { status.finished, status.expectedNumMessages };
}

bool PlannerClient::waitBatchDone(int32_t appId, int timeoutMs)
{
    // A process without a FunctionCallServer cannot receive the push —
    // registering would stall the planner's result workers on a dead
    // connection. Poll with a gentle backoff instead (matches the
    // pre-push behavior for driver/test processes).
    if (!functionCallServerRunning()) {
        auto deadline = std::chrono::steady_clock::now() +
                        std::chrono::milliseconds(timeoutMs);
        int sleepUs = 200;
        while (std::chrono::steady_clock::now() < deadline) {
            auto [finished, n] = getBatchStatusCounts(appId);
            (void)n;
            if (finished) {
                return true;
            }
            std::this_thread::sleep_for(std::chrono::microseconds(sleepUs));
            sleepUs = std::min(sleepUs * 2, 20000);
        }
        return false;
    }

    // Local flag FIRST: a push that lands between the register RPC and
    // the wait below still sets it
    auto waiter = batchDoneWaiterPrepare(appId);

    PbWriter w;
    w.putInt32(1, appId);
    w.putString(2, getSystemConfig().endpointHost);
    std::string resp =
      rpc.syncSend((uint8_t)PlannerCalls::WaitBatchDone, w.take());
    bool already = false;
    {
        PbReader r(resp);
        uint32_t f;
        WireType t;
        while (r.next(f, t)) {
            if (f == 1) {
                already = r.asBool();
            } else {
                r.skip(t);
            }
        }
    }
    if (already) {
        batchDoneWaiterDiscard(appId);
        return true;
    }

    // Sleep on the flag; wake every 500 ms for a status re-check so a
    // lost push (e.g. planner restart) cannot hang the caller
    auto deadline = std::chrono::steady_clock::now() +
                    std::chrono::milliseconds(timeoutMs);
    bool done = false;
    while (!done) {
        if (waiter->isSet()) {
            done = true;
            break;
        }
        if (std::chrono::steady_clock::now() > deadline) {
            break;
        }
        if (waiter->waitMs(500)) {
            done = true;
            break;
        }
        auto [finished, n] = getBatchStatusCounts(appId);
        (void)n;
        if (finished) {
            done = true;
            break;
        }
    }
    batchDoneWaiterDiscard(appId);
    return done;
}

SchedulingDecision PlannerClient::getSchedulingDecision(int32_t appId)
{
    BatchExecuteRequest req;
    req.appId = appId;
    std::string resp = rpc.syncSend(
      (uint8_t)PlannerCalls::GetSchedulingDecision, req.encode());
    return SchedulingDecision::fromPointToPointMappings(
      PointToPointMappings::decode(resp));
}

void PlannerClient::preloadSchedulingDecision(
  int32_t appId,
  const SchedulingDecision& decision)
{
    PointToPointMappings mappings = decision.toPointToPointMappings();
    mappings.appId = appId;
    rpc.syncSend((uint8_t)PlannerCalls::PreloadSchedulingDecision,
                 mappings.encode());
}

int PlannerClient::getNumMigrations()
{
    std::string resp =
      rpc.syncSend((uint8_t)PlannerCalls::GetNumMigrations, "");
    PbReader r(resp);
    uint32_t f;
    WireType t;
    while (r.next(f, t)) {
        if (f == 1) {
            return r.asInt32();
        }
        r.skip(t);
    }
    return 0;
}

void PlannerClient::startKeepAlive()
{
    if (!keepAlive) {
        keepAlive = std::make_shared<KeepAliveThread>();
        keepAlive->startMillis(
          std::max(500, Planner::get().hostTimeoutMs / 2));
    }
}

void PlannerClient::stopKeepAlive()
{
    if (keepAlive) {
        keepAlive->stop();
        keepAlive = nullptr;
    }
}

void PlannerClient::clearCache()
{
    std::lock_guard<std::mutex> lock(resultsMx);
    localResults.clear();
}

static std::shared_ptr<PlannerClient> plannerClientInstance;
static std::mutex plannerClientMx;

PlannerClient& getPlannerClient()
{
    std::lock_guard<std::mutex> lock(plannerClientMx);
    if (!plannerClientInstance) {
        plannerClientInstance = std::make_shared<PlannerClient>();
    }
    return *plannerClientInstance;
}

void resetPlannerClient()
{
    std::lock_guard<std::mutex> lock(plannerClientMx);
    plannerClientInstance = nullptr;
}

} // namespace faabricamd
