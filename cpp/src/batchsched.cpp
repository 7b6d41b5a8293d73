// Placement policies. Behavioral parity with the reference's
// bin-pack / compact / spot schedulers (reference:
// src/batch-scheduler/BinPackScheduler.cpp:299-363,
// CompactScheduler.cpp:339-400, SpotScheduler.cpp:255-330,
// BatchScheduler.cpp:50-66) — implemented fresh around a shared
// sort-then-fill core instead of three near-identical classes.
#include "faabricamd/scheduling.h"
#include "faabricamd/util.h"

#include <algorithm>
#include <mutex>

namespace faabricamd {

// ------------------------- SchedulingDecision ------------------------------

SchedulingDecision SchedulingDecision::fromPointToPointMappings(
  const PointToPointMappings& mappings)
{
    SchedulingDecision d(mappings.appId, mappings.groupId);
    for (const auto& m : mappings.mappings) {
        d.addMessageInPosition(
          d.nFunctions, m.host, m.messageId, m.appIdx, m.groupIdx, m.mpiPort);
    }
    return d;
}

PointToPointMappings SchedulingDecision::toPointToPointMappings() const
{
    PointToPointMappings out;
    out.appId = appId;
    out.groupId = groupId;
    for (int i = 0; i < nFunctions; i++) {
        PointToPointMapping m;
        m.host = hosts[i];
        m.messageId = messageIds[i];
        m.appIdx = appIdxs[i];
        m.groupIdx = groupIdxs[i];
        m.mpiPort = mpiPorts[i];
        out.mappings.push_back(std::move(m));
    }
    return out;
}

bool SchedulingDecision::isSingleHost() const
{
    const std::string& thisHost = getSystemConfig().endpointHost;
    return std::all_of(hosts.begin(), hosts.end(), [&](const auto& h) {
        return h == thisHost;
    });
}

void SchedulingDecision::addMessage(const std::string& host,
                                    const Message& msg)
{
    addMessage(host, msg.id, msg.appIdx, msg.groupIdx);
}

void SchedulingDecision::addMessage(const std::string& host,
                                    int32_t messageId,
                                    int32_t appIdx,
                                    int32_t groupIdx)
{
    nFunctions++;
    hosts.push_back(host);
    messageIds.push_back(messageId);
    appIdxs.push_back(appIdx);
    groupIdxs.push_back(groupIdx);
    mpiPorts.push_back(0);
}

void SchedulingDecision::addMessageInPosition(int32_t pos,
                                              const std::string& host,
                                              int32_t messageId,
                                              int32_t appIdx,
                                              int32_t groupIdx,
                                              int32_t mpiPort)
{
    int32_t desiredSize = std::max<int32_t>(pos + 1, nFunctions);
    nFunctions = desiredSize;
    if ((int32_t)hosts.size() < desiredSize) {
        hosts.resize(desiredSize);
        messageIds.resize(desiredSize, 0);
        appIdxs.resize(desiredSize, 0);
        groupIdxs.resize(desiredSize, 0);
        mpiPorts.resize(desiredSize, 0);
    }
    hosts[pos] = host;
    messageIds[pos] = messageId;
    appIdxs[pos] = appIdx;
    groupIdxs[pos] = groupIdx;
    mpiPorts[pos] = mpiPort;
}

int32_t SchedulingDecision::removeMessage(int32_t messageId)
{
    auto it = std::find(messageIds.begin(), messageIds.end(), messageId);
    if (it == messageIds.end()) {
        return 0;
    }
    size_t idx = (size_t)std::distance(messageIds.begin(), it);
    int32_t port = mpiPorts[idx];
    nFunctions--;
    // Swap-and-pop: results arrive per message and vector erases shift
    // every later entry; position alignment with the in-flight request
    // is preserved because Planner::setMessageResult removes the SAME
    // message from both structures the same way
    size_t last = messageIds.size() - 1;
    hosts[idx] = std::move(hosts[last]);
    messageIds[idx] = messageIds[last];
    appIdxs[idx] = appIdxs[last];
    groupIdxs[idx] = groupIdxs[last];
    mpiPorts[idx] = mpiPorts[last];
    hosts.pop_back();
    messageIds.pop_back();
    appIdxs.pop_back();
    groupIdxs.pop_back();
    mpiPorts.pop_back();
    return port;
}

std::set<std::string> SchedulingDecision::uniqueHosts() const
{
    return { hosts.begin(), hosts.end() };
}

void SchedulingDecision::print() const
{
    FAM_DEBUG("decision app=%d group=%d n=%d", appId, groupId, nFunctions);
    for (int i = 0; i < nFunctions; i++) {
        FAM_DEBUG("  msg=%d appIdx=%d grpIdx=%d host=%s port=%d",
                  messageIds[i],
                  appIdxs[i],
                  groupIdxs[i],
                  hosts[i].c_str(),
                  mpiPorts[i]);
    }
}

// ------------------------- shared helpers -----------------------------------

DecisionType BatchScheduler::getDecisionType(const InFlightReqs& inFlightReqs,
                                             const BatchExecuteRequest& req)
{
    if (inFlightReqs.find(req.appId) == inFlightReqs.end()) {
        return DecisionType::NEW;
    }
    if (req.type == BatchExecuteType::MIGRATION) {
        return DecisionType::DIST_CHANGE;
    }
    return DecisionType::SCALE_CHANGE;
}

using HostPtr = std::shared_ptr<HostState>;

static std::map<std::string, int> hostFreqCountOf(
  const SchedulingDecision& decision)
{
    std::map<std::string, int> freq;
    for (const auto& h : decision.hosts) {
        freq[h] += 1;
    }
    return freq;
}

// Sort key: more free slots first, then bigger host, then larger ip
static bool moreCapacityFirst(const HostPtr& a, const HostPtr& b)
{
    int availA = std::max(0, a->slots - a->usedSlots);
    int availB = std::max(0, b->slots - b->usedSlots);
    if (availA != availB) {
        return availA > availB;
    }
    if (a->slots != b->slots) {
        return a->slots > b->slots;
    }
    return a->ip > b->ip;
}

static auto makeFreqComparator(const std::map<std::string, int>& freq)
{
    return [&freq](const HostPtr& a, const HostPtr& b) {
        auto count = [&freq](const HostPtr& h) {
            auto it = freq.find(h->ip);
            return it == freq.end() ? 0 : it->second;
        };
        int fa = count(a);
        int fb = count(b);
        if (fa != fb) {
            return fa > fb;
        }
        return moreCapacityFirst(a, b);
    };
}

// Fill hosts in order until all messages are placed; empty optional = out of
// slots
static bool binPackInto(std::vector<HostPtr>& sortedHosts,
                        const BatchExecuteRequest& req,
                        SchedulingDecision& decision)
{
    int numLeft = (int)req.messages.size();
    int msgIdx = 0;
    for (auto& h : sortedHosts) {
        int avail = std::max(0, h->slots - h->usedSlots);
        int n = std::min(numLeft, avail);
        for (int i = 0; i < n; i++) {
            decision.addMessage(h->ip, req.messages[msgIdx++]);
        }
        numLeft -= n;
        if (numLeft == 0) {
            return true;
        }
    }
    return false;
}

std::shared_ptr<SchedulingDecision> minimiseNumOfMigrations(
  std::shared_ptr<SchedulingDecision> newDecision,
  std::shared_ptr<SchedulingDecision> oldDecision)
{
    auto out = std::make_shared<SchedulingDecision>(oldDecision->appId,
                                                    oldDecision->groupId);
    auto freq = hostFreqCountOf(*newDecision);

    // Keep each message on its old host where the new histogram allows it
    for (int i = 0; i < oldDecision->nFunctions; i++) {
        const auto& oldHost = oldDecision->hosts[i];
        auto it = freq.find(oldHost);
        if (it != freq.end() && it->second > 0) {
            out->addMessageInPosition(i,
                                      oldHost,
                                      oldDecision->messageIds[i],
                                      oldDecision->appIdxs[i],
                                      oldDecision->groupIdxs[i],
                                      oldDecision->mpiPorts[i]);
            it->second--;
        }
    }

    // Place the rest onto whatever the histogram still has
    for (int i = 0; i < oldDecision->nFunctions; i++) {
        if (out->nFunctions <= i || out->hosts[i].empty()) {
            std::string nextHost;
            for (auto& [ip, n] : freq) {
                if (n > 0) {
                    nextHost = ip;
                    break;
                }
            }
            if (nextHost.empty()) {
                throw FaabricException("migration histogram exhausted");
            }
            out->addMessageInPosition(i,
                                      nextHost,
                                      oldDecision->messageIds[i],
                                      oldDecision->appIdxs[i],
                                      oldDecision->groupIdxs[i],
                                      -1);
            freq[nextHost]--;
        }
    }
    return out;
}

// ------------------------- BinPack ------------------------------------------

// Locality score: (num hosts, cross-host links in the fully-connected gang)
static std::pair<int, int> localityScore(const SchedulingDecision& d)
{
    auto freq = hostFreqCountOf(d);
    if (freq.size() <= 1) {
        return { (int)freq.size(), 0 };
    }
    int total = 0;
    for (auto& [h, n] : freq) {
        total += n;
    }
    int score = 0;
    for (auto& [h, n] : freq) {
        score += n * (total - n);
    }
    return { (int)freq.size(), score / 2 };
}

std::shared_ptr<SchedulingDecision> BinPackScheduler::makeSchedulingDecision(
  HostMap& hostMap,
  const InFlightReqs& inFlightReqs,
  const BatchExecuteRequest& req)
{
    auto decision = std::make_shared<SchedulingDecision>(req.appId, 0);
    auto decisionType = getDecisionType(inFlightReqs, req);

    std::vector<HostPtr> sorted;
    for (auto& [ip, h] : hostMap) {
        sorted.push_back(h);
    }

    std::map<std::string, int> freq;
    if (decisionType != DecisionType::NEW) {
        freq = hostFreqCountOf(*inFlightReqs.at(req.appId).second);
    }

    switch (decisionType) {
        case DecisionType::NEW:
            std::sort(sorted.begin(), sorted.end(), moreCapacityFirst);
            break;
        case DecisionType::SCALE_CHANGE:
            std::sort(sorted.begin(), sorted.end(), makeFreqComparator(freq));
            break;
        case DecisionType::DIST_CHANGE: {
            // Fresh shot at scheduling: release this app's current slots,
            // then sort by capacity breaking ties on the app's histogram
            for (auto& h : sorted) {
                auto it = freq.find(h->ip);
                if (it != freq.end()) {
                    h->usedSlots = std::max(0, h->usedSlots - it->second);
                }
            }
            auto freqCmp = makeFreqComparator(freq);
            std::sort(sorted.begin(),
                      sorted.end(),
                      [&](const HostPtr& a, const HostPtr& b) {
                          int availA = std::max(0, a->slots - a->usedSlots);
                          int availB = std::max(0, b->slots - b->usedSlots);
                          if (availA != availB) {
                              return availA > availB;
                          }
                          return freqCmp(a, b);
                      });
            break;
        }
        default:
            throw FaabricException("unrecognised decision type");
    }

    if (!binPackInto(sorted, req, *decision)) {
        return std::make_shared<SchedulingDecision>(NOT_ENOUGH_SLOTS,
                                                    NOT_ENOUGH_SLOTS);
    }

    if (decisionType == DecisionType::DIST_CHANGE) {
        auto oldDecision = inFlightReqs.at(req.appId).second;
        auto newScore = localityScore(*decision);
        auto oldScore = localityScore(*oldDecision);
        bool better = newScore.first != oldScore.first
                        ? newScore.first < oldScore.first
                        : newScore.second < oldScore.second;
        if (better) {
            return minimiseNumOfMigrations(decision, oldDecision);
        }
        return std::make_shared<SchedulingDecision>(DO_NOT_MIGRATE,
                                                    DO_NOT_MIGRATE);
    }
    return decision;
}

// ------------------------- Compact ------------------------------------------

std::shared_ptr<SchedulingDecision> CompactScheduler::makeSchedulingDecision(
  HostMap& hostMap,
  const InFlightReqs& inFlightReqs,
  const BatchExecuteRequest& req)
{
    auto decision = std::make_shared<SchedulingDecision>(req.appId, 0);
    auto decisionType = getDecisionType(inFlightReqs, req);

    // Multi-tenancy: drop hosts running other users' apps (the subType
    // field carries a user id in the reference's simulations)
    HostMap filtered = hostMap;
    for (const auto& [appId, pair] : inFlightReqs) {
        if (pair.first->subType == req.subType) {
            continue;
        }
        for (const auto& ip : pair.second->hosts) {
            filtered.erase(ip);
        }
    }

    std::vector<HostPtr> sorted;
    for (auto& [ip, h] : filtered) {
        sorted.push_back(h);
    }

    std::map<std::string, int> freq;
    if (decisionType != DecisionType::NEW) {
        freq = hostFreqCountOf(*inFlightReqs.at(req.appId).second);
    }

    switch (decisionType) {
        case DecisionType::NEW:
            std::sort(sorted.begin(), sorted.end(), moreCapacityFirst);
            break;
        case DecisionType::SCALE_CHANGE:
            std::sort(sorted.begin(), sorted.end(), makeFreqComparator(freq));
            break;
        case DecisionType::DIST_CHANGE: {
            for (auto& h : sorted) {
                auto it = freq.find(h->ip);
                if (it != freq.end()) {
                    h->usedSlots = std::max(0, h->usedSlots - it->second);
                }
            }
            // Compact: fill the fullest hosts first to empty out VMs
            std::sort(sorted.begin(),
                      sorted.end(),
                      [&](const HostPtr& a, const HostPtr& b) {
                          if (a->usedSlots != b->usedSlots) {
                              return a->usedSlots > b->usedSlots;
                          }
                          return moreCapacityFirst(a, b);
                      });
            break;
        }
        default:
            throw FaabricException("unrecognised decision type");
    }

    if (!binPackInto(sorted, req, *decision)) {
        return std::make_shared<SchedulingDecision>(NOT_ENOUGH_SLOTS,
                                                    NOT_ENOUGH_SLOTS);
    }

    if (decisionType == DecisionType::DIST_CHANGE) {
        auto oldDecision = inFlightReqs.at(req.appId).second;

        // Better = more completely-free hosts after the move
        auto countFreeWith = [&](const SchedulingDecision& d) {
            std::map<std::string, int> used;
            for (auto& [ip, h] : filtered) {
                used[ip] = h->usedSlots;
            }
            for (const auto& ip : d.hosts) {
                used[ip]++;
            }
            int free = 0;
            for (auto& [ip, u] : used) {
                if (u == 0) {
                    free++;
                }
            }
            return free;
        };
        if (countFreeWith(*decision) > countFreeWith(*oldDecision)) {
            return minimiseNumOfMigrations(decision, oldDecision);
        }
        return std::make_shared<SchedulingDecision>(DO_NOT_MIGRATE,
                                                    DO_NOT_MIGRATE);
    }
    return decision;
}

// ------------------------- Spot ---------------------------------------------

std::shared_ptr<SchedulingDecision> SpotScheduler::makeSchedulingDecision(
  HostMap& hostMap,
  const InFlightReqs& inFlightReqs,
  const BatchExecuteRequest& req)
{
    auto decision = std::make_shared<SchedulingDecision>(req.appId, 0);
    auto decisionType = getDecisionType(inFlightReqs, req);

    // Remove the to-be-evicted VMs (tagged MUST_EVICT_IP by the planner)
    std::set<std::string> evictedIps;
    HostMap filtered;
    for (auto& [ip, h] : hostMap) {
        if (h->ip == MUST_EVICT_IP) {
            evictedIps.insert(ip);
        } else {
            filtered[ip] = h;
        }
    }

    std::vector<HostPtr> sorted;
    for (auto& [ip, h] : filtered) {
        sorted.push_back(h);
    }

    std::map<std::string, int> freq;
    if (decisionType != DecisionType::NEW) {
        freq = hostFreqCountOf(*inFlightReqs.at(req.appId).second);
    }

    switch (decisionType) {
        case DecisionType::NEW:
            std::sort(sorted.begin(), sorted.end(), moreCapacityFirst);
            break;
        case DecisionType::SCALE_CHANGE:
            std::sort(sorted.begin(), sorted.end(), makeFreqComparator(freq));
            break;
        case DecisionType::DIST_CHANGE: {
            for (auto& h : sorted) {
                auto it = freq.find(h->ip);
                if (it != freq.end()) {
                    h->usedSlots = std::max(0, h->usedSlots - it->second);
                }
            }
            std::sort(sorted.begin(), sorted.end(), makeFreqComparator(freq));
            break;
        }
        default:
            throw FaabricException("unrecognised decision type");
    }

    bool fits = binPackInto(sorted, req, *decision);
    bool isDistChange = decisionType == DecisionType::DIST_CHANGE;

    if (!fits && !isDistChange) {
        return std::make_shared<SchedulingDecision>(NOT_ENOUGH_SLOTS,
                                                    NOT_ENOUGH_SLOTS);
    }

    if (isDistChange) {
        if (!fits) {
            // Messages on the doomed VM cannot be moved: freeze the app
            return std::make_shared<SchedulingDecision>(MUST_FREEZE,
                                                        MUST_FREEZE);
        }
        auto oldDecision = inFlightReqs.at(req.appId).second;
        for (const auto& ip : oldDecision->hosts) {
            if (evictedIps.count(ip) > 0) {
                return minimiseNumOfMigrations(decision, oldDecision);
            }
        }
        return std::make_shared<SchedulingDecision>(DO_NOT_MIGRATE,
                                                    DO_NOT_MIGRATE);
    }
    return decision;
}

// ------------------------- decision cache ------------------------------------

DecisionCache& DecisionCache::get()
{
    static DecisionCache cache;
    return cache;
}

std::string DecisionCache::keyOf(const BatchExecuteRequest& req)
{
    return req.user + "/" + req.function + "/" +
           std::to_string(req.messages.size());
}

std::shared_ptr<SchedulingDecision> DecisionCache::getCachedDecision(
  const BatchExecuteRequest& req)
{
    std::lock_guard<std::mutex> lock(mx);
    auto it = cache.find(keyOf(req));
    return it == cache.end() ? nullptr : it->second;
}

void DecisionCache::addCachedDecision(const BatchExecuteRequest& req,
                                      const SchedulingDecision& decision)
{
    std::lock_guard<std::mutex> lock(mx);
    cache[keyOf(req)] = std::make_shared<SchedulingDecision>(decision);
}

void DecisionCache::clear()
{
    std::lock_guard<std::mutex> lock(mx);
    cache.clear();
}

size_t DecisionCache::size()
{
    std::lock_guard<std::mutex> lock(mx);
    return cache.size();
}

// ------------------------- registry -----------------------------------------

static std::shared_ptr<BatchScheduler> currentScheduler;
static std::string currentMode;
static std::mutex schedulerMx;

std::shared_ptr<BatchScheduler> getBatchScheduler()
{
    std::lock_guard<std::mutex> lock(schedulerMx);
    if (currentScheduler != nullptr) {
        return currentScheduler;
    }
    currentMode = getSystemConfig().batchSchedulerMode;
    if (currentMode == "bin-pack") {
        currentScheduler = std::make_shared<BinPackScheduler>();
    } else if (currentMode == "compact") {
        currentScheduler = std::make_shared<CompactScheduler>();
    } else if (currentMode == "spot") {
        currentScheduler = std::make_shared<SpotScheduler>();
    } else {
        throw FaabricException("unrecognised batch scheduler mode: " +
                               currentMode);
    }
    return currentScheduler;
}

void resetBatchScheduler()
{
    std::lock_guard<std::mutex> lock(schedulerMx);
    currentScheduler = nullptr;
}

void resetBatchScheduler(const std::string& newMode)
{
    {
        std::lock_guard<std::mutex> lock(schedulerMx);
        currentScheduler = nullptr;
        getSystemConfig().batchSchedulerMode = newMode;
    }
    getBatchScheduler();
}

std::string getBatchSchedulerMode()
{
    getBatchScheduler();
    std::lock_guard<std::mutex> lock(schedulerMx);
    return currentMode;
}

} // namespace faabricamd
