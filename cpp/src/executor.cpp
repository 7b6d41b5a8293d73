// Executor implementation (reference behavior: src/executor/Executor.cpp
// :38-212 pool lifecycle, :307-576 threadPoolThread, :580-590 claims).
#include "faabricamd/executor.h"
#include "faabricamd/utilextras.h"
#include "faabricamd/mpi.h"
#include "faabricamd/ops.h"
#include "faabricamd/planner.h"

#include <hip/hip_runtime.h>
#include "faabricamd/snapshot.h"
#include "faabricamd/util.h"

#include <algorithm>
#include <cstring>

namespace faabricamd {

// Defined below: group-commit result reporting to the planner
void resultBatcherEnqueue(std::shared_ptr<Message> msg);

// ----------------------------- context -------------------------------------

static thread_local std::shared_ptr<ExecutorContext> currentContext;

bool ExecutorContext::isSet()
{
    return currentContext != nullptr;
}

void ExecutorContext::set(Executor* executor,
                          std::shared_ptr<BatchExecuteRequest> req,
                          int msgIdx)
{
    auto ctx = std::make_shared<ExecutorContext>();
    ctx->executor = executor;
    ctx->req = std::move(req);
    ctx->msgIdx = msgIdx;
    currentContext = ctx;
}

void ExecutorContext::unset()
{
    currentContext = nullptr;
}

ExecutorContext& ExecutorContext::get()
{
    if (!currentContext) {
        throw FaabricException("no executor context set");
    }
    return *currentContext;
}

Message& ExecutorContext::getMsg()
{
    return req->messages.at(msgIdx);
}

// ----------------------------- registry ------------------------------------

FunctionRegistry& FunctionRegistry::get()
{
    static FunctionRegistry reg;
    return reg;
}

void FunctionRegistry::registerFunction(const std::string& user,
                                        const std::string& function,
                                        FaabricFunction fn)
{
    std::lock_guard<std::mutex> lock(mx);
    functions[user + "/" + function] = std::move(fn);
}

FaabricFunction* FunctionRegistry::getFunction(const std::string& user,
                                               const std::string& function)
{
    std::lock_guard<std::mutex> lock(mx);
    auto it = functions.find(user + "/" + function);
    return it == functions.end() ? nullptr : &it->second;
}

void FunctionRegistry::clear()
{
    std::lock_guard<std::mutex> lock(mx);
    functions.clear();
}

// ----------------------------- executor ------------------------------------

Executor::Executor(Message& msg)
  : id(funcToString(msg.user, msg.function, 0) + "_" +
       std::to_string(generateGid()))
  , boundMsg(msg)
{
    threadPoolSize = std::min(getUsableCores(), 64);
    threadTaskQueues.resize(threadPoolSize);
    for (auto& q : threadTaskQueues) {
        q = std::make_shared<Queue<ExecutorTask>>();
    }
    threadPoolThreads.resize(threadPoolSize);
    lastExecMs = getGlobalClockEpochMillis();
}

void Executor::cleanupForkSnapshot()
{
    if (lastForkSnapshotKey.empty()) {
        return;
    }
    DeviceSnapshotRegistry::get().deleteSnapshot(lastForkSnapshotKey);
    SnapshotRegistry::get().deleteSnapshot(lastForkSnapshotKey);
    for (const auto& host : lastForkRemoteHosts) {
        try {
            getSnapshotClient(host)->deleteSnapshot(lastForkSnapshotKey);
        } catch (const std::exception& e) {
            FAM_DEBUG("remote fork-snapshot delete failed: %s", e.what());
        }
    }
    lastForkSnapshotKey.clear();
    lastForkRemoteHosts.clear();
}

Executor::~Executor()
{
    try {
        cleanupForkSnapshot();
    } catch (...) {
    }
    shutdown();
    if (deviceArena != nullptr) {
        hipFree(deviceArena);
    }
}

void Executor::setDeviceMemorySize(size_t newSize)
{
    if (!gpuAvailable()) {
        throw FaabricException("device arena requires a GPU");
    }
    size_t rounded =
      (newSize + DEVICE_PAGE - 1) / DEVICE_PAGE * DEVICE_PAGE;
    if (rounded <= deviceArenaSize) {
        deviceArenaSize = rounded;
        return;
    }
    int dev = gpuDevice >= 0 ? gpuDevice : 0;
    if (hipSetDevice(dev) != hipSuccess) {
        throw FaabricException("hipSetDevice failed");
    }
    uint8_t* fresh = nullptr;
    if (hipMalloc(&fresh, rounded) != hipSuccess) {
        throw FaabricException("device arena alloc failed");
    }
    hipMemset(fresh, 0, rounded);
    if (deviceArena != nullptr) {
        hipMemcpy(fresh, deviceArena, deviceArenaSize,
                  hipMemcpyDeviceToDevice);
        hipFree(deviceArena);
    }
    deviceArena = fresh;
    deviceArenaSize = rounded;
}

std::pair<uint8_t*, size_t> Executor::getDeviceMemoryView()
{
    return { deviceArena, deviceArenaSize };
}

void Executor::shutdown()
{
    std::lock_guard<std::mutex> lock(threadsMx);
    for (int i = 0; i < threadPoolSize; i++) {
        if (threadPoolThreads[i].joinable()) {
            ExecutorTask stopTask;
            stopTask.stop = true;
            threadTaskQueues[i]->enqueue(stopTask);
        }
    }
    for (auto& t : threadPoolThreads) {
        if (t.joinable()) {
            t.join();
        }
    }
}

long Executor::getMillisSinceLastExec() const
{
    return (long)(getGlobalClockEpochMillis() - lastExecMs.load());
}

bool Executor::tryClaim()
{
    bool expected = false;
    return claimed.compare_exchange_strong(expected, true);
}

void Executor::claim()
{
    claimed.store(true);
}

void Executor::releaseClaim()
{
    claimed.store(false);
}

void Executor::executeTasks(std::vector<int> msgIdxs,
                            std::shared_ptr<BatchExecuteRequest> req)
{
    lastExecMs = getGlobalClockEpochMillis();
    batchCounter = std::make_shared<std::atomic<int>>((int)msgIdxs.size());

    // Restore from snapshot where requested (THREADS fork-join and
    // un-freeze paths; reference: src/executor/Executor.cpp:142-167)
    if (!req->snapshotKey.empty()) {
        restore(req->snapshotKey);
    } else if (!req->messages.empty() &&
               !req->messages[msgIdxs[0]].snapshotKey.empty() &&
               req->type != BatchExecuteType::THREADS) {
        const std::string& key = req->messages[msgIdxs[0]].snapshotKey;
        restore(key);
        // Migration/freeze snapshots are single-use: drop them after
        // landing in the arena (one leaked arena per migration
        // otherwise). Fork (THREADS) snapshots are reused and cleaned
        // by their own lifecycle.
        if (startsWith(key, "migration_")) {
            SnapshotRegistry::get().deleteSnapshot(key);
            DeviceSnapshotRegistry::get().deleteSnapshot(key);
        }
    }

    // Fault-driven dirty tracking of the restored arena (segfault mode;
    // compare mode diffs against the snapshot instead —
    // reference: src/executor/Executor.cpp:142-162 startTracking)
    if (req->type == BatchExecuteType::THREADS &&
        !req->snapshotKey.empty()) {
        auto tracker = getDirtyTracker();
        if (tracker->getType() == "segfault") {
            auto [base, size] = getMemoryView();
            if (size > 0) {
                tracker->startTracking(base, size);
            }
        }
    }

    std::lock_guard<std::mutex> lock(threadsMx);
    for (int msgIdx : msgIdxs) {
        int poolIdx = threadPoolSize == 0 ? 0 : msgIdx % threadPoolSize;
        if (!threadPoolThreads[poolIdx].joinable()) {
            threadPoolThreads[poolIdx] =
              std::thread([this, poolIdx] { threadPoolThread(poolIdx); });
        }
        threadTaskQueues[poolIdx]->enqueue(ExecutorTask(msgIdx, req));
    }
}

std::vector<std::pair<int32_t, int32_t>> Executor::executeThreads(
  std::shared_ptr<BatchExecuteRequest> req,
  const std::vector<SnapshotMergeRegion>& mergeRegions)
{
    // (elasticScaleHint on req lets the planner grow the fork to every
    // free slot on this host)
    // Fork-join over a shared memory snapshot
    // (reference: SURVEY §3.4; src/executor/Executor.cpp executeThreads)
    Message& parentMsg = ExecutorContext::get().getMsg();
    std::string key = getMainThreadSnapshotKey(
      parentMsg.user, parentMsg.function, parentMsg.appId);

    bool onDevice = hasDeviceArena();
    // A fork for a new app supersedes the previous app's snapshot
    {
        Message& pm = ExecutorContext::get().getMsg();
        std::string newKey = getMainThreadSnapshotKey(
          pm.user, pm.function, pm.appId);
        if (!lastForkSnapshotKey.empty() &&
            lastForkSnapshotKey != newKey) {
            cleanupForkSnapshot();
        }
    }
    std::shared_ptr<SnapshotData> snap;
    std::shared_ptr<DeviceSnapshot> dsnap;
    if (onDevice) {
        // GPU fork-join: the snapshot lives in HBM; diffs are XOR page
        // diffs from the gfx950 kernels (typed merge regions are a host
        // feature — reject them loudly rather than silently ignoring)
        if (!mergeRegions.empty()) {
            throw FaabricException(
              "typed merge regions unsupported on the device arena "
              "(XOR page diffing applies)");
        }
        auto& dreg = DeviceSnapshotRegistry::get();
        if (dreg.snapshotExists(key) &&
            dreg.getSnapshot(key)->size() == deviceArenaSize) {
            dsnap = dreg.getSnapshot(key);
        } else {
            dsnap = std::make_shared<DeviceSnapshot>(
              deviceArenaSize, gpuDevice >= 0 ? gpuDevice : 0);
            dreg.registerSnapshot(key, dsnap);
        }
        dsnap->captureFromDevice(deviceArena);
    } else {
        auto [memBase, memSize] = getMemoryView();
        auto& reg = SnapshotRegistry::get();
        if (reg.snapshotExists(key)) {
            snap = reg.getSnapshot(key);
            snap->copyInData(memBase, memSize, 0);
        } else {
            snap = std::make_shared<SnapshotData>(
              std::vector<uint8_t>(memBase, memBase + memSize));
            reg.registerSnapshot(key, snap);
        }
        snap->clearMergeRegions();
        for (const auto& r : mergeRegions) {
            snap->addMergeRegion(r.offset, r.length, r.dataType,
                                 r.operation);
        }
    }

    // Thread messages scale-change onto the running app
    req->appId = parentMsg.appId;
    req->type = BatchExecuteType::THREADS;
    req->snapshotKey = key;
    for (size_t i = 0; i < req->messages.size(); i++) {
        auto& m = req->messages[i];
        m.appId = parentMsg.appId;
        m.appIdx = (int32_t)i + 1;
        m.groupIdx = (int32_t)i + 1;
        m.snapshotKey = key;
        m.mainHost = getSystemConfig().endpointHost;
    }

    auto decision = getPlannerClient().callFunctions(req);
    if (decision->appId == NOT_ENOUGH_SLOTS) {
        throw FaabricException("not enough slots for thread fork");
    }

    // Ship the snapshot to the other hosts in the (updated) decision —
    // the planner cannot: the snapshot lives here, not in its registry
    lastForkSnapshotKey = key;
    {
        const std::string& thisHost = getSystemConfig().endpointHost;
        for (const auto& host : decision->uniqueHosts()) {
            if (host == thisHost) {
                continue;
            }
            if (std::find(lastForkRemoteHosts.begin(),
                          lastForkRemoteHosts.end(),
                          host) == lastForkRemoteHosts.end()) {
                lastForkRemoteHosts.push_back(host);
            }
            if (onDevice) {
                // Same-node workers stream this over xGMI (HIP IPC);
                // only cross-node peers fall back to a host copy
                getSnapshotClient(host)->pushDeviceSnapshotFromDevice(
                  key, dsnap->data(), dsnap->size());
            } else {
                getSnapshotClient(host)->pushSnapshot(key, *snap);
            }
        }
    }

    // Await every thread result, then merge the queued diffs and re-map
    // the merged snapshot over this executor's memory. The decision is
    // the source of truth for WHICH messages to wait on: an elastic
    // scale-up may have grown the fork beyond what was requested
    // (reference: src/planner/Planner.cpp:832-891)
    std::vector<std::pair<int32_t, int32_t>> results;
    const auto& conf = getSystemConfig();
    std::set<int32_t> waitIds;
    for (const auto& m : req->messages) {
        waitIds.insert(m.id);
    }
    for (int i = 0; i < decision->nFunctions; i++) {
        int32_t id = decision->messageIds[i];
        if (id != 0 && id != parentMsg.id) {
            waitIds.insert(id);
        }
    }
    for (int32_t id : waitIds) {
        Message result = getPlannerClient().getMessageResult(
          parentMsg.appId, id, conf.globalMessageTimeout);
        results.emplace_back(id, result.returnValue);
    }
    if (onDevice) {
        dsnap->applyQueuedPackedDiffs();
        hipError_t err = hipMemcpy(deviceArena,
                                   dsnap->data(),
                                   dsnap->size(),
                                   hipMemcpyDeviceToDevice);
        if (err != hipSuccess ||
            hipStreamSynchronize(nullptr) != hipSuccess) {
            throw FaabricException("device snapshot merge-back failed");
        }
    } else {
        snap->writeQueuedDiffs();
        auto [memBase2, memSize2] = getMemoryView();
        snap->mapToMemory(memBase2, memSize2);
        snap->clearMergeRegions();
    }
    return results;
}

int32_t Executor::executeTask(int threadPoolIdx,
                              int msgIdx,
                              std::shared_ptr<BatchExecuteRequest> req)
{
    (void)threadPoolIdx;
    Message& msg = req->messages.at(msgIdx);
    FaabricFunction* fn =
      FunctionRegistry::get().getFunction(msg.user, msg.function);
    if (fn == nullptr) {
        throw FaabricException("function not registered: " + msg.user + "/" +
                               msg.function);
    }
    return (*fn)(msg);
}

std::pair<uint8_t*, size_t> Executor::getMemoryView()
{
    return { dummyMemory.data(), dummyMemory.size() };
}

void Executor::setMemorySize(size_t newSize)
{
    dummyMemory.resize(newSize);
}

void Executor::restore(const std::string& snapshotKey)
{
    // Copy the snapshot over this executor's memory view (reference:
    // src/executor/Executor.cpp:640-654 mapToMemory). The snapshot may
    // arrive from the forking host slightly after the dispatch, so wait
    // for it (bounded by the bound timeout); device snapshots restore
    // into the HBM arena via a D2D copy.
    const auto& conf = getSystemConfig();
    int64_t deadline = getGlobalClockEpochMillis() + conf.boundTimeout;
    while (true) {
        if (DeviceSnapshotRegistry::get().snapshotExists(snapshotKey)) {
            auto dsnap =
              DeviceSnapshotRegistry::get().getSnapshot(snapshotKey);
            if (deviceArenaSize < dsnap->size()) {
                setDeviceMemorySize(dsnap->size());
            }
            hipError_t err = hipMemcpy(deviceArena,
                                       dsnap->data(),
                                       dsnap->size(),
                                       hipMemcpyDeviceToDevice);
            if (err != hipSuccess ||
                hipStreamSynchronize(nullptr) != hipSuccess) {
                throw FaabricException("device snapshot restore failed");
            }
            return;
        }
        if (SnapshotRegistry::get().snapshotExists(snapshotKey)) {
            auto snap = SnapshotRegistry::get().getSnapshot(snapshotKey);
            if (dummyMemory.size() < snap->getSize()) {
                setMemorySize(snap->getSize());
            }
            auto [base, size] = getMemoryView();
            snap->mapToMemory(base, size);
            return;
        }
        if (getGlobalClockEpochMillis() >= deadline) {
            throw FaabricException("snapshot not found: " + snapshotKey);
        }
        usleep(10 * 1000);
    }
}

void Executor::reset(Message& msg)
{
    (void)msg;
}

void Executor::flush()
{
    FunctionRegistry::get().clear();
}

void Executor::setThreadResult(Message& msg,
                               int32_t returnValue,
                               const std::string& key,
                               const std::vector<SnapshotDiffMsg>& diffs)
{
    (void)key;
    (void)diffs;
    msg.returnValue = returnValue;
}

void Executor::threadPoolThread(int poolIdx)
{
    const auto& conf = getSystemConfig();
    while (true) {
        ExecutorTask task;
        try {
            task = threadTaskQueues[poolIdx]->dequeue(conf.boundTimeout);
        } catch (const QueueTimeoutException&) {
            continue; // executor reaping handles true idleness
        }
        if (task.stop) {
            break;
        }

        Message& msg = task.req->messages.at(task.msgIdx);
        int32_t returnValue = 0;
        ExecutorContext::set(this, task.req, task.msgIdx);
        try {
            PROF_START(exec_task)
            returnValue = executeTask(poolIdx, task.msgIdx, task.req);
            PROF_END(exec_task)
            // Exec-graph detail: per-rank MPI message counters ride the
            // result (reference: mpi/MpiWorld.h:13-18 + Executor exec
            // graph details)
            if (msg.isMpi && msg.recordExecGraph &&
                MpiWorldRegistry::get().worldExists(msg.mpiWorldId)) {
                auto counts = MpiWorldRegistry::get()
                                .getWorld(msg.mpiWorldId)
                                .getMsgCountDetails(msg.mpiRank);
                for (const auto& [k, v] : counts) {
                    msg.intExecGraphDetails[k] = v;
                }
            }
        } catch (const FunctionMigratedException&) {
            FAM_DEBUG("task %d migrated", msg.id);
            returnValue = MIGRATED_FUNCTION_RETURN_VALUE;
        } catch (const FunctionFrozenException&) {
            FAM_DEBUG("task %d frozen", msg.id);
            returnValue = FROZEN_FUNCTION_RETURN_VALUE;
        } catch (const std::exception& e) {
            FAM_ERROR("task %d failed: %s", msg.id, e.what());
            returnValue = 1;
            msg.outputData = std::string("Task failed: ") + e.what();
        }
        ExecutorContext::unset();

        int remaining = batchCounter->fetch_sub(1) - 1;
        bool isLastInBatch = remaining == 0;
        handleTaskResult(msg, returnValue, task.req, isLastInBatch);
    }
}

void Executor::handleTaskResult(Message& msg,
                                int32_t returnValue,
                                std::shared_ptr<BatchExecuteRequest> req,
                                bool isLastInBatch)
{
    const auto& conf = getSystemConfig();
    msg.returnValue = returnValue;
    msg.executedHost = conf.endpointHost;
    msg.finishTimestamp = getGlobalClockEpochMillis();
    lastExecMs = getGlobalClockEpochMillis();

    bool isThreads = req->type == BatchExecuteType::THREADS;

    // Last local MPI rank done: release this host's world state (RCCL
    // comms, streams, queues) and drop the registry entry
    if (msg.isMpi && returnValue != MIGRATED_FUNCTION_RETURN_VALUE &&
        returnValue != FROZEN_FUNCTION_RETURN_VALUE &&
        MpiWorldRegistry::get().worldExists(msg.mpiWorldId)) {
        if (MpiWorldRegistry::get()
              .getWorld(msg.mpiWorldId)
              .rankFinished(msg.mpiRank)) {
            MpiWorldRegistry::get().clearWorld(msg.mpiWorldId);
        }
    }

    // THREADS: the last local thread diffs this executor's memory against
    // the shared snapshot and ships the typed diffs to the main host
    // (reference: src/executor/Executor.cpp:509-516, :684 mergeDirtyRegions)
    std::vector<SnapshotDiff> threadDiffs;
    bool deviceThreads =
      isThreads && !req->snapshotKey.empty() &&
      DeviceSnapshotRegistry::get().snapshotExists(req->snapshotKey);
    if (deviceThreads && isLastInBatch) {
        try {
            // GPU fork-join: diff the HBM arena against the device
            // snapshot, ship the compact XOR page diff
            auto dsnap =
              DeviceSnapshotRegistry::get().getSnapshot(req->snapshotKey);
            uint32_t nd = dsnap->diffXor(deviceArena);
            if (nd > 0) {
                std::vector<uint32_t> pages;
                std::vector<uint8_t> payload;
                dsnap->gatherLastDiffToHost(pages, payload);
                std::vector<uint8_t> packed(4 + pages.size() * 4 +
                                            payload.size());
                uint32_t n = (uint32_t)pages.size();
                std::memcpy(packed.data(), &n, 4);
                std::memcpy(packed.data() + 4,
                            pages.data(),
                            pages.size() * 4);
                std::memcpy(packed.data() + 4 + pages.size() * 4,
                            payload.data(),
                            payload.size());
                threadDiffs.emplace_back(SnapshotDataType::Raw,
                                         SnapshotMergeOperation::XorPages,
                                         0,
                                         packed.data(),
                                         packed.size());
            }
        } catch (const std::exception& e) {
            FAM_ERROR("device thread diff failed: %s", e.what());
        }
    } else if (isThreads && isLastInBatch && !req->snapshotKey.empty()) {
        try {
            auto snap = SnapshotRegistry::get().getSnapshot(req->snapshotKey);
            auto [base, size] = getMemoryView();
            snap->fillGapsWithBytewiseRegions();
            auto tracker = getDirtyTracker();
            if (tracker->getType() == "segfault" && size > 0) {
                // Fault-tracked pages only
                tracker->stopTracking(base, size);
                auto dirty = tracker->getDirtyPages(base, size);
                threadDiffs =
                  snap->diffWithDirtyRegions(base, size, dirty);
            } else {
                // Compare mode: every page is a candidate and the typed
                // merge regions refine (no mprotect on the HBM path)
                threadDiffs = snap->diffWithMemory(base, size);
            }
        } catch (const std::exception& e) {
            FAM_ERROR("thread diff failed: %s", e.what());
        }
    }

    // Claim-reset-release order matters
    // (reference: src/executor/Executor.cpp:537-552)
    if (isLastInBatch) {
        if (!isThreads) {
            try {
                reset(msg);
            } catch (const std::exception& e) {
                FAM_ERROR("executor reset failed: %s", e.what());
            }
        }
        releaseClaim();
    }

    if (isThreads && !req->snapshotKey.empty()) {
        // Route via the snapshot channel so the main host can queue the
        // diffs before the result lands (reference: setThreadResult
        // src/executor/Executor.cpp:271-299)
        const std::string& mainHost = msg.mainHost;
        if (mainHost == conf.endpointHost || mainHost.empty()) {
            if (!threadDiffs.empty()) {
                try {
                    if (deviceThreads) {
                        auto dsnap = DeviceSnapshotRegistry::get()
                                       .getSnapshot(req->snapshotKey);
                        for (auto& d : threadDiffs) {
                            dsnap->queuePackedDiff(d.dataCopy);
                        }
                    } else {
                        auto snap = SnapshotRegistry::get().getSnapshot(
                          req->snapshotKey);
                        snap->queueDiffs(threadDiffs);
                    }
                } catch (const std::exception& e) {
                    FAM_ERROR("queueing thread diffs failed: %s", e.what());
                }
            }
            auto resultMsg = std::make_shared<Message>(msg);
            PROF_START(result_rpc)
            getPlannerClient().setMessageResult(resultMsg);
            PROF_END(result_rpc)
        } else {
            getSnapshotClient(mainHost)->pushThreadResult(
              msg.appId, msg.id, returnValue, req->snapshotKey, threadDiffs);
        }
        return;
    }

    // Report the result to the planner through the result batcher:
    // group commit — a lone result flushes immediately, a burst of
    // finishing executors coalesces into one RPC while the previous
    // send is in flight
    resultBatcherEnqueue(std::make_shared<Message>(msg));
}

// ------------------------- result batcher -----------------------------------

namespace {

struct ResultBatcher
{
    std::mutex mx;
    std::condition_variable cv;
    std::vector<std::shared_ptr<Message>> queue;
    std::thread worker;
    bool stop = false;
    bool started = false;

    void ensureStarted()
    {
        std::lock_guard<std::mutex> lock(mx);
        if (started) {
            return;
        }
        started = true;
        worker = std::thread([this] { run(); });
    }

    void run()
    {
        std::vector<std::shared_ptr<Message>> batch;
        while (true) {
            {
                std::unique_lock<std::mutex> lock(mx);
                cv.wait(lock, [&] { return stop || !queue.empty(); });
                if (stop && queue.empty()) {
                    return;
                }
                batch.swap(queue);
            }
            try {
                PROF_START(result_batch_send)
                getPlannerClient().setMessageResultsBatch(batch);
                PROF_END(result_batch_send)
            } catch (const std::exception& e) {
                FAM_ERROR("result batch send failed: %s", e.what());
            }
            batch.clear();
        }
    }

    ~ResultBatcher()
    {
        {
            std::lock_guard<std::mutex> lock(mx);
            stop = true;
        }
        cv.notify_all();
        if (worker.joinable()) {
            worker.join();
        }
    }
};

ResultBatcher& resultBatcher()
{
    static ResultBatcher instance;
    return instance;
}

} // namespace

void resultBatcherEnqueue(std::shared_ptr<Message> msg)
{
    auto& b = resultBatcher();
    b.ensureStarted();
    {
        std::lock_guard<std::mutex> lock(b.mx);
        b.queue.push_back(std::move(msg));
    }
    b.cv.notify_one();
}

// ----------------------------- chaining -------------------------------------

int32_t chainFunction(const std::string& user,
                      const std::string& function,
                      const std::vector<uint8_t>& input)
{
    // Chained call: a SCALE_CHANGE single-message batch on the caller's
    // app, recorded in the parent's chainedMsgIds for the exec graph
    // (reference: Executor::addChainedMessage src/executor/Executor.cpp:656)
    Message& parent = ExecutorContext::get().getMsg();
    auto req = std::make_shared<BatchExecuteRequest>();
    req->appId = parent.appId;
    req->user = user;
    req->function = function;
    Message m = messageFactory(user, function);
    m.appId = parent.appId;
    // Chained messages need a group idx that cannot collide with gang
    // members (broker mappings are keyed per (groupId, idx))
    m.appIdx = parent.appIdx;
    m.groupIdx = 1000 + (int32_t)(generateGid() % 14000);
    m.inputData = input;
    m.recordExecGraph = parent.recordExecGraph;
    req->messages.push_back(m);

    auto decision = getPlannerClient().callFunctions(req);
    if (decision->appId == NOT_ENOUGH_SLOTS) {
        throw FaabricException("not enough slots for chained call");
    }
    parent.chainedMsgIds.push_back(m.id);
    return m.id;
}

Message awaitChainedCall(int32_t msgId, int timeoutMs)
{
    Message& parent = ExecutorContext::get().getMsg();
    return getPlannerClient().getMessageResult(parent.appId, msgId,
                                               timeoutMs);
}

// ----------------------------- factory -------------------------------------

static std::shared_ptr<ExecutorFactory> executorFactory =
  std::make_shared<ExecutorFactory>();

void setExecutorFactory(std::shared_ptr<ExecutorFactory> factory)
{
    executorFactory = std::move(factory);
}

std::shared_ptr<ExecutorFactory> getExecutorFactory()
{
    return executorFactory;
}

} // namespace faabricamd
