// Executor implementation (reference behavior: src/executor/Executor.cpp
// :38-212 pool lifecycle, :307-576 threadPoolThread, :580-590 claims).
#include "faabricamd/executor.h"
#include "faabricamd/planner.h"
#include "faabricamd/util.h"

#include <algorithm>

namespace faabricamd {

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

Executor::~Executor()
{
    shutdown();
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
        restore(req->messages[msgIdxs[0]].snapshotKey);
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
    (void)snapshotKey; // wired to the snapshot registry in snapshot.cpp users
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
            returnValue = executeTask(poolIdx, task.msgIdx, task.req);
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

    // Report the result to the planner
    auto resultMsg = std::make_shared<Message>(msg);
    getPlannerClient().setMessageResult(resultMsg);
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
