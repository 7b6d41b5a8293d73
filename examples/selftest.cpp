// Self-contained C++ integration sweep, used for the sanitizer builds
// (`make asan-check` / `make tsan-check` — the reference runs its Catch2
// suites under Address/Thread/Undefined sanitizers in CI,
// .github/workflows/tests.yml). Exercises: batches, MPI world with all
// collectives (host plane), THREADS fork-join with a Sum merge region,
// snapshot diff/apply, chaining.
#include <faabricamd/executor.h>
#include <faabricamd/planner.h>
#include <faabricamd/runner.h>
#include <faabricamd/scheduler.h>
#include <faabricamd/snapshot.h>
#include <faabricamd/queue.h>
#include <faabricamd/util.h>

#include <cstdio>
#include <cstring>
#include <thread>

using namespace faabricamd;

namespace faabricamd {
void registerMpiExampleFunctions();
}

#define CHECK(cond)                                                            \
    do {                                                                       \
        if (!(cond)) {                                                         \
            fprintf(stderr, "SELFTEST FAILED at %s:%d: %s\n", __FILE__,        \
                    __LINE__, #cond);                                          \
            return 1;                                                          \
        }                                                                      \
    } while (0)

static int32_t forkParent(Message& msg)
{
    Executor* exec = ExecutorContext::get().getExecutor();
    exec->setMemorySize(8192);
    auto [base, size] = exec->getMemoryView();
    for (int i = 0; i < 8; i++) {
        int32_t v = 50;
        std::memcpy(base + i * 4, &v, 4);
    }
    auto req = std::make_shared<BatchExecuteRequest>(
      batchExecFactory("self", "forkchild", 3));
    std::vector<SnapshotMergeRegion> regions = {
        SnapshotMergeRegion(0, 32, SnapshotDataType::Int,
                            SnapshotMergeOperation::Sum),
    };
    auto results = exec->executeThreads(req, regions);
    if (results.size() != 3) {
        return 1;
    }
    auto [base2, size2] = exec->getMemoryView();
    for (int i = 0; i < 3; i++) {
        int32_t v = 0;
        std::memcpy(&v, base2 + i * 4, 4);
        if (v != 50 + (i + 1)) {
            return 2;
        }
    }
    return 0;
}

static int32_t forkChild(Message& msg)
{
    Executor* exec = ExecutorContext::get().getExecutor();
    auto [base, size] = exec->getMemoryView();
    int idx = msg.groupIdx;
    int32_t v = 0;
    std::memcpy(&v, base + (idx - 1) * 4, 4);
    v += idx;
    std::memcpy(base + (idx - 1) * 4, &v, 4);
    return 0;
}

static int32_t chainParent(Message& msg)
{
    int32_t childId = chainFunction("self", "noop", {});
    Message result = awaitChainedCall(childId, 20000);
    return result.returnValue;
}

int main()
{
    setLogLevel(LogLevel::error);
    int off = getEnvVarInt("FAABRIC_PORT_OFFSET", 6500);
    setPortOffset(off);
    std::string ident = "127.0.0.1@" + std::to_string(off);
    getSystemConfig().endpointHost = ident;
    getSystemConfig().plannerHost = ident;

    HostResources res;
    res.slots = 8;
    Scheduler::get().setThisHostResources(res);

    PlannerRuntime planner;
    planner.start(false);
    FaabricMain w(getExecutorFactory());
    w.startBackground();

    auto& reg = FunctionRegistry::get();
    reg.registerFunction("self", "noop", [](Message&) { return 0; });
    reg.registerFunction("self", "forkparent", forkParent);
    reg.registerFunction("self", "forkchild", forkChild);
    reg.registerFunction("self", "chainparent", chainParent);
    registerMpiExampleFunctions();

    auto& client = getPlannerClient();
    const auto& conf = getSystemConfig();

    // 1. Plain batch
    {
        auto ber = std::make_shared<BatchExecuteRequest>(
          batchExecFactory("self", "noop", 8));
        auto decision = client.callFunctions(ber);
        CHECK(decision->appId == ber->appId);
        for (const auto& m : ber->messages) {
            Message r = client.getMessageResult(ber->appId, m.id, 20000);
            CHECK(r.returnValue == 0);
        }
    }

    // 2. THREADS fork-join with Sum merge
    {
        auto ber = std::make_shared<BatchExecuteRequest>(
          batchExecFactory("self", "forkparent", 1));
        auto decision = client.callFunctions(ber);
        CHECK(decision->appId == ber->appId);
        Message r = client.getMessageResult(
          ber->appId, ber->messages[0].id, 30000);
        CHECK(r.returnValue == 0);
    }

    // 3. MPI world, all example programs (host data plane)
    for (const char* fn : { "allreduce", "ring", "async", "vcollectives" }) {
        auto ber = std::make_shared<BatchExecuteRequest>(
          batchExecFactory("mpi-cpp", fn, 1));
        ber->messages[0].isMpi = true;
        ber->messages[0].mpiWorldSize = 4;
        auto decision = client.callFunctions(ber);
        CHECK(decision->appId == ber->appId);
        // Wait for the whole world
        int64_t deadline = getGlobalClockEpochMillis() + 60000;
        while (true) {
            auto [finished, n] = client.getBatchStatusCounts(ber->appId);
            if (finished && n >= 4) {
                break;
            }
            CHECK(getGlobalClockEpochMillis() < deadline);
            usleep(5000);
        }
        auto status = client.getBatchResults(ber->appId);
        for (const auto& m : status.messageResults) {
            CHECK(m.returnValue == 0);
        }
    }

    // 4. Chaining
    {
        auto ber = std::make_shared<BatchExecuteRequest>(
          batchExecFactory("self", "chainparent", 1));
        client.callFunctions(ber);
        Message r = client.getMessageResult(
          ber->appId, ber->messages[0].id, 30000);
        CHECK(r.returnValue == 0);
    }

    // 5. SPSC FixedCapacityQueue under a producer/consumer pair (the
    // sanitizer builds verify the acquire/release protocol)
    {
        FixedCapacityQueue<int64_t> q(64);
        int64_t sum = 0;
        std::thread consumer([&] {
            for (int i = 0; i < 10000; i++) {
                sum += q.dequeue(10000);
            }
        });
        for (int i = 0; i < 10000; i++) {
            q.enqueue(i, 10000);
        }
        consumer.join();
        CHECK(sum == 10000LL * 9999 / 2);
        CHECK(q.size() == 0);
        bool threw = false;
        try {
            q.dequeue(30);
        } catch (QueueTimeoutException&) {
            threw = true;
        }
        CHECK(threw);
    }

    // 5b. MPMC SpinLockQueue: 4 producers x 4 consumers
    {
        SpinLockQueue<int64_t> q(128);
        std::atomic<int64_t> total{ 0 };
        std::vector<std::thread> ts;
        for (int c = 0; c < 4; c++) {
            ts.emplace_back([&] {
                for (int i = 0; i < 2500; i++) {
                    total += q.dequeue(10000);
                }
            });
        }
        for (int p = 0; p < 4; p++) {
            ts.emplace_back([&, p] {
                for (int i = 0; i < 2500; i++) {
                    q.enqueue(p * 2500 + i, 10000);
                }
            });
        }
        for (auto& t : ts) {
            t.join();
        }
        CHECK(total.load() == 10000LL * 9999 / 2);
        CHECK(q.size() == 0);
    }

    // 5c. ConcurrentMap under concurrent writers + getOrCreate races
    {
        ConcurrentMap<int, std::shared_ptr<int>> m;
        std::vector<std::thread> ts;
        std::atomic<int> creations{ 0 };
        std::atomic<int> mismatches{ 0 };
        for (int t = 0; t < 8; t++) {
            ts.emplace_back([&] {
                for (int k = 0; k < 256; k++) {
                    auto v = m.getOrCreate(k, [&] {
                        creations++;
                        return std::make_shared<int>(k * 7);
                    });
                    // No CHECK in lambdas: its return-1 path plus
                    // fall-through would be UB (value-returning lambda
                    // flowing off the end)
                    if (*v != k * 7) {
                        mismatches++;
                    }
                }
            });
        }
        for (auto& t : ts) {
            t.join();
        }
        CHECK(mismatches.load() == 0);
        CHECK(m.size() == 256);
        // getOrCreate must have created each key exactly once
        CHECK(creations.load() == 256);
        std::shared_ptr<int> got;
        CHECK(m.tryGet(100, got) && *got == 700);
        CHECK(m.erase(100));
        CHECK(!m.tryGet(100, got));
        int visited = 0;
        int feMismatches = 0;
        m.forEach([&](int k, const std::shared_ptr<int>& v) {
            if (*v != k * 7) {
                feMismatches++;
            }
            visited++;
        });
        CHECK(feMismatches == 0);
        CHECK(visited == 255);
    }

    // 6. Snapshot diff/apply semantics
    {
        std::vector<uint8_t> base(8192, 0);
        SnapshotData snap(base);
        snap.addMergeRegion(0, 16, SnapshotDataType::Int,
                            SnapshotMergeOperation::Sum);
        snap.fillGapsWithBytewiseRegions();
        std::vector<uint8_t> updated = base;
        int32_t v = 5;
        std::memcpy(updated.data(), &v, 4);
        updated[5000] = 9;
        auto diffs = snap.diffWithMemory(updated.data(), updated.size());
        CHECK(!diffs.empty());
        snap.applyDiffs(diffs);
        int32_t got = 0;
        std::memcpy(&got, snap.getDataPtr(), 4);
        CHECK(got == 5);
    }

    w.shutdown();
    planner.shutdown();
    (void)conf;
    printf("SELFTEST OK\n");
    return 0;
}
