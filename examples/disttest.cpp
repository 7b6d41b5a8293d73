// Multi-process dist scenarios under sanitizers: a planner + two worker
// PROCESSES from this same (ASan/TSan-instrumented) binary, exercising
// the cross-process transport, leader collectives, and live MPI-world
// migration — the paths the single-process selftest cannot reach.
// (Reference CI sanitizes its dist suite: .github/workflows/tests.yml;
// run via `make asan-dist` / `make tsan-dist`.)
//
// Roles: no args = main (planner + worker A + orchestration);
//        "worker2" = worker B, exits when the stop file appears.
#include "faabricamd/executor.h"
#include "faabricamd/messages.h"
#include "faabricamd/planner.h"
#include "faabricamd/runner.h"
#include "faabricamd/scheduler.h"
#include "faabricamd/scheduling.h"
#include "faabricamd/util.h"

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <string>
#include <sys/stat.h>
#include <sys/types.h>
#include <sys/wait.h>
#include <unistd.h>

using namespace faabricamd;

#define CHECK(cond)                                                         \
    do {                                                                    \
        if (!(cond)) {                                                      \
            fprintf(stderr, "DISTTEST FAIL %s:%d: %s\n", __FILE__,          \
                    __LINE__, #cond);                                       \
            exit(1);                                                        \
        }                                                                   \
    } while (0)

namespace faabricamd {
void registerMpiExampleFunctions();
}

static std::string stopFile()
{
    const char* f = getenv("DISTTEST_STOP_FILE");
    return f != nullptr ? f : "/tmp/disttest.stop";
}

static int basePortOffset()
{
    const char* e = getenv("DISTTEST_BASE_OFFSET");
    return e != nullptr ? atoi(e) : 6800;
}

static void awaitBatch(int32_t appId, int total, int timeoutMs)
{
    auto& client = getPlannerClient();
    int64_t deadline = getGlobalClockEpochMillis() + timeoutMs;
    while (true) {
        auto [finished, n] = client.getBatchStatusCounts(appId);
        if (finished && n >= total) {
            break;
        }
        CHECK(getGlobalClockEpochMillis() < deadline);
        usleep(5000);
    }
}

static int runWorker2()
{
    // Hard backstop: if the parent dies (aborted test run) or shutdown
    // wedges, never outlive the scenario and squat on the ports
    alarm(150);
    int off = basePortOffset() + 200;
    setPortOffset(off);
    getSystemConfig().endpointHost = "127.0.0.1@" + std::to_string(off);
    getSystemConfig().plannerHost =
      "127.0.0.1@" + std::to_string(basePortOffset());
    {
        HostResources res;
        res.slots = 2;
        Scheduler::get().setThisHostResources(res);
    }

    FaabricMain w(getExecutorFactory());
    w.startBackground();
    registerMpiExampleFunctions();

    std::string stop = stopFile();
    for (int i = 0; i < 2400; i++) { // up to 2 min
        struct stat st;
        if (::stat(stop.c_str(), &st) == 0) {
            break;
        }
        // Orphaned (parent died without writing the stop file)? Exit.
        if (getppid() == 1) {
            break;
        }
        usleep(50000);
    }
    w.shutdown();
    return 0;
}

int main(int argc, char** argv)
{
    if (argc > 1 && strcmp(argv[1], "worker2") == 0) {
        return runWorker2();
    }

    int base = basePortOffset();
    setPortOffset(base);
    getSystemConfig().endpointHost = "127.0.0.1@" + std::to_string(base);
    getSystemConfig().plannerHost = getSystemConfig().endpointHost;
    {
        HostResources res;
        res.slots = 2;
        Scheduler::get().setThisHostResources(res);
    }
    ::unlink(stopFile().c_str());

    PlannerRuntime planner;
    planner.start(false);
    FaabricMain w(getExecutorFactory());
    w.startBackground();
    registerMpiExampleFunctions();

    // Second worker: a fresh PROCESS from this same sanitized binary
    pid_t child = fork();
    CHECK(child >= 0);
    if (child == 0) {
        execl(argv[0], argv[0], "worker2", (char*)nullptr);
        _exit(127);
    }

    auto& client = getPlannerClient();
    {
        int64_t deadline = getGlobalClockEpochMillis() + 60000;
        while (client.getAvailableHosts().size() < 2) {
            CHECK(getGlobalClockEpochMillis() < deadline);
            usleep(20000);
        }
    }
    std::string w1 = "127.0.0.1@" + std::to_string(base);
    std::string w2 = "127.0.0.1@" + std::to_string(base + 200);

    // A. 4-rank MPI worlds spanning both workers: allreduce (leader
    // trees), cartesian topology, MPI_IN_PLACE collectives
    for (const char* fn : { "allreduce", "cartesian", "inplace" }) {
        auto ber = std::make_shared<BatchExecuteRequest>(
          batchExecFactory("mpi-cpp", fn, 1));
        ber->messages[0].isMpi = true;
        ber->messages[0].mpiWorldSize = 4;
        auto decision = client.callFunctions(ber);
        CHECK(decision->appId == ber->appId);
        awaitBatch(ber->appId, 4, 60000);
        auto status = client.getBatchResults(ber->appId);
        CHECK((int)status.messageResults.size() == 4);
        for (const auto& m : status.messageResults) {
            if (m.returnValue != 0) {
                fprintf(stderr, "rank %d: %s\n", m.mpiRank,
                        m.outputData.c_str());
            }
            CHECK(m.returnValue == 0);
        }
        printf("disttest: %s over 2 workers OK\n", fn);
    }

    // B. Live MPI-world migration: preload a 1+1 split; bin-pack
    // consolidates onto w1's free slot mid-app (DIST_CHANGE path,
    // exception unwind, world re-init on the destination)
    {
        auto ber = std::make_shared<BatchExecuteRequest>(
          batchExecFactory("mpi-cpp", "migrate", 1));
        ber->messages[0].isMpi = true;
        ber->messages[0].mpiWorldSize = 2;
        SchedulingDecision preload(ber->appId, 0);
        preload.addMessage(w1, 0, 0, 0);
        preload.addMessage(w2, 0, 1, 1);
        client.preloadSchedulingDecision(ber->appId, preload);
        auto decision = client.callFunctions(ber);
        CHECK(decision->appId == ber->appId);
        awaitBatch(ber->appId, 2, 90000);
        auto status = client.getBatchResults(ber->appId);
        int migrated = 0;
        for (const auto& m : status.messageResults) {
            CHECK(m.returnValue == 0);
            if (m.outputData == "migrated+rejoined") {
                migrated++;
            }
        }
        CHECK(migrated == 1);
        printf("disttest: live MPI migration OK\n");
    }

    // Stop the second worker, reap it
    {
        FILE* f = fopen(stopFile().c_str(), "w");
        CHECK(f != nullptr);
        fclose(f);
    }
    int st = 0;
    CHECK(waitpid(child, &st, 0) == child);
    CHECK(WIFEXITED(st) && WEXITSTATUS(st) == 0);
    ::unlink(stopFile().c_str());

    w.shutdown();
    planner.shutdown();
    printf("DISTTEST OK\n");
    return 0;
}
