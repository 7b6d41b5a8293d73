// Standalone planner binary (reference: src/planner/planner_server.cpp:9-43
// — the control-plane container's entrypoint). Runs the planner RPC
// server, its snapshot server and the HTTP ops endpoint until SIGINT/TERM.
#include <faabricamd/endpoint.h>
#include <faabricamd/runner.h>
#include <faabricamd/util.h>

#include <atomic>
#include <csignal>
#include <cstdio>
#include <thread>

using namespace faabricamd;

static std::atomic<bool> stop{ false };

static void onSignal(int)
{
    stop.store(true);
}

int main()
{
    getSystemConfig().print();
    PlannerRuntime planner;
    planner.start(/*withSnapshotServer=*/true);
    PlannerEndpoint http;
    http.start();

    signal(SIGINT, onSignal);
    signal(SIGTERM, onSignal);
    printf("planner running; ctrl-c to stop\n");
    while (!stop.load()) {
        std::this_thread::sleep_for(std::chrono::milliseconds(200));
    }
    http.stop();
    planner.shutdown();
    return 0;
}
