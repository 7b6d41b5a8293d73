// Per-host bring-up (reference: src/runner/FaabricMain.cpp:18-41,
// src/planner/planner_server.cpp:9-43).
#include "faabricamd/runner.h"
#include "faabricamd/planner.h"
#include "faabricamd/ptp.h"
#include "faabricamd/scheduler.h"
#include "faabricamd/snapshot.h"
#include "faabricamd/state.h"
#include "faabricamd/util.h"

namespace faabricamd {

FaabricMain::FaabricMain(std::shared_ptr<ExecutorFactory> factory)
{
    setExecutorFactory(std::move(factory));
}

FaabricMain::~FaabricMain()
{
    shutdown();
}

void FaabricMain::startBackground()
{
    if (started) {
        return;
    }
    const auto& conf = getSystemConfig();
    FAM_INFO("starting worker runtime on %s (port offset %d)",
             conf.endpointHost.c_str(),
             getPortOffset());

    // Drop any cached outbound connections from a previous runtime in this
    // process (tests restart runtimes in-process)
    clearFunctionCallClients();
    clearSnapshotClients();
    clearStateClients();
    getPointToPointBroker().clearClients();

    functionServer = std::make_unique<FunctionCallServer>();
    functionServer->start();
    snapshotServer = std::make_unique<SnapshotServer>();
    snapshotServer->start();
    ptpServer = std::make_unique<PointToPointServer>();
    ptpServer->start();
    stateServer = std::make_unique<StateServer>();
    stateServer->start();

    Scheduler::get().startReaper();

    // Register with the planner and keep the registration alive
    Host host;
    host.ip = conf.endpointHost;
    host.slots = Scheduler::get().getThisHostResources().slots;
    getPlannerClient().registerHost(host, true);
    getPlannerClient().startKeepAlive();

    started = true;
}

void FaabricMain::shutdown()
{
    if (!started) {
        return;
    }
    started = false;
    getPlannerClient().stopKeepAlive();
    try {
        Host host;
        host.ip = getSystemConfig().endpointHost;
        getPlannerClient().removeHost(host);
    } catch (const std::exception& e) {
        FAM_WARN("host removal failed: %s", e.what());
    }
    Scheduler::get().shutdown();
    if (functionServer) {
        functionServer->stop();
    }
    if (snapshotServer) {
        snapshotServer->stop();
    }
    if (ptpServer) {
        ptpServer->stop();
    }
    if (stateServer) {
        stateServer->stop();
    }
    functionServer.reset();
    snapshotServer.reset();
    ptpServer.reset();
    stateServer.reset();
}

PlannerRuntime::PlannerRuntime() = default;

PlannerRuntime::~PlannerRuntime()
{
    shutdown();
}

void PlannerRuntime::start(bool withSnapshotServer, bool withStateServer)
{
    if (started) {
        return;
    }
    server = std::make_unique<PlannerServer>();
    server->start();
    if (withSnapshotServer) {
        snapshotServer = std::make_unique<SnapshotServer>();
        snapshotServer->start();
    }
    if (withStateServer) {
        stateServer = std::make_unique<StateServer>();
        stateServer->start();
    }
    started = true;
    FAM_INFO("planner runtime started (port offset %d)", getPortOffset());
}

void PlannerRuntime::shutdown()
{
    if (!started) {
        return;
    }
    started = false;
    if (server) {
        server->stop();
    }
    if (snapshotServer) {
        snapshotServer->stop();
    }
    if (stateServer) {
        stateServer->stop();
    }
    server.reset();
    snapshotServer.reset();
    stateServer.reset();
}

} // namespace faabricamd
