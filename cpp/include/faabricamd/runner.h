// Per-host bring-up: start all RPC servers, register with the planner
// (reference: include/faabric/runner/FaabricMain.h:11-37,
//  src/runner/FaabricMain.cpp:11-41, src/planner/planner_server.cpp:9-43).
#pragma once

#include <memory>

#include "faabricamd/executor.h"

namespace faabricamd {

class FunctionCallServer;
class SnapshotServer;
class PointToPointServer;
class StateServer;
class PlannerServer;

// Worker-side runtime: the five per-host servers + planner registration
class FaabricMain
{
  public:
    explicit FaabricMain(std::shared_ptr<ExecutorFactory> factory);
    ~FaabricMain();

    void startBackground();
    void shutdown();

  private:
    std::unique_ptr<FunctionCallServer> functionServer;
    std::unique_ptr<SnapshotServer> snapshotServer;
    std::unique_ptr<PointToPointServer> ptpServer;
    std::unique_ptr<StateServer> stateServer; // wired in state.cpp phase
    bool started = false;
};

// Control-plane runtime: the planner server (plus its own snapshot server
// used to stage THREADS / freeze snapshots)
class PlannerRuntime
{
  public:
    PlannerRuntime();
    ~PlannerRuntime();
    // withStateServer hosts the global KV + scripted-lock store for the
    // "planner" state mode (the reference's Redis-service role)
    void start(bool withSnapshotServer = true,
               bool withStateServer = false);
    void shutdown();

  private:
    std::unique_ptr<PlannerServer> server;
    std::unique_ptr<SnapshotServer> snapshotServer;
    std::unique_ptr<StateServer> stateServer;
    bool started = false;
};

} // namespace faabricamd
