// Migration / freeze points called from inside running functions
// (reference behavior: tests/dist/mpi/mpi_native.cpp:783-905
// mpiMigrationPoint — in Faasm this glue lives in the WASM host layer;
// here it is part of the runtime so any registered function can call it).
#include "faabricamd/executor.h"
#include <hip/hip_runtime.h>
#include "faabricamd/mpi.h"
#include "faabricamd/planner.h"
#include "faabricamd/ptp.h"
#include "faabricamd/scheduler.h"
#include "faabricamd/snapshot.h"
#include "faabricamd/util.h"

namespace faabricamd {

// Returns 0 (continue), MIGRATED_FUNCTION_RETURN_VALUE (this function was
// shipped to another host and must unwind) or FROZEN_FUNCTION_RETURN_VALUE
// (the app is freezing). The reentry input is what the re-started function
// receives as inputData on the destination host.
int32_t migrationPoint(const std::vector<uint8_t>& reentryInput)
{
    auto& ctx = ExecutorContext::get();
    Message& call = ctx.getMsg();
    Executor* exec = ctx.getExecutor();

    auto migration = Scheduler::get().checkForMigrationOpportunities(call);
    if (migration == nullptr) {
        return 0;
    }

    bool mustFreeze = migration->appId == MUST_FREEZE;
    if (mustFreeze) {
        // Snapshot this function's memory into the planner so it can be
        // restored on un-freeze (reference: mpi_native.cpp:795-821)
        std::string snapKey = "migration_" + std::to_string(call.id);
        call.inputData = reentryInput;
        call.snapshotKey = snapKey;

        if (exec->hasDeviceArena()) {
            // GPU-resident function: freeze the HBM arena. Same-node it
            // streams over xGMI (HIP IPC); otherwise it travels as bytes
            // and lands as a DeviceSnapshot wherever the app thaws
            auto [dbase, dsize] = exec->getDeviceMemoryView();
            try {
                getSnapshotClient(getSystemConfig().plannerHost)
                  ->pushDeviceSnapshotFromDevice(snapKey, dbase, dsize);
            } catch (const std::exception& e) {
                FAM_ERROR("freeze device snapshot push failed: %s",
                          e.what());
            }
        } else {
            auto [base, size] = exec->getMemoryView();
            auto snap = std::make_shared<SnapshotData>(
              std::vector<uint8_t>(base, base + size));
            try {
                getSnapshotClient(getSystemConfig().plannerHost)
                  ->pushSnapshot(snapKey, *snap);
            } catch (const std::exception& e) {
                FAM_ERROR("freeze snapshot push failed: %s", e.what());
            }
            // The planner owns the frozen state now; on thaw it pushes
            // to whichever host is chosen (keeping a local copy here
            // would leak one arena per freeze)
        }
        if (call.isMpi &&
            MpiWorldRegistry::get().worldExists(call.mpiWorldId)) {
            MpiWorldRegistry::get().getWorld(call.mpiWorldId).destroy();
            MpiWorldRegistry::get().clearWorld(call.mpiWorldId);
        }
        return FROZEN_FUNCTION_RETURN_VALUE;
    }

    bool funcMustMigrate = migration->srcHost != migration->dstHost;

    // The app has a new distribution and hence a new PTP group
    call.groupId = migration->groupId;
    if (call.isMpi &&
        MpiWorldRegistry::get().worldExists(call.mpiWorldId)) {
        MpiWorldRegistry::get()
          .getWorld(call.mpiWorldId)
          .prepareMigration(call.mpiRank);
    }

    if (!funcMustMigrate) {
        return 0;
    }

    // Ship this function to its new host: snapshot + MIGRATION batch sent
    // DIRECTLY to the destination (the planner already re-accounted the
    // slots during the DIST_CHANGE; reference: mpi_native.cpp:846-905)
    auto req = std::make_shared<BatchExecuteRequest>(
      batchExecFactory(call.user, call.function, 1));
    req->type = BatchExecuteType::MIGRATION;
    updateBatchExecAppId(*req, call.appId);
    updateBatchExecGroupId(*req, migration->groupId);

    Message& msg = req->messages[0];
    msg.inputData = reentryInput;
    msg.id = call.id;
    msg.appIdx = call.appIdx;
    msg.groupIdx = call.groupIdx;
    msg.mainHost = call.mainHost;
    msg.recordExecGraph = call.recordExecGraph;
    if (call.isMpi) {
        msg.isMpi = true;
        msg.mpiWorldId = call.mpiWorldId;
        msg.mpiWorldSize = call.mpiWorldSize;
        msg.mpiRank = call.mpiRank;
    }

    std::string snapKey = "migration_" + std::to_string(msg.id);
    if (exec->hasDeviceArena()) {
        // Ship the HBM arena; the destination lands it in ITS GPU and
        // restore() D2D-copies it into the fresh executor's arena.
        // Same-node destinations stream over xGMI (HIP IPC)
        auto [dbase, dsize] = exec->getDeviceMemoryView();
        getSnapshotClient(migration->dstHost)
          ->pushDeviceSnapshotFromDevice(snapKey, dbase, dsize);
        msg.snapshotKey = snapKey;
    } else {
        auto [base, size] = exec->getMemoryView();
        if (size > 0) {
            auto snap = std::make_shared<SnapshotData>(
              std::vector<uint8_t>(base, base + size));
            getSnapshotClient(migration->dstHost)
              ->pushSnapshot(snapKey, *snap);
            msg.snapshotKey = snapKey;
            // No local registration: the destination restores and
            // deletes it (single-use)
        }
    }

    FAM_INFO("migrating %s idx %d from %s to %s",
             funcToString(call.user, call.function, 0).c_str(),
             call.groupIdx,
             migration->srcHost.c_str(),
             migration->dstHost.c_str());
    getFunctionCallClient(migration->dstHost)->executeFunctions(*req);

    // MPI world on this host must drop the evacuated rank
    if (call.isMpi &&
        MpiWorldRegistry::get().worldExists(call.mpiWorldId)) {
        MpiWorldRegistry::get().getWorld(call.mpiWorldId).destroy();
        MpiWorldRegistry::get().clearWorld(call.mpiWorldId);
    }
    return MIGRATED_FUNCTION_RETURN_VALUE;
}

} // namespace faabricamd
