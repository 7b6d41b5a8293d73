// Native benchmark payload functions (registered into the FunctionRegistry
// so the hot paths never touch Python). These implement the BASELINE.json
// workload shapes:
//  - "bench/kvtouch":   HBM/host state-KV touch, used by the 1024-function
//                       batch-throughput config
//  - "bench/rankstep":  one MPI rank of the composite benchmark step:
//                       256 MB fp32 allreduce over RCCL/xGMI (+ optional
//                       small alltoall) and, on rank 0, a batch submission
//                       of kvtouch functions per step
// (reference harness shapes: tests/dist/mpi/benchmarks/mpi_allreduce.cpp,
//  mpi_bench.cpp:18-45, tests/dist/scheduler/test_funcs.cpp)
#include "faabricamd/executor.h"
#include "faabricamd/messages.h"
#include "faabricamd/mpi.h"
#include "faabricamd/ops.h"
#include "faabricamd/planner.h"
#include "faabricamd/state.h"
#include "faabricamd/util.h"

#include <hip/hip_runtime.h>

#include <cmath>
#include <cstring>
#include <sstream>

namespace faabricamd {

// Parse "k1=v1;k2=v2;..." from inputData
static std::map<std::string, int64_t> parseParams(const Message& msg)
{
    std::map<std::string, int64_t> out;
    std::string s(msg.inputData.begin(), msg.inputData.end());
    std::stringstream ss(s);
    std::string item;
    while (std::getline(ss, item, ';')) {
        auto eq = item.find('=');
        if (eq != std::string::npos) {
            out[item.substr(0, eq)] = atoll(item.c_str() + eq + 1);
        }
    }
    return out;
}

static int32_t benchKvTouch(Message& msg)
{
    auto params = parseParams(msg);
    int64_t kvBytes = params.count("kvbytes") ? params["kvbytes"] : 4096;
    // Each host owns one KV; functions write disjoint-ish chunks
    static thread_local std::vector<uint8_t> buf;
    buf.resize(kvBytes);
    std::memset(buf.data(), (int)(msg.id & 0xff), kvBytes);

    // State lives in HBM when a GPU is present (north star), host
    // memory otherwise
    int nGpus = 0;
    (void)hipGetDeviceCount(&nGpus);
    auto kv = nGpus > 0
                ? State::get().getKVDevice("bench", "kv", 1024 * 1024)
                : State::get().getKV("bench", "kv", 1024 * 1024);
    uint64_t offset =
      ((uint64_t)(uint32_t)msg.id * 4096) % (1024 * 1024 - kvBytes);
    kv->setChunk(offset, buf.data(), kvBytes);
    kv->getChunk(offset, buf.data(), kvBytes);
    if (buf[0] != (uint8_t)(msg.id & 0xff)) {
        return 1;
    }
    return 0;
}

static int32_t benchRankStep(Message& msg)
{
    auto& ctx = getMpiContext();
    if (msg.mpiRank == 0) {
        ctx.createWorld(msg);
    } else {
        ctx.joinWorld(msg);
    }
    MpiWorld& world = ctx.getWorld();
    int rank = ctx.getRank();
    int worldSize = world.getSize();

    auto params = parseParams(msg);
    int steps = (int)(params.count("steps") ? params["steps"] : 5);
    int warmup = (int)(params.count("warmup") ? params["warmup"] : 2);
    int64_t bytes =
      params.count("bytes") ? params["bytes"] : 256LL * 1024 * 1024;
    int batchPerHost = (int)(params.count("batch") ? params["batch"] : 0);
    int64_t kvBytes = params.count("kvbytes") ? params["kvbytes"] : 4096;
    int64_t a2aBytes = params.count("a2abytes") ? params["a2abytes"] : 0;
    int64_t ppBytes = params.count("ppbytes") ? params["ppbytes"] : 0;
    int64_t snapBytes =
      params.count("snapbytes") ? params["snapbytes"] : 0;

    int nGpus = 0;
    (void)hipGetDeviceCount(&nGpus);
    bool onGpu = nGpus > 0;

    size_t count = (size_t)bytes / sizeof(float);
    uint8_t* sendBuf = nullptr;
    uint8_t* recvBuf = nullptr;
    uint8_t* a2aSend = nullptr;
    uint8_t* a2aRecv = nullptr;
    std::vector<uint8_t> hostSend;
    std::vector<uint8_t> hostRecv2;
    std::vector<uint8_t> hostA2a;
    std::vector<uint8_t> hostA2aRecv;

    if (onGpu) {
        if (hipSetDevice(getSystemConfig().gpuDevice) != hipSuccess ||
            hipMalloc(&sendBuf, bytes) != hipSuccess ||
            hipMalloc(&recvBuf, bytes) != hipSuccess) {
            msg.outputData = "hip alloc failed";
            return 1;
        }
        // Non-trivial payload (DVFS give-back on zero-filled data)
        hipMemset(sendBuf, 0x3f, bytes);
        if (a2aBytes > 0) {
            hipMalloc(&a2aSend, a2aBytes * worldSize);
            hipMalloc(&a2aRecv, a2aBytes * worldSize);
            hipMemset(a2aSend, 0x11, a2aBytes * worldSize);
        }
    } else {
        hostSend.assign(bytes, 0x3f);
        hostRecv2.assign(bytes, 0);
        sendBuf = hostSend.data();
        recvBuf = hostRecv2.data();
        if (a2aBytes > 0) {
            hostA2a.assign(a2aBytes * worldSize, 0x11);
            hostA2aRecv.assign(a2aBytes * worldSize, 0);
            a2aSend = hostA2a.data();
            a2aRecv = hostA2aRecv.data();
        }
    }
    MpiBufferLoc loc = onGpu ? MpiBufferLoc::DEVICE : MpiBufferLoc::HOST;

    std::vector<double> stepMs;
    std::vector<double> allreduceMs;
    std::vector<double> batchMs;
    std::vector<double> pingpongMs;
    std::vector<double> ringMs;

    // Config 4 on this rank's GPU: diff+merge of a RANDOM-BYTE region
    // with RANDOMLY SCATTERED dirty pages (default 25%), measured before
    // the timed steps. Two accountings are reported: "algorithmic"
    // (2×region + dirty bytes — what the diff kernel touches) and
    // "region-normalized" (region bytes ÷ diff time).
    double snapDiffGbps = 0;
    double snapApplyGbps = 0;
    double snapDiffRegionGbps = 0;
    int64_t snapDirtyPct =
      params.count("snapdirty") ? params["snapdirty"] : 25;
    if (onGpu && snapBytes > 0) {
        try {
            DeviceSnapshot snap((size_t)snapBytes);
            uint8_t* updatedBuf = nullptr;
            if (hipMalloc(&updatedBuf, snapBytes) == hipSuccess) {
                (void)famFillRandom(updatedBuf, (uint64_t)snapBytes,
                                    0xfeedULL, nullptr);
                (void)hipDeviceSynchronize();
                snap.captureFromDevice(updatedBuf);
                // Scattered dirty set: every page index hashed, take the
                // fraction (uniform pseudo-random, no contiguity)
                uint32_t nPages = (uint32_t)(snapBytes / 4096);
                uint32_t nDirty =
                  (uint32_t)((uint64_t)nPages * snapDirtyPct / 100);
                std::vector<uint32_t> dirtyPages;
                dirtyPages.reserve(nDirty);
                uint64_t h = 0x2545F4914F6CDD1DULL;
                for (uint32_t p = 0; p < nPages && nDirty > 0; p++) {
                    h ^= h << 13;
                    h ^= h >> 7;
                    h ^= h << 17;
                    if ((h % 100) < (uint64_t)snapDirtyPct &&
                        dirtyPages.size() < nDirty) {
                        dirtyPages.push_back(p);
                    }
                }
                uint32_t* pagesDev = nullptr;
                (void)hipMalloc(&pagesDev,
                                dirtyPages.size() * sizeof(uint32_t));
                for (int it = 0; it < 3; it++) {
                    (void)hipMemcpy(pagesDev, dirtyPages.data(),
                                    dirtyPages.size() * sizeof(uint32_t),
                                    hipMemcpyHostToDevice);
                    (void)famTouchPages(updatedBuf, pagesDev,
                                        (uint32_t)dirtyPages.size(),
                                        0xbeef + it, nullptr);
                    (void)hipDeviceSynchronize();
                    int64_t s0 = getEpochMicros();
                    uint32_t nd = snap.diffXor(updatedBuf);
                    int64_t s1 = getEpochMicros();
                    snap.applyLastDiff();
                    int64_t s2 = getEpochMicros();
                    if (it == 2 && s1 > s0 && s2 > s1) {
                        double dirtyB = (double)nd * 4096;
                        snapDiffGbps = (2.0 * snapBytes + dirtyB) /
                                       ((s1 - s0) / 1e6) / 1e9;
                        snapDiffRegionGbps =
                          (double)snapBytes / ((s1 - s0) / 1e6) / 1e9;
                        snapApplyGbps =
                          (3.0 * dirtyB) / ((s2 - s1) / 1e6) / 1e9;
                    }
                }
                (void)hipFree(pagesDev);
                (void)hipFree(updatedBuf);
            }
        } catch (const std::exception& e) {
            FAM_ERROR("snapshot sub-bench failed: %s", e.what());
        }
    }

    for (int iter = 0; iter < warmup + steps; iter++) {
        world.barrier(rank);
        int64_t t0 = getEpochMicros();

        world.allReduce(rank,
                        sendBuf,
                        recvBuf,
                        MpiDataType::FLOAT,
                        (int)count,
                        MpiOp::SUM,
                        loc);
        int64_t tAr = getEpochMicros();

        if (a2aBytes > 0) {
            world.allToAll(rank,
                           a2aSend,
                           a2aRecv,
                           MpiDataType::BYTE,
                           (int)a2aBytes,
                           loc);
        }

        // Config 2: rank0<->rank1 ping-pong over xGMI (measured when the
        // world has at least 2 ranks; uses the allreduce buffers)
        int64_t tPp0 = getEpochMicros();
        if (ppBytes > 0 && worldSize >= 2 && rank <= 1) {
            int peer = 1 - rank;
            if (rank == 0) {
                world.send(0, peer, sendBuf, MpiDataType::BYTE,
                           (int)ppBytes, MpiMessageType::NORMAL, loc);
                world.recv(peer, 0, recvBuf, MpiDataType::BYTE,
                           (int)ppBytes, MpiMessageType::NORMAL, loc);
            } else {
                world.recv(peer, 1, recvBuf, MpiDataType::BYTE,
                           (int)ppBytes, MpiMessageType::NORMAL, loc);
                world.send(1, peer, sendBuf, MpiDataType::BYTE,
                           (int)ppBytes, MpiMessageType::NORMAL, loc);
            }
        }
        int64_t tPp1 = getEpochMicros();

        // Ring exchange over xGMI (the sendRecv ring is the
        // primitive a ring-attention/sequence-parallel layer would use):
        // every rank sendRecvs ppBytes to next/from prev in one step
        int64_t tRing0 = getEpochMicros();
        if (ppBytes > 0 && worldSize >= 2) {
            int next = (rank + 1) % worldSize;
            int prev = (rank + worldSize - 1) % worldSize;
            world.sendRecv(sendBuf, (int)ppBytes, MpiDataType::BYTE, next,
                           recvBuf, (int)ppBytes, MpiDataType::BYTE, prev,
                           rank);
        }
        int64_t tRing1 = getEpochMicros();

        // Batch-throughput half of the composite step: rank 0 submits a
        // batch of kvtouch functions across all hosts and waits
        int64_t tBatch0 = getEpochMicros();
        if (batchPerHost > 0 && rank == 0) {
            int total = batchPerHost * worldSize;
            auto ber = std::make_shared<BatchExecuteRequest>(
              batchExecFactory("bench", "kvtouch", total));
            std::string kvParams = "kvbytes=" + std::to_string(kvBytes);
            for (auto& m : ber->messages) {
                m.inputData.assign(kvParams.begin(), kvParams.end());
            }
            auto decision = getPlannerClient().callFunctions(ber);
            if (decision->appId == NOT_ENOUGH_SLOTS) {
                msg.outputData = "batch bench: not enough slots";
                break;
            }
            // Event-driven completion: planner pushes BATCH_DONE when
            // the last result lands (no status polls contending with
            // result ingestion on the planner lock)
            getPlannerClient().waitBatchDone(ber->appId, 120000);
        }
        world.barrier(rank); // every host's batch functions returned
        // Group-commit: drain this host's pending HBM KV writes inside
        // the timed region, then re-sync the world so t1 covers every
        // host's durability point
        State::get().syncAll();
        world.barrier(rank);
        int64_t t1 = getEpochMicros();

        if (iter >= warmup) {
            stepMs.push_back((t1 - t0) / 1000.0);
            allreduceMs.push_back((tAr - t0) / 1000.0);
            batchMs.push_back((t1 - tBatch0) / 1000.0);
            pingpongMs.push_back((tPp1 - tPp0) / 1000.0);
            ringMs.push_back((tRing1 - tRing0) / 1000.0);
        }
    }

    if (onGpu) {
        hipFree(sendBuf);
        hipFree(recvBuf);
        if (a2aSend != nullptr) {
            hipFree(a2aSend);
            hipFree(a2aRecv);
        }
    }

    std::ostringstream out;
    out << "step:";
    for (size_t i = 0; i < stepMs.size(); i++) {
        out << (i ? "," : "") << stepMs[i];
    }
    out << ";ar:";
    for (size_t i = 0; i < allreduceMs.size(); i++) {
        out << (i ? "," : "") << allreduceMs[i];
    }
    out << ";snapdiff:" << snapDiffGbps << ";snapapply:" << snapApplyGbps
        << ";snapdiffregion:" << snapDiffRegionGbps
        << ";snapdirtypct:" << snapDirtyPct;
    out << ";ring:";
    for (size_t i = 0; i < ringMs.size(); i++) {
        out << (i ? "," : "") << ringMs[i];
    }
    out << ";pp:";
    for (size_t i = 0; i < pingpongMs.size(); i++) {
        out << (i ? "," : "") << pingpongMs[i];
    }
    out << ";batch:";
    for (size_t i = 0; i < batchMs.size(); i++) {
        out << (i ? "," : "") << batchMs[i];
    }
    msg.outputData = out.str();
    return 0;
}

void registerBenchFunctions()
{
    FunctionRegistry::get().registerFunction(
      "bench", "sleep", [](Message& msg) {
          int ms = 1000;
          if (!msg.inputData.empty()) {
              ms = atoi(std::string(msg.inputData.begin(),
                                    msg.inputData.end())
                          .c_str());
          }
          std::this_thread::sleep_for(std::chrono::milliseconds(ms));
          return 0;
      });
    FunctionRegistry::get().registerFunction("bench", "kvtouch",
                                             benchKvTouch);
    FunctionRegistry::get().registerFunction("bench", "rankstep",
                                             benchRankStep);
}

} // namespace faabricamd
