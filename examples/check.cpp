// Minimal C++ embedder check: subclass Executor, bring up a planner and a
// worker in-process, run a batch end-to-end (reference: examples/check.cpp
// + examples/server.cpp:7-56 — the embedder API parity check, BASELINE
// config 1).
#include <faabricamd/executor.h>
#include <faabricamd/planner.h>
#include <faabricamd/runner.h>
#include <faabricamd/scheduler.h>
#include <faabricamd/util.h>

#include <cstdio>

using namespace faabricamd;

class ExampleExecutor : public Executor
{
  public:
    using Executor::Executor;

    int32_t executeTask(int threadPoolIdx,
                        int msgIdx,
                        std::shared_ptr<BatchExecuteRequest> req) override
    {
        Message& msg = req->messages.at(msgIdx);
        msg.outputData = "Example executor ran " + msg.user + "/" +
                         msg.function + " idx " +
                         std::to_string(msg.appIdx);
        return 0;
    }
};

class ExampleExecutorFactory : public ExecutorFactory
{
  public:
    std::shared_ptr<Executor> createExecutor(Message& msg) override
    {
        return std::make_shared<ExampleExecutor>(msg);
    }
};

int main()
{
    // Identity carries the port offset so multiple deployments share an IP
    std::string ident =
      "127.0.0.1@" + std::to_string(getPortOffset());
    getSystemConfig().endpointHost = ident;
    getSystemConfig().plannerHost = ident;

    HostResources res;
    res.slots = 4;
    Scheduler::get().setThisHostResources(res);

    PlannerRuntime planner;
    planner.start(/*withSnapshotServer=*/false);

    FaabricMain w(std::make_shared<ExampleExecutorFactory>());
    w.startBackground();

    auto ber = std::make_shared<BatchExecuteRequest>(
      batchExecFactory("demo", "hello", 4));
    auto decision = getPlannerClient().callFunctions(ber);
    if (decision->appId != ber->appId) {
        fprintf(stderr, "scheduling failed\n");
        return 1;
    }

    int failures = 0;
    for (const auto& m : ber->messages) {
        Message result =
          getPlannerClient().getMessageResult(ber->appId, m.id, 10000);
        printf("msg %d -> rv=%d output=\"%s\"\n",
               result.id,
               result.returnValue,
               result.outputData.c_str());
        if (result.returnValue != 0) {
            failures++;
        }
    }

    w.shutdown();
    planner.shutdown();

    if (failures == 0) {
        printf("CHECK OK\n");
        return 0;
    }
    return 1;
}
