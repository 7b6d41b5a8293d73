// Long-running worker binary with a custom Executor (reference:
// examples/server.cpp:7-56 — the embedder's worker container entrypoint).
// Registers with the planner (PLANNER_HOST env), serves batches until
// SIGINT/TERM. Also registers the native benchmark + MPI example payloads
// so a planner-driven deployment can exercise the full surface.
#include <faabricamd/executor.h>
#include <faabricamd/runner.h>
#include <faabricamd/util.h>

#include <atomic>
#include <csignal>
#include <cstdio>
#include <thread>

using namespace faabricamd;

namespace faabricamd {
void registerBenchFunctions();
void registerMpiExampleFunctions();
}

class ExampleExecutor : public Executor
{
  public:
    using Executor::Executor;

    int32_t executeTask(int threadPoolIdx,
                        int msgIdx,
                        std::shared_ptr<BatchExecuteRequest> req) override
    {
        Message& msg = req->messages.at(msgIdx);
        // Registry functions first; unknown functions echo like the
        // reference's example executor
        if (FunctionRegistry::get().getFunction(msg.user, msg.function) !=
            nullptr) {
            return Executor::executeTask(threadPoolIdx, msgIdx, req);
        }
        msg.outputData = "Example executor ran " + msg.user + "/" +
                         msg.function;
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

static std::atomic<bool> stop{ false };

static void onSignal(int)
{
    stop.store(true);
}

int main()
{
    getSystemConfig().print();
    registerBenchFunctions();
    registerMpiExampleFunctions();

    FaabricMain w(std::make_shared<ExampleExecutorFactory>());
    w.startBackground();

    signal(SIGINT, onSignal);
    signal(SIGTERM, onSignal);
    printf("worker running; ctrl-c to stop\n");
    while (!stop.load()) {
        std::this_thread::sleep_for(std::chrono::milliseconds(200));
    }
    w.shutdown();
    return 0;
}
