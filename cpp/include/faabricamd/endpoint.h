// Planner HTTP ops API (reference: src/endpoint/FaabricEndpoint.cpp:12-60
// Boost.Beast server; src/planner/PlannerEndpointHandler.cpp:54-381
// handlers; planner.proto:34-66 HttpMessage). Re-implemented as a minimal
// threaded HTTP/1.1 server — the ops API is low-rate control traffic.
#pragma once

#include <atomic>
#include <string>
#include <thread>
#include <vector>

#include "faabricamd/transport.h"

namespace faabricamd {

inline constexpr int PLANNER_HTTP_PORT = 8080;

class PlannerEndpoint
{
  public:
    explicit PlannerEndpoint(int port = PLANNER_HTTP_PORT);
    ~PlannerEndpoint();

    void start();
    void stop();
    int boundPort() const { return port; }

    // (status code, body) — exposed for in-process tests
    std::pair<int, std::string> handle(const std::string& jsonBody);

  private:
    void acceptLoop();
    void connectionLoop(TcpConnection conn);

    int port;
    TcpListener listener;
    std::thread acceptThread;
    std::mutex threadsMx;
    std::vector<std::thread> connThreads;
    std::atomic<bool> running{ false };
};

// Execution-graph JSON for a finished (or running) app
// (reference: util/ExecGraph.h:19-59, src/util/ExecGraph.cpp:62)
std::string getExecGraphJson(int32_t appId, int32_t msgId);

} // namespace faabricamd
