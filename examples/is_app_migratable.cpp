// `is_app_migratable <appId> [plannerHost]` — deployment CLI that asks the
// planner's HTTP ops API for the cluster state and answers whether the
// given in-flight app could be improved by a migration under the current
// scheduling policy (reference: src/planner/is_app_migratable.cpp:14 —
// same contract: exit 0 and print "YES" if a DIST_CHANGE pass would move
// the app, exit 0 + "NO" if not, exit 1 on lookup errors).
//
// Re-designed for this runtime: talks plain HTTP/1.1 to the planner
// endpoint (no Boost.Beast), then replays the decision locally with the
// same BatchScheduler code the planner itself runs, so the answer can't
// drift from planner behaviour.
#include <faabricamd/endpoint.h>
#include <faabricamd/json.h>
#include <faabricamd/messages.h>
#include <faabricamd/scheduling.h>
#include <faabricamd/transport.h>
#include <faabricamd/util.h>

#include <cstdio>
#include <cstdlib>
#include <cstring>
#include <sys/socket.h>

using namespace faabricamd;

static std::string httpOps(const std::string& host,
                           int port,
                           HttpMessageType type,
                           const std::string& payload)
{
    Json req = Json::object();
    req["http_type"] = Json((int64_t)type);
    req["payload"] = Json(payload);
    std::string body = req.dump();
    std::string msg = "POST / HTTP/1.1\r\nHost: " + host +
                      "\r\nContent-Type: application/json"
                      "\r\nContent-Length: " +
                      std::to_string(body.size()) +
                      "\r\nConnection: close\r\n\r\n" + body;

    TcpConnection conn = TcpConnection::dial(host, port);
    conn.sendAll(msg.data(), msg.size());

    std::string resp;
    char tmp[4096];
    while (true) {
        ssize_t n = ::recv(conn.rawFd(), tmp, sizeof(tmp), 0);
        if (n <= 0) {
            break;
        }
        resp.append(tmp, (size_t)n);
    }
    auto bodyAt = resp.find("\r\n\r\n");
    if (bodyAt == std::string::npos) {
        throw FaabricException("malformed HTTP response from planner");
    }
    if (resp.find("200") == std::string::npos ||
        resp.find("200") > resp.find("\r\n")) {
        throw FaabricException("planner ops call failed: " +
                               resp.substr(0, resp.find("\r\n")));
    }
    if (getenv("CLI_DEBUG") != nullptr) {
        fprintf(stderr, "RESP[%s]\n", resp.c_str());
    }
    return resp.substr(bodyAt + 4);
}

int main(int argc, char** argv)
{
    if (argc < 2) {
        fprintf(stderr, "usage: %s <appId> [plannerHost]\n", argv[0]);
        return 1;
    }
    setLogLevel(LogLevel::error);
    int32_t appId = (int32_t)atoll(argv[1]);
    std::string host = argc > 2 ? argv[2] : "127.0.0.1";
    int port = PLANNER_HTTP_PORT + getPortOffset();

    try {
        // 1. Find the app among in-flight apps
        Json inFlight =
          Json::parse(httpOps(host, port, HttpMessageType::GET_IN_FLIGHT_APPS,
                              ""));
        const Json* apps = inFlight.find("apps");
        std::vector<std::string> appHosts;
        bool found = false;
        if (apps != nullptr) {
            for (const Json& a : apps->items()) {
                if ((int32_t)a.getInt("appId") != appId) {
                    continue;
                }
                found = true;
                const Json* ips = a.find("hostIps");
                if (ips != nullptr) {
                    for (const Json& ip : ips->items()) {
                        appHosts.push_back(ip.asString());
                    }
                }
            }
        }
        if (!found) {
            fprintf(stderr, "app %d is not in flight\n", appId);
            return 1;
        }

        // 2. Cluster load
        Json hostsJson = Json::parse(
          httpOps(host, port, HttpMessageType::GET_AVAILABLE_HOSTS, ""));
        HostMap hostMap;
        const Json* hosts = hostsJson.find("hosts");
        if (hosts != nullptr) {
            for (const Json& h : hosts->items()) {
                std::string ip = h.getString("ip");
                hostMap[ip] = std::make_shared<HostState>(
                  ip, (int)h.getInt("slots"), (int)h.getInt("usedSlots"));
            }
        }

        // 3. Replay a DIST_CHANGE pass with the active policy
        std::string policy = "bin-pack";
        try {
            policy = Json::parse(httpOps(host, port,
                                         HttpMessageType::GET_POLICY, ""))
                       .getString("policy");
        } catch (...) {
        }
        resetBatchScheduler(policy);

        auto ber = std::make_shared<BatchExecuteRequest>(
          batchExecFactory("ops", "is-app-migratable", (int)appHosts.size()));
        ber->appId = appId;
        auto current = std::make_shared<SchedulingDecision>(appId, 0);
        for (size_t i = 0; i < appHosts.size(); i++) {
            current->addMessage(appHosts[i], ber->messages[i].id, (int32_t)i,
                                (int32_t)i);
        }
        InFlightReqs inFlightReqs;
        inFlightReqs[appId] = { ber, current };

        auto decision = getBatchScheduler()->makeSchedulingDecision(
          hostMap, inFlightReqs, *ber);
        bool migratable = decision->appId != DO_NOT_MIGRATE &&
                          decision->appId != NOT_ENOUGH_SLOTS &&
                          decision->hosts != appHosts;
        printf("%s\n", migratable ? "YES" : "NO");
        return 0;
    } catch (const std::exception& e) {
        fprintf(stderr, "is_app_migratable failed: %s\n", e.what());
        return 1;
    }
}
