// HTTP ops endpoint + exec graph (reference:
// src/planner/PlannerEndpointHandler.cpp:54-381, src/util/ExecGraph.cpp).
#include "faabricamd/endpoint.h"
#include "faabricamd/json.h"
#include "faabricamd/planner.h"
#include "faabricamd/ptp.h"
#include "faabricamd/scheduling.h"
#include "faabricamd/util.h"

#include <cstring>
#include <sys/socket.h>

namespace faabricamd {

// ------------------------- exec graph ---------------------------------------

static Json execGraphNode(
  const std::map<int32_t, std::shared_ptr<Message>>& results,
  int32_t msgId)
{
    Json node = Json::object();
    auto it = results.find(msgId);
    if (it == results.end()) {
        Json stub = Json::object();
        stub["id"] = Json((int64_t)msgId);
        node["msg"] = stub;
        return node;
    }
    node["msg"] = messageToJson(*it->second);
    Json chained = Json::array();
    for (int32_t childId : it->second->chainedMsgIds) {
        chained.push(execGraphNode(results, childId));
    }
    if (chained.size() > 0) {
        node["chained"] = chained;
    }
    return node;
}

std::string getExecGraphJson(int32_t appId, int32_t msgId)
{
    auto status = Planner::get().getBatchResults(appId);
    std::map<int32_t, std::shared_ptr<Message>> results;
    if (status) {
        for (auto& m : status->messageResults) {
            results[m.id] = std::make_shared<Message>(m);
        }
    }
    Json root = Json::object();
    root["root"] = execGraphNode(results, msgId);
    return root.dump();
}

// ------------------------- endpoint -----------------------------------------

PlannerEndpoint::PlannerEndpoint(int portIn)
  : port(portIn)
{}

PlannerEndpoint::~PlannerEndpoint()
{
    stop();
}

void PlannerEndpoint::start()
{
    if (running.exchange(true)) {
        return;
    }
    listener.listen(port + getPortOffset());
    acceptThread = std::thread([this] { acceptLoop(); });
    FAM_INFO("planner HTTP endpoint on %d", port + getPortOffset());
}

void PlannerEndpoint::stop()
{
    if (!running.exchange(false)) {
        return;
    }
    listener.close();
    if (acceptThread.joinable()) {
        acceptThread.join();
    }
    std::vector<std::thread> toJoin;
    {
        std::lock_guard<std::mutex> lock(threadsMx);
        toJoin.swap(connThreads);
    }
    for (auto& t : toJoin) {
        if (t.joinable()) {
            t.join();
        }
    }
}

void PlannerEndpoint::acceptLoop()
{
    while (running.load()) {
        auto conn = listener.accept();
        if (!conn.has_value()) {
            break;
        }
        std::lock_guard<std::mutex> lock(threadsMx);
        if (!running.load()) {
            break;
        }
        connThreads.emplace_back([this, c = std::move(*conn)]() mutable {
            connectionLoop(std::move(c));
        });
    }
}

void PlannerEndpoint::connectionLoop(TcpConnection conn)
{
    // One request per connection (Connection: close)
    std::string buf;
    char tmp[4096];
    size_t contentLen = 0;
    size_t headerEnd = std::string::npos;
    try {
        while (true) {
            ssize_t n = ::recv(conn.rawFd(), tmp, sizeof(tmp), 0);
            if (n <= 0) {
                return;
            }
            buf.append(tmp, (size_t)n);
            if (headerEnd == std::string::npos) {
                headerEnd = buf.find("\r\n\r\n");
                if (headerEnd != std::string::npos) {
                    // Parse Content-Length
                    std::string headers = buf.substr(0, headerEnd);
                    for (auto& c : headers) {
                        c = (char)tolower(c);
                    }
                    auto pos = headers.find("content-length:");
                    if (pos != std::string::npos) {
                        contentLen =
                          (size_t)atoll(headers.c_str() + pos + 15);
                    }
                }
            }
            if (headerEnd != std::string::npos &&
                buf.size() >= headerEnd + 4 + contentLen) {
                break;
            }
        }
        std::string body = buf.substr(headerEnd + 4, contentLen);
        auto [status, respBody] = handle(body);
        std::string statusLine =
          status == 200 ? "HTTP/1.1 200 OK" : "HTTP/1.1 400 Bad Request";
        std::string resp = statusLine +
                           "\r\nContent-Type: application/json"
                           "\r\nContent-Length: " +
                           std::to_string(respBody.size()) +
                           "\r\nConnection: close\r\n\r\n" + respBody;
        conn.sendAll(resp.data(), resp.size());
    } catch (const std::exception& e) {
        FAM_ERROR("http connection error: %s", e.what());
    }
}

std::pair<int, std::string> PlannerEndpoint::handle(
  const std::string& jsonBody)
{
    auto& planner = Planner::get();
    try {
        Json req = Json::parse(jsonBody);
        auto type = (HttpMessageType)req.getInt("http_type");
        std::string payload = req.getString("payload");

        switch (type) {
            case HttpMessageType::RESET: {
                planner.flushSchedulingState();
                return { 200, "Planner fully reset!" };
            }
            case HttpMessageType::FLUSH_AVAILABLE_HOSTS: {
                planner.reset();
                return { 200, "Flushed available hosts!" };
            }
            case HttpMessageType::FLUSH_EXECUTORS: {
                planner.flushExecutors();
                return { 200, "Flushed executors!" };
            }
            case HttpMessageType::FLUSH_SCHEDULING_STATE: {
                planner.flushSchedulingState();
                return { 200, "Flushed scheduling state!" };
            }
            case HttpMessageType::GET_AVAILABLE_HOSTS: {
                Json out = Json::object();
                Json hosts = Json::array();
                for (const auto& h : planner.getAvailableHosts()) {
                    Json hj = Json::object();
                    hj["ip"] = Json(h.ip);
                    hj["slots"] = Json((int64_t)h.slots);
                    hj["usedSlots"] = Json((int64_t)h.usedSlots);
                    hosts.push(hj);
                }
                out["hosts"] = hosts;
                return { 200, out.dump() };
            }
            case HttpMessageType::GET_CONFIG: {
                auto conf = planner.getConfig();
                Json out = Json::object();
                out["ip"] = Json(conf.ip);
                out["hostTimeout"] = Json((int64_t)conf.hostTimeout);
                out["numThreadsHttpServer"] =
                  Json((int64_t)conf.numThreadsHttpServer);
                return { 200, out.dump() };
            }
            case HttpMessageType::GET_EXEC_GRAPH: {
                Message msg = messageFromJson(Json::parse(payload));
                return { 200, getExecGraphJson(msg.appId, msg.id) };
            }
            case HttpMessageType::GET_IN_FLIGHT_APPS: {
                auto resp = planner.getInFlightApps();
                Json out = Json::object();
                Json apps = Json::array();
                for (const auto& a : resp.apps) {
                    Json aj = Json::object();
                    aj["appId"] = Json((int64_t)a.appId);
                    aj["subType"] = Json((int64_t)a.subType);
                    aj["size"] = Json((int64_t)a.size);
                    Json ips = Json::array();
                    for (const auto& ip : a.hostIps) {
                        ips.push(Json(ip));
                    }
                    aj["hostIps"] = ips;
                    apps.push(aj);
                }
                out["apps"] = apps;
                out["numMigrations"] = Json((int64_t)resp.numMigrations);
                Json evicted = Json::array();
                for (const auto& ip : resp.nextEvictedVmIps) {
                    evicted.push(Json(ip));
                }
                out["nextEvictedVmIps"] = evicted;
                Json frozen = Json::array();
                for (const auto& a : resp.frozenApps) {
                    Json aj = Json::object();
                    aj["appId"] = Json((int64_t)a.appId);
                    aj["size"] = Json((int64_t)a.size);
                    frozen.push(aj);
                }
                out["frozenApps"] = frozen;
                return { 200, out.dump() };
            }
            case HttpMessageType::EXECUTE_BATCH: {
                auto ber = std::make_shared<BatchExecuteRequest>(
                  berFromJson(Json::parse(payload)));
                if (ber->appId == 0) {
                    updateBatchExecAppId(*ber, generateGidInt32());
                }
                for (auto& m : ber->messages) {
                    if (m.id == 0) {
                        m.id = generateGidInt32();
                    }
                    m.user = ber->user;
                    m.function = ber->function;
                    m.appId = ber->appId;
                }
                if (!isBatchExecRequestValid(*ber)) {
                    return { 400, "Bad BatchExecRequest" };
                }
                auto decision = planner.callBatch(ber);
                if (decision->appId == NOT_ENOUGH_SLOTS) {
                    return { 400, "No available hosts" };
                }
                return { 200, berToJson(*ber).dump() };
            }
            case HttpMessageType::EXECUTE_BATCH_STATUS: {
                BatchExecuteRequest query =
                  berFromJson(Json::parse(payload));
                auto status = planner.getBatchResults(query.appId);
                if (!status) {
                    return { 400, "App not registered in results" };
                }
                Json out = berStatusToJson(*status);
                return { 200, out.dump() };
            }
            case HttpMessageType::PRELOAD_SCHEDULING_DECISION: {
                // Payload is a BER whose messages carry the target host in
                // executedHost (reference handler convention)
                BatchExecuteRequest ber =
                  berFromJson(Json::parse(payload));
                auto decision = std::make_shared<SchedulingDecision>(
                  ber.appId, ber.groupId);
                for (const auto& m : ber.messages) {
                    decision->addMessage(
                      m.executedHost, m.id, m.appIdx, m.groupIdx);
                }
                planner.preloadSchedulingDecision(ber.appId, decision);
                return { 200, "Decision pre-loaded to planner" };
            }
            case HttpMessageType::SET_POLICY: {
                planner.setPolicy(payload);
                return { 200, "Policy set correctly" };
            }
            case HttpMessageType::GET_RUNTIME_METRICS: {
                Json out = Json::object();
                out["appResults"] =
                  Json((int64_t)planner.debugAppResultsCount());
                out["doneApps"] = Json((int64_t)planner.debugDoneAppsCount());
                out["inFlightApps"] =
                  Json((int64_t)planner.debugInFlightCount());
                auto& broker = getPointToPointBroker();
                out["ptpMappings"] =
                  Json((int64_t)broker.debugMappingsCount());
                out["ptpChannels"] =
                  Json((int64_t)broker.debugChannelsCount());
                out["ptpSendSeqs"] =
                  Json((int64_t)broker.debugSendSeqsCount());
                out["decisionCache"] =
                  Json((int64_t)DecisionCache::get().size());
                return { 200, out.dump() };
            }
            case HttpMessageType::GET_POLICY: {
                return { 200, planner.getPolicy() };
            }
            case HttpMessageType::SET_NEXT_EVICTED_VM: {
                std::set<std::string> ips;
                if (!payload.empty() && payload[0] == '[') {
                    Json arr = Json::parse(payload);
                    for (const auto& v : arr.items()) {
                        ips.insert(v.asString());
                    }
                } else {
                    ips.insert(payload);
                }
                planner.setNextEvictedVm(ips);
                return { 200, "Next evicted VM set" };
            }
            default:
                return { 400, "Unrecognised http type" };
        }
    } catch (const std::exception& e) {
        return { 400, std::string("Error: ") + e.what() };
    }
}

} // namespace faabricamd
