// Runtime bindings: planner / scheduler / executor / state / snapshot / ptp.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "faabricamd/endpoint.h"
#include "faabricamd/executor.h"
#include "faabricamd/flat.h"
#include "faabricamd/hipipc.h"
#include "faabricamd/planner.h"
#include "faabricamd/ptp.h"
#include "faabricamd/runner.h"
#include "faabricamd/scheduler.h"
#include "faabricamd/snapshot.h"
#include "faabricamd/state.h"
#include "faabricamd/util.h"
#include "faabricamd/utilextras.h"
#include "faabricamd/dirty.h"
#include <hip/hip_runtime.h>

namespace py = pybind11;
using namespace faabricamd;

namespace faabricamd {
void registerBenchFunctions();       // bench_funcs.cpp
void registerMpiExampleFunctions();  // mpi_examples.cpp
}

void initRuntimeBindings(py::module_& m)
{
    // ---------------- config ----------------
    m.def("set_endpoint_host", [](const std::string& host) {
        getSystemConfig().endpointHost = host;
    });
    m.def("get_endpoint_host",
          [] { return getSystemConfig().endpointHost; });
    m.def("set_planner_host", [](const std::string& host) {
        getSystemConfig().plannerHost = host;
        resetPlannerClient();
    });
    m.def("set_batch_scheduler_mode", [](const std::string& mode) {
        resetBatchScheduler(mode);
    });
    // Pure policy evaluation for unit tests: hosts = [(ip, slots, used)],
    // in-flight = {appId: ([msgIds], [hosts])}; returns the decision
    m.def("test_make_scheduling_decision",
          [](const std::string& mode,
             const std::vector<std::tuple<std::string, int, int>>& hosts,
             int nMessages,
             int32_t appId,
             bool migration,
             const std::vector<std::pair<int32_t, std::vector<std::string>>>&
               inFlight) {
              std::shared_ptr<BatchScheduler> sched;
              if (mode == "bin-pack") {
                  sched = std::make_shared<BinPackScheduler>();
              } else if (mode == "compact") {
                  sched = std::make_shared<CompactScheduler>();
              } else if (mode == "spot") {
                  sched = std::make_shared<SpotScheduler>();
              } else {
                  throw FaabricException("bad mode");
              }
              HostMap hostMap;
              for (const auto& [ip, slots, used] : hosts) {
                  hostMap[ip] =
                    std::make_shared<HostState>(ip, slots, used);
              }
              InFlightReqs reqs;
              for (const auto& [ifAppId, ifHosts] : inFlight) {
                  auto ber = std::make_shared<BatchExecuteRequest>();
                  ber->appId = ifAppId;
                  ber->user = "t";
                  ber->function = "t";
                  auto dec =
                    std::make_shared<SchedulingDecision>(ifAppId, 1);
                  int idx = 0;
                  for (const auto& h : ifHosts) {
                      Message m;
                      m.id = ifAppId * 1000 + idx;
                      m.appId = ifAppId;
                      m.appIdx = idx;
                      m.groupIdx = idx;
                      ber->messages.push_back(m);
                      dec->addMessage(h, m);
                      idx++;
                  }
                  reqs[ifAppId] = { ber, dec };
              }
              BatchExecuteRequest req;
              req.appId = appId;
              req.user = "t";
              req.function = "t";
              req.type = migration ? BatchExecuteType::MIGRATION
                                   : BatchExecuteType::FUNCTIONS;
              for (int i = 0; i < nMessages; i++) {
                  Message m;
                  m.id = appId * 1000 + i;
                  m.appId = appId;
                  m.appIdx = i;
                  m.groupIdx = i;
                  req.messages.push_back(m);
              }
              return *sched->makeSchedulingDecision(hostMap, reqs, req);
          },
          py::arg("mode"),
          py::arg("hosts"),
          py::arg("n_messages"),
          py::arg("app_id") = 12345,
          py::arg("migration") = false,
          py::arg("in_flight") =
            std::vector<std::pair<int32_t, std::vector<std::string>>>{});
    m.def("decision_cache_size",
          [] { return DecisionCache::get().size(); });
    m.def("decision_cache_clear", [] { DecisionCache::get().clear(); });

    // ---------------- scheduling decision ----------------
    py::class_<SchedulingDecision>(m, "SchedulingDecision")
      .def(py::init<>())
      .def_readwrite("app_id", &SchedulingDecision::appId)
      .def_readwrite("group_id", &SchedulingDecision::groupId)
      .def_readwrite("n_functions", &SchedulingDecision::nFunctions)
      .def_readwrite("hosts", &SchedulingDecision::hosts)
      .def_readwrite("message_ids", &SchedulingDecision::messageIds)
      .def_readwrite("app_idxs", &SchedulingDecision::appIdxs)
      .def_readwrite("group_idxs", &SchedulingDecision::groupIdxs)
      .def_readwrite("mpi_ports", &SchedulingDecision::mpiPorts);

    m.def("NOT_ENOUGH_SLOTS", [] { return NOT_ENOUGH_SLOTS; });
    m.def("DO_NOT_MIGRATE", [] { return DO_NOT_MIGRATE; });
    m.def("MUST_FREEZE", [] { return MUST_FREEZE; });

    // ---------------- runtimes ----------------
    py::class_<PlannerRuntime>(m, "PlannerRuntime")
      .def(py::init<>())
      .def("start",
           &PlannerRuntime::start,
           py::arg("with_snapshot_server") = true,
           py::arg("with_state_server") = false,
           py::call_guard<py::gil_scoped_release>())
      .def("shutdown",
           &PlannerRuntime::shutdown,
           py::call_guard<py::gil_scoped_release>());

    py::class_<FaabricMain>(m, "FaabricMain")
      .def(py::init([]() {
          return new FaabricMain(getExecutorFactory());
      }))
      .def("start_background",
           &FaabricMain::startBackground,
           py::call_guard<py::gil_scoped_release>())
      .def("shutdown",
           &FaabricMain::shutdown,
           py::call_guard<py::gil_scoped_release>());

    // ---------------- function registry ----------------
    m.def("register_function",
          [](const std::string& user,
             const std::string& function,
             py::function fn) {
              FunctionRegistry::get().registerFunction(
                user, function, [fn](Message& msg) -> int32_t {
                    py::gil_scoped_acquire gil;
                    py::object result = fn(&msg);
                    if (result.is_none()) {
                        return 0;
                    }
                    return result.cast<int32_t>();
                });
          });
    m.def("clear_function_registry",
          [] { FunctionRegistry::get().clear(); });

    // Native no-op / sleep functions for benchmarking the scheduling path
    // without touching Python from executor threads
    m.def("register_native_noop",
          [](const std::string& user, const std::string& function) {
              FunctionRegistry::get().registerFunction(
                user, function, [](Message&) -> int32_t { return 0; });
          });
    m.def("register_native_echo",
          [](const std::string& user, const std::string& function) {
              FunctionRegistry::get().registerFunction(
                user, function, [](Message& msg) -> int32_t {
                    msg.outputData.assign(msg.inputData.begin(),
                                          msg.inputData.end());
                    return 0;
                });
          });
    m.def("register_native_sleep",
          [](const std::string& user,
             const std::string& function,
             int sleepMs) {
              FunctionRegistry::get().registerFunction(
                user, function, [sleepMs](Message&) -> int32_t {
                    std::this_thread::sleep_for(
                      std::chrono::milliseconds(sleepMs));
                    return 0;
                });
          });

    // ---------------- planner client ----------------
    m.def(
      "call_functions",
      [](const BatchExecuteRequest& reqIn) {
          auto req = std::make_shared<BatchExecuteRequest>(reqIn);
          py::gil_scoped_release release;
          return *getPlannerClient().callFunctions(req);
      });
    m.def(
      "get_message_result",
      [](int32_t appId, int32_t msgId, int timeoutMs) {
          py::gil_scoped_release release;
          return getPlannerClient().getMessageResult(appId, msgId, timeoutMs);
      });
    m.def("get_batch_results", [](int32_t appId) {
        py::gil_scoped_release release;
        return getPlannerClient().getBatchResults(appId);
    });
    m.def("get_available_hosts", [] {
        py::gil_scoped_release release;
        return getPlannerClient().getAvailableHosts();
    });
    m.def("get_num_migrations", [] {
        py::gil_scoped_release release;
        return getPlannerClient().getNumMigrations();
    });
    m.def("preload_scheduling_decision",
          [](int32_t appId, const SchedulingDecision& decision) {
              py::gil_scoped_release release;
              getPlannerClient().preloadSchedulingDecision(appId, decision);
          });
    m.def("planner_ping", [] {
        py::gil_scoped_release release;
        getPlannerClient().ping();
    });

    // Direct planner access (valid in the planner process only)
    m.def("planner_reset", [] { Planner::get().reset(); });
    m.def("planner_flush_scheduling_state",
          [] { Planner::get().flushSchedulingState(); });
    m.def("planner_set_next_evicted_vms",
          [](const std::vector<std::string>& ips) {
              Planner::get().setNextEvictedVm(
                { ips.begin(), ips.end() });
          });
    m.def("planner_set_policy",
          [](const std::string& p) { Planner::get().setPolicy(p); });
    m.def("planner_get_policy", [] { return Planner::get().getPolicy(); });
    m.def("planner_num_in_flight_apps", [] {
        return (int)Planner::get().getInFlightApps().apps.size();
    });

    // ---------------- scheduler ----------------
    m.def("set_this_host_resources", [](int slots, int usedSlots) {
        HostResources res;
        res.slots = slots;
        res.usedSlots = usedSlots;
        Scheduler::get().setThisHostResources(res);
    });
    m.def("get_this_host_slots", [] {
        return Scheduler::get().getThisHostResources().slots;
    });
    m.def("get_executor_count",
          [] { return (int)Scheduler::get().getExecutorCount(); });
    m.def("scheduler_reset", [] {
        py::gil_scoped_release release;
        Scheduler::get().reset();
    });
    m.def("get_recorded_messages",
          [] { return Scheduler::get().getRecordedMessages(); });
    m.def("set_test_mode", &setTestMode);

    // ---------------- point-to-point ----------------
    m.def("ptp_send",
          [](int32_t appId,
             int32_t groupId,
             int32_t sendIdx,
             int32_t recvIdx,
             const py::bytes& data,
             bool ordered) {
              std::string s = data;
              py::gil_scoped_release release;
              getPointToPointBroker().sendMessage(appId,
                                                  groupId,
                                                  sendIdx,
                                                  recvIdx,
                                                  (const uint8_t*)s.data(),
                                                  s.size(),
                                                  ordered);
          },
          py::arg("app_id"),
          py::arg("group_id"),
          py::arg("send_idx"),
          py::arg("recv_idx"),
          py::arg("data"),
          py::arg("ordered") = false);
    m.def("ptp_recv",
          [](int32_t groupId,
             int32_t sendIdx,
             int32_t recvIdx,
             bool ordered,
             int timeoutMs) {
              std::vector<uint8_t> out;
              {
                  py::gil_scoped_release release;
                  out = getPointToPointBroker().recvMessage(
                    groupId, sendIdx, recvIdx, ordered, timeoutMs);
              }
              return py::bytes((const char*)out.data(), out.size());
          },
          py::arg("group_id"),
          py::arg("send_idx"),
          py::arg("recv_idx"),
          py::arg("ordered") = false,
          py::arg("timeout_ms") = DEFAULT_QUEUE_TIMEOUT_MS);
    m.def("ptp_send_device",
          [](int32_t appId,
             int32_t groupId,
             int32_t sendIdx,
             int32_t recvIdx,
             uintptr_t devPtr,
             size_t size,
             bool ordered) {
              py::gil_scoped_release release;
              getPointToPointBroker().sendMessageDevice(
                appId, groupId, sendIdx, recvIdx, (const void*)devPtr,
                size, ordered);
          },
          py::arg("app_id"),
          py::arg("group_id"),
          py::arg("send_idx"),
          py::arg("recv_idx"),
          py::arg("dev_ptr"),
          py::arg("size"),
          py::arg("ordered") = false);
    m.def("ptp_recv_device",
          [](int32_t groupId,
             int32_t sendIdx,
             int32_t recvIdx,
             uintptr_t devPtr,
             size_t capacity,
             bool ordered,
             int timeoutMs) {
              py::gil_scoped_release release;
              return getPointToPointBroker().recvMessageDevice(
                groupId, sendIdx, recvIdx, (void*)devPtr, capacity,
                ordered, timeoutMs);
          },
          py::arg("group_id"),
          py::arg("send_idx"),
          py::arg("recv_idx"),
          py::arg("dev_ptr"),
          py::arg("capacity"),
          py::arg("ordered") = false,
          py::arg("timeout_ms") = DEFAULT_QUEUE_TIMEOUT_MS);
    m.def("ptp_group_barrier", [](int32_t groupId, int32_t groupIdx) {
        py::gil_scoped_release release;
        PointToPointGroup::getOrAwaitGroup(groupId)->barrier(groupIdx);
    });
    m.def("ptp_group_lock",
          [](int32_t groupId, int32_t groupIdx, bool recursive) {
              py::gil_scoped_release release;
              PointToPointGroup::getOrAwaitGroup(groupId)->lock(groupIdx,
                                                                recursive);
          },
          py::arg("group_id"),
          py::arg("group_idx"),
          py::arg("recursive") = false);
    m.def("ptp_group_unlock",
          [](int32_t groupId, int32_t groupIdx, bool recursive) {
              py::gil_scoped_release release;
              PointToPointGroup::getOrAwaitGroup(groupId)->unlock(groupIdx,
                                                                  recursive);
          },
          py::arg("group_id"),
          py::arg("group_idx"),
          py::arg("recursive") = false);
    m.def("ptp_group_notify", [](int32_t groupId, int32_t groupIdx) {
        py::gil_scoped_release release;
        PointToPointGroup::getOrAwaitGroup(groupId)->notify(groupIdx);
    });
    m.def("ptp_setup_local_mappings", [](const SchedulingDecision& d) {
        getPointToPointBroker().setUpLocalMappingsFromSchedulingDecision(d);
    });
    m.def("ptp_clear", [] { getPointToPointBroker().clear(); });
    // Inject a message with an explicit sequence number, as the network
    // would deliver it — lets tests exercise the out-of-order
    // resequencing buffer directly
    m.def("_test_ptp_deliver_seq",
          [](int32_t appId,
             int32_t groupId,
             int32_t sendIdx,
             int32_t recvIdx,
             const py::bytes& data,
             uint32_t seq) {
              PointToPointMessage msg;
              msg.appId = appId;
              msg.groupId = groupId;
              msg.sendIdx = sendIdx;
              msg.recvIdx = recvIdx;
              std::string s = data;
              msg.data.assign(s.begin(), s.end());
              py::gil_scoped_release release;
              getPointToPointBroker().deliverRemoteMessage(msg, seq);
          });
    // Standalone servers (multi-process PTP/IPC tests run these without
    // the rest of the worker fabric)
    py::class_<PointToPointServer>(m, "PointToPointServerHandle")
      .def(py::init<>())
      .def("start",
           [](PointToPointServer& s) {
               py::gil_scoped_release release;
               s.start();
           })
      .def("stop", [](PointToPointServer& s) {
          py::gil_scoped_release release;
          s.stop();
      });
    py::class_<StateServer>(m, "StateServerHandle")
      .def(py::init<>())
      .def("start",
           [](StateServer& s) {
               py::gil_scoped_release release;
               s.start();
           })
      .def("stop", [](StateServer& s) {
          py::gil_scoped_release release;
          s.stop();
      });
    py::class_<SnapshotServer>(m, "SnapshotServerHandle")
      .def(py::init<>())
      .def("start",
           [](SnapshotServer& s) {
               py::gil_scoped_release release;
               s.start();
           })
      .def("stop", [](SnapshotServer& s) {
          py::gil_scoped_release release;
          s.stop();
      });
    m.def("snapshot_push_device_from_ptr",
          [](const std::string& host,
             const std::string& key,
             uintptr_t devPtr,
             size_t size) {
              py::gil_scoped_release release;
              getSnapshotClient(host)->pushDeviceSnapshotFromDevice(
                key, (const void*)devPtr, size);
          },
          py::arg("host"),
          py::arg("key"),
          py::arg("dev_ptr"),
          py::arg("size"));
    m.def("ipc_shipped", [] {
        return py::make_tuple(IpcSender::get().shippedSegments(),
                              IpcSender::get().shippedBytes());
    });
    m.def("ipc_available",
          [](const std::string& host) {
              py::gil_scoped_release release;
              return IpcSender::get().available(host);
          },
          py::arg("host"));
    m.def("ptp_wait_for_mappings", [](int32_t groupId) {
        py::gil_scoped_release release;
        getPointToPointBroker().waitForMappingsOnThisHost(groupId);
    });

    // ---------------- state ----------------
    py::class_<StateKeyValue, std::shared_ptr<StateKeyValue>>(m,
                                                              "StateKeyValue")
      .def_property_readonly("user", &StateKeyValue::getUser)
      .def_property_readonly("key", &StateKeyValue::getKey)
      .def_property_readonly("size", &StateKeyValue::size)
      .def_property_readonly("is_master", &StateKeyValue::isMaster)
      .def_property_readonly("master_host", &StateKeyValue::getMasterHost)
      .def_property_readonly("on_device", &StateKeyValue::isOnDevice)
      .def_property_readonly("data_ptr",
                             [](StateKeyValue& kv) {
                                 return (uintptr_t)kv.getDataPtr();
                             })
      .def("get",
           [](StateKeyValue& kv) {
               std::vector<uint8_t> out;
               {
                   py::gil_scoped_release release;
                   out = kv.get();
               }
               return py::bytes((const char*)out.data(), out.size());
           })
      .def("set",
           [](StateKeyValue& kv, const py::bytes& data) {
               std::string s = data;
               py::gil_scoped_release release;
               kv.set((const uint8_t*)s.data(), s.size());
           })
      .def("get_chunk",
           [](StateKeyValue& kv, uint64_t offset, size_t len) {
               std::vector<uint8_t> out(len);
               {
                   py::gil_scoped_release release;
                   kv.getChunk(offset, out.data(), len);
               }
               return py::bytes((const char*)out.data(), out.size());
           })
      .def("set_chunk",
           [](StateKeyValue& kv, uint64_t offset, const py::bytes& data) {
               std::string s = data;
               py::gil_scoped_release release;
               kv.setChunk(offset, (const uint8_t*)s.data(), s.size());
           })
      .def("append",
           [](StateKeyValue& kv, const py::bytes& data) {
               std::string s = data;
               py::gil_scoped_release release;
               kv.append((const uint8_t*)s.data(), s.size());
           })
      .def("get_appended",
           [](StateKeyValue& kv, size_t n) {
               std::vector<std::vector<uint8_t>> vals;
               {
                   py::gil_scoped_release release;
                   vals = kv.getAppended(n);
               }
               py::list out;
               for (auto& v : vals) {
                   out.append(py::bytes((const char*)v.data(), v.size()));
               }
               return out;
           })
      .def("clear_appended", [](StateKeyValue& kv) {
          py::gil_scoped_release release;
          kv.clearAppended();
      })
      .def("pull",
           [](StateKeyValue& kv) {
               py::gil_scoped_release release;
               kv.pull();
           })
      .def("push_partial", [](StateKeyValue& kv) {
          py::gil_scoped_release release;
          kv.pushPartial();
      })
      .def("push_full", [](StateKeyValue& kv) {
          py::gil_scoped_release release;
          kv.pushFull();
      })
      .def("sync", [](StateKeyValue& kv) {
          py::gil_scoped_release release;
          kv.sync();
      });
    m.def("state_sync_all", [] {
        py::gil_scoped_release release;
        State::get().syncAll();
    });
    m.def("set_state_mode", [](const std::string& mode) {
        getSystemConfig().stateMode = mode;
    });
    m.def("state_acquire_lock",
          [](const std::string& user, const std::string& key,
             int expiryMs) {
              py::gil_scoped_release release;
              return State::get().acquireLock(user, key, expiryMs);
          },
          py::arg("user"),
          py::arg("key"),
          py::arg("expiry_ms") = 10000);
    m.def("state_release_lock",
          [](const std::string& user, const std::string& key,
             uint64_t token) {
              py::gil_scoped_release release;
              State::get().releaseLock(user, key, token);
          });

    m.def("state_get_kv_device",
          [](const std::string& user,
             const std::string& key,
             size_t size,
             int device) {
              return State::get().getKVDevice(user, key, size, device);
          },
          py::arg("user"),
          py::arg("key"),
          py::arg("size"),
          py::arg("device") = 0);
    m.def("state_get_kv",
          [](const std::string& user, const std::string& key, size_t size) {
              return State::get().getKV(user, key, size);
          });
    m.def("state_set_master_host",
          [](const std::string& user,
             const std::string& key,
             const std::string& host) {
              State::get().setMasterHost(user, key, host);
          });
    m.def("state_size",
          [](const std::string& user, const std::string& key) {
              py::gil_scoped_release release;
              return State::get().getStateSize(user, key);
          });
    m.def("state_clear_all", [] { State::get().forceClearAll(false); });
    m.def("state_kv_count", [] { return State::get().getKVCount(); });

    // ---------------- snapshots ----------------
    py::enum_<SnapshotDataType>(m, "SnapshotDataType")
      .value("Raw", SnapshotDataType::Raw)
      .value("Bool", SnapshotDataType::Bool)
      .value("Int", SnapshotDataType::Int)
      .value("Long", SnapshotDataType::Long)
      .value("Float", SnapshotDataType::Float)
      .value("Double", SnapshotDataType::Double);

    py::enum_<SnapshotMergeOperation>(m, "SnapshotMergeOperation")
      .value("Bytewise", SnapshotMergeOperation::Bytewise)
      .value("Sum", SnapshotMergeOperation::Sum)
      .value("Product", SnapshotMergeOperation::Product)
      .value("Subtract", SnapshotMergeOperation::Subtract)
      .value("Max", SnapshotMergeOperation::Max)
      .value("Min", SnapshotMergeOperation::Min)
      .value("XOR", SnapshotMergeOperation::XOR);

    py::class_<SnapshotDiff>(m, "SnapshotDiff")
      .def(py::init<>())
      .def_readwrite("data_type", &SnapshotDiff::dataType)
      .def_readwrite("operation", &SnapshotDiff::operation)
      .def_readwrite("offset", &SnapshotDiff::offset)
      .def_property(
        "data",
        [](const SnapshotDiff& d) {
            return py::bytes((const char*)d.getData(), d.size());
        },
        [](SnapshotDiff& d, const py::bytes& b) {
            std::string s = b;
            d.dataCopy.assign(s.begin(), s.end());
        });

    py::class_<SnapshotData, std::shared_ptr<SnapshotData>>(m,
                                                            "SnapshotData")
      .def(py::init([](const py::bytes& data, size_t maxSize) {
               std::string s = data;
               std::vector<uint8_t> v(s.begin(), s.end());
               if (maxSize == 0) {
                   return std::make_shared<SnapshotData>(v);
               }
               return std::make_shared<SnapshotData>(v, maxSize);
           }),
           py::arg("data"),
           py::arg("max_size") = 0)
      .def(py::init([](size_t size) {
          return std::make_shared<SnapshotData>(size);
      }))
      .def_property_readonly("size", &SnapshotData::getSize)
      .def_property_readonly("max_size", &SnapshotData::getMaxSize)
      .def("get_data",
           [](SnapshotData& s) {
               auto v = s.getDataCopy();
               return py::bytes((const char*)v.data(), v.size());
           })
      .def("copy_in_data",
           [](SnapshotData& s, const py::bytes& data, uint32_t offset) {
               std::string str = data;
               s.copyInData((const uint8_t*)str.data(), str.size(), offset);
           },
           py::arg("data"),
           py::arg("offset") = 0)
      .def("add_merge_region",
           &SnapshotData::addMergeRegion,
           py::arg("offset"),
           py::arg("length"),
           py::arg("data_type"),
           py::arg("operation"))
      .def("fill_gaps_with_bytewise_regions",
           &SnapshotData::fillGapsWithBytewiseRegions)
      .def("clear_merge_regions", &SnapshotData::clearMergeRegions)
      .def("diff_with_memory",
           [](SnapshotData& s, const py::bytes& updated) {
               std::string str = updated;
               return s.diffWithMemory((const uint8_t*)str.data(),
                                       str.size());
           })
      .def("diff_with_dirty_regions",
           [](SnapshotData& s,
              const py::bytes& updated,
              const std::vector<char>& dirtyPages) {
               std::string str = updated;
               return s.diffWithDirtyRegions(
                 (const uint8_t*)str.data(), str.size(), dirtyPages);
           })
      .def("apply_diffs", &SnapshotData::applyDiffs)
      .def("queue_diffs", &SnapshotData::queueDiffs)
      .def("write_queued_diffs", &SnapshotData::writeQueuedDiffs);

    m.def("snapshot_register",
          [](const std::string& key, std::shared_ptr<SnapshotData> snap) {
              SnapshotRegistry::get().registerSnapshot(key, snap);
          });
    m.def("snapshot_get", [](const std::string& key) {
        return SnapshotRegistry::get().getSnapshot(key);
    });
    m.def("snapshot_exists", [](const std::string& key) {
        return SnapshotRegistry::get().snapshotExists(key);
    });
    m.def("snapshot_delete", [](const std::string& key) {
        SnapshotRegistry::get().deleteSnapshot(key);
    });
    m.def("snapshot_count",
          [] { return SnapshotRegistry::get().getSnapshotCount(); });
    m.def("snapshot_clear", [] { SnapshotRegistry::get().clear(); });

    // Mock recordings
    m.def("get_batch_requests_sent_mock", [] {
        return getBatchRequestsSentMock();
    });
    m.def("clear_mocked_function_calls", &clearMockedFunctionCalls);

    // ---------------- util extras ------------------------------------------
    m.def("delta_default_config", [] { return DeltaConfig{}.str(); });
    // FlatBuffers snapshot wire format (tests pin the encoding)
    m.def("flat_encode_push",
          [](const std::string& key, uint64_t maxSize,
             const py::bytes& contents,
             const std::vector<std::tuple<int32_t, uint64_t, int32_t,
                                          int32_t>>& regions) {
              FlatSnapshotPush req;
              req.key = key;
              req.maxSize = maxSize;
              std::string c = contents;
              req.contents.assign(c.begin(), c.end());
              for (auto& [off, len, dt, op] : regions) {
                  FlatMergeRegion m;
                  m.offset = off;
                  m.length = len;
                  m.dataType = dt;
                  m.mergeOp = op;
                  req.mergeRegions.push_back(m);
              }
              std::string out = req.encode();
              return py::bytes(out);
          });
    m.def("flat_decode_push", [](const py::bytes& buf) {
        std::string s = buf;
        auto req = FlatSnapshotPush::decode(s);
        py::list regions;
        for (auto& m : req.mergeRegions) {
            regions.append(py::make_tuple(m.offset, m.length, m.dataType,
                                          m.mergeOp));
        }
        return py::make_tuple(
          req.key, req.maxSize,
          py::bytes((const char*)req.contents.data(),
                    req.contents.size()),
          regions);
    });
    m.def("flat_encode_thread_result",
          [](int32_t appId, int32_t messageId, int32_t returnValue,
             const std::string& key,
             const std::vector<std::tuple<int32_t, int32_t, int32_t,
                                          py::bytes>>& diffs) {
              FlatThreadResult req;
              req.appId = appId;
              req.messageId = messageId;
              req.returnValue = returnValue;
              req.key = key;
              for (auto& [off, dt, op, data] : diffs) {
                  FlatSnapshotDiff d;
                  d.offset = off;
                  d.dataType = dt;
                  d.mergeOp = op;
                  std::string s = data;
                  d.data.assign(s.begin(), s.end());
                  req.diffs.push_back(std::move(d));
              }
              return py::bytes(req.encode());
          });
    m.def("flat_decode_thread_result", [](const py::bytes& buf) {
        std::string s = buf;
        auto req = FlatThreadResult::decode(s);
        py::list diffs;
        for (auto& d : req.diffs) {
            diffs.append(py::make_tuple(
              d.offset, d.dataType, d.mergeOp,
              py::bytes((const char*)d.data.data(), d.data.size())));
        }
        return py::make_tuple(req.appId, req.messageId, req.returnValue,
                              req.key, diffs, req.executedHost);
    });
    m.def("delta_encode",
          [](const py::bytes& oldData,
             const py::bytes& newData,
             const std::string& config) {
              std::string o = oldData;
              std::string n = newData;
              DeltaConfig conf = config.empty() ? DeltaConfig{}
                                                : DeltaConfig::parse(config);
              auto out = deltaEncode({ o.begin(), o.end() },
                                     { n.begin(), n.end() },
                                     conf);
              return py::bytes((const char*)out.data(), out.size());
          },
          py::arg("old_data"),
          py::arg("new_data"),
          py::arg("config") = "");
    m.def("delta_apply",
          [](const py::bytes& oldData, const py::bytes& delta) {
              std::string o = oldData;
              std::string d = delta;
              auto out = deltaApply({ o.begin(), o.end() },
                                    { d.begin(), d.end() });
              return py::bytes((const char*)out.data(), out.size());
          });
    m.def("prof_summary", &profSummary);
    m.def("prof_clear", &profClear);
    m.def("set_up_crash_handler", &setUpCrashHandler);
    m.def("pin_thread_to_free_cpu", &pinThreadToFreeCpu);

    // Self-contained native check of the segfault dirty tracker (mprotect
    // + SIGSEGV interplay is best kept out of Python)
    m.def("_selftest_segfault_tracker", [] {
        py::gil_scoped_release release;
        SegfaultDirtyTracker tracker;
        PageAlignedBuffer buf;
        buf.resize(16 * 4096);
        tracker.startTracking(buf.data(), buf.size());
        buf.data()[3 * 4096 + 5] = 42;
        buf.data()[9 * 4096] = 7;
        tracker.stopTracking(buf.data(), buf.size());
        auto dirty = tracker.getDirtyPages(buf.data(), buf.size());
        int nDirty = 0;
        for (char c : dirty) {
            nDirty += c != 0;
        }
        return nDirty == 2 && dirty[3] == 1 && dirty[9] == 1;
    });

    // Same check for the userfaultfd write-protect tracker; returns None
    // (skip) when the kernel lacks uffd-wp
    m.def("_selftest_uffd_tracker", []() -> py::object {
        if (!UffdDirtyTracker::isAvailable()) {
            return py::none();
        }
        bool ok;
        {
            py::gil_scoped_release release;
            UffdDirtyTracker tracker;
            PageAlignedBuffer buf;
            buf.resize(16 * 4096);
            tracker.startTracking(buf.data(), buf.size());
            buf.data()[2 * 4096 + 11] = 42;
            buf.data()[14 * 4096] = 7;
            // Writes block until the poller clears WP, so they are
            // ordered before stopTracking on this thread
            tracker.stopTracking(buf.data(), buf.size());
            auto dirty = tracker.getDirtyPages(buf.data(), buf.size());
            int nDirty = 0;
            for (char d : dirty) {
                nDirty += d != 0;
            }
            ok = nDirty == 2 && dirty[2] == 1 && dirty[14] == 1;
        }
        return py::bool_(ok);
    });

    // Same check for the soft-dirty PTE tracker; returns None (skip)
    // when the kernel lacks CONFIG_MEM_SOFT_DIRTY
    m.def("_selftest_softpte_tracker", []() -> py::object {
        if (!SoftPTEDirtyTracker::isAvailable()) {
            return py::none();
        }
        bool ok;
        {
            py::gil_scoped_release release;
            SoftPTEDirtyTracker tracker;
            PageAlignedBuffer buf;
            buf.resize(16 * 4096);
            // Fault pages in BEFORE the reset so the dirty set below is
            // from writes, not first-touch population
            std::memset(buf.data(), 1, buf.size());
            tracker.startTracking(buf.data(), buf.size());
            buf.data()[5 * 4096 + 3] = 42;
            buf.data()[12 * 4096] = 7;
            tracker.stopTracking(buf.data(), buf.size());
            auto dirty = tracker.getDirtyPages(buf.data(), buf.size());
            // Soft-dirty may over-report (kernel can flag extra pages,
            // e.g. on THP boundaries) but must include the two writes
            ok = dirty.size() == 16 && dirty[5] == 1 && dirty[12] == 1;
        }
        return py::bool_(ok);
    });

    // Native benchmark payloads (cpp/src/bench_funcs.cpp)
    // Runtime map sizes for leak hunting / observability
    m.def("_debug_runtime_sizes", [] {
        py::dict d;
        d["planner_app_results"] = Planner::get().debugAppResultsCount();
        d["planner_done_apps"] = Planner::get().debugDoneAppsCount();
        d["planner_in_flight"] = Planner::get().debugInFlightCount();
        d["broker_mappings"] = getPointToPointBroker().debugMappingsCount();
        d["broker_channels"] = getPointToPointBroker().debugChannelsCount();
        d["broker_send_seqs"] = getPointToPointBroker().debugSendSeqsCount();
        d["decision_cache"] = DecisionCache::get().size();
        return d;
    });
    m.def("wait_batch_done",
          [](int32_t appId, int timeoutMs) {
              py::gil_scoped_release release;
              return getPlannerClient().waitBatchDone(appId, timeoutMs);
          },
          py::arg("app_id"),
          py::arg("timeout_ms") = 30000);
    m.def("set_bound_timeout",
          [](int ms) { getSystemConfig().boundTimeout = ms; });
    m.def("reap_stale_executors", [] {
        py::gil_scoped_release release;
        return Scheduler::get().reapStaleExecutors();
    });
    m.def("register_bench_functions", [] { registerBenchFunctions(); });
    m.def("register_mpi_example_functions",
          [] { registerMpiExampleFunctions(); });

    // ---------------- THREADS fork-join (from inside a running task) -------
    m.def("execute_threads",
          [](const std::string& user,
             const std::string& function,
             int nThreads,
             const std::vector<std::tuple<uint32_t, size_t, int, int>>&
               regions,
             const py::bytes& inputData,
             bool elastic) {
              Executor* exec = ExecutorContext::get().getExecutor();
              std::string input = inputData;
              std::vector<std::pair<int32_t, int32_t>> results;
              {
                  py::gil_scoped_release release;
                  auto req = std::make_shared<BatchExecuteRequest>(
                    batchExecFactory(user, function, nThreads));
                  req->elasticScaleHint = elastic;
                  for (auto& m2 : req->messages) {
                      m2.inputData.assign(input.begin(), input.end());
                  }
                  std::vector<SnapshotMergeRegion> mr;
                  for (const auto& [off, len, dt, op] : regions) {
                      mr.emplace_back(off,
                                      len,
                                      (SnapshotDataType)dt,
                                      (SnapshotMergeOperation)op);
                  }
                  results = exec->executeThreads(req, mr);
              }
              py::list out;
              for (auto& [msgId, rv] : results) {
                  out.append(py::make_tuple(msgId, rv));
              }
              return out;
          },
          py::arg("user"),
          py::arg("function"),
          py::arg("n_threads"),
          py::arg("merge_regions") =
            std::vector<std::tuple<uint32_t, size_t, int, int>>{},
          py::arg("input_data") = py::bytes(""),
          py::arg("elastic") = false);

    // ---------------- chaining + exec graph --------------------------------
    m.def("chain_function",
          [](const std::string& user,
             const std::string& function,
             const py::bytes& input) {
              std::string s = input;
              std::vector<uint8_t> in(s.begin(), s.end());
              py::gil_scoped_release release;
              return chainFunction(user, function, in);
          },
          py::arg("user"),
          py::arg("function"),
          py::arg("input") = py::bytes(""));
    m.def("await_chained_call", [](int32_t msgId, int timeoutMs) {
        py::gil_scoped_release release;
        return awaitChainedCall(msgId, timeoutMs);
    },
          py::arg("msg_id"),
          py::arg("timeout_ms") = 60000);
    m.def("get_exec_graph_json", [](int32_t appId, int32_t msgId) {
        return getExecGraphJson(appId, msgId);
    });
    m.def("migration_point",
          [](const py::bytes& reentryInput) {
              std::string s = reentryInput;
              std::vector<uint8_t> in(s.begin(), s.end());
              py::gil_scoped_release release;
              return migrationPoint(in);
          },
          py::arg("reentry_input") = py::bytes(""));
    m.def("MIGRATED_FUNCTION_RETURN_VALUE",
          [] { return MIGRATED_FUNCTION_RETURN_VALUE; });
    m.def("FROZEN_FUNCTION_RETURN_VALUE",
          [] { return FROZEN_FUNCTION_RETURN_VALUE; });

    // ---------------- HTTP ops endpoint ------------------------------------
    py::class_<PlannerEndpoint>(m, "PlannerEndpoint")
      .def(py::init<int>(), py::arg("port") = PLANNER_HTTP_PORT)
      .def("start",
           &PlannerEndpoint::start,
           py::call_guard<py::gil_scoped_release>())
      .def("stop",
           &PlannerEndpoint::stop,
           py::call_guard<py::gil_scoped_release>())
      .def("handle", [](PlannerEndpoint& ep, const std::string& body) {
          std::pair<int, std::string> out;
          {
              py::gil_scoped_release release;
              out = ep.handle(body);
          }
          return py::make_tuple(out.first, out.second);
      });

    // Executor memory access for thread bodies / parents
    m.def("executor_set_memory_size", [](size_t n) {
        ExecutorContext::get().getExecutor()->setMemorySize(n);
    });
    m.def("executor_memory_size", [] {
        return ExecutorContext::get().getExecutor()->getMemoryView().second;
    });
    m.def("executor_write_memory", [](size_t offset, const py::bytes& data) {
        std::string s = data;
        auto [base, size] = ExecutorContext::get().getExecutor()
                              ->getMemoryView();
        if (offset + s.size() > size) {
            throw FaabricException("executor memory write out of bounds");
        }
        std::memcpy(base + offset, s.data(), s.size());
    });
    m.def("executor_read_memory", [](size_t offset, size_t n) {
        auto [base, size] = ExecutorContext::get().getExecutor()
                              ->getMemoryView();
        if (offset + n > size) {
            throw FaabricException("executor memory read out of bounds");
        }
        return py::bytes((const char*)base + offset, n);
    });

    // HBM executor arena (device THREADS flow)
    m.def("executor_set_device_memory_size", [](size_t n) {
        py::gil_scoped_release release;
        ExecutorContext::get().getExecutor()->setDeviceMemorySize(n);
    });
    m.def("executor_device_ptr", [] {
        return (uintptr_t)ExecutorContext::get()
          .getExecutor()
          ->getDeviceMemoryView()
          .first;
    });
    m.def("executor_device_write_memory",
          [](size_t offset, const py::bytes& data) {
              std::string s = data;
              auto [base, size] = ExecutorContext::get()
                                    .getExecutor()
                                    ->getDeviceMemoryView();
              if (base == nullptr || offset + s.size() > size) {
                  throw FaabricException("device arena write out of bounds");
              }
              py::gil_scoped_release release;
              if (hipMemcpy(base + offset, s.data(), s.size(),
                            hipMemcpyHostToDevice) != hipSuccess) {
                  throw FaabricException("device arena write failed");
              }
          });
    m.def("executor_device_read_memory", [](size_t offset, size_t n) {
        auto [base, size] =
          ExecutorContext::get().getExecutor()->getDeviceMemoryView();
        if (base == nullptr || offset + n > size) {
            throw FaabricException("device arena read out of bounds");
        }
        std::vector<uint8_t> out(n);
        {
            py::gil_scoped_release release;
            if (hipMemcpy(out.data(), base + offset, n,
                          hipMemcpyDeviceToHost) != hipSuccess) {
                throw FaabricException("device arena read failed");
            }
        }
        return py::bytes((const char*)out.data(), out.size());
    });
}
