// pybind11 bindings for the faabric-mi355x core. The Python layer is a thin
// driver (benchmarks, tests, deployment glue) over the native C++ runtime —
// the reference has no Python runtime at all, and neither
// does the compute path here.
#include <pybind11/functional.h>
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "faabricamd/executor.h"
#include "faabricamd/messages.h"
#include "faabricamd/queue.h"
#include "faabricamd/transport.h"
#include "faabricamd/util.h"

namespace py = pybind11;
using namespace faabricamd;

void initRuntimeBindings(py::module_& m); // bindings_runtime.cpp
void initOpsBindings(py::module_& m);     // bindings_ops.cpp
void initMpiBindings(py::module_& m);     // bindings_mpi.cpp

namespace {

py::bytes toPyBytes(const std::string& s)
{
    return py::bytes(s);
}

std::vector<uint8_t> fromPyBytes(const py::bytes& b)
{
    std::string s = b;
    return { s.begin(), s.end() };
}

} // namespace

PYBIND11_MODULE(_core, m)
{
    m.doc() = "faabric-mi355x native core (C++20 + HIP gfx950 + RCCL)";

    // ---------------- util ----------------
    m.def("generate_gid", &generateGidInt32);
    m.def("get_primary_ip",
          [] { return getPrimaryIPForThisHost(); });
    m.def("set_log_level", [](const std::string& lvl) {
        if (lvl == "trace") setLogLevel(LogLevel::trace);
        else if (lvl == "debug") setLogLevel(LogLevel::debug);
        else if (lvl == "info") setLogLevel(LogLevel::info);
        else if (lvl == "warn") setLogLevel(LogLevel::warn);
        else if (lvl == "error") setLogLevel(LogLevel::error);
        else setLogLevel(LogLevel::off);
    });
    m.def("set_mock_mode", &setMockMode);
    m.def("is_mock_mode", &isMockMode);
    m.def("set_port_offset", &setPortOffset);
    m.def("get_port_offset", &getPortOffset);
    m.def("get_main_thread_snapshot_key", &getMainThreadSnapshotKey);
    m.def("get_usable_cores", &getUsableCores);

    // ---------------- messages ----------------
    py::enum_<MessageType>(m, "MessageType")
      .value("CALL", MessageType::CALL)
      .value("KILL", MessageType::KILL)
      .value("EMPTY", MessageType::EMPTY)
      .value("FLUSH", MessageType::FLUSH);

    py::enum_<BatchExecuteType>(m, "BatchExecuteType")
      .value("FUNCTIONS", BatchExecuteType::FUNCTIONS)
      .value("THREADS", BatchExecuteType::THREADS)
      .value("PROCESSES", BatchExecuteType::PROCESSES)
      .value("MIGRATION", BatchExecuteType::MIGRATION);

    py::class_<Message>(m, "Message")
      .def(py::init<>())
      .def_readwrite("id", &Message::id)
      .def_readwrite("app_id", &Message::appId)
      .def_readwrite("app_idx", &Message::appIdx)
      .def_readwrite("main_host", &Message::mainHost)
      .def_readwrite("type", &Message::type)
      .def_readwrite("user", &Message::user)
      .def_readwrite("function", &Message::function)
      .def_property(
        "input_data",
        [](const Message& msg) {
            return py::bytes((const char*)msg.inputData.data(),
                             msg.inputData.size());
        },
        [](Message& msg, const py::bytes& b) {
            msg.inputData = fromPyBytes(b);
        })
      .def_readwrite("output_data", &Message::outputData)
      .def_readwrite("return_value", &Message::returnValue)
      .def_readwrite("snapshot_key", &Message::snapshotKey)
      .def_readwrite("start_timestamp", &Message::startTimestamp)
      .def_readwrite("finish_timestamp", &Message::finishTimestamp)
      .def_readwrite("executed_host", &Message::executedHost)
      .def_readwrite("group_id", &Message::groupId)
      .def_readwrite("group_idx", &Message::groupIdx)
      .def_readwrite("group_size", &Message::groupSize)
      .def_readwrite("is_mpi", &Message::isMpi)
      .def_readwrite("mpi_world_id", &Message::mpiWorldId)
      .def_readwrite("mpi_rank", &Message::mpiRank)
      .def_readwrite("mpi_world_size", &Message::mpiWorldSize)
      .def_readwrite("record_exec_graph", &Message::recordExecGraph)
      .def_readwrite("chained_msg_ids", &Message::chainedMsgIds)
      .def_readwrite("int_exec_graph_details", &Message::intExecGraphDetails)
      .def_readwrite("exec_graph_details", &Message::execGraphDetails)
      .def("encode", [](const Message& msg) { return toPyBytes(msg.encode()); })
      .def_static("decode", [](const py::bytes& b) {
          return Message::decode(std::string(b));
      });

    py::class_<BatchExecuteRequest>(m, "BatchExecuteRequest")
      .def(py::init<>())
      .def_readwrite("app_id", &BatchExecuteRequest::appId)
      .def_readwrite("group_id", &BatchExecuteRequest::groupId)
      .def_readwrite("user", &BatchExecuteRequest::user)
      .def_readwrite("function", &BatchExecuteRequest::function)
      .def_readwrite("type", &BatchExecuteRequest::type)
      .def_readwrite("snapshot_key", &BatchExecuteRequest::snapshotKey)
      .def_readwrite("messages", &BatchExecuteRequest::messages)
      .def_readwrite("sub_type", &BatchExecuteRequest::subType)
      .def_readwrite("single_host", &BatchExecuteRequest::singleHost)
      .def_readwrite("single_host_hint", &BatchExecuteRequest::singleHostHint)
      .def_readwrite("elastic_scale_hint",
                     &BatchExecuteRequest::elasticScaleHint)
      .def("encode",
           [](const BatchExecuteRequest& b) { return toPyBytes(b.encode()); })
      .def_static("decode", [](const py::bytes& b) {
          return BatchExecuteRequest::decode(std::string(b));
      });

    py::class_<BatchExecuteRequestStatus>(m, "BatchExecuteRequestStatus")
      .def(py::init<>())
      .def_readwrite("app_id", &BatchExecuteRequestStatus::appId)
      .def_readwrite("finished", &BatchExecuteRequestStatus::finished)
      .def_readwrite("message_results",
                     &BatchExecuteRequestStatus::messageResults)
      .def_readwrite("expected_num_messages",
                     &BatchExecuteRequestStatus::expectedNumMessages);

    py::class_<Host>(m, "Host")
      .def(py::init<>())
      .def_readwrite("ip", &Host::ip)
      .def_readwrite("slots", &Host::slots)
      .def_readwrite("used_slots", &Host::usedSlots)
      .def_readwrite("register_ts_epoch_ms", &Host::registerTsEpochMs);

    py::class_<PointToPointMapping>(m, "PointToPointMapping")
      .def(py::init<>())
      .def_readwrite("host", &PointToPointMapping::host)
      .def_readwrite("message_id", &PointToPointMapping::messageId)
      .def_readwrite("app_idx", &PointToPointMapping::appIdx)
      .def_readwrite("group_idx", &PointToPointMapping::groupIdx)
      .def_readwrite("mpi_port", &PointToPointMapping::mpiPort);

    py::class_<PointToPointMappings>(m, "PointToPointMappings")
      .def(py::init<>())
      .def_readwrite("app_id", &PointToPointMappings::appId)
      .def_readwrite("group_id", &PointToPointMappings::groupId)
      .def_readwrite("mappings", &PointToPointMappings::mappings)
      .def("encode",
           [](const PointToPointMappings& p) { return toPyBytes(p.encode()); })
      .def_static("decode", [](const py::bytes& b) {
          return PointToPointMappings::decode(std::string(b));
      });

    m.def("message_factory", &messageFactory);
    m.def("batch_exec_factory", &batchExecFactory);
    m.def("is_batch_exec_request_valid", &isBatchExecRequestValid);

    initRuntimeBindings(m);
    initOpsBindings(m);
    initMpiBindings(m);

    // Python callables captured by the C++ function registry must be
    // released while the interpreter is still alive (the registry is a
    // C++ static, destroyed after Py_Finalize otherwise).
    m.add_object("_registry_cleanup", py::capsule([]() {
                     faabricamd::FunctionRegistry::get().clear();
                 }));
}
