"""Planner HTTP ops API + chaining + exec graph (reference coverage:
tests/test/planner/test_planner_endpoint.cpp, test_mpi_exec_graph.cpp)."""

import json
import urllib.request

import pytest

from faabric_amd import _core
from faabric_amd.runtime import LocalRuntime, wait_for_batch

SLOTS = 8
HTTP_PORT = 8080 + 600

# HttpMessage types (reference: planner.proto:34-66)
RESET = 1
GET_AVAILABLE_HOSTS = 5
GET_CONFIG = 6
GET_EXEC_GRAPH = 7
GET_IN_FLIGHT_APPS = 8
EXECUTE_BATCH = 10
EXECUTE_BATCH_STATUS = 11
SET_POLICY = 13
GET_POLICY = 14
GET_RUNTIME_METRICS = 16
SET_NEXT_EVICTED_VM = 15


def chain_parent(msg):
    msg_id = _core.chain_function("http", "child", b"from-parent")
    result = _core.await_chained_call(msg_id)
    if result.return_value != 0:
        return 1
    msg.output_data = "parent:" + result.output_data
    return 0


@pytest.fixture(scope="module")
def runtime():
    rt = LocalRuntime(slots=SLOTS, port_offset=600, planner_port_offset=600)
    rt.start_planner(with_snapshot_server=False)
    rt.start_worker()
    _core.register_native_echo("http", "echo")
    _core.register_native_echo("http", "child")
    _core.register_function("http", "parent", chain_parent)
    ep = _core.PlannerEndpoint(8080)
    ep.start()
    yield rt
    ep.stop()
    rt.stop()


def post(http_type, payload=""):
    body = json.dumps({"http_type": http_type, "payload": payload}).encode()
    req = urllib.request.Request(
        f"http://127.0.0.1:{HTTP_PORT}/", data=body, method="POST"
    )
    try:
        with urllib.request.urlopen(req, timeout=10) as resp:
            return resp.status, resp.read().decode()
    except urllib.error.HTTPError as e:
        return e.code, e.read().decode()


def test_get_available_hosts(runtime):
    status, body = post(GET_AVAILABLE_HOSTS)
    assert status == 200
    hosts = json.loads(body)["hosts"]
    assert len(hosts) == 1
    assert hosts[0]["slots"] == SLOTS


def test_get_config(runtime):
    status, body = post(GET_CONFIG)
    assert status == 200
    assert "hostTimeout" in json.loads(body)


def test_execute_batch_over_http(runtime):
    ber_json = {
        "user": "http",
        "function": "echo",
        "messages": [{"input_data": "aGVsbG8="}],  # b64 "hello"
    }
    status, body = post(EXECUTE_BATCH, json.dumps(ber_json))
    assert status == 200, body
    ber = json.loads(body)
    app_id = ber["appId"]
    assert app_id != 0

    # Poll status over HTTP
    import time

    deadline = time.monotonic() + 15
    while time.monotonic() < deadline:
        status, body = post(
            EXECUTE_BATCH_STATUS, json.dumps({"appId": app_id})
        )
        if status == 200:
            parsed = json.loads(body)
            if parsed.get("finished") and parsed["messageResults"]:
                results = parsed["messageResults"]
                assert results[0]["output_data"] == "hello"
                return
        time.sleep(0.02)
    pytest.fail("batch did not finish over HTTP")


def test_policy_roundtrip(runtime):
    status, body = post(GET_POLICY)
    assert status == 200 and body == "bin-pack"
    status, _ = post(SET_POLICY, "compact")
    assert status == 200
    _, body = post(GET_POLICY)
    assert body == "compact"
    post(SET_POLICY, "bin-pack")


def test_set_next_evicted_vm(runtime):
    status, _ = post(SET_NEXT_EVICTED_VM, "10.9.9.9")
    assert status == 200
    _, body = post(GET_IN_FLIGHT_APPS)
    assert "10.9.9.9" in json.loads(body)["nextEvictedVmIps"]
    post(SET_NEXT_EVICTED_VM, json.dumps([]))


def test_chaining_and_exec_graph(runtime):
    ber = _core.batch_exec_factory("http", "parent", 1)
    msgs = ber.messages
    msgs[0].record_exec_graph = True
    ber.messages = msgs
    root_id = ber.messages[0].id
    _core.call_functions(ber)
    # Parent + 1 chained child
    results = wait_for_batch(ber.app_id, 2, timeout_ms=30_000)
    parent = [r for r in results if r.id == root_id][0]
    assert parent.return_value == 0
    assert parent.output_data == "parent:from-parent"
    assert len(parent.chained_msg_ids) == 1

    status, body = post(
        GET_EXEC_GRAPH, json.dumps({"appId": ber.app_id, "id": root_id})
    )
    assert status == 200
    graph = json.loads(body)
    assert graph["root"]["msg"]["id"] == root_id
    assert len(graph["root"]["chained"]) == 1
    child = graph["root"]["chained"][0]["msg"]
    assert child["output_data"] == "from-parent"


def test_bad_request(runtime):
    status, _ = post(99)
    assert status == 400


def test_get_runtime_metrics(runtime):
    """GET_RUNTIME_METRICS exposes planner/broker table sizes (leak
    monitoring; extension beyond the reference op set)."""
    import json

    status, body = post(GET_RUNTIME_METRICS)
    assert status == 200
    m = json.loads(body)
    for key in ("appResults", "doneApps", "inFlightApps", "ptpMappings",
                "ptpChannels", "ptpSendSeqs", "decisionCache"):
        assert key in m and m[key] >= 0, m
