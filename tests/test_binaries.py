"""Deployment parity: the standalone planner_server + worker binaries run
as real processes and serve an EXECUTE_BATCH over the HTTP ops API
(reference: docker-compose deployment of planner + worker containers)."""

import base64
import json
import os
import signal
import subprocess
import time
import urllib.request

import pytest

REPO_ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BUILD = os.path.join(REPO_ROOT, "build")

PLANNER_OFF = 7000
WORKER_OFF = 7200

pytestmark = pytest.mark.skipif(
    not os.path.exists(os.path.join(BUILD, "planner_server")),
    reason="example binaries not built (run `make examples`)",
)


def post(http_type, payload="", timeout=10):
    body = json.dumps({"http_type": http_type, "payload": payload}).encode()
    req = urllib.request.Request(
        f"http://127.0.0.1:{8080 + PLANNER_OFF}/", data=body, method="POST"
    )
    try:
        with urllib.request.urlopen(req, timeout=timeout) as resp:
            return resp.status, resp.read().decode()
    except urllib.error.HTTPError as e:
        return e.code, e.read().decode()


@pytest.fixture(scope="module")
def binaries():
    env_common = {**os.environ, "LOG_LEVEL": "error"}
    planner = subprocess.Popen(
        [os.path.join(BUILD, "planner_server")],
        env={
            **env_common,
            "FAABRIC_PORT_OFFSET": str(PLANNER_OFF),
            "ENDPOINT_HOST": f"127.0.0.1@{PLANNER_OFF}",
        },
        stdout=subprocess.DEVNULL,
        stderr=subprocess.DEVNULL,
    )
    worker = subprocess.Popen(
        [os.path.join(BUILD, "server")],
        env={
            **env_common,
            "FAABRIC_PORT_OFFSET": str(WORKER_OFF),
            "ENDPOINT_HOST": f"127.0.0.1@{WORKER_OFF}",
            "PLANNER_HOST": f"127.0.0.1@{PLANNER_OFF}",
            "OVERRIDE_CPU_COUNT": "4",
            "FAABRIC_USE_GPU": "0",
        },
        stdout=subprocess.DEVNULL,
        stderr=subprocess.DEVNULL,
    )
    # Wait for registration
    deadline = time.monotonic() + 20
    ready = False
    while time.monotonic() < deadline:
        try:
            status, body = post(5)  # GET_AVAILABLE_HOSTS
            if status == 200 and len(json.loads(body)["hosts"]) == 1:
                ready = True
                break
        except Exception:
            pass
        time.sleep(0.1)
    yield ready
    for p in (worker, planner):
        p.send_signal(signal.SIGTERM)
    for p in (worker, planner):
        try:
            p.wait(timeout=10)
        except subprocess.TimeoutExpired:
            p.kill()


def test_binaries_serve_batch(binaries):
    assert binaries, "worker did not register with planner binary"
    ber = {
        "user": "demo",
        "function": "anything",
        "messages": [{}, {}, {}],
    }
    # The worker keep-alive is half the 5s host timeout; under a loaded
    # test machine a beat can slip, so tolerate a transient no-hosts
    deadline = time.monotonic() + 10
    while True:
        status, body = post(10, json.dumps(ber))  # EXECUTE_BATCH
        if status == 200 or time.monotonic() > deadline:
            break
        time.sleep(0.2)
    assert status == 200, body
    app_id = json.loads(body)["appId"]

    deadline = time.monotonic() + 15
    while time.monotonic() < deadline:
        status, body = post(11, json.dumps({"appId": app_id}))
        if status == 200:
            parsed = json.loads(body)
            if parsed.get("finished") and len(parsed["messageResults"]) == 3:
                outs = [m["output_data"] for m in parsed["messageResults"]]
                assert all("Example executor ran" in o for o in outs)
                return
        time.sleep(0.05)
    pytest.fail("batch did not finish through the binaries")


def test_is_app_migratable_cli(binaries):
    """Deployment CLI parity (reference src/planner/is_app_migratable.cpp):
    replays a DIST_CHANGE scheduling pass against the live planner state."""
    assert binaries
    cli = os.path.join(BUILD, "is_app_migratable")
    if not os.path.exists(cli):
        pytest.skip("is_app_migratable not built")
    env = {**os.environ, "FAABRIC_PORT_OFFSET": str(PLANNER_OFF)}

    # Unknown app -> exit 1
    out = subprocess.run(
        [cli, "999999"], env=env, capture_output=True, text=True, timeout=20
    )
    assert out.returncode == 1, out.stderr

    # In-flight app on a single host -> "NO" (bin-pack has nowhere better)
    ber = {
        "user": "bench",
        "function": "sleep",
        "messages": [
            {"input_data": base64.b64encode(b"3000").decode()},
            {"input_data": base64.b64encode(b"3000").decode()},
        ],
    }
    status, body = post(10, json.dumps(ber))
    assert status == 200, body
    app_id = json.loads(body)["appId"]
    out = subprocess.run(
        [cli, str(app_id)], env=env, capture_output=True, text=True,
        timeout=20,
    )
    assert out.returncode == 0, out.stderr
    assert out.stdout.strip() == "NO"
    # Drain the batch so the module fixture tears down cleanly
    deadline = time.monotonic() + 15
    while time.monotonic() < deadline:
        status, body = post(11, json.dumps({"appId": app_id}))
        if status == 200 and json.loads(body).get("finished"):
            break
        time.sleep(0.2)


def test_check_binary():
    """The minimal C++ embedder check binary (BASELINE config 1) runs
    end-to-end under a port offset."""
    check = os.path.join(BUILD, "check")
    if not os.path.exists(check):
        pytest.skip("check not built")
    out = subprocess.run(
        [check],
        env={**os.environ, "FAABRIC_PORT_OFFSET": "7600",
             "LOG_LEVEL": "error"},
        capture_output=True, text=True, timeout=60,
    )
    assert out.returncode == 0, out.stderr[-500:]
    assert "CHECK OK" in out.stdout


def test_disttest_binary():
    """Multi-process dist scenarios (two worker processes, leader
    collectives, live MPI migration) — same binary the asan-dist /
    tsan-dist targets run."""
    disttest = os.path.join(BUILD, "disttest")
    if not os.path.exists(disttest):
        pytest.skip("disttest not built")
    out = subprocess.run(
        [disttest],
        env={**os.environ, "DISTTEST_BASE_OFFSET": "7400",
             "DISTTEST_STOP_FILE": "/tmp/disttest-pytest.stop",
             "LOG_LEVEL": "error"},
        capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, (out.stdout[-300:], out.stderr[-500:])
    assert "DISTTEST OK" in out.stdout


def test_selftest_binary():
    """The C++ integration sweep (batches, THREADS merge, MPI world,
    chaining, SPSC queue, snapshot semantics) — same binary the
    sanitizer targets run."""
    selftest = os.path.join(BUILD, "selftest")
    if not os.path.exists(selftest):
        pytest.skip("selftest not built")
    out = subprocess.run(
        [selftest],
        env={**os.environ, "FAABRIC_PORT_OFFSET": "7700"},
        capture_output=True, text=True, timeout=120,
    )
    assert out.returncode == 0, out.stderr[-500:]
    assert "SELFTEST OK" in out.stdout
