"""Deployment glue: start planner/worker runtimes and submit batches.

The native core (cpp/) does all the work; these helpers wire processes
together the way the reference's docker-compose deployment does
(reference: docker-compose.yml:1-62 — one planner, N workers), except the
MI355X deployment shape is N single-GPU worker processes per node
distinguished by a per-process port offset ("ip@offset" host identities).
"""

import time

from faabric_amd import _core

DEFAULT_TIMEOUT_MS = 30_000


class LocalRuntime:
    """A planner and/or worker running inside this process."""

    def __init__(
        self,
        host: str = "127.0.0.1",
        port_offset: int = 0,
        planner_host: str = "127.0.0.1",
        planner_port_offset: int = 0,
        slots: int | None = None,
    ):
        self.host = host
        self.port_offset = port_offset
        self.planner_host = planner_host
        self.planner_port_offset = planner_port_offset
        self.slots = slots
        self._planner = None
        self._worker = None

    @property
    def identity(self) -> str:
        if self.port_offset:
            return f"{self.host}@{self.port_offset}"
        return self.host

    def _configure(self):
        _core.set_port_offset(self.port_offset)
        _core.set_endpoint_host(self.identity)
        if self.planner_port_offset:
            _core.set_planner_host(
                f"{self.planner_host}@{self.planner_port_offset}"
            )
        else:
            _core.set_planner_host(self.planner_host)
        if self.slots is not None:
            _core.set_this_host_resources(self.slots, 0)

    def start_planner(
        self,
        with_snapshot_server: bool = True,
        with_state_server: bool = False,
    ):
        """with_state_server hosts the global KV + scripted-lock store
        for the "planner" state mode (the reference's Redis role)."""
        self._configure()
        self._planner = _core.PlannerRuntime()
        self._planner.start(with_snapshot_server, with_state_server)
        return self

    def start_worker(self):
        self._configure()
        self._worker = _core.FaabricMain()
        self._worker.start_background()
        return self

    def stop(self):
        if self._worker is not None:
            self._worker.shutdown()
            self._worker = None
        if self._planner is not None:
            self._planner.shutdown()
            self._planner = None


def execute_batch(
    user: str,
    function: str,
    count: int,
    input_data: bytes = b"",
    batch_type=None,
    timeout_ms: int = DEFAULT_TIMEOUT_MS,
):
    """Submit a batch and wait for all results (like the reference's
    EXECUTE_BATCH + EXECUTE_BATCH_STATUS HTTP flow)."""
    ber = _core.batch_exec_factory(user, function, count)
    if batch_type is not None:
        ber.type = batch_type
    if input_data:
        msgs = ber.messages
        for m in msgs:
            m.input_data = input_data
        ber.messages = msgs
    decision = _core.call_functions(ber)
    if decision.app_id in (_core.NOT_ENOUGH_SLOTS(), _core.MUST_FREEZE()):
        raise RuntimeError(f"batch not scheduled: {decision.app_id}")
    return wait_for_batch(ber.app_id, count, timeout_ms)


def wait_for_batch(app_id: int, count: int, timeout_ms: int = DEFAULT_TIMEOUT_MS):
    deadline = time.monotonic() + timeout_ms / 1000.0
    # Event-driven: the planner pushes BATCH_DONE when the last result
    # lands; wait_batch_done sleeps on a local flag (with its own coarse
    # fallback poll), then one status fetch collects the results
    _core.wait_batch_done(app_id, timeout_ms)
    while time.monotonic() < deadline:
        status = _core.get_batch_results(app_id)
        if (
            status.expected_num_messages != -1
            and status.finished
            and len(status.message_results) >= count
        ):
            return sorted(status.message_results, key=lambda m: m.app_idx)
        time.sleep(0.002)
    raise TimeoutError(f"batch {app_id} did not finish in {timeout_ms} ms")
