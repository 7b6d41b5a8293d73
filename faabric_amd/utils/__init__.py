"""Config, ids, logging and introspection helpers (faabric util parity)."""

from faabric_amd._core import (  # noqa: F401
    generate_gid,
    get_endpoint_host,
    get_port_offset,
    get_primary_ip,
    get_usable_cores,
    prof_clear,
    prof_summary,
    set_endpoint_host,
    set_log_level,
    set_mock_mode,
    set_planner_host,
    set_port_offset,
    set_up_crash_handler,
)
