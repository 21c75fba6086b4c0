"""registrar_amd — MI355X-host-native service-registration framework.

A from-scratch re-implementation of the capabilities of Joyent/Triton's
`registrar` (the ZooKeeper-backed DNS service-discovery sidecar read by
Binder), built MI355X-first: one registrar process per GPU on an 8-GPU node,
a native C++ core (epoll event loops, a hand-written ZooKeeper jute-protocol
client, a synthetic in-process ZK ensemble for hermetic tests/benches), a
GPU-liveness health gate via rocm-smi/amdsmi, and xGMI-local-rank
advertisement in the registration payload.

The native core lives in `registrar_amd._core` (pybind11 extension built
in-tree from `registrar_amd/csrc/`); `registrard` is the standalone daemon
binary built from the same sources.

Reference parity map: SURVEY.md §2 (component inventory) and §2.4/§2.5
(client-verb surface and config schema).
"""

from registrar_amd._version import __version__

try:
    from registrar_amd import _core
except ImportError as _e:  # pragma: no cover
    raise ImportError(
        "registrar_amd._core native extension is not built. "
        "Run `python setup.py build_ext --inplace` (or `make`) at the repo "
        "root. Original error: %s" % _e
    ) from _e

# Re-export the native surface flat, mirroring the reference's lib/index.js
# which re-exports zk/register/health flat (reference: lib/index.js:180-186).
from registrar_amd._core import (  # noqa: F401
    ZOK,
    ZNONODE,
    ZNODEEXISTS,
    ZNOTEMPTY,
    ZBADVERSION,
    ZSESSIONEXPIRED,
    ZCONNECTIONLOSS,
    ZNOCHILDRENFOREPHEMERALS,
    Ensemble,
    HealthCheck,
    Orchestrator,
    PreparedRegistration,
    ZkClient,
    build_host_record,
    build_node_list,
    build_service_record,
    discover_gpus,
    domain_to_path,
    error_name,
    exec_with_timeout,
    gpu_alive,
    gpu_count,
    gpu_health_command,
    register_node,
    self_address,
    self_hostname,
    unregister_node,
    xgmi_local_rank,
)

__all__ = [
    "__version__",
    "ZOK",
    "ZNONODE",
    "ZNODEEXISTS",
    "ZNOTEMPTY",
    "ZBADVERSION",
    "ZSESSIONEXPIRED",
    "ZCONNECTIONLOSS",
    "ZNOCHILDRENFOREPHEMERALS",
    "Ensemble",
    "HealthCheck",
    "Orchestrator",
    "PreparedRegistration",
    "ZkClient",
    "build_host_record",
    "build_node_list",
    "build_service_record",
    "discover_gpus",
    "domain_to_path",
    "error_name",
    "exec_with_timeout",
    "gpu_alive",
    "gpu_count",
    "gpu_health_command",
    "register_node",
    "self_address",
    "self_hostname",
    "unregister_node",
    "xgmi_local_rank",
]
