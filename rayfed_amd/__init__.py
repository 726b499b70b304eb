"""rayfed_amd — an MI355X-native cross-party federated execution engine.

A from-scratch implementation of the RayFed programming model
(``fed.init`` / ``@fed.remote`` / ``.party()`` / ``fed.get`` /
``fed.shutdown``, multi-controller symmetric execution, push-based
cross-silo data movement) built for one AMD MI355X (gfx950) node:

- in-process task/actor substrate instead of Ray (``rayfed_amd.runtime``);
- cross-party transport over raw-bytes gRPC frames with an optional TLS
  perimeter (``rayfed_amd.proxy``);
- GPU data plane: HIP pack/unpack + CRC32 kernels, pinned staging,
  side-stream overlap (``rayfed_amd.ops``);
- intra-party data parallelism over RCCL/xGMI (``rayfed_amd.parallel``).

Usage is a drop-in for the reference::

    import rayfed_amd as fed
    fed.init(addresses={...}, party="alice")

Reference API surface: /root/reference/fed/__init__.py:15-30.
"""

from rayfed_amd.api import get, init, kill, remote, shutdown, stats
from rayfed_amd.exceptions import FedRemoteError
from rayfed_amd.fed_object import FedObject
from rayfed_amd.proxy.barriers import recv, send

__version__ = "0.2.0"

__all__ = [
    "get",
    "init",
    "kill",
    "remote",
    "shutdown",
    "stats",
    "send",
    "recv",
    "FedObject",
    "FedRemoteError",
]
