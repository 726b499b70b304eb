"""Shared helpers: dependency resolution, logging, address validation, TLS.

Parity: /root/reference/fed/utils.py:48-254 (same helper surface; fresh
implementations).
"""
from __future__ import annotations

import ipaddress
import logging
import re
import subprocess
import sys
from typing import Dict, List, Tuple

from rayfed_amd import tree_util
from rayfed_amd.fed_object import FedObject
from rayfed_amd.runtime.object_ref import ObjectRef

logger = logging.getLogger(__name__)


def resolve_dependencies(current_party: str, current_fed_task_id, *args, **kwargs):
    """Replace every FedObject in (args, kwargs) with a local ObjectRef.

    Owned objects contribute their live ref; foreign objects get a recv
    barrier ref (cached on the FedObject so repeated consumption receives
    once).  Parity: reference utils.py:48-83.
    """
    flattened_args, args_tree = tree_util.tree_flatten((args, kwargs))
    indexes = []
    resolved = []
    for idx, arg in enumerate(flattened_args):
        if isinstance(arg, FedObject):
            indexes.append(idx)
            if arg.get_ray_object_ref() is not None:
                resolved.append(arg.get_ray_object_ref())
            else:
                from rayfed_amd.proxy.barriers import recv

                ref = recv(
                    current_party,
                    arg.get_party(),
                    arg.get_fed_task_id(),
                    current_fed_task_id,
                )
                arg._cache_ray_object_ref(ref)
                resolved.append(ref)
    if indexes:
        for idx, ref in zip(indexes, resolved):
            flattened_args[idx] = ref
    resolved_args, resolved_kwargs = tree_util.tree_unflatten(
        flattened_args, args_tree
    )
    return resolved_args, resolved_kwargs


def materialize(tree):
    """Block on every ObjectRef leaf in ``tree`` and substitute its value.
    (Run inside the task-submission wrapper so user functions see values,
    the way Ray resolves ObjectRef args before invoking a task.)"""
    leaves, spec = tree_util.tree_flatten(tree)
    out = [
        leaf.result() if isinstance(leaf, ObjectRef) else leaf for leaf in leaves
    ]
    return tree_util.tree_unflatten(out, spec)


# -- logging ------------------------------------------------------------------
class _ContextFilter(logging.Filter):
    def __init__(self, party: str, job_name: str):
        super().__init__()
        self._party = party
        self._job_name = job_name

    def filter(self, record: logging.LogRecord) -> bool:
        record.party = self._party
        record.jobname = self._job_name
        return True


def setup_logger(
    logging_level="info",
    logging_format=None,
    date_format=None,
    party: str = "",
    job_name: str = "",
) -> None:
    """Install the party/job-aware log format on the rayfed logger tree.
    Parity: reference utils.py:99-146."""
    from rayfed_amd._private import constants

    logging_format = logging_format or constants.RAYFED_LOG_FMT
    date_format = date_format or constants.RAYFED_DATE_FMT
    root = logging.getLogger("rayfed_amd")
    if isinstance(logging_level, str):
        logging_level = getattr(logging, logging_level.upper())
    root.setLevel(logging_level)
    for h in list(root.handlers):
        root.removeHandler(h)
    handler = logging.StreamHandler(stream=sys.stderr)
    handler.setFormatter(logging.Formatter(logging_format, date_format))
    handler.addFilter(_ContextFilter(party, job_name))
    root.addHandler(handler)
    root.propagate = False


# -- address validation --------------------------------------------------------
_HOSTPORT_RE = re.compile(
    r"^(?P<host>[A-Za-z0-9._-]+)?:(?P<port>\d{1,5})$"
)


def validate_address(address: str) -> None:
    """Accept ip:port, hostname:port, http(s)://..., or 'local'.
    Parity: reference utils.py:198-228."""
    if not isinstance(address, str):
        raise ValueError(f"address must be a str, got {type(address)}")
    if address == "local":
        return
    if address.startswith("http://") or address.startswith("https://"):
        return
    m = _HOSTPORT_RE.match(address)
    if m:
        port = int(m.group("port"))
        if not 0 < port < 65536:
            raise ValueError(f"invalid port in address {address!r}")
        host = m.group("host")
        if host:
            try:
                ipaddress.ip_address(host)
                return
            except ValueError:
                pass  # not an IP — treat as hostname
            if re.match(r"^[A-Za-z0-9]([A-Za-z0-9._-]*[A-Za-z0-9])?$", host):
                return
        raise ValueError(f"invalid host in address {address!r}")
    raise ValueError(
        f"invalid address {address!r}: expect 'host:port', 'http(s)://…' or "
        f"'local'"
    )


def validate_addresses(addresses: Dict[str, str]) -> None:
    if not isinstance(addresses, dict) or not addresses:
        raise ValueError("addresses must be a non-empty dict of party -> address")
    for party, address in addresses.items():
        if not isinstance(party, str):
            raise ValueError(f"party name must be a str, got {party!r}")
        validate_address(address)


# -- TLS -----------------------------------------------------------------------
def load_cert_config(cert_config: Dict[str, str]) -> Tuple[bytes, bytes, bytes]:
    """Read (ca_cert, private_key, cert_chain) bytes from a tls_config dict
    with keys ``ca_cert`` / ``key`` / ``cert``.  Parity: utils.py:153-163."""
    ca_cert, private_key, cert_chain = None, None, None
    if "ca_cert" in cert_config:
        with open(cert_config["ca_cert"], "rb") as f:
            ca_cert = f.read()
    if "key" in cert_config:
        with open(cert_config["key"], "rb") as f:
            private_key = f.read()
    if "cert" in cert_config:
        with open(cert_config["cert"], "rb") as f:
            cert_chain = f.read()
    return ca_cert, private_key, cert_chain


# -- misc ----------------------------------------------------------------------
def dict2tuple(dic) -> List[Tuple]:
    """Channel-options dict → list of (k, v) tuples (utils.py:182-195)."""
    if dic is None:
        return []
    if isinstance(dic, dict):
        return [(k, v) for k, v in dic.items()]
    return list(dic)


def is_cython(obj) -> bool:
    """Detect cython functions/methods (utils.py:166-179)."""

    def check(x):
        return (
            type(x).__name__ == "cython_function_or_method"
            or (hasattr(x, "__func__") and check(x.__func__))
        )

    return check(obj)


def start_command(command: str, timeout: int = 60) -> str:
    """Run a shell command, returning stdout; raise on stderr output
    (utils.py:242-254)."""
    proc = subprocess.Popen(
        command, shell=True, stdout=subprocess.PIPE, stderr=subprocess.PIPE
    )
    out, err = proc.communicate(timeout=timeout)
    if err:
        raise RuntimeError(f"command failed: {err.decode()}")
    return out.decode()
