"""Serialization: cloudpickle + receiver-side restricted unpickling.

Parity: /root/reference/fed/_private/serialization_utils.py:41-83 — the
receiver enforces a class whitelist while unpickling cross-party payloads
(defense against pickle gadget injection from a peer silo).  Whitelist
format: ``{module_name: [class, ...] | "*"}``; a module entry of ``"*"``
allows every attribute of that module, and a listed class name must match
exactly.  Unlisted → ``pickle.UnpicklingError``.

Unlike the reference (which monkey-patches ``cloudpickle.loads`` process-wide,
serialization_utils.py:83) the restriction here is applied explicitly on the
receive path only — the sender and intra-party paths keep full pickle.
"""
from __future__ import annotations

import io
import pickle
from typing import Dict, Optional

import cloudpickle

dumps = cloudpickle.dumps


# Always-allowed internals: the engine's own wire types (tensor placeholder
# rebuild hook, cross-party error carrier).  Without these a whitelist-using
# job could never receive a tensor payload or an error object.
_IMPLICIT_ALLOWED = {
    "rayfed_amd.ops.tensor_codec": ["_rebuild_placeholder", "_TensorPlaceholder"],
    "rayfed_amd.exceptions": ["FedRemoteError"],
}


class RestrictedUnpickler(pickle.Unpickler):
    def __init__(self, file, allowed_list: Dict[str, object]):
        super().__init__(file)
        self._allowed_list = dict(_IMPLICIT_ALLOWED)
        self._allowed_list.update(allowed_list or {})

    def _is_allowed(self, module: str, name: str) -> bool:
        for mod, classes in self._allowed_list.items():
            if module == mod or module.startswith(mod + "."):
                if classes == "*" or classes is None:
                    return True
                if isinstance(classes, (list, tuple, set)) and (
                    name in classes or "*" in classes
                ):
                    return True
                if classes == name:
                    return True
        return False

    def find_class(self, module: str, name: str):
        if self._is_allowed(module, name):
            return super().find_class(module, name)
        raise pickle.UnpicklingError(
            f"global '{module}.{name}' is forbidden by the serializing "
            f"allowed list"
        )


def restricted_loads(data: bytes, allowed_list: Dict[str, object]):
    return RestrictedUnpickler(io.BytesIO(data), allowed_list).load()


def loads(data: bytes, allowed_list: Optional[Dict[str, object]] = None):
    """Deserialize a cross-party payload, honoring the whitelist if set."""
    if allowed_list:
        return restricted_loads(data, allowed_list)
    return cloudpickle.loads(data)
