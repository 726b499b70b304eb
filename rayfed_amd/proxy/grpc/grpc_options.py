"""gRPC channel/server option defaults.

Parity: /root/reference/fed/proxy/grpc/grpc_options.py:19-83 — same defaults
(500 MiB message cap, retry policy, reuseport off, retries on) expressed for
the generic-method transport (the retry policy rides in a ``grpc.service_config``
JSON matched against our generic method name).
"""
from __future__ import annotations

import json
from typing import Dict, List, Optional, Tuple

from rayfed_amd.proxy.grpc import frames

_GRPC_MAX_SEND_MESSAGE_LENGTH = 500 * 1024 * 1024
_GRPC_MAX_RECEIVE_MESSAGE_LENGTH = 500 * 1024 * 1024

_DEFAULT_GRPC_RETRY_POLICY = {
    "maxAttempts": 5,
    "initialBackoff": "5s",
    "maxBackoff": "30s",
    "backoffMultiplier": 2,
    "retryableStatusCodes": ["UNAVAILABLE"],
}

_GRPC_SERVICE = frames.SERVICE_NAME


def get_default_grpc_retry_policy() -> Dict:
    return dict(_DEFAULT_GRPC_RETRY_POLICY)


def _make_service_config(retry_policy: Optional[Dict]) -> str:
    return json.dumps(
        {
            "methodConfig": [
                {
                    "name": [{"service": _GRPC_SERVICE}],
                    "retryPolicy": retry_policy or get_default_grpc_retry_policy(),
                }
            ]
        }
    )


def get_grpc_options(
    retry_policy: Optional[Dict] = None,
    max_send_message_length: Optional[int] = None,
    max_receive_message_length: Optional[int] = None,
) -> List[Tuple[str, object]]:
    return [
        (
            "grpc.max_send_message_length",
            max_send_message_length or _GRPC_MAX_SEND_MESSAGE_LENGTH,
        ),
        (
            "grpc.max_receive_message_length",
            max_receive_message_length or _GRPC_MAX_RECEIVE_MESSAGE_LENGTH,
        ),
        ("grpc.so_reuseport", 0),
        ("grpc.enable_retries", 1),
        ("grpc.service_config", _make_service_config(retry_policy)),
    ]


def parse_grpc_options(proxy_config) -> List[Tuple[str, object]]:
    """Map the job's cross-silo config onto channel args.

    Precedence (parity with grpc_options.py:50-99 /
    test_grpc_options_on_proxies): explicit ``grpc_channel_options`` >
    ``messages_max_size_in_bytes`` > defaults.
    """
    retry_policy = None
    max_msg = None
    explicit: List[Tuple[str, object]] = []
    if proxy_config is not None:
        max_msg = getattr(proxy_config, "messages_max_size_in_bytes", None)
        retry_policy = getattr(proxy_config, "grpc_retry_policy", None)
        raw = getattr(proxy_config, "grpc_channel_options", None)
        if raw:
            explicit = [tuple(o) for o in raw]
    options = get_grpc_options(
        retry_policy=retry_policy,
        max_send_message_length=max_msg,
        max_receive_message_length=max_msg,
    )
    if explicit:
        keys = {k for k, _ in explicit}
        options = [(k, v) for k, v in options if k not in keys] + explicit
    return options
