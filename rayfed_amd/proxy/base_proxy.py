"""Proxy SPI — pluggable cross-party transports.

Parity: /root/reference/fed/proxy/base_proxy.py:21-106 — same three abstract
roles and constructor signature ``(addresses, party, job_name, tls_config,
proxy_config)``, pluggable via ``fed.init(sender_proxy_cls=...,
receiver_proxy_cls=..., receiver_sender_proxy_cls=...)``.

Difference from the reference: proxies here are **async services hosted on the
driver's I/O event loop** (see ``rayfed_amd.proxy.barriers``), not Ray actor
processes — methods are coroutines.
"""
from __future__ import annotations

import abc
from typing import Any, Dict, Optional

from rayfed_amd.config import CrossSiloMessageConfig


class SenderProxy(abc.ABC):
    def __init__(
        self,
        addresses: Dict,
        party: str,
        job_name: str,
        tls_config: Optional[Dict],
        proxy_config: Optional[CrossSiloMessageConfig] = None,
    ) -> None:
        self._addresses = addresses
        self._party = party
        self._job_name = job_name
        self._tls_config = tls_config
        self._proxy_config = proxy_config

    @abc.abstractmethod
    async def send(
        self,
        dest_party: str,
        data: Any,
        upstream_seq_id,
        downstream_seq_id,
    ):
        """Push serialized ``data`` to ``dest_party``; return truthy on ack."""

    async def is_ready(self):
        return True

    async def get_proxy_config(self, dest_party: Optional[str] = None):
        return self._proxy_config

    async def stop(self):
        pass


class ReceiverProxy(abc.ABC):
    def __init__(
        self,
        listening_address: str,
        party: str,
        job_name: str,
        tls_config: Optional[Dict],
        proxy_config: Optional[CrossSiloMessageConfig] = None,
    ) -> None:
        self._listening_address = listening_address
        self._party = party
        self._job_name = job_name
        self._tls_config = tls_config
        self._proxy_config = proxy_config

    @abc.abstractmethod
    async def start(self):
        """Bind and start serving; raise if the address is unavailable."""

    @abc.abstractmethod
    async def get_data(self, src_party: str, upstream_seq_id, curr_seq_id):
        """Await the (upstream_seq_id, curr_seq_id) mailbox slot and return
        the deserialized payload."""

    async def is_ready(self):
        return True

    async def get_proxy_config(self):
        return self._proxy_config

    async def stop(self):
        pass


class SenderReceiverProxy(abc.ABC):
    """Combined single-service variant (reference base_proxy.py:77-106)."""

    def __init__(
        self,
        addresses: Dict,
        listening_address: str,
        party: str,
        job_name: str,
        tls_config: Optional[Dict],
        proxy_config: Optional[CrossSiloMessageConfig] = None,
    ) -> None:
        self._addresses = addresses
        self._listening_address = listening_address
        self._party = party
        self._job_name = job_name
        self._tls_config = tls_config
        self._proxy_config = proxy_config

    @abc.abstractmethod
    async def start(self):
        ...

    @abc.abstractmethod
    async def send(self, dest_party, data, upstream_seq_id, downstream_seq_id):
        ...

    @abc.abstractmethod
    async def get_data(self, src_party, upstream_seq_id, curr_seq_id):
        ...

    async def is_ready(self):
        return True

    async def stop(self):
        pass
