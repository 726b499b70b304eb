"""Combined sender+receiver proxy — the single-service variant.

Parity: /root/reference/fed/proxy/barriers.py:339-412 (SenderReceiverProxyActor)
and base_proxy.py:77-106 — one service owns both directions, selected via
``fed.init(receiver_sender_proxy_cls=...)``.  Useful when a deployment wants
one port/identity per party for both roles.
"""
from __future__ import annotations

from typing import Dict, Optional

from rayfed_amd.config import CrossSiloMessageConfig
from rayfed_amd.proxy.base_proxy import SenderReceiverProxy
from rayfed_amd.proxy.tcp.tcp_proxy import TcpReceiverProxy, TcpSenderProxy


class TcpSenderReceiverProxy(SenderReceiverProxy):
    """Composes the TCP sender and receiver behind the combined SPI."""

    def __init__(
        self,
        addresses: Dict,
        listening_address: str,
        party: str,
        job_name: str,
        tls_config: Optional[Dict],
        proxy_config: Optional[CrossSiloMessageConfig] = None,
    ) -> None:
        super().__init__(
            addresses, listening_address, party, job_name, tls_config, proxy_config
        )
        self._sender = TcpSenderProxy(
            addresses, party, job_name, tls_config, proxy_config
        )
        self._receiver = TcpReceiverProxy(
            listening_address, party, job_name, tls_config, proxy_config
        )

    @property
    def gpu_plane(self):
        return self._receiver.gpu_plane

    @gpu_plane.setter
    def gpu_plane(self, plane):
        self._sender.gpu_plane = plane
        self._receiver.gpu_plane = plane

    @property
    def received_op_count(self) -> int:
        return self._receiver.received_op_count

    @property
    def _mailbox(self):
        return self._receiver._mailbox

    async def start(self):
        await self._receiver.start()

    async def send(self, dest_party, data, upstream_seq_id, downstream_seq_id):
        return await self._sender.send(
            dest_party, data, upstream_seq_id, downstream_seq_id
        )

    async def get_data(self, src_party, upstream_seq_id, curr_seq_id):
        return await self._receiver.get_data(src_party, upstream_seq_id, curr_seq_id)

    async def stop(self):
        await self._sender.stop()
        await self._receiver.stop()
