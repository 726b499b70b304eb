"""gRPC cross-silo transport: sender/receiver proxies over generic methods.

Parity: /root/reference/fed/proxy/grpc/grpc_proxy.py (GrpcSenderProxy
:119-220, GrpcReceiverProxy :223-298, SendDataService :301-342,
_run_grpc_server :345-381) with these deliberate differences:

- **No protobuf**: frames are hand-packed (``frames.py``) and registered via
  ``grpc.method_handlers_generic_handler`` with identity serializers, so a
  multi-GiB tensor payload crosses the stack with no protobuf copy.
- **asyncio futures instead of event+dict pairs** for the receive mailbox:
  one ``{(up, down): Future}`` map; a send that arrives before its reader
  parks the payload on a resolved future, a reader that arrives first awaits.
- **Tensor payloads** ride the tensor codec (``rayfed_amd.ops.tensor_codec``)
  — manifest in the frame header, raw bytes in the payload — with optional
  HIP pack/CRC staging when a GPU data plane is attached.
"""
from __future__ import annotations

import logging
from typing import Dict, Optional

import grpc

from rayfed_amd import config as fed_config
from rayfed_amd.proxy import base_proxy
from rayfed_amd.proxy._mailbox import Mailbox
from rayfed_amd.proxy.grpc import frames, grpc_options
from rayfed_amd.utils import load_cert_config

logger = logging.getLogger(__name__)


class GrpcSenderProxy(base_proxy.SenderProxy):
    def __init__(self, addresses, party, job_name, tls_config, proxy_config=None):
        if proxy_config is not None and not isinstance(
            proxy_config, fed_config.CrossSiloMessageConfig
        ):
            proxy_config = fed_config.GrpcCrossSiloMessageConfig.from_dict(proxy_config)
        super().__init__(addresses, party, job_name, tls_config, proxy_config)
        self._channels: Dict[str, grpc.aio.Channel] = {}
        self._stubs: Dict[str, grpc.aio.UnaryUnaryMultiCallable] = {}
        self.gpu_plane = None  # attached by barriers when a GPU is present
        self.last_sent_bytes = 0
        self._metadata = []
        if proxy_config is not None and getattr(proxy_config, "http_header", None):
            self._metadata = [
                (k.lower(), v) for k, v in proxy_config.http_header.items()
            ]
        self._metadata.append(("x-rayfed-job", job_name))

    # -- channel management ---------------------------------------------------
    def _get_stub(self, dest_party: str) -> grpc.aio.UnaryUnaryMultiCallable:
        stub = self._stubs.get(dest_party)
        if stub is not None:
            return stub
        if dest_party not in self._addresses:
            raise ValueError(f"unknown dest party {dest_party!r}")
        address = self._addresses[dest_party]
        options = grpc_options.parse_grpc_options(self._proxy_config)
        if self._tls_config:
            ca_cert, private_key, cert_chain = load_cert_config(self._tls_config)
            credentials = grpc.ssl_channel_credentials(
                root_certificates=ca_cert,
                private_key=private_key,
                certificate_chain=cert_chain,
            )
            # Test certs are usually issued for a DNS name while parties dial
            # bare IPs; allow overriding the expected server name.
            override = self._tls_config.get("target_name_override")
            if override:
                options = options + [("grpc.ssl_target_name_override", override)]
            channel = grpc.aio.secure_channel(address, credentials, options=options)
        else:
            channel = grpc.aio.insecure_channel(address, options=options)
        self._channels[dest_party] = channel
        stub = channel.unary_unary(
            frames.SEND_DATA_METHOD,
            request_serializer=frames.identity_serializer,
            response_deserializer=frames.identity_deserializer,
        )
        self._stubs[dest_party] = stub
        return stub

    # -- sending --------------------------------------------------------------
    async def send(self, dest_party, data, upstream_seq_id, downstream_seq_id):
        stub = self._get_stub(dest_party)
        req = await self._encode_request(data, upstream_seq_id, downstream_seq_id)
        request = req.to_bytes()
        self.last_sent_bytes = len(request)
        req.release()  # joined into one protobuf-free bytes body
        timeout = 60.0
        if self._proxy_config is not None and self._proxy_config.timeout_in_ms:
            timeout = self._proxy_config.timeout_in_ms / 1000.0
        response_bytes = await stub(
            request, metadata=self._metadata, timeout=timeout
        )
        response = frames.decode_response(response_bytes)
        code = response.get("code", 500)
        if 400 <= code < 500:
            # Client-class error (e.g. 417 job-name mismatch): not retryable.
            raise RuntimeError(
                f"[{code}] send to {dest_party} rejected: "
                f"{response.get('result', '')}. The receiver may be serving a "
                f"different job than {self._job_name!r}."
            )
        if code >= 500:
            raise RuntimeError(
                f"[{code}] send to {dest_party} failed: {response.get('result', '')}"
            )
        return True

    async def _encode_request(self, data, upstream_seq_id, downstream_seq_id) -> bytes:
        from rayfed_amd.proxy._encode import encode_request

        return await encode_request(
            self._job_name, data, upstream_seq_id, downstream_seq_id, self.gpu_plane
        )

    async def get_proxy_config(self, dest_party: Optional[str] = None):
        return self._proxy_config

    async def stop(self):
        for ch in self._channels.values():
            await ch.close()
        self._channels.clear()
        self._stubs.clear()


class GrpcReceiverProxy(base_proxy.ReceiverProxy):
    def __init__(self, listening_address, party, job_name, tls_config, proxy_config=None):
        if proxy_config is not None and not isinstance(
            proxy_config, fed_config.CrossSiloMessageConfig
        ):
            proxy_config = fed_config.GrpcCrossSiloMessageConfig.from_dict(proxy_config)
        super().__init__(listening_address, party, job_name, tls_config, proxy_config)
        self._server: Optional[grpc.aio.Server] = None
        self._mailbox = Mailbox(
            job_name,
            proxy_config.serializing_allowed_list if proxy_config else None,
        )

    # -- server ---------------------------------------------------------------
    async def start(self):
        port = self._listening_address[self._listening_address.index(":") + 1 :]
        options = grpc_options.parse_grpc_options(self._proxy_config)
        server = grpc.aio.server(options=options)
        handler = grpc.unary_unary_rpc_method_handler(
            self._handle_send_data,
            request_deserializer=frames.identity_deserializer,
            response_serializer=frames.identity_serializer,
        )
        generic = grpc.method_handlers_generic_handler(
            frames.SERVICE_NAME, {"SendData": handler}
        )
        server.add_generic_rpc_handlers((generic,))
        try:
            if self._tls_config:
                ca_cert, private_key, cert_chain = load_cert_config(self._tls_config)
                credentials = grpc.ssl_server_credentials(
                    [(private_key, cert_chain)],
                    root_certificates=ca_cert,
                    require_client_auth=ca_cert is not None,
                )
                bound = server.add_secure_port(f"[::]:{port}", credentials)
            else:
                bound = server.add_insecure_port(f"[::]:{port}")
        except RuntimeError as e:  # grpc ≥1.60 raises instead of returning 0
            raise AssertionError(
                f"Failed to listen on port {port}: it is in use ({e}). "
                f"Choose another port in the cluster addresses."
            ) from e
        assert str(bound) == str(port), (
            f"Failed to listen on port {port}: it is in use (got {bound}). "
            f"Choose another port in the cluster addresses."
        )
        await server.start()
        self._server = server
        logger.info("Receiver proxy of %s listening on %s", self._party, port)

    @property
    def gpu_plane(self):
        return self._mailbox.gpu_plane

    @gpu_plane.setter
    def gpu_plane(self, plane):
        self._mailbox.gpu_plane = plane

    @property
    def received_op_count(self) -> int:
        return self._mailbox.received_op_count

    async def _handle_send_data(self, request: bytes, context) -> bytes:
        try:
            kind, header, payload = frames.decode_frame(request)
        except ValueError as e:
            return frames.encode_response(400, f"bad frame: {e}")
        code, result = self._mailbox.deliver(kind, header, payload)
        if code == 417:
            logger.warning("Rejected message: %s", result)
        return frames.encode_response(code, result)

    # -- consumption ----------------------------------------------------------
    async def get_data(self, src_party, upstream_seq_id, curr_seq_id):
        return await self._mailbox.get_data(upstream_seq_id, curr_seq_id)

    async def stop(self):
        if self._server is not None:
            await self._server.stop(grace=None)
            self._server = None
