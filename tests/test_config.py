"""Config dataclasses + internal KV (parity: reference test_api.py config
readback, test_internal_kv.py, test_retry_policy.py plumb-through)."""
import cloudpickle

from rayfed_amd import config as fed_config
from rayfed_amd._private import constants, kv as kv_mod
from rayfed_amd.config import (
    CrossSiloMessageConfig,
    GpuDataPlaneConfig,
    GrpcCrossSiloMessageConfig,
)


def test_from_dict_filters_unknown_keys():
    cfg = CrossSiloMessageConfig.from_dict(
        {"timeout_in_ms": 1234, "bogus_key": "x", "exit_on_sending_failure": True}
    )
    assert cfg.timeout_in_ms == 1234
    assert cfg.exit_on_sending_failure is True
    assert not hasattr(cfg, "bogus_key")


def test_grpc_config_fields():
    cfg = GrpcCrossSiloMessageConfig.from_dict(
        {
            "grpc_retry_policy": {"maxAttempts": 2},
            "grpc_channel_options": [("grpc.max_send_message_length", 100)],
            "messages_max_size_in_bytes": 1000,
        }
    )
    assert cfg.grpc_retry_policy == {"maxAttempts": 2}
    assert cfg.messages_max_size_in_bytes == 1000


def test_json_roundtrip():
    cfg = CrossSiloMessageConfig(timeout_in_ms=77)
    again = CrossSiloMessageConfig.from_json(cfg.__json__())
    assert again.timeout_in_ms == 77


def test_gpu_plane_config_defaults():
    cfg = GpuDataPlaneConfig.from_dict({})
    assert cfg.chunk_bytes == 64 << 20
    assert cfg.verify_crc is True
    cfg2 = GpuDataPlaneConfig.from_dict({"verify_crc": False, "junk": 1})
    assert cfg2.verify_crc is False


def test_internal_kv_job_prefixed_keys():
    kv = kv_mod._init_internal_kv("myjob")
    try:
        kv.put("k1", b"v1")
        assert kv.get("k1") == b"v1"
        # The raw store key carries the job prefix (reference
        # compatible_utils.py:68-74 scheme RAYFED#{job}#{key}).
        assert b"RAYFED#myjob#k1" in kv_mod._store
        # Another job's KV does not see it.
        other = kv_mod.InternalKv("otherjob")
        assert other.get("k1") is None
        kv.delete("k1")
        assert kv.get("k1") is None
    finally:
        kv_mod._clear_internal_kv()
    assert kv_mod.kv is None


def test_cluster_config_readback():
    kv = kv_mod._init_internal_kv("j")
    try:
        kv.put(
            constants.KEY_OF_CLUSTER_CONFIG,
            cloudpickle.dumps(
                {
                    constants.KEY_OF_CLUSTER_ADDRESSES: {"alice": "127.0.0.1:1"},
                    constants.KEY_OF_CURRENT_PARTY_NAME: "alice",
                    constants.KEY_OF_TLS_CONFIG: None,
                }
            ),
        )
        fed_config._clear_cached_config()
        cc = fed_config.get_cluster_config()
        assert cc.cluster_addresses == {"alice": "127.0.0.1:1"}
        assert cc.current_party == "alice"
    finally:
        kv_mod._clear_internal_kv()
        fed_config._clear_cached_config()


def test_retry_policy_plumbs_into_channel_options():
    from rayfed_amd.proxy.grpc import grpc_options

    cfg = GrpcCrossSiloMessageConfig.from_dict(
        {"grpc_retry_policy": {"maxAttempts": 2, "initialBackoff": "1s"}}
    )
    options = grpc_options.parse_grpc_options(cfg)
    svc = dict(options)["grpc.service_config"]
    assert '"maxAttempts": 2' in svc


def test_channel_option_precedence_explicit_over_max_size():
    """Explicit grpc_channel_options win over messages_max_size_in_bytes
    (parity: reference test_grpc_options_on_proxies.py)."""
    from rayfed_amd.proxy.grpc import grpc_options

    cfg = GrpcCrossSiloMessageConfig.from_dict(
        {
            "messages_max_size_in_bytes": 1000,
            "grpc_channel_options": [("grpc.max_send_message_length", 77)],
        }
    )
    options = dict(grpc_options.parse_grpc_options(cfg))
    assert options["grpc.max_send_message_length"] == 77
    # The non-overridden one still reflects messages_max_size_in_bytes.
    assert options["grpc.max_receive_message_length"] == 1000


def test_gpu_plane_config_from_dict_filters_unknown_keys():
    from rayfed_amd.config import GpuDataPlaneConfig

    cfg = GpuDataPlaneConfig.from_dict(
        {"chunk_bytes": 123, "lazy_ipc": True, "device_checksum": "crc32",
         "not_a_knob": 1}
    )
    assert cfg.chunk_bytes == 123
    assert cfg.lazy_ipc is True
    assert cfg.device_checksum == "crc32"
    assert not hasattr(cfg, "not_a_knob")
    # Defaults preserved for unspecified fields.
    assert cfg.verify_crc is True and cfg.place_on_gpu is True
