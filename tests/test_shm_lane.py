"""Same-host shared-memory lane (CPU paths; GPU variants in test_gpu_plane)."""
import pytest

torch = pytest.importorskip("torch")

import rayfed_amd as fed  # noqa: E402
from rayfed_amd.ops import shm_pool, tensor_codec  # noqa: E402
from tests._util import run_parties  # noqa: E402


def test_codec_shm_roundtrip_cpu():
    big = torch.arange(1 << 20, dtype=torch.float32)  # 4 MiB >= threshold
    small = torch.arange(10)
    obj = {"big": big, "small": small, "x": 42}
    extras, parts = tensor_codec.encode(obj, None, shm=True)
    mans = extras["tensors"]
    assert any("shm" in m for m in mans), "big tensor should ride shm"
    big_man = next(m for m in mans if "shm" in m)
    assert big_man["nbytes"] == big.numel() * 4
    # Small tensor stays in the payload.
    assert sum("shm" not in m for m in mans) == 1
    payload = b"".join(bytes(p) for p in parts)
    out = tensor_codec.decode(extras, memoryview(payload), None, None)
    assert torch.equal(out["big"], big)
    assert torch.equal(out["small"], small)
    assert out["x"] == 42
    tensor_codec.release_parts(extras)
    shm_pool.detach_all()


def test_shm_segments_recycled():
    t = torch.ones(1 << 19, dtype=torch.float32)  # 2 MiB
    extras1, _ = tensor_codec.encode(t, None, shm=True)
    name1 = extras1["tensors"][0]["shm"]
    tensor_codec.release_parts(extras1)
    extras2, _ = tensor_codec.encode(t, None, shm=True)
    name2 = extras2["tensors"][0]["shm"]
    tensor_codec.release_parts(extras2)
    assert name1 == name2, "pooled segment should be reused"
    shm_pool.detach_all()


def test_below_threshold_stays_on_socket():
    t = torch.ones(100)
    extras, parts = tensor_codec.encode(t, None, shm=True)
    assert all("shm" not in m for m in extras["tensors"])
    assert len(parts) == 2  # skeleton + tensor bytes


def _driver_big_tensor(party, addresses):
    fed.init(addresses=addresses, party=party, logging_level="warning")

    @fed.remote
    def make():
        return {"w": torch.arange(1 << 21, dtype=torch.float32), "tag": "big"}

    @fed.remote
    def consume(d):
        return float(d["w"].sum()) if d["tag"] == "big" else -1.0

    o = make.party("alice").remote()
    r = consume.party("bob").remote(o)
    n = 1 << 21
    assert fed.get(r) == float(n * (n - 1) / 2)
    fed.shutdown()


def test_two_party_big_tensor_over_shm_lane():
    """8 MiB tensor alice→bob on loopback: rides the shm lane end to end
    (ack-after-consume, segment recycling) and decodes identically."""
    run_parties(_driver_big_tensor)


def _driver_many_pushes(party, addresses):
    fed.init(addresses=addresses, party=party, logging_level="warning")

    @fed.remote
    def make(i):
        return torch.full((1 << 19,), float(i))

    @fed.remote
    def total(*vals):
        return sum(float(v[0]) for v in vals)

    objs = [make.party("alice").remote(i) for i in range(6)]
    r = total.party("bob").remote(*objs)
    assert fed.get(r) == sum(range(6))
    fed.shutdown()


def test_repeated_pushes_recycle_segments():
    run_parties(_driver_many_pushes)
