"""Tensor-aware codec: CPU wire-format round-trips + CRC behavior.

GPU variants (HIP pack kernel / pinned staging) live in test_gpu_plane.py
and are marked ``gpu``.
"""
import pytest

torch = pytest.importorskip("torch")

from rayfed_amd.ops import tensor_codec


def _roundtrip(obj, allowed=None):
    extras, parts = tensor_codec.encode(obj)
    payload = b"".join(bytes(p) for p in parts)
    return extras, tensor_codec.decode(extras, memoryview(payload), None, allowed)


def test_plain_object_no_tensors():
    extras, out = _roundtrip({"a": 1, "b": [2, 3]})
    assert extras["tensors"] == []
    assert out == {"a": 1, "b": [2, 3]}


@pytest.mark.parametrize(
    "dtype",
    [torch.float32, torch.float64, torch.bfloat16, torch.float16,
     torch.int64, torch.uint8, torch.bool, torch.complex64],
)
def test_single_tensor_roundtrip(dtype):
    if dtype is torch.bool:
        t = (torch.arange(257) % 2).to(dtype)
    else:
        t = (torch.arange(257) % 128).to(dtype)
    extras, out = _roundtrip(t)
    assert len(extras["tensors"]) == 1
    man = extras["tensors"][0]
    assert man["dtype"] == str(dtype).replace("torch.", "")
    assert man["shape"] == [257]
    assert torch.equal(out, t)
    assert out.dtype == dtype


def test_nested_tensors_anywhere():
    class Holder:
        def __init__(self, t):
            self.t = t

    a = torch.randn(4, 5)
    b = torch.randn(3)
    obj = {"x": [a, {"y": Holder(b)}], "z": "keep"}
    extras, out = _roundtrip(obj)
    assert len(extras["tensors"]) == 2
    assert torch.equal(out["x"][0], a)
    assert torch.equal(out["x"][1]["y"].t, b)
    assert out["z"] == "keep"


def test_empty_and_scalar_tensors():
    e = torch.empty(0, 3)
    s = torch.tensor(7.5)
    extras, out = _roundtrip([e, s])
    assert out[0].shape == (0, 3)
    assert out[1].item() == 7.5


def test_noncontiguous_tensor():
    t = torch.arange(24).reshape(4, 6).t()  # transposed view
    _, out = _roundtrip(t)
    assert torch.equal(out, t)


def test_crc_mismatch_detected():
    import zlib

    t = torch.arange(16, dtype=torch.float32)
    extras, parts = tensor_codec.encode(t)
    payload = bytearray(b"".join(bytes(p) for p in parts))
    extras["tensors"][0]["crc32"] = zlib.crc32(b"something-else") & 0xFFFFFFFF
    with pytest.raises(ValueError, match="CRC"):
        tensor_codec.decode(extras, memoryview(bytes(payload)), None, None)


def test_crc_match_passes():
    import zlib

    t = torch.arange(16, dtype=torch.float32)
    extras, parts = tensor_codec.encode(t)
    raw = bytes(parts[1])
    extras["tensors"][0]["crc32"] = zlib.crc32(raw) & 0xFFFFFFFF
    payload = b"".join(bytes(p) for p in parts)
    out = tensor_codec.decode(extras, memoryview(payload), None, None)
    assert torch.equal(out, t)


def test_shared_tensor_sent_twice_is_two_entries():
    t = torch.randn(3)
    extras, out = _roundtrip([t, t])
    # Two references → placeholder memoization may dedupe via pickle memo;
    # either way both decoded slots must equal the original.
    assert torch.equal(out[0], t) and torch.equal(out[1], t)
