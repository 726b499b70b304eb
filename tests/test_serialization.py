"""Restricted unpickling whitelist (parity: reference
serializations_tests/test_unpickle_with_whitelist.py) + frame codec units."""
import pickle

import numpy as np
import pytest

from rayfed_amd._private import serialization
from rayfed_amd.proxy.grpc import frames


class Unlisted:
    pass


def test_whitelisted_numpy_roundtrip():
    allowed = {"numpy": "*"}
    data = serialization.dumps(np.array([1, 2, 3]))
    out = serialization.loads(data, allowed)
    assert (out == np.array([1, 2, 3])).all()


def test_unlisted_class_rejected():
    allowed = {"numpy": "*"}
    data = serialization.dumps(Unlisted())
    with pytest.raises(pickle.UnpicklingError):
        serialization.loads(data, allowed)


def test_exact_class_name_whitelist():
    allowed = {"numpy": ["dtype"]}
    # dtype alone is allowed...
    out = serialization.loads(serialization.dumps(np.dtype("int32")), allowed)
    assert out == np.dtype("int32")
    # ...but an ndarray (requires numpy reconstruct globals) is not.
    with pytest.raises(pickle.UnpicklingError):
        serialization.loads(serialization.dumps(np.zeros(3)), allowed)


def test_no_whitelist_allows_everything():
    obj = Unlisted()
    out = serialization.loads(serialization.dumps(obj))
    assert isinstance(out, Unlisted)


def test_fed_remote_error_implicitly_allowed():
    from rayfed_amd.exceptions import FedRemoteError

    data = serialization.dumps(FedRemoteError("alice", None))
    out = serialization.loads(data, {"numpy": "*"})
    assert isinstance(out, FedRemoteError)
    assert out.src_party == "alice"


# -- wire frames ---------------------------------------------------------------
def test_frame_roundtrip():
    header = {"job": "j", "up": "1#0", "down": "2"}
    payload = b"x" * 1000
    raw = frames.encode_frame(frames.KIND_PICKLE, header, payload)
    kind, hdr, view = frames.decode_frame(raw)
    assert kind == frames.KIND_PICKLE
    assert hdr == header
    assert bytes(view) == payload


def test_frame_empty_payload_and_bad_magic():
    raw = frames.encode_frame(frames.KIND_ERROR, {"job": "j", "up": "p", "down": "p"})
    kind, hdr, view = frames.decode_frame(raw)
    assert kind == frames.KIND_ERROR and len(view) == 0
    with pytest.raises(ValueError):
        frames.decode_frame(b"XXXX" + raw[4:])
    with pytest.raises(ValueError):
        frames.decode_frame(b"RF")


def test_response_roundtrip():
    raw = frames.encode_response(417, "mismatch")
    out = frames.decode_response(raw)
    assert out == {"code": 417, "result": "mismatch"}
