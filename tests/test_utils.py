"""Address validation + misc helpers (parity: reference without_ray_tests/test_utils.py)."""
import pytest

from rayfed_amd.utils import dict2tuple, validate_address, validate_addresses


@pytest.mark.parametrize(
    "address",
    [
        "127.0.0.1:8080",
        "localhost:8080",
        "my-host.example.com:443",
        "http://example.com",
        "https://example.com:9999",
        "local",
    ],
)
def test_valid_addresses(address):
    validate_address(address)


@pytest.mark.parametrize(
    "address",
    [
        "127.0.0.1",          # no port
        "127.0.0.1:0",        # port 0
        "127.0.0.1:99999",    # port out of range
        ":8080:extra",
        "ftp//no",
        12345,
        "",
        "-badhost-:80",
    ],
)
def test_invalid_addresses(address):
    with pytest.raises(ValueError):
        validate_address(address)


def test_validate_addresses_dict():
    validate_addresses({"alice": "127.0.0.1:1", "bob": "h:2"})
    with pytest.raises(ValueError):
        validate_addresses({})
    with pytest.raises(ValueError):
        validate_addresses({"alice": "nope"})


def test_dict2tuple():
    assert dict2tuple(None) == []
    assert sorted(dict2tuple({"a": 1, "b": 2})) == [("a", 1), ("b", 2)]
    assert dict2tuple([("x", 3)]) == [("x", 3)]


def test_start_command():
    from rayfed_amd.utils import start_command

    assert start_command("echo hello").strip() == "hello"
    import pytest as _pytest

    with _pytest.raises(RuntimeError):
        start_command("echo oops 1>&2")


def test_materialize_resolves_nested_refs():
    from rayfed_amd.runtime.object_ref import ObjectRef
    from rayfed_amd.utils import materialize

    tree = {"a": [ObjectRef.from_value(1), 2], "b": (ObjectRef.from_value(3),)}
    assert materialize(tree) == {"a": [1, 2], "b": (3,)}


def test_is_cython_false_for_plain_function():
    from rayfed_amd.utils import is_cython

    assert is_cython(lambda: None) is False


def test_setup_logger_injects_party_fields(capsys):
    import logging

    from rayfed_amd.utils import setup_logger

    setup_logger(logging_level="info", party="alice", job_name="jobx")
    logging.getLogger("rayfed_amd.test").info("hello-log")
    err = capsys.readouterr().err
    assert "[alice]" in err and "[jobx]" in err and "hello-log" in err
