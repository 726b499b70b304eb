"""Pytree round-trips (coverage parity: reference without_ray_tests/test_tree_utils.py)."""
from collections import OrderedDict, namedtuple

import pytest

from rayfed_amd.tree_util import tree_flatten, tree_map, tree_unflatten

Point = namedtuple("Point", ["x", "y"])


@pytest.mark.parametrize(
    "tree",
    [
        1,
        "leaf",
        None,
        [1, 2, 3],
        (1, 2),
        {"a": 1, "b": 2},
        OrderedDict([("z", 1), ("a", 2)]),
        Point(1, 2),
        {"a": [1, (2, {"b": 3})], "c": Point(4, [5, 6])},
        [],
        {},
        [[], {}, ()],
    ],
)
def test_roundtrip(tree):
    leaves, spec = tree_flatten(tree)
    rebuilt = tree_unflatten(leaves, spec)
    assert rebuilt == tree
    assert type(rebuilt) is type(tree)


def test_leaf_count_and_order():
    tree = {"b": [1, 2], "a": (3, {"x": 4})}
    leaves, spec = tree_flatten(tree)
    assert leaves == [1, 2, 3, 4]  # insertion order of dict keys
    assert spec.num_leaves == 4


def test_ordered_dict_key_order_preserved():
    od = OrderedDict([("z", 1), ("a", 2)])
    leaves, spec = tree_flatten(od)
    out = tree_unflatten(leaves, spec)
    assert list(out.keys()) == ["z", "a"]


def test_unflatten_wrong_leaf_count():
    _, spec = tree_flatten([1, 2, 3])
    with pytest.raises(ValueError):
        tree_unflatten([1, 2], spec)


def test_tree_map():
    assert tree_map(lambda x: x * 2, {"a": [1, 2]}) == {"a": [2, 4]}


def test_namedtuple_nested():
    t = Point([1, 2], {"k": Point(3, 4)})
    leaves, spec = tree_flatten(t)
    assert leaves == [1, 2, 3, 4]
    assert tree_unflatten(leaves, spec) == t
