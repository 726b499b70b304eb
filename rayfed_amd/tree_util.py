"""Minimal pytree: flatten/unflatten arbitrary nested containers.

Used to locate :class:`~rayfed_amd.fed_object.FedObject` instances nested in
task arguments and to rebuild the container with resolved values.

Parity: /root/reference/fed/tree_util.py:61-231 (same capability — registry of
container nodes, ``tree_flatten``/``tree_unflatten`` round-trip).  This is a
fresh, smaller implementation: a single recursive walk with a type registry,
no TreeSpec string format.
"""
from __future__ import annotations

from collections import OrderedDict
from typing import Any, Callable, Dict, List, NamedTuple, Tuple, Type

# A node "spec" is (type, context, child_specs) where leaves use type None.
FlattenFn = Callable[[Any], Tuple[List[Any], Any]]
UnflattenFn = Callable[[List[Any], Any], Any]


class _NodeDef(NamedTuple):
    flatten: FlattenFn
    unflatten: UnflattenFn


_NODE_REGISTRY: Dict[Type, _NodeDef] = {}


def register_pytree_node(ty: Type, flatten: FlattenFn, unflatten: UnflattenFn) -> None:
    _NODE_REGISTRY[ty] = _NodeDef(flatten, unflatten)


def _is_namedtuple(obj: Any) -> bool:
    return isinstance(obj, tuple) and hasattr(obj, "_fields")


register_pytree_node(
    list, lambda x: (list(x), None), lambda ch, _ctx: list(ch)
)
register_pytree_node(
    tuple, lambda x: (list(x), None), lambda ch, _ctx: tuple(ch)
)
register_pytree_node(
    dict,
    lambda x: ([x[k] for k in x.keys()], list(x.keys())),
    lambda ch, keys: dict(zip(keys, ch)),
)
register_pytree_node(
    OrderedDict,
    lambda x: ([x[k] for k in x.keys()], list(x.keys())),
    lambda ch, keys: OrderedDict(zip(keys, ch)),
)


class TreeSpec:
    """Structure descriptor produced by :func:`tree_flatten`."""

    __slots__ = ("type", "context", "children", "num_leaves")

    def __init__(self, ty, context, children: List["TreeSpec"]):
        self.type = ty
        self.context = context
        self.children = children
        self.num_leaves = (
            1 if ty is None else sum(c.num_leaves for c in children)
        )

    def is_leaf(self) -> bool:
        return self.type is None

    def __eq__(self, other) -> bool:
        return (
            isinstance(other, TreeSpec)
            and self.type == other.type
            and self.context == other.context
            and self.children == other.children
        )

    def __repr__(self) -> str:  # pragma: no cover - debugging aid
        if self.is_leaf():
            return "*"
        name = getattr(self.type, "__name__", str(self.type))
        return f"{name}({', '.join(map(repr, self.children))})"


_LEAF = TreeSpec(None, None, [])


def _lookup(obj: Any):
    ty = type(obj)
    node = _NODE_REGISTRY.get(ty)
    if node is not None:
        # namedtuples subclass tuple but must be rebuilt via their own ctor.
        if ty is not tuple and _is_namedtuple(obj):
            return None
        return node
    if _is_namedtuple(obj):
        return _NodeDef(
            lambda x: (list(x), type(x)),
            lambda ch, ctor: ctor(*ch),
        )
    if ty is tuple:
        return _NODE_REGISTRY[tuple]
    return None


def tree_flatten(tree: Any) -> Tuple[List[Any], TreeSpec]:
    """Flatten ``tree`` into (leaves, spec); unflatten restores it exactly."""
    node = _lookup(tree)
    if node is None and not _is_namedtuple(tree):
        return [tree], _LEAF
    if node is None:  # namedtuple
        node = _NodeDef(lambda x: (list(x), type(x)), lambda ch, c: c(*ch))
    children, context = node.flatten(tree)
    leaves: List[Any] = []
    child_specs: List[TreeSpec] = []
    for child in children:
        sub_leaves, sub_spec = tree_flatten(child)
        leaves.extend(sub_leaves)
        child_specs.append(sub_spec)
    return leaves, TreeSpec(type(tree), context, child_specs)


def tree_unflatten(leaves: List[Any], spec: TreeSpec) -> Any:
    if not isinstance(spec, TreeSpec):
        raise TypeError(f"tree_unflatten expects a TreeSpec, got {type(spec)}")
    if len(leaves) != spec.num_leaves:
        raise ValueError(
            f"tree_unflatten: {len(leaves)} leaves for a spec of "
            f"{spec.num_leaves}"
        )
    return _unflatten(iter(leaves), spec)


def _unflatten(it, spec: TreeSpec) -> Any:
    if spec.is_leaf():
        return next(it)
    children = [_unflatten(it, c) for c in spec.children]
    if spec.type is tuple:
        return tuple(children)
    if _node_is_namedtuple(spec):
        return spec.context(*children)
    node = _NODE_REGISTRY.get(spec.type)
    if node is None:
        raise TypeError(f"unregistered pytree node type {spec.type}")
    return node.unflatten(children, spec.context)


def _node_is_namedtuple(spec: TreeSpec) -> bool:
    return (
        isinstance(spec.context, type)
        and issubclass(spec.type, tuple)
        and hasattr(spec.type, "_fields")
    )


def tree_map(fn: Callable[[Any], Any], tree: Any) -> Any:
    leaves, spec = tree_flatten(tree)
    return tree_unflatten([fn(x) for x in leaves], spec)
