"""Minimal pytree utilities for nested dict/list/tuple containers of tensors.

The reference (HoagyC/sparse_coding) relies on ``optree`` for stacking model
parameter trees (reference ``autoencoders/ensemble.py:50-65``).  optree is not
part of this environment and the reference only ever uses flat/one-level dict
trees, so a small deterministic implementation is both sufficient and cheaper
to import.  Keys are traversed in sorted order so that flatten order is stable
across processes (important for the RCCL gradient buckets in
``sparse_coding_amd.parallel``).
"""

from __future__ import annotations

from typing import Any, Callable, List, Tuple

__all__ = [
    "tree_flatten",
    "tree_unflatten",
    "tree_map",
    "tree_map_",
    "tree_leaves",
]

_LEAF = "leaf"


def _is_container(x: Any) -> bool:
    return isinstance(x, (dict, list, tuple))


def tree_flatten(tree: Any) -> Tuple[List[Any], Any]:
    """Flatten a nested dict/list/tuple into (leaves, treespec).

    Dict keys are visited in sorted order for determinism.
    """
    leaves: List[Any] = []

    def build(node: Any) -> Any:
        if isinstance(node, dict):
            keys = sorted(node.keys())
            return ("dict", keys, [build(node[k]) for k in keys])
        if isinstance(node, (list, tuple)):
            tag = "list" if isinstance(node, list) else "tuple"
            return (tag, None, [build(v) for v in node])
        leaves.append(node)
        return (_LEAF, None, None)

    spec = build(tree)
    return leaves, spec


def tree_unflatten(spec: Any, leaves: List[Any]) -> Any:
    it = iter(leaves)

    def build(node: Any) -> Any:
        tag, keys, children = node
        if tag == _LEAF:
            return next(it)
        if tag == "dict":
            return {k: build(c) for k, c in zip(keys, children)}
        vals = [build(c) for c in children]
        return vals if tag == "list" else tuple(vals)

    out = build(spec)
    rest = list(it)
    if rest:
        raise ValueError(f"tree_unflatten: {len(rest)} unused leaves")
    return out


def tree_leaves(tree: Any) -> List[Any]:
    return tree_flatten(tree)[0]


def tree_map(fn: Callable, tree: Any, *rest: Any) -> Any:
    leaves, spec = tree_flatten(tree)
    if rest:
        rest_leaves = [tree_flatten(r)[0] for r in rest]
        mapped = [fn(*args) for args in zip(leaves, *rest_leaves)]
    else:
        mapped = [fn(leaf) for leaf in leaves]
    return tree_unflatten(spec, mapped)


def tree_map_(fn: Callable, tree: Any) -> Any:
    """In-place map (for .share_memory_() style calls); returns the tree."""
    for leaf in tree_leaves(tree):
        fn(leaf)
    return tree
