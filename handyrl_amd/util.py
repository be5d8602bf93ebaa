"""Nested-structure helpers and small numerics used across the framework.

Semantics match the reference utilities (reference handyrl/util.py:7-63):
recursive map over arbitrarily nested list/tuple/dict structures, an
axis-rotation for nested containers, and a numerically-stable softmax.
"""

import numpy as np


def map_r(x, fn=None):
    """Recursively apply ``fn`` to every leaf of a nested structure."""
    if isinstance(x, (list, tuple, set)):
        return type(x)(map_r(v, fn) for v in x)
    if isinstance(x, dict):
        return type(x)((k, map_r(v, fn)) for k, v in x.items())
    return fn(x) if fn is not None else None


def bimap_r(x, y, fn=None):
    """Recursively apply ``fn`` over two parallel nested structures.

    The first argument drives the recursion (so ``y`` must be at least as
    deep as ``x`` along the visited path).
    """
    if isinstance(x, (list, tuple)):
        return type(x)(bimap_r(v, y[i], fn) for i, v in enumerate(x))
    if isinstance(x, dict):
        return type(x)((k, bimap_r(v, y[k], fn)) for k, v in x.items())
    return fn(x, y) if fn is not None else None


def trimap_r(x, y, z, fn=None):
    """Three-structure variant of :func:`bimap_r`."""
    if isinstance(x, (list, tuple)):
        return type(x)(trimap_r(v, y[i], z[i], fn) for i, v in enumerate(x))
    if isinstance(x, dict):
        return type(x)((k, trimap_r(v, y[k], z[k], fn)) for k, v in x.items())
    return fn(x, y, z) if fn is not None else None


def rotate(x, max_depth=1024):
    """Swap the outermost two container levels of a nested structure.

    ``rotate([{k: v}])`` -> ``{k: [v]}`` and so on, recursing into the
    result, mirroring reference util.py:32-58.  Used by the batch maker to
    turn per-timestep-per-player nests into per-leaf arrays.
    """
    if max_depth == 0:
        return x
    if isinstance(x, (list, tuple)):
        if len(x) == 0:
            return x
        head = x[0]
        if isinstance(head, (list, tuple)):
            return type(head)(
                rotate(type(x)(inner[i] for inner in x), max_depth - 1)
                for i in range(len(head))
            )
        if isinstance(head, dict):
            return type(head)(
                (k, rotate(type(x)(inner[k] for inner in x), max_depth - 1))
                for k in head
            )
    elif isinstance(x, dict):
        if len(x) == 0:
            return x
        head = next(iter(x.values()))
        if isinstance(head, (list, tuple)):
            return type(head)(
                rotate(type(x)((k, inner[i]) for k, inner in x.items()), max_depth - 1)
                for i in range(len(head))
            )
        if isinstance(head, dict):
            return type(head)(
                (k2, rotate(type(x)((k1, inner[k2]) for k1, inner in x.items()), max_depth - 1))
                for k2 in head
            )
    return x


def softmax(x):
    """Numerically-stable softmax along the last axis (numpy)."""
    x = np.asarray(x)
    e = np.exp(x - np.max(x, axis=-1, keepdims=True))
    return e / e.sum(axis=-1, keepdims=True)
