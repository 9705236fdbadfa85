"""Nested-structure helpers (map/flatten/zip over dict/list/tuple trees).

Capability parity with the reference's examples/common/nest.py; used both by
the library (batching) and by example code. Implementation is our own:
multi-nest operations flatten every nest to a parallel leaf list and rebuild
along the first nest's structure, rather than threading iterators through a
recursive map.
"""

__all__ = ["map", "map_many", "flatten", "zip"]

_builtin_map = map
_builtin_zip = zip


def map(f, n):
    """Apply f to every leaf of nest n, preserving structure."""
    t = type(n)
    if t is dict:
        return {k: map(f, v) for k, v in n.items()}
    if t is list:
        return [map(f, x) for x in n]
    if t is tuple:
        return tuple(map(f, x) for x in n)
    if isinstance(n, (list, tuple)):  # namedtuples / subclasses
        return t(map(f, x) for x in n)
    if isinstance(n, dict):
        return t((k, map(f, v)) for k, v in n.items())
    return f(n)


def flatten(n):
    """Yield every leaf of nest n in deterministic order."""
    if isinstance(n, (list, tuple)):
        for x in n:
            yield from flatten(x)
    elif isinstance(n, dict):
        for k in n:
            yield from flatten(n[k])
    else:
        yield n


def _rebuild(structure, leaves):
    """Shape the flat iterator `leaves` like `structure` (inverse of flatten)."""
    return map(lambda _: next(leaves), structure)


def _leaf_groups(nests):
    """Transpose several same-shaped nests into per-position leaf lists."""
    columns = [list(flatten(n)) for n in nests]
    return [list(group) for group in _builtin_zip(*columns)]


def zip(*nests):
    """Zip leaves of several same-shaped nests into lists, shaped like nests[0]."""
    return _rebuild(nests[0], iter(_leaf_groups(nests)))


def map_many(f, *nests):
    """Like zip, but applies f to each leaf-list."""
    return _rebuild(nests[0], iter([f(g) for g in _leaf_groups(nests)]))
