"""Nested-structure helpers (map/flatten/zip over dict/list/tuple trees).

Capability parity with the reference's examples/common/nest.py; used both by
the library (batching) and by example code.
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


def zip(*nests):
    """Zip leaves of several same-shaped nests into lists, shaped like nests[0]."""
    first, *rest = nests
    iters = [flatten(n) for n in rest]

    def g(leaf):
        return [leaf] + [next(i) for i in iters]

    return map(g, first)


def map_many(f, *nests):
    """Like zip, but applies f to each leaf-list."""
    first, *rest = nests
    iters = [flatten(n) for n in rest]

    def g(leaf):
        return f([leaf] + [next(i) for i in iters])

    return map(g, first)
