#!/usr/bin/env python
"""One-command API parity check against the reference's public surface.

Parses the reference's py/moolib/__init__.py (read-only mount) for its
__all__ and asserts every symbol exists in moolib_amd with the same broad
kind (class/callable). Run:  python tools/verify_parity.py
"""
import ast
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

REFERENCE_INIT = "/root/reference/py/moolib/__init__.py"


def reference_exports(path=REFERENCE_INIT):
    tree = ast.parse(open(path).read())
    for node in ast.walk(tree):
        if isinstance(node, ast.Assign):
            for t in node.targets:
                if isinstance(t, ast.Name) and t.id == "__all__":
                    return [
                        e.value
                        for e in node.value.elts
                        if isinstance(e, ast.Constant) and isinstance(e.value, str)
                    ]
    raise SystemExit("could not find __all__ in %s" % path)


def main():
    import moolib_amd

    want = [s for s in reference_exports() if not s.startswith("__")]
    missing = [s for s in want if not hasattr(moolib_amd, s)]
    print("reference exports: %d; present: %d; missing: %s" % (
        len(want), len(want) - len(missing), missing or "none"))
    # key method surfaces (reference src/moolib.cc bindings)
    surfaces = {
        "Rpc": ["set_name", "listen", "connect", "define", "define_deferred",
                "define_queue", "sync", "async_", "async_callback", "set_timeout",
                "set_transports", "undefine", "debug_info"],
        "Accumulator": ["update", "connected", "wants_state", "has_new_state",
                        "set_state", "state", "wants_gradients", "has_gradients",
                        "skip_gradients", "reduce_gradients", "zero_gradients",
                        "model_version", "set_model_version", "get_leader",
                        "is_leader", "get_gradient_stats", "set_virtual_batch_size",
                        "set_parallel_gradients"],
        "Group": ["update", "active", "members", "sync_id", "all_reduce",
                  "set_timeout", "set_sort_order"],
        "Broker": ["update", "listen", "set_name"],
        "Batcher": ["stack", "cat", "empty", "size", "get"],
        "EnvPool": ["step"],
        "Future": ["result", "wait", "done", "exception", "cancel"],
        "Queue": ["enqueue", "size"],
    }
    bad = []
    for cls, methods in surfaces.items():
        c = getattr(moolib_amd, cls, None)
        if c is None:
            bad.append(cls)
            continue
        for meth in methods:
            if not hasattr(c, meth):
                bad.append("%s.%s" % (cls, meth))
    print("method surface: %s" % ("ok" if not bad else "MISSING %s" % bad))
    if missing or bad:
        sys.exit(1)
    print("parity: OK")


if __name__ == "__main__":
    main()
