"""Run-directory utilities: tsv logging, symlinks, metadata.

Capability parity with the reference's examples/common/record.py.
"""
import csv
import json
import os
import time

_writers = {}


def log_to_file(path="logs.tsv", **fields):
    """Incrementally append rows to a tsv file (header on first write)."""
    state = _writers.get(path)
    if state is None:
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        writeheader = not os.path.exists(path)
        f = open(path, "a", buffering=1)
        writer = csv.DictWriter(f, list(fields.keys()), delimiter="\t", extrasaction="ignore", restval="")
        if writeheader:
            writer.writeheader()
        _writers[path] = state = (f, writer)
    state[1].writerow(fields)


def symlink_path(target, link_name):
    """Create/replace a symlink; returns True on success."""
    try:
        os.makedirs(os.path.dirname(link_name) or ".", exist_ok=True)
        tmp = link_name + ".tmp"
        if os.path.islink(tmp) or os.path.exists(tmp):
            os.remove(tmp)
        os.symlink(target, tmp)
        os.replace(tmp, link_name)
        return True
    except OSError:
        return False


def write_metadata(localdir, cwd=None, **kwargs):
    os.makedirs(localdir, exist_ok=True)
    meta = {
        "time": time.time(),
        "cwd": cwd or os.getcwd(),
        "env": {k: v for k, v in os.environ.items() if k.startswith(("MASTER_", "RANK", "WORLD"))},
        **kwargs,
    }
    with open(os.path.join(localdir, "metadata.json"), "w") as f:
        json.dump(meta, f, indent=2, default=str)
