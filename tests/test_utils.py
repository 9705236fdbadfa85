"""Unit tests for moolib_amd.utils (stats, record, nest, batch_size_finder).

Mirrors the reference's examples/common helpers (stat math, delta-based
global aggregation, tsv logging)."""
import math
import os

import torch

from moolib_amd.utils import nest
from moolib_amd.utils.record import log_to_file, symlink_path, write_metadata
from moolib_amd.utils.stats import RunningMeanStd, StatMean, StatSum


class TestStats:
    def test_statmean_math(self):
        s = StatMean()
        assert s.result() is None
        s += 2.0
        s += 4.0
        assert s.result() == 3.0
        d = s - StatMean(2.0, 1)
        assert d.result() == 4.0 and d.n == 1
        s.reset()
        assert s.result() is None

    def test_statsum_math(self):
        s = StatSum()
        s += 5
        s += StatSum(2.0)
        assert s.result() == 7.0
        s.reset()  # sums never reset (reference semantics)
        assert s.result() == 7.0
        assert (s - StatSum(3.0)).result() == 4.0

    def test_running_mean_std(self):
        torch.manual_seed(0)
        rms = RunningMeanStd(shape=(3,))
        data = torch.randn(1000, 3) * 2.5 + 1.0
        for i in range(0, 1000, 100):
            rms.update(data[i : i + 100])
        assert torch.allclose(rms.mean.float(), data.mean(0), atol=0.05)
        assert torch.allclose(rms.var.sqrt().float(), data.std(0), atol=0.05)


class TestNest:
    def test_map_flatten_zip(self):
        x = {"a": [1, 2], "b": {"c": 3}}
        y = nest.map(lambda v: v * 10, x)
        assert y == {"a": [10, 20], "b": {"c": 30}}
        assert list(nest.flatten(x)) == [1, 2, 3]

    def test_map_many(self):
        a = {"x": 1, "y": (2, 3)}
        b = {"x": 10, "y": (20, 30)}
        s = nest.map_many(lambda leaves: leaves[0] + leaves[1], a, b)
        assert s == {"x": 11, "y": (22, 33)}

    def test_preserves_types(self):
        x = (1, [2], {"k": (3,)})
        y = nest.map(lambda v: v, x)
        assert isinstance(y, tuple) and isinstance(y[1], list)
        assert isinstance(y[2]["k"], tuple)


class TestRecord:
    def test_tsv_roundtrip(self, tmp_path):
        p = str(tmp_path / "logs.tsv")
        log_to_file(p, step=1, loss=0.5)
        log_to_file(p, step=2, loss=0.25)
        log_to_file(p, step=3, loss=None, extra="ignored-not-in-header")
        lines = open(p).read().strip().split("\n")
        assert lines[0].split("\t") == ["step", "loss"]
        assert lines[1].split("\t") == ["1", "0.5"]
        assert len(lines) == 4

    def test_symlink_replace(self, tmp_path):
        t1 = tmp_path / "a.txt"
        t2 = tmp_path / "b.txt"
        t1.write_text("1")
        t2.write_text("2")
        link = str(tmp_path / "latest")
        assert symlink_path(str(t1), link)
        assert os.readlink(link) == str(t1)
        assert symlink_path(str(t2), link)  # atomic replace
        assert os.readlink(link) == str(t2)

    def test_write_metadata(self, tmp_path):
        import json

        d = str(tmp_path / "run")
        write_metadata(d, run="x", lr=1e-3)
        m = json.load(open(os.path.join(d, "metadata.json")))
        assert m["run"] == "x" and "time" in m and "cwd" in m


class TestBatchSizeFinder:
    def test_latency_model_prefers_throughput(self):
        import time as _t

        from moolib_amd.utils.batch_size_finder import find

        # Synthetic workload: fixed 1ms overhead + 2us per item. Bigger
        # batches amortize overhead; the finder should pick a large one.
        def fn(batch):
            _t.sleep(0.001 + len(batch) * 2e-6)

        best, results = find(fn, make_batch=lambda bs: list(range(bs)), max_batch_size=256)
        assert best >= 32, (best, results)
