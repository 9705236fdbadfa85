"""GraphedCall behavior that is CPU-testable: eager warmup phase and the
permanent eager fallback when capture fails (on CPU, CUDAGraph capture
always fails -> exercised for real)."""
import torch

from moolib_amd.parallel.graphs import GraphedCall


def test_warmup_runs_eager_then_falls_back():
    calls = []

    def fn(inputs):
        calls.append(1)
        return {"y": inputs["x"] * 2}

    g = GraphedCall(fn, warmup=2, name="t")
    x = {"x": torch.ones(3)}
    for i in range(5):
        out = g(x)
        assert torch.equal(out["y"], torch.full((3,), 2.0))
    # 2 warmups eager, then capture fails on CPU -> permanent eager
    assert len(calls) == 5
    assert g.failed and g.graph is None


def test_eager_results_track_inputs():
    g = GraphedCall(lambda i: {"y": i["x"] + 1}, warmup=1, name="t2")
    assert torch.equal(g({"x": torch.zeros(2)})["y"], torch.ones(2))
    assert torch.equal(g({"x": torch.ones(2)})["y"], torch.full((2,), 2.0))
