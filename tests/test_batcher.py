"""Batcher tests: randomized equivalence against torch.stack/cat.

Mirrors the reference's test/unit/test_batcher.py strategy (randomized
stack/cat equivalence including cat overflow carry).
"""
import random

import torch

import moolib_amd


class TestBatcherStack:
    def test_stack_dim0(self):
        b = moolib_amd.Batcher(5)
        items = [torch.randn(3, 4) for _ in range(5)]
        for t in items:
            assert b.empty()
            b.stack({"x": t, "meta": "m"})
        assert not b.empty()
        out = b.get()
        assert torch.equal(out["x"], torch.stack(items))
        assert out["meta"] == "m"
        assert b.empty()

    def test_stack_dim1(self):
        b = moolib_amd.Batcher(3, dim=1)
        items = [torch.randn(2, 4) for _ in range(3)]
        for t in items:
            b.stack([t])
        out = b.get()
        assert torch.equal(out[0], torch.stack(items, dim=1))

    def test_stack_multiple_batches(self):
        b = moolib_amd.Batcher(2)
        items = [torch.randn(3) for _ in range(6)]
        for t in items:
            b.stack(t)
        assert b.size() == 3
        for i in range(3):
            out = b.get()
            assert torch.equal(out, torch.stack(items[2 * i : 2 * i + 2]))

    def test_nested_structures(self):
        b = moolib_amd.Batcher(2)
        n1 = {"a": torch.ones(2), "b": [torch.zeros(1), (torch.full((2, 2), 3.0),)]}
        n2 = {"a": torch.ones(2) * 2, "b": [torch.ones(1), (torch.full((2, 2), 4.0),)]}
        b.stack(n1)
        b.stack(n2)
        out = b.get()
        assert torch.equal(out["a"], torch.stack([n1["a"], n2["a"]]))
        assert torch.equal(out["b"][0], torch.stack([n1["b"][0], n2["b"][0]]))
        assert torch.equal(out["b"][1][0], torch.stack([n1["b"][1][0], n2["b"][1][0]]))


class TestBatcherCat:
    def test_cat_exact(self):
        b = moolib_amd.Batcher(6, dim=0)
        x1, x2 = torch.randn(2, 5), torch.randn(4, 5)
        b.cat({"x": x1})
        assert b.empty()
        b.cat({"x": x2})
        out = b.get()
        assert torch.equal(out["x"], torch.cat([x1, x2]))

    def test_cat_overflow_carry(self):
        b = moolib_amd.Batcher(4, dim=0)
        x = torch.randn(10, 3)
        b.cat(x)  # 10 -> two full batches of 4, carry 2
        assert b.size() == 2
        assert torch.equal(b.get(), x[0:4])
        assert torch.equal(b.get(), x[4:8])
        assert b.empty()
        y = torch.randn(2, 3)
        b.cat(y)
        out = b.get()
        assert torch.equal(out, torch.cat([x[8:10], y]))

    def test_cat_dim1(self):
        # The learn-batcher shape: [T, B, ...] catted along dim 1.
        b = moolib_amd.Batcher(4, dim=1)
        x = torch.randn(5, 8, 3)  # batch of 8 -> 2 batches of 4
        b.cat({"obs": x, "tag": 7})
        assert b.size() == 2
        o1 = b.get()
        o2 = b.get()
        assert torch.equal(o1["obs"], x[:, 0:4])
        assert torch.equal(o2["obs"], x[:, 4:8])
        assert o1["tag"] == 7

    def test_randomized_equivalence(self):
        rng = random.Random(7)
        for trial in range(10):
            size = rng.randint(1, 7)
            dim = rng.randint(0, 1)
            b = moolib_amd.Batcher(size, dim=dim)
            chunks = []
            total = 0
            for _ in range(rng.randint(1, 8)):
                k = rng.randint(1, 9)
                shape = [4, 5]
                shape.insert(dim, k) if False else None
                t = torch.randn(*([k, 4] if dim == 0 else [4, k]))
                chunks.append(t)
                total += k
                b.cat(t)
            want = torch.cat(chunks, dim=dim)
            n_full = total // size
            assert b.size() == n_full
            for i in range(n_full):
                got = b.get()
                want_i = want.narrow(dim, i * size, size)
                assert torch.equal(got, want_i), f"trial {trial} batch {i}"


class TestBatcherAwait:
    def test_await(self):
        import asyncio

        b = moolib_amd.Batcher(2)

        async def run():
            b.stack(torch.ones(3))
            b.stack(torch.zeros(3))
            return await b

        out = asyncio.run(run())
        assert out.shape == (2, 3)
