"""GPU (MI355X) tests: HIP kernel numerics vs plain fp32 torch references,
plus device paths of batcher/accumulator/impala."""
import time

import pytest
import torch

import moolib_amd

gpu = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")


@gpu
@requires_gpu
class TestVtraceKernel:
    def test_matches_cpu_reference(self):
        from moolib_amd.ops import vtrace

        torch.manual_seed(0)
        for T, B in [(20, 32), (1, 1), (80, 640), (20, 1024)]:
            log_rhos = torch.randn(T, B) * 0.4
            discounts = (torch.rand(T, B) > 0.05).float() * 0.99
            rewards = torch.randn(T, B)
            values = torch.randn(T, B)
            bootstrap = torch.randn(B)
            cpu = vtrace.from_importance_weights(log_rhos, discounts, rewards, values, bootstrap)
            dev = vtrace.from_importance_weights(
                log_rhos.cuda(), discounts.cuda(), rewards.cuda(), values.cuda(), bootstrap.cuda()
            )
            assert torch.allclose(dev.vs.cpu(), cpu.vs, atol=1e-4), (T, B)
            assert torch.allclose(dev.pg_advantages.cpu(), cpu.pg_advantages, atol=1e-4)

    def test_kernel_used_on_gpu(self):
        """The fused kernel must actually run (no silent eager fallback)."""
        from moolib_amd import _kernels

        out = _kernels.vtrace_from_log_rhos(
            torch.zeros(4, 8, device="cuda"),
            torch.full((4, 8), 0.9, device="cuda"),
            torch.ones(4, 8, device="cuda"),
            torch.zeros(4, 8, device="cuda"),
            torch.zeros(8, device="cuda"),
            1.0,
            1.0,
        )
        torch.cuda.synchronize()
        assert out[0].shape == (4, 8)


@gpu
@requires_gpu
class TestFusedLoss:
    def test_matches_eager(self):
        from moolib_amd.ops import fused_loss, losses

        torch.manual_seed(1)
        T, B, A = 20, 32, 18
        for ec, bc in [(0.0006, 0.5), (0.01, 1.0)]:
            logits = (torch.randn(T, B, A, device="cuda") * 2).requires_grad_()
            baseline = torch.randn(T, B, device="cuda").requires_grad_()
            actions = torch.randint(0, A, (T, B), device="cuda")
            pg_adv = torch.randn(T, B, device="cuda")
            vs = torch.randn(T, B, device="cuda")

            total = fused_loss.impala_total_loss(logits, baseline, actions, pg_adv, vs, ec, bc)
            total.backward()
            g_logits, g_baseline = logits.grad.clone(), baseline.grad.clone()

            logits2 = logits.detach().clone().requires_grad_()
            baseline2 = baseline.detach().clone().requires_grad_()
            pg = losses.policy_gradient_loss(logits2, actions, pg_adv)
            bl = bc * losses.baseline_loss(vs - baseline2)
            en = ec * losses.entropy_loss(logits2)
            ref = pg + bl + en
            ref.backward()

            assert torch.allclose(total.detach(), ref.detach(), atol=1e-4)
            assert torch.allclose(g_logits, logits2.grad, atol=1e-5)
            assert torch.allclose(g_baseline, baseline2.grad, atol=1e-5)


@gpu
@requires_gpu
class TestDevicePaths:
    def test_batcher_on_gpu(self):
        b = moolib_amd.Batcher(4, "cuda:0", dim=1)
        x = torch.randn(5, 8, 3)
        b.cat({"x": x})
        o1 = b.get()
        assert o1["x"].is_cuda
        assert torch.allclose(o1["x"].cpu(), x[:, :4])

    def test_accumulator_gpu_bucket(self):
        """Single-peer accumulator with CUDA params: flat bucket on device,
        RPC-tree staging path."""
        broker_rpc = moolib_amd.Rpc()
        broker_rpc.set_name("broker")
        broker = moolib_amd.Broker(broker_rpc)
        addr = broker_rpc.listen("127.0.0.1:0")[0]
        params = [torch.randn(64, 32, device="cuda").requires_grad_()]
        rpc = moolib_amd.Rpc()
        rpc.set_name("p0")
        group = moolib_amd.Group(rpc, "gputest")
        acc = moolib_amd.Accumulator("acc", params, [], group=group)
        acc.connect(addr)
        t0 = time.time()
        applied = False
        while time.time() - t0 < 30 and not applied:
            broker.update()
            acc.update()
            if acc.wants_state():
                acc.set_state({})
            if acc.connected() and acc.wants_gradients():
                params[0].grad = torch.full_like(params[0], 2.0)
                acc.reduce_gradients(1)
            if acc.has_gradients():
                assert torch.allclose(params[0].grad, torch.full_like(params[0], 2.0))
                acc.zero_gradients()
                applied = True
            time.sleep(0.005)
        assert applied

    def test_impala_gpu_short(self):
        from moolib_amd.envs import SyntheticAtariEnv
        from moolib_amd.impala import ImpalaConfig, ImpalaPeer

        broker_rpc = moolib_amd.Rpc()
        broker_rpc.set_name("broker")
        broker = moolib_amd.Broker(broker_rpc)
        addr = broker_rpc.listen("127.0.0.1:0")[0]
        cfg = ImpalaConfig(
            num_actions=6,
            actor_batch_size=16,
            num_actor_batches=2,
            num_actor_cpus=2,
            batch_size=8,
            unroll_length=5,
            virtual_batch_size=8,
            device="cuda:0",
            connect=addr,
            total_steps=1e6,
        )
        peer = ImpalaPeer(cfg, lambda: SyntheticAtariEnv(num_actions=6), broker=broker)
        t0 = time.time()
        opt_steps = 0
        while opt_steps < 3 and time.time() - t0 < 120:
            if peer.step_once() == "optimize":
                opt_steps += 1
        assert opt_steps >= 3
        assert peer.stats["env_train_steps"].result() >= 3 * 5 * 8


@gpu
@requires_gpu
class TestMaxPoolKernel:
    def test_matches_torch(self):
        import torch.nn.functional as F

        from moolib_amd.ops.pool import maxpool3x3s2

        torch.manual_seed(3)
        for shape in [(2, 16, 84, 84), (4, 32, 42, 42), (672, 32, 21, 21), (1, 8, 11, 11)]:
            x = torch.randn(*shape, device="cuda", dtype=torch.float32).to(
                memory_format=torch.channels_last
            ).requires_grad_()
            x_ref = x.detach().clone().requires_grad_()
            y = maxpool3x3s2(x)
            y_ref = F.max_pool2d(x_ref, 3, stride=2, padding=1)
            assert torch.equal(y, y_ref), shape
            g = torch.randn_like(y)
            y.backward(g)
            y_ref.backward(g)
            assert torch.allclose(x.grad, x_ref.grad, atol=1e-5), shape

    def test_bf16(self):
        import torch.nn.functional as F

        from moolib_amd.ops.pool import maxpool3x3s2

        x = torch.randn(8, 16, 84, 84, device="cuda", dtype=torch.bfloat16).to(
            memory_format=torch.channels_last
        ).requires_grad_()
        x_ref = x.detach().clone().requires_grad_()
        y = maxpool3x3s2(x)
        y_ref = F.max_pool2d(x_ref, 3, stride=2, padding=1)
        assert torch.equal(y, y_ref)
        y.sum().backward()
        y_ref.sum().backward()
        assert torch.allclose(x.grad.float(), x_ref.grad.float())


@gpu
@requires_gpu
class TestFramesKernel:
    def test_matches_eager(self):
        from moolib_amd import _kernels

        x = torch.randint(0, 256, (7, 4, 84, 84), dtype=torch.uint8, device="cuda")
        got = _kernels.frames_u8_to_bf16_nhwc(x, 1.0 / 255.0)
        want = (x.float() / 255.0).bfloat16().contiguous(memory_format=torch.channels_last)
        assert got.dtype == torch.bfloat16
        assert got.is_contiguous(memory_format=torch.channels_last)
        assert torch.allclose(got.float(), want.float(), atol=1 / 255.0)


def _ref_lstm_scan(X, notdone, h0, c0, w_hh):
    """fp32 reference of the masked LSTM recurrence (torch ops)."""
    T, B, _ = X.shape
    h, c = h0, c0
    Hs = []
    for t in range(T):
        nd = notdone[t].unsqueeze(-1)
        h = h * nd
        c = c * nd
        G = X[t] + h @ w_hh.t()
        i, f, g, o = G.chunk(4, dim=-1)
        i, f, g, o = torch.sigmoid(i), torch.sigmoid(f), torch.tanh(g), torch.sigmoid(o)
        c = f * c + i * g
        h = o * torch.tanh(c)
        Hs.append(h)
    return torch.stack(Hs), h, c


@gpu
@requires_gpu
class TestFusedLSTM:
    def test_forward_matches_reference(self):
        from moolib_amd.ops.lstm import fused_lstm_scan

        torch.manual_seed(5)
        for T, B in [(21, 32), (1, 128), (7, 40)]:
            X = (torch.randn(T, B, 1024, device="cuda") * 0.5)
            nd = (torch.rand(T, B, device="cuda") > 0.1).float()
            h0 = torch.randn(B, 256, device="cuda") * 0.3
            c0 = torch.randn(B, 256, device="cuda") * 0.3
            w = torch.randn(1024, 256, device="cuda") * 0.05
            H, hT, cT = fused_lstm_scan(
                X.bfloat16(), nd, h0.bfloat16(), c0.float(), w.bfloat16()
            )
            Hr, hr, cr = _ref_lstm_scan(X, nd, h0, c0, w)
            assert torch.allclose(H.float(), Hr, atol=3e-2), (T, B, (H.float() - Hr).abs().max())
            assert torch.allclose(cT, cr, atol=3e-2)
            assert torch.allclose(hT.float(), hr, atol=3e-2)

    def test_backward_matches_reference(self):
        from moolib_amd.ops.lstm import fused_lstm_scan

        torch.manual_seed(6)
        T, B = 12, 32
        X0 = (torch.randn(T, B, 1024, device="cuda") * 0.5)
        nd = (torch.rand(T, B, device="cuda") > 0.1).float()
        h00 = torch.randn(B, 256, device="cuda") * 0.3
        c00 = torch.randn(B, 256, device="cuda") * 0.3
        w0 = torch.randn(1024, 256, device="cuda") * 0.05

        X = X0.bfloat16().requires_grad_()
        h0 = h00.bfloat16().requires_grad_()
        c0 = c00.clone().requires_grad_()
        w = w0.bfloat16().requires_grad_()
        H, hT, cT = fused_lstm_scan(X, nd, h0, c0, w)
        gH = torch.randn_like(H.float())
        (H.float() * gH).sum().backward()

        Xr = X0.clone().requires_grad_()
        h0r = h00.clone().requires_grad_()
        c0r = c00.clone().requires_grad_()
        wr = w0.clone().requires_grad_()
        Hr, hr, cr = _ref_lstm_scan(Xr, nd, h0r, c0r, wr)
        (Hr * gH).sum().backward()

        def close(a, b, name, atol):
            d = (a.float() - b).abs().max().item()
            s = b.abs().max().item()
            assert d <= atol + 0.05 * s, f"{name}: maxdiff {d} scale {s}"

        close(X.grad, Xr.grad, "dX", 5e-2)
        close(h0.grad, h0r.grad, "dh0", 1e-1)
        close(c0.grad, c0r.grad, "dc0", 1e-1)
        close(w.grad, wr.grad, "dW", 2.0)  # sum over T*B: absolute scale is larger

    def test_model_path_matches_loop(self):
        """Full AtariNet LSTM forward: fused vs python-loop fallback."""
        import os

        from moolib_amd.models.atari import AtariNet

        torch.manual_seed(7)
        model = AtariNet(num_actions=6, use_lstm=True).to("cuda").to(torch.bfloat16)
        T, B = 5, 8
        inputs = {
            "state": torch.randint(0, 256, (T, B, 4, 84, 84), dtype=torch.uint8, device="cuda"),
            "reward": torch.randn(T, B, device="cuda"),
            "done": torch.rand(T, B, device="cuda") > 0.8,
            "prev_action": torch.randint(0, 6, (T, B), device="cuda"),
        }
        core = tuple(t.cuda().bfloat16() for t in model.initial_state(batch_size=B))
        with torch.no_grad():
            out_fused, cs_fused = model(inputs, core)
            os.environ["MOOLIB_AMD_NO_LSTM_KERNEL"] = "1"
            try:
                out_loop, cs_loop = model(inputs, core)
            finally:
                del os.environ["MOOLIB_AMD_NO_LSTM_KERNEL"]
        assert torch.allclose(
            out_fused["policy_logits"], out_loop["policy_logits"], atol=0.15
        ), (out_fused["policy_logits"] - out_loop["policy_logits"]).abs().max()
        assert torch.allclose(out_fused["baseline"], out_loop["baseline"], atol=0.15)


@gpu
@requires_gpu
class TestConv1Kernel:
    def test_matches_conv2d(self):
        from moolib_amd import _kernels

        torch.manual_seed(9)
        conv = torch.nn.Conv2d(4, 16, 3, stride=1, padding=1).cuda()
        x = torch.randint(0, 256, (6, 4, 84, 84), dtype=torch.uint8, device="cuda")
        w = conv.weight.detach().to(torch.bfloat16).permute(2, 3, 1, 0).contiguous()
        b = conv.bias.detach().to(torch.bfloat16)
        got = _kernels.conv1_u8_nhwc(x, w, b, 1.0 / 255.0)
        with torch.no_grad():
            want = conv(x.float() / 255.0)
        assert got.is_contiguous(memory_format=torch.channels_last)
        d = (got.float() - want).abs().max().item()
        assert d < 0.03, d  # bf16 weights vs fp32 reference

    def test_actor_model_path_consistent(self):
        """AtariNet no-grad forward: fused-conv1 path vs frames-kernel path."""
        import os

        from moolib_amd.models.atari import AtariNet

        torch.manual_seed(10)
        model = AtariNet(num_actions=6).cuda().to(torch.bfloat16)
        inputs = {
            "state": torch.randint(0, 256, (1, 16, 4, 84, 84), dtype=torch.uint8, device="cuda"),
            "reward": torch.randn(1, 16, device="cuda"),
            "done": torch.zeros(1, 16, dtype=torch.bool, device="cuda"),
            "prev_action": torch.zeros(1, 16, dtype=torch.int64, device="cuda"),
        }
        with torch.no_grad():
            out_fused, _ = model(inputs, tuple())
            os.environ["MOOLIB_AMD_NO_CONV1_KERNEL"] = "1"
            try:
                out_plain, _ = model(inputs, tuple())
            finally:
                del os.environ["MOOLIB_AMD_NO_CONV1_KERNEL"]
        d = (out_fused["policy_logits"] - out_plain["policy_logits"]).abs().max().item()
        assert d < 0.25, d  # bf16 accumulation-order differences through the net


@gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
class TestFusedBias:
    """bias_relu / bias_add2 vs plain fp32 torch (exact in fp32)."""

    def test_bias_relu_fwd_bwd_fp32(self):
        torch.manual_seed(0)
        from moolib_amd.ops.fused_bias import bias_relu

        x = torch.randn(6, 16, 21, 21, device="cuda").contiguous(
            memory_format=torch.channels_last
        )
        b = torch.randn(16, device="cuda")
        x1 = x.clone().requires_grad_(True)
        b1 = b.clone().requires_grad_(True)
        x2 = x.clone().requires_grad_(True)
        b2 = b.clone().requires_grad_(True)
        y1 = bias_relu(x1, b1)
        y2 = torch.relu(x2 + b2.view(1, -1, 1, 1))
        assert torch.equal(y1, y2)
        g = torch.randn_like(y1)
        y1.backward(g)
        y2.backward(g)
        assert torch.equal(x1.grad, x2.grad)
        assert torch.allclose(b1.grad, b2.grad, atol=1e-3, rtol=1e-4), (
            (b1.grad - b2.grad).abs().max()
        )

    def test_bias_add2_fwd_bwd_fp32(self):
        torch.manual_seed(1)
        from moolib_amd.ops.fused_bias import bias_add2

        mk = lambda: torch.randn(4, 32, 11, 11, device="cuda").contiguous(
            memory_format=torch.channels_last
        )
        x, s = mk(), mk()
        b1 = torch.randn(32, device="cuda")
        b2 = torch.randn(32, device="cuda")
        args1 = [t.clone().requires_grad_(True) for t in (x, b1, s, b2)]
        args2 = [t.clone().requires_grad_(True) for t in (x, b1, s, b2)]
        y1 = bias_add2(*args1)
        y2 = args2[0] + args2[1].view(1, -1, 1, 1) + args2[2] + args2[3].view(1, -1, 1, 1)
        assert torch.equal(y1, y2)
        g = torch.randn_like(y1)
        y1.backward(g)
        y2.backward(g)
        for a1, a2 in zip(args1, args2):
            assert torch.allclose(a1.grad, a2.grad, atol=1e-3, rtol=1e-4)

    def test_channel_sum_matches_torch(self):
        from moolib_amd import _kernels

        torch.manual_seed(5)
        for shape, dtype in [
            ((672, 16, 42, 42), torch.bfloat16),
            ((672, 32, 21, 21), torch.bfloat16),
            ((672, 32, 11, 11), torch.bfloat16),
            ((3, 8, 5, 7), torch.float32),
        ]:
            x = torch.randn(shape, device="cuda", dtype=dtype).contiguous(
                memory_format=torch.channels_last
            )
            got = _kernels.channel_sum_fp32(x)
            want = x.float().sum(dim=(0, 2, 3))
            assert got.dtype == torch.float32
            assert torch.allclose(got, want, atol=2e-1, rtol=1e-4), (
                shape, dtype, (got - want).abs().max())

    def test_model_fused_matches_eager_with_grads(self):
        """Full AtariNet learner-style fwd+bwd: fused-bias path vs eager
        (MOOLIB_AMD_NO_FUSED_BIAS) — outputs and conv-bias grads agree."""
        import os

        from moolib_amd.models.atari import AtariNet

        torch.manual_seed(2)
        T, B = 4, 8
        inputs = {
            "state": torch.randint(0, 255, (T, B, 4, 84, 84), dtype=torch.uint8, device="cuda"),
            "reward": torch.randn(T, B, device="cuda"),
            "prev_action": torch.randint(0, 6, (T, B), device="cuda"),
            "done": torch.zeros(T, B, dtype=torch.bool, device="cuda"),
        }

        def run():
            torch.manual_seed(3)
            model = AtariNet(num_actions=6).to("cuda").to(torch.bfloat16)
            model.float()  # fp32 for exactness; channels_last conv path
            model = model.to(memory_format=torch.channels_last)
            out, _ = model(inputs, tuple())
            loss = out["policy_logits"].square().mean() + out["baseline"].square().mean()
            loss.backward()
            gs = {
                n: p.grad.detach().clone()
                for n, p in model.named_parameters()
                if "bias" in n and p.grad is not None
            }
            return out, gs

        os.environ.pop("MOOLIB_AMD_NO_FUSED_BIAS", None)
        out_f, g_f = run()
        os.environ["MOOLIB_AMD_NO_FUSED_BIAS"] = "1"
        try:
            out_e, g_e = run()
        finally:
            del os.environ["MOOLIB_AMD_NO_FUSED_BIAS"]
        d = (out_f["policy_logits"] - out_e["policy_logits"]).abs().max().item()
        assert d < 1e-3, d
        for n in g_e:
            dg = (g_f[n] - g_e[n]).abs().max().item()
            rel = dg / (g_e[n].abs().max().item() + 1e-8)
            assert rel < 1e-2, (n, dg, rel)


@gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
class TestConv3x3Fused:
    """MFMA implicit-GEMM 3x3 conv vs fp32 F.conv2d (bf16 tolerance)."""

    def _ref(self, x, w, relu_in=False, bias_in=None, epi=0, bias1=None, res=None, bias2=None):
        import torch.nn.functional as F

        xf = x.float()
        if relu_in:
            if bias_in is not None:
                xf = xf + bias_in.float().view(1, -1, 1, 1)
            xf = torch.relu(xf)
        y = F.conv2d(xf, w.float(), None, padding=1)
        if epi == 1:
            y = y + bias1.float().view(1, -1, 1, 1)
        elif epi == 2:
            y = torch.relu(y + bias1.float().view(1, -1, 1, 1))
        elif epi == 3:
            y = y + bias1.float().view(1, -1, 1, 1) + res.float()
            if bias2 is not None:
                y = y + bias2.float().view(1, -1, 1, 1)
        return y

    @pytest.mark.parametrize("C,K", [(16, 16), (16, 32), (32, 32)])
    def test_plain_conv(self, C, K):
        from moolib_amd.ops import conv3x3 as c3

        torch.manual_seed(C * 100 + K)
        x = torch.randn(3, C, 11, 13, device="cuda", dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last
        )
        w = torch.randn(K, C, 3, 3, device="cuda") * 0.2
        y = c3.conv3x3(x, c3.pack_weight(w), K)
        ref = self._ref(x, w)
        err = (y.float() - ref).abs().max().item()
        scale = ref.abs().max().item()
        assert err / scale < 0.02, (err, scale)

    @pytest.mark.parametrize("rt", [1, 2, 4])
    def test_row_tile_variants_match(self, rt):
        """Every RT instantiation (v2 row-tiled chains) must agree with the
        fp32 reference, including the M-tail where pixBase+64*RT > M."""
        from moolib_amd.ops import conv3x3 as c3

        torch.manual_seed(40 + rt)
        C = K = 32
        # H*W*N chosen so M = 3*21*21 = 1323 is NOT a multiple of 64*rt
        x = torch.randn(3, C, 21, 21, device="cuda", dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last
        )
        w = torch.randn(K, C, 3, 3, device="cuda") * 0.2
        b1 = torch.randn(K, device="cuda") * 0.5
        y = c3.conv3x3(x, c3.pack_weight(w), K, relu_in=True, epi=c3.EPI_BIAS, bias1=b1, rt=rt)
        ref = self._ref(x, w, relu_in=True, epi=1, bias1=b1)
        err = (y.float() - ref).abs().max().item()
        scale = max(ref.abs().max().item(), 1.0)
        assert err / scale < 0.02, (rt, err, scale)

    def test_fused_block_variants(self):
        from moolib_amd.ops import conv3x3 as c3

        torch.manual_seed(7)
        C = K = 32
        x = torch.randn(2, C, 21, 21, device="cuda", dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last
        )
        w = torch.randn(K, C, 3, 3, device="cuda") * 0.2
        b_in = torch.randn(C, device="cuda") * 0.5
        b1 = torch.randn(K, device="cuda") * 0.5
        b2 = torch.randn(K, device="cuda") * 0.5
        res = torch.randn(2, K, 21, 21, device="cuda", dtype=torch.bfloat16).contiguous(
            memory_format=torch.channels_last
        )
        wp = c3.pack_weight(w)
        for kwargs in (
            dict(relu_in=True),
            dict(relu_in=True, bias_in=b_in),
            dict(epi=c3.EPI_BIAS, bias1=b1),
            dict(epi=c3.EPI_BIAS_RELU, bias1=b1),
            dict(relu_in=True, bias_in=b_in, epi=c3.EPI_BIAS_ADD, bias1=b1, res=res, bias2=b2),
        ):
            y = c3.conv3x3(x, wp, K, **kwargs)
            ref = self._ref(x, w, **kwargs)
            err = (y.float() - ref).abs().max().item()
            scale = max(ref.abs().max().item(), 1.0)
            assert err / scale < 0.02, (kwargs, err, scale)

    def test_actor_model_path_matches(self):
        """AtariNet no_grad forward: conv3x3 path (opt-in env) vs MIOpen."""
        import os

        from moolib_amd.models.atari import AtariNet

        os.environ["MOOLIB_AMD_CONV3_KERNEL"] = "1"
        torch.manual_seed(11)
        model = AtariNet(num_actions=6).to("cuda").to(torch.bfloat16)
        model = model.to(memory_format=torch.channels_last)
        inputs = {
            "state": torch.randint(0, 255, (1, 16, 4, 84, 84), dtype=torch.uint8, device="cuda"),
            "reward": torch.randn(1, 16, device="cuda"),
            "prev_action": torch.randint(0, 6, (1, 16), device="cuda"),
            "done": torch.zeros(1, 16, dtype=torch.bool, device="cuda"),
        }
        try:
            with torch.no_grad():
                out_f, _ = model(inputs, tuple())
                os.environ["MOOLIB_AMD_CONV3_KERNEL"] = "0"  # force MIOpen path
                out_e, _ = model(inputs, tuple())
        finally:
            os.environ.pop("MOOLIB_AMD_CONV3_KERNEL", None)
        d = (out_f["policy_logits"].float() - out_e["policy_logits"].float()).abs().max().item()
        assert d < 0.3, d  # bf16 accumulation-order differences through the net


@gpu
@requires_gpu
class TestBatchedCopy:
    """batched_copy must be exact vs per-pair copy_ across the slice
    layouts Batcher produces (select/narrow of contiguous buffers), odd
    sizes (1-byte path), and fall back cleanly for layouts it rejects."""

    def test_slice_layouts_exact(self):
        from moolib_amd import _kernels

        torch.manual_seed(3)
        cases = []
        # stack-style: select(dim, i) targets
        big = torch.empty(8, 16, 84, 84, 4, device="cuda", dtype=torch.uint8)
        srcs = [torch.randint(0, 255, (16, 84, 84, 4), dtype=torch.uint8, device="cuda") for _ in range(3)]
        cases += [(big.select(0, i), s) for i, s in enumerate(srcs)]
        # cat-style: narrow along dim 1 (strided rows)
        tgt = torch.empty(21, 40, 7, device="cuda")
        src = torch.randn(21, 128, 7, device="cuda")
        cases.append((tgt.narrow(1, 4, 32), src.narrow(1, 96, 32)))
        # odd rowBytes -> byte path
        t3 = torch.empty(5, 13, device="cuda", dtype=torch.uint8)
        s3 = torch.randint(0, 255, (5, 13), dtype=torch.uint8, device="cuda")
        cases.append((t3, s3))
        # dtype variety
        t4 = torch.empty(6, 11, device="cuda", dtype=torch.bfloat16)
        s4 = torch.randn(6, 11, device="cuda", dtype=torch.bfloat16)
        cases.append((t4, s4))
        refs = [s.clone() for _, s in cases]
        _kernels.batched_copy([d for d, _ in cases], [s for _, s in cases])
        torch.cuda.synchronize()
        for (d, _), r in zip(cases, refs):
            assert torch.equal(d, r)

    def test_fallback_pairs(self):
        from moolib_amd import _kernels

        # CPU pair and a transposed (non-coalescible) dst view both fall
        # back to copy_ inside the call.
        d1 = torch.empty(4, 4)
        s1 = torch.randn(4, 4)
        d2 = torch.empty(8, 6, device="cuda").t()  # strides reversed
        s2 = torch.randn(6, 8, device="cuda")
        _kernels.batched_copy([d1, d2], [s1, s2])
        torch.cuda.synchronize()
        assert torch.equal(d1, s1)
        assert torch.equal(d2, s2)

    def test_batcher_uses_fused_path_on_gpu(self):
        """End-to-end: GPU Batcher.stack/cat equivalence with the hook
        registered (registration happens at import when a GPU exists)."""
        b = moolib_amd.Batcher(4, "cuda:0", dim=0)
        xs = [
            {"f": torch.randn(16, 7, device="cuda"), "a": torch.randint(0, 5, (16,), device="cuda")}
            for _ in range(4)
        ]
        for x in xs:
            b.stack(x)
        out = b.get()
        torch.cuda.synchronize()
        for i, x in enumerate(xs):
            assert torch.equal(out["f"][i], x["f"])
            assert torch.equal(out["a"][i], x["a"])
        c = moolib_amd.Batcher(6, "cuda:0", dim=0)
        src = {"v": torch.randn(10, 3, device="cuda")}
        c.cat(src)  # 6 fill + 4 carry
        out = c.get()
        assert torch.equal(out["v"], src["v"][:6])


@gpu
@requires_gpu
class TestAsyncActionStaging:
    def test_cuda_action_roundtrip(self):
        """step() with a CUDA action must not block on the device in the
        caller and must deliver the same actions to the workers."""
        from moolib_amd.envs import SyntheticAtariEnv

        pool = moolib_amd.EnvPool(
            lambda: SyntheticAtariEnv(num_actions=6),
            num_processes=2,
            batch_size=8,
            num_batches=2,
        )
        try:
            # prime both batches (CPU path)
            for b in range(2):
                pool.step(b, torch.zeros(8, dtype=torch.int64)).result()
            for i in range(5):
                act = torch.randint(0, 6, (8,), device="cuda")
                fut = pool.step(i % 2, act)
                out = fut.result()
                assert out["state"].shape[0] == 8
        finally:
            del pool


@gpu
@requires_gpu
class TestBf16LearningParity:
    def test_gradient_direction_bf16_vs_fp32(self):
        """The shipped precision mode (bf16 weights + fp32 V-trace/loss)
        must produce gradients pointing the same way as a full-fp32 pass
        through the SAME learn pipeline (end-to-end
        bf16 evidence, not just per-kernel tolerances)."""
        import copy

        from moolib_amd.models.atari import AtariNet
        from moolib_amd.ops import vtrace
        from moolib_amd.ops import fused_loss

        torch.manual_seed(7)
        T, B, A = 10, 8, 6
        m32 = AtariNet(num_actions=A).to("cuda")
        m16 = copy.deepcopy(m32).to(torch.bfloat16)
        m16 = m16.to(memory_format=torch.channels_last)

        batch = {
            "state": torch.randint(0, 255, (T + 1, B, 4, 84, 84), dtype=torch.uint8, device="cuda"),
            "reward": torch.randn(T + 1, B, device="cuda").clamp(-1, 1),
            "prev_action": torch.randint(0, A, (T + 1, B), device="cuda"),
            "done": torch.rand(T + 1, B, device="cuda") > 0.95,
        }
        behavior_logits = torch.randn(T, B, A, device="cuda")
        actions = torch.randint(0, A, (T, B), device="cuda")

        def grads(model):
            model.zero_grad(set_to_none=True)
            out, _ = model(batch, tuple())
            bootstrap = out["baseline"][-1].float()
            logits = out["policy_logits"][:-1].float()
            baseline = out["baseline"][:-1].float()
            rewards = batch["reward"][1:].float()
            discounts = (~batch["done"][1:]).float() * 0.99
            vt = vtrace.from_logits(
                behavior_policy_logits=behavior_logits,
                target_policy_logits=logits.detach(),
                actions=actions,
                discounts=discounts,
                rewards=rewards,
                values=baseline.detach(),
                bootstrap_value=bootstrap.detach(),
            )
            loss = fused_loss.impala_total_loss(
                logits, baseline, actions, vt.pg_advantages, vt.vs, 0.0006, 0.5
            )
            loss.backward()
            return {
                n: p.grad.detach().float().clone()
                for n, p in model.named_parameters()
                if p.grad is not None
            }

        g32 = grads(m32)
        g16 = grads(m16)
        torch.cuda.synchronize()
        assert set(g32) == set(g16)
        bad = []
        for n in g32:
            a, b = g32[n].flatten(), g16[n].flatten()
            denom = a.norm() * b.norm()
            if denom < 1e-12:
                continue
            cos = float((a @ b) / denom)
            if cos < 0.9:
                bad.append((n, cos))
        assert not bad, bad


@gpu
@requires_gpu
class TestConv3x3Autograd:
    def test_forward_and_grads_match_fp32(self):
        """conv3x3_autograd (MFMA fwd + MFMA dgrad + MIOpen wrw) vs the
        fp32 autograd reference, at a learner-like shape."""
        import torch.nn.functional as F

        from moolib_amd.ops import conv3x3 as c3

        torch.manual_seed(5)
        for C, K, H in [(16, 16, 42), (32, 32, 21), (16, 32, 42)]:
            conv = torch.nn.Conv2d(C, K, 3, padding=1, bias=False).to("cuda").to(torch.bfloat16)
            x = (
                torch.randn(6, C, H, H, device="cuda", dtype=torch.bfloat16)
                .contiguous(memory_format=torch.channels_last)
                .requires_grad_()
            )
            y = c3.conv3x3_autograd(x, conv)
            g = torch.randn_like(y)
            y.backward(g)
            dx, dw = x.grad.clone(), conv.weight.grad.clone()

            xf = x.detach().float().clone().requires_grad_()
            wf = conv.weight.detach().float().clone().requires_grad_()
            yf = F.conv2d(xf, wf, None, padding=1)
            yf.backward(g.float())

            for name, got, want in [
                ("y", y.float(), yf.detach()),
                ("dx", dx.float(), xf.grad),
                ("dw", dw.float(), wf.grad),
            ]:
                err = (got - want).abs().max().item()
                scale = max(want.abs().max().item(), 1.0)
                assert err / scale < 0.03, (C, K, name, err, scale)

    def test_learner_step_with_custom_conv(self):
        """A full fwd+bwd of AtariNet in bf16 produces finite gradients for
        every parameter with the custom conv on the learner path."""
        from moolib_amd.models.atari import AtariNet

        torch.manual_seed(6)
        model = AtariNet(num_actions=6).to("cuda").to(torch.bfloat16)
        model = model.to(memory_format=torch.channels_last)
        inputs = {
            "state": torch.randint(0, 255, (5, 4, 4, 84, 84), dtype=torch.uint8, device="cuda"),
            "reward": torch.randn(5, 4, device="cuda"),
            "prev_action": torch.randint(0, 6, (5, 4), device="cuda"),
            "done": torch.zeros(5, 4, dtype=torch.bool, device="cuda"),
        }
        out, _ = model(inputs, tuple())
        loss = out["policy_logits"].float().square().mean() + out["baseline"].float().square().mean()
        loss.backward()
        for n, p in model.named_parameters():
            assert p.grad is not None, n
            assert torch.isfinite(p.grad.float()).all(), n


@gpu
@requires_gpu
class TestWgradKernel:
    def test_matches_fp32_reference(self):
        """wgrad3x3_nhwc vs aten.convolution_backward in fp32, across
        shapes with W%32 tails, halo rows, and both C/K combos."""
        from moolib_amd import _kernels

        torch.manual_seed(9)
        for N, C, H, W, K in [
            (3, 16, 42, 42, 16),
            (2, 32, 21, 21, 32),
            (2, 32, 11, 11, 32),
            (1, 16, 84, 84, 32),
            (2, 16, 5, 7, 16),
        ]:
            x = torch.randn(N, C, H, W, device="cuda", dtype=torch.bfloat16).contiguous(
                memory_format=torch.channels_last
            )
            dy = torch.randn(N, K, H, W, device="cuda", dtype=torch.bfloat16).contiguous(
                memory_format=torch.channels_last
            )
            g = _kernels.wgrad3x3_nhwc(x, dy)[: 9 * C]
            dw = g.view(3, 3, C, K).permute(3, 2, 0, 1)
            w = torch.zeros(K, C, 3, 3, device="cuda")
            ref = torch.ops.aten.convolution_backward(
                dy.float(), x.float(), w, None, [1, 1], [1, 1], [1, 1], False, [0, 0], 1,
                [False, True, False],
            )[1]
            err = (dw - ref).abs().max().item()
            scale = max(ref.abs().max().item(), 1.0)
            assert err / scale < 0.02, ((N, C, H, W, K), err, scale)


@gpu
@requires_gpu
class TestConv1Autograd:
    def test_first_layer_grads_match_fp32(self):
        """conv1_u8_autograd: fused fwd + wgrad-kernel dW + bias-sum db vs
        the fp32 reference (frames/255 -> conv2d)."""
        import torch.nn.functional as F

        from moolib_amd.ops import conv3x3 as c3

        torch.manual_seed(12)
        conv = torch.nn.Conv2d(4, 16, 3, padding=1).to("cuda").to(torch.bfloat16)
        x = torch.randint(0, 256, (6, 4, 84, 84), dtype=torch.uint8, device="cuda")
        y = c3.conv1_u8_autograd(x, conv, 1.0 / 255.0)
        g = torch.randn_like(y)
        y.backward(g)
        dw, db = conv.weight.grad.float(), conv.bias.grad.float()

        xf = (x.float() / 255.0).requires_grad_(False)
        wf = conv.weight.detach().float().clone().requires_grad_()
        bf = conv.bias.detach().float().clone().requires_grad_()
        yf = F.conv2d(xf, wf, bf, padding=1)
        yf.backward(g.float())

        for name, got, want in [("y", y.float(), yf.detach()), ("dw", dw, wf.grad), ("db", db, bf.grad)]:
            err = (got - want).abs().max().item()
            scale = max(want.abs().max().item(), 1.0)
            assert err / scale < 0.03, (name, err, scale)


@gpu
@requires_gpu
class TestBatchedRepack:
    def test_matches_torch_pack(self):
        """repack3x3_batched must reproduce pack_weight AND
        pack_weight_dgrad bit-exactly, incl. channels_last weights and the
        C=4 first conv (no dgrad pack)."""
        from moolib_amd import _kernels
        from moolib_amd.ops import conv3x3 as c3

        torch.manual_seed(21)
        specs = [(16, 16), (16, 32), (32, 32), (4, 16)]
        ws, fs, ds, refs = [], [], [], []
        for C, K in specs:
            w = torch.randn(K, C, 3, 3, device="cuda", dtype=torch.bfloat16)
            if C != 4:
                w = w.to(memory_format=torch.channels_last)
            f_ref = c3.pack_weight(w)
            d_ref = c3.pack_weight_dgrad(w) if C != 4 else None
            ws.append(w)
            fs.append(torch.empty_like(f_ref))
            ds.append(torch.empty_like(d_ref) if d_ref is not None else w.new_empty(0))
            refs.append((f_ref, d_ref))
        _kernels.repack3x3_batched(ws, fs, ds, [0] * len(ws))
        torch.cuda.synchronize()
        for (C, K), fbuf, dbuf, (f_ref, d_ref) in zip(specs, fs, ds, refs):
            assert torch.equal(fbuf, f_ref), (C, K, "fwd")
            if d_ref is not None:
                assert torch.equal(dbuf, d_ref), (C, K, "dgrad")

    def test_padded_first_conv_pack(self):
        """pack_channels=8 for a C=4 weight must equal packing the
        zero-padded weight (the conv1 C=8 path)."""
        import torch.nn.functional as F

        from moolib_amd import _kernels
        from moolib_amd.ops import conv3x3 as c3

        w = torch.randn(16, 4, 3, 3, device="cuda", dtype=torch.bfloat16)
        ref = c3.pack_weight(F.pad(w, (0, 0, 0, 0, 0, 4)))
        buf = torch.empty_like(ref)
        _kernels.repack3x3_batched([w], [buf], [w.new_empty(0)], [8])
        torch.cuda.synchronize()
        assert torch.equal(buf, ref)


@gpu
@requires_gpu
class TestConv3x3LdsVariant:
    def test_matches_flat(self):
        """The LDS band-slab kernel must agree with the flat kernel across
        shapes/epilogues (subprocess: the variant gate is read once)."""
        import subprocess
        import sys
        import os as _os

        code = """
import torch
from moolib_amd.ops import conv3x3 as c3
torch.manual_seed(31)
for C, K, H, W in [(16, 16, 42, 42), (16, 32, 42, 42), (32, 32, 21, 21), (32, 32, 11, 11), (16, 16, 5, 37)]:
    x = torch.randn(3, C, H, W, device="cuda", dtype=torch.bfloat16).contiguous(memory_format=torch.channels_last)
    w = torch.randn(K, C, 3, 3, device="cuda") * 0.2
    b_in = (torch.randn(C, device="cuda") * 0.5).bfloat16()
    b1 = (torch.randn(K, device="cuda") * 0.5).bfloat16()
    res = torch.randn(3, K, H, W, device="cuda", dtype=torch.bfloat16).contiguous(memory_format=torch.channels_last)
    wp = c3.pack_weight(w)
    for kwargs in (dict(), dict(relu_in=True, bias_in=b_in), dict(epi=c3.EPI_BIAS_RELU, bias1=b1), dict(relu_in=True, epi=c3.EPI_BIAS_ADD, bias1=b1, res=res)):
        y_lds = c3.conv3x3(x, wp, K, **kwargs)            # env: LDS variant
        y_flat = c3.conv3x3(x, wp, K, rt=1, **kwargs)     # rt>0 forces... no:
        # rt>0 uses the override path which skips the LDS gate only when
        # rtOverride<0; use it to force flat via rt=-1? conv3x3 wrapper
        # passes rt straight through; use rt=-1
        y_flat = c3.conv3x3(x, wp, K, rt=-1, **kwargs)
        assert torch.equal(y_lds.float().cpu(), y_lds.float().cpu())
        d = (y_lds.float() - y_flat.float()).abs().max().item()
        assert d == 0.0, ((C, K, H, W), kwargs.keys(), d)
print("OK")
"""
        repo = _os.path.dirname(_os.path.dirname(_os.path.abspath(__file__)))
        env = dict(_os.environ, MOOLIB_AMD_CONV3_LDS="1")
        env["PYTHONPATH"] = repo + _os.pathsep + env.get("PYTHONPATH", "")
        r = subprocess.run([sys.executable, "-c", code], capture_output=True, text=True,
                           timeout=240, env=env, cwd=repo)
        assert r.returncode == 0 and "OK" in r.stdout, r.stderr[-2000:]


@gpu
@requires_gpu
class TestFramesPad8:
    def test_matches_generic(self):
        from moolib_amd import _kernels

        x = torch.randint(0, 256, (5, 4, 84, 84), dtype=torch.uint8, device="cuda")
        got = _kernels.frames_u8_to_bf16_nhwc(x, 1.0 / 255.0, 8)
        want4 = (x.float() / 255.0).bfloat16()
        assert got.shape == (5, 8, 84, 84)
        assert torch.equal(got[:, :4].float().cpu(), want4.float().cpu())
        assert got[:, 4:].abs().sum().item() == 0.0
