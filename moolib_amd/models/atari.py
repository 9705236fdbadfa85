"""IMPALA deep ResNet for Atari (Espeholt et al. 2018, "deep" variant).

Architecture parity with the reference model (examples/atari/models.py:9-153)
— this exact config is what BASELINE.json's headline benchmark names:
3 blocks of (3x3 conv -> 3x3/2 maxpool -> 2 residual blocks) with 16/32/32
channels, FC 3872->256, one-hot prev action + clipped reward appended,
optional LSTM(256), policy + baseline heads, multinomial sampling inside
forward.

MI355X notes: forward runs under bf16 autocast from the IMPALA loop (MFMA
conv/GEMM via MIOpen/hipBLASLt); the actor path is hipGraph-captured by the
learner loop (fixed [1, B] shapes), which removes the ~70-kernel launch
overhead that dominates small-CNN inference.
"""
import torch
import torch.nn.functional as F
from torch import nn


def _conv3(x, conv):
    """Bias-free 3x3/s1/p1 conv: the MFMA kernel when the shape/dtype
    allows (forward AND input-gradient on the custom kernel — measured
    faster than MIOpen at every IMPALA shape, profiles/r2_conv3x3_micro),
    F.conv2d otherwise."""
    from moolib_amd.ops import conv3x3 as c3

    if (
        x.is_cuda
        and c3.available(conv.in_channels, conv.out_channels)
        and (x.dtype == torch.bfloat16 or torch.is_autocast_enabled())
    ):
        return c3.conv3x3_autograd(x, conv)
    return F.conv2d(x, conv.weight, None, padding=1)


class ResidualBlock(nn.Module):
    def __init__(self, ch):
        super().__init__()
        self.conv0 = nn.Conv2d(ch, ch, 3, stride=1, padding=1)
        self.conv1 = nn.Conv2d(ch, ch, 3, stride=1, padding=1)

    def forward(self, x):
        y = self.conv0(F.relu(x))
        y = self.conv1(F.relu(y))
        return x + y

    def forward_fused(self, x, pending_bias=None):
        """Same math with the conv biases folded into the elementwise ops
        (ops/fused_bias): convs run bias-free, relu/add are one pass each.
        `pending_bias` is a per-channel bias the CALLER still owes `x`
        (section conv bias carried through the maxpool).

        Inference fast path (actor, no_grad): the whole block is TWO fused
        MFMA conv kernels (ops/conv3x3) — producer relu+bias on load,
        bias+residual on store.
        """
        from moolib_amd.ops import conv3x3 as c3
        from moolib_amd.ops.fused_bias import bias_add2, bias_relu

        ch = self.conv0.out_channels
        if (
            not torch.is_grad_enabled()
            and x.is_cuda
            and c3.available(x.size(1), ch)
            and (x.dtype == torch.bfloat16 or torch.is_autocast_enabled())
        ):
            u = c3.conv3x3(
                x, c3.packed_buffer(self.conv0), ch,
                relu_in=True, bias_in=pending_bias,
            )
            return c3.conv3x3(
                u, c3.packed_buffer(self.conv1), ch,
                relu_in=True, bias_in=self.conv0.bias,
                epi=c3.EPI_BIAS_ADD, bias1=self.conv1.bias,
                res=x, bias2=pending_bias,
            )

        if pending_bias is not None:
            t = bias_relu(x, pending_bias)
        else:
            t = F.relu(x)
        u = _conv3(t, self.conv0)
        t = bias_relu(u, self.conv0.bias)
        u = _conv3(t, self.conv1)
        return bias_add2(u, self.conv1.bias, x, pending_bias)


class ConvSection(nn.Module):
    def __init__(self, in_ch, out_ch):
        super().__init__()
        self.conv = nn.Conv2d(in_ch, out_ch, 3, stride=1, padding=1)
        self.pool = nn.MaxPool2d(3, stride=2, padding=1)  # CPU fallback
        self.res0 = ResidualBlock(out_ch)
        self.res1 = ResidualBlock(out_ch)

    def forward(self, x):
        from moolib_amd.ops import fused_bias
        from moolib_amd.ops.pool import maxpool3x3s2

        if fused_bias.available(x, self.conv.out_channels):
            # bias-free conv; per-channel bias commutes with the per-channel
            # spatial max, so it rides into res0 as a pending bias.
            from moolib_amd.ops import conv3x3 as c3

            if (
                not torch.is_grad_enabled()
                and c3.available(x.size(1), self.conv.out_channels)
                and (x.dtype == torch.bfloat16 or torch.is_autocast_enabled())
            ):
                u = c3.conv3x3(
                    x, c3.packed_buffer(self.conv), self.conv.out_channels
                )
            else:
                u = _conv3(x, self.conv)
            u = maxpool3x3s2(u)
            u = self.res0.forward_fused(u, self.conv.bias)
            return self.res1.forward_fused(u)
        x = maxpool3x3s2(self.conv(x))
        x = self.res0(x)
        x = self.res1(x)
        return x


class AtariNet(nn.Module):
    def __init__(self, num_actions=18, input_channels=4, use_lstm=False):
        super().__init__()
        self.num_actions = num_actions
        self.use_lstm = use_lstm

        chans = [16, 32, 32]
        sections = []
        ch_in = input_channels
        for ch in chans:
            sections.append(ConvSection(ch_in, ch))
            ch_in = ch
        self.sections = nn.ModuleList(sections)

        self.fc = nn.Linear(3872, 256)
        core_in = self.fc.out_features + num_actions + 1
        if use_lstm:
            self.core = nn.LSTM(core_in, 256, num_layers=1)
            core_in = 256
        self.policy = nn.Linear(core_in, num_actions)
        self.baseline = nn.Linear(core_in, 1)
        # Optional dedicated RNG for the in-forward multinomial: hipGraph
        # replays of two captures racing on the DEFAULT generator's device
        # state is what forced the actor side stream off; per-capture
        # generators make concurrent replays sound.
        self.sample_generator = None

    def initial_state(self, batch_size=1):
        if not self.use_lstm:
            return tuple()
        return tuple(
            torch.zeros(self.core.num_layers, batch_size, self.core.hidden_size)
            for _ in range(2)
        )

    def forward(self, inputs, core_state=None):
        x = inputs["state"]
        reward = inputs["reward"]
        T, B = x.shape[:2]
        x = x.flatten(0, 1)
        import os as _os

        bf16_compute = torch.is_autocast_enabled() or self.fc.weight.dtype == torch.bfloat16
        kernels = None
        if x.is_cuda and x.dtype == torch.uint8 and bf16_compute:
            try:
                from moolib_amd import _kernels as kernels
            except ImportError:
                kernels = None

        first = 0
        if (
            kernels is not None
            and self.sections[0].conv.in_channels == 4
            and not _os.environ.get("MOOLIB_AMD_NO_CONV1_KERNEL")
        ):
            # Fused uint8 frames -> conv1 + bias in one kernel (no separate
            # preprocessing pass, no 19 MB fp intermediate). Under grad the
            # same kernel runs through an autograd Function whose backward
            # is our wgrad kernel + a bias-sum — the first layer never
            # touches MIOpen in either direction.
            sec0 = self.sections[0]
            c1 = sec0.conv
            from moolib_amd.ops import conv3x3 as c3m

            if torch.is_grad_enabled() and c1.weight.requires_grad:
                x = c3m.conv1_u8_autograd(x, c1, 1.0 / 255.0)
            elif _os.environ.get("MOOLIB_AMD_CONV1_SCALAR"):
                # the original fused scalar kernel (kept for A/Bs)
                w = c1.weight.detach().to(torch.bfloat16).permute(2, 3, 1, 0).contiguous()
                b = c1.bias.detach().to(torch.bfloat16)
                x = kernels.conv1_u8_nhwc(x, w, b, 1.0 / 255.0)
            else:
                # same C=8 MFMA route as the learner (graph-capturable:
                # the packed weights live in a stable version-checked
                # buffer refreshed by repack())
                x8 = kernels.frames_u8_to_bf16_nhwc(x, 1.0 / 255.0, 8)
                x = c3m.conv3x3(
                    x8, c3m.conv1_packed_buffer(c1), c1.out_channels,
                    epi=c3m.EPI_BIAS, bias1=c1.bias.detach(),
                )
            from moolib_amd.ops.pool import maxpool3x3s2

            x = maxpool3x3s2(x)
            from moolib_amd.ops import fused_bias

            if fused_bias.available(x):
                x = sec0.res1.forward_fused(sec0.res0.forward_fused(x))
            else:
                x = sec0.res1(sec0.res0(x))
            first = 1
        elif kernels is not None and not _os.environ.get("MOOLIB_AMD_NO_FRAMES_KERNEL"):
            x = kernels.frames_u8_to_bf16_nhwc(x, 1.0 / 255.0)
        else:
            x = x.to(self.fc.weight.dtype).mul_(1.0 / 255.0)
            if x.is_cuda:
                # NHWC: MIOpen's bf16 igemm kernels are NHWC-native; NCHW
                # input inserts a batched_transpose around every conv.
                # (Note: the FC input ordering then differs from the
                # CPU/NCHW path — a self-consistent permutation of learned
                # features.)
                x = x.contiguous(memory_format=torch.channels_last)

        for s in self.sections[first:]:
            x = s(x)
        x = F.relu(x)
        x = x.reshape(T * B, -1)
        x = F.relu(self.fc(x))

        prev_action_onehot = F.one_hot(
            inputs["prev_action"].view(T * B).to(torch.int64), self.num_actions
        ).to(x.dtype)
        clipped_reward = reward.view(T * B, 1).clamp(-1, 1).to(x.dtype)
        core_input = torch.cat([x, clipped_reward, prev_action_onehot], dim=-1)

        if self.use_lstm:
            done = inputs["done"]
            core_input = core_input.view(T, B, -1)
            from moolib_amd.ops import lstm as lstm_ops

            if lstm_ops.available(self.core.hidden_size, core_input.device) and (
                torch.is_autocast_enabled() or core_input.dtype == torch.bfloat16
            ):
                # fused MFMA sequence scan: one GEMM for x@W_ih over all T,
                # one kernel for the masked recurrence (fwd and bwd each).
                Xg = F.linear(
                    core_input.reshape(T * B, -1),
                    self.core.weight_ih_l0,
                    self.core.bias_ih_l0 + self.core.bias_hh_l0,
                ).view(T, B, -1).to(torch.bfloat16)
                notdone_f = (~done).float()
                h0 = core_state[0][0].to(torch.bfloat16)
                c0 = core_state[1][0].float()
                H, hT, cT = lstm_ops.fused_lstm_scan(
                    Xg, notdone_f, h0, c0, self.core.weight_hh_l0
                )
                core_output = H.flatten(0, 1)
                core_state = (
                    hT.unsqueeze(0).to(core_state[0].dtype),
                    cT.unsqueeze(0).to(core_state[1].dtype),
                )
            else:
                notdone = (~done).to(core_input.dtype)
                outputs = []
                for t in range(T):
                    nd = notdone[t].view(1, -1, 1)
                    core_state = tuple(nd * s for s in core_state)
                    out_t, core_state = self.core(core_input[t : t + 1], core_state)
                    outputs.append(out_t)
                core_output = torch.cat(outputs).flatten(0, 1)
        else:
            core_output = core_input

        policy_logits = self.policy(core_output).float()
        baseline = self.baseline(core_output).float()
        out = dict(
            policy_logits=policy_logits.view(T, B, self.num_actions),
            baseline=baseline.view(T, B),
        )
        if not torch.is_grad_enabled():
            # Actor path: sample the action in-forward (reference
            # models.py:134-140). The learner discards the sample (it
            # trains on the ACTOR's actions), and multinomial is one of
            # the slowest ops in the step (~230 us CPU+GPU at [672, 18]) —
            # skip it whenever autograd is recording.
            action = torch.multinomial(
                F.softmax(policy_logits, dim=-1), num_samples=1,
                generator=self.sample_generator,
            )
            out["action"] = action.view(T, B)
        return out, core_state


def create_model(num_actions=18, use_lstm=False, device="cpu"):
    model = AtariNet(num_actions=num_actions, use_lstm=use_lstm)
    model.to(device)
    return model
