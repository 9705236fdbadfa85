"""IMPALA/V-trace training engine: the flagship cooperative loop.

Capability parity with the reference's examples/vtrace/experiment.py main
loop (acting, time/learn batching, V-trace learning, elastic Accumulator
protocol), packaged as a reusable engine so the benchmark, the example CLI
and the tests share one implementation.

MI355X design:
  - model compute runs under bf16 autocast (MFMA via MIOpen/hipBLASLt);
    V-trace + losses stay fp32 (fused HIP kernels when available);
  - gradient allreduce runs on RCCL over xGMI via the Accumulator's
    collective hook when the peers form a torch.distributed world;
  - actor-batch H2D staging and learner batching overlap the compute stream
    (Batcher non-blocking copies);
  - the actor forward (fixed [1, B] shape) is hipGraph-captured when
    enabled (cfg.graph_actor) to eliminate launch overhead.
"""
import logging
import contextlib
import dataclasses
import os
import time
from typing import Any, Optional

import torch

import moolib_amd
from moolib_amd import parallel
from moolib_amd.models.atari import AtariNet
from moolib_amd.ops import fused_loss, vtrace
from moolib_amd.utils import nest
from moolib_amd.utils.stats import GlobalStatsAccumulator, StatMean, StatSum


@dataclasses.dataclass
class ImpalaConfig:
    def __post_init__(self):
        # Coerce numeric fields that arrived as strings (e.g. yaml "6e-4",
        # which YAML 1.1 parses as a string) to the declared field type.
        for f in dataclasses.fields(self):
            v = getattr(self, f.name)
            if isinstance(v, str) and f.type in ("int", "float", int, float):
                cast = int if f.type in ("int", int) else float
                try:
                    setattr(self, f.name, cast(float(v)))
                except ValueError:
                    pass

    num_actions: int = 18
    actor_batch_size: int = 128
    num_actor_batches: int = 2
    num_actor_cpus: int = 10
    batch_size: int = 32
    unroll_length: int = 20
    virtual_batch_size: int = 32
    discounting: float = 0.99
    baseline_cost: float = 0.5
    entropy_cost: float = 0.0006
    reward_clip: float = 1.0
    grad_norm_clipping: float = 40.0
    learning_rate: float = 6e-4
    adam_beta1: float = 0.9
    adam_beta2: float = 0.999
    adam_eps: float = 1e-8
    total_steps: float = 50e6
    device: str = "cuda:0"
    use_lstm: bool = False
    autocast_bf16: bool = True
    connect: str = "127.0.0.1:4431"
    local_name: str = ""
    group_name: str = "impala"
    lr_schedule: bool = True
    channels_last: bool = True   # NHWC convs (MIOpen direct, no transposes)
    graph_actor: bool = True     # hipGraph-capture the actor forward
    graph_optimizer: bool = True  # hipGraph-capture clip_grad_norm + Adam step
    #   (no autograd in the capture; only used when the LR is constant —
    #   python-float hyperparameters are baked into the graph)
    graph_learner: bool = False  # hipGraph-capture the learner fwd+bwd.
    #   OFF by default: replays of the captured learner (autograd backward
    #   incl. MIOpen bwd-weight convs) intermittently fault on ROCm 7.0
    #   (HSA_STATUS_ERROR_EXCEPTION, ~1 in 6 runs; actor-graph-only is 8/8
    #   stable across every stress batch). Costs ~8% vs full graphs;
    #   re-enable via graph_learner=True to investigate (round 2).
    pinned_staging: bool = True  # pinned bounce buffers for shm->HBM copies
    shm_host_register: bool = False  # hipHostRegister the env shm (measured SLOWER on MI355X)
    prewarm: bool = True  # construction-time pipeline priming: act until the
    #   learn backlog is at its steady level, so measurement windows start
    #   at steady state instead of spending ~30 steps filling it
    actor_side_stream: bool = False  # overlap actor work on a side HIP stream.
    #   OFF by default: concurrent hipGraph replays on two streams still hit
    #   an intermittent HSA memory fault (~1 in 8 runs) on ROCm 7.0 even
    #   with per-capture RNG generators registered; single-stream replay is
    #   9/9 stable and costs only ~2%. Re-enable with
    #   MOOLIB_AMD_FORCE_SIDE_STREAM=1 to investigate (round 2).
    prefetch_h2d: bool = True        # DMA env results to HBM on a copy stream as soon as
    #   the workers finish (poll + persistent device buffers + events), so the
    #   ~2 ms/step of frame H2D overlaps learner/actor compute instead of
    #   serializing in the act loop.
    max_learn_backlog: int = 4       # stop acting when this many learn batches are queued
    group_timeout: float = 10.0      # broker eviction / allreduce timeout (seconds)
    bf16_weights: bool = True        # forward on bf16 shadow weights (fp32 master for optimizer/sync):
    #   removes ~250 autocast weight-cast kernels per step from the graph replays (+3.5%)


class EnvBatchState:
    """Per-actor-batch rollout state (reference: examples/common EnvBatchState)."""

    def __init__(self, cfg: ImpalaConfig, model):
        B = cfg.actor_batch_size
        device = cfg.device
        self.prev_action = torch.zeros(B, dtype=torch.int64, device=device)
        self.future = None
        dt = next(model.parameters()).dtype
        self.core_state = tuple(
            s.to(device=device, dtype=dt) for s in model.initial_state(batch_size=B)
        )
        self.initial_core_state = self.core_state
        self.running_reward = torch.zeros(B)
        self.step_count = torch.zeros(B)
        self.time_batcher = moolib_amd.Batcher(cfg.unroll_length + 1, device)
        self.pinned = {}  # field -> pinned staging tensor
        # H2D prefetch (copy-stream pipeline): persistent device buffers the
        # prefetcher DMAs shm results into while the GPU is busy elsewhere.
        self.staged = None           # field -> device tensor (stable addresses)
        self.prefetched = None       # cpu result dict already consumed from the future
        self.h2d_event = None        # recorded on the copy stream after staging
        self.consumed_event = None   # recorded on the compute stream after last read

    def update(self, env_outputs, action, stats):
        # prev_action is a persistent device buffer (stable address for
        # hipGraph replay); copy instead of rebinding.
        self.prev_action.copy_(action, non_blocking=True)
        self.running_reward += env_outputs["reward"]
        self.step_count += 1
        done = env_outputs["done"]
        n_done = int(done.sum().item())
        if n_done > 0:
            stats["mean_episode_return"] += (self.running_reward * done).sum().item() / n_done
            stats["mean_episode_step"] += (self.step_count * done).sum().item() / n_done
        stats["steps_done"] += done.numel()
        stats["episodes_done"] += n_done
        not_done = ~done
        self.running_reward *= not_done
        self.step_count *= not_done


def make_stats():
    return {
        "mean_episode_return": StatMean(),
        "mean_episode_step": StatMean(),
        "SPS": StatMean(),
        "env_act_steps": StatSum(),
        "env_train_steps": StatSum(),
        "optimizer_steps": StatSum(),
        "steps_done": StatSum(),
        "episodes_done": StatSum(),
        "unclipped_grad_norm": StatMean(),
        "model_version": StatSum(),
        "virtual_batch_size": StatMean(),
        "num_gradients": StatMean(),
    }


def flatten_master_shadow(master_params, fwd_params):
    """Re-home master (fp32) and shadow (bf16) params and their grads as
    views into four contiguous flat buffers.

    Cross-dtype ``_foreach_copy_`` decomposes into one copy kernel per
    tensor, so the per-optimizer-step fp32->bf16 weight sync and
    bf16->fp32 grad cast each cost ~1 launch per parameter (measured 38
    aten::copy_ calls/step, profiles/evidence/r4d_stepprof.txt). With
    flat buffers each cast is ONE kernel: ``flat_shadow.copy_(flat_master)``
    and ``flat_grad32.copy_(flat_grad16)``. Grad views are preassigned so
    autograd accumulates straight into the flat buffer.

    Returns (flat_master, flat_shadow, flat_grad32, flat_grad16).
    Note: views are contiguous — a channels_last weight loses its NHWC
    strides, which is fine here because every forward path reads the
    MFMA-packed weight caches (ops/conv3x3), not the raw strides.
    """
    total = sum(p.numel() for p in fwd_params)
    dev = fwd_params[0].device if fwd_params else "cpu"
    flat_master = torch.empty(total, dtype=torch.float32, device=dev)
    flat_shadow = torch.empty(total, dtype=torch.bfloat16, device=dev)
    flat_grad32 = torch.zeros(total, dtype=torch.float32, device=dev)
    flat_grad16 = torch.zeros(total, dtype=torch.bfloat16, device=dev)
    off = 0
    for pm, pf in zip(master_params, fwd_params):
        n = pm.numel()
        mv = flat_master[off : off + n].view_as(pm)
        mv.copy_(pm.detach())
        pm.data = mv
        pm.grad = flat_grad32[off : off + n].view_as(pm)
        sv = flat_shadow[off : off + n].view_as(pf)
        sv.copy_(pf.detach())
        pf.data = sv
        pf.grad = flat_grad16[off : off + n].view_as(pf)
        off += n
    return flat_master, flat_shadow, flat_grad32, flat_grad16


class ImpalaPeer:
    """One learner+actor peer (one GPU)."""

    def __init__(
        self,
        cfg: ImpalaConfig,
        create_env=None,
        model: Optional[torch.nn.Module] = None,
        use_collective_backend: bool = False,
        broker: Optional[Any] = None,
        envs: Optional["moolib_amd.EnvPool"] = None,
    ):
        self.cfg = cfg
        # EnvPool forks workers: create it before anything that spawns
        # threads or touches the GPU (same constraint as the reference,
        # env.cc:150-157). Callers that must initialize torch.distributed
        # can pre-build the pool and pass it via `envs`.
        if envs is not None:
            self.envs = envs
        else:
            self.envs = moolib_amd.EnvPool(
                create_env,
                num_processes=cfg.num_actor_cpus,
                batch_size=cfg.actor_batch_size,
                num_batches=cfg.num_actor_batches,
            )
        self.broker = broker

        self.model = model if model is not None else AtariNet(
            num_actions=cfg.num_actions, use_lstm=cfg.use_lstm
        )
        self.model.to(cfg.device)
        self.optimizer = torch.optim.Adam(
            self.model.parameters(),
            lr=cfg.learning_rate,
            betas=(cfg.adam_beta1, cfg.adam_beta2),
            eps=cfg.adam_eps,
            # multi-tensor kernels: one launch per op across all 40 params
            # instead of 40 tiny latency-bound launches each
            foreach=True,
        )
        if cfg.lr_schedule:
            factor = cfg.unroll_length * cfg.virtual_batch_size / cfg.total_steps
            self.scheduler = torch.optim.lr_scheduler.LambdaLR(
                self.optimizer, lambda epoch: max(1 - epoch * factor, 0)
            )
        else:
            self.scheduler = None
        self._graph_opt = (
            cfg.graph_optimizer
            and self.scheduler is None
            and torch.device(cfg.device).type == "cuda"
            and not os.environ.get("MOOLIB_AMD_NO_OPT_GRAPH")
        )
        if self._graph_opt:
            # capturable: the Adam step counter must live on-device so graph
            # replays advance bias correction
            for g in self.optimizer.param_groups:
                g["capturable"] = True

        self.rpc = moolib_amd.Rpc()
        self.rpc.set_name(cfg.local_name or ("peer-" + moolib_amd.create_uid()[:8]))
        self.rpc.set_timeout(20)
        self.group = moolib_amd.Group(self.rpc, cfg.group_name)
        self.group.set_timeout(cfg.group_timeout)
        self.accumulator = moolib_amd.Accumulator(
            "model", self.model.parameters(), self.model.buffers(), group=self.group
        )
        self.accumulator.set_virtual_batch_size(cfg.virtual_batch_size)
        self.accumulator.connect(cfg.connect)
        if use_collective_backend:
            parallel.install_collective_backend(self.accumulator)

        self.learn_batcher = moolib_amd.Batcher(cfg.batch_size, cfg.device, dim=1)
        self.stats = make_stats()
        self.global_stats_accumulator = GlobalStatsAccumulator(self.group, make_stats())
        self.next_env_index = 0
        self.model_version = 0
        self.is_cuda = torch.device(cfg.device).type == "cuda"
        # env kill-switches for debugging/bisection
        if os.environ.get("MOOLIB_AMD_NO_GRAPHS"):
            cfg.graph_actor = cfg.graph_learner = False
        if os.environ.get("MOOLIB_AMD_NO_ACTOR_GRAPH"):
            cfg.graph_actor = False
        if os.environ.get("MOOLIB_AMD_NO_LEARNER_GRAPH"):
            cfg.graph_learner = False
        if os.environ.get("MOOLIB_AMD_FORCE_LEARNER_GRAPH"):
            cfg.graph_learner = True
        if os.environ.get("MOOLIB_AMD_NO_SIDE_STREAM"):
            cfg.actor_side_stream = False
        if os.environ.get("MOOLIB_AMD_FORCE_SIDE_STREAM"):
            cfg.actor_side_stream = True
        if os.environ.get("MOOLIB_AMD_BF16_WEIGHTS"):
            cfg.bf16_weights = True
        if os.environ.get("MOOLIB_AMD_NO_BF16_WEIGHTS"):
            cfg.bf16_weights = False
        self.autocast = cfg.autocast_bf16 and self.is_cuda
        if self.is_cuda and cfg.channels_last:
            self.model.to(memory_format=torch.channels_last)
        # bf16 shadow: forwards/backwards run on bf16 weights (no per-use
        # autocast casts inside every graph replay); the fp32 master owns
        # the optimizer, the accumulator and model sync. Gradients are cast
        # bf16->fp32 once per learn; weights fp32->bf16 once per optimizer
        # step.
        self.bf16_shadow = self.autocast and cfg.bf16_weights
        if self.bf16_shadow:
            import copy as _copy

            self.fwd_model = _copy.deepcopy(self.model).to(torch.bfloat16)
            if cfg.channels_last:
                self.fwd_model.to(memory_format=torch.channels_last)
            self.autocast = False
            self._master_params = [p for p in self.model.parameters() if p.requires_grad]
            self._fwd_params = [p for p in self.fwd_model.parameters() if p.requires_grad]
            # Flat-buffer layout: master weights/grads and shadow
            # weights/grads are views into four contiguous buffers, so the
            # per-step fp32<->bf16 casts are ONE copy kernel each instead
            # of one per tensor (cross-dtype _foreach_copy_ decomposes to
            # per-tensor copies — measured 38 copy launches per optimizer
            # step in profiles/evidence/r4d_stepprof.txt).
            (
                self._flat_master,
                self._flat_shadow,
                self._flat_grad32,
                self._flat_grad16,
            ) = flatten_master_shadow(self._master_params, self._fwd_params)
        else:
            self.fwd_model = self.model
            self._master_params = []
            self._fwd_params = []
        self.env_states = [
            EnvBatchState(cfg, self.fwd_model) for _ in range(cfg.num_actor_batches)
        ]
        self.actor_stream = (
            torch.cuda.Stream() if (self.is_cuda and cfg.actor_side_stream) else None
        )
        self._batch_events = []
        if self.is_cuda and hasattr(self.fwd_model, "sample_generator"):
            self._actor_rng = torch.Generator(device=cfg.device)
            self._actor_rng.manual_seed(int(torch.seed()) % (2**62))
            self._learn_rng = torch.Generator(device=cfg.device)
            self._learn_rng.manual_seed(int(torch.seed()) % (2**62))
        else:
            self._actor_rng = None
            self._learn_rng = None
        from moolib_amd.parallel.graphs import GraphedCall

        self._opt_call = (
            GraphedCall(self._opt_fn, warmup=3, name="optimizer_step")
            if self._graph_opt
            else self._opt_fn
        )

        self._actor_call = (
            GraphedCall(
                self._actor_fn,
                warmup=3,
                name="actor_fwd",
                generators=[self._actor_rng] if self._actor_rng is not None else (),
            )
            if (self.is_cuda and cfg.graph_actor)
            else self._actor_fn
        )
        self._learn_call = (
            GraphedCall(
                self._learn_fn,
                warmup=3,
                name="learner_fwd_bwd",
                generators=[self._learn_rng] if self._learn_rng is not None else (),
            )
            if (self.is_cuda and cfg.graph_learner)
            else self._learn_fn
        )
        self._h2d_stream = (
            torch.cuda.Stream()
            if (
                self.is_cuda
                and cfg.prefetch_h2d
                and not os.environ.get("MOOLIB_AMD_NO_PREFETCH")
            )
            else None
        )
        self._shm_registered = False
        if self.is_cuda and cfg.shm_host_register:
            try:
                from moolib_amd import _kernels

                _kernels.register_host_memory(self.envs.shared_buffer())
                self._shm_registered = True
            except Exception:  # noqa: BLE001
                pass  # fall back to pinned bounce buffers
        # Phase profiling (bench --breakdown): cumulative seconds per phase,
        # with device sync at boundaries so GPU time is attributed correctly.
        self.profile = False
        # Env-fed scheduling (opportunistic acting before learn/optimize)
        # smooths the periodic act bursts but LOSES throughput overall —
        # same-box interleaved A/B: off 81.1/82.9k short, 88.5k sustained
        # vs on 79.5/71.1k short, 84.9k sustained (evidence/r3y_*.json).
        # The inline acts' CPU orchestration serializes with the learner's
        # and costs more than the bursts they spread out. Off by default;
        # MOOLIB_AMD_ENVFED=1 enables it for experiments.
        self._envfed = os.environ.get("MOOLIB_AMD_ENVFED", "0") == "1"
        self.phase_times = {}
        if self.is_cuda and cfg.prewarm and not os.environ.get("MOOLIB_AMD_NO_PREWARM"):
            try:
                self._prewarm()
            except Exception as e:  # noqa: BLE001 — e.g. a custom model whose
                # observation shape differs from the synthetic warm batch
                logging.warning("prewarm skipped: %s", e)

    def _prewarm(self):
        """Fill the actor->learner pipeline to its steady backlog at
        construction.

        Measured (profiles/evidence/r3t_*.json, one box): a 20-step window
        after 5 warmup steps reads ~80k f/s while the same window after 20
        warmup steps reads 92-95k. The gap is NOT one-time init (a dry
        fwd/bwd/opt prewarm changed nothing — MIOpen find and the graph
        captures already settle inside 5 warmup steps); it is the learn
        backlog filling: the actor gets ahead of the learner only by its
        small per-step surplus, so reaching the steady max_learn_backlog
        takes ~30-40 optimizer steps. Acting the pipeline full at
        construction starts ANY window at steady state — no borrowed work,
        since in steady state the backlog is level across the window (the
        actor replaces every batch the learner consumes)."""
        target = self.cfg.max_learn_backlog
        t0 = time.time()
        while self.learn_batcher.size() < target and time.time() - t0 < 120.0:
            self.act_once()
        torch.cuda.synchronize()

    def _t(self, name, t0):
        if self.profile:
            if torch.device(self.cfg.device).type == "cuda":
                torch.cuda.synchronize()
            t1 = time.perf_counter()
            self.phase_times[name] = self.phase_times.get(name, 0.0) + (t1 - t0)
            return t1
        return t0

    # ------------------------------------------------------------ learning

    def _sync_shadow(self):
        if not self.bf16_shadow:
            return
        with torch.no_grad():
            # single cast-copy for ALL params (flat fp32 -> flat bf16)
            self._flat_shadow.copy_(self._flat_master, non_blocking=True)
            bufs_dst = list(self.fwd_model.buffers())
            bufs_src = list(self.model.buffers())
            if bufs_dst:
                torch._foreach_copy_(bufs_dst, bufs_src)

    def _learn_fn(self, data):
        """Forward + V-trace + fused loss + backward. hipGraph-capturable:
        fixed shapes, writes gradients into the (stable) param.grad tensors."""
        cfg = self.cfg
        model = self.fwd_model
        if self._learn_rng is not None:
            model.sample_generator = self._learn_rng
        # (shadow-grad zeroing happens in compute_gradients, outside any
        # captured region, as ONE foreach launch)
        env_outputs = data["env_outputs"]
        actor_outputs = data["actor_outputs"]
        initial_core_state = data["initial_core_state"]

        with torch.autocast(
            "cuda", dtype=torch.bfloat16, enabled=self.autocast, cache_enabled=False
        ):
            learner_outputs, _ = model(env_outputs, initial_core_state)

        bootstrap_value = learner_outputs["baseline"][-1]
        learner_outputs = nest.map(lambda t: t[:-1], learner_outputs)
        env_outputs_t = nest.map(lambda t: t[1:], env_outputs)
        actor_outputs = nest.map(lambda t: t[:-1], actor_outputs)

        rewards = env_outputs_t["reward"].float()
        if cfg.reward_clip:
            rewards = rewards.clamp(-cfg.reward_clip, cfg.reward_clip)
        discounts = (~env_outputs_t["done"]).float() * cfg.discounting

        vt = vtrace.from_logits(
            behavior_policy_logits=actor_outputs["policy_logits"].float(),
            target_policy_logits=learner_outputs["policy_logits"].float(),
            actions=actor_outputs["action"],
            discounts=discounts,
            rewards=rewards,
            values=learner_outputs["baseline"].float(),
            bootstrap_value=bootstrap_value.float(),
        )

        total = fused_loss.impala_total_loss(
            learner_outputs["policy_logits"].float(),
            learner_outputs["baseline"].float(),
            actor_outputs["action"],
            vt.pg_advantages,
            vt.vs,
            cfg.entropy_cost,
            cfg.baseline_cost,
        )
        total.backward()
        return ()

    def compute_gradients(self, data):
        self.fwd_model.train()
        # Shadow-grad bookkeeping stays OUTSIDE the captured region: zeroing
        # and cast-copying the shadow grads inside the learner graph is what
        # made replays fault on ROCm 7.0 (captured pure fwd+bwd is stable).
        if self.bf16_shadow:
            with torch.no_grad():
                self._flat_grad16.zero_()
        self._learn_call(data)
        if self.bf16_shadow:
            with torch.no_grad():
                # single cast-copy for ALL grads (flat bf16 -> flat fp32);
                # shadow .grad tensors are preassigned views of the flat
                # buffer, so autograd accumulates straight into it
                self._flat_grad32.copy_(self._flat_grad16, non_blocking=True)
        self.stats["env_train_steps"] += self.cfg.unroll_length * self.cfg.batch_size

    def _actor_fn(self, inputs):
        """No-grad actor forward on [1, B]. hipGraph-capturable."""
        env_outputs, core_state = inputs["env"], inputs["core"]
        if self._actor_rng is not None:
            self.fwd_model.sample_generator = self._actor_rng
        with torch.no_grad(), torch.autocast(
            "cuda", dtype=torch.bfloat16, enabled=self.autocast, cache_enabled=False
        ):
            actor_outputs, core_out = self.fwd_model(
                nest.map(lambda t: t.unsqueeze(0), env_outputs), core_state
            )
        return {"out": actor_outputs, "core": core_out if core_out else ()}

    def _opt_fn(self, _inputs):
        norm = torch.nn.utils.clip_grad_norm_(
            self.model.parameters(), self.cfg.grad_norm_clipping
        )
        self.optimizer.step()
        self._sync_shadow()
        # Refresh cached MFMA-packed conv weights (ops/conv3x3) so actor
        # graph replays read the updated weights; capturable ops only.
        from moolib_amd.ops import conv3x3 as _c3

        if self.is_cuda and _c3.available(16, 16):
            _c3.repack(self.fwd_model)
        return norm

    def step_optimizer(self):
        cfg = self.cfg
        norm = self._opt_call({})
        if self.scheduler is not None:
            self.scheduler.step()
        self.model_version += 1
        # .item() is a device sync: sample the norm stat instead of paying
        # the sync every step.
        if not self.is_cuda or self.model_version % 16 == 0:
            self.stats["unclipped_grad_norm"] += float(norm)
        self.stats["optimizer_steps"] += 1
        self.stats["model_version"] += 1

    # --------------------------------------------------------------- state

    def save_state(self):
        return {
            "model": {k: v.cpu() for k, v in self.model.state_dict().items()},
            "optimizer": self.optimizer.state_dict(),
            "scheduler": self.scheduler.state_dict() if self.scheduler else None,
            "model_version": self.model_version,
        }

    def load_state(self, state):
        # params/buffers were synced by the accumulator already; adopt the rest
        self.optimizer.load_state_dict(state["optimizer"])
        if self.scheduler is not None and state.get("scheduler"):
            self.scheduler.load_state_dict(state["scheduler"])
        self.model_version = state["model_version"]

    # ----------------------------------------------------------------- loop

    def step_once(self):
        """One cooperative iteration. Returns the event performed."""
        cfg = self.cfg
        if self.broker is not None:
            self.broker.update()
        self.group.update()
        acc = self.accumulator
        acc.update()

        if acc.wants_state():
            acc.set_state(self.save_state())
        if acc.has_new_state():
            self.load_state(acc.state())
            self._sync_shadow()  # master params were replaced by the leader's

        if not acc.connected():
            time.sleep(0.05)
            return "idle"

        # Opportunistic H2D: if the next batch's envs are done, start the
        # copy-stream DMA before committing this iteration to learn or act.
        self._try_prefetch()

        # Keep the env workers FED: if an env batch's results are ready,
        # act on them before learning. Without this the strict
        # learn-over-act priority drains the backlog while the workers sit
        # idle, and acting then happens as a serialized ~40-round burst
        # (measured: steady learner steps are 5.3 ms but every ~8th step
        # stalled 24 ms in the burst — profiles/evidence/r3v_ramp.txt).
        # The act's GPU work is small and queues behind the learner's on
        # the same stream; its CPU orchestration overlaps learner GPU time.
        # (~5 acts sustain one optimizer step; a single act per iteration
        # still left ~35-act bursts at the drain point)
        if self._envfed:
            for _ in range(2 * cfg.num_actor_batches):
                if (
                    self.learn_batcher.size() >= cfg.max_learn_backlog
                    or not self._env_ready()
                ):
                    break
                self.act_once()

        t0 = time.perf_counter() if self.profile else 0.0
        if acc.has_gradients():
            gstats = acc.get_gradient_stats()
            self.stats["virtual_batch_size"] += gstats["batch_size"]
            self.stats["num_gradients"] += gstats["num_gradients"]
            self.step_optimizer()
            acc.zero_gradients()
            if self.actor_stream is not None:
                # next actor forwards must see the updated weights
                self.actor_stream.wait_stream(torch.cuda.current_stream())
            self._t("optimize", t0)
            return "optimize"
        elif not self.learn_batcher.empty() and acc.wants_gradients():
            data = self.learn_batcher.get()
            if self.actor_stream is not None:
                # wait only for the event recorded when THIS batch completed,
                # and tell the allocator the tensors are now used on the
                # learner stream (they were written on the actor stream).
                if self._batch_events:
                    torch.cuda.current_stream().wait_event(self._batch_events.pop(0))
                else:
                    torch.cuda.current_stream().wait_stream(self.actor_stream)
                cur = torch.cuda.current_stream()
                for t in nest.flatten(data):
                    if isinstance(t, torch.Tensor) and t.is_cuda:
                        t.record_stream(cur)
            self.compute_gradients(data)
            t0 = self._t("learn_fwd_bwd", t0)
            acc.reduce_gradients(cfg.batch_size)
            self._t("learn_reduce", t0)
            return "learn"
        else:
            if acc.wants_gradients():
                acc.skip_gradients()
            if self.learn_batcher.size() >= cfg.max_learn_backlog:
                # Backpressure: the learner is behind (slow peer / stalled
                # round) — unbounded acting would grow the ready queue (and
                # HBM use) without limit.
                time.sleep(0.001)
                return "throttle"
            self.act_once()
            return "act"

    def _env_ready(self):
        """True if the next env batch's results can be consumed without
        blocking (already prefetched, or the workers have posted)."""
        st = self.env_states[self.next_env_index]
        if st.prefetched is not None:
            return True
        return st.future is not None and self.envs.poll(self.next_env_index)

    def _try_prefetch(self):
        """If the next batch's env step has completed, DMA its results into
        persistent device buffers on the copy stream NOW — the transfer
        overlaps whatever the compute stream is doing (learner fwd+bwd,
        the other batch's actor forward)."""
        if self._h2d_stream is None:
            return
        cur = self.next_env_index
        st = self.env_states[cur]
        if st.future is None or st.prefetched is not None:
            return
        if not self.envs.poll(cur):
            return
        cpu_out = st.future.result()  # poll() said this returns immediately
        st.future = None
        if st.h2d_event is None:
            st.h2d_event = torch.cuda.Event()
            st.consumed_event = torch.cuda.Event()
        # Don't overwrite buffers the compute stream may still be reading.
        self._h2d_stream.wait_event(st.consumed_event)
        with torch.cuda.stream(self._h2d_stream):
            if st.staged is None:
                st.staged = {
                    k: torch.empty(t.shape, dtype=t.dtype, device=self.cfg.device)
                    for k, t in cpu_out.items()
                }
            for k, t in cpu_out.items():
                st.staged[k].copy_(t, non_blocking=True)
            st.h2d_event.record()
        st.prefetched = cpu_out

    def act_once(self):
        cfg = self.cfg
        cur = self.next_env_index
        self.next_env_index = (self.next_env_index + 1) % cfg.num_actor_batches
        env_state = self.env_states[cur]
        t0 = time.perf_counter() if self.profile else 0.0
        if env_state.future is None and env_state.prefetched is None:
            env_state.future = self.envs.step(cur, env_state.prev_action)
        prefetched = env_state.prefetched is not None
        if prefetched:
            cpu_env_outputs = env_state.prefetched
            env_state.prefetched = None
        else:
            cpu_env_outputs = env_state.future.result()
        t0 = self._t("act_env_wait", t0)
        stream_ctx = (
            torch.cuda.stream(self.actor_stream)
            if self.actor_stream is not None
            else contextlib.nullcontext()
        )
        with stream_ctx:
            self._act_body(cfg, cur, env_state, cpu_env_outputs, t0, prefetched)

    def _act_body(self, cfg, cur, env_state, cpu_env_outputs, t0, prefetched=False):

        if prefetched:
            # results already live in the staged device buffers (copy stream)
            torch.cuda.current_stream().wait_event(env_state.h2d_event)
            env_outputs = dict(env_state.staged)
        elif self._shm_registered:
            # shm region is hipHostRegistered: direct async DMA, no bounce.
            env_outputs = {
                k: t.to(cfg.device, non_blocking=True) for k, t in cpu_env_outputs.items()
            }
        elif self.is_cuda and cfg.pinned_staging:
            # shm -> pinned -> HBM: the shm-backed views are pageable, so a
            # direct .to() is a slow synchronous copy; the pinned bounce
            # buffer makes the H2D leg an async DMA.
            env_outputs = {}
            for k, t in cpu_env_outputs.items():
                pin = env_state.pinned.get(k)
                if pin is None:
                    pin = torch.empty_like(t).pin_memory()
                    env_state.pinned[k] = pin
                pin.copy_(t)
                env_outputs[k] = pin.to(cfg.device, non_blocking=True)
        else:
            env_outputs = nest.map(
                lambda t: t.to(cfg.device, copy=True, non_blocking=True), cpu_env_outputs
            )
        env_outputs["prev_action"] = env_state.prev_action
        t0 = self._t("act_h2d", t0)
        use_lstm = cfg.use_lstm
        prev_core_state = (
            tuple(t.clone() for t in env_state.core_state) if use_lstm else tuple()
        )
        self.fwd_model.eval()
        res = self._actor_call({"env": env_outputs, "core": env_state.core_state})
        actor_outputs = nest.map(lambda t: t.squeeze(0), res["out"])
        if use_lstm:
            # keep persistent core buffers (stable addresses for graph replay)
            new_core = res["core"]
            for dst, src in zip(env_state.core_state, new_core):
                if dst.data_ptr() != src.data_ptr():
                    dst.copy_(src, non_blocking=True)
        action = actor_outputs["action"]
        t0 = self._t("act_forward", t0)
        env_state.update(cpu_env_outputs, action, self.stats)
        del cpu_env_outputs  # aliases shm; next step() overwrites it
        env_state.future = self.envs.step(cur, action)
        self.stats["env_act_steps"] += action.numel()
        t0 = self._t("act_stats_step", t0)

        last_data = {"env_outputs": env_outputs, "actor_outputs": actor_outputs}
        env_state.time_batcher.stack(last_data)
        if not env_state.time_batcher.empty():
            data = env_state.time_batcher.get()
            data["initial_core_state"] = env_state.initial_core_state
            n_before = self.learn_batcher.size()
            self.learn_batcher.cat(data)
            if self.actor_stream is not None:
                for _ in range(self.learn_batcher.size() - n_before):
                    ev = torch.cuda.Event()
                    ev.record(self.actor_stream)
                    self._batch_events.append(ev)
            # Carry the last entry of the previous unroll into the next one.
            env_state.initial_core_state = prev_core_state
            env_state.time_batcher.stack(last_data)
        if prefetched:
            # all reads of the staged buffers are now issued on this stream;
            # the next prefetch overwrite waits on this event
            env_state.consumed_event.record()
        self._t("act_batch", t0)
