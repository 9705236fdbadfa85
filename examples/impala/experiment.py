"""Elastic multi-peer IMPALA/V-trace experiment.

Capability parity with the reference's examples/vtrace/experiment.py:
hydra-free (yaml + argparse overrides), periodic + versioned checkpoints
with atomic rename, leader symlinks, warmup, local + cluster-global stats
with delta allreduce, SPS reporting, graceful SIGINT/SIGTERM handling,
elastic join/leave via the broker.

Run one peer:
    python examples/impala/experiment.py --connect 127.0.0.1:4431
(start a broker first: python -m moolib_amd.broker)
Multiple peers on one node: run once per GPU with --device cuda:N.
"""
import argparse
import dataclasses
import logging
import os
import signal
import time

import torch
import yaml

import moolib_amd
from moolib_amd.envs import SyntheticAtariEnv
from moolib_amd.impala import ImpalaConfig, ImpalaPeer
from moolib_amd.utils.record import log_to_file, symlink_path, write_metadata


def parse_args():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--config", default=os.path.join(os.path.dirname(__file__), "config.yaml"))
    ap.add_argument("--connect", default=None)
    ap.add_argument("--broker", action="store_true", help="host the broker in this process")
    ap.add_argument("--device", default=None)
    ap.add_argument("--savedir", default=None)
    ap.add_argument("--local-name", default=None)
    ap.add_argument("--total-steps", type=float, default=None)
    ap.add_argument("--use-lstm", action="store_true", default=None)
    ap.add_argument("-o", "--override", action="append", default=[], metavar="KEY=VALUE",
                    help="override any config key, e.g. -o batch_size=16")
    return ap.parse_args()


def _yaml_value(v):
    out = yaml.safe_load(v)
    # PyYAML (YAML 1.1) parses "2e9"/"6e-4" as STRINGS (needs "2.0e+9");
    # users write scientific notation constantly, so coerce numeric-looking
    # strings here.
    if isinstance(out, str):
        for cast in (int, float):
            try:
                return cast(out)
            except ValueError:
                pass
    return out


def load_flags(args):
    with open(args.config) as f:
        flags = yaml.safe_load(f)
    flags = {k: (_yaml_value(v) if isinstance(v, str) else v) for k, v in flags.items()}
    for kv in args.override:
        k, v = kv.split("=", 1)
        flags[k] = _yaml_value(v)
    for k in ("connect", "device", "savedir", "total_steps"):
        v = getattr(args, k.replace("-", "_"), None)
        if v is not None:
            flags[k] = v
    if args.use_lstm:
        flags["use_lstm"] = True
    if args.local_name:
        flags["local_name"] = args.local_name
    flags.setdefault("local_name", "peer-%s" % moolib_amd.create_uid()[:8])
    return flags


def save_checkpoint(path, peer, flags):
    """Atomic tmp+rename, same layout as the reference (experiment.py:186)."""
    tmp = "%s.tmp.%s" % (path, moolib_amd.create_uid()[:8])
    checkpoint = {
        "learner_state": peer.save_state(),
        "flags": flags,
    }
    os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
    torch.save(checkpoint, tmp)
    os.replace(tmp, path)
    logging.info("checkpoint saved to %s", path)


def load_checkpoint(path, peer):
    checkpoint = torch.load(path, map_location="cpu", weights_only=False)
    state = checkpoint["learner_state"]
    peer.model.load_state_dict(state["model"])
    peer.load_state(state)
    peer._sync_shadow()
    peer.accumulator.set_model_version(state["model_version"])
    logging.info("loaded checkpoint %s at model version %d", path, state["model_version"])


def main():
    args = parse_args()
    flags = load_flags(args)
    logging.basicConfig(
        level=logging.INFO,
        format="[%(levelname)s:" + flags["local_name"] + " %(module)s:%(lineno)d %(asctime)s] %(message)s",
    )
    logging.info("flags: %s", flags)

    savedir = flags.get("savedir") or "./impala_runs/%s" % flags.get("group", "group")
    localdir = os.path.join(savedir, "peers", flags["local_name"])
    os.makedirs(localdir, exist_ok=True)
    write_metadata(localdir, flags=flags)
    checkpoint_path = os.path.join(savedir, "checkpoint.tar")

    cfg_fields = {f.name for f in dataclasses.fields(ImpalaConfig)}
    cfg = ImpalaConfig(**{k: v for k, v in flags.items() if k in cfg_fields})

    broker = None
    if args.broker:
        broker_rpc = moolib_amd.Rpc()
        broker_rpc.set_name("broker")
        broker = moolib_amd.Broker(broker_rpc)
        bound = broker_rpc.listen(cfg.connect)
        cfg.connect = ([a for a in bound if a.startswith("tcp://127")] or bound)[0]
        logging.info("hosting broker at %s", cfg.connect)

    peer = ImpalaPeer(
        cfg,
        create_env=lambda: SyntheticAtariEnv(num_actions=cfg.num_actions),
        broker=broker,
    )

    if os.path.exists(checkpoint_path):
        load_checkpoint(checkpoint_path, peer)

    terminate = []
    for sig in (signal.SIGINT, signal.SIGTERM):
        signal.signal(sig, lambda s, f: terminate.append(s))

    log_interval = flags.get("log_interval", 10)
    checkpoint_interval = flags.get("checkpoint_interval", 600)
    checkpoint_history_interval = flags.get("checkpoint_history_interval", 3600)

    now = time.time()
    last_log = now
    last_reduce = now
    train_time = 0.0
    last_checkpoint = 0.0
    last_checkpoint_history = 0.0
    prev_steps = 0
    was_leader = False
    num_previous_leaders = 0

    while not terminate:
        prev_now, now = now, time.time()
        steps = peer.stats["env_train_steps"].result()
        if steps >= cfg.total_steps:
            logging.info("stopping after %d steps", steps)
            break

        ev = peer.step_once()
        if ev == "idle":
            continue
        train_time += now - prev_now

        is_leader = peer.accumulator.is_leader()
        if is_leader and not was_leader:
            link = os.path.join(savedir, "leader-%03d" % num_previous_leaders)
            symlink_path(localdir, link)
            num_previous_leaders += 1
        was_leader = is_leader

        if now - last_reduce >= 2:
            last_reduce = now
            peer.global_stats_accumulator.reduce(peer.stats)

        if now - last_log >= log_interval:
            delta = now - last_log
            last_log = now
            peer.global_stats_accumulator.reduce(peer.stats)
            peer.global_stats_accumulator.reset()
            cur_steps = peer.stats["env_train_steps"].result()
            sps = (cur_steps - prev_steps) / delta
            prev_steps = cur_steps
            peer.stats["SPS"] += sps
            values = {k: v.result() for k, v in peer.stats.items()}
            logging.info("steps %d SPS %.0f stats %s", cur_steps, sps, values)
            log_to_file(
                os.path.join(localdir, "logs.tsv"),
                step=cur_steps,
                sps=sps,
                **{k: ("" if v is None else v) for k, v in values.items()},
            )

        if is_leader and train_time - last_checkpoint >= checkpoint_interval:
            last_checkpoint = train_time
            save_checkpoint(checkpoint_path, peer, flags)
        if is_leader and train_time - last_checkpoint_history >= checkpoint_history_interval:
            last_checkpoint_history = train_time
            save_checkpoint(
                os.path.join(savedir, "checkpoint_v%d.tar" % peer.model_version), peer, flags
            )

    if peer.accumulator.connected() and peer.accumulator.is_leader():
        save_checkpoint(checkpoint_path, peer, flags)
    logging.info("graceful exit")


if __name__ == "__main__":
    main()
