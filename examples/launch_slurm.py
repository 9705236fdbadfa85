#!/usr/bin/env python
"""Submit a multi-peer IMPALA experiment to SLURM (one peer per array task).

Capability parity with the reference's examples/sbatch_experiment.py:
a job array of N peers, 1 GPU + num_actor_cpus CPUs per task, all joining
one broker; --dry prints the sbatch command instead of submitting. Design
is our own: the sbatch command is assembled by build_sbatch() (unit-tested
without SLURM) and the broker liveness probe uses the RPC error taxonomy
(a "does not exist" reply proves a live broker; a timeout proves none).

Typical use (broker already running on the submit host):
    python -m moolib_amd.broker 0.0.0.0:4431 &
    python examples/launch_slurm.py -n 8 --partition mi355x
"""
import argparse
import os
import socket
import subprocess
import sys
import uuid

DEFAULT_PORT = 4431


def default_broker_address():
    """The submit host's routable address — peers on other nodes must be
    able to dial it."""
    try:
        return "%s:%d" % (socket.gethostbyname(socket.gethostname()), DEFAULT_PORT)
    except socket.gaierror:
        return "127.0.0.1:%d" % DEFAULT_PORT


def broker_is_alive(address, timeout=3.0):
    """True if an Rpc peer named 'broker' answers at `address`.

    Calling an undefined function on a LIVE peer returns a remote
    'does not exist' error; only an unreachable peer times out."""
    import moolib_amd

    rpc = moolib_amd.Rpc()
    rpc.set_name("launch-probe-" + uuid.uuid4().hex[:8])
    rpc.set_timeout(timeout)
    rpc.connect(address)
    try:
        rpc.sync("broker", "__launch_probe__")
        return True  # unexpectedly defined — still alive
    except Exception as e:  # noqa: BLE001
        return "does not exist" in str(e)


def build_sbatch(
    num_peers,
    broker,
    savedir,
    job_name,
    partition="",
    constraint="",
    time_min=1440,
    cpus_per_task=10,
    mem_per_cpu="8G",
    peer_cmd=None,
):
    """Assemble the sbatch argv for a job array of IMPALA peers."""
    if peer_cmd is None:
        peer_cmd = (
            "python examples/impala/experiment.py"
            " --connect {broker} --savedir {savedir}"
            " --local-name peer$SLURM_ARRAY_TASK_ID"
        )
    wrap = peer_cmd.format(broker=broker, savedir=savedir)
    out = os.path.join(savedir, "slurm-%A_%a.out")
    argv = ["sbatch", "--job-name", job_name, "--array", "0-%d" % (num_peers - 1)]
    if partition:
        argv += ["--partition", partition]
    if constraint:
        argv += ["--constraint", constraint]
    argv += [
        "--ntasks", "1",
        "--gpus-per-task", "1",
        "--cpus-per-task", str(cpus_per_task),
        "--mem-per-cpu", mem_per_cpu,
        "--time", str(time_min),
        "--output", out,
        "--error", out,
        "--export", "ALL",
        "--wrap", wrap,
    ]
    return argv


def main(argv=None):
    ap = argparse.ArgumentParser(description=__doc__.splitlines()[0])
    ap.add_argument("-n", "--num-peers", type=int, default=1)
    ap.add_argument("--broker", default="", help="broker addr:port (default: this host)")
    ap.add_argument("--project", default="moolib-amd-atari")
    ap.add_argument("--group", default=uuid.uuid4().hex[:8], help="run group name")
    ap.add_argument("--partition", default="")
    ap.add_argument("--constraint", default="")
    ap.add_argument("--time", type=int, default=1440, help="minutes")
    ap.add_argument("--cpus-per-task", type=int, default=10)
    ap.add_argument("--mem-per-cpu", default="8G")
    ap.add_argument("--savedir", default="", help="default: ./impala_runs/<project>/<group>")
    ap.add_argument("--peer-cmd", default=None, help="override the per-peer command")
    ap.add_argument("--dry", action="store_true", help="print, do not submit")
    ap.add_argument("--no-checks", action="store_true")
    args = ap.parse_args(argv)

    broker = args.broker or default_broker_address()
    savedir = args.savedir or os.path.join("impala_runs", args.project, args.group)
    os.makedirs(savedir, exist_ok=True)

    if not args.no_checks and not broker_is_alive(broker):
        print(
            "no broker answering at %s — start one first:\n"
            "    python -m moolib_amd.broker 0.0.0.0:%d" % (broker, DEFAULT_PORT),
            file=sys.stderr,
        )
        return 1

    argv_out = build_sbatch(
        args.num_peers,
        broker,
        savedir,
        job_name="%s/%s" % (args.project, args.group),
        partition=args.partition,
        constraint=args.constraint,
        time_min=args.time,
        cpus_per_task=args.cpus_per_task,
        mem_per_cpu=args.mem_per_cpu,
        peer_cmd=args.peer_cmd,
    )
    print(" ".join(repr(a) if " " in a else a for a in argv_out))
    if args.dry:
        return 0
    return subprocess.call(argv_out)


if __name__ == "__main__":
    sys.exit(main())
