"""Launch a multi-peer IMPALA run on one node: a broker + one experiment
peer per GPU.

Capability parity with the reference's examples/sbatch_experiment.py
(which submits one broker + N peers to SLURM). On a SLURM cluster the
equivalent submission is:

    #SBATCH --ntasks=N --gpus-per-task=1
    srun --ntasks=1 python -m moolib_amd.broker 0.0.0.0:4431 &
    srun python examples/impala/experiment.py --connect head-node:4431 \
         --device cuda:$SLURM_LOCALID

Usage: python examples/launch_local.py --gpus 8 [-- experiment args...]
"""
import argparse
import os
import signal
import subprocess
import sys
import time

HERE = os.path.dirname(os.path.abspath(__file__))


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--port", type=int, default=4431)
    ap.add_argument("--savedir", default="./impala_runs/local")
    ap.add_argument("rest", nargs=argparse.REMAINDER)
    args = ap.parse_args()
    extra = [a for a in args.rest if a != "--"]

    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.dirname(HERE) + os.pathsep + env.get("PYTHONPATH", "")

    procs = []
    broker = subprocess.Popen(
        [sys.executable, "-m", "moolib_amd.broker", "127.0.0.1:%d" % args.port], env=env
    )
    procs.append(broker)
    time.sleep(1.0)
    try:
        for i in range(args.gpus):
            cmd = [
                sys.executable,
                os.path.join(HERE, "impala", "experiment.py"),
                "--connect", "127.0.0.1:%d" % args.port,
                "--device", "cuda:%d" % i,
                "--savedir", args.savedir,
                "--local-name", "peer%d" % i,
            ] + extra
            procs.append(subprocess.Popen(cmd, env=env))
        for p in procs[1:]:
            p.wait()
    finally:
        for p in procs:
            if p.poll() is None:
                p.send_signal(signal.SIGINT)
        for p in procs:
            try:
                p.wait(timeout=30)
            except subprocess.TimeoutExpired:
                p.kill()


if __name__ == "__main__":
    main()
