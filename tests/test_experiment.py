"""IMPALA experiment CLI: runs, logs, checkpoints, resumes (subprocess)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.join(os.path.dirname(__file__), "..")
EXP = os.path.join(ROOT, "examples", "impala", "experiment.py")

TINY = [
    "-o", "device=cpu", "-o", "actor_batch_size=8", "-o", "num_actor_cpus=2",
    "-o", "batch_size=4", "-o", "unroll_length=5", "-o", "virtual_batch_size=4",
    "-o", "total_steps=4000", "-o", "num_actions=6", "-o", "log_interval=1",
    "-o", "checkpoint_interval=1", "-o", "autocast_bf16=false",
]


@pytest.mark.timeout(300)
def test_experiment_runs_and_resumes(tmp_path):
    savedir = str(tmp_path / "run")
    env = dict(os.environ)
    env["PYTHONPATH"] = os.path.abspath(ROOT) + os.pathsep + env.get("PYTHONPATH", "")
    cmd = [sys.executable, EXP, "--broker", "--connect", "127.0.0.1:0",
           "--savedir", savedir, "--local-name", "p0"] + TINY
    out = subprocess.run(cmd, capture_output=True, text=True, timeout=240, cwd=ROOT, env=env)
    assert out.returncode == 0, out.stderr[-3000:]
    assert os.path.exists(os.path.join(savedir, "checkpoint.tar")), out.stderr[-2000:]
    assert os.path.exists(os.path.join(savedir, "peers", "p0", "logs.tsv"))
    assert os.path.islink(os.path.join(savedir, "leader-000"))

    # resume: loads the checkpoint, should exit immediately (steps done is
    # per-run stats, so give it a small extra budget and just check load)
    cmd2 = [sys.executable, EXP, "--broker", "--connect", "127.0.0.1:0",
            "--savedir", savedir, "--local-name", "p1"] + TINY
    out2 = subprocess.run(cmd2, capture_output=True, text=True, timeout=240, cwd=ROOT, env=env)
    assert out2.returncode == 0, out2.stderr[-3000:]
    assert "loaded checkpoint" in out2.stderr


class TestSlurmLauncher:
    def test_build_sbatch_structure(self):
        import sys
        sys.path.insert(0, "examples")
        import launch_slurm

        argv = launch_slurm.build_sbatch(
            8, "10.0.0.1:4431", "/save/dir", job_name="p/g", partition="mi355x"
        )
        assert argv[0] == "sbatch"
        assert argv[argv.index("--array") + 1] == "0-7"
        assert argv[argv.index("--gpus-per-task") + 1] == "1"
        assert argv[argv.index("--partition") + 1] == "mi355x"
        wrap = argv[argv.index("--wrap") + 1]
        assert "--connect 10.0.0.1:4431" in wrap
        assert "--savedir /save/dir" in wrap

    def test_broker_probe(self, tmp_path):
        import sys
        sys.path.insert(0, "examples")
        import launch_slurm

        import moolib_amd

        broker_rpc = moolib_amd.Rpc()
        broker_rpc.set_name("broker")
        addr = [a for a in broker_rpc.listen("127.0.0.1:0") if a.startswith("tcp://")][0]
        hostport = addr[len("tcp://"):]
        assert launch_slurm.broker_is_alive(hostport, timeout=5)
        assert not launch_slurm.broker_is_alive("127.0.0.1:1", timeout=1.5)

    def test_dry_run(self, tmp_path, capsys):
        import sys
        sys.path.insert(0, "examples")
        import launch_slurm

        rc = launch_slurm.main(
            ["-n", "2", "--dry", "--no-checks", "--savedir", str(tmp_path / "sv")]
        )
        assert rc == 0
        out = capsys.readouterr().out
        assert "sbatch" in out and "--array 0-1" in out
