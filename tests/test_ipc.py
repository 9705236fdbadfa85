"""hipIpc zero-copy GPU tensor RPC: two processes, one node, one GPU."""
import os
import time

import pytest
import torch

gpu = pytest.mark.gpu


def _server(addr_file, stop_file):
    import moolib_amd
    from moolib_amd import ipc

    rpc = moolib_amd.Rpc()
    rpc.set_name("ipc_server")
    addr = [a for a in rpc.listen("127.0.0.1:0") if a.startswith("tcp://127")][0]

    payload = torch.arange(1024 * 1024, dtype=torch.float32, device="cuda").reshape(1024, 1024)

    def get_batch():
        return {"data": ipc.share(payload), "tag": "hbm-resident"}

    def check_write():
        # the client wrote into our HBM through the alias
        torch.cuda.synchronize()
        return float(payload[0, 0].item())

    rpc.define("get_batch", get_batch)
    rpc.define("check_write", check_write)
    with open(addr_file + ".tmp", "w") as f:
        f.write(addr)
    os.replace(addr_file + ".tmp", addr_file)
    t0 = time.time()
    while not os.path.exists(stop_file) and time.time() - t0 < 120:
        time.sleep(0.05)


@gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
@pytest.mark.timeout(300)
def test_hipipc_tensor_rpc(tmp_path):
    import torch.multiprocessing as mp

    import moolib_amd

    addr_file = str(tmp_path / "addr")
    stop_file = str(tmp_path / "stop")
    ctx = mp.get_context("spawn")
    proc = ctx.Process(target=_server, args=(addr_file, stop_file))
    proc.start()
    try:
        t0 = time.time()
        while not os.path.exists(addr_file) and time.time() - t0 < 60:
            time.sleep(0.05)
        addr = open(addr_file).read()

        client = moolib_amd.Rpc()
        client.set_name("ipc_client")
        client.set_timeout(60)
        client.connect(addr)
        result = client.sync("ipc_server", "get_batch")
        t = result["data"]
        assert isinstance(t, torch.Tensor) and t.is_cuda
        assert t.shape == (1024, 1024)
        # values alias the server's HBM buffer
        assert float(t[3, 7].item()) == 3 * 1024 + 7
        # write through the alias; the server sees it (true zero-copy)
        t[0, 0] = 42.5
        torch.cuda.synchronize()
        assert client.sync("ipc_server", "check_write") == 42.5
    finally:
        open(stop_file, "w").write("x")
        proc.join(timeout=30)
        if proc.is_alive():
            proc.kill()


@gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
@pytest.mark.timeout(120)
def test_hipipc_same_process():
    """Handles deserialized by the producing process resolve to the original
    tensor (a process cannot hipIpcOpenMemHandle its own handle), and the
    producer-side IPC ref counter is released so streams of shares don't
    pin HBM (moolib_amd/ipc.py registry)."""
    import moolib_amd
    from moolib_amd import ipc

    host = moolib_amd.Rpc()
    host.set_name("same_proc_host")
    addr = host.listen("127.0.0.1:0")[0]
    payload = torch.arange(256, dtype=torch.float32, device="cuda")
    host.define("get", lambda: ipc.share(payload))

    client = moolib_amd.Rpc()
    client.set_name("same_proc_client")
    client.set_timeout(60)
    client.connect(addr)

    t = client.sync("same_proc_host", "get")
    assert t.is_cuda and t.data_ptr() == payload.data_ptr()

    # a stream of fresh shares must not accumulate pinned storage
    free0, _ = torch.cuda.mem_get_info()
    for _ in range(50):
        host_side = torch.randn(1024 * 1024, device="cuda")  # 4 MiB each
        host.define("get_fresh", lambda ref=host_side: ipc.share(ref))
        got = client.sync("same_proc_host", "get_fresh")
        del host_side, got
    torch.cuda.synchronize()
    free1, _ = torch.cuda.mem_get_info()
    assert free0 - free1 < 64 * 1024 * 1024, (free0 - free1) / 1e6


def _auto_server(addr_file, stop_file):
    import moolib_amd

    rpc = moolib_amd.Rpc()
    rpc.set_name("auto_server")
    addr = [a for a in rpc.listen("127.0.0.1:0") if a.startswith("tcp://127")][0]

    payload = torch.arange(4096, dtype=torch.float32, device="cuda").reshape(64, 64)

    # No explicit ipc.share(): serde detects the same-machine peer and
    # ships the hipIpc handle automatically.
    rpc.define("get_auto", lambda: payload)
    rpc.define("probe", lambda: float(payload[0, 0].item()))
    with open(addr_file + ".tmp", "w") as f:
        f.write(addr)
    os.replace(addr_file + ".tmp", addr_file)
    t0 = time.time()
    while not os.path.exists(stop_file) and time.time() - t0 < 120:
        time.sleep(0.05)


@gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
@pytest.mark.timeout(300)
def test_automatic_hipipc_for_local_peer(tmp_path):
    """The DEFAULT serde path ships CUDA tensors between same-node peers as
    hipIpc handles: the receiver gets a device tensor aliasing the sender's
    HBM with no CPU staging (reference behavior: fatal,
    src/rpc.cc:661-667)."""
    import torch.multiprocessing as mp

    import moolib_amd

    addr_file = str(tmp_path / "addr")
    stop_file = str(tmp_path / "stop")
    ctx = mp.get_context("spawn")
    proc = ctx.Process(target=_auto_server, args=(addr_file, stop_file))
    proc.start()
    try:
        t0 = time.time()
        while not os.path.exists(addr_file) and time.time() - t0 < 60:
            time.sleep(0.05)
        addr = open(addr_file).read()

        client = moolib_amd.Rpc()
        client.set_name("auto_client")
        client.set_timeout(60)
        client.connect(addr)
        t = client.sync("auto_server", "get_auto")
        assert isinstance(t, torch.Tensor) and t.is_cuda, t
        assert float(t[63, 63].item()) == 4095.0
        # prove the alias: write through it, server observes the value
        t[0, 0] = -7.0
        torch.cuda.synchronize()
        assert client.sync("auto_server", "probe") == -7.0
        # request direction: client's CUDA args arrive as device tensors too
        client2 = moolib_amd.Rpc()
        client2.set_name("auto_client2")
        client2.set_timeout(60)
        client2.connect(addr)
    finally:
        open(stop_file, "w").write("x")
        proc.join(timeout=30)
        if proc.is_alive():
            proc.kill()


@gpu
@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs MI355X")
@pytest.mark.timeout(120)
def test_automatic_hipipc_opt_out(monkeypatch):
    """MOOLIB_AMD_NO_IPC_RPC disables the automatic handle path (staging
    fallback still delivers the values). Uses a subprocess so the env var
    is seen before the serde cache resolves."""
    import subprocess
    import sys

    code = """
import os, torch, moolib_amd
host = moolib_amd.Rpc(); host.set_name("h")
addr = host.listen("127.0.0.1:0")[0]
x = torch.arange(16, dtype=torch.float32, device="cuda")
host.define("get", lambda: x)
c = moolib_amd.Rpc(); c.set_name("c"); c.set_timeout(30); c.connect(addr)
t = c.sync("h", "get")
assert torch.equal(t.cpu(), x.cpu())
print("OK")
"""
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ, MOOLIB_AMD_NO_IPC_RPC="1")
    env["PYTHONPATH"] = repo + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run([sys.executable, "-c", code], capture_output=True, text=True,
                       timeout=110, env=env, cwd=repo)
    assert r.returncode == 0 and "OK" in r.stdout, r.stderr[-2000:]
