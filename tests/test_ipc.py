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
