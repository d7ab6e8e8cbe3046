"""RCCL (nccl backend) on real hardware — VERDICT r1 #1.

Round-1 left the RCCL path gloo-proven only.  These tests run on the
1-GPU lease:

* world_size=2 with BOTH ranks on device 0 (HIP_VISIBLE_DEVICES pinning
  makes each process see one device): real RCCL communicator init,
  bucketed all-reduce via the engine, broadcast, loss parity vs the
  emulated 2-shard reference — the gloo parity test's GPU twin.
* dist_async double-buffer cycling over RCCL.
* hipGraph capture of a step CONTAINING an RCCL all-reduce (ws=2), the
  unknown that decides whether the 8-GPU bench can run graphed.

If this RCCL build refuses two ranks on one device the tests record that
as a skip with the exact error, so the 8-GPU driver run is informed
either way.
"""

import json
import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

pytestmark = pytest.mark.gpu

PY = sys.executable
REPO = Path(__file__).resolve().parent.parent

WORKER = REPO / "tools" / "rccl_worker.py"


def _spawn_two_ranks(mode, out_dir, extra_env=None, timeout=240):
    """Two processes, both pinned to device 0, nccl rendezvous on
    127.0.0.1 (env contract: the container hostname may not resolve)."""
    procs = []
    port = 29711
    for rank in range(2):
        env = dict(os.environ)
        env.update({
            "RANK": str(rank), "LOCAL_RANK": str(rank), "WORLD_SIZE": "2",
            "MASTER_ADDR": "127.0.0.1", "MASTER_PORT": str(port),
            "HIP_VISIBLE_DEVICES": "0",
            "HSA_ENABLE_IPC_MODE_LEGACY": "0",
            "DLX_RCCL_MODE": mode,
            "DLX_RCCL_OUT": str(out_dir),
        })
        if extra_env:
            env.update(extra_env)
        procs.append(subprocess.Popen(
            [PY, str(WORKER)], env=env, cwd=str(REPO),
            stdout=subprocess.PIPE, stderr=subprocess.STDOUT, text=True))
    outs, rcs = [], []
    for p in procs:
        try:
            out, _ = p.communicate(timeout=timeout)
        except subprocess.TimeoutExpired:
            for q in procs:
                q.kill()
            raise
        outs.append(out)
        rcs.append(p.returncode)
    return rcs, outs


def _check_dup_gpu(outs):
    text = "\n".join(outs)
    for marker in ("Duplicate GPU", "duplicate GPU", "DUP_GPU_REFUSED"):
        if marker in text:
            pytest.skip("RCCL refuses two ranks on one device: "
                        + text[-500:])


def test_rccl_two_ranks_one_device_sync_parity(tmp_path):
    rcs, outs = _spawn_two_ranks("sync", tmp_path)
    _check_dup_gpu(outs)
    assert rcs == [0, 0], outs
    res = json.loads((tmp_path / "rank0.json").read_text())
    # parity vs the in-process emulated 2-shard reference computed by
    # the worker itself (same construction as test_train_parity)
    for a, b in zip(res["dp_losses"], res["ref_losses"]):
        assert abs(a - b) < 5e-3, res
    assert res["dp_losses"][0] != res["dp_losses"][-1]


def test_rccl_two_ranks_dist_async_cycles(tmp_path):
    rcs, outs = _spawn_two_ranks("async", tmp_path)
    _check_dup_gpu(outs)
    assert rcs == [0, 0], outs
    res = json.loads((tmp_path / "rank0.json").read_text())
    assert all(map(lambda x: x == x and abs(x) < 1e6, res["losses"]))


def test_rccl_graph_captured_allreduce_ws2(tmp_path):
    rcs, outs = _spawn_two_ranks("graph", tmp_path)
    _check_dup_gpu(outs)
    text = "\n".join(outs)
    if "GRAPH_CAPTURE_UNSUPPORTED" in text:
        pytest.skip("RCCL graph capture unsupported: " + text[-500:])
    assert rcs == [0, 0], outs
    res = json.loads((tmp_path / "rank0.json").read_text())
    # replayed graph must produce the same reduced values as eager
    assert res["graph_matches_eager"], res


def test_nccl_ws1_init_and_collectives():
    """world 1 over the real nccl backend: init, all_reduce, broadcast."""
    import torch.distributed as dist
    if dist.is_initialized():
        dist.destroy_process_group()
    store = dist.TCPStore("127.0.0.1", 29755, 1, True)
    dist.init_process_group("nccl", store=store, rank=0, world_size=1)
    try:
        t = torch.ones(1024, device="cuda")
        dist.all_reduce(t)
        dist.broadcast(t, src=0)
        torch.cuda.synchronize()
        assert t.sum().item() == 1024

        # hipGraph capture CONTAINING an RCCL collective (the ws>1
        # variant skips while two-ranks-one-device is refused; this at
        # least proves RCCL ops capture+replay on this stack)
        x = torch.ones(1 << 16, device="cuda")
        buf = torch.empty_like(x)

        def step():
            buf.copy_(x * 2.0)
            dist.all_reduce(buf)
            buf.mul_(0.25)

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                step()
        torch.cuda.current_stream().wait_stream(s)
        torch.cuda.synchronize()
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            step()
        for _ in range(3):
            g.replay()
        torch.cuda.synchronize()
        assert torch.allclose(buf, torch.full_like(buf, 0.5)), buf[:4]
    finally:
        dist.destroy_process_group()
