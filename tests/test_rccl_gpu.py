"""RCCL-backend GPU tests (VERDICT round-1 item 1).

Round 1 never executed an RCCL collective on hardware (the gloo world-2
CPU tests were the only coverage).  These run in a single-GPU lease:

* world-1 ``nccl`` process-group init + every fanout primitive on
  cuda:0 — exercises RCCL init, dtype handling, and stream interaction
  with the landing pipeline (world-1 collectives still launch RCCL
  kernels);
* ``torch.distributed.run --nproc-per-node 1`` over bench.py's
  collective modes — the exact launch shape the driver's 8-GPU SCALE
  run uses, so rendezvous/env handling is validated before round end.
"""

import json
import os
import socket
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.gpu


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.fixture()
def nccl_world1():
    import torch.distributed as dist

    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.cuda.set_device(0)
    dist.init_process_group(
        "nccl", init_method=f"tcp://127.0.0.1:{_free_port()}",
        rank=0, world_size=1,
        device_id=torch.device("cuda", 0))
    yield dist
    dist.destroy_process_group()


def test_rccl_world1_fanout_primitives(nccl_world1):
    from demodel_amd.parallel.fanout import (broadcast_blob,
                                             range_sharded_allgather,
                                             shard_assignment,
                                             sharded_pull_fanout)

    n = 3 << 20
    t = (torch.arange(n, dtype=torch.int64) % 251).to(torch.uint8).cuda()
    broadcast_blob(t, src=0, bucket_bytes=1 << 20)
    torch.cuda.synchronize()
    want = (torch.arange(n, dtype=torch.int64) % 251).to(torch.uint8)
    assert torch.equal(t.cpu(), want)

    files = [("a.bin", 300_000), ("b.bin", 100_000)]
    plan = shard_assignment(files, 1, 0)

    def pull_one(name):
        nb = dict(files)[name]
        return torch.full((nb,), fill_value=len(name) % 251,
                          dtype=torch.uint8, device="cuda")

    out = sharded_pull_fanout(
        plan, pull_one,
        lambda nb: torch.zeros(nb, dtype=torch.uint8, device="cuda"),
        bucket_bytes=128 << 10)
    assert {k: v.numel() for k, v in out.items()} == dict(files)

    total = 1_000_000
    blob = (torch.arange(total, dtype=torch.int64) * 7 % 253
            ).to(torch.uint8).cuda()

    def pull_range(lo, want_n, dest, bucket_done):
        bucket = 100_000
        done = 0
        flushed = 0
        while done < want_n:
            take = min(bucket, want_n - done)
            dest[done:done + take] = blob[lo + done:lo + done + take]
            done += take
            while (flushed + 1) * bucket <= done:
                bucket_done(flushed)
                flushed += 1
        while flushed * bucket < want_n:
            bucket_done(flushed)
            flushed += 1

    full = range_sharded_allgather(
        total, pull_range,
        lambda n2: torch.zeros(n2, dtype=torch.uint8, device="cuda"),
        bucket_bytes=100_000)
    torch.cuda.synchronize()
    assert torch.equal(full[:total].cpu(), blob.cpu())


@pytest.mark.parametrize("mode", ["broadcast", "shard", "allgather"])
def test_torchrun_bench_collective_modes_world1(mode, tmp_path):
    """The driver's exact launch shape, nproc=1: must produce one valid
    JSON line per run with no code changes."""
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    env["MASTER_PORT"] = str(_free_port())
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "1", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()),
         os.path.join(REPO, "bench.py"),
         "--model", "tiny", "--mode", mode, "--steps", "1",
         "--warmup", "1",
         "--data-dir", str(tmp_path / "tinydata")],
        cwd=REPO, capture_output=True, text=True, timeout=600, env=env)
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-2000:])
    line = [ln for ln in r.stdout.splitlines()
            if ln.startswith("{")][-1]
    out = json.loads(line)
    assert out["metric"] == "pull_gbps_into_hbm"
    assert out["value"] > 0
    assert out["n_gpus"] == 1


def test_torchrun_bench_world2_nccl(tmp_path):
    """Real 2-GPU RCCL run of the allgather mode — auto-skips on 1-GPU
    boxes (collective tests gated on visible devices)."""
    if torch.cuda.device_count() < 2:
        pytest.skip("needs >=2 GPUs")
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()),
         os.path.join(REPO, "bench.py"),
         "--model", "tiny", "--mode", "allgather", "--steps", "1",
         "--warmup", "1",
         "--data-dir", str(tmp_path / "tinydata")],
        cwd=REPO, capture_output=True, text=True, timeout=600, env=env)
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-2000:])
    out = json.loads([ln for ln in r.stdout.splitlines()
                      if ln.startswith("{")][-1])
    assert out["n_gpus"] == 2 and out["value"] > 0


def test_bench_dp_world1_direct(tmp_path):
    """Plain `python bench.py` (the driver's N=1 BENCH call) with
    scatter wired into the timed step."""
    r = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"),
         "--model", "tiny", "--steps", "2", "--warmup", "1",
         "--data-dir", str(tmp_path / "tinydata")],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, (r.stdout[-2000:], r.stderr[-2000:])
    line = [ln for ln in r.stdout.splitlines()
            if ln.startswith("{")][-1]
    out = json.loads(line)
    assert out["config"]["scatter"] is True
    assert out["value"] > 0
