"""RCCL/gloo fan-out tests — multi-process (world_size=2, gloo backend)
on CPU, per SURVEY.md §4's distributed-test plan."""

import hashlib
import os
import socket

import pytest
import torch

from demodel_amd.parallel.fanout import shard_assignment


def test_shard_assignment_balanced():
    files = [(f"f{i}", (i + 1) * 1000) for i in range(10)]
    plans = [shard_assignment(files, 4, r) for r in range(4)]
    # every file owned exactly once, order preserved
    assert list(plans[0].owners) == [f"f{i}" for i in range(10)]
    loads = [0] * 4
    for name, (owner, nb) in plans[0].owners.items():
        loads[owner] += nb
    assert max(loads) - min(loads) <= max(nb for _, nb in files)
    # my_files consistent with owners
    for r, p in enumerate(plans):
        assert all(p.owners[n][0] == r for n in p.my_files)
        assert p.owners == plans[0].owners


def test_broadcast_order_deterministic_round_robin():
    from demodel_amd.parallel.fanout import broadcast_order

    files = [(f"f{i}", (7 * i + 3) % 11 * 1000 + 100) for i in range(9)]
    plans = [shard_assignment(files, 4, r) for r in range(4)]
    orders = [broadcast_order(p) for p in plans]
    # identical on every rank (NCCL requires identical enqueue order)
    assert all(o == orders[0] for o in orders)
    assert sorted(orders[0]) == sorted(n for n, _ in files)
    # round 0 has one file per owner that owns anything
    owners = plans[0].owners
    n_owners = len({r for r, _ in owners.values()})
    head = [owners[n][0] for n in orders[0][:n_owners]]
    assert len(set(head)) == n_owners


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, world, port, tmpdir):
    import torch.distributed as dist

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world)
    try:
        from demodel_amd.parallel.fanout import (broadcast_blob,
                                                 shard_assignment,
                                                 sharded_pull_fanout)

        # --- bucketed broadcast ---
        n = 3 << 20  # 3 MiB, several 1 MiB buckets
        if rank == 0:
            t = torch.arange(n, dtype=torch.int64) % 251
            t = t.to(torch.uint8)
        else:
            t = torch.zeros(n, dtype=torch.uint8)
        broadcast_blob(t, src=0, bucket_bytes=1 << 20)
        want = (torch.arange(n, dtype=torch.int64) % 251).to(torch.uint8)
        assert torch.equal(t, want), "broadcast content mismatch"

        # --- sharded pull + fan-out ---
        def content(name, nbytes):
            seed = int.from_bytes(
                hashlib.sha256(name.encode()).digest()[:4], "little")
            g = torch.Generator().manual_seed(seed)
            return torch.randint(0, 256, (nbytes,), generator=g,
                                 dtype=torch.uint8)

        files = [("a.bin", 100_000), ("b.bin", 400_000),
                 ("c.bin", 50_000), ("d.bin", 250_000)]
        plan = shard_assignment(files, world, rank)

        def pull_one(name):
            nb = dict(files)[name]
            return content(name, nb)

        out = sharded_pull_fanout(
            plan, pull_one, lambda nb: torch.zeros(nb, dtype=torch.uint8),
            bucket_bytes=128 << 10)
        assert set(out) == {n for n, _ in files}
        for name, nb in files:
            assert torch.equal(out[name], content(name, nb)), name
        # --- byte-range shards + bucketed all-gather (R2) ---
        from demodel_amd.parallel.fanout import range_sharded_allgather

        total = 1_000_000  # not divisible by bucket or world
        blob = (torch.arange(total, dtype=torch.int64) * 7 % 253
                ).to(torch.uint8)

        def pull_range(lo, want, dest, bucket_done):
            bucket = 100_000
            done = 0
            flushed = 0
            while done < want:
                take = min(bucket, want - done)
                dest[done:done + take] = blob[lo + done:lo + done + take]
                done += take
                while (flushed + 1) * bucket <= done:
                    bucket_done(flushed)
                    flushed += 1
            while flushed * bucket < want:
                bucket_done(flushed)
                flushed += 1

        full = range_sharded_allgather(
            total, pull_range,
            lambda n2: torch.zeros(n2, dtype=torch.uint8),
            bucket_bytes=100_000)
        assert torch.equal(full[:total], blob), "allgather reassembly"

        # write a success marker per rank
        with open(os.path.join(tmpdir, f"ok{rank}"), "w") as f:
            f.write("ok")
    finally:
        dist.destroy_process_group()


def test_sharded_fanout_gloo_world2(tmp_path):
    port = _free_port()
    torch.multiprocessing.spawn(
        _worker, args=(2, port, str(tmp_path)), nprocs=2, join=True)
    assert (tmp_path / "ok0").exists() and (tmp_path / "ok1").exists()


def _worker_http_resume(rank, world, port, tmpdir):
    """Sharded fan-out where each rank's FIRST owned blob drops
    mid-stream: Range-resume must complete inside the collective run
    (the realistic multi-GPU failure mode — one flaky origin conn must
    not poison the broadcast schedule)."""
    import torch.distributed as dist

    dist.init_process_group(
        "gloo", init_method=f"tcp://127.0.0.1:{port}",
        rank=rank, world_size=world)
    try:
        from demodel_amd.engine.pull import LanderPool, _pull_blob
        from demodel_amd.parallel.fanout import (shard_assignment,
                                                 sharded_pull_fanout)
        from demodel_amd.testing.origin import FakeOrigin
        from helpers import LoopThread

        names = sorted(n for n in os.listdir(tmpdir)
                       if n.endswith(".bin"))
        files = {n: os.path.join(tmpdir, n) for n in names}
        sizes = [(n, os.path.getsize(p)) for n, p in files.items()]
        lt = LoopThread()
        origin = FakeOrigin(tmpdir)
        origin.add_hf_repo("org/m", files)
        oport = lt.call(origin.start())
        landers = LanderPool(0, gpu=False,
                             slab_bytes=64 << 10)  # small slabs: real
        plan = shard_assignment(sizes, world, rank)  # resume mid-file
        if plan.my_files:
            origin.drop_once[plan.my_files[0]] = 100 << 10

        def pull_one(name):
            pf = _pull_blob(
                landers, name,
                f"http://127.0.0.1:{oport}/org/m/resolve/main/{name}",
                None, "chunked", None, False)
            return pf.blob.torch_u8()

        def alloc(nb):
            return torch.zeros(nb, dtype=torch.uint8)

        out = sharded_pull_fanout(plan, pull_one, alloc,
                                  bucket_bytes=128 << 10)
        assert not origin.drop_once  # the fault fired on this rank
        for n, p in files.items():
            want = open(p, "rb").read()
            assert bytes(out[n].numpy().tobytes()) == want, n
        with open(os.path.join(tmpdir, f"okr{rank}"), "w") as f:
            f.write("ok")
        lt.call(origin.close())
        lt.stop()
    finally:
        dist.destroy_process_group()


def test_fanout_survives_mid_pull_drops_world2(tmp_path):
    rng = __import__("random").Random(5)
    for i in range(4):
        (tmp_path / f"m{i}.bin").write_bytes(
            bytes(rng.getrandbits(8) for _ in range(300_000 + i * 70_000)))
    port = _free_port()
    torch.multiprocessing.spawn(
        _worker_http_resume, args=(2, port, str(tmp_path)), nprocs=2,
        join=True)
    assert (tmp_path / "okr0").exists() and (tmp_path / "okr1").exists()
