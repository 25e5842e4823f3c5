"""bench.py contract tests: the driver launches it directly (N=1) and
under torch.distributed.run (N>1, one rank per GPU) — keep both paths
green on CPU so round-end GPU runs don't hit launcher bugs."""

import json
import os
import socket
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _check_line(line: str, world: int):
    out = json.loads(line)
    assert out["metric"] == "pull_gbps_into_hbm"
    assert out["n_gpus"] == world
    assert out["unit"] == "GB/s"
    assert out["higher_is_better"] is True
    assert out["value"] > 0
    assert out["data"] == "synthetic"
    assert "ms_per_step" in out and out["ms_per_step"] > 0
    assert "config" in out and "model" in out["config"]
    return out


def test_bench_single_process(tmp_path):
    r = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--steps", "2",
         "--warmup", "1", "--data-dir", str(tmp_path / "d")],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert len(line) == 1, r.stdout  # EXACTLY one JSON line
    _check_line(line[0], world=1)


def test_bench_via_proxy(tmp_path):
    """--via proxy: engine pulls THROUGH the demodel proxy (primed
    cache) — the client-facing data-plane measurement."""
    r = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--steps", "1",
         "--warmup", "1", "--via", "proxy",
         "--data-dir", str(tmp_path / "d")],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    out = _check_line(line[0], world=1)
    assert out["config"]["via"] == "proxy"


def test_bench_via_proxy_miss(tmp_path):
    r = subprocess.run(
        [sys.executable, "bench.py", "--model", "tiny", "--steps", "1",
         "--warmup", "1", "--via", "proxy-miss",
         "--data-dir", str(tmp_path / "d")],
        cwd=REPO, capture_output=True, text=True, timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    out = _check_line(line[0], world=1)
    assert out["config"]["via"] == "proxy-miss"


def test_bench_torchrun_world2(tmp_path):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), "bench.py", "--gpus", "2",
         "--model", "tiny", "--steps", "2", "--warmup", "1",
         "--data-dir", str(tmp_path / "d")],
        cwd=REPO, capture_output=True, text=True, timeout=300, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert len(line) == 1, r.stdout
    _check_line(line[0], world=2)


def test_bench_torchrun_shard_mode(tmp_path):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), "bench.py", "--gpus", "2",
         "--model", "tiny", "--mode", "shard", "--steps", "1",
         "--warmup", "1", "--data-dir", str(tmp_path / "d")],
        cwd=REPO, capture_output=True, text=True, timeout=300, env=env)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    out = _check_line(line[0], world=2)
    assert out["scaling"] == "strong"


def test_demodel_pull_cli(tmp_path):
    """`demodel pull hf://...` end-to-end via subprocess."""
    sys.path.insert(0, os.path.join(REPO, "tests"))
    from helpers import Stack

    stack = Stack(tmp_path)
    try:
        blob = tmp_path / "w.bin"
        blob.write_bytes(os.urandom(50_000))
        stack.origin.add_hf_repo("org/cli", {"w.bin": str(blob)})
        out_dir = tmp_path / "out"
        r = subprocess.run(
            [sys.executable, "-m", "demodel_amd", "pull", "hf://org/cli",
             "--cpu", "--endpoint", stack.origin_base,
             "--out", str(out_dir)],
            cwd=REPO, capture_output=True, text=True, timeout=120)
        assert r.returncode == 0, r.stderr[-1500:]
        result = json.loads(r.stdout)
        assert result["total_bytes"] == 50_000
        assert (out_dir / "w.bin").read_bytes() == blob.read_bytes()
    finally:
        stack.close()


def test_demodel_stats_cli(tmp_path, capsys):
    """`demodel stats` prints a running proxy's transfer metrics."""
    sys.path.insert(0, os.path.join(REPO, "tests"))
    from helpers import Stack

    from demodel_amd.cli import main as cli_main

    stack = Stack(tmp_path)
    try:
        blob = tmp_path / "s.bin"
        blob.write_bytes(b"x" * 1000)
        stack.origin.add_hf_repo("org/st", {"s.bin": str(blob)})
        import urllib.request

        urllib.request.urlopen(
            f"{stack.endpoint}/org/st/resolve/main/s.bin").read()
        rc = cli_main(["stats", "--endpoint", stack.endpoint])
        assert rc == 0
        out = json.loads(capsys.readouterr().out)
        assert out["requests"] >= 1
        assert out["cache_misses"] >= 1
    finally:
        stack.close()
