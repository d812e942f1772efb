"""bench.py driver-contract test (CPU): one JSON line with the exact field
surface the driver parses; whole-job aggregate value; timed-region math."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_bench_json_contract():
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "bench.py"), "--steps", "3",
         "--warmup", "1", "--batch", "32", "--dtype", "fp32"],
        capture_output=True, text=True, timeout=280, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.strip().splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout
    d = json.loads(lines[0])

    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["steps"] == 3 and d["warmup"] == 1
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["vs_baseline"] is None  # reference publishes no numbers
    assert d["n_gpus"] == 1
    assert "images/sec" in d["metric"] and "LeNet" in d["metric"]
    for key in ("model", "global_batch", "seq_len", "parallelism"):
        assert key in d["config"], key
    assert d["config"]["global_batch"] == 32
    assert d["config"]["parallelism"] == "dp1"
    # value is the whole-job aggregate: images/s == global_batch * steps / t
    t_s = d["ms_per_step"] * d["steps"] / 1000.0
    assert abs(d["value"] - 32 * 3 / t_s) / d["value"] < 1e-6


@pytest.mark.timeout(300)
def test_bench_two_rank_gloo_contract():
    """The driver launches multi-GPU benches via torch.distributed.run —
    rehearse that exact path on CPU/gloo: exactly ONE JSON line (rank 0
    only), n_gpus==2, whole-job aggregate value, dp2 parallelism tag."""
    import socket

    with socket.socket() as s:
        s.bind(("127.0.0.1", 0))
        port = s.getsockname()[1]
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(REPO, "bench.py"), "--gpus", "2", "--steps", "3",
         "--warmup", "1", "--batch", "32", "--dtype", "fp32"],
        capture_output=True, text=True, timeout=280, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [ln for ln in out.stdout.strip().splitlines() if ln.startswith("{")]
    assert len(lines) == 1, out.stdout  # rank 0 prints, rank 1 must not
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 64  # 32 per rank x 2
    t_s = d["ms_per_step"] * d["steps"] / 1000.0
    assert abs(d["value"] - 64 * 3 / t_s) / d["value"] < 1e-6
