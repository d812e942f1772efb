"""CLI end-to-end tests: the dist_mnist.py entrypoint
runs with env-var launch, prints the reference's per-epoch line, and writes a
loadable checkpoint."""

import os
import re
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


@pytest.mark.timeout(300)
def test_cli_single_rank_trains_and_checkpoints(tmp_path, free_port):
    path = str(tmp_path / "model.pt")
    env = dict(os.environ,
               MASTER_ADDR="127.0.0.1", MASTER_PORT=str(free_port),
               RANK="0", WORLD_SIZE="1")
    out = subprocess.run(
        [sys.executable, os.path.join(REPO, "dist_mnist.py"),
         "--backend", "gloo", "--epochs", "2", "--batch-size", "32",
         "--dataset-size", "128", "--modelpath", path, "--no-cuda"],
        env=env, capture_output=True, text=True, timeout=280, cwd=REPO)
    assert out.returncode == 0, out.stderr[-2000:]
    # the reference's per-epoch line: "Rank <r>, epoch <e>: <loss>"
    lines = [ln for ln in out.stdout.splitlines() if re.match(r"Rank\s+0\s*,\s*epoch", ln)]
    assert len(lines) == 2, out.stdout
    sd = torch.load(path, map_location="cpu", weights_only=True)
    assert "conv1.weight" in sd and sd["fc2.bias"].shape == (10,)


@pytest.mark.timeout(300)
def test_cli_two_rank_gloo(tmp_path, free_port):
    """BASELINE.json config 1: world_size=2 gloo CPU plumbing."""
    env_base = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(free_port),
                    WORLD_SIZE="2")
    procs = []
    for rank in range(2):
        env = dict(env_base, RANK=str(rank))
        procs.append(subprocess.Popen(
            [sys.executable, os.path.join(REPO, "dist_mnist.py"),
             "--backend", "gloo", "--epochs", "1", "--batch-size", "32",
             "--dataset-size", "64", "--no-cuda"],
            env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE, text=True, cwd=REPO))
    outs = [p.communicate(timeout=280) for p in procs]
    for i, p in enumerate(procs):
        assert p.returncode == 0, outs[i][1][-2000:]
    for rank in range(2):
        assert f"Rank  {rank} , epoch  0" in outs[rank][0] or \
            re.search(rf"Rank\s+{rank}\s*,\s*epoch\s+0", outs[rank][0]), outs[rank][0]


@pytest.mark.timeout(300)
def test_cli_ddp_reference_mode_matches_bucketed(tmp_path, free_port):
    """--ddp reference (the literal per-param all-reduce) must
    train to the same loss as the default bucketed engine — same seed, same
    shards, same SUM/world semantics."""
    def run_mode(mode, port):
        env_base = dict(os.environ, MASTER_ADDR="127.0.0.1",
                        MASTER_PORT=str(port), WORLD_SIZE="2")
        procs = []
        for rank in range(2):
            env = dict(env_base, RANK=str(rank))
            procs.append(subprocess.Popen(
                [sys.executable, os.path.join(REPO, "dist_mnist.py"),
                 "--backend", "gloo", "--epochs", "1", "--batch-size", "32",
                 "--dataset-size", "64", "--no-cuda", "--seed", "5",
                 "--ddp", mode],
                env=env, stdout=subprocess.PIPE, stderr=subprocess.PIPE,
                text=True, cwd=REPO))
        outs = [p.communicate(timeout=280) for p in procs]
        for i, p in enumerate(procs):
            assert p.returncode == 0, outs[i][1][-2000:]
        m = re.search(r"Rank\s+0\s*,\s*epoch\s+0\s*:\s*([0-9.]+)",
                      outs[0][0])
        assert m, outs[0][0]
        return float(m.group(1))

    l_buck = run_mode("bucketed", free_port)
    l_ref = run_mode("reference", free_port + 1)
    assert abs(l_buck - l_ref) <= 1e-6 * max(1.0, abs(l_buck)), (l_buck, l_ref)


@pytest.mark.timeout(300)
def test_cli_resume(tmp_path, free_port):
    """--resume loads a prior checkpoint and continues (framework extension
    on the reference's save-only surface, SURVEY.md §5 checkpoint/resume)."""
    path = str(tmp_path / "m.pt")
    env = dict(os.environ, MASTER_ADDR="127.0.0.1", MASTER_PORT=str(free_port),
               RANK="0", WORLD_SIZE="1")
    a = subprocess.run(
        [sys.executable, os.path.join(REPO, "dist_mnist.py"), "--backend", "gloo",
         "--epochs", "1", "--batch-size", "32", "--dataset-size", "64",
         "--modelpath", path, "--no-cuda"],
        env=env, capture_output=True, text=True, timeout=280, cwd=REPO)
    assert a.returncode == 0, a.stderr[-1500:]
    env["MASTER_PORT"] = str(free_port + 1)
    b = subprocess.run(
        [sys.executable, os.path.join(REPO, "dist_mnist.py"), "--backend", "gloo",
         "--epochs", "1", "--batch-size", "32", "--dataset-size", "64",
         "--resume", path, "--modelpath", path + ".2", "--no-cuda"],
        env=env, capture_output=True, text=True, timeout=280, cwd=REPO)
    assert b.returncode == 0, b.stderr[-1500:]
    sd1 = torch.load(path, map_location="cpu", weights_only=True)
    sd2 = torch.load(path + ".2", map_location="cpu", weights_only=True)
    assert not torch.equal(sd1["fc2.weight"], sd2["fc2.weight"])  # training continued
