"""The driver launches bench.py via torch.distributed.run for N>1; this
runs the REAL entrypoint with world_size=2 over gloo on CPU so rendezvous,
env parsing, GradAllReduce wiring and the JSON contract are exercised
end-to-end before any multi-GPU run."""
import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


import pytest


@pytest.mark.parametrize("world,port", [(2, 29575), (4, 29576)])
def test_bench_under_torchrun_cpu(world, port):
    env = dict(os.environ)
    env["MASTER_ADDR"] = "127.0.0.1"
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", str(world), "--config", "cpu", "--steps", "2",
         "--warmup", "1"],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=420)
    assert out.returncode == 0, out.stderr[-2000:]
    lines = [l for l in out.stdout.splitlines() if l.startswith("{")]
    assert len(lines) == 1, out.stdout  # exactly one JSON line from rank 0
    rec = json.loads(lines[0])
    assert rec["n_gpus"] == world
    assert rec["config"]["parallelism"] == f"dp{world}"
    assert rec["value"] > 0 and rec["scaling"] == "weak"
    assert rec["config"]["global_batch"] == world * 8  # per-rank CPU cap
    # full driver-contract key set
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in rec, key
    assert rec["data"] == "synthetic" and rec["higher_is_better"] is True
    for ckey in ("model", "global_batch", "seq_len", "parallelism"):
        assert ckey in rec["config"], ckey
