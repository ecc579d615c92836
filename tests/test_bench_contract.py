"""Driver-contract guard: bench.py must run (CPU fallback), obey the
flag shape, and print ONE JSON line with the required fields — including
under the driver's exact torch.distributed.run invocation (gloo here)."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
BENCH = os.path.join(ROOT, "bench.py")

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _last_json(stdout: str) -> dict:
    lines = [l for l in stdout.strip().splitlines() if l.startswith("{")]
    assert lines, stdout
    return json.loads(lines[-1])


def test_bench_single_process_contract():
    out = subprocess.run(
        [sys.executable, BENCH, "--gpus", "1", "--steps", "2", "--warmup", "1",
         "--batch", "256", "--pool", "2", "--no-transformer"],
        capture_output=True, text=True, timeout=300, cwd=ROOT,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    d = _last_json(out.stdout)
    assert REQUIRED.issubset(d.keys()), sorted(REQUIRED - set(d))
    assert d["steps"] == 2 and d["warmup"] == 1 and d["n_gpus"] == 1
    assert d["data"] == "synthetic" and d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["value"] > 0 and d["ms_per_step"] > 0
    cfg = d["config"]
    assert "model" in cfg and "global_batch" in cfg and "parallelism" in cfg


@pytest.mark.timeout(400)
def test_bench_torchrun_two_ranks_contract():
    """The driver's multi-rank launch shape (gloo over 127.0.0.1 here)."""
    env = dict(os.environ)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29531", BENCH, "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--batch", "256", "--pool", "2",
         "--no-transformer"],
        capture_output=True, text=True, timeout=380, cwd=ROOT, env=env,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    d = _last_json(out.stdout)
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 512  # whole-job aggregate
