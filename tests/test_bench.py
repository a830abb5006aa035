"""bench.py driver-contract tests: the JSON line schema and the exact
torch.distributed.run launch path the round driver uses (CPU/gloo here)."""
import json
import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent
NEED = ["metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
        "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config"]


def _last_json(stdout):
    for line in reversed(stdout.strip().splitlines()):
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output: {stdout[-500:]}")


def test_bench_json_contract_single():
    r = subprocess.run([sys.executable, str(REPO / "bench.py"), "--device", "cpu",
                        "--rows", "100000", "--steps", "3", "--warmup", "1"],
                       capture_output=True, text=True, timeout=600, cwd=REPO)
    assert r.returncode == 0, r.stderr[-2000:]
    d = _last_json(r.stdout)
    assert not [k for k in NEED if k not in d]
    assert d["higher_is_better"] is False and d["scaling"] == "weak"
    assert d["data"] == "synthetic" and d["ms_per_step"] > 0
    assert d["config"]["max_bin"] == 63 and d["config"]["num_leaves"] == 255


def test_bench_driver_launch_world2():
    env = dict(os.environ)
    env.pop("RANK", None)
    r = subprocess.run([sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
                        "--nproc-per-node=2", "--master-addr", "127.0.0.1",
                        "--master-port", "29382", str(REPO / "bench.py"), "--gpus", "2",
                        "--device", "cpu", "--rows", "100000", "--steps", "3",
                        "--warmup", "1"],
                       capture_output=True, text=True, timeout=600, env=env, cwd=REPO)
    assert r.returncode == 0, r.stdout[-1500:] + r.stderr[-1500:]
    d = _last_json(r.stdout)
    assert d["config"]["parallelism"] == "dp2"
