"""bench.py driver-contract tests: single-rank and multi-rank (gloo, CPU) runs must
produce the one-line JSON with the required fields and sane values."""
import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED_FIELDS = ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                   "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                   "dtype", "data", "config")


def _last_json_line(out: str):
    for line in reversed(out.strip().splitlines()):
        line = line.strip()
        if line.startswith("{"):
            return json.loads(line)
    raise AssertionError(f"no JSON line in output:\n{out[-2000:]}")


@pytest.mark.timeout(240)
def test_bench_single_rank():
    out = subprocess.run(
        [sys.executable, "bench.py", "--steps", "3", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=220)
    assert out.returncode == 0, out.stderr[-2000:]
    res = _last_json_line(out.stdout)
    for f in REQUIRED_FIELDS:
        assert f in res, f"missing field {f}"
    assert res["metric"] == "podgangs_per_sec"
    assert res["n_gpus"] == 1 and res["steps"] == 3 and res["warmup"] == 1
    assert res["value"] > 0 and res["higher_is_better"] is True
    assert res["scaling"] == "weak" and res["data"] == "synthetic"
    assert res["p50_time_to_running_ms"] > 0
    # headline is the deployable (wire) shape, with the in-process number nested
    assert res["config"]["transport"] == "http"
    assert res["inproc"]["value"] > 0
    assert res["calibration"]["spin_ms"] > 0


@pytest.mark.timeout(300)
@pytest.mark.parametrize("gang_size", [1, 2, 4, 8])
def test_bench_gang_size_matrix(gang_size):
    """Every gang size of the driver's scaling sweep produces a valid record
    (be ready for the 8-GPU sweep on first contact)."""
    out = subprocess.run(
        [sys.executable, "bench.py", "--gpus", str(gang_size), "--steps", "2",
         "--warmup", "0", "--gangs-per-step", "2", "--transport", "inproc"],
        cwd=REPO, capture_output=True, text=True, timeout=280)
    assert out.returncode == 0, out.stderr[-2000:]
    res = _last_json_line(out.stdout)
    assert res["n_gpus"] == gang_size
    assert res["config"]["gang_size"] == gang_size
    assert res["value"] > 0 and res["p50_time_to_running_ms"] > 0


@pytest.mark.timeout(300)
def test_bench_two_ranks_gloo():
    env = dict(os.environ, MASTER_ADDR="127.0.0.1")
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29517", "bench.py", "--gpus", "2",
         "--steps", "3", "--warmup", "1"],
        cwd=REPO, capture_output=True, text=True, timeout=280, env=env)
    assert out.returncode == 0, (out.stdout[-1500:], out.stderr[-1500:])
    res = _last_json_line(out.stdout)
    assert res["n_gpus"] == 2
    assert res["config"]["gang_size"] == 2
    assert res["value"] > 0
