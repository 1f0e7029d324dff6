"""bench.py contract tests (CPU): the driver parses one JSON line from
rank 0 with a fixed schema; variant flags must not break it."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = {"metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"}


def _run_bench(*flags, timeout=420):
    r = subprocess.run(
        [sys.executable, "bench.py", "--steps", "2", "--warmup", "1",
         "--batch", "64", *flags],
        cwd=REPO, capture_output=True, text=True, timeout=timeout)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines() if ln.startswith("{")][-1]
    return json.loads(line)


def test_default_contract():
    j = _run_bench()
    assert REQUIRED <= set(j)
    assert j["n_gpus"] == 1 and j["steps"] == 2 and j["warmup"] == 1
    assert j["higher_is_better"] is True and j["scaling"] == "weak"
    assert j["value"] > 0 and j["ms_per_step"] > 0
    # value is the whole-job aggregate: batch * n_gpus * steps / elapsed
    assert abs(j["value"] - 64 * 1 * 1000.0 / j["ms_per_step"]) / j["value"] < 0.01
    assert j["config"]["model"] == "deepfm-dim9"
    assert j["config"]["global_batch"] == 64
    # vs_baseline only for the exact headline config (batch 4096)
    assert j["vs_baseline"] is None


def test_hash_mode_flag():
    j = _run_bench("--hash")
    assert j["config"]["table"] == "hash"


def test_cache_tier_flag():
    j = _run_bench("--cache-mb", "16")
    assert j["config"]["table"] == "hash" and j["config"]["cache_mb"] == 16


@pytest.mark.parametrize("model", ["wdl", "lr"])
def test_model_variants(model):
    j = _run_bench("--model", model)
    assert j["config"]["model"].startswith(model)


def test_xdeepfm_variant():
    j = _run_bench("--model", "xdeepfm", "--dim", "4")
    assert j["config"]["model"] == "xdeepfm-dim4"


def test_benchmark_grid_script():
    # the grid driver must run cells and write JSON lines (CPU, tiny cells)
    out = os.path.join(REPO, "gpurun_out", "grid_test.jsonl")
    r = subprocess.run(
        [sys.executable, "scripts/benchmark_grid.py", "--models", "lr,wdl",
         "--dims", "4", "--gpus", "1", "--steps", "2", "--warmup", "1",
         "--batch", "64", "--out", out],
        cwd=REPO, capture_output=True, text=True, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    rows = [json.loads(l) for l in open(out)]
    assert len(rows) == 2
    assert {row["config"]["model"] for row in rows} == {"lr-dim4", "wdl-dim4"}


def test_prefetch_flag():
    j = _run_bench("--prefetch", "--graph", "off")
    assert j["config"]["prefetch"] is True
    assert j["value"] > 0


@pytest.mark.timeout(420)
def test_torchrun_world2_contract():
    """The driver's SCALE launch shape: torchrun --nproc-per-node N
    bench.py --gpus N. Runs on CPU/gloo here; rank 0 must print the one
    JSON line with whole-job value and multi-rank config."""
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node=2", "--master-addr=127.0.0.1",
         "--master-port=29655", "bench.py", "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--batch", "32", "--data-pool", "2"],
        cwd=REPO, capture_output=True, text=True, timeout=390)
    assert r.returncode == 0, r.stderr[-2000:]
    line = [ln for ln in r.stdout.splitlines()
            if ln.strip().startswith("{") and '"metric"' in ln][-1]
    j = json.loads(line)
    assert j["n_gpus"] == 2
    assert j["config"]["global_batch"] == 64          # whole-job
    assert "all_to_all" in j["config"]["parallelism"]
