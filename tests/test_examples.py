"""Smoke-run every script in examples/ on CPU (tiny sizes).

The reference kept its examples runnable as part of the release checks
(examples/run/*.sh); here each example is a subprocess with shrunk args so
doc drift or API breakage surfaces in CI, not in a user's terminal.
"""

import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EXAMPLES = os.path.join(REPO, "examples")


def _run(script, *args, timeout=300):
    env = dict(os.environ)
    env.setdefault("OMP_NUM_THREADS", "2")
    # scripts live in examples/, so sys.path[0] is examples/ — put the repo
    # root on the path the way a `pip install -e` user would have it
    env["PYTHONPATH"] = REPO + os.pathsep + env.get("PYTHONPATH", "")
    r = subprocess.run(
        [sys.executable, os.path.join(EXAMPLES, script), *args],
        cwd=REPO, env=env, capture_output=True, text=True, timeout=timeout)
    assert r.returncode == 0, (
        f"{script} failed rc={r.returncode}\n"
        f"stdout:\n{r.stdout[-2000:]}\nstderr:\n{r.stderr[-2000:]}")
    return r.stdout


def test_three_line_change():
    out = _run("three_line_change.py")
    assert "rows materialized" in out


def test_criteo_deepfm():
    # logs every 20 steps — run exactly 20 so one line appears
    out = _run("criteo_deepfm.py", "--steps", "20", "--batch", "128")
    assert "loss" in out


def test_criteo_lr_hash():
    out = _run("criteo_lr_hash.py", "--steps", "10", "--batch", "128")
    assert "loss" in out


def test_wide_deep_small_vocab():
    # the 1B-vocab flagship shrunk to CPU scale; same code path (hash-mode
    # CombinedEmbedding + reserve, WDL wide/deep split)
    out = _run("wide_deep_1b.py", "--vocab", "100000", "--dim", "8",
               "--batch", "128", "--steps", "5")
    assert "loss" in out


def test_checkpoint_and_serve():
    pytest.importorskip("fastapi")
    out = _run("checkpoint_and_serve.py")
    assert "serving matches training: OK" in out
