"""Bit-rot guards: every example driver runs a couple of iterations as
a subprocess (single process, CPU, tiny synthetic data)."""

import os
import subprocess
import sys

import pytest

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

EXAMPLES = ["cnn.py", "cnn_bsc.py", "cnn_fp16.py", "cnn_mpq.py",
            "cnn_hfa.py", "cnn_dgt.py", "resnet50.py"]


@pytest.mark.parametrize("script", EXAMPLES)
def test_example_runs(script):
    r = subprocess.run(
        [sys.executable, os.path.join(HERE, "examples", script),
         "--max-iters", "2", "-bs", "8", "--data-n", "64",
         "--image-size", "20"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.join(HERE, "examples"))
    assert r.returncode == 0, r.stderr[-2000:]
    assert "Iteration 2" in r.stdout, r.stdout[-500:]


def test_example_cnn_flags():
    r = subprocess.run(
        [sys.executable, os.path.join(HERE, "examples", "cnn.py"),
         "--dcasgd", "--max-iters", "2", "-bs", "8", "--data-n", "64",
         "--image-size", "20"],
        capture_output=True, text=True, timeout=600,
        cwd=os.path.join(HERE, "examples"))
    assert r.returncode == 0, r.stderr[-2000:]
