"""Guards the driver contract: bench.py must emit ONE JSON line with
the agreed schema, on CPU and under torchrun-style env."""

import json
import os
import subprocess
import sys

HERE = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

REQUIRED = ["metric", "value", "unit", "n_gpus", "steps", "warmup",
            "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
            "dtype", "data", "config"]


def run_bench(extra, env_extra=None):
    env = dict(os.environ)
    if env_extra:
        env.update(env_extra)
    r = subprocess.run(
        [sys.executable, os.path.join(HERE, "bench.py"), "--steps", "2",
         "--warmup", "1", "--batch-size", "4", "--image-size", "32"] + extra,
        capture_output=True, text=True, timeout=600, env=env, cwd=HERE)
    assert r.returncode == 0, r.stderr[-2000:]
    lines = [l for l in r.stdout.strip().splitlines()
             if l.startswith("{")]
    assert len(lines) == 1, r.stdout
    return json.loads(lines[0])


def check_schema(d, n_gpus=1):
    for k in REQUIRED:
        assert k in d, k
    assert d["metric"] == "samples/sec (whole node)"
    assert d["n_gpus"] == n_gpus
    assert d["value"] > 0
    assert d["higher_is_better"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
    assert d["dtype"] in ("bf16", "fp32")
    assert "model" in d["config"] and "global_batch" in d["config"]
    assert "parallelism" in d["config"]


def test_bench_default_cpu():
    d = run_bench([])
    check_schema(d)
    assert d["config"]["parallelism"] == "dp1"


def test_bench_hips_flags_cpu():
    d = run_bench(["--mode", "hips", "--compress", "fp16",
                   "--wan-gbps", "10"])
    check_schema(d)  # single process -> falls back to dp1 topology


def test_bench_value_consistency():
    d = run_bench([])
    # value == n_gpus * per_gpu_batch * 1000 / ms_per_step
    expect = d["n_gpus"] * d["config"]["per_gpu_batch"] * 1e3 / d["ms_per_step"]
    assert abs(expect - d["value"]) / expect < 0.01
