"""bench.py contract: runs on one GPU and prints the JSON line."""

import json
import os
import subprocess
import sys

import pytest

pytestmark = pytest.mark.gpu

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_bench_single_gpu_json():
    r = subprocess.run(
        [sys.executable, os.path.join(ROOT, "bench.py"), "--gpus", "1",
         "--steps", "3", "--warmup", "2", "--batch", "256"],
        capture_output=True,
        text=True,
        timeout=600,
        cwd=ROOT,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    line = [l for l in r.stdout.splitlines() if l.startswith("{")][-1]
    d = json.loads(line)
    assert d["unit"] == "samples/sec"
    assert d["n_gpus"] == 1
    assert d["steps"] == 3
    assert d["value"] > 0
    assert d["dtype"] == "bf16"
    assert d["data"] == "synthetic"
    assert d["higher_is_better"] is True
    assert d["config"]["model"] == "resnet18-cifar"
    assert d["loss"] == d["loss"]  # not NaN
