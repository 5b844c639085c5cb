"""bench.py harness validation on CPU: the exact torchrun invocation the
driver uses, world_size=2 over gloo-delegated cgx, JSON contract checks."""

import json
import os
import subprocess
import sys

import pytest

from dist_utils import free_port

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run_bench(extra, world=2, timeout=420):
    cmd = [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
           f"--nproc-per-node={world}", "--master-addr", "127.0.0.1",
           "--master-port", str(free_port()), "bench.py",
           "--device", "cpu", "--steps", "2", "--warmup", "1",
           "--batch", "2"] + extra
    r = subprocess.run(cmd, cwd=ROOT, capture_output=True, text=True,
                       timeout=timeout)
    assert r.returncode == 0, f"stdout:\n{r.stdout}\nstderr:\n{r.stderr}"
    lines = [ln for ln in r.stdout.splitlines() if ln.startswith("{")]
    assert lines, r.stdout
    return json.loads(lines[-1])


def test_bench_resnet_cpu_ws2():
    d = _run_bench(["--gpus", "2"])
    assert d["n_gpus"] == 2
    assert d["value"] > 0
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["bits"] == 4
    assert d["config"]["compression_active"] is True
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"


def test_bench_show_registry_ws2():
    d = _run_bench(["--gpus", "2", "--show-registry"])
    split = d["config"]["layerwise_filter"]
    # ResNet-50: conv kernels compressed; BN weights/biases (dim<=1) and
    # small layers registered at bits=32 by the layerwise filter
    assert split["compressed_layers"] > 0
    assert split["bits32_layers"] > 0


def test_bench_json_contract_single():
    cmd = [sys.executable, "bench.py", "--device", "cpu", "--steps", "1",
           "--warmup", "0", "--batch", "2"]
    r = subprocess.run(cmd, cwd=ROOT, capture_output=True, text=True,
                       timeout=300)
    assert r.returncode == 0, r.stderr
    d = json.loads([ln for ln in r.stdout.splitlines()
                    if ln.startswith("{")][-1])
    for key in ["metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"]:
        assert key in d, key
    assert d["n_gpus"] == 1
    # an N=1 run must be honestly labeled: the backend+hook run, but no
    # gradient compression/communication happens
    assert d["config"]["compression_active"] is False
