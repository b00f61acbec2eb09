"""Smoke-run every benchmark harness on CPU with tiny shapes: the driver
and the judge call these scripts; they must never bitrot."""
import json
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=240):
    return subprocess.run(
        [sys.executable] + args,
        cwd=ROOT,
        capture_output=True,
        text=True,
        timeout=timeout,
    )


def test_bench_default_contract():
    r = _run(["bench.py", "--steps", "2", "--warmup", "1", "--d", "20000"])
    assert r.returncode == 0, r.stderr[-800:]
    line = r.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in (
        "metric",
        "value",
        "unit",
        "n_gpus",
        "steps",
        "warmup",
        "ms_per_step",
        "higher_is_better",
        "scaling",
        "vs_baseline",
        "dtype",
        "data",
        "config",
    ):
        assert key in d, f"bench.py JSON missing {key}"
    assert d["n_gpus"] == 1 and d["steps"] == 2
    assert d["data"] == "synthetic"
    assert d["config"]["global_batch"] == d["config"]["n_workers"]


def test_run_suite_single_workload():
    r = _run(
        ["benchmarks/run_suite.py", "--only", "Median n=64", "--repeat", "1",
         "--warmup", "0"]
    )
    assert r.returncode == 0, r.stderr[-800:]
    assert "Median n=64" in r.stdout


def test_config3_harness():
    r = _run(
        ["benchmarks/config3_ps_resnet50.py", "--rounds", "1", "--warmup", "0",
         "--d", "5000", "--device", "cpu"]
    )
    assert r.returncode == 0, r.stderr[-800:]
    assert "config3" in r.stdout


def test_config4_harness():
    r = _run(
        ["benchmarks/config4_p2p_ring.py", "--rounds", "1", "--warmup", "0",
         "--d", "4000", "--device", "cpu"]
    )
    assert r.returncode == 0, r.stderr[-800:]
    assert "config4" in r.stdout


def test_config1_harness():
    r = _run(["benchmarks/config1_thread_pool.py", "--repeat", "2"])
    assert r.returncode == 0, r.stderr[-800:]


def test_kernels_bench_requires_gpu_gracefully():
    import torch

    if torch.cuda.is_available():
        pytest.skip("GPU visible: kernels_bench would run for real")
    r = _run(["benchmarks/kernels_bench.py"])
    # no GPU here: must fail via the assert, not crash weirdly
    assert r.returncode != 0


def test_graft_entry_build_and_importable():
    r = _run(["-c", "import __graft_entry__; __graft_entry__.build()"], timeout=900)
    assert r.returncode == 0, r.stderr[-800:]


def test_bench_metric_matches_baseline_contract():
    """bench.py's metric string must describe BASELINE.json's metric
    (aggregated-grads/sec for Median & Krum) and the config must name the
    BASELINE config-2 shape."""
    import json

    r = _run(["bench.py", "--steps", "1", "--warmup", "0", "--d", "10000"])
    assert r.returncode == 0, r.stderr[-500:]
    d = json.loads(r.stdout.strip().splitlines()[-1])
    assert "aggregated-grads/sec" in d["metric"]
    assert "Median" in d["metric"] and "Krum" in d["metric"]
    assert d["higher_is_better"] is True
    assert d["dtype"] in ("bf16", "f32")
    assert "Multi-Krum" in d["config"]["model"]
    assert d["config"]["n_workers"] == 64
