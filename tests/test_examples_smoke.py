"""Every example script must run end-to-end on CPU (small args)."""
import os
import subprocess
import sys

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _run(args, timeout=300):
    return subprocess.run(
        [sys.executable] + args,
        cwd=ROOT,
        capture_output=True,
        text=True,
        timeout=timeout,
    )


def test_ps_training_example():
    r = _run(["examples/ps_training.py", "--rounds", "2"])
    assert r.returncode == 0, r.stderr[-800:]
    assert "loss" in r.stdout


def test_p2p_gossip_example():
    r = _run(["examples/p2p_gossip.py"])
    assert r.returncode == 0, r.stderr[-800:]


def test_heterogeneous_pool_example():
    r = _run(["examples/heterogeneous_pool.py"])
    assert r.returncode == 0, r.stderr[-800:]
    assert "cpu results" in r.stdout


def test_mesh_decentralized_example():
    r = _run(["examples/mesh_decentralized.py"])
    assert r.returncode == 0, r.stderr[-800:]
    assert "consensus=True" in r.stdout


def test_rccl_ps_example_single_process():
    r = _run(["examples/rccl_multigpu_ps.py"])
    assert r.returncode == 0, r.stderr[-800:]
    assert "RCCL PS" in r.stdout


def test_gossip_serving_example():
    r = _run(["examples/gossip_serving.py", "--rounds", "2", "--d", "20000"])
    assert r.returncode == 0, r.stderr
    assert "gossip serving" in r.stdout
