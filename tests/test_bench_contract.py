"""The driver depends on bench.py's CLI + one-line JSON contract."""

import json
import os
import subprocess
import sys

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def _run(*extra):
    out = subprocess.run(
        [sys.executable, "bench.py", "--npsr", "2", "--ntoa", "120",
         "--ntm", "4", "--rn-comps", "3", "--gwb-comps", "3",
         "--freqs", "8", "--draws-per-step", "4", "--steps", "1",
         "--warmup", "0", "--device", "cpu", *extra],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    return json.loads(line)

def test_bench_json_contract():
    j = _run()
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling",
                "vs_baseline", "dtype", "data", "config"):
        assert key in j, key
    assert j["n_gpus"] == 1 and j["steps"] == 1
    assert j["dtype"] == "fp64" and j["data"] == "synthetic"
    assert j["higher_is_better"] is True and j["scaling"] == "weak"
    assert j["value"] > 0 and j["vs_baseline"] == j["value"] / 6.0
    assert j["config"]["global_batch"] == 4

def test_bench_presets_parse():
    for p in ("ecorr67", "fp45", "ska200", "nmfp67"):
        j = _run("--preset", p)
        assert j["value"] > 0, p


def test_bench_torchrun_ws2_gloo():
    """The driver's multi-rank launch path end to end (2 ranks, gloo on
    CPU): rendezvous, per-rank pools, collectives inside the timed
    region, MAX-over-ranks timing, one JSON line from rank 0 with the
    whole-job aggregate."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29877", os.path.join(REPO, "bench.py"),
         "--gpus", "2", "--npsr", "2", "--ntoa", "120", "--ntm", "4",
         "--rn-comps", "3", "--gwb-comps", "3", "--freqs", "8",
         "--draws-per-step", "4", "--steps", "1", "--warmup", "0",
         "--device", "cpu"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    j = json.loads(line)
    assert j["n_gpus"] == 2
    assert j["config"]["global_batch"] == 8  # 2 ranks x 4 draws
    assert j["config"]["spectrum_shape"] == [8, 8]  # gathered over ranks


import pytest


@pytest.mark.parametrize("world,port", [(4, 29878), (8, 29879)])
def test_bench_torchrun_multirank_gloo(world, port):
    """4 and 8 ranks — the driver's N=4/N=8 scaling launch shapes.
    Exercises the >2-rank collective path (size exchange + padded
    all-gather over N shards, MAX-over-ranks timing) that the 4- and
    8-GPU round-end runs hit."""
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", str(world), "--master-addr", "127.0.0.1",
         "--master-port", str(port), os.path.join(REPO, "bench.py"),
         "--gpus", str(world), "--npsr", "2", "--ntoa", "120", "--ntm", "4",
         "--rn-comps", "3", "--gwb-comps", "3", "--freqs", "8",
         "--draws-per-step", "3", "--steps", "1", "--warmup", "0",
         "--device", "cpu"],
        cwd=REPO, capture_output=True, text=True, timeout=600,
    )
    assert out.returncode == 0, out.stderr[-2000:]
    line = [l for l in out.stdout.strip().splitlines()
            if l.startswith("{")][-1]
    j = json.loads(line)
    assert j["n_gpus"] == world
    assert j["config"]["global_batch"] == 3 * world
    assert j["config"]["spectrum_shape"] == [3 * world, 8]
