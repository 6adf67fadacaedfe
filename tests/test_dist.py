"""Multi-process data-parallel tests on the gloo backend (world_size 2):
sharded + all-gathered spectrum must equal the single-process result
(SURVEY.md §4(e))."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from fastfp_amd import FpEngine, get_mats_fp, initialize_pta, make_synthetic_pta
from fastfp_amd.parallel import all_gather_concat, init_distributed, shard_slice


def _build():
    psrs = make_synthetic_pta(npsr=2, ntoa=60, ntm=3, seed=0)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=3, gwb_comps=3)
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    return psrs, Nvecs, Ts, sigmas


def _worker(rank, world, port, freqs, out_file):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    psrs, Nvecs, Ts, sigmas = _build()
    local = freqs[shard_slice(len(freqs), rank, world)]
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(local)
    fp_local = eng.sweep(sigmas=sigmas)
    fp = all_gather_concat(fp_local.reshape(-1), world)
    if rank == 0:
        np.save(out_file, fp.numpy())
    torch.distributed.destroy_process_group()


def test_freq_sharded_fp_matches_single(tmp_path):
    freqs = np.linspace(3e-9, 6e-8, 7)  # odd count -> uneven shards
    out_file = str(tmp_path / "fp.npy")
    port = 29841
    mp.spawn(_worker, args=(2, port, freqs, out_file), nprocs=2, join=True)
    got = np.load(out_file)

    psrs, Nvecs, Ts, sigmas = _build()
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(freqs)
    want = eng.sweep(sigmas=sigmas).numpy()
    # identical partition of identical per-frequency computations:
    # bitwise equality expected (fp64, per-frequency independence)
    np.testing.assert_array_equal(got, want)


def _ws1_env(port):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    os.environ["LOCAL_RANK"] = "0"


def _ws1_env_clear():
    for k in ("MASTER_ADDR", "MASTER_PORT", "RANK", "WORLD_SIZE", "LOCAL_RANK"):
        os.environ.pop(k, None)


def test_ws1_initialized_group_collectives_cpu():
    """A torchrun-style WORLD_SIZE=1 launch now initializes the process
    group, and every collective call site runs (one-rank collectives):
    the gathered spectrum must equal the local one bitwise."""
    _ws1_env(29861)
    try:
        rank, world, dev = init_distributed(device=torch.device("cpu"))
        assert (rank, world) == (0, 1)
        assert torch.distributed.is_initialized()
        x = torch.arange(12, dtype=torch.float64).reshape(3, 4)
        got = all_gather_concat(x.clone(), world, dim=0)
        np.testing.assert_array_equal(got.numpy(), x.numpy())
        t = torch.tensor([3.5], dtype=torch.float64)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        assert float(t.item()) == 3.5
        torch.distributed.barrier()
    finally:
        if torch.distributed.is_initialized():
            torch.distributed.destroy_process_group()
        _ws1_env_clear()


@pytest.mark.gpu
def test_ws1_rccl_collectives_on_device():
    """RCCL (nccl backend on ROCm) communicator init + collectives on
    HIP tensors at world_size 1 — exercises the exact code path the
    8-GPU bench uses (init_distributed -> all_gather_concat ->
    all_reduce MAX on device tensors) on a 1-GPU box."""
    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    _ws1_env(29862)
    try:
        rank, world, dev = init_distributed()
        assert dev.type == "cuda"
        assert torch.distributed.get_backend() in ("nccl", "cclx")
        x = torch.randn(512, 7, dtype=torch.float64, device=dev)
        got = all_gather_concat(x.clone(), world, dim=0)
        torch.cuda.synchronize()
        np.testing.assert_array_equal(got.cpu().numpy(), x.cpu().numpy())
        t = torch.tensor([2.25], dtype=torch.float64, device=dev)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        torch.distributed.barrier()
        torch.cuda.synchronize()
        assert float(t.item()) == 2.25
    finally:
        if torch.distributed.is_initialized():
            torch.distributed.destroy_process_group()
        _ws1_env_clear()


def test_shard_slice_partitions():
    for n in (1, 5, 8, 17):
        for world in (1, 2, 3, 8):
            idx = []
            for r in range(world):
                s = shard_slice(n, r, world)
                idx.extend(range(n)[s])
            assert idx == list(range(n))


def _nmfp_build():
    from fastfp_amd import get_mats_nmfp, initialize_pta
    from fastfp_amd.nmfp import NMFp

    psrs = make_synthetic_pta(npsr=2, ntoa=60, ntm=3, seed=0)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=3, gwb_comps=3)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    D = 5
    rng = np.random.default_rng(2)
    samples = {
        n: (rng.uniform(2, 6, D) if n.endswith("gamma")
            else rng.uniform(-16, -14, D))
        for n in pta.params
    }
    return psrs, pta, Nvecs, Ts, samples, D


def _nmfp_worker(rank, world, port, out_file):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    torch.distributed.init_process_group("gloo", rank=rank, world_size=world)
    from fastfp_amd.nmfp import NMFp

    psrs, pta, Nvecs, Ts, samples, D = _nmfp_build()
    my = shard_slice(D, rank, world)
    local_samples = {k: v[my] for k, v in samples.items()}
    nm = NMFp(psrs, pta.rn_containers)
    freqs = np.linspace(4e-9, 5e-8, 6)
    vals = nm.sweep(freqs, local_samples, Nvecs, Ts, device="cpu")
    t = torch.as_tensor(vals, dtype=torch.float64)
    full = all_gather_concat(t, world, dim=0)
    if rank == 0:
        np.save(out_file, full.numpy())
    torch.distributed.destroy_process_group()


def test_draw_sharded_nmfp_matches_single(tmp_path):
    """Draw-sharded + all-gathered NM-Fp == single-process (uneven
    shards: 5 draws over 2 ranks)."""
    out_file = str(tmp_path / "nm.npy")
    mp.spawn(_nmfp_worker, args=(2, 29853, out_file), nprocs=2, join=True)
    got = np.load(out_file)

    from fastfp_amd.nmfp import NMFp

    psrs, pta, Nvecs, Ts, samples, D = _nmfp_build()
    nm = NMFp(psrs, pta.rn_containers)
    freqs = np.linspace(4e-9, 5e-8, 6)
    want = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu")
    np.testing.assert_allclose(got, want, rtol=1e-12)
