"""Multi-process data-parallel tests on the gloo backend (world_size 2):
sharded + all-gathered spectrum must equal the single-process result
(SURVEY.md §4(e))."""

import os

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

from fastfp_amd import FpEngine, get_mats_fp, initialize_pta, make_synthetic_pta
from fastfp_amd.parallel import all_gather_concat, shard_slice


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


def test_shard_slice_partitions():
    for n in (1, 5, 8, 17):
        for world in (1, 2, 3, 8):
            idx = []
            for r in range(world):
                s = shard_slice(n, r, world)
                idx.extend(range(n)[s])
            assert idx == list(range(n))
