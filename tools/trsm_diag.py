"""Isolated diagnosis of the trsm_fp kernel at the BENCH compressed
shape (P=67, mv=60 -> mp=64, F=1000), plus the chol kernel.

Run on a GPU box, once per algo arm:
    python tools/trsm_diag.py                      # default (ll)
    FASTFP_TRSM_ALGO=res python tools/trsm_diag.py
    FASTFP_TRSM_ALGO=res FASTFP_TRSM_MODE=1 python tools/trsm_diag.py
    FASTFP_TRSM_ALGO=res FASTFP_TRSM_MODE=2 python tools/trsm_diag.py

Prints us/launch and effective TF/s for the P-batched stacked launch
exactly as the bench issues it.
"""

import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))
from fastfp_amd.ops import _fastfp_hip as ext  # noqa: E402

DEV = "cuda:0"


def timeit(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    start = torch.cuda.Event(enable_timing=True)
    end = torch.cuda.Event(enable_timing=True)
    start.record()
    for _ in range(iters):
        fn()
    end.record()
    torch.cuda.synchronize()
    return start.elapsed_time(end) / iters * 1e-3  # seconds


def main():
    P, mv, mp = 67, 60, 64
    rng = np.random.default_rng(0)
    for F, D in ((1000, 1000), (1000, 250), (4000, 1000)):
        # synthetic SPD G + phi
        A = rng.normal(size=(P, mv, mv))
        G = np.einsum("pij,pkj->pik", A, A) + mv * np.eye(mv)
        G = torch.as_tensor(G, dtype=torch.float64, device=DEV).contiguous()
        phi = torch.as_tensor(
            rng.uniform(0.5, 2.0, (P, D, mv)), dtype=torch.float64, device=DEV
        ).contiguous()
        K = torch.as_tensor(
            rng.normal(size=(P, mp, 2 * F + 1)), dtype=torch.float64, device=DEV
        ).contiguous()
        sNs = (torch.ones((P, 3, F), dtype=torch.float64, device=DEV) * 10).contiguous()
        sNr = torch.ones((P, 2, F), dtype=torch.float64, device=DEV).contiguous()
        fp = torch.zeros((P, D, F), dtype=torch.float64, device=DEV)

        t_chol = timeit(lambda: ext.chol_batch(G, phi, mp))
        L, invd = ext.chol_batch(G, phi, mp)
        t_trsm = timeit(lambda: ext.trsm_fp_accum(L, invd, K, sNs, sNr, fp, -1.0))
        # per-launch MFMA flop count of the solve (strictly-lower updates
        # + diagonal inverse-apply, 128 cols per ftile incl pad)
        ftiles = (F + 62) // 63
        nbt = mp // 16
        mfma = (nbt * (nbt - 1) // 2 + nbt) * 4  # per wave
        flops = P * D * ftiles * 8 * mfma * 2048
        print(
            f"F={F} D={D}: chol {t_chol*1e3:8.3f} ms   "
            f"trsm {t_trsm*1e3:8.3f} ms  ({flops/t_trsm/1e12:5.1f} TF/s MFMA-issued)"
        )


if __name__ == "__main__":
    main()
