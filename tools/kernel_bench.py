"""Isolated micro-benchmarks of the hot HIP kernels on one MI355X.

Usage (GPU box): python tools/kernel_bench.py
Prints per-kernel wall time and fp64 TFLOP/s at the NMFp bench shape.
"""

import sys

import numpy as np
import torch

sys.path.insert(0, __import__("os").path.abspath(
    __import__("os").path.join(__import__("os").path.dirname(__file__), "..")))
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
    m, mp = 120, 128
    F = 1000
    ntoa = 5000
    rng = np.random.default_rng(0)

    T = torch.as_tensor(rng.normal(size=(ntoa, m)), dtype=torch.float64, device=DEV)
    nvec = torch.full((ntoa,), 1e-12, dtype=torch.float64, device=DEV)
    TNT = (T.T @ (T / nvec[:, None])).contiguous()
    toas = torch.as_tensor(
        np.sort(rng.uniform(0, 4.7e8, ntoa)), dtype=torch.float64, device=DEV
    )
    ninv = (1.0 / nvec).contiguous()
    freqs = torch.as_tensor(
        np.arange(1, F + 1) / 4.7e8, dtype=torch.float64, device=DEV
    )
    RHS = torch.zeros((mp, 2 * F + 1), dtype=torch.float64, device=DEV)
    sNs = torch.ones((3, F), dtype=torch.float64, device=DEV) * 1e13
    sNr = torch.ones((2, F), dtype=torch.float64, device=DEV)

    for D in (128, 256, 512, 1024):
        phiinv = torch.as_tensor(
            rng.uniform(0.5, 2.0, (D, m)) * 1e10, dtype=torch.float64, device=DEV
        )
        t = timeit(lambda: ext.chol_batch(TNT, phiinv, mp))
        # flops: assemble+factor: m^3/3 chol + panel/trailing MFMA work
        flops = D * (mp**3 / 3 + mp**3 / 6) * 2
        print(f"chol  D={D:5d}: {t*1e6:9.1f} us  {flops/t/1e12:6.2f} TF/s"
              f"  ({t/D*1e6:6.2f} us/matrix)")

        L, invd = ext.chol_batch(TNT, phiinv, mp)
        fp = torch.zeros((D, F), dtype=torch.float64, device=DEV)
        t = timeit(lambda: ext.trsm_fp_accum(L, invd, RHS, sNs, sNr, fp, 1.0))
        flops = D * (mp * mp / 2) * (2 * F + 2) * 2  # TRSM MACs*2
        print(f"trsm  D={D:5d}: {t*1e6:9.1f} us  {flops/t/1e12:6.2f} TF/s"
              f"  ({flops:.3g} flops)")

    # sbgemm at the two headline shapes
    for Fs in (1000, 10000):
        fr = torch.as_tensor(
            np.arange(1, Fs + 1) / 4.7e8, dtype=torch.float64, device=DEV
        )
        out = torch.zeros((mp, 2 * Fs + 1), dtype=torch.float64, device=DEV)
        ctiles = (2 * Fs + 63) // 64
        ks = max(1, min(8, (512 + ctiles - 1) // ctiles, (ntoa + 255) // 256))
        if ks == 1:
            fn = lambda: ext.sbgemm(T, toas, ninv, fr, out, 0, 2 * Fs + 1, mp, 1)  # noqa: E731
        else:
            part = torch.empty((ks, mp, 2 * Fs), dtype=torch.float64, device=DEV)
            fn = lambda: ext.sbgemm(T, toas, ninv, fr, part, mp * 2 * Fs, 2 * Fs, mp, ks)  # noqa: E731
        t = timeit(fn, iters=10)
        flops = 2.0 * mp * 2 * Fs * ntoa
        print(f"sbgemm F={Fs:6d} ks={ks}: {t*1e6:9.1f} us  {flops/t/1e12:6.2f} TF/s")


if __name__ == "__main__":
    main()
