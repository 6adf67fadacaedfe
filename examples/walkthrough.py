"""End-to-end walkthrough — the native equivalent of the reference's
``examples/run_fp.ipynb``: build a synthetic PTA, run the Fp sweep,
validate the null distribution (2Fp ~ chi^2(2 N_psr)), and run a small
noise-marginalized sweep.

    python examples/walkthrough.py [--device cuda:0] [--outdir out]

Writes ``fp_spectrum.json`` (the run_fp output format), the chi^2
summary, and ``nmfp.npy`` (the run_nmfp output format) to ``--outdir``.
"""

import argparse
import json
import os
import sys
import time

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

import numpy as np
import scipy.stats as ss
import torch

from fastfp_amd import (
    FastFp,
    NMFp,
    get_mats_fp,
    get_mats_nmfp,
    initialize_pta,
    make_synthetic_pta,
)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--device", default=None)
    ap.add_argument("--outdir", default="out")
    ap.add_argument("--npsr", type=int, default=12)
    ap.add_argument("--ntoa", type=int, default=500)
    ap.add_argument("--nfreqs", type=int, default=200)
    args = ap.parse_args()
    device = args.device or ("cuda:0" if torch.cuda.is_available() else "cpu")
    os.makedirs(args.outdir, exist_ok=True)

    # 1. synthetic noise-only PTA (random init; no network, no datasets)
    psrs = make_synthetic_pta(npsr=args.npsr, ntoa=args.ntoa, ntm=5, seed=42)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 13.0 / 3.0
        noise[f"{p.name}_red_noise_log10_A"] = -15.0

    # 2. model + precompute (the reference's initialize_pta/get_mats_fp
    #    flow, /root/reference/examples/run_fp.py:47-51)
    t0 = time.perf_counter()
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=10, gwb_comps=10)
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    print(f"precompute: {time.perf_counter() - t0:.3f} s")

    # 3. Fp sweep over the reference's default grid
    freqs = np.linspace(2e-9, 3e-7, args.nfreqs)
    t0 = time.perf_counter()
    fp = FastFp(psrs).sweep(freqs, Nvecs, Ts, sigmas, device=device)
    print(f"Fp sweep ({args.nfreqs} freqs, {args.npsr} psrs, {device}): "
          f"{time.perf_counter() - t0:.3f} s")
    with open(os.path.join(args.outdir, "fp_spectrum.json"), "w") as f:
        json.dump({float(fr): float(v) for fr, v in zip(freqs, fp)}, f)

    # 4. null-distribution check: 2Fp ~ chi^2(2 N_psr) for noise-only
    #    data (the reference notebook's cell-5 validation)
    k = 2 * args.npsr
    mean, var = float(np.mean(2 * fp)), float(np.var(2 * fp))
    ks = ss.kstest((2 * fp - 0) / 1.0, ss.chi2(df=k).cdf)
    print(f"2Fp sample mean {mean:.1f} (chi2 k={k}); var {var:.1f} (2k={2*k})")
    print(f"KS test vs chi2({k}): D={ks.statistic:.3f} p={ks.pvalue:.3f}")

    # 5. small noise-marginalized sweep (run_nmfp flow)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    D = 32
    rng = np.random.default_rng(7)
    samples = {
        n: (rng.uniform(2, 6, D) if n.endswith("gamma")
            else rng.uniform(-16, -14, D))
        for n in pta.params
    }
    nm = NMFp(psrs, pta.rn_containers)
    cwfreqs = np.arange(1, 101) / pta.Tspan
    t0 = time.perf_counter()
    vals = nm.sweep(cwfreqs, samples, Nvecs, Ts, device=device)
    dt = time.perf_counter() - t0
    print(f"NM-Fp ({D} draws x 100 freqs): {dt:.3f} s "
          f"({D * 100 / dt:.0f} evals/s)")
    np.save(os.path.join(args.outdir, "nmfp.npy"), vals)
    print("outputs in", args.outdir)


if __name__ == "__main__":
    main()
