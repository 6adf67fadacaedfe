"""Phase timing inside the CLI sweep path (NMFp.sweep) at the BASELINE
config-3 shape — locates the CLI-vs-bench gap.

Run on a GPU box: python tools/cli_sweep_phases.py [D]
"""

import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from fastfp_amd import FpEngine, get_mats_nmfp, initialize_pta, make_synthetic_pta
from fastfp_amd.cli.run_nmfp import setup_fp_model
from fastfp_amd.noise import batch_phiinv, check_batch_homogeneous

DEV = "cuda:0"


def tsec(fn):
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    out = fn()
    torch.cuda.synchronize()
    return time.perf_counter() - t0, out


def main():
    D = int(sys.argv[1]) if len(sys.argv) > 1 else 10000
    F = 1000
    psrs = make_synthetic_pta(npsr=67, ntoa=5000, ntm=60, seed=1234, ragged=True)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 13.0 / 3.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=30, gwb_comps=30)
    nmfp = setup_fp_model(psrs, noise, pta=pta)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    for c in pta.rn_containers:
        c.to(DEV)
    eng = FpEngine(psrs, Nvecs, Ts, device=DEV)
    freqs = np.arange(1, F + 1) / pta.Tspan
    eng.precompute(freqs)
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    print("comp stack:", eng._comp_stack is not None)

    # CLI-style numpy samples
    rng = np.random.default_rng(0)
    samples = {
        n: (rng.uniform(2, 6, D) if n.endswith("gamma")
            else rng.uniform(-16, -13.5, D))
        for n in pta.params
    }

    homog = check_batch_homogeneous(nmfp.rn_sigs)
    t, phiinvs = tsec(lambda: batch_phiinv(nmfp.rn_sigs, samples,
                                           homogeneous=homog))
    print(f"batch_phiinv (numpy in): {t*1e3:8.1f} ms  homog={homog}")
    print("  phi device:", phiinvs[0].device, phiinvs[0].shape)
    t, _ = tsec(lambda: eng.compression_margin(phiinvs))
    print(f"compression_margin     : {t*1e3:8.1f} ms")
    for chunk in (512, 1024):
        t, fp = tsec(lambda: eng.sweep(phiinvs=phiinvs, draw_chunk=chunk))
        print(f"eng.sweep chunk={chunk:5d} : {t*1e3:8.1f} ms "
              f"({D*F/t/1e6:.1f}M evals/s)")
    t, _ = tsec(lambda: fp.cpu().numpy())
    print(f"fp .cpu().numpy()      : {t*1e3:8.1f} ms")
    # device-tensor samples for comparison (the bench's pool format)
    dsamples = {k: torch.as_tensor(v, dtype=torch.float64, device=DEV)
                for k, v in samples.items()}
    t, phiinvs2 = tsec(lambda: batch_phiinv(nmfp.rn_sigs, dsamples,
                                            homogeneous=homog))
    print(f"batch_phiinv (dev in)  : {t*1e3:8.1f} ms")
    t, _ = tsec(lambda: eng.sweep(phiinvs=phiinvs2, draw_chunk=1024))
    print(f"eng.sweep (dev phi)    : {t*1e3:8.1f} ms")


if __name__ == "__main__":
    main()
