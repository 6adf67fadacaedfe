"""Per-pulsar compression-probe error distribution at the benchmark
shape — decides the probe tolerance (docs/TUNING_NOTES.md).

    python tools/probe_diag.py [npsr] [ntoa]
"""

import os
import sys

import numpy as np
import torch

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from fastfp_amd.data import make_synthetic_pta  # noqa: E402
from fastfp_amd.engine import FpEngine, _t64  # noqa: E402
from fastfp_amd.model import get_mats_nmfp, initialize_pta  # noqa: E402


def main():
    npsr = int(sys.argv[1]) if len(sys.argv) > 1 else 67
    ntoa = int(sys.argv[2]) if len(sys.argv) > 2 else 5000
    device = "cuda" if torch.cuda.is_available() else "cpu"
    psrs = make_synthetic_pta(npsr=npsr, ntoa=ntoa, tspan_yr=15.0, ntm=8,
                              seed=1234, ragged=True)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 13.0 / 3.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=30, gwb_comps=14)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    eng = FpEngine(psrs, Nvecs, Ts, device=device)
    freqs = np.arange(1, 1001) / pta.Tspan
    eng.precompute(freqs)
    probe = {k: v for k, v in noise.items() if k in pta.params}
    var_slices = [c.var_slice for c in pta.rn_containers]
    pfs = [c.get_phiinv(probe).to(device) for c in pta.rn_containers]
    eng.enable_draw_compression(var_slices, pfs)

    errs = []
    F = eng.freqs.shape[0]
    for i, (blk, pf) in enumerate(zip(eng.blocks, pfs)):
        if blk.comp is None:
            errs.append((i, float("nan")))
            continue
        pinv = _t64(pf, eng.device).reshape(1, -1)
        c = blk.comp
        fpA = torch.zeros((1, F), dtype=torch.float64, device=eng.device)
        fpB = torch.zeros_like(fpA)
        phi_var = (1.0 / (pinv[:, c["var"]] - c["delta0"][None, :])).contiguous()
        if eng._use_hip:
            from fastfp_amd import ops

            ops.chol_trsm_fp_accum(c["G"], phi_var, c["K"], c["M0"], c["N0"],
                                   fpA, gsign=-1.0)
            ops.chol_trsm_fp_accum(blk.TNT, pinv.contiguous(), blk.RHS,
                                   blk.sNs, blk.sNr, fpB, gsign=1.0)
        else:
            sigc = c["G"][None, :, :] + torch.diag_embed(phi_var)
            eng._accum_eager_mats(sigc, c["K"], c["M0"], c["N0"], fpA, -1.0)
            sigma = blk.TNT[None, :, :] + torch.diag_embed(pinv)
            eng._accum_eager(blk, sigma, fpB)
        scale = fpB.abs().max().clamp_min(1e-30)
        errs.append((i, float((fpA - fpB).abs().max() / scale)))

    vals = np.array([e for _, e in errs])
    ncomp = sum(1 for b in eng.blocks if b.comp is not None)
    print(f"device={device} npsr={npsr} kept={ncomp}/{npsr}")
    print(f"err: median={np.nanmedian(vals):.3e} p90={np.nanpercentile(vals, 90):.3e} "
          f"max={np.nanmax(vals):.3e}")
    worst = sorted(errs, key=lambda t: -(t[1] if t[1] == t[1] else 0))[:10]
    print("worst pulsars:", [(i, f"{e:.2e}") for i, e in worst])


if __name__ == "__main__":
    main()
