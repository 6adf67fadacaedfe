"""Memory-stability soak: repeated sweeps watching HBM allocation.

    python tools/soak.py [seconds]
"""

import sys
import time

import numpy as np
import torch

sys.path.insert(0, __import__("os").path.abspath(
    __import__("os").path.join(__import__("os").path.dirname(__file__), "..")))
from fastfp_amd import FpEngine, get_mats_nmfp, initialize_pta, make_synthetic_pta  # noqa: E402
from fastfp_amd.noise import batch_phiinv, check_batch_homogeneous  # noqa: E402


def main():
    secs = float(sys.argv[1]) if len(sys.argv) > 1 else 60.0
    psrs = make_synthetic_pta(npsr=67, ntoa=5000, ntm=60, seed=1)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.4
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    eng = FpEngine(psrs, Nvecs, Ts, device="cuda:0")
    eng.precompute(np.arange(1, 1001) / pta.Tspan)
    for c in pta.rn_containers:
        c.to("cuda:0")
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    homog = check_batch_homogeneous(pta.rn_containers)
    rng = np.random.default_rng(0)
    D = 1000
    t0 = time.time()
    it = 0
    m0 = None
    while time.time() - t0 < secs:
        pars = {
            n: torch.as_tensor(
                rng.uniform(2, 6, D) if n.endswith("gamma")
                else rng.uniform(-16, -14, D),
                dtype=torch.float64, device="cuda:0",
            )
            for n in pta.params
        }
        phiinvs = batch_phiinv(pta.rn_containers, pars, homogeneous=homog)
        fp = eng.sweep(phiinvs=phiinvs, draw_chunk=1024)
        assert torch.isfinite(fp).all()
        it += 1
        if it == 3:
            torch.cuda.synchronize()
            m0 = torch.cuda.memory_allocated()
    torch.cuda.synchronize()
    m1 = torch.cuda.memory_allocated()
    dt = time.time() - t0
    print(f"{it} sweeps of {D}x1000 in {dt:.1f}s "
          f"({it * D * 1000 / dt / 1e6:.1f}M evals/s); "
          f"mem after warmup {m0/1e9:.2f} GB -> final {m1/1e9:.2f} GB "
          f"(growth {(m1-m0)/1e6:.1f} MB)")
    assert m1 - m0 < 100e6, "memory growth!"


if __name__ == "__main__":
    main()
