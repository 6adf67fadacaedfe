"""Measure the production-CLI vs bench throughput gap at the BASELINE
config-3 shape (67 psr, 1e4 draws x 1e3 freqs) — VERDICT r01 item 7.

Generates bench-shape inputs on disk, runs ``run_nmfp.main`` end to
end (chain load, model build, precompute, sharded sweep, checkpoint
shards, .npy output) and reports the sweep-phase throughput to compare
against bench.py's ~47M evals/s.

Run on a GPU box:  python tools/cli_gap.py [--no-checkpoint]
"""

import argparse
import json
import os
import sys
import time

import numpy as np

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from fastfp_amd.cli import run_nmfp
from fastfp_amd.data import make_synthetic_pta, save_pulsars
from fastfp_amd.model import initialize_pta


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--no-checkpoint", action="store_true")
    ap.add_argument("--nsamples", type=int, default=10000)
    ap.add_argument("--ncwfreqs", type=int, default=1000)
    ap.add_argument("--batch-size", type=int, default=1000)
    ap.add_argument("--workdir", default="/tmp/cli_gap")
    args = ap.parse_args()

    os.makedirs(args.workdir, exist_ok=True)
    psrs = make_synthetic_pta(npsr=67, ntoa=5000, ntm=60, seed=1234,
                              ragged=True)
    psrfile = os.path.join(args.workdir, "psrs.npz")
    save_pulsars(psrs, psrfile)
    noise = {}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 13.0 / 3.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    noisefile = os.path.join(args.workdir, "noise.json")
    with open(noisefile, "w") as f:
        json.dump(noise, f)
    # synthetic chain with the right column count: params sorted by the
    # PTA + 4 bookkeeping cols
    pta = initialize_pta(psrs, dict(noise, gw_gamma=13 / 3,
                                    gw_log10_A=float(np.log10(2e-15))),
                         inc_cp=True, rn_comps=30, gwb_comps=30)
    nparams = len(pta.params)
    rng = np.random.default_rng(0)
    chain = np.zeros((2000, nparams + 4))
    for i, name in enumerate(pta.params):
        chain[:, i] = (rng.uniform(2.0, 6.0, 2000) if name.endswith("gamma")
                       else rng.uniform(-16.0, -13.5, 2000))
    chainfile = os.path.join(args.workdir, "chain.txt")
    np.savetxt(chainfile, chain)

    t0 = time.perf_counter()
    run_nmfp.main(
        psrfile, noisefile, chainfile, "nm",
        inc_cp=True, nrncomps=30, ngwbcomps=30,
        ncwfreqs=args.ncwfreqs, nsamples=args.nsamples,
        batch_size=args.batch_size,
        outdir=os.path.join(args.workdir, "res"),
        checkpoint=not args.no_checkpoint,
    )
    wall = time.perf_counter() - t0
    evals = args.nsamples * args.ncwfreqs
    print(f"TOTAL wall {wall:.2f} s; {evals} evals; "
          f"end-to-end {evals / wall / 1e6:.1f}M evals/s "
          f"(sweep-phase rate in the log line above)")


if __name__ == "__main__":
    main()
