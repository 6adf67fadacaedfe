"""Fp-statistic sweep CLI.

Argument and output parity with the reference script
(``/root/reference/examples/run_fp.py``): three positionals
(psrfile, noisefile, savefile), GW frequency grid
``linspace(2e-9, 3e-7, 200)`` by default, output ``{savefile}.json``
mapping frequency -> Fp.  Additions: ``--device``, ``--nfreqs``,
``--fmin/--fmax``, and multi-GPU frequency sharding via torchrun
(``python -m torch.distributed.run --nproc-per-node N -m
fastfp_amd.cli.run_fp ...``).
"""

import argparse
import json
import logging
import time

import numpy as np
import torch

from fastfp_amd.data import load_pulsars
from fastfp_amd.engine import FpEngine
from fastfp_amd.model import get_mats_fp, initialize_pta
from fastfp_amd.parallel import (
    all_gather_concat,
    cleanup,
    init_distributed,
    shard_slice,
)


def main(
    psrfile,
    noisefile,
    savefile,
    nfreqs=200,
    fmin=2e-9,
    fmax=3e-7,
    device=None,
    rn_comps=30,
    gwb_comps=30,
    ecorr_kernel=False,
):
    logging.basicConfig(format="%(levelname)s: %(message)s", level=logging.INFO)
    logger = logging.getLogger(__name__)

    rank, world, dev = init_distributed(
        device=torch.device(device) if device else None
    )
    if world > 1:
        # CPU-pinned precompute under N concurrent ranks: avoid
        # oversubscribed LAPACK threading (see bench.py)
        import os

        torch.set_num_threads(max(1, (os.cpu_count() or world) // world))
    logger.info(f"fastfp_amd backend device {dev} (rank {rank}/{world})")

    psrs = load_pulsars(psrfile)
    with open(noisefile, "r") as f:
        noise = json.load(f)

    # CURN parameters fixed as in the reference script
    # (/root/reference/examples/run_fp.py:43-44)
    noise["gw_gamma"] = 13 / 3
    noise["gw_log10_A"] = float(np.log10(2e-15))

    pta = initialize_pta(
        psrs, noise, inc_cp=True, rn_comps=rn_comps, gwb_comps=gwb_comps,
        ecorr_kernel=ecorr_kernel,
    )

    t0 = time.perf_counter()
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    phiinvs = pta.get_phiinv(noise)
    logger.info(f"Precompute matrix wall time: {time.perf_counter() - t0:.4f} s")

    freqs = np.linspace(fmin, fmax, nfreqs)
    local = freqs[shard_slice(nfreqs, rank, world)]

    t0 = time.perf_counter()
    eng = FpEngine(psrs, Nvecs, Ts, device=dev)
    eng.precompute(local)
    # the phiinv form lets homogeneous models use the pulsar-stacked
    # single-launch path; heterogeneous models fall back per pulsar
    fp_local = eng.sweep(phiinvs=phiinvs)
    fp = all_gather_concat(fp_local.reshape(-1), world).cpu().numpy()
    logger.info(f"Fp-statistic wall time: {time.perf_counter() - t0:.4f} s")

    if rank == 0:
        res = {float(fr): float(v) for fr, v in zip(freqs, fp)}
        with open(f"{savefile}.json", "w") as f:
            json.dump(res, f)
    cleanup()
    return


def cli():
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("psrfile", type=str, help="pulsars file (.pkl/.npz/dir)")
    parser.add_argument("noisefile", type=str, help="noise dictionary json")
    parser.add_argument("savefile", type=str, help="output json path (no ext)")
    parser.add_argument("--nfreqs", type=int, default=200)
    parser.add_argument("--fmin", type=float, default=2e-9)
    parser.add_argument("--fmax", type=float, default=3e-7)
    parser.add_argument("--device", type=str, default=None)
    parser.add_argument("--rn_comps", type=int, default=30)
    parser.add_argument("--gwb_comps", type=int, default=30)
    parser.add_argument("--ecorr_kernel", action="store_true",
                        help="model ECORR as block-diagonal white noise")
    main(**vars(parser.parse_args()))


if __name__ == "__main__":
    cli()
