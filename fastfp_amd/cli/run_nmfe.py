"""Noise-marginalized Fe CLI: sky-coherent CW search over MCMC draws.

Completes the CLI family (run_fp / run_nmfp / run_fe / run_nmfe); no
reference counterpart exists (the reference has neither Fe nor its
marginalized form).  Input contract mirrors ``run_nmfp`` (psrfile,
noisefile, chainfile, savefile; 25% burn-in, last 4 bookkeeping
columns stripped); the sky grid mirrors ``run_fe``.  Output:
``{outdir}/{savefile}.npy`` of shape (nsamples, nsky, nfreqs) plus
``{outdir}/{savefile}.meta.json`` with the freqs/sky axes.
"""

import argparse
import json
import logging
import os
import time

import numpy as np
import torch

from fastfp_amd.cli.run_fe import fibonacci_sky
from fastfp_amd.data import get_tspan, load_pulsars
from fastfp_amd.festat import NMFe
from fastfp_amd.model import get_mats_nmfp, initialize_pta
from fastfp_amd.parallel import (
    all_gather_concat,
    cleanup,
    init_distributed,
    shard_slice,
)


def main(
    psrfile,
    noisefile,
    chainfile,
    savefile,
    inc_cp=False,
    nrncomps=30,
    ngwbcomps=30,
    ncwfreqs=100,
    nsamples=100,
    nsky=48,
    theta=None,
    phi=None,
    outdir="res",
    device=None,
    seed=0,
    batch_size=64,
):
    logging.basicConfig(format="%(levelname)s: %(message)s", level=logging.INFO)
    logger = logging.getLogger(__name__)

    rank, world, dev = init_distributed(
        device=torch.device(device) if device else None
    )
    if world > 1:
        torch.set_num_threads(max(1, (os.cpu_count() or world) // world))
    logger.info(f"fastfp_amd backend device {dev} (rank {rank}/{world})")

    psrs = load_pulsars(psrfile)
    with open(noisefile, "r") as f:
        noise = json.load(f)
    chain = np.loadtxt(chainfile)
    if chain.ndim == 1:
        chain = chain[None, :]
    burn = int(0.25 * chain.shape[0])

    noise["gw_gamma"] = 13 / 3
    noise["gw_log10_A"] = float(np.log10(2e-15))

    Tspan = get_tspan(psrs)
    pta = initialize_pta(
        psrs, noise, inc_cp=inc_cp, rn_comps=nrncomps, gwb_comps=ngwbcomps
    )

    t0 = time.perf_counter()
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    logger.info(f"Precompute matrix wall time: {time.perf_counter() - t0:.2f} s")

    freqs = np.arange(1, ncwfreqs + 1) / Tspan
    sky = (
        [(float(theta), float(phi))]
        if theta is not None and phi is not None
        else fibonacci_sky(nsky)
    )

    rng = np.random.default_rng(seed)
    n_avail = chain.shape[0] - burn
    idxs = rng.choice(np.arange(burn, chain.shape[0]), size=nsamples,
                      replace=nsamples > n_avail)
    rns_full = chain[idxs, :-4].T  # (nparams, nsamples)

    my = shard_slice(nsamples, rank, world)
    my_idx = np.arange(nsamples)[my]

    t0 = time.perf_counter()
    nm = NMFe(psrs, pta.rn_containers)
    parts = []
    for lo in range(0, len(my_idx), batch_size):
        sel = my_idx[lo : lo + batch_size]
        samples = {
            p: rns_full[ct, sel] for ct, p in enumerate(pta.params)
        }
        parts.append(nm.sweep(freqs, sky, samples, Nvecs, Ts, device=dev))
    local = (
        np.concatenate(parts, axis=0) if parts
        else np.zeros((0, len(sky), ncwfreqs))
    )
    t_local = torch.as_tensor(local, dtype=torch.float64)
    full = all_gather_concat(t_local, world, dim=0).numpy()
    logger.info(
        "Noise marginalized Fe-statistic wall time: "
        f"{time.perf_counter() - t0:.2f} s"
    )

    if rank == 0:
        os.makedirs(outdir, exist_ok=True)
        with open(os.path.join(outdir, f"{savefile}.npy"), "wb") as f:
            np.save(f, full)
        with open(os.path.join(outdir, f"{savefile}.meta.json"), "w") as f:
            json.dump({"freqs": [float(x) for x in freqs],
                       "sky": [[t, p] for t, p in sky]}, f)
    cleanup()
    return


def cli():
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("psrfile", type=str, help="pulsars file (.pkl/.npz/dir)")
    parser.add_argument("noisefile", type=str, help="noise dictionary json")
    parser.add_argument("chainfile", type=str, help="MCMC chain text file")
    parser.add_argument("savefile", type=str, help="output .npy name (no ext)")
    parser.add_argument("--inc_cp", action="store_true", help="include CURN")
    parser.add_argument("--nrncomps", type=int, default=30)
    parser.add_argument("--ngwbcomps", type=int, default=30)
    parser.add_argument("--ncwfreqs", type=int, default=100)
    parser.add_argument("--nsamples", type=int, default=100)
    parser.add_argument("--nsky", type=int, default=48)
    parser.add_argument("--theta", type=float, default=None)
    parser.add_argument("--phi", type=float, default=None)
    parser.add_argument("--outdir", type=str, default="res")
    parser.add_argument("--device", type=str, default=None)
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--batch_size", type=int, default=64)
    main(**vars(parser.parse_args()))


if __name__ == "__main__":
    cli()
