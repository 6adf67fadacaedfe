"""Fe-statistic sweep CLI: sky-coherent Earth-term CW search.

Companion to ``run_fp`` for the Fe statistic (the reference's open
to-do, ``/root/reference/README.md:23`` — no reference counterpart
exists).  Same input contract as ``run_fp`` (psrfile, noisefile,
savefile; CURN parameters fixed as in the reference script); sweeps
the (frequency x sky) grid and writes ``{savefile}.json``:

    {"freqs": [...], "sky": [[theta, phi], ...],
     "fe": [[Fe(sky0, f0), ...], ...]}        # (nsky, nfreqs)

The sky grid is an equal-area-ish lattice of ``--nsky`` points
(Fibonacci sphere), or a single ``--theta/--phi`` location.  Pulsars
must carry sky positions (``PulsarData.pos``).
"""

import argparse
import json
import logging
import time

import numpy as np
import torch

from fastfp_amd.data import load_pulsars
from fastfp_amd.festat import FastFe
from fastfp_amd.model import get_mats_fp, initialize_pta
from fastfp_amd.parallel import (
    all_gather_concat,
    cleanup,
    init_distributed,
    shard_slice,
)


def fibonacci_sky(nsky: int) -> list:
    """~Equal-area sky lattice: (theta, phi) colatitude/longitude pairs
    on the Fibonacci sphere."""
    ga = np.pi * (3.0 - np.sqrt(5.0))
    k = np.arange(nsky)
    z = 1.0 - (2.0 * k + 1.0) / nsky
    theta = np.arccos(z)
    phi = np.mod(ga * k, 2.0 * np.pi)
    return [(float(t), float(p)) for t, p in zip(theta, phi)]


def main(
    psrfile,
    noisefile,
    savefile,
    nfreqs=200,
    fmin=2e-9,
    fmax=3e-7,
    nsky=48,
    theta=None,
    phi=None,
    device=None,
    rn_comps=30,
    gwb_comps=30,
):
    logging.basicConfig(format="%(levelname)s: %(message)s", level=logging.INFO)
    logger = logging.getLogger(__name__)

    rank, world, dev = init_distributed(
        device=torch.device(device) if device else None
    )
    if world > 1:
        import os

        torch.set_num_threads(max(1, (os.cpu_count() or world) // world))
    logger.info(f"fastfp_amd backend device {dev} (rank {rank}/{world})")

    psrs = load_pulsars(psrfile)
    with open(noisefile, "r") as f:
        noise = json.load(f)
    noise["gw_gamma"] = 13 / 3
    noise["gw_log10_A"] = float(np.log10(2e-15))

    pta = initialize_pta(
        psrs, noise, inc_cp=True, rn_comps=rn_comps, gwb_comps=gwb_comps
    )

    t0 = time.perf_counter()
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    logger.info(f"Precompute matrix wall time: {time.perf_counter() - t0:.4f} s")

    freqs = np.linspace(fmin, fmax, nfreqs)
    sky = (
        [(float(theta), float(phi))]
        if theta is not None and phi is not None
        else fibonacci_sky(nsky)
    )
    # shard the SKY axis across ranks (the engine pass is per-rank;
    # each sky point is O(P) assembly on top of it)
    local_sky = sky[shard_slice(len(sky), rank, world)]

    t0 = time.perf_counter()
    fe = FastFe(psrs, pta)
    local = fe.sweep(freqs, local_sky, Nvecs, Ts, sigmas, device=dev) \
        if local_sky else np.zeros((0, nfreqs))
    t_local = torch.as_tensor(local, dtype=torch.float64)
    full = all_gather_concat(t_local, world, dim=0).numpy()
    logger.info(f"Fe-statistic wall time: {time.perf_counter() - t0:.4f} s")

    if rank == 0:
        out = {
            "freqs": [float(f) for f in freqs],
            "sky": [[t, p] for t, p in sky],
            "fe": full.tolist(),
        }
        with open(f"{savefile}.json", "w") as f:
            json.dump(out, f)
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
    parser.add_argument("--nsky", type=int, default=48,
                        help="Fibonacci-sphere sky points")
    parser.add_argument("--theta", type=float, default=None,
                        help="single-sky colatitude (with --phi)")
    parser.add_argument("--phi", type=float, default=None,
                        help="single-sky longitude (with --theta)")
    parser.add_argument("--device", type=str, default=None)
    parser.add_argument("--rn_comps", type=int, default=30)
    parser.add_argument("--gwb_comps", type=int, default=30)
    main(**vars(parser.parse_args()))


if __name__ == "__main__":
    cli()
