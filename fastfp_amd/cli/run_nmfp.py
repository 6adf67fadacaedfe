"""Noise-marginalized Fp CLI.

Argument and output parity with the reference script
(``/root/reference/examples/run_nmfp.py``): positionals
(psrfile, noisefile, chainfile, savefile), flags ``--inc_ecorr
--inc_cp --nrncomps --ngwbcomps --ncwfreqs --nsamples --batch_size``,
chain handling (25% burn-in, last 4 bookkeeping columns stripped,
``run_nmfp.py:221-254``), CW frequency grid ``k/Tspan``, and output
``{outdir}/{savefile}.npy`` of shape (nsamples, nfreqs).

Additions over the reference: ``--outdir`` (the reference hard-codes
``res/``), ``--device``, ``--seed``, checkpoint/resume of completed
draw batches (``--resume``; per-batch shards are written to
``{outdir}/.{savefile}.batches/``), and multi-GPU draw sharding via
torchrun.
"""

import argparse
import json
import logging
import os
import time

import numpy as np
import torch

from fastfp_amd.data import get_tspan, load_pulsars
from fastfp_amd.engine import FpEngine
from fastfp_amd.model import get_mats_nmfp, initialize_pta
from fastfp_amd.nmfp import NMFp
from fastfp_amd.parallel import (
    all_gather_concat,
    cleanup,
    init_distributed,
    shard_slice,
)


def setup_fp_model(psrs, noise, Tspan=None, add_ecorr=False, nrncomps=30,
                   add_curn=False, ngwbcomps=5, pta=None):
    """Build the NMFp object and its containers — signature parity with
    the reference's ``setup_fp_model`` (``run_nmfp.py:73-171``).  Here
    the containers already live on the PTAModel; reuse them when a
    ``pta`` is passed, otherwise build a fresh model."""
    if pta is None:
        pta = initialize_pta(
            psrs,
            noise,
            inc_cp=add_curn,
            rn_comps=nrncomps,
            gwb_comps=ngwbcomps,
            inc_ecorr=add_ecorr,
            # reference semantics: Tspan=None -> per-pulsar bases
            # (run_nmfp.py:94-98); a set Tspan -> one shared grid
            per_psr_tspan=Tspan is None,
        )
    return NMFp(psrs, pta.rn_containers)


def map_params(pta, xs):
    """2-D sample batch -> dict of per-parameter arrays (parity with
    ``run_nmfp.py:174-186``)."""
    xs = np.asarray(xs)
    if xs.ndim > 1:
        return {p: xs[ct, :] for ct, p in enumerate(pta.params)}
    return pta.map_params(xs)


def main(
    psrfile,
    noisefile,
    chainfile,
    savefile,
    inc_ecorr=False,
    inc_cp=False,
    nrncomps=30,
    ngwbcomps=30,
    ncwfreqs=100,
    nsamples=1000,
    batch_size=100,
    outdir="res",
    device=None,
    seed=0,
    resume=False,
    ecorr_kernel=False,
    checkpoint=True,
):
    logging.basicConfig(format="%(levelname)s: %(message)s", level=logging.INFO)
    logger = logging.getLogger(__name__)

    rank, world, dev = init_distributed(
        device=torch.device(device) if device else None
    )
    if world > 1:
        # CPU-pinned precompute under N concurrent ranks: avoid
        # oversubscribed LAPACK threading (see bench.py)
        torch.set_num_threads(max(1, (os.cpu_count() or world) // world))
    logger.info(f"fastfp_amd backend device {dev} (rank {rank}/{world})")
    logger.info(f"number of CW frequencies: {ncwfreqs}")
    logger.info(f"number of samples: {nsamples}")
    logger.info(f"batch_size: {batch_size}")

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
        psrs,
        noise,
        inc_cp=inc_cp,
        rn_comps=nrncomps,
        gwb_comps=ngwbcomps,
        inc_ecorr=inc_ecorr,
        ecorr_kernel=ecorr_kernel,
    )
    nmfp = setup_fp_model(psrs, noise, pta=pta)

    t0 = time.perf_counter()
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    logger.info(f"Precompute matrix wall time: {time.perf_counter() - t0:.2f} s")

    freqs = np.arange(1, ncwfreqs + 1) / Tspan

    # draw nsamples rows from the chain (after burn-in, bookkeeping
    # columns stripped)
    rng = np.random.default_rng(seed)
    idxs = rng.choice(
        np.arange(burn, chain.shape[0]),
        size=min(nsamples, chain.shape[0] - burn),
        replace=nsamples > chain.shape[0] - burn,
    )
    if len(idxs) < nsamples:
        idxs = rng.choice(np.arange(burn, chain.shape[0]), nsamples, replace=True)
    rns_full = chain[idxs, :-4].T  # (nparams, nsamples)

    # shard draws across ranks
    my = shard_slice(nsamples, rank, world)
    my_idx = np.arange(nsamples)[my]

    batch_dir = os.path.join(outdir, f".{savefile}.batches")
    # Resume safety: batch shards are keyed by (rank, batch offset)
    # only, so a manifest fingerprints everything that changes their
    # CONTENT — re-running with a different seed/chain/model silently
    # mixing stale shards into the output would be far worse than
    # recomputing.  On mismatch the stale shards are ignored (resume
    # effectively restarts) and the manifest is rewritten.
    import hashlib

    try:
        chain_stat = os.stat(chainfile)
        chain_fp = f"{chain_stat.st_size}:{chain_stat.st_mtime_ns}"
    except OSError:
        chain_fp = "unknown"
    manifest_key = hashlib.sha256(
        repr(
            dict(
                seed=seed, nsamples=nsamples, batch_size=batch_size,
                world=world, ncwfreqs=ncwfreqs, chain=chain_fp,
                inc_ecorr=inc_ecorr, inc_cp=inc_cp, nrncomps=nrncomps,
                ngwbcomps=ngwbcomps, ecorr_kernel=ecorr_kernel,
                psrfile=os.path.abspath(psrfile),
                noisefile=os.path.abspath(noisefile),
            )
        ).encode()
    ).hexdigest()[:16]
    manifest_path = os.path.join(batch_dir, "MANIFEST")
    if rank == 0:
        os.makedirs(outdir, exist_ok=True)
    if rank == 0 and checkpoint:
        os.makedirs(batch_dir, exist_ok=True)
        prev = None
        if checkpoint and os.path.exists(manifest_path):
            with open(manifest_path) as f:
                prev = f.read().strip()
        if checkpoint and prev != manifest_key:
            stale = [
                fn for fn in os.listdir(batch_dir)
                if fn.endswith(".npy")
            ]
            if stale and prev is not None:
                logger.warning(
                    "resume: run configuration changed (manifest %s -> %s); "
                    "removing %d stale batch shards",
                    prev, manifest_key, len(stale),
                )
            for fn in stale:
                os.remove(os.path.join(batch_dir, fn))
            with open(manifest_path, "w") as f:
                f.write(manifest_key)
    if world > 1:
        torch.distributed.barrier()

    for cont in pta.rn_containers:
        cont.to(dev)
    eng = FpEngine(psrs, Nvecs, Ts, device=dev)
    eng.precompute(freqs)
    # Schur draw compression: per-draw solves run at the variable-bin
    # dimension only (docs/DESIGN.md)
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )

    t0 = time.perf_counter()
    parts = []
    try:
        from tqdm import tqdm

        batches = tqdm(range(0, len(my_idx), batch_size),
                       disable=rank != 0, desc="draw batches")
    except ImportError:
        batches = range(0, len(my_idx), batch_size)
    # checkpoint writes happen on a background thread so disk I/O
    # overlaps the next batch's GPU sweep (the CLI-vs-bench gap was
    # dominated by synchronous np.save + host gather; docs/PERFORMANCE.md)
    from concurrent.futures import ThreadPoolExecutor

    writer = ThreadPoolExecutor(max_workers=1) if checkpoint else None
    pending = []
    timing = os.environ.get("FASTFP_CLI_TIMING") == "1"
    tacc = {"map": 0.0, "sweep": 0.0, "save": 0.0}
    for lo in batches:
        sel = my_idx[lo : lo + batch_size]
        ck = os.path.join(batch_dir, f"r{rank}_b{lo}.npy")
        if checkpoint and resume and os.path.exists(ck):
            parts.append(np.load(ck))
            continue
        tm0 = time.perf_counter()
        samples = map_params(pta, rns_full[:, sel])
        tm1 = time.perf_counter()
        vals = nmfp.sweep(freqs, samples, Nvecs, Ts, engine=eng)
        tm2 = time.perf_counter()
        if writer is not None:
            pending.append(writer.submit(np.save, ck, vals))
        parts.append(vals)
        if timing:
            tacc["map"] += tm1 - tm0
            tacc["sweep"] += tm2 - tm1
    if timing:
        logger.info(f"CLI timing: map {tacc['map']:.3f} s, "
                    f"sweep {tacc['sweep']:.3f} s")
    if writer is not None:
        for fut in pending:
            fut.result()  # surface write errors before declaring success
        writer.shutdown()
    local_vals = (
        np.vstack(parts) if parts else np.zeros((0, ncwfreqs))
    )
    t_local = torch.as_tensor(local_vals, dtype=torch.float64, device=dev)
    full = all_gather_concat(t_local, world, dim=0).cpu().numpy()
    logger.info(
        f"Noise marginalized Fp-statistic wall time: {time.perf_counter() - t0:.2f} s"
    )

    if rank == 0:
        if not np.isfinite(full).all():
            logger.warning(
                "non-finite Fp values in output (%d of %d) — check the "
                "noise model / chain for invalid parameters",
                int(np.sum(~np.isfinite(full))), full.size,
            )
        with open(os.path.join(outdir, f"{savefile}.npy"), "wb") as f:
            np.save(f, full)
    cleanup()
    return


def cli():
    parser = argparse.ArgumentParser(description=__doc__)
    parser.add_argument("psrfile", type=str, help="pulsars file (.pkl/.npz/dir)")
    parser.add_argument("noisefile", type=str, help="noise dictionary json")
    parser.add_argument("chainfile", type=str, help="MCMC chain text file")
    parser.add_argument("savefile", type=str, help="output .npy name (no ext)")
    parser.add_argument("--inc_ecorr", action="store_true", help="include ECORR")
    parser.add_argument("--inc_cp", action="store_true", help="include CURN process")
    parser.add_argument("--nrncomps", type=int, default=30)
    parser.add_argument("--ngwbcomps", type=int, default=30)
    parser.add_argument("--ncwfreqs", type=int, default=100)
    parser.add_argument("--nsamples", type=int, default=1000)
    parser.add_argument("--batch_size", type=int, default=100)
    parser.add_argument("--outdir", type=str, default="res")
    parser.add_argument("--device", type=str, default=None)
    parser.add_argument("--seed", type=int, default=0)
    parser.add_argument("--resume", action="store_true",
                        help="skip draw batches already on disk")
    parser.add_argument("--ecorr_kernel", action="store_true",
                        help="model ECORR as block-diagonal white noise "
                             "(EcorrKernelNoise; the reference's "
                             "unsupported case)")
    parser.add_argument("--no-checkpoint", dest="checkpoint",
                        action="store_false",
                        help="disable per-batch checkpoint shards "
                             "(faster; --resume unavailable)")
    main(**vars(parser.parse_args()))


if __name__ == "__main__":
    cli()
