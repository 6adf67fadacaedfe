"""fastfp_amd flagship benchmark — the BASELINE.json headline metric.

Metric: **Fp evals/sec** (= frequencies x draws evaluated per second) on
the 67-pulsar noise-marginalized Fp sweep (BASELINE.json config 3:
1e4 draws x 1e3 freqs at 8 GPUs).  One "step" = one draw batch of
``--draws-per-step`` MCMC noise draws evaluated at all ``--freqs`` CW
frequencies across all pulsars (phi(theta) assembly + Sigma assembly +
batched Cholesky + fused triangular-solve/2x2-reduction, all on device).
Weak scaling: each rank owns its own draw batch; the final RCCL
all-gather of the per-rank (D_local, F) spectrum shards is included in
the timed region.

Reference baseline: ~6.0 Fp evals/s (200 freqs / 33.56 s, 1 draw,
unknown hardware — BASELINE.md); vs_baseline = value / 6.0.

Launch (driver contract):
  python bench.py --gpus 1 --steps K --warmup W
  python -m torch.distributed.run --nnodes=1 --nproc-per-node N \
      --master-addr 127.0.0.1 bench.py --gpus N --steps K --warmup W

Data: synthetic epoch-structured PTA (random-init) of the BASELINE
config shape — there is no network for real datasets.
"""

import argparse
import json
import time

import numpy as np
import torch

from fastfp_amd.data import make_synthetic_pta
from fastfp_amd.engine import FpEngine
from fastfp_amd.model import get_mats_nmfp, initialize_pta
from fastfp_amd.noise import batch_phiinv, check_batch_homogeneous
from fastfp_amd.parallel import all_gather_concat, cleanup, init_distributed


#: BASELINE.json config presets (see BASELINE.md / docs/DESIGN.md §5)
PRESETS = {
    # config 3 (default): 67-psr NMFp, 1e3 freqs, draw batches
    "nmfp67": {},
    # config 2: 45-psr 1e4-frequency Fp sweep, one noise draw
    # (draw compression only pays for D > 1)
    "fp45": dict(npsr=45, freqs=10000, draws_per_step=1, no_compress=True),
    # config 4: 67-psr NMFp with ECORR as block-diagonal white noise
    "ecorr67": dict(ecorr_kernel=True),
    # config 5: 200-psr SKA-scale, 1e5-frequency sweep (HBM sizing)
    "ska200": dict(npsr=200, ntoa=10000, freqs=100000, draws_per_step=1,
                   no_compress=True),
}


def build_problem(args, device, rank):
    psrs = make_synthetic_pta(
        npsr=args.npsr,
        ntoa=args.ntoa,
        tspan_yr=15.0,
        ntm=args.ntm,
        seed=1234,  # same PTA on every rank
        ragged=True,
    )
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 13.0 / 3.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
        if args.ecorr_kernel:
            for b in np.unique(p.backend_flags):
                noise[f"{p.name}_basis_ecorr_{b}_log10_ecorr"] = -6.5
    pta = initialize_pta(
        psrs, noise, inc_cp=True, rn_comps=args.rn_comps,
        gwb_comps=args.gwb_comps, ecorr_kernel=args.ecorr_kernel,
    )
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)

    eng = FpEngine(psrs, Nvecs, Ts, device=device)
    freqs = np.arange(1, args.freqs + 1) / pta.Tspan
    eng.precompute(freqs, freq_chunk=args.freq_chunk)
    if not args.no_compress:
        probe = {k: v for k, v in noise.items() if k in pta.params}
        eng.enable_draw_compression(
            [c.var_slice for c in pta.rn_containers],
            [c.get_phiinv(probe).to(device) for c in pta.rn_containers],
        )

    for cont in pta.rn_containers:
        cont.to(device)

    # pre-generate a pool of noise-parameter draws on device (rank-seeded:
    # each rank marginalizes over its own draws -- weak scaling).  The
    # pool holds ``--pool-rotations`` independent draw batches; every
    # step runs on a DIFFERENT batch (rotated into the active buffers
    # before each graph replay) so no timed step recomputes the
    # previous step's inputs.
    rng = np.random.default_rng(1000 + rank)
    D = args.draws_per_step
    R = max(1, args.pool_rotations)
    pool = {}
    active = {}
    for name in pta.params:
        if name.endswith("gamma"):
            v = rng.uniform(1.0, 6.5, (R, D))
        else:
            v = rng.uniform(-16.0, -13.5, (R, D))
        pool[name] = torch.as_tensor(v, dtype=torch.float64, device=device)
        active[name] = pool[name][0].clone()
    # syncing check, done once here so the captured step never syncs
    pta._phi_homog = check_batch_homogeneous(pta.rn_containers)
    # one-time validity guard on the WHOLE pool (outside the timed
    # region): every draw must clear the compression margin the
    # accuracy evidence covers (engine.compression_margin docstring)
    # Pool validity: the captured bench step runs the compressed path
    # unconditionally, so every pool draw must clear the compression
    # guard (production's NMFp.sweep splits such draws onto the direct
    # path per draw; a graph-captured step cannot).  Prior-corner draws
    # (margin < 1.5; ~0.1-1% of the uniform box) are RE-DRAWN from the
    # same prior — the pool is the prior conditioned on the
    # compressed-valid region, and the redraw count is reported in the
    # output JSON.  The finite-spectrum assert backstops validity.
    args._pool_margin = None
    args._pool_redrawn = 0
    if not args.no_compress and any(b.comp is not None for b in eng.blocks):
        margin = None
        for _ in range(50):
            flat = {k: v.reshape(-1) for k, v in pool.items()}
            piv = batch_phiinv(pta.rn_containers, flat,
                               homogeneous=pta._phi_homog)
            piv = [p if p.dim() == 2 else p[None] for p in piv]
            margins = eng.compression_margin_per_draw(piv)
            bad = margins < 1.5
            nbad = int(bad.sum())
            if nbad == 0:
                margin = float(margins.min())
                break
            args._pool_redrawn += nbad
            idx = torch.nonzero(bad).reshape(-1)
            for name in pta.params:
                if name.endswith("gamma"):
                    v = rng.uniform(1.0, 6.5, nbad)
                else:
                    v = rng.uniform(-16.0, -13.5, nbad)
                pool[name].view(-1)[idx] = torch.as_tensor(
                    v, dtype=torch.float64, device=device
                )
        assert margin is not None and margin > 1.5, \
            f"pool margin {margin} after redraws"
        args._pool_margin = margin
        for name in active:
            active[name].copy_(pool[name][0])
    return pta, eng, pool, active


def run_step(pta, eng, pool, args, fp_accum):
    phiinvs = batch_phiinv(pta.rn_containers, pool, homogeneous=pta._phi_homog)
    fp_accum.zero_()
    eng.sweep(phiinvs=phiinvs, draw_chunk=args.draw_chunk, accumulate_to=fp_accum)
    return fp_accum


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--warmup", type=int, default=2)
    ap.add_argument("--npsr", type=int, default=67)
    ap.add_argument("--ntoa", type=int, default=5000)
    ap.add_argument("--ntm", type=int, default=60)
    ap.add_argument("--rn-comps", type=int, default=30)
    ap.add_argument("--gwb-comps", type=int, default=30)
    ap.add_argument("--freqs", type=int, default=1000)
    # one step = one full BASELINE config-3 draw batch (1e4 draws x 1e3
    # freqs); also sizes the driver's 20-step run to a >=5 s timed
    # region so SMI utilization sampling is meaningful
    ap.add_argument("--draws-per-step", type=int, default=10000)
    ap.add_argument("--pool-rotations", type=int, default=4,
                    help="independent draw batches rotated across steps")
    ap.add_argument("--draw-chunk", type=int, default=1024)
    ap.add_argument("--freq-chunk", type=int, default=4096)
    ap.add_argument("--device", type=str, default=None)
    ap.add_argument("--no-graph", action="store_true",
                    help="disable hipGraph capture of the step")
    ap.add_argument("--no-compress", action="store_true",
                    help="disable the Schur draw compression")
    ap.add_argument("--ecorr-kernel", action="store_true",
                    help="ECORR as block-diagonal white noise (config 4)")
    ap.add_argument("--preset", choices=sorted(PRESETS), default=None,
                    help="BASELINE.json config preset")
    args = ap.parse_args()
    if args.preset:
        defaults = {a.dest: a.default for a in ap._actions}
        for k, v in PRESETS[args.preset].items():
            if getattr(args, k) == defaults.get(k):
                setattr(args, k, v)

    rank, world, device = init_distributed(
        device=torch.device(args.device) if args.device else None
    )
    on_gpu = device.type == "cuda"
    if world > 1:
        # the compression precompute is pinned to CPU LAPACK; with N
        # ranks each defaulting to all cores, setup would thrash on
        # oversubscribed threads.  Timed-region work is all on-GPU, so
        # this only protects the (untimed) setup wall time.
        import os as _os

        torch.set_num_threads(max(1, (_os.cpu_count() or world) // world))

    pta, eng, pool, active = build_problem(args, device, rank)
    F = args.freqs
    D = args.draws_per_step
    R = max(1, args.pool_rotations)
    fp_accum = torch.zeros((D, F), dtype=torch.float64, device=device)
    dist_on = torch.distributed.is_initialized()

    def barrier_sync():
        if dist_on:
            torch.distributed.barrier()
        if on_gpu:
            torch.cuda.synchronize()

    # hipGraph-capture the whole step (67 pulsars x ~2 kernel launches
    # + the batched phi assembly): launch gaps between the many small
    # dispatches otherwise cost ~15% of the step.  The graph reads the
    # ACTIVE draw buffers; each step rotates a fresh pool batch into
    # them (device-to-device, ~1 MB) before replay, so timed steps never
    # reuse inputs.
    rot = {"i": 0}

    def rotate():
        r = rot["i"] % R
        rot["i"] += 1
        if R > 1:
            for name in active:
                active[name].copy_(pool[name][r])

    inner = lambda: run_step(pta, eng, active, args, fp_accum)  # noqa: E731
    if on_gpu and not args.no_graph:
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            inner()
        torch.cuda.current_stream().wait_stream(side)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            run_step(pta, eng, active, args, fp_accum)
        inner = graph.replay

    def step():
        rotate()
        inner()

    for _ in range(args.warmup):
        step()

    barrier_sync()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        step()
    # the spectrum all-gather is part of the job
    full = all_gather_concat(fp_accum, world, dim=0)
    barrier_sync()
    elapsed = time.perf_counter() - t0

    # max over ranks
    t = torch.tensor([elapsed], dtype=torch.float64, device=device if on_gpu else "cpu")
    if dist_on:
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
    elapsed = float(t.item())

    evals = args.steps * D * F * world  # whole-job evals
    value = evals / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    # honesty guard (outside the timed region): the gathered spectrum
    # must be finite everywhere
    finite = bool(torch.isfinite(full).all().item())
    assert finite, "non-finite Fp values in the benchmark spectrum"

    # secondary (untimed-region) measurement: the DIRECT full-m path
    # with the Schur draw compression off — reported alongside the
    # compressed headline so both numbers are on record.
    direct = None
    if (
        not args.no_compress
        and any(blk.comp is not None for blk in eng.blocks)
        and max(blk.m for blk in eng.blocks) <= 128
    ):
        eng.disable_draw_compression()
        eng._stack_direct()
        dsteps = max(1, args.steps // 4)
        fp2 = torch.zeros_like(fp_accum)

        def dstep():
            phiinvs = batch_phiinv(
                pta.rn_containers, active, homogeneous=pta._phi_homog
            )
            fp2.zero_()
            eng.sweep(phiinvs=phiinvs, draw_chunk=args.draw_chunk,
                      accumulate_to=fp2)

        dstep()  # warm
        barrier_sync()
        t0 = time.perf_counter()
        for _ in range(dsteps):
            dstep()
        barrier_sync()
        del_t = time.perf_counter() - t0
        t = torch.tensor([del_t], dtype=torch.float64,
                         device=device if on_gpu else "cpu")
        if dist_on:
            torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        del_t = float(t.item())
        direct = {
            "value": dsteps * D * F * world / del_t,
            "ms_per_step": del_t / dsteps * 1000.0,
            "steps": dsteps,
        }

    if rank == 0:
        out = {
            "metric": "Fp evals/sec (freqs*draws/s), 67-psr NMFp sweep",
            "value": value,
            "unit": "evals/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": value / 6.0,
            "dtype": "fp64",
            "data": "synthetic",
            "config": {
                "model": "nmfp",
                "npsr": args.npsr,
                "ntoa": args.ntoa,
                "basis_size": args.ntm + 2 * args.rn_comps,
                "freqs": F,
                "draws_per_step_per_gpu": D,
                "global_batch": D * world,
                "parallelism": f"dp{world} draw-sharded",
                "spectrum_shape": list(full.shape),
                "pool_rotations": R,
                "spectrum_finite": finite,
                "pool_compression_margin": args._pool_margin,
                "pool_draws_redrawn": args._pool_redrawn,
            },
        }
        if direct is not None:
            out["direct_path"] = direct
        print(json.dumps(out))
    cleanup()


if __name__ == "__main__":
    main()
