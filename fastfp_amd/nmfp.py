"""Noise-marginalized Fp statistic: :class:`NMFp`.

API parity with the reference's ``NMFP``
(``/root/reference/fastfp/nmfp.py:22-128``): construction from
``(psrs, rn_sigs)``, per-(freq, draw) ``calculate_nmfp`` and
``_get_sigmas``.  The production entry point :meth:`sweep` runs the
whole (draws x freqs) grid through the restructured engine: phi(theta)
is evaluated draw-vectorized by the containers, and each (pulsar, draw)
costs one Cholesky + one fused triangular-solve/reduction instead of
6*F Woodbury solves.
"""

from __future__ import annotations

import math
import os

import numpy as np
import torch

from fastfp_amd.engine import FpEngine
from fastfp_amd.xcy import get_xCy


class _GraphedEngineSweep:
    """One captured hipGraph: static (nparams, D) input buffer ->
    batched phi assembly -> zeroed fp -> stacked compressed sweep.
    Replays give the per-batch cost of ONE host-to-device copy + one
    graph launch (the bench's step structure, brought to the CLI)."""

    def __init__(self, nmfp, engine, names, D, draw_chunk):
        from fastfp_amd.noise import batch_phiinv

        self.engine = engine
        self.names = list(names)
        dev = engine.device
        self.buf = torch.empty((len(self.names), D), dtype=torch.float64,
                               device=dev)
        pars = {n: self.buf[i] for i, n in enumerate(self.names)}
        F = engine.freqs.shape[0]
        self.fp = torch.zeros((D, F), dtype=torch.float64, device=dev)
        self.phiinvs = None

        def run():
            piv = batch_phiinv(nmfp.rn_sigs, pars, homogeneous=True)
            piv = [p[None, :] if p.dim() == 1 else p for p in piv]
            self.phiinvs = piv
            self.fp.zero_()
            engine.sweep(phiinvs=piv, draw_chunk=draw_chunk,
                         accumulate_to=self.fp)

        # benign warmup values so the capture's kernels see valid input
        self.buf.fill_(-14.5)
        for i, n in enumerate(self.names):
            if n.endswith("gamma"):
                self.buf[i].fill_(13.0 / 3.0)
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            run()
        torch.cuda.current_stream().wait_stream(side)
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            run()

    def __call__(self, samples):
        mat = np.stack(
            [np.asarray(samples[n], dtype=np.float64) for n in self.names]
        )
        self.buf.copy_(torch.from_numpy(mat))
        self.graph.replay()
        return self.fp, self.phiinvs


class NMFp:
    """Noise-marginalized Fp over red-noise parameter draws."""

    def __init__(self, psrs, rn_sigs):
        self.psrs = psrs
        self.rn_sigs = rn_sigs
        self.toas = [np.asarray(p.toas, dtype=np.float64) for p in psrs]
        self.residuals = [np.asarray(p.residuals, dtype=np.float64) for p in psrs]
        self._phi_homog = None  # cached check_batch_homogeneous result
        self._graphs = {}  # (id(engine), D, draw_chunk, names) -> _GraphedEngineSweep
        self._graph_seen = {}  # shape-repeat counts (capture on 2nd)

    def __call__(self, fgw, samples, Nvecs, Ts, TNTs):
        return self.calculate_nmfp(fgw, samples, Nvecs, Ts, TNTs)

    # ------------------------------------------------------------------
    def _get_sigmas(self, pars: dict, TNTs) -> list:
        """Sigma = TNT + diag(phi^-1) per pulsar at one parameter point
        (parity with ``/root/reference/fastfp/nmfp.py:57-74``)."""
        sigmas = []
        for rn_sig, TNT in zip(self.rn_sigs, TNTs):
            phiinv = rn_sig.get_phiinv(pars)
            if isinstance(phiinv, torch.Tensor):
                phiinv = phiinv.cpu().numpy()
            sigmas.append(np.asarray(TNT) + np.diag(phiinv))
        return sigmas

    def calculate_nmfp(self, fgw, samples: dict, Nvecs, Ts, TNTs) -> float:
        """Single (frequency, parameter-point) NM-Fp — parity path."""
        sigmas = self._get_sigmas(samples, TNTs)
        amp = 1.0 / fgw ** (1.0 / 3.0)
        fstat = 0.0
        for Nvec, T, sigma, toa, resid in zip(
            Nvecs, Ts, sigmas, self.toas, self.residuals
        ):
            arg = 2.0 * math.pi * fgw * toa
            A0 = amp * np.sin(arg)
            A1 = amp * np.cos(arg)
            ip1 = get_xCy(Nvec, T, sigma, A0, resid)
            ip2 = get_xCy(Nvec, T, sigma, A1, resid)
            N = np.array([ip1, ip2])
            M = np.empty((2, 2))
            M[0, 0] = get_xCy(Nvec, T, sigma, A0, A0)
            M[0, 1] = get_xCy(Nvec, T, sigma, A0, A1)
            M[1, 0] = M[0, 1]
            M[1, 1] = get_xCy(Nvec, T, sigma, A1, A1)
            fstat += 0.5 * float(N @ np.linalg.solve(M, N))
        return fstat

    # ------------------------------------------------------------------
    def _direct_rows(self, engine, phiinvs, bad_idx, draw_chunk, Nvecs, Ts):
        """Exact direct-path evaluation of the draws at ``bad_idx``.

        On GPU with basis size m <= 256 this is the direct HIP solve;
        above 256 (no GPU kernel) the rows run through a cached CPU
        LAPACK eager engine — slow but exact, and only the rare
        prior-corner draws pay it."""
        sub = [p[bad_idx.to(p.device)] for p in phiinvs]
        if engine._use_hip and any(b.m > 256 for b in engine.blocks):
            fr = engine.freqs.cpu().numpy()
            ce = getattr(self, "_cpu_eng", None)
            if ce is None or len(ce[0]) != len(fr) or not np.array_equal(ce[0], fr):
                eng = FpEngine(self.psrs, Nvecs, Ts, device="cpu")
                eng.precompute(fr)
                self._cpu_eng = ce = (fr, eng)
            vals = ce[1].sweep(phiinvs=[p.cpu() for p in sub],
                               draw_chunk=draw_chunk)
            return vals.to(engine.device)
        return engine.sweep(phiinvs=sub, draw_chunk=draw_chunk,
                            force_direct=True)

    def _sweep_graphed(self, engine, samples, draw_chunk, compress,
                       Nvecs, Ts):
        """hipGraph-replayed batch sweep, or ``None`` when any
        precondition fails (CPU engine, heterogeneous containers, no
        stacked compression, device-tensor samples, odd shapes) — the
        caller then runs the eager path.

        Per-draw margin safety is preserved: the replayed graph
        evaluates EVERY draw through the compressed path, margins are
        checked afterwards from the captured phi buffers, and any
        prior-corner draws (rare) are recomputed through the exact
        direct path and overwritten — per-draw independence makes the
        other rows valid."""
        if (
            engine is None
            or not engine._use_hip
            or not compress
            or not self._phi_homog
            or getattr(engine, "_comp_stack", None) is None
            or os.environ.get("FASTFP_NO_GRAPH") == "1"
        ):
            return None
        names = list(samples.keys())
        vals = list(samples.values())
        if any(isinstance(v, torch.Tensor) and v.is_cuda for v in vals):
            return None
        if any(np.ndim(v) != 1 for v in vals):
            return None
        D = len(vals[0])
        if any(len(v) != D for v in vals):
            return None
        # the captured graph closes over ONE engine (its freq grid, its
        # compression stack), so the engine's identity must be part of
        # the key: a same-shaped sweep against a different engine (new
        # freqs, same D) must not replay a stale capture.  Cached
        # entries hold a strong reference to their engine, so a live
        # id() here can only name that same object.  engine=None
        # callers get a fresh engine per sweep and therefore never
        # reuse a capture — they stay on the eager path (correct: their
        # per-call precompute dwarfs the launch overhead anyway).
        key = (id(engine), D, draw_chunk, tuple(names))
        g = self._graphs.get(key)
        if g is None:
            # capture costs ~2 sweeps + graph instantiation, so only
            # capture once a shape REPEATS (batch 2 of a CLI loop);
            # one-shot sweeps stay eager
            if len(self._graph_seen) > 64:  # engine-per-call callers
                self._graph_seen.clear()    # never repeat a key: bound it
            seen = self._graph_seen.get(key, 0) + 1
            self._graph_seen[key] = seen
            if seen < 2 or len(self._graphs) >= 4:  # bound graph pool
                return None
            try:
                g = _GraphedEngineSweep(self, engine, names, D, draw_chunk)
            except Exception:  # capture failure: permanent eager fallback
                g = False
            self._graphs[key] = g
        if g is False:
            return None
        fp, phiinvs = g(samples)
        margins = engine.compression_margin_per_draw(phiinvs)
        bad = margins < 1.5
        if bool(bad.any()):
            bad_idx = torch.nonzero(bad).reshape(-1)
            fp = fp.clone()
            fp[bad_idx] = self._direct_rows(engine, phiinvs, bad_idx,
                                            draw_chunk, Nvecs, Ts)
        return fp.cpu().numpy()

    # ------------------------------------------------------------------
    def sweep(
        self,
        freqs,
        samples: dict,
        Nvecs,
        Ts,
        TNTs=None,
        device: str = None,
        draw_chunk: int = 512,
        freq_chunk: int = 2048,
        engine: FpEngine = None,
        compress: bool = True,
    ) -> np.ndarray:
        """NM-Fp over (draws x freqs).

        ``samples``: dict of parameter-name -> (D,) arrays (the
        ``map_params`` format of ``/root/reference/examples/run_nmfp.py:174-186``).
        Returns (D, F) numpy array — the reference's output shape
        (``/root/reference/examples/run_nmfp.py:270``).
        """
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        if engine is None:
            engine = FpEngine(self.psrs, Nvecs, Ts, device=device)
            engine.precompute(freqs, freq_chunk=freq_chunk)
            if compress:
                # fixed phi^-1 from any parameter point; variable bins
                # are zeroed inside enable_draw_compression
                probe = {k: (v[0] if np.ndim(v) else v) for k, v in samples.items()}
                engine.enable_draw_compression(
                    [sig.var_slice for sig in self.rn_sigs],
                    [sig.get_phiinv(probe) for sig in self.rn_sigs],
                )
        from fastfp_amd.noise import batch_phiinv, check_batch_homogeneous

        # the homogeneity check device-syncs; cache it across the
        # per-batch calls of a long CLI sweep
        if self._phi_homog is None:
            self._phi_homog = check_batch_homogeneous(self.rn_sigs)

        # hipGraph fast path for repeated same-shape draw batches (the
        # CLI hot loop): phi assembly + the stacked compressed sweep
        # are captured once and replayed per batch — the per-batch
        # eager launch overhead was the remaining CLI-vs-bench gap.
        out = self._sweep_graphed(engine, samples, draw_chunk, compress,
                                  Nvecs, Ts)
        if out is not None:
            return out

        phiinvs = batch_phiinv(self.rn_sigs, samples,
                               homogeneous=self._phi_homog)
        # scalar-parameter dicts produce (m,) vectors; promote to (1, m)
        phiinvs = [p[None, :] if p.dim() == 1 else p for p in phiinvs]
        # Per-draw margin guard: the compressed path is ALGEBRAICALLY
        # exact for any Delta_d = phiinv_d - delta0 > 0; the only
        # numerical hazard is the subtraction's precision loss as
        # margin -> 1 (rel err = eps/(1 - 1/margin)) and the Delta <= 0
        # cliff (indefinite C_d -> NaN from the Cholesky solver).
        # Measured: at margin 1.7 the compressed-vs-direct error is
        # 2e-10 of the spectrum scale (docs/PERFORMANCE.md r02) — the
        # round-1 threshold of 1e3 was ~3 orders too conservative and
        # silently pushed whole uniform-prior sweeps onto the 4x-slower
        # direct path.  The guard is PER DRAW: prior-corner draws
        # (margin < 1.5) run the exact direct path, the rest stay
        # compressed — one extreme draw no longer slows a whole batch.
        if compress and any(b.comp is not None for b in engine.blocks):
            margins = engine.compression_margin_per_draw(phiinvs)
            bad = margins < 1.5
            nbad = int(bad.sum())
            if nbad:
                good_idx = torch.nonzero(~bad).reshape(-1)
                bad_idx = torch.nonzero(bad).reshape(-1)
                D = margins.shape[0]
                F = engine.freqs.shape[0]
                fp = torch.zeros((D, F), dtype=torch.float64,
                                 device=engine.device)
                if len(good_idx):
                    sub = [p[good_idx.to(p.device)] for p in phiinvs]
                    fp[good_idx] = engine.sweep(phiinvs=sub,
                                                draw_chunk=draw_chunk)
                fp[bad_idx] = self._direct_rows(engine, phiinvs, bad_idx,
                                                draw_chunk, Nvecs, Ts)
                return fp.cpu().numpy()
        fp = engine.sweep(phiinvs=phiinvs, draw_chunk=draw_chunk)
        return fp.cpu().numpy()


#: reference-compatible class name
NMFP = NMFp
