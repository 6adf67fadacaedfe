"""The Fp-statistic compute engine (restructured math, device-aware).

The reference evaluates, for every (frequency, draw, pulsar), six
independent Woodbury products ``x^T C^-1 y`` each re-solving the same
Sigma (``/root/reference/fastfp/fastfp.py:81-88``,
``/root/reference/fastfp/utils.py:49-54``) and relies on XLA CSE.  This
engine restructures the computation around what is invariant along each
batch axis:

per pulsar (fixed):
    ``TNT = T^T N^-1 T``, ``TNr = T^T N^-1 r``
per (pulsar, frequency) (fixed across draws):
    ``B = T^T N^-1 [s, c]`` for the whole frequency grid at once — one
    GEMM — plus the diagonal-weighted dots ``sNs`` (3 per f) and ``sNr``
    (2 per f).  The common amplitude ``f^-1/3`` of the reference's filter
    (``/root/reference/fastfp/fastfp.py:78-79``) cancels exactly in
    ``N^T M^-1 N`` and is omitted here (proved in docs/DESIGN.md,
    verified against the oracle).
per (pulsar, draw):
    one Cholesky ``L L^T = Sigma = TNT + diag(phi^-1)`` and one
    triangular solve ``W = L^-1 [B | TNr]``; then for every frequency
    the 2x2 system is closed-form:

    ``M = sNs - W_s/c^T W_s/c``, ``N = sNr - W_s/c^T w_u``,
    ``Fp += 1/2 N^T M^-1 N``.

This turns 6*F*D*P Woodbury solves into P GEMMs + D*P Cholesky/TRSMs —
the MFMA-DGEMM-shaped form the MI355X wants.  On CPU the engine runs
vectorized torch eager (the unit-test oracle path); on a ROCm GPU the
hot ops are hand-written fp64 HIP kernels (fastfp_amd.ops), and falling
back silently to eager on GPU is an error.
"""

from __future__ import annotations

import contextlib
import math
import os

import numpy as np
import torch


@contextlib.contextmanager
def _roctx(name: str):
    """Optional roctx ranges (shown in rocprof timelines) around the
    engine phases; enable with FASTFP_ROCTX=1."""
    if os.environ.get("FASTFP_ROCTX") == "1" and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


def _t64(x, device):
    if isinstance(x, torch.Tensor):
        return x.to(device=device, dtype=torch.float64)
    return torch.as_tensor(np.asarray(x, dtype=np.float64), device=device)


class PulsarBlock:
    """Per-pulsar device tensors + fixed precompute.

    ``Nvec`` may be a per-TOA variance vector (diagonal N) or a
    :class:`fastfp_amd.blocknoise.BlockNoise` (block-diagonal N, the
    EcorrKernelNoise case).  For block N the TOAs are re-ordered by the
    BlockNoise permutation (a pure relabeling — every inner product is
    permutation-invariant) and the engine precomputes ``V = N^{-1} T``
    so the frequency GEMM is unchanged: ``T^T N^{-1} S = V^T S``.
    """

    def __init__(self, toas, resid, Nvec, T, device):
        from fastfp_amd.blocknoise import BlockNoise

        toas = np.asarray(toas, dtype=np.float64)
        resid = np.asarray(resid, dtype=np.float64)
        T = np.asarray(T, dtype=np.float64)
        self.block_noise = Nvec if isinstance(Nvec, BlockNoise) else None
        if self.block_noise is not None:
            perm = self.block_noise.perm
            toas, resid, T = toas[perm], resid[perm], T[perm, :]
        self.toas = _t64(toas, device)
        self.r = _t64(resid, device)
        self.T = _t64(T, device).contiguous()
        self.ntoa, self.m = self.T.shape

        if self.block_noise is None:
            self.Nvec = _t64(Nvec, device)
            TN = self.T / self.Nvec[:, None]  # N^-1 T  (ntoa, m)
            self.Nr = self.r / self.Nvec
        else:
            self.Nvec = None
            TN = _t64(self.block_noise.solve(T), device)  # V = N^-1 T
            self.Nr = _t64(self.block_noise.solve(resid), device)
        self.V = TN.contiguous()
        self.TNT = self.T.transpose(0, 1) @ TN
        self.TNr = TN.transpose(0, 1) @ self.r

        # filled by freq precompute:
        self.RHS = None  # (m, 2F+1): interleaved [s_f, c_f] columns + TNr
        self.sNs = None  # (3, F): ss, cc, sc
        self.sNr = None  # (2, F): s.r, c.r
        # filled by enable_draw_compression (docs/DESIGN.md):
        self.comp = None  # dict(G, K, M0, N0, var)


class FpEngine:
    """Restructured Fp/NM-Fp compute over a list of pulsars.

    Parameters
    ----------
    psrs : list of PulsarData (or anything with .toas/.residuals)
    Nvecs, Ts : per-pulsar white-noise diagonals and basis matrices
        (the ``get_mats_*`` outputs, keeping the reference's data flow).
    device : torch device ("cpu" or "cuda:N").
    """

    def __init__(self, psrs, Nvecs, Ts, device="cpu", force_eager=False):
        self.device = torch.device(device)
        self.blocks = [
            PulsarBlock(p.toas, p.residuals, nv, T, self.device)
            for p, nv, T in zip(psrs, Nvecs, Ts)
        ]
        self.freqs = None
        self._use_hip = False
        self._side_stream = None
        if self.device.type == "cuda" and not force_eager:
            from fastfp_amd import ops

            ops.require_hip()  # fail loudly if the extension is missing
            self._use_hip = True
            # second stream: per-pulsar chol->trsm chains are independent,
            # so alternating pulsars across two streams overlaps each
            # pulsar's Cholesky with the previous pulsar's solve
            self._side_stream = torch.cuda.Stream(device=self.device)

    # ------------------------------------------------------------------
    # frequency precompute
    # ------------------------------------------------------------------
    def precompute(self, freqs, freq_chunk: int = 2048):
        """Compute B, sNs, sNr for all pulsars on the engine's frequency
        grid.  Chunked over frequencies to bound the (Fc, ntoa) sin/cos
        intermediate."""
        freqs = _t64(freqs, self.device).reshape(-1)
        self.freqs = freqs
        F = freqs.shape[0]
        # any existing compression state was built against the OLD
        # frequency grid; callers re-enable after re-precomputing
        self.disable_draw_compression()
        with _roctx("fastfp:freq_precompute"):
            for blk in self.blocks:
                if self._use_hip:
                    self._precompute_hip(blk, freqs)
                else:
                    self._precompute_eager(blk, freqs, freq_chunk)
        self._stack_direct()
        return self

    def _precompute_eager(self, blk: PulsarBlock, freqs, freq_chunk):
        F = freqs.shape[0]
        m = blk.m
        RHS = torch.empty((m, 2 * F + 1), dtype=torch.float64, device=self.device)
        sNs = torch.empty((3, F), dtype=torch.float64, device=self.device)
        sNr = torch.empty((2, F), dtype=torch.float64, device=self.device)
        for lo in range(0, F, freq_chunk):
            hi = min(lo + freq_chunk, F)
            arg = 2.0 * math.pi * freqs[lo:hi, None] * blk.toas[None, :]  # (Fc, ntoa)
            S = torch.sin(arg)
            C = torch.cos(arg)
            if blk.block_noise is None:
                NS = S / blk.Nvec[None, :]
                NC = C / blk.Nvec[None, :]
            else:
                NS = blk.block_noise.solve(S.transpose(0, 1)).transpose(0, 1)
                NC = blk.block_noise.solve(C.transpose(0, 1)).transpose(0, 1)
            # B columns: interleaved sin,cos  (B = S N^-1 T = S V)
            Bs = S @ blk.V  # (Fc, m)
            Bc = C @ blk.V
            RHS[:, 2 * lo : 2 * hi : 2] = Bs.transpose(0, 1)
            RHS[:, 2 * lo + 1 : 2 * hi : 2] = Bc.transpose(0, 1)
            sNs[0, lo:hi] = (S * NS).sum(dim=1)
            sNs[1, lo:hi] = (C * NC).sum(dim=1)
            sNs[2, lo:hi] = (S * NC).sum(dim=1)
            sNr[0, lo:hi] = S @ blk.Nr
            sNr[1, lo:hi] = C @ blk.Nr
        RHS[:, -1] = blk.TNr
        blk.RHS, blk.sNs, blk.sNr = RHS, sNs, sNr

    def _precompute_hip(self, blk: PulsarBlock, freqs):
        from fastfp_amd import ops

        if blk.block_noise is None:
            # no freq_chunk: the HIP kernels generate the S panel in
            # LDS and need no chunking (ops.freq_precompute docstring)
            blk.RHS, blk.sNs, blk.sNr = ops.freq_precompute(
                blk.toas, blk.Nvec, blk.r, blk.T, blk.TNr, freqs
            )
        else:
            blk.RHS, blk.sNs, blk.sNr = ops.freq_precompute_block(
                blk.toas, blk.block_noise, blk.Nr, blk.V, blk.TNr, freqs
            )

    # ------------------------------------------------------------------
    # Schur draw-compression
    # ------------------------------------------------------------------
    def enable_draw_compression(self, var_slices, phiinv_fixed,
                                jitter_rel: float = 1e-8):
        """Compress the per-draw solve to the VARIABLE prior bins.

        The draw-dependent part of ``Sigma(theta) = TNT + diag(phi^-1)``
        lives only on the bins whose prior varies (the red-noise(+CURN)
        Fourier bins); the timing-model 1e-40 and fixed GP-ECORR bins
        never change.  With ``Sigma_0 = TNT + diag(phiinv_fixed)``
        (variable bins zeroed) and Woodbury,

          B^T Sigma_d^-1 B = B^T Sigma_0^-1 B - K^T C_d^-1 K,
          C_d = diag(phi_d[var]) + G,  K = (Sigma_0^-1 [B|TNr])[var],
          G = (Sigma_0^-1)[var, var],

        so the per-draw Cholesky/TRSM run at dimension m_v = |var bins|
        instead of m — 4x fewer TRSM flops at the benchmark shape
        (m=120, m_v=60).  The per-frequency baseline M0/N0 replaces
        sNs/sNr, and the fused reduction flips sign (M = M0 + W.W).

        ``var_slices``: per-pulsar slice of variable basis columns;
        ``phiinv_fixed``: per-pulsar (m,) fixed phi^-1 (values on the
        variable bins are ignored/zeroed).
        """
        assert self.freqs is not None, "call precompute(freqs) first"
        # The K/M0/N0/G setup runs on CPU LAPACK even for GPU engines:
        # rocBLAS trsm is inversion-based rather than backward-stable
        # substitution and measured ~50x less accurate here (1e-4 vs
        # 2e-6 of spectrum scale at the benchmark conditioning) — a
        # one-time ~2 MB/pulsar round trip buys LAPACK accuracy.
        wdev = torch.device("cpu")
        for blk, sl, pf in zip(self.blocks, var_slices, phiinv_fixed):
            m = blk.m
            pf = _t64(pf, wdev).clone()
            # Sigma_0 must be SPD even when timing-model columns are
            # (nearly) degenerate with red-noise Fourier columns, so the
            # variable bins keep a TINY reference prior delta_i
            # proportional to the Sigma diagonal; the per-draw
            # correction then uses Delta_d = phiinv_d - delta, valid
            # while phiinv_d >> delta (checked by compression_margin).
            # delta_i trades conditioning against margin EXACTLY (the
            # split is algebraically exact; only round-off depends on
            # it): the compression error scales ~ cond(Sigma_0)*eps ~
            # eps/jitter_rel, so 1e-8 keeps errors ~1e-9..1e-7 of the
            # spectrum scale (measured: 100x better than the r01
            # 1e-10, which left 6/67 bench pulsars over the probe tol).
            # The margin guard (compression_margin) falls back to the
            # direct path below margin 1.5.
            # rank-deficient TNT (basis larger than the TOA count)
            # leaves Sigma_0 supported only by the jitter along its
            # null space — keep such pulsars on the exact direct path
            # (checked BEFORE factoring: their Sigma_0 may not even be
            # numerically PD).  Near-degeneracy SHORT of deficiency is
            # handled by the empirical probe below: Cholesky-pivot
            # heuristics measured uncorrelated with the actual error.
            if blk.ntoa < m:
                blk.comp = None
                continue
            TNTc = blk.TNT.to(wdev)
            delta0 = jitter_rel * torch.diagonal(TNTc)[sl].abs()
            pf[sl] = delta0
            sigma0 = TNTc + torch.diag(pf)
            try:
                L0 = torch.linalg.cholesky(sigma0)
            except torch.linalg.LinAlgError:
                blk.comp = None  # fail safe to the exact direct path
                continue
            RHSe = blk.RHS[:m, :].to(wdev)  # (m, 2F+1)
            ncols = RHSe.shape[1]
            mv = len(range(*sl.indices(m)))
            sNs_c = blk.sNs.to(wdev)
            sNr_c = blk.sNr.to(wdev)
            # column-chunked solves (bounds LAPACK workspace/temporaries
            # at the 2e5-frequency SKA shape)
            K = torch.empty((mv, ncols), dtype=torch.float64)
            M0 = torch.empty((3, ncols // 2), dtype=torch.float64)
            N0 = torch.empty((2, ncols // 2), dtype=torch.float64)
            CH = 16384  # even: chunks stay aligned to sin/cos pairs
            # first pass: the u column (last), needed by every chunk
            wcol = torch.linalg.solve_triangular(
                L0, RHSe[:, -1:], upper=False
            )
            for lo in range(0, ncols, CH):
                hi = min(lo + CH, ncols)
                W0c = torch.linalg.solve_triangular(
                    L0, RHSe[:, lo:hi], upper=False
                )
                # frequency-column bookkeeping: chunks are aligned to
                # even columns (CH is even), so cols lo..hi-1 pair up;
                # the final chunk also carries the (already solved) u col
                fs, fe = lo // 2, (hi - (1 if hi == ncols else 0)) // 2
                Ws = W0c[:, 0 : 2 * (fe - fs) : 2]
                Wc = W0c[:, 1 : 2 * (fe - fs) : 2]
                wu = wcol[:, 0]
                M0[0, fs:fe] = sNs_c[0, fs:fe] - (Ws * Ws).sum(0)
                M0[1, fs:fe] = sNs_c[1, fs:fe] - (Wc * Wc).sum(0)
                M0[2, fs:fe] = sNs_c[2, fs:fe] - (Ws * Wc).sum(0)
                N0[0, fs:fe] = sNr_c[0, fs:fe] - Ws.T @ wu
                N0[1, fs:fe] = sNr_c[1, fs:fe] - Wc.T @ wu
                S0c = torch.linalg.solve_triangular(
                    L0.transpose(0, 1), W0c, upper=True
                )
                K[:, lo:hi] = S0c[sl, :]
            S0inv = torch.cholesky_inverse(L0)
            G = S0inv[sl, sl]
            if self._use_hip:
                from fastfp_amd import ops

                mvp = ops.check_m(mv)
                Kp = torch.zeros((mvp, K.shape[1]), dtype=torch.float64)
                Kp[:mv] = K
                K = Kp
            blk.comp = dict(
                G=G.to(self.device).contiguous(),
                K=K.to(self.device).contiguous(),
                M0=M0.to(self.device).contiguous(),
                N0=N0.to(self.device).contiguous(),
                var=sl, mv=mv,
                delta0=delta0.to(self.device),
            )
        self._probe_compression(phiinv_fixed)
        self._stack_compression()
        return self

    def _probe_compression(self, phiinv_fixed, tol: float = 1e-6):
        """Empirical accuracy guard: evaluate one probe draw per pulsar
        through BOTH the compressed and the direct path and drop
        compression where they disagree beyond ``tol`` — conditioning
        heuristics (Cholesky pivot ratios vs the jitter) measured
        uncorrelated with the real error, so measure instead.

        tol 1e-6 (10x tighter than round 1): with jitter_rel=1e-8 the
        measured bench-shape errors are 1e-8..7e-8 of the spectrum
        scale, so healthy pulsars pass with two orders of margin."""
        F = self.freqs.shape[0]
        for blk, pf in zip(self.blocks, phiinv_fixed):
            if blk.comp is None:
                continue
            pinv = _t64(pf, self.device).reshape(1, -1)
            c = blk.comp
            fpA = torch.zeros((1, F), dtype=torch.float64, device=self.device)
            fpB = torch.zeros_like(fpA)
            phi_var = (1.0 / (pinv[:, c["var"]] - c["delta0"][None, :])).contiguous()
            try:
                if self._use_hip and blk.m > 128:
                    # above m=128 the GPU direct factor routes through
                    # rocSOLVER; for the probe's REFERENCE arm prefer
                    # the CPU LAPACK eager path on this one draw
                    # (compression precompute is already pinned to CPU,
                    # so this adds one more small host round trip at
                    # setup time only)
                    from fastfp_amd import ops

                    ops.chol_trsm_fp_accum(
                        c["G"], phi_var, c["K"], c["M0"], c["N0"], fpA,
                        gsign=-1.0,
                    )
                    pinv_c = pinv.cpu()
                    sigma_c = blk.TNT.cpu()[None, :, :] + torch.diag_embed(pinv_c)
                    fpB_c = torch.zeros((1, F), dtype=torch.float64)
                    self._accum_eager_mats(
                        sigma_c, blk.RHS[: blk.m].cpu(), blk.sNs.cpu(),
                        blk.sNr.cpu(), fpB_c, 1.0,
                    )
                    fpB = fpB_c.to(self.device)
                elif self._use_hip:
                    from fastfp_amd import ops

                    ops.chol_trsm_fp_accum(
                        c["G"], phi_var, c["K"], c["M0"], c["N0"], fpA,
                        gsign=-1.0,
                    )
                    ops.chol_trsm_fp_accum(
                        blk.TNT, pinv.contiguous(), blk.RHS, blk.sNs,
                        blk.sNr, fpB, gsign=1.0,
                    )
                else:
                    sigc = c["G"][None, :, :] + torch.diag_embed(phi_var)
                    self._accum_eager_mats(
                        sigc, c["K"], c["M0"], c["N0"], fpA, -1.0
                    )
                    sigma = blk.TNT[None, :, :] + torch.diag_embed(pinv)
                    self._accum_eager(blk, sigma, fpB)
            except torch.linalg.LinAlgError:
                blk.comp = None  # pathological probe (e.g. phi < jitter)
                continue
            # error against the SPECTRUM scale: per-frequency relative
            # error is meaningless at cancellation-dominated frequencies
            # (docs/DESIGN.md §8) and mis-flags healthy models
            scale = fpB.abs().max().clamp_min(1e-30)
            err = (fpA - fpB).abs().max() / scale
            blk.probe_err = float(err)  # kept for diagnostics
            # NaN-safe: a non-finite probe (GPU kernels don't raise) or
            # a too-large error both disqualify the compressed path
            if not bool(torch.isfinite(err)) or float(err) > tol:
                blk.comp = None

    def _stack_direct(self):
        """Stack per-pulsar TNT/RHS/sNs/sNr for single-launch DIRECT
        sweeps (same dimensions across pulsars; used by the plain-Fp
        phiinv path and the compression-margin fallback)."""
        self._direct_stack = None
        if not self._use_hip:
            return
        ms = {blk.m for blk in self.blocks}
        rs = {tuple(blk.RHS.shape) for blk in self.blocks if blk.RHS is not None}
        if len(ms) != 1 or len(rs) != 1 or self.blocks[0].m > 256:
            return
        self._direct_stack = dict(
            TNT=torch.stack([b.TNT for b in self.blocks]).contiguous(),
            RHS=torch.stack([b.RHS for b in self.blocks]).contiguous(),
            sNs=torch.stack([b.sNs for b in self.blocks]).contiguous(),
            sNr=torch.stack([b.sNr for b in self.blocks]).contiguous(),
        )

    def _stack_compression(self):
        """When every pulsar compresses to the same variable dimension,
        stack the per-pulsar G/K/M0/N0 so the whole sweep is ONE chol
        launch + ONE trsm launch per draw chunk (grid.z = pulsar) — no
        per-pulsar launch gaps or tail quantization.  Per-pulsar Fp
        planes are summed deterministically by torch."""
        self._comp_stack = None
        if not self._use_hip:
            return
        comps = [blk.comp for blk in self.blocks]
        if any(c is None for c in comps):
            return
        mvs = {c["G"].shape[0] for c in comps}
        kshapes = {tuple(c["K"].shape) for c in comps}
        if len(mvs) != 1 or len(kshapes) != 1:
            return
        vars_ = [c["var"] for c in comps]
        self._comp_stack = dict(
            G=torch.stack([c["G"] for c in comps]).contiguous(),
            K=torch.stack([c["K"] for c in comps]).contiguous(),
            M0=torch.stack([c["M0"] for c in comps]).contiguous(),
            N0=torch.stack([c["N0"] for c in comps]).contiguous(),
            delta0=torch.stack([c["delta0"] for c in comps]).contiguous(),
            vars=vars_,
            mv=comps[0]["G"].shape[0],
        )

    def compression_margin(self, phiinvs) -> float:
        """min over pulsars/draws/bins of phiinv_d / delta_0.  The
        compressed path is numerically safe above ~1.5 (the split is
        algebraically exact; precision of the Delta = phiinv - delta0
        subtraction degrades as eps/(1 - 1/margin) and Delta <= 0 is
        the NaN cliff — measured 2e-10 error at margin 1.7).  Callers
        fall back to the exact direct path below 1.5.

        One device sync total: per-pulsar mins are stacked and reduced
        on device (a per-pulsar ``float()`` cost 67 syncs per draw
        batch in the CLI hot loop).  When every pulsar compresses with
        the same variable slice (the stacked homogeneous case), the
        whole check is ~4 kernels."""
        st = getattr(self, "_comp_stack", None)
        if (
            st is not None
            and all(sl == st["vars"][0] for sl in st["vars"])
            and all(
                isinstance(p, torch.Tensor)
                and p.device == self.device
                and p.dim() == 2
                for p in phiinvs
            )
        ):
            pall = torch.stack(list(phiinvs))  # (P, D, m)
            ratio = pall[:, :, st["vars"][0]] / st["delta0"][:, None, :]
            return float(ratio.min())
        mins = []
        for blk, pinv in zip(self.blocks, phiinvs):
            if blk.comp is None:
                continue
            p = _t64(pinv, self.device)
            p = p[None, :] if p.dim() == 1 else p
            mins.append(
                (p[:, blk.comp["var"]] / blk.comp["delta0"][None, :]).min()
            )
        if not mins:
            return float("inf")
        return float(torch.stack(mins).min())

    def compression_margin_per_draw(self, phiinvs) -> torch.Tensor:
        """Per-draw compression margin: (D,) tensor of
        min over pulsars/bins of phiinv_d / delta0.  Basis of the
        per-draw hybrid split (NMFp.sweep): draws above the guard run
        compressed, the (rare) prior-corner draws run the exact direct
        path — one extreme draw no longer forces a whole batch off the
        fast path.  Returns +inf per draw when nothing is compressed."""
        mins = []
        for blk, pinv in zip(self.blocks, phiinvs):
            if blk.comp is None:
                continue
            p = _t64(pinv, self.device)
            p = p[None, :] if p.dim() == 1 else p
            mins.append(
                (p[:, blk.comp["var"]] / blk.comp["delta0"][None, :])
                .min(dim=1).values
            )
        if not mins:
            D = phiinvs[0].shape[0] if phiinvs[0].dim() == 2 else 1
            return torch.full((D,), float("inf"), dtype=torch.float64,
                              device=self.device)
        return torch.stack(mins).min(dim=0).values

    def disable_draw_compression(self):
        for blk in self.blocks:
            blk.comp = None
        self._comp_stack = None  # the stacked path must also fall back
        return self

    # ------------------------------------------------------------------
    # sweeps
    # ------------------------------------------------------------------
    def sweep(
        self,
        phiinvs=None,
        sigmas=None,
        draw_chunk: int = 32,
        accumulate_to=None,
        force_direct: bool = False,
    ) -> torch.Tensor:
        """Run the Fp sweep over the precomputed frequency grid.

        ``phiinvs``: per-pulsar diagonal phi^-1, shape (m,) for the plain
        Fp path or (D, m) for D noise draws.  Alternatively pass dense
        ``sigmas`` (m, m) / (D, m, m) directly (the ``get_mats_fp``
        contract).  Returns Fp of shape (F,) or (D, F).

        ``force_direct``: bypass the Schur compression for THIS call
        (used by the per-draw hybrid for prior-corner draws whose
        margin is below the guard).
        """
        assert self.freqs is not None, "call precompute(freqs) first"
        F = self.freqs.shape[0]
        if phiinvs is not None:
            first = _t64(phiinvs[0], self.device)
            batched = first.dim() == 2
            D = first.shape[0] if batched else 1
        else:
            first = _t64(sigmas[0], self.device)
            batched = first.dim() == 3
            D = first.shape[0] if batched else 1

        fp = accumulate_to
        if fp is None:
            fp = torch.zeros((D, F), dtype=torch.float64, device=self.device)

        stack = getattr(self, "_comp_stack", None)
        if stack is not None and phiinvs is not None and not force_direct:
            return self._sweep_stacked(phiinvs, fp, D, F, draw_chunk, batched)
        dstack = getattr(self, "_direct_stack", None)
        if (
            dstack is not None
            and phiinvs is not None
            and (force_direct
                 or all(blk.comp is None for blk in self.blocks))
        ):
            return self._sweep_stacked_direct(
                phiinvs, fp, D, F, draw_chunk, batched
            )

        fp_side = None
        main_stream = None
        if self._use_hip:
            main_stream = torch.cuda.current_stream(self.device)
            fp_side = torch.zeros_like(fp)
            ev = torch.cuda.Event()
            ev.record(main_stream)
            self._side_stream.wait_event(ev)

        for lo in range(0, D, draw_chunk):
            hi = min(lo + draw_chunk, D)
            for i, blk in enumerate(self.blocks):
                side = self._use_hip and (i & 1) == 1
                stream_ctx = (
                    torch.cuda.stream(self._side_stream)
                    if side
                    else contextlib.nullcontext()
                )
                fp_tgt = (fp_side if side else fp)[lo:hi]
                with stream_ctx:
                    if phiinvs is not None:
                        pinv = _t64(phiinvs[i], self.device)
                        pinv = pinv[None, :] if pinv.dim() == 1 else pinv[lo:hi]
                        sigma = None
                    else:
                        sg = _t64(sigmas[i], self.device)
                        sigma = sg[None, :, :] if sg.dim() == 2 else sg[lo:hi]
                        # get_mats_fp contract: sigma = TNT + diag(phi^-1)
                        pinv = (
                            torch.diagonal(sigma, dim1=-2, dim2=-1)
                            - torch.diagonal(blk.TNT)[None, :]
                        )
                    if blk.comp is not None and not force_direct:
                        # Schur-compressed: C_d = diag(1/Delta_d) + G
                        c = blk.comp
                        phi_var = (
                            1.0 / (pinv[:, c["var"]] - c["delta0"][None, :])
                        ).contiguous()
                        if self._use_hip:
                            from fastfp_amd import ops

                            ops.chol_trsm_fp_accum(
                                c["G"], phi_var, c["K"], c["M0"], c["N0"],
                                fp_tgt, gsign=-1.0,
                            )
                        else:
                            sigc = c["G"][None, :, :] + torch.diag_embed(phi_var)
                            self._accum_eager_mats(
                                sigc, c["K"], c["M0"], c["N0"], fp_tgt, -1.0
                            )
                    elif self._use_hip:
                        self._accum_hip(blk, pinv.contiguous(), fp_tgt)
                    else:
                        if sigma is None:
                            sigma = blk.TNT[None, :, :] + torch.diag_embed(pinv)
                        self._accum_eager(blk, sigma, fp_tgt)

        if self._use_hip:
            ev2 = torch.cuda.Event()
            ev2.record(self._side_stream)
            main_stream.wait_event(ev2)
            fp += fp_side

        return fp[0] if not batched else fp

    def _pipeline_chunks(self, chunks, P, mp, F, fp, fp_pp, factor_into,
                         solve):
        """Factor/solve software pipeline over draw chunks with
        PING-PONG factor buffers and two-way event fencing.

        Chunk k+1's factor (side stream, buffer (k+1)&1) overlaps chunk
        k's solve (main stream); before overwriting a buffer, the side
        stream waits on the event recorded after the solve that last
        read it.  Explicit reusable buffers instead of per-chunk
        allocations + record_stream: the latter defers caching-
        allocator reuse across streams and measured as multi-GB
        allocator churn (2.2 s/step) on the 9.5 GB direct-path chunks.
        The structure is hipGraph-capturable (events + two streams
        forked from the capture stream)."""
        dev = self.device
        chunk0 = chunks[0][1] - chunks[0][0]
        nmax = P * chunk0
        key = ("pipebuf", nmax, mp)
        bufs = getattr(self, "_pipe_bufs", None)
        if bufs is None or bufs[0] != key:
            L0 = torch.empty((nmax, mp, mp), dtype=torch.float64, device=dev)
            L1 = torch.empty_like(L0)
            i0 = torch.empty((nmax, mp // 16, 16, 16), dtype=torch.float64,
                             device=dev)
            i1 = torch.empty_like(i0)
            bufs = (key, (L0, L1), (i0, i1))
            self._pipe_bufs = bufs
        _, Lb, Ib = bufs

        main = torch.cuda.current_stream(dev)
        side = self._side_stream
        side.wait_stream(main)  # inputs visible to side
        solve_done = [None, None]  # event after the solve reading buf b

        def factor(i, lo, hi):
            b = i & 1
            with torch.cuda.stream(side):
                if solve_done[b] is not None:
                    side.wait_event(solve_done[b])
                factor_into(Lb[b], Ib[b], lo, hi)
                ev = torch.cuda.Event()
                ev.record(side)
            return ev

        pending = factor(0, *chunks[0])
        for i, (lo, hi) in enumerate(chunks):
            ev = pending
            if i + 1 < len(chunks):
                pending = factor(i + 1, *chunks[i + 1])
            main.wait_event(ev)
            b = i & 1
            if hi - lo == fp_pp.shape[1]:
                pp = fp_pp
            else:
                pp = torch.empty((P, hi - lo, F), dtype=torch.float64,
                                 device=dev)
            pp.zero_()
            solve(Lb[b], Ib[b], lo, hi, pp)
            fp[lo:hi] += pp.sum(dim=0)
            ev2 = torch.cuda.Event()
            ev2.record(main)
            solve_done[b] = ev2

    def _sweep_stacked(self, phiinvs, fp, D, F, draw_chunk, batched):
        """Pulsar-batched compressed sweep: one chol + one trsm launch
        per draw chunk across ALL pulsars."""
        from fastfp_amd import ops

        st = self._comp_stack
        P = len(self.blocks)
        mv = st["mv"]
        pin = []
        for i in range(P):
            p = _t64(phiinvs[i], self.device)
            p = p[None, :] if p.dim() == 1 else p
            pin.append(p[:, st["vars"][i]])
        pinv_var = torch.stack(pin)  # (P, D, mv)
        fp_pp = torch.empty((P, min(draw_chunk, D), F),
                            dtype=torch.float64, device=self.device)
        chunks = [(lo, min(lo + draw_chunk, D))
                  for lo in range(0, D, draw_chunk)]
        if self._use_hip and len(chunks) > 1:
            # SOFTWARE PIPELINE across draw chunks: the Cholesky is
            # latency-bound (~82% parked waves), the solve is
            # throughput-bound, so chunk k+1's factor runs on a side
            # stream UNDER chunk k's solve and its parked cycles are
            # filled with solve work.
            mvp = ops.pad16(st["G"].shape[-1])

            def factor_into(L, invd, lo, hi):
                from fastfp_amd.ops import _fastfp_hip as ext

                phi_var = (
                    1.0 / (pinv_var[:, lo:hi, :]
                           - st["delta0"][:, None, :])
                ).contiguous()
                n = P * (hi - lo)
                ext.chol_batch_into(st["G"], phi_var, mvp,
                                    L[:n], invd[:n])

            def solve(L, invd, lo, hi, pp):
                from fastfp_amd.ops import _fastfp_hip as ext

                n = P * (hi - lo)
                ext.trsm_fp_accum(L[:n], invd[:n], st["K"], st["M0"],
                                  st["N0"], pp, -1.0)

            self._pipeline_chunks(chunks, P, mvp, F, fp, fp_pp,
                                  factor_into, solve)
            return fp[0] if not batched else fp

        for lo, hi in chunks:
            phi_var = (
                1.0 / (pinv_var[:, lo:hi, :] - st["delta0"][:, None, :])
            ).contiguous()
            if hi - lo == fp_pp.shape[1]:
                pp = fp_pp
            else:  # tail chunk: a dim-1 slice would be non-contiguous
                pp = torch.empty((P, hi - lo, F), dtype=torch.float64,
                                 device=self.device)
            pp.zero_()
            ops.chol_trsm_fp_accum(
                st["G"], phi_var, st["K"], st["M0"], st["N0"], pp,
                gsign=-1.0,
            )
            fp[lo:hi] += pp.sum(dim=0)
        return fp[0] if not batched else fp

    def _sweep_stacked_direct(self, phiinvs, fp, D, F, draw_chunk, batched):
        """Pulsar-batched DIRECT sweep (no compression): one chol + one
        trsm launch per draw chunk on the full-m system."""
        from fastfp_amd import ops

        st = self._direct_stack
        P = len(self.blocks)
        pin = []
        for i in range(P):
            p = _t64(phiinvs[i], self.device)
            p = p[None, :] if p.dim() == 1 else p
            pin.append(p)
        pinv_all = torch.stack(pin)  # (P, D, m)
        fp_pp = torch.empty((P, min(draw_chunk, D), F),
                            dtype=torch.float64, device=self.device)
        chunks = [(lo, min(lo + draw_chunk, D))
                  for lo in range(0, D, draw_chunk)]
        m = st["TNT"].shape[-1]
        if self._use_hip and len(chunks) > 1 and ops.pad16(m) <= ops.MAX_MP_CHOL:
            # same factor/solve software pipeline as the compressed path
            mp = ops.pad16(m)

            def factor_into(L, invd, lo, hi):
                from fastfp_amd.ops import _fastfp_hip as ext

                n = P * (hi - lo)
                ext.chol_batch_into(
                    st["TNT"], pinv_all[:, lo:hi, :].contiguous(), mp,
                    L[:n], invd[:n],
                )

            def solve(L, invd, lo, hi, pp):
                from fastfp_amd.ops import _fastfp_hip as ext

                n = P * (hi - lo)
                ext.trsm_fp_accum(L[:n], invd[:n], st["RHS"], st["sNs"],
                                  st["sNr"], pp, 1.0)

            self._pipeline_chunks(chunks, P, mp, F, fp, fp_pp,
                                  factor_into, solve)
            return fp[0] if not batched else fp

        for lo, hi in chunks:
            if hi - lo == fp_pp.shape[1]:
                pp = fp_pp
            else:
                pp = torch.empty((P, hi - lo, F), dtype=torch.float64,
                                 device=self.device)
            pp.zero_()
            ops.chol_trsm_fp_accum(
                st["TNT"], pinv_all[:, lo:hi, :].contiguous(), st["RHS"],
                st["sNs"], st["sNr"], pp, gsign=1.0,
            )
            fp[lo:hi] += pp.sum(dim=0)
        return fp[0] if not batched else fp

    def sweep_products(self, sigmas=None, phiinvs=None) -> torch.Tensor:
        """Per-pulsar corrected inner products
        [s|s, c|c, s|c, s|r, c|r], each ``(x|y) = x^T C_p^{-1} y``
        over the precomputed frequency grid.

        Fixed noise ((m,) phiinvs / (m, m) sigmas): returns (P, 5, F).
        Draw-batched ((D, m) phiinvs / (D, m, m) sigmas): returns
        (P, D, 5, F) — the noise-marginalized Fe input.

        These are exactly the quantities the Fp reduction consumes
        before its per-pulsar 2x2 solve; exposing them lets sky-
        coherent statistics (the Fe assembly, ``fastfp_amd.festat``)
        reuse the engine's precompute + one m-dim solve per (pulsar,
        draw) and pay only O(P) per additional sky location.  Runs on
        the engine's device (torch batched Cholesky/solve — setup-
        scale work, not the draw-batched Fp hot path)."""
        assert self.freqs is not None, "call precompute(freqs) first"
        F = self.freqs.shape[0]
        vals = sigmas if sigmas is not None else phiinvs
        first = _t64(vals[0], self.device)
        batched = first.dim() == (3 if sigmas is not None else 2)
        D = first.shape[0] if batched else 1
        out = torch.empty((len(self.blocks), D, 5, F),
                          dtype=torch.float64, device=self.device)
        for i, blk in enumerate(self.blocks):
            if sigmas is not None:
                sigma = _t64(sigmas[i], self.device)
                sigma = sigma[None] if sigma.dim() == 2 else sigma
            else:
                pinv = _t64(phiinvs[i], self.device)
                pinv = pinv[None, :] if pinv.dim() == 1 else pinv
                sigma = blk.TNT[None] + torch.diag_embed(pinv)
            m = blk.m
            L = torch.linalg.cholesky(sigma)  # (D, m, m)
            W = torch.linalg.solve_triangular(
                L, blk.RHS[:m, :][None].expand(D, -1, -1), upper=False
            )  # (D, m, 2F+1)
            wu = W[:, :, -1]
            Ws, Wc = W[:, :, 0:-1:2], W[:, :, 1:-1:2]
            out[i, :, 0] = blk.sNs[0][None] - (Ws * Ws).sum(1)
            out[i, :, 1] = blk.sNs[1][None] - (Wc * Wc).sum(1)
            out[i, :, 2] = blk.sNs[2][None] - (Ws * Wc).sum(1)
            out[i, :, 3] = blk.sNr[0][None] - torch.einsum(
                "dmf,dm->df", Ws, wu)
            out[i, :, 4] = blk.sNr[1][None] - torch.einsum(
                "dmf,dm->df", Wc, wu)
        return out[:, 0] if not batched else out

    def _accum_eager(self, blk: PulsarBlock, sigma, fp_out):
        """Eager per-pulsar accumulation: Cholesky + TRSM + fused 2x2."""
        self._accum_eager_mats(sigma, blk.RHS, blk.sNs, blk.sNr, fp_out, 1.0)

    def _accum_eager_mats(self, sigma, RHS, sNs, sNr, fp_out, gsign):
        Dc = sigma.shape[0]
        L = torch.linalg.cholesky(sigma)  # (Dc, m, m)
        RHSb = RHS.unsqueeze(0).expand(Dc, -1, -1)
        W = torch.linalg.solve_triangular(L, RHSb, upper=False)
        wu = W[:, :, -1]  # (Dc, m)
        Ws = W[:, :, 0:-1:2]  # (Dc, m, F)
        Wc = W[:, :, 1:-1:2]
        g_ss = (Ws * Ws).sum(dim=1)  # (Dc, F)
        g_cc = (Wc * Wc).sum(dim=1)
        g_sc = (Ws * Wc).sum(dim=1)
        n_s = torch.einsum("dmf,dm->df", Ws, wu)
        n_c = torch.einsum("dmf,dm->df", Wc, wu)

        M11 = sNs[0][None, :] - gsign * g_ss
        M22 = sNs[1][None, :] - gsign * g_cc
        M12 = sNs[2][None, :] - gsign * g_sc
        N1 = sNr[0][None, :] - gsign * n_s
        N2 = sNr[1][None, :] - gsign * n_c
        det = M11 * M22 - M12 * M12
        fp_out += 0.5 * (N1 * N1 * M22 - 2.0 * N1 * N2 * M12 + N2 * N2 * M11) / det

    def _accum_hip(self, blk: PulsarBlock, phiinv, fp_out):
        from fastfp_amd import ops

        ops.chol_trsm_fp_accum(
            blk.TNT, phiinv, blk.RHS, blk.sNs, blk.sNr, fp_out
        )
