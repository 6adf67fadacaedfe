"""Plain Fp-statistic API: :class:`FastFp`.

API parity with the reference's ``FastFp``
(``/root/reference/fastfp/fastfp.py:22-101``): construction from
``(psrs, pta)``, a per-frequency ``calculate_Fp(fgw, Nvecs, Ts, sigmas)``
and ``__call__``.  Additionally provides the production entry point
:meth:`sweep` which evaluates the whole frequency grid through the
restructured :class:`fastfp_amd.engine.FpEngine` (one Cholesky per
pulsar, one GEMM per pulsar — not 6 Woodbury solves per (f, pulsar)).
"""

from __future__ import annotations

import math

import numpy as np
import torch

from fastfp_amd.engine import FpEngine
from fastfp_amd.xcy import get_xCy


class FastFp:
    """Fp detection statistic (Ellis, Siemens & Creighton 2012)."""

    def __init__(self, psrs, pta=None):
        self.psrs = psrs
        self.pta = pta
        self.toas = [np.asarray(p.toas, dtype=np.float64) for p in psrs]
        self.residuals = [np.asarray(p.residuals, dtype=np.float64) for p in psrs]

    def __call__(self, fgw, Nvecs, Ts, sigmas):
        return self.calculate_Fp(fgw, Nvecs, Ts, sigmas)

    # ------------------------------------------------------------------
    # parity path: single frequency, explicit Woodbury products
    # ------------------------------------------------------------------
    def calculate_Fp(self, fgw, Nvecs, Ts, sigmas) -> float:
        """Single-frequency Fp — same evaluation order as the reference
        (filter amplitude ``fgw^-1/3`` included for bit-level parity,
        ``/root/reference/fastfp/fastfp.py:69-92``)."""
        fstat = 0.0
        amp = 1.0 / fgw ** (1.0 / 3.0)
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
    # production path: whole grid at once on the engine
    # ------------------------------------------------------------------
    def sweep(
        self,
        freqs,
        Nvecs,
        Ts,
        sigmas,
        device: str = None,
        freq_chunk: int = 2048,
    ) -> np.ndarray:
        """Fp over a frequency grid.  Returns (F,) numpy array.

        The direct GPU solve covers basis sizes m <= 256 (rocSOLVER-
        factored above 128, fastfp_amd.ops.chol_trsm_fp_accum).  For a
        larger basis (extreme GP-ECORR models) the sweep routes through
        the Schur compression when the PTAModel's variable-bin info is
        available, else falls back to the CPU LAPACK engine."""
        if device is None:
            device = "cuda" if torch.cuda.is_available() else "cpu"
        eng = FpEngine(self.psrs, Nvecs, Ts, device=device)
        eng.precompute(freqs, freq_chunk=freq_chunk)
        max_m = max(T.shape[1] for T in Ts)
        conts = getattr(self.pta, "rn_containers", None)
        if eng._use_hip and max_m > 256 and conts is not None:
            phiinvs = [
                np.diag(np.asarray(sg)) - np.diag(np.asarray(TNT))
                for sg, TNT in zip(
                    sigmas, (blk.TNT.cpu().numpy() for blk in eng.blocks)
                )
            ]
            eng.enable_draw_compression(
                [c.var_slice for c in conts], phiinvs
            )
            # the compressed route is the ONLY m>256 GPU path, so guard
            # it twice: the accuracy probe may have dropped pulsars
            # (blk.comp None -> their direct solve would exceed the
            # kernel cap), and near-degenerate priors shrink the
            # phi/jitter margin the Woodbury correction relies on.  In
            # either case fall back to the CPU LAPACK engine — slower
            # but always exact.
            # threshold 1.5: see the margin-guard note in nmfp.sweep
            margin = eng.compression_margin(phiinvs)
            if any(blk.comp is None for blk in eng.blocks) or margin < 1.5:
                import warnings

                warnings.warn(
                    "m > 256 GPU sweep: compression unsafe "
                    f"(margin={margin:.1e}); falling back to the CPU engine"
                )
                eng = FpEngine(self.psrs, Nvecs, Ts, device="cpu")
                eng.precompute(freqs, freq_chunk=freq_chunk)
                fp = eng.sweep(sigmas=sigmas)
            else:
                fp = eng.sweep(phiinvs=phiinvs)
        elif eng._use_hip and max_m > 256:
            import warnings

            warnings.warn(
                "basis size m > 256 without PTAModel variable-bin info; "
                "falling back to the CPU engine"
            )
            eng = FpEngine(self.psrs, Nvecs, Ts, device="cpu")
            eng.precompute(freqs, freq_chunk=freq_chunk)
            fp = eng.sweep(sigmas=sigmas)
        else:
            fp = eng.sweep(sigmas=sigmas)
        return fp.cpu().numpy()


def compute_Fp(psrs, pta, noise, freqs, device=None) -> np.ndarray:
    """One-call convenience: precompute mats and sweep the grid."""
    from fastfp_amd.model import get_mats_fp

    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    return FastFp(psrs, pta).sweep(freqs, Nvecs, Ts, sigmas, device=device)
