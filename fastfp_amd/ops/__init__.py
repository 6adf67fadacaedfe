"""HIP kernel op layer for the MI355X (gfx950).

Loads the in-tree extension ``_fastfp_hip`` built by ``setup.py
build_ext --inplace`` (or ``__graft_entry__.build()``).  On a GPU box
the HIP path is mandatory: :func:`require_hip` raises if the extension
is missing so eager fallback can never silently absorb GPU runs.
Set ``FASTFP_ALLOW_EAGER_GPU=1`` only for debugging comparisons.
"""

from __future__ import annotations

import os

import torch

_ext = None
_load_err = None


def _try_load():
    global _ext, _load_err
    if _ext is not None:
        return _ext
    try:
        from fastfp_amd.ops import _fastfp_hip as ext  # built in-tree

        _ext = ext
    except ImportError as e:  # pragma: no cover - exercised on GPU boxes
        _load_err = e
        _ext = None
    return _ext


def hip_available() -> bool:
    return _try_load() is not None


def require_hip() -> None:
    if os.environ.get("FASTFP_ALLOW_EAGER_GPU") == "1":
        return
    if _try_load() is None:
        raise RuntimeError(
            "fastfp_amd HIP extension (_fastfp_hip) is not built but a GPU "
            "run was requested. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950) "
            f"or set FASTFP_ALLOW_EAGER_GPU=1 to debug with eager torch. "
            f"Original import error: {_load_err}"
        )


# ----------------------------------------------------------------------
# op wrappers (thin; shapes documented in the kernels)
# ----------------------------------------------------------------------
def freq_precompute(toas, Nvec, r, T, TNr, freqs, freq_chunk: int = 8192):
    """GPU frequency precompute for one pulsar.

    Returns (RHS (m, 2F+1), sNs (3, F), sNr (2, F)) — the fused
    sincos signal-basis kernel computes the diagonal-weighted dots and
    the N^-1-scaled sin/cos panel; the MFMA DGEMM computes
    B = T^T (N^-1 S).
    """
    ext = _try_load()
    F = freqs.shape[0]
    ntoa, m = T.shape
    device = T.device
    RHS = torch.empty((m, 2 * F + 1), dtype=torch.float64, device=device)
    sNs = torch.empty((3, F), dtype=torch.float64, device=device)
    sNr = torch.empty((2, F), dtype=torch.float64, device=device)
    Nr = r / Nvec
    Ninv = 1.0 / Nvec
    for lo in range(0, F, freq_chunk):
        hi = min(lo + freq_chunk, F)
        fchunk = freqs[lo:hi].contiguous()
        # NS panel (2Fc, ntoa) row-major + dots, one fused kernel
        NS, sNs_c, sNr_c = ext.sigbasis(toas, Ninv, Nr, fchunk)
        sNs[:, lo:hi] = sNs_c
        sNr[:, lo:hi] = sNr_c
        # B = NS @ T -> (2Fc, m); store transposed into RHS columns
        Bc = ext.dgemm_nn(NS, T)  # (2Fc, m)
        RHS[:, 2 * lo : 2 * hi].copy_(Bc.transpose(0, 1))
    RHS[:, -1] = TNr
    return RHS, sNs, sNr


def chol_trsm_fp_accum(sigma, RHS, sNs, sNr, fp_out):
    """Batched Cholesky of sigma (D, m, m) in place, then fused
    triangular solve of RHS (m, 2F+1) + 2x2 Fp reduction, accumulating
    into fp_out (D, F)."""
    ext = _try_load()
    ext.chol_batch(sigma)
    ext.trsm_fp_accum(sigma, RHS, sNs, sNr, fp_out)


def dgemm_tn(A, B):
    """C = A^T @ B for fp64 (K, M) x (K, N) -> (M, N)."""
    ext = _try_load()
    return ext.dgemm_tn(A, B)
