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


MAX_MP_CHOL = 128  # LDS-resident limit of the chol_batch kernel
MAX_MP = 256  # solve limit: trsm W registers (NBT*4 f64/lane at NBT=16)


def pad16(m: int) -> int:
    return max(16, (m + 15) // 16 * 16)


def check_m(m: int) -> int:
    """Solve-dimension check.  Up to mp=128 the batched Cholesky runs
    in the LDS-resident chol_batch kernel; 128 < mp <= 256 factors via
    batched rocSOLVER Cholesky + the diag_inv kernel and solves in the
    same register-resident trsm (W = NBT*4 doubles per lane).  Beyond
    256 use the Schur-compressed path (per-draw dimension = variable
    bins) or the CPU engine."""
    mp = pad16(m)
    if mp > MAX_MP:
        raise NotImplementedError(
            f"the direct per-draw solve supports m <= {MAX_MP} (got m={m}); "
            "use the Schur-compressed path (NMFp.sweep compress=True, on by "
            "default) whose per-draw dimension is the variable-bin count, "
            "or the CPU engine"
        )
    return mp


# ----------------------------------------------------------------------
# op wrappers (thin; shapes documented in the kernels)
# ----------------------------------------------------------------------
def freq_precompute(toas, Nvec, r, T, TNr, freqs):
    """GPU frequency precompute for one pulsar.

    Returns (RHS (mp, 2F+1), sNs (3, F), sNr (2, F)).  The fused
    sigdots kernel computes the diagonal-weighted sin/cos dots; the
    fused signal-basis MFMA DGEMM (sbgemm) computes B = T^T (N^-1 S)
    with the S panel generated in LDS — NS is never materialized.
    RHS rows are zero-padded to mp (multiple of 16).

    Unlike the eager path there is no ``freq_chunk`` knob: the sin/cos
    S panel is generated tile-by-tile in LDS inside the kernels and no
    (F, ntoa) intermediate ever exists to bound.
    """
    ext = _try_load()
    F = int(freqs.shape[0])
    ntoa, m = T.shape
    mp = pad16(m)  # sbgemm is M-tiled: no 128 cap on the basis size
    device = T.device
    Ninv = (1.0 / Nvec).contiguous()
    Nr = (r / Nvec).contiguous()
    freqs = freqs.contiguous()

    sNs, sNr = ext.sigdots(toas.contiguous(), Ninv, Nr, freqs)

    RHS = torch.zeros((mp, 2 * F + 1), dtype=torch.float64, device=device)
    F2 = 2 * F
    ctiles = (F2 + 63) // 64
    # split K to fill the 256-CU chip when the frequency grid is small;
    # partial planes + a deterministic torch reduction keep fp64
    # bitwise-reproducible (no atomics)
    ksplit = max(1, min(8, (512 + ctiles - 1) // ctiles, (ntoa + 255) // 256))
    if ksplit == 1:
        ext.sbgemm(T, toas, Ninv, freqs, RHS, 0, 2 * F + 1, mp, 1)
    else:
        part = torch.empty((ksplit, mp, F2), dtype=torch.float64, device=device)
        ext.sbgemm(T, toas, Ninv, freqs, part, mp * F2, F2, mp, ksplit)
        RHS[:, :F2] = part.sum(dim=0)
    RHS[:m, -1] = TNr
    return RHS, sNs, sNr


def freq_precompute_block(toas, block_noise, Nr, V, TNr, freqs):
    """Block-diagonal-N (EcorrKernelNoise) frequency precompute.

    ``B = T^T N^-1 S = V^T S`` reuses the sbgemm kernel with V in place
    of T and unit weights; the per-frequency quadratics go through the
    sigdots_block kernel, which applies the exact Sherman–Morrison
    rank-1 correction per epoch block (any epoch size — no cap; the
    per-epoch blocks are diagonal + rank-1, see fastfp_amd.blocknoise).
    """
    ext = _try_load()
    F = int(freqs.shape[0])
    ntoa, m = V.shape
    mp = pad16(m)  # sbgemm is M-tiled: no 128 cap on the basis size
    device = V.device
    freqs = freqs.contiguous()

    bt = block_noise.tensors(device)
    sNs, sNr = ext.sigdots_block(
        toas.contiguous(), bt["uvec"].contiguous(), Nr.contiguous(), freqs,
        bt["beta"].contiguous(), bt["offsets"], bt["sizes"],
    )

    RHS = torch.zeros((mp, 2 * F + 1), dtype=torch.float64, device=device)
    F2 = 2 * F
    ones = torch.ones(ntoa, dtype=torch.float64, device=device)
    ctiles = (F2 + 63) // 64
    ksplit = max(1, min(8, (512 + ctiles - 1) // ctiles, (ntoa + 255) // 256))
    if ksplit == 1:
        ext.sbgemm(V, toas, ones, freqs, RHS, 0, 2 * F + 1, mp, 1)
    else:
        part = torch.empty((ksplit, mp, F2), dtype=torch.float64, device=device)
        ext.sbgemm(V, toas, ones, freqs, part, mp * F2, F2, mp, ksplit)
        RHS[:, :F2] = part.sum(dim=0)
    RHS[:m, -1] = TNr
    return RHS, sNs, sNr


def chol_trsm_fp_accum(TNT, phiinv, RHS, sNs, sNr, fp_out, gsign=1.0):
    """Batched Cholesky of Sigma = TNT + diag(phiinv), then the fused
    triangular solve of RHS (mp, 2F+1) + 2x2 Fp reduction, accumulating
    into fp_out (D, F).

    mp <= 128: LDS-resident chol_batch kernel.  128 < mp <= 256
    (GP-ECORR direct path): Sigma no longer fits LDS — factor through
    batched rocSOLVER Cholesky (substitution-based, accurate; the
    rocBLAS accuracy caveat of docs/TUNING_NOTES.md concerns its
    inversion-based trsm, not potrf) + the diag_inv kernel, then the
    same register-resident trsm.

    ``gsign``: +1 direct path (M = sNs - W.W); -1 Schur-compressed draw
    path (M = M0 + W.W) — docs/DESIGN.md §draw compression."""
    ext = _try_load()
    m = TNT.shape[-1]
    mp = check_m(m)
    if mp <= MAX_MP_CHOL:
        L, invd = ext.chol_batch(TNT.contiguous(), phiinv.contiguous(), mp)
    else:
        batched = TNT.dim() == 3
        TNTb = TNT if batched else TNT.unsqueeze(0)  # (P, m, m)
        pib = phiinv if batched else phiinv.unsqueeze(0)  # (P, D, m)
        P, D = pib.shape[0], pib.shape[1]
        sigma = TNTb[:, None, :, :] + torch.diag_embed(pib)  # (P, D, m, m)
        Lt = torch.linalg.cholesky(sigma.reshape(P * D, m, m))
        L = torch.zeros((P * D, mp, mp), dtype=TNT.dtype, device=TNT.device)
        L[:, :m, :m] = Lt
        # identity padding: pad rows of RHS are zero, so results are
        # unchanged (same convention as chol_batch)
        idx = torch.arange(m, mp, device=TNT.device)
        L[:, idx, idx] = 1.0
        invd = ext.diag_inv(L.contiguous())
    ext.trsm_fp_accum(L, invd, RHS, sNs, sNr, fp_out, gsign)
