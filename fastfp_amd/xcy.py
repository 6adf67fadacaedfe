"""Woodbury inner product ``x^T C^-1 y`` — API-parity scalar path.

Same math as the reference's ``get_xCy``
(``/root/reference/fastfp/utils.py:26-54``): ``C = N + T B T^T`` with
diagonal N, evaluated as
``x^T N^-1 y - (T^T N^-1 x)^T Sigma^-1 (T^T N^-1 y)``.

This is the *parity* entry point (used by ``FastFp.calculate_Fp`` and
tests); the production sweep path in :mod:`fastfp_amd.engine` factors
Sigma once per (pulsar, draw) instead of re-solving per call.
"""

from __future__ import annotations

import numpy as np
import torch


def get_xCy(Nvec, T, sigma, x, y):
    """Compute ``x^T C^-1 y``.  ``Nvec`` is a diagonal-variance vector,
    or a :class:`fastfp_amd.blocknoise.BlockNoise` for block-diagonal N
    (the EcorrKernelNoise case the reference does not support,
    ``/root/reference/fastfp/utils.py:30-31``).  Accepts numpy arrays or
    torch tensors; returns a python float (numpy inputs) or 0-dim torch
    tensor (torch inputs)."""
    from fastfp_amd.blocknoise import BlockNoise

    if isinstance(Nvec, BlockNoise):
        perm = Nvec.perm
        xp = np.asarray(x, dtype=np.float64)[perm]
        yp = np.asarray(y, dtype=np.float64)[perm]
        Tp = np.asarray(T, dtype=np.float64)[perm, :]
        Nx = Nvec.solve(xp)
        Ny = Nvec.solve(yp)
        TNx = Tp.T @ Nx
        TNy = Tp.T @ Ny
        xNy = float(xp @ Ny)
        return xNy - float(TNx @ np.linalg.solve(np.asarray(sigma), TNy))

    torch_in = any(isinstance(a, torch.Tensor) for a in (Nvec, T, sigma, x, y))

    def cv(a):
        if isinstance(a, torch.Tensor):
            return a.to(dtype=torch.float64)
        return torch.as_tensor(np.asarray(a, dtype=np.float64))

    Nvec, T, sigma, x, y = map(cv, (Nvec, T, sigma, x, y))
    Nx = x / Nvec
    Ny = y / Nvec
    TNx = T.transpose(0, 1) @ Nx
    TNy = T.transpose(0, 1) @ Ny
    xNy = torch.dot(x, Ny)
    out = xNy - TNx @ torch.linalg.solve(sigma, TNy)
    return out if torch_in else float(out)


def get_xCy_blockdiag(Nblocks, block_index, T, sigma, x, y):
    """``x^T C^-1 y`` with BLOCK-diagonal white noise N (ECORR modeled as
    white noise) — the reference's documented unsupported case
    (``/root/reference/fastfp/utils.py:30-31``, ``README.md:22``).

    ``Nblocks``: list of per-epoch dense SPD blocks; ``block_index``:
    list of index arrays selecting each epoch's TOAs (every TOA must be
    covered exactly once).
    """

    def cv(a):
        if isinstance(a, torch.Tensor):
            return a.to(dtype=torch.float64)
        return torch.as_tensor(np.asarray(a, dtype=np.float64))

    T, sigma, x, y = map(cv, (T, sigma, x, y))
    Nx = torch.empty_like(x)
    Ny = torch.empty_like(y)
    for blk, idx in zip(Nblocks, block_index):
        blk = cv(blk)
        L = torch.linalg.cholesky(blk)
        Nx[idx] = torch.cholesky_solve(x[idx, None], L)[:, 0]
        Ny[idx] = torch.cholesky_solve(y[idx, None], L)[:, 0]
    TNx = T.transpose(0, 1) @ Nx
    TNy = T.transpose(0, 1) @ Ny
    xNy = torch.dot(x, Ny)
    return float(xNy - TNx @ torch.linalg.solve(sigma, TNy))
