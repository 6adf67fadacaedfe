"""Block-diagonal white noise: ECORR modeled as white noise
(EcorrKernelNoise) — the reference's documented unsupported case
(``/root/reference/fastfp/utils.py:30-31``, ``README.md:22``;
BASELINE.json config 4).

Per observing epoch (and backend), ECORR adds a fully-correlated
component to the white noise: the per-epoch block is
``N_b = diag(nvec_b) + 10^(2*log10_ecorr) * J`` (J = all-ones) —
**diagonal plus rank-1**, so Sherman–Morrison gives the exact inverse
in closed form with O(n) work and no per-block factorization:

    N_b^{-1} = D^{-1} - beta_b * d d^T,   d = D^{-1} 1,
    beta_b   = e2 / (1 + e2 * s_b),       s_b = sum_i 1/nvec_i,
    logdet   = sum_i log(nvec_i) + log1p(e2 * s_b).

This replaces the round-1 per-block Cholesky + dense-inverse design
(and removes its 32-TOA-per-epoch GPU cap: the correction is a single
weighted sum per block at any epoch size).  Numerical safety: the
subtracted diagonal term is ``beta * d_i^2 <= (1/n_i) * w_i`` with
``w_i = (1/n_i)/s_b`` the TOA's weight fraction in its block, so the
cancellation is bounded by the dominance of a single TOA — benign for
PTA data; validated against the dense oracle in tests at rtol 1e-9.

This module partitions the TOAs into such blocks, permutes them
contiguous, precomputes (u = 1/nvec, beta), and applies ``N^{-1}`` to
vectors/panels.

Engine integration (see ``fastfp_amd/engine.py``): with
``V = N^{-1} T`` precomputed once, the frequency-domain GEMM is
unchanged (``T^T N^{-1} S = V^T S``), and only the per-frequency
quadratics ``s^T N^{-1} s`` need the block structure — the
``sigdots_block`` HIP kernel (CPU eager equivalent here).
"""

from __future__ import annotations

import numpy as np
import torch

from fastfp_amd.constants import day
from fastfp_amd.noise import white_noise_nvec


class BlockNoise:
    """Block-diagonal white-noise covariance for one pulsar.

    Attributes (after construction):
      perm       : (ntoa,) permutation making blocks contiguous
      sizes      : (nblk,) block sizes (1 for un-quantized TOAs)
      offsets    : (nblk,) start index of each block (permuted order)
      nvec       : (ntoa,) permuted diagonal variances
      ecorr2     : (nblk,) per-block ECORR variance (0 for singletons)
    """

    def __init__(self, psr, noise: dict = None, simple_wn: bool = True,
                 select: str = "backend", dt: float = day, nmin: int = 2):
        self.psr = psr
        nvec_full = white_noise_nvec(psr, noise, simple_wn=simple_wn,
                                     select=select)
        noise = noise or {}

        # partition: per backend, epoch buckets (>= nmin TOAs within dt
        # of the bucket's first TOA) become correlated blocks; all other
        # TOAs are 1x1 blocks.  Same bucketing as the GP-ECORR basis
        # (fastfp_amd.bases.create_quantization_matrix).
        blocks = []  # (indices, ecorr2)
        used = np.zeros(psr.ntoa, dtype=bool)
        backends = np.unique(psr.backend_flags)
        for b in backends:
            key = "_".join([psr.name, "basis", "ecorr", str(b), "log10_ecorr"])
            if key not in noise:
                continue
            e2 = 10.0 ** (2.0 * float(noise[key]))
            idx = np.nonzero(np.asarray(psr.backend_flags == b))[0]
            toas_b = psr.toas[idx]
            isort = np.argsort(toas_b, kind="stable")
            ref = toas_b[isort[0]]
            bucket = [idx[isort[0]]]
            buckets = []
            for k in isort[1:]:
                if toas_b[k] - ref < dt:
                    bucket.append(idx[k])
                else:
                    buckets.append(bucket)
                    bucket = [idx[k]]
                    ref = toas_b[k]
            buckets.append(bucket)
            for bk in buckets:
                if len(bk) >= nmin:
                    blocks.append((np.asarray(bk), e2))
                    used[bk] = True
        for i in np.nonzero(~used)[0]:
            blocks.append((np.asarray([i]), 0.0))
        # order blocks by first TOA for locality
        blocks.sort(key=lambda be: psr.toas[be[0][0]])

        self.perm = np.concatenate([b[0] for b in blocks])
        self.sizes = np.asarray([len(b[0]) for b in blocks], dtype=np.int64)
        self.offsets = np.concatenate(([0], np.cumsum(self.sizes)[:-1]))
        self.ecorr2 = np.asarray([b[1] for b in blocks], dtype=np.float64)
        self.nvec = nvec_full[self.perm]
        self.max_block = int(self.sizes.max())

        self._factor()

    # ------------------------------------------------------------------
    def _factor(self):
        """Sherman–Morrison precompute: u = 1/nvec, per-block
        ``s = sum(u)``, ``beta = e2/(1 + e2*s)``, logdet."""
        self.uvec = 1.0 / self.nvec
        s = np.add.reduceat(self.uvec, self.offsets)
        self.beta = self.ecorr2 / (1.0 + self.ecorr2 * s)
        self.logdet = float(
            np.log(self.nvec).sum() + np.log1p(self.ecorr2 * s).sum()
        )
        # per-TOA block id for the vectorized segment sums in solve()
        self.blk_id = np.repeat(
            np.arange(len(self.sizes), dtype=np.int64), self.sizes
        )
        self._t = {}

    def tensors(self, device):
        """Device tensors of the block data (cached): u = 1/nvec per
        TOA, per-block beta, offsets, sizes (int64)."""
        key = str(device)
        if key not in self._t:
            self._t[key] = dict(
                uvec=torch.as_tensor(self.uvec, device=device),
                beta=torch.as_tensor(self.beta, device=device),
                sizes=torch.as_tensor(self.sizes, device=device),
                offsets=torch.as_tensor(self.offsets, device=device),
                blk_id=torch.as_tensor(self.blk_id, device=device),
            )
        return self._t[key]

    def solve(self, X):
        """``N^{-1} X`` for X of shape (ntoa,) or (ntoa, k), in the
        PERMUTED TOA order.  Torch or numpy in, same type out.

        Sherman–Morrison: ``Z = D^{-1}X``; per block,
        ``out = Z - u * beta_b * (1^T Z)_b`` (one segment sum per
        column, no dense blocks)."""
        is_np = not isinstance(X, torch.Tensor)
        Xt = torch.as_tensor(np.asarray(X) if is_np else X)
        vec = Xt.dim() == 1
        if vec:
            Xt = Xt[:, None]
        bt = self.tensors(Xt.device)
        u = bt["uvec"].to(Xt.dtype)
        beta = bt["beta"].to(Xt.dtype)
        blk_id = bt["blk_id"]
        Z = Xt * u[:, None]
        g = torch.zeros(
            (len(self.sizes), Xt.shape[1]), dtype=Xt.dtype, device=Xt.device
        )
        g.index_add_(0, blk_id, Z)
        out = Z - u[:, None] * (beta[:, None] * g)[blk_id]
        out = out[:, 0] if vec else out
        return out.numpy() if is_np else out

    def quad(self, x, y) -> float:
        """``x^T N^{-1} y`` (permuted order)."""
        return float(np.dot(np.asarray(x), self.solve(np.asarray(y))))

    def dense(self) -> np.ndarray:
        """Full dense N (permuted order) — tests only."""
        n = int(self.sizes.sum())
        N = np.zeros((n, n))
        for b in range(len(self.sizes)):
            s, o = int(self.sizes[b]), int(self.offsets[b])
            N[o : o + s, o : o + s] = (
                np.diag(self.nvec[o : o + s]) + self.ecorr2[b]
            )
        return N
