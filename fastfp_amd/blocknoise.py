"""Block-diagonal white noise: ECORR modeled as white noise
(EcorrKernelNoise) — the reference's documented unsupported case
(``/root/reference/fastfp/utils.py:30-31``, ``README.md:22``;
BASELINE.json config 4).

Per observing epoch (and backend), ECORR adds a fully-correlated
component to the white noise: the per-epoch block is
``N_b = diag(nvec_b) + 10^(2*log10_ecorr) * J`` (J = all-ones).  This
module partitions the TOAs into such blocks, permutes them contiguous,
factors each block (Cholesky) and precomputes the dense block inverses,
and applies ``N^{-1}`` to vectors/panels.

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
        """Cholesky-factor every block and store the dense inverses,
        packed (CSR-style: values of block b at packed offset
        ``poff[b] .. poff[b] + sizes[b]^2``, row-major)."""
        self.poff = np.concatenate(([0], np.cumsum(self.sizes**2)[:-1]))
        total = int((self.sizes**2).sum())
        inv = np.empty(total, dtype=np.float64)
        logdet = 0.0
        for b in range(len(self.sizes)):
            s = int(self.sizes[b])
            o = int(self.offsets[b])
            blk = np.diag(self.nvec[o : o + s]) + self.ecorr2[b]
            L = np.linalg.cholesky(blk)
            logdet += 2.0 * np.log(np.diag(L)).sum()
            inv[self.poff[b] : self.poff[b] + s * s] = np.linalg.inv(blk).ravel()
        self.inv_packed = inv
        self.logdet = logdet
        self._t = {}

    def tensors(self, device):
        """Device tensors of the packed block data (cached)."""
        key = str(device)
        if key not in self._t:
            self._t[key] = dict(
                inv_packed=torch.as_tensor(self.inv_packed, device=device),
                sizes=torch.as_tensor(self.sizes, device=device),
                offsets=torch.as_tensor(self.offsets, device=device),
                poff=torch.as_tensor(self.poff, device=device),
            )
        return self._t[key]

    # ------------------------------------------------------------------
    def _size_groups(self, device, dtype):
        """Cached per-(device) grouped gather indices and stacked dense
        inverse blocks for the batched solve."""
        key = (str(device), str(dtype))
        cache = getattr(self, "_groups", None)
        if cache is None:
            cache = self._groups = {}
        if key not in cache:
            groups = []
            for s in np.unique(self.sizes):
                sel = np.nonzero(self.sizes == s)[0]
                offs = self.offsets[sel]
                idx = torch.as_tensor(
                    (offs[:, None] + np.arange(s)[None, :]).ravel(),
                    device=device,
                )
                invs = torch.as_tensor(
                    np.stack(
                        [
                            self.inv_packed[self.poff[b] : self.poff[b] + s * s]
                            .reshape(s, s)
                            for b in sel
                        ]
                    ),
                    device=device,
                    dtype=dtype,
                )
                groups.append((int(s), len(sel), idx, invs))
            cache[key] = groups
        return cache[key]

    def solve(self, X):
        """``N^{-1} X`` for X of shape (ntoa,) or (ntoa, k), in the
        PERMUTED TOA order.  Torch or numpy in, same type out."""
        is_np = not isinstance(X, torch.Tensor)
        Xt = torch.as_tensor(np.asarray(X) if is_np else X)
        vec = Xt.dim() == 1
        if vec:
            Xt = Xt[:, None]
        out = torch.empty_like(Xt)
        for s, nblk, idx, invs in self._size_groups(Xt.device, Xt.dtype):
            xb = Xt[idx].reshape(nblk, s, -1)
            out[idx] = torch.bmm(invs, xb).reshape(nblk * s, -1)
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
