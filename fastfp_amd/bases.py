"""Gaussian-process basis construction: timing model, Fourier red noise,
and ECORR quantization bases.

The reference obtains these from ``enterprise`` (``pta.get_basis``,
``/root/reference/fastfp/utils.py:75``); this module builds them natively.
Conventions (self-consistent with :mod:`fastfp_amd.noise` phi ordering,
mirroring the reference's phi-block order tm|ecorr|rn at
``/root/reference/fastfp/nmfp.py:282``):

- timing-model basis = left singular vectors of Mmat (``use_svd=True``
  semantics, ``/root/reference/fastfp/utils.py:146``),
- Fourier basis columns interleave sin/cos per frequency with
  ``f_k = k/Tspan`` each repeated twice (``/root/reference/fastfp/nmfp.py:201-215``),
- ECORR quantization basis: per sorted backend, per epoch bucket
  (>= nmin TOAs within dt), an indicator column.
"""

from __future__ import annotations

import numpy as np

from fastfp_amd.constants import day


def timing_model_basis_svd(Mmat: np.ndarray) -> np.ndarray:
    """Orthonormal timing-model basis: U from the thin SVD of Mmat.

    Matches enterprise's ``TimingModel(use_svd=True)``: the returned
    basis spans col(Mmat) with orthonormal columns, which keeps the
    1e40 improper-flat prior block of Sigma well-scaled.
    """
    U, _s, _Vt = np.linalg.svd(np.asarray(Mmat, dtype=np.float64), full_matrices=False)
    return U


def create_freqarray(Tspan: float, ncomps: int = 30) -> np.ndarray:
    """Fourier frequencies ``k/Tspan`` for ``k=1..ncomps``, each repeated
    twice (sin & cos) — parity with ``RN_container._create_freqarray``
    (``/root/reference/fastfp/nmfp.py:201-215``)."""
    f = np.arange(1, ncomps + 1, dtype=np.float64) / Tspan
    return np.repeat(f, 2)


def fourier_basis(toas: np.ndarray, Ffreqs: np.ndarray) -> np.ndarray:
    """Fourier sin/cos design matrix, columns ordered
    ``[sin f1, cos f1, sin f2, cos f2, ...]`` to match ``Ffreqs``
    (frequencies repeated pairwise).

    Shape (ntoa, len(Ffreqs)).
    """
    toas = np.asarray(toas, dtype=np.float64)
    f = np.asarray(Ffreqs, dtype=np.float64)[::2]  # unique frequencies
    arg = 2.0 * np.pi * np.outer(toas, f)
    F = np.empty((toas.shape[0], 2 * f.shape[0]), dtype=np.float64)
    F[:, ::2] = np.sin(arg)
    F[:, 1::2] = np.cos(arg)
    return F


def create_quantization_matrix(
    toas: np.ndarray, dt: float = day, nmin: int = 2
) -> tuple:
    """Epoch quantization of TOAs: buckets of TOAs within ``dt`` seconds
    of the bucket's first TOA; buckets with fewer than ``nmin`` TOAs are
    dropped.  Returns ``(U, weights)`` with U (ntoa, nepoch) the 0/1
    indicator matrix and weights = ones(nepoch).

    Mirrors the bucketing logic of the reference's
    ``create_quantization_array`` (``/root/reference/examples/run_nmfp.py:38-57``)
    but takes ``dt`` in seconds (default one day); the reference passes
    ``dt=1`` against TOAs in seconds, which buckets 1-second epochs — a
    quirk we deliberately fix (documented in SURVEY.md §2.5).
    """
    toas = np.asarray(toas, dtype=np.float64)
    isort = np.argsort(toas, kind="stable")

    bucket_ref = [toas[isort[0]]]
    bucket_ind = [[isort[0]]]
    for i in isort[1:]:
        if toas[i] - bucket_ref[-1] < dt:
            bucket_ind[-1].append(i)
        else:
            bucket_ref.append(toas[i])
            bucket_ind.append([i])

    kept = [ind for ind in bucket_ind if len(ind) >= nmin]
    U = np.zeros((toas.shape[0], len(kept)), dtype=np.float64)
    for j, ind in enumerate(kept):
        U[ind, j] = 1.0
    weights = np.ones(len(kept), dtype=np.float64)
    return U, weights


def create_quantization_array(toas, dt: float = day, nmin: int = 2):
    """Per-epoch quantization weights only — name/signature parity with
    the reference helper (``/root/reference/examples/run_nmfp.py:38-57``;
    weights hardcoded to 1.0 there and here)."""
    _, weights = create_quantization_matrix(toas, dt=dt, nmin=nmin)
    return weights


def ecorr_weights_by_backend(psr, dt: float = day, nmin: int = 2) -> list:
    """Quantization weights split by receiver backend — parity with
    ``/root/reference/examples/run_nmfp.py:60-70``."""
    _, weights = ecorr_basis_by_backend(psr, dt=dt, nmin=nmin)
    return weights


def ecorr_basis_by_backend(psr, dt: float = day, nmin: int = 2) -> tuple:
    """Build the per-backend ECORR quantization basis for one pulsar.

    Backends are processed in sorted (``np.unique``) order to match the
    phi ordering of :class:`fastfp_amd.noise.GPEcorrContainer`
    (reference reads backends the same way,
    ``/root/reference/fastfp/nmfp.py:443-452``).

    Returns ``(U, weights_list)``: U (ntoa, total_epochs) block columns
    per backend, and the per-backend weight arrays.
    """
    backends = np.unique(psr.backend_flags)
    cols = []
    weights_list = []
    for b in backends:
        mask = np.asarray(psr.backend_flags == b)
        idx = np.nonzero(mask)[0]
        Ub, wb = create_quantization_matrix(psr.toas[idx], dt=dt, nmin=nmin)
        Ufull = np.zeros((psr.ntoa, Ub.shape[1]), dtype=np.float64)
        Ufull[idx, :] = Ub
        cols.append(Ufull)
        weights_list.append(wb)
    U = (
        np.concatenate(cols, axis=1)
        if cols
        else np.zeros((psr.ntoa, 0), dtype=np.float64)
    )
    return U, weights_list
