"""Basis construction tests."""

import numpy as np

from fastfp_amd.bases import (
    create_freqarray,
    create_quantization_matrix,
    ecorr_basis_by_backend,
    fourier_basis,
    timing_model_basis_svd,
)
from fastfp_amd.constants import day
from fastfp_amd.data import make_synthetic_pta


def test_create_freqarray():
    Ff = create_freqarray(100.0, 3)
    np.testing.assert_allclose(
        Ff, [0.01, 0.01, 0.02, 0.02, 0.03, 0.03], rtol=1e-15
    )


def test_fourier_basis_ordering():
    toas = np.array([0.0, 10.0, 25.0])
    Ff = create_freqarray(100.0, 2)
    F = fourier_basis(toas, Ff)
    assert F.shape == (3, 4)
    np.testing.assert_allclose(F[:, 0], np.sin(2 * np.pi * 0.01 * toas), atol=1e-14)
    np.testing.assert_allclose(F[:, 1], np.cos(2 * np.pi * 0.01 * toas), atol=1e-14)
    np.testing.assert_allclose(F[:, 2], np.sin(2 * np.pi * 0.02 * toas), atol=1e-14)
    np.testing.assert_allclose(F[:, 3], np.cos(2 * np.pi * 0.02 * toas), atol=1e-14)


def test_svd_basis_orthonormal_and_spans():
    psr = make_synthetic_pta(npsr=1, ntoa=100, ntm=5, seed=0, ragged=False)[0]
    U = timing_model_basis_svd(psr.Mmat)
    assert U.shape == (100, 5)
    np.testing.assert_allclose(U.T @ U, np.eye(5), atol=1e-12)
    # span check: projector onto col(U) reproduces Mmat
    P = U @ U.T
    np.testing.assert_allclose(P @ psr.Mmat, psr.Mmat, atol=1e-8)


def test_quantization_buckets():
    # three epochs: [0, .2, .4d], [5d, 5.1d], lone TOA at 20d dropped
    toas = np.array([0.0, 0.2 * day, 0.4 * day, 5.0 * day, 5.1 * day, 20.0 * day])
    U, w = create_quantization_matrix(toas, dt=day, nmin=2)
    assert U.shape == (6, 2)
    np.testing.assert_allclose(w, [1.0, 1.0])
    np.testing.assert_allclose(U[:, 0], [1, 1, 1, 0, 0, 0])
    np.testing.assert_allclose(U[:, 1], [0, 0, 0, 1, 1, 0])


def test_quantization_bucket_by_first_ref():
    # bucketing compares to the FIRST toa of the bucket (reference
    # semantics, run_nmfp.py:47-52): 0, 0.6d, 1.2d -> [0, .6d], [1.2d]
    toas = np.array([0.0, 0.6 * day, 1.2 * day, 1.3 * day])
    U, w = create_quantization_matrix(toas, dt=day, nmin=1)
    assert U.shape[1] == 2
    np.testing.assert_allclose(U[:, 0], [1, 1, 0, 0])
    np.testing.assert_allclose(U[:, 1], [0, 0, 1, 1])


def test_ecorr_basis_by_backend():
    psr = make_synthetic_pta(npsr=1, ntoa=60, ntm=3, seed=1)[0]
    U, weights = ecorr_basis_by_backend(psr)
    backends = np.unique(psr.backend_flags)
    assert len(weights) == len(backends)
    assert U.shape == (psr.ntoa, sum(len(w) for w in weights))
    # every column is an indicator within a single backend
    col = 0
    for b, w in zip(backends, weights):
        mask = np.asarray(psr.backend_flags == b)
        for _ in range(len(w)):
            assert set(np.unique(U[:, col])) <= {0.0, 1.0}
            assert U[~mask, col].sum() == 0
            col += 1
