"""Woodbury inner product vs the dense oracle."""

import numpy as np
import pytest

from fastfp_amd.xcy import get_xCy, get_xCy_blockdiag
from oracle import dense_xCy


def _tiny_system(rng, ntoa=60, m=9):
    Nvec = rng.uniform(0.5, 2.0, ntoa)
    T = rng.normal(size=(ntoa, m))
    # moderate prior scale: the dense oracle cannot survive 1e40
    phi = rng.uniform(0.1, 10.0, m)
    x = rng.normal(size=ntoa)
    y = rng.normal(size=ntoa)
    return Nvec, T, phi, x, y


def _sigma(Nvec, T, phi):
    TNT = T.T @ (T / Nvec[:, None])
    return TNT + np.diag(1.0 / phi)


@pytest.mark.parametrize("seed", [0, 1, 2])
def test_get_xcy_matches_dense(seed):
    rng = np.random.default_rng(seed)
    Nvec, T, phi, x, y = _tiny_system(rng)
    sigma = _sigma(Nvec, T, phi)
    got = get_xCy(Nvec, T, sigma, x, y)
    want = dense_xCy(Nvec, T, phi, x, y)
    assert got == pytest.approx(want, rel=1e-9)


def test_get_xcy_symmetric():
    rng = np.random.default_rng(3)
    Nvec, T, phi, x, y = _tiny_system(rng)
    sigma = _sigma(Nvec, T, phi)
    assert get_xCy(Nvec, T, sigma, x, y) == pytest.approx(
        get_xCy(Nvec, T, sigma, y, x), rel=1e-10
    )


def test_get_xcy_blockdiag_reduces_to_diag():
    """Block-diagonal path with 1x1 blocks == diagonal path."""
    rng = np.random.default_rng(4)
    Nvec, T, phi, x, y = _tiny_system(rng, ntoa=30, m=5)
    sigma = _sigma(Nvec, T, phi)
    blocks = [np.array([[v]]) for v in Nvec]
    index = [np.array([i]) for i in range(len(Nvec))]
    got = get_xCy_blockdiag(blocks, index, T, sigma, x, y)
    want = get_xCy(Nvec, T, sigma, x, y)
    assert got == pytest.approx(want, rel=1e-9)


def test_get_xcy_blockdiag_dense_oracle():
    """Real 2x2/3x3 blocks vs a dense solve with the full block-diag N."""
    rng = np.random.default_rng(5)
    ntoa, m = 24, 5
    T = rng.normal(size=(ntoa, m))
    phi = rng.uniform(0.1, 10.0, m)
    x = rng.normal(size=ntoa)
    y = rng.normal(size=ntoa)
    # random SPD blocks covering all TOAs
    sizes = [3, 2, 3, 2, 3, 2, 3, 2, 2, 2]
    assert sum(sizes) == ntoa
    blocks, index, Nfull = [], [], np.zeros((ntoa, ntoa))
    pos = 0
    for s in sizes:
        A = rng.normal(size=(s, s))
        blk = A @ A.T + s * np.eye(s)
        idx = np.arange(pos, pos + s)
        blocks.append(blk)
        index.append(idx)
        Nfull[np.ix_(idx, idx)] = blk
        pos += s
    # Sigma with block-diag N: TNT = T^T Nfull^-1 T
    Ninv = np.linalg.inv(Nfull)
    sigma = T.T @ Ninv @ T + np.diag(1.0 / phi)
    got = get_xCy_blockdiag(blocks, index, T, sigma, x, y)
    C = Nfull + T @ np.diag(phi) @ T.T
    want = float(x @ np.linalg.solve(C, y))
    assert got == pytest.approx(want, rel=1e-9)
