"""EcorrKernelNoise block-diagonal white-noise path (BASELINE config 4):
the case the reference documents as unsupported
(/root/reference/fastfp/utils.py:30-31, README.md:22)."""

import numpy as np
import pytest

from fastfp_amd import (
    FastFp,
    get_mats_fp,
    initialize_pta,
    make_synthetic_pta,
)
from fastfp_amd.blocknoise import BlockNoise
from fastfp_amd.xcy import get_xCy


def _psr_and_noise(seed=0, npsr=1, ntoa=80):
    psrs = make_synthetic_pta(npsr=npsr, ntoa=ntoa, ntm=3, seed=seed)
    noise = {}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
        for b in np.unique(p.backend_flags):
            noise[f"{p.name}_basis_ecorr_{b}_log10_ecorr"] = -6.3
    return psrs, noise


def test_blocknoise_partition_covers_all_toas():
    psrs, noise = _psr_and_noise()
    bn = BlockNoise(psrs[0], noise)
    assert sorted(bn.perm.tolist()) == list(range(psrs[0].ntoa))
    assert bn.sizes.sum() == psrs[0].ntoa
    assert (bn.ecorr2[bn.sizes >= 2] > 0).all()
    assert (bn.ecorr2[bn.sizes == 1] == 0).all()


def test_blocknoise_solve_vs_dense():
    psrs, noise = _psr_and_noise()
    bn = BlockNoise(psrs[0], noise)
    N = bn.dense()
    rng = np.random.default_rng(1)
    X = rng.normal(size=(psrs[0].ntoa, 3))
    got = bn.solve(X)
    want = np.linalg.solve(N, X)
    np.testing.assert_allclose(got, want, rtol=1e-9, atol=1e-3)
    x, y = X[:, 0], X[:, 1]
    assert bn.quad(x, y) == pytest.approx(float(x @ np.linalg.solve(N, y)), rel=1e-9)


def test_blocknoise_solve_vs_dense_large_epochs():
    """Sherman–Morrison solve on epochs far beyond 32 TOAs (the old
    GPU block-Cholesky cap) and with heterogeneous toaerrs — exactness
    against the dense oracle."""
    rng = np.random.default_rng(7)
    psrs = make_synthetic_pta(npsr=1, ntoa=350, tspan_yr=0.02, ntm=3, seed=7)
    psr = psrs[0]
    psr.toaerrs = rng.uniform(0.3e-6, 3e-6, psr.ntoa)
    noise = {}
    for b in np.unique(psr.backend_flags):
        noise[f"{psr.name}_basis_ecorr_{b}_log10_ecorr"] = -6.0
    bn = BlockNoise(psr, noise)
    assert bn.max_block > 32
    N = bn.dense()
    X = rng.normal(size=(psr.ntoa, 4))
    got = bn.solve(X)
    want = np.linalg.solve(N, X)
    np.testing.assert_allclose(got, want, rtol=1e-9, atol=1e-4)
    assert bn.logdet == pytest.approx(float(np.linalg.slogdet(N)[1]), rel=1e-12)


def test_blocknoise_solve_extreme_ecorr():
    """Stress the S-M cancellation bound: ECORR variance 1e6x the
    white-noise level (beta*d_i^2 approaches 1/n_i)."""
    rng = np.random.default_rng(8)
    psrs = make_synthetic_pta(npsr=1, ntoa=120, tspan_yr=0.01, ntm=3, seed=8)
    psr = psrs[0]
    noise = {}
    for b in np.unique(psr.backend_flags):
        noise[f"{psr.name}_basis_ecorr_{b}_log10_ecorr"] = -3.0  # huge
    bn = BlockNoise(psr, noise)
    N = bn.dense()
    X = rng.normal(size=(psr.ntoa, 2))
    got = bn.solve(X)
    want = np.linalg.solve(N, X)
    np.testing.assert_allclose(got, want, rtol=1e-7, atol=1e-2)


def test_get_xcy_blocknoise_vs_dense_oracle():
    psrs, noise = _psr_and_noise(seed=2)
    psr = psrs[0]
    bn = BlockNoise(psr, noise)
    rng = np.random.default_rng(3)
    m = 7
    T = rng.normal(size=(psr.ntoa, m))
    # phi at the noise scale so the DENSE oracle stays well-conditioned
    phi = rng.uniform(0.1, 10.0, m) * 1e-12
    x = rng.normal(size=psr.ntoa)
    y = rng.normal(size=psr.ntoa)
    Tp = T[bn.perm]
    sigma = Tp.T @ bn.solve(Tp) + np.diag(1.0 / phi)
    got = get_xCy(bn, T, sigma, x, y)
    # dense C in the PERMUTED frame
    C = bn.dense() + Tp @ np.diag(phi) @ Tp.T
    want = float(x[bn.perm] @ np.linalg.solve(C, y[bn.perm]))
    assert got == pytest.approx(want, rel=1e-8)


def test_kernel_ecorr_equals_gp_ecorr():
    """ECORR as block-diagonal N is mathematically identical to ECORR as
    a quantization-basis GP (same per-epoch fully-correlated component)
    -> the two model paths must produce the same Fp spectrum."""
    psrs, noise = _psr_and_noise(seed=4, npsr=2, ntoa=70)

    pta_gp = initialize_pta(
        psrs, noise, inc_cp=False, rn_comps=3, simple_wn=True, inc_ecorr=True
    )
    pta_bk = initialize_pta(
        psrs, noise, inc_cp=False, rn_comps=3, simple_wn=True,
        ecorr_kernel=True,
    )
    freqs = np.linspace(4e-9, 5e-8, 6)
    fps = []
    for pta in (pta_gp, pta_bk):
        Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
        fp = FastFp(psrs).sweep(freqs, Nvecs, Ts, sigmas, device="cpu")
        fps.append(fp)
    np.testing.assert_allclose(fps[0], fps[1], rtol=1e-6)


def test_engine_blocknoise_matches_parity_path():
    psrs, noise = _psr_and_noise(seed=5, npsr=2, ntoa=60)
    pta = initialize_pta(
        psrs, noise, inc_cp=False, rn_comps=3, ecorr_kernel=True
    )
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    freqs = np.linspace(4e-9, 5e-8, 5)
    fp_obj = FastFp(psrs)
    want = np.array([fp_obj.calculate_Fp(f, Nvecs, Ts, sigmas) for f in freqs])
    got = fp_obj.sweep(freqs, Nvecs, Ts, sigmas, device="cpu")
    np.testing.assert_allclose(got, want, rtol=1e-7)
