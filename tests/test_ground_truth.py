"""Arbitrary-precision ground truth for the Fp statistic.

Every other numerics test compares two fp64 computations (engine vs
parity vs dense-fp64 oracle) — all of which share fp64's round-off
floor.  This test computes Fp through a 40-digit mpmath dense
``C = diag(N) + T diag(phi) T^T`` evaluation and asserts the engine
agrees to ~1e-9 relative at a well-conditioned tiny model, pinning the
ABSOLUTE accuracy, not just path agreement.  (At cancellation-prone
corners — f*Tspan near an integer with extreme phi^-1 dynamic range —
fp64 itself floors at ~1e-6; see docs/STATUS.md round-2 closing.)
"""

import numpy as np
import pytest

mp = pytest.importorskip("mpmath")

from fastfp_amd import FastFp, FpEngine, make_synthetic_pta
from fastfp_amd.bases import (
    create_freqarray,
    fourier_basis,
    timing_model_basis_svd,
)
from fastfp_amd.noise import white_noise_nvec


def _mp_fp(psr, Nvec, T, phi, freqs, dps=40):
    """Fp(f) per pulsar through dps-digit dense C^{-1} solves."""
    mp.mp.dps = dps
    n = len(psr.toas)
    Tm = np.asarray(T)
    C = mp.matrix(n, n)
    for a in range(n):
        for b in range(n):
            acc = mp.mpf(float(Nvec[a])) if a == b else mp.mpf(0)
            for k in range(Tm.shape[1]):
                acc += mp.mpf(Tm[a, k]) * mp.mpf(phi[k]) * mp.mpf(Tm[b, k])
            C[a, b] = acc
    out = []
    for f in freqs:
        amp = mp.mpf(1.0) / mp.mpf(float(f)) ** (mp.mpf(1) / 3)
        s = [amp * mp.sin(2 * mp.pi * mp.mpf(float(f)) * mp.mpf(t))
             for t in psr.toas]
        c = [amp * mp.cos(2 * mp.pi * mp.mpf(float(f)) * mp.mpf(t))
             for t in psr.toas]
        r = [mp.mpf(x) for x in psr.residuals]
        Ci_s = mp.lu_solve(C, mp.matrix(s))
        Ci_c = mp.lu_solve(C, mp.matrix(c))
        Ci_r = mp.lu_solve(C, mp.matrix(r))
        dot = lambda u, v: mp.fsum(ui * vi for ui, vi in zip(u, v))  # noqa: E731
        N1, N2 = dot(s, Ci_r), dot(c, Ci_r)
        M11, M12, M22 = dot(s, Ci_s), dot(s, Ci_c), dot(c, Ci_c)
        det = M11 * M22 - M12 * M12
        out.append(
            float((N1 * N1 * M22 - 2 * N1 * N2 * M12 + N2 * N2 * M11)
                  / det / 2)
        )
    return np.asarray(out)


def test_engine_matches_arbitrary_precision_reference():
    psr = make_synthetic_pta(npsr=1, ntoa=24, ntm=3, seed=5,
                             ragged=False)[0]
    Nvec = white_noise_nvec(psr)
    U = timing_model_basis_svd(psr.Mmat)
    Fb = fourier_basis(psr.toas, create_freqarray(psr.Tspan, 3))
    T = np.concatenate([U, Fb], axis=1)
    rng = np.random.default_rng(3)
    # moderate prior scales: tm 1e5 x residual^2, rn O(residual^2)
    phi = np.concatenate([
        np.full(U.shape[1], 1e5) * 1e-12,
        rng.uniform(0.3, 3.0, Fb.shape[1]) * 1e-12,
    ])
    # frequencies away from the k/Tspan cancellation corners
    freqs = np.array([4.6e-9, 1.13e-8, 2.71e-8])

    want = _mp_fp(psr, Nvec, T, phi, freqs)

    eng = FpEngine([psr], [Nvec], [T], device="cpu")
    eng.precompute(freqs)
    got = eng.sweep(phiinvs=[1.0 / phi]).numpy()
    np.testing.assert_allclose(got, want, rtol=2e-9)

    # the get_xCy parity path hits the same truth
    sigma = T.T @ (T / Nvec[:, None]) + np.diag(1.0 / phi)
    fp_obj = FastFp([psr])
    par = np.array([
        fp_obj.calculate_Fp(f, [Nvec], [T], [sigma]) for f in freqs
    ])
    np.testing.assert_allclose(par, want, rtol=2e-9)


@pytest.mark.gpu
def test_hip_kernels_match_arbitrary_precision_reference():
    """The full HIP path (sigdots + sbgemm precompute, chol_batch +
    trsm_fp solve) against the 40-digit oracle: pins the KERNELS'
    absolute accuracy, not just their agreement with CPU fp64."""
    import torch

    if not torch.cuda.is_available():
        pytest.skip("needs a GPU")
    psr = make_synthetic_pta(npsr=1, ntoa=24, ntm=3, seed=5,
                             ragged=False)[0]
    Nvec = white_noise_nvec(psr)
    U = timing_model_basis_svd(psr.Mmat)
    Fb = fourier_basis(psr.toas, create_freqarray(psr.Tspan, 3))
    T = np.concatenate([U, Fb], axis=1)
    rng = np.random.default_rng(3)
    phi = np.concatenate([
        np.full(U.shape[1], 1e5) * 1e-12,
        rng.uniform(0.3, 3.0, Fb.shape[1]) * 1e-12,
    ])
    freqs = np.array([4.6e-9, 1.13e-8, 2.71e-8])

    want = _mp_fp(psr, Nvec, T, phi, freqs)

    eng = FpEngine([psr], [Nvec], [T], device="cuda:0")
    eng.precompute(freqs)
    got = eng.sweep(phiinvs=[1.0 / phi]).cpu().numpy()
    np.testing.assert_allclose(got, want, rtol=2e-9)
