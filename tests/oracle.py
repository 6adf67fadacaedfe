"""Dense NumPy oracle for the Fp statistic.

Computes everything the slow, obviously-correct way: the full dense
covariance ``C = diag(N) + T diag(phi) T^T`` per pulsar, solved directly
— no Woodbury identity, no restructuring.  Used by the unit tests to
validate both the Woodbury parity path (`get_xCy`) and the restructured
engine (SURVEY.md §4's consequence (a)/(b)).

Tiny systems only (O(ntoa^3)).
"""

import numpy as np


def dense_xCy(Nvec, T, phi, x, y):
    """x^T C^-1 y via a dense solve of C = diag(N) + T diag(phi) T^T."""
    C = np.diag(np.asarray(Nvec, dtype=np.float64))
    T = np.asarray(T, dtype=np.float64)
    C = C + T @ np.diag(np.asarray(phi, dtype=np.float64)) @ T.T
    return float(np.asarray(x) @ np.linalg.solve(C, np.asarray(y)))


def dense_fp_single(psrs, Nvecs, Ts, phis, fgw):
    """Fp at one frequency via dense C^-1 solves (amplitude included,
    matching the reference's filter definition)."""
    fstat = 0.0
    amp = 1.0 / fgw ** (1.0 / 3.0)
    for psr, Nvec, T, phi in zip(psrs, Nvecs, Ts, phis):
        toa = psr.toas
        resid = psr.residuals
        s = amp * np.sin(2 * np.pi * fgw * toa)
        c = amp * np.cos(2 * np.pi * fgw * toa)
        C = np.diag(Nvec) + T @ np.diag(phi) @ T.T
        Ci = np.linalg.inv(C)
        N = np.array([s @ Ci @ resid, c @ Ci @ resid])
        M = np.array(
            [[s @ Ci @ s, s @ Ci @ c], [c @ Ci @ s, c @ Ci @ c]]
        )
        fstat += 0.5 * float(N @ np.linalg.solve(M, N))
    return fstat


def dense_fp_sweep(psrs, Nvecs, Ts, phis, freqs):
    return np.array(
        [dense_fp_single(psrs, Nvecs, Ts, phis, f) for f in freqs]
    )
