"""phi(theta) container layer vs hand-computed reference formulas
(conventions of /root/reference/fastfp/nmfp.py:217-292)."""

import numpy as np
import pytest

from fastfp_amd.bases import create_freqarray
from fastfp_amd.constants import fyr
from fastfp_amd.data import make_synthetic_pta
from fastfp_amd.noise import (
    CURNContainer,
    GPEcorrContainer,
    RNContainer,
    TM_PRIOR,
    powerlaw_psd,
    white_noise_nvec,
)


def _ref_powerlaw(Ffreqs, log10_A, gamma):
    """The reference's formula, transcribed
    (/root/reference/fastfp/nmfp.py:226-234)."""
    df = np.diff(np.concatenate(([0.0], Ffreqs[::2])))
    return (
        Ffreqs ** (-gamma)
        * (10.0**log10_A) ** 2
        / 12.0
        / np.pi**2
        * fyr ** (gamma - 3.0)
        * np.repeat(df, 2)
    )


@pytest.fixture
def psr():
    return make_synthetic_pta(npsr=1, ntoa=120, ntm=3, seed=0)[0]


def test_powerlaw_matches_reference_formula(psr):
    Ff = create_freqarray(psr.Tspan, 5)
    pars = {
        f"{psr.name}_red_noise_log10_A": -14.2,
        f"{psr.name}_red_noise_gamma": 3.7,
    }
    cont = RNContainer(psr, Ffreqs=Ff, ncomps=5, inc_tm=False)
    got = cont.get_phi_rn(pars).numpy()
    want = _ref_powerlaw(Ff, -14.2, 3.7)
    np.testing.assert_allclose(got, want, rtol=1e-12)
    np.testing.assert_allclose(powerlaw_psd(Ff, -14.2, 3.7), want, rtol=1e-12)


def test_phi_tm_rn_block_order(psr):
    Ff = create_freqarray(psr.Tspan, 4)
    pars = {
        f"{psr.name}_red_noise_log10_A": -14.0,
        f"{psr.name}_red_noise_gamma": 4.0,
    }
    cont = RNContainer(psr, Ffreqs=Ff, ncomps=4)
    phi = cont.update_phi(pars).numpy()
    assert phi.shape[0] == psr.ntm + 8
    np.testing.assert_allclose(phi[: psr.ntm], TM_PRIOR)
    np.testing.assert_allclose(phi[psr.ntm :], _ref_powerlaw(Ff, -14.0, 4.0))
    np.testing.assert_allclose(
        cont.get_phiinv(pars).numpy(), 1.0 / phi, rtol=1e-14
    )


def test_phi_curn_added_on_first_bins(psr):
    """CURN phi ADDS onto the first 2*ngwb rn bins
    (/root/reference/fastfp/nmfp.py:247)."""
    Ff = create_freqarray(psr.Tspan, 5)
    Ffg = Ff[:6]  # 3 gwb comps on the same Tspan grid
    curn = CURNContainer(Ffg)
    pars = {
        f"{psr.name}_red_noise_log10_A": -14.0,
        f"{psr.name}_red_noise_gamma": 4.0,
        "gw_log10_A": -14.5,
        "gw_gamma": 13.0 / 3.0,
    }
    cont = RNContainer(psr, Ffreqs=Ff, ncomps=5, add_curn=True, curn_container=curn)
    phi = cont.update_phi(pars).numpy()
    rn = _ref_powerlaw(Ff, -14.0, 4.0)
    gw = _ref_powerlaw(Ffg, -14.5, 13.0 / 3.0)
    want = rn.copy()
    want[:6] += gw
    np.testing.assert_allclose(phi[psr.ntm :], want, rtol=1e-12)


def test_phi_draw_vectorized_matches_scalar(psr):
    Ff = create_freqarray(psr.Tspan, 4)
    curn = CURNContainer(Ff[:4])
    cont = RNContainer(psr, Ffreqs=Ff, ncomps=4, add_curn=True, curn_container=curn)
    D = 5
    rng = np.random.default_rng(1)
    pars = {
        f"{psr.name}_red_noise_log10_A": rng.uniform(-16, -13, D),
        f"{psr.name}_red_noise_gamma": rng.uniform(2, 6, D),
        "gw_log10_A": rng.uniform(-16, -13, D),
        "gw_gamma": rng.uniform(2, 6, D),
    }
    batch = cont.get_phiinv(pars).numpy()
    assert batch.shape == (D, psr.ntm + 8)
    for d in range(D):
        scalar = cont.get_phiinv({k: float(v[d]) for k, v in pars.items()}).numpy()
        np.testing.assert_allclose(batch[d], scalar, rtol=1e-13)


def test_gp_ecorr_fixed_phi(psr):
    noise = {}
    for b in np.unique(psr.backend_flags):
        noise[f"{psr.name}_basis_ecorr_{b}_log10_ecorr"] = -6.5
    ec = GPEcorrContainer(psr, fix_wn_vals=noise)
    phi = ec.get_phi().numpy()
    assert phi.ndim == 1 and phi.shape[0] > 0
    np.testing.assert_allclose(phi, 10.0 ** (2 * -6.5))

    Ff = create_freqarray(psr.Tspan, 3)
    cont = RNContainer(
        psr, Ffreqs=Ff, ncomps=3, gp_ecorr=True, ecorr_container=ec
    )
    pars = {
        f"{psr.name}_red_noise_log10_A": -14.0,
        f"{psr.name}_red_noise_gamma": 4.0,
    }
    full = cont.update_phi(pars).numpy()
    nec = phi.shape[0]
    assert full.shape[0] == psr.ntm + nec + 6
    np.testing.assert_allclose(full[psr.ntm : psr.ntm + nec], phi)


def test_white_noise_nvec_backends(psr):
    simple = white_noise_nvec(psr, simple_wn=True)
    np.testing.assert_allclose(simple, psr.toaerrs**2)

    noise = {}
    for b in np.unique(psr.backend_flags):
        noise[f"{psr.name}_{b}_efac"] = 1.5
        noise[f"{psr.name}_{b}_log10_t2equad"] = -6.0
    full = white_noise_nvec(psr, noise, simple_wn=False)
    want = 1.5**2 * (psr.toaerrs**2 + 1e-12)
    np.testing.assert_allclose(full, want, rtol=1e-12)


def test_white_noise_nvec_tnequad_and_legacy(psr):
    """TempoNest (tnequad) and legacy (log10_equad) conventions on a
    real-shaped NANOGrav noise dict — enterprise's white_noise_block
    admits all three (/root/reference/fastfp/utils.py:151-155).  The
    legacy key uses tnequad semantics (old EquadNoise: quadrature AFTER
    the EFAC scaling)."""
    backends = np.unique(psr.backend_flags)
    # tnequad convention: N = (efac*sig)^2 + tnequad^2
    noise = {}
    for b in backends:
        noise[f"{psr.name}_{b}_efac"] = 1.3
        noise[f"{psr.name}_{b}_log10_tnequad"] = -6.2
    nv = white_noise_nvec(psr, noise, simple_wn=False)
    want = (1.3 * psr.toaerrs) ** 2 + 10.0 ** (2 * -6.2)
    np.testing.assert_allclose(nv, want, rtol=1e-12)

    # legacy log10_equad keys: same tnequad semantics
    noise = {}
    for b in backends:
        noise[f"{psr.name}_{b}_efac"] = 1.1
        noise[f"{psr.name}_{b}_log10_equad"] = -6.8
    nv = white_noise_nvec(psr, noise, simple_wn=False)
    want = (1.1 * psr.toaerrs) ** 2 + 10.0 ** (2 * -6.8)
    np.testing.assert_allclose(nv, want, rtol=1e-12)


def test_white_noise_nvec_key_precedence(psr):
    """Mixed dicts: t2equad wins over tnequad wins over legacy equad;
    conventions may differ per backend group."""
    backends = np.unique(psr.backend_flags)
    assert len(backends) >= 2
    b0, b1 = backends[0], backends[1]
    noise = {
        # b0 carries BOTH t2equad and legacy equad -> t2equad wins
        f"{psr.name}_{b0}_efac": 1.4,
        f"{psr.name}_{b0}_log10_t2equad": -6.0,
        f"{psr.name}_{b0}_log10_equad": -5.0,
        # b1 carries tnequad and legacy equad -> tnequad wins
        f"{psr.name}_{b1}_efac": 0.9,
        f"{psr.name}_{b1}_log10_tnequad": -6.4,
        f"{psr.name}_{b1}_log10_equad": -5.5,
    }
    nv = white_noise_nvec(psr, noise, simple_wn=False)
    m0 = psr.backend_flags == b0
    m1 = psr.backend_flags == b1
    np.testing.assert_allclose(
        nv[m0], 1.4**2 * (psr.toaerrs[m0] ** 2 + 1e-12), rtol=1e-12
    )
    np.testing.assert_allclose(
        nv[m1], (0.9 * psr.toaerrs[m1]) ** 2 + 10.0 ** (2 * -6.4), rtol=1e-12
    )
    # untouched backends stay at efac=1, equad=0
    rest = ~(m0 | m1)
    if rest.any():
        np.testing.assert_allclose(nv[rest], psr.toaerrs[rest] ** 2, rtol=1e-12)


def test_white_noise_no_selection(psr):
    """select != backend: one parameter set per pulsar with the
    un-selected enterprise naming ({psr}_efac)."""
    noise = {f"{psr.name}_efac": 1.2, f"{psr.name}_log10_t2equad": -6.5}
    nv = white_noise_nvec(psr, noise, simple_wn=False, select="none")
    want = 1.2**2 * (psr.toaerrs**2 + 10.0 ** (2 * -6.5))
    np.testing.assert_allclose(nv, want, rtol=1e-12)


def test_phi_fn_bound_variant(psr):
    """Reference parity: RN_container binds the matching phi-variant
    method at init (/root/reference/fastfp/nmfp.py:185-199) — users may
    call ``cont.phi_fn(pars)`` directly."""
    from fastfp_amd.noise import CURNContainer, RNContainer
    from fastfp_amd.bases import create_freqarray

    curn = CURNContainer(create_freqarray(psr.Tspan, 2))
    pars = {
        f"{psr.name}_red_noise_gamma": 4.0,
        f"{psr.name}_red_noise_log10_A": -14.5,
        "gw_gamma": 13.0 / 3.0,
        "gw_log10_A": -14.8,
    }
    plain = RNContainer(psr, ncomps=3)
    assert plain.phi_fn == plain.get_phi_tm_rn
    np.testing.assert_array_equal(
        plain.phi_fn(pars).numpy(), plain.update_phi(pars).numpy()
    )
    withc = RNContainer(psr, ncomps=3, add_curn=True, curn_container=curn)
    assert withc.phi_fn == withc.get_phi_tm_rn_curn
    np.testing.assert_array_equal(
        withc.phi_fn(pars).numpy(), withc.update_phi(pars).numpy()
    )
