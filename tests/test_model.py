"""PTAModel surface tests (the enterprise-compatible API)."""

import numpy as np
import pytest

from fastfp_amd import (
    get_mats_fp,
    get_mats_nmfp,
    initialize_pta,
    make_synthetic_pta,
)
from fastfp_amd.data import get_tspan, load_pulsars, save_pulsars


@pytest.fixture(scope="module")
def setup():
    psrs = make_synthetic_pta(npsr=3, ntoa=80, ntm=3, seed=0)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": np.log10(2e-15)}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=4, gwb_comps=3)
    return psrs, noise, pta


def test_params_sorted(setup):
    psrs, noise, pta = setup
    assert pta.params == sorted(pta.params)
    assert len(pta.params) == 2 * len(psrs) + 2
    assert "gw_gamma" in pta.params and "gw_log10_A" in pta.params


def test_map_params(setup):
    _, _, pta = setup
    xs = np.arange(len(pta.params), dtype=float)
    d = pta.map_params(xs)
    for i, name in enumerate(pta.params):
        assert d[name] == i


def test_get_mats_shapes(setup):
    psrs, noise, pta = setup
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    TNTs, Nvecs2, Ts2 = get_mats_nmfp(pta, noise)
    for p, nv, T, sg, TNT in zip(psrs, Nvecs, Ts, sigmas, TNTs):
        m = p.ntm + 8  # tm + 2*rn_comps
        assert T.shape == (p.ntoa, m)
        assert sg.shape == (m, m)
        assert TNT.shape == (m, m)
        np.testing.assert_allclose(TNT, TNT.T, atol=1e-9 * np.abs(TNT).max())
        # sigma = TNT + diag(phiinv), so sigma - TNT is diagonal
        d = sg - TNT
        np.testing.assert_allclose(d, np.diag(np.diag(d)), atol=1e-30)
        # SPD check
        np.linalg.cholesky(sg)


def test_basis_block_order(setup):
    psrs, noise, pta = setup
    Ts = pta.get_basis(noise)
    for p, T, sl_tm, sl_rn in zip(psrs, Ts, pta.tm_slices, pta.rn_slices):
        U = T[:, sl_tm]
        np.testing.assert_allclose(U.T @ U, np.eye(p.ntm), atol=1e-10)
        assert sl_rn.stop == T.shape[1]


def test_pulsar_npz_roundtrip(tmp_path, setup):
    psrs, _, _ = setup
    path = str(tmp_path / "psrs.npz")
    save_pulsars(psrs, path)
    back = load_pulsars(path)
    assert len(back) == len(psrs)
    for a, b in zip(psrs, back):
        assert a.name == b.name
        np.testing.assert_allclose(a.toas, b.toas)
        np.testing.assert_allclose(a.residuals, b.residuals)
        np.testing.assert_allclose(a.Mmat, b.Mmat)
        assert list(a.backend_flags) == list(b.backend_flags)
    assert get_tspan(back) == pytest.approx(get_tspan(psrs))


def test_ecorr_model_builds():
    psrs = make_synthetic_pta(npsr=2, ntoa=60, ntm=3, seed=1)
    noise = {}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
        for b in np.unique(p.backend_flags):
            noise[f"{p.name}_basis_ecorr_{b}_log10_ecorr"] = -7.0
            noise[f"{p.name}_{b}_efac"] = 1.1
            noise[f"{p.name}_{b}_log10_t2equad"] = -7.0
    pta = initialize_pta(
        psrs, noise, inc_cp=False, rn_comps=3, simple_wn=False, inc_ecorr=True
    )
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    for p, T, sl_ec in zip(psrs, Ts, pta.ecorr_slices):
        n_ec = sl_ec.stop - sl_ec.start
        assert n_ec > 0
        assert T.shape[1] == p.ntm + n_ec + 6
    for sg in sigmas:
        np.linalg.cholesky(sg)


def test_per_psr_tspan_bases():
    """The reference's setup_fp_model(Tspan=None) mode: red-noise bases
    on each pulsar's own Tspan (run_nmfp.py:94-98)."""
    from fastfp_amd.bases import create_freqarray

    psrs = make_synthetic_pta(npsr=3, ntoa=70, ntm=3, seed=7)
    noise = {}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=False, rn_comps=4,
                         per_psr_tspan=True)
    for p, cont in zip(psrs, pta.rn_containers):
        np.testing.assert_allclose(
            cont.Ffreqs.numpy(), create_freqarray(p.Tspan, 4)
        )
    # engine still runs end to end
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    from fastfp_amd import FastFp

    fp = FastFp(psrs).sweep(np.linspace(4e-9, 4e-8, 3), Nvecs, Ts,
                            sigmas, device="cpu")
    assert np.isfinite(fp).all()
    # the shared-basis CURN combination is rejected, not silently wrong
    with pytest.raises(ValueError):
        initialize_pta(psrs, noise, inc_cp=True, per_psr_tspan=True)


def test_from_object_duck_typed_pickle_surface(tmp_path):
    """The enterprise-pickle converter: with and without backend_flags
    / toaerrs (missing flags must fall back to the per-TOA default,
    not a 0-d None array)."""
    import pickle
    import types

    from fastfp_amd.data import PulsarData

    src = make_synthetic_pta(npsr=1, ntoa=40, ntm=3, seed=7)[0]
    full = types.SimpleNamespace(
        name=src.name, toas=src.toas, toaerrs=src.toaerrs,
        residuals=src.residuals, Mmat=src.Mmat,
        backend_flags=src.backend_flags,
    )
    minimal = types.SimpleNamespace(  # no backend_flags, no toaerrs
        name=src.name, toas=src.toas, residuals=src.residuals,
        Mmat=src.Mmat,
    )
    pkl = tmp_path / "psrs.pkl"
    with open(pkl, "wb") as f:
        pickle.dump([full, minimal], f)
    a, b = load_pulsars(str(pkl))
    assert isinstance(a, PulsarData) and isinstance(b, PulsarData)
    np.testing.assert_array_equal(a.backend_flags, src.backend_flags)
    # missing flags -> one default backend per TOA (1-d, right length)
    assert b.backend_flags.shape == (b.ntoa,)
    assert set(np.unique(b.backend_flags)) == {"backend"}
    np.testing.assert_array_equal(b.toaerrs, np.full(b.ntoa, 1e-6))
    # the converted pulsar must drive the full model build
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": np.log10(2e-15),
             f"{b.name}_red_noise_gamma": 4.0,
             f"{b.name}_red_noise_log10_A": -14.5}
    pta = initialize_pta([b], noise, inc_cp=True, rn_comps=3, gwb_comps=2)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    assert Ts[0].shape[0] == b.ntoa
