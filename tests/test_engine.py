"""Engine correctness: the restructured sweep vs the dense oracle and vs
the reference-shaped parity path."""

import torch
import numpy as np
import pytest

from fastfp_amd import (
    FastFp,
    FpEngine,
    NMFp,
    get_mats_fp,
    get_mats_nmfp,
    initialize_pta,
    make_synthetic_pta,
)
from fastfp_amd.bases import create_freqarray, fourier_basis, timing_model_basis_svd
from fastfp_amd.noise import white_noise_nvec
from oracle import dense_fp_sweep


def _tiny_setup(seed=0, npsr=3, ntoa=70, ncomps=4, ntm=3, tm_prior=1e5):
    """Small PTA with MODERATE priors so the dense oracle is well
    conditioned."""
    psrs = make_synthetic_pta(
        npsr=npsr, ntoa=ntoa, tspan_yr=10.0, ntm=ntm, seed=seed, toaerr=1e-6
    )
    rng = np.random.default_rng(seed + 100)
    Nvecs, Ts, phis = [], [], []
    for p in psrs:
        Nvecs.append(white_noise_nvec(p))
        U = timing_model_basis_svd(p.Mmat)
        F = fourier_basis(p.toas, create_freqarray(p.Tspan, ncomps))
        T = np.concatenate([U, F], axis=1)
        Ts.append(T)
        # phi: tm block (moderate), rn block random positive
        phi = np.concatenate(
            [
                np.full(U.shape[1], tm_prior) * 1e-12,  # tm, scaled to residual^2
                rng.uniform(0.5, 2.0, F.shape[1]) * 1e-12,
            ]
        )
        phis.append(phi)
    return psrs, Nvecs, Ts, phis


def test_engine_fp_matches_dense_oracle():
    psrs, Nvecs, Ts, phis = _tiny_setup()
    freqs = np.linspace(3e-9, 5e-8, 7)
    want = dense_fp_sweep(psrs, Nvecs, Ts, phis, freqs)

    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(freqs, freq_chunk=3)
    phiinvs = [1.0 / p for p in phis]
    got = eng.sweep(phiinvs=phiinvs).numpy()
    np.testing.assert_allclose(got, want, rtol=1e-7)


def test_engine_sigma_and_phiinv_paths_agree():
    psrs, Nvecs, Ts, phis = _tiny_setup(seed=1)
    freqs = np.linspace(3e-9, 5e-8, 5)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(freqs)
    phiinvs = [1.0 / p for p in phis]
    sigmas = [
        T.T @ (T / nv[:, None]) + np.diag(pi)
        for T, nv, pi in zip(Ts, Nvecs, phiinvs)
    ]
    a = eng.sweep(phiinvs=phiinvs).numpy()
    b = eng.sweep(sigmas=sigmas).numpy()
    np.testing.assert_allclose(a, b, rtol=1e-10)


def test_parity_calculate_fp_matches_engine():
    """The reference-shaped get_xCy path (with the f^-1/3 amplitude)
    equals the engine sweep (amplitude dropped) — proves the amplitude
    cancellation."""
    psrs, Nvecs, Ts, phis = _tiny_setup(seed=2)
    freqs = np.linspace(4e-9, 4e-8, 4)
    phiinvs = [1.0 / p for p in phis]
    sigmas = [
        T.T @ (T / nv[:, None]) + np.diag(pi)
        for T, nv, pi in zip(Ts, Nvecs, phiinvs)
    ]
    fp_obj = FastFp(psrs)
    want = np.array(
        [fp_obj.calculate_Fp(f, Nvecs, Ts, sigmas) for f in freqs]
    )
    got = fp_obj.sweep(freqs, Nvecs, Ts, sigmas, device="cpu")
    np.testing.assert_allclose(got, want, rtol=1e-7)


def test_fp_invariant_to_tm_prior_scale():
    """1e40 improper-flat tm prior acts as a projection: Fp must be
    insensitive to the exact (huge) scale."""
    psrs, Nvecs, Ts, phis = _tiny_setup(seed=3)
    freqs = np.linspace(4e-9, 4e-8, 3)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(freqs)
    ntm = 3
    out = []
    for scale in (1e30, 1e40):
        phiinvs = []
        for phi in phis:
            p = phi.copy()
            p[:ntm] = scale
            phiinvs.append(1.0 / p)
        out.append(eng.sweep(phiinvs=phiinvs).numpy())
    np.testing.assert_allclose(out[0], out[1], rtol=1e-5)


def test_nmfp_sweep_matches_per_draw_parity():
    """Draw-vectorized NM-Fp sweep == per-draw calculate_nmfp (which goes
    through get_xCy), on a full model with tm(1e40)+rn+curn."""
    psrs = make_synthetic_pta(npsr=2, ntoa=60, tspan_yr=10.0, ntm=3, seed=4)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": np.log10(2e-15)}
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=4, gwb_comps=3)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)

    D = 3
    rng = np.random.default_rng(5)
    samples = {}
    for name in pta.params:
        if name.endswith("gamma"):
            samples[name] = rng.uniform(2.0, 6.0, D)
        else:
            samples[name] = rng.uniform(-15.0, -13.0, D)

    nm = NMFp(psrs, pta.rn_containers)
    freqs = np.linspace(4e-9, 4e-8, 4)
    got = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu", draw_chunk=2)
    assert got.shape == (D, len(freqs))

    for d in range(D):
        pars = {k: float(v[d]) for k, v in samples.items()}
        for j, f in enumerate(freqs):
            want = nm.calculate_nmfp(f, pars, Nvecs, Ts, TNTs)
            assert got[d, j] == pytest.approx(want, rel=1e-6)


def test_chi2_null_distribution():
    """Statistical integration test: for noise-only
    data 2*Fp ~ chi^2(2 * npsr); check the sample mean loosely."""
    npsr = 6
    psrs = make_synthetic_pta(npsr=npsr, ntoa=150, tspan_yr=12.0, ntm=3, seed=6)
    noise = {}
    pta = initialize_pta(psrs, noise, inc_cp=False, rn_comps=5)
    Nvecs, Ts, sigmas = get_mats_fp(
        pta, {f"{p.name}_red_noise_gamma": 4.0 for p in psrs}
        | {f"{p.name}_red_noise_log10_A": -16.0 for p in psrs}
    )
    freqs = np.linspace(5e-9, 8e-8, 40)
    fp = FastFp(psrs).sweep(freqs, Nvecs, Ts, sigmas, device="cpu")
    mean2fp = float(np.mean(2.0 * fp))
    expect = 2.0 * npsr
    assert abs(mean2fp - expect) < 0.45 * expect, (mean2fp, expect)


def test_draw_compression_matches_direct():
    """The Schur-compressed sweep (per-draw solve on the variable bins
    only, docs/DESIGN.md) must match the direct full-m sweep."""
    psrs = make_synthetic_pta(npsr=3, ntoa=90, ntm=4, seed=9)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=5, gwb_comps=4)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    D = 6
    rng = np.random.default_rng(10)
    samples = {
        n: (rng.uniform(2, 6, D) if n.endswith("gamma") else rng.uniform(-16, -14, D))
        for n in pta.params
    }
    from fastfp_amd.nmfp import NMFp

    nm = NMFp(psrs, pta.rn_containers)
    freqs = np.linspace(4e-9, 5e-8, 8)
    direct = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu", compress=False)
    comp = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu", compress=True)
    np.testing.assert_allclose(comp, direct, rtol=1e-7)


@pytest.mark.parametrize("seed,inc_cp,inc_ecorr,simple_wn,rn", [
    (11, True, False, True, 3),
    (12, False, True, True, 4),
    (13, True, True, False, 5),
    (14, False, False, False, 6),
])
def test_engine_fuzz_configs(seed, inc_cp, inc_ecorr, simple_wn, rn):
    """Random model configurations: engine sweep == get_xCy parity path."""
    psrs = make_synthetic_pta(npsr=2, ntoa=80, ntm=3, seed=seed)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    rng = np.random.default_rng(seed)
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = rng.uniform(2, 6)
        noise[f"{p.name}_red_noise_log10_A"] = rng.uniform(-16, -14)
        for b in np.unique(p.backend_flags):
            noise[f"{p.name}_basis_ecorr_{b}_log10_ecorr"] = -6.5
            noise[f"{p.name}_{b}_efac"] = rng.uniform(0.9, 1.3)
            noise[f"{p.name}_{b}_log10_t2equad"] = -6.8
    pta = initialize_pta(psrs, noise, inc_cp=inc_cp, rn_comps=rn,
                         gwb_comps=min(rn, 3), simple_wn=simple_wn,
                         inc_ecorr=inc_ecorr)
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    freqs = np.linspace(4e-9, 5e-8, 4)
    fp_obj = FastFp(psrs)
    want = np.array([fp_obj.calculate_Fp(f, Nvecs, Ts, sigmas) for f in freqs])
    got = fp_obj.sweep(freqs, Nvecs, Ts, sigmas, device="cpu")
    np.testing.assert_allclose(got, want, rtol=1e-6)


def test_fp_detects_injected_cw_signal():
    """Physics validation: a continuous wave injected at f_inj must
    produce a strong Fp peak AT f_inj (detection works end to end)."""
    f_inj = 2.2e-8
    psrs = make_synthetic_pta(
        npsr=8, ntoa=300, tspan_yr=12.0, ntm=3, seed=77,
        toaerr=1e-7, cw_amp=5e-7, cw_freq=f_inj,
    )
    noise = {}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -16.0
    pta = initialize_pta(psrs, noise, inc_cp=False, rn_comps=5)
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    freqs = np.linspace(5e-9, 6e-8, 56)
    fp = FastFp(psrs).sweep(freqs, Nvecs, Ts, sigmas, device="cpu")
    peak_f = freqs[int(np.argmax(fp))]
    # peak within one grid step of the injection, and strongly above
    # the chi^2(16) background (mean 8)
    assert abs(peak_f - f_inj) <= (freqs[1] - freqs[0]), (peak_f, f_inj)
    assert fp.max() > 50.0, fp.max()


def test_compression_low_margin_draws_stay_compressed_and_exact():
    """Draws whose margin is small-but-safe (>1.5; e.g. uniform-prior
    tails with large A and gamma) must STAY on the compressed path and
    agree with the exact direct sweep — the r01 threshold of 1e3
    silently pushed whole uniform-prior sweeps onto the direct path."""
    psrs = make_synthetic_pta(npsr=2, ntoa=400, ntm=8, seed=16)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 13.0 / 3.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=15, gwb_comps=15)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    D = 4
    samples = {
        n: (np.array([6.5, 6.0, 5.0, 13 / 3]) if n.endswith("gamma")
            else np.array([-13.5, -13.8, -14.0, -14.5]))
        for n in pta.params
    }
    nm = NMFp(psrs, pta.rn_containers)
    freqs = np.linspace(3e-9, 5e-8, 12)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(freqs)
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    assert all(blk.comp is not None for blk in eng.blocks)
    from fastfp_amd.noise import batch_phiinv

    piv = [p if p.dim() == 2 else p[None]
           for p in batch_phiinv(pta.rn_containers, samples)]
    margin = eng.compression_margin(piv)
    assert 1.5 < margin < 1e3, f"test shape should sit in (1.5, 1e3): {margin}"
    got = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu", engine=eng)
    # engine still compressed after the sweep (the old threshold
    # disabled it here)
    assert all(blk.comp is not None for blk in eng.blocks)
    direct = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu",
                      compress=False)
    np.testing.assert_allclose(got, direct, rtol=1e-7)


def test_per_draw_hybrid_split_matches_direct():
    """A batch mixing normal draws with ONE prior-corner draw (margin
    below the guard): the corner draw runs the exact direct path, the
    rest stay compressed, and the combined output equals the all-direct
    reference."""
    psrs = make_synthetic_pta(npsr=2, ntoa=400, ntm=8, seed=17)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 13.0 / 3.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=10, gwb_comps=10)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    samples = {
        n: (np.array([4.0, 9.0, 13 / 3]) if n.endswith("gamma")
            else np.array([-14.5, -12.5, -15.0]))
        for n in pta.params
    }
    nm = NMFp(psrs, pta.rn_containers)
    freqs = np.linspace(3e-9, 5e-8, 8)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(freqs)
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    assert all(blk.comp is not None for blk in eng.blocks)
    from fastfp_amd.noise import batch_phiinv

    piv = [p if p.dim() == 2 else p[None]
           for p in batch_phiinv(pta.rn_containers, samples)]
    margins = eng.compression_margin_per_draw(piv).numpy()
    assert margins[1] < 1.5 <= min(margins[0], margins[2]), margins
    got = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu", engine=eng)
    # compression must survive the sweep (the old guard disabled it)
    assert all(blk.comp is not None for blk in eng.blocks)
    direct = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu",
                      compress=False)
    assert np.isfinite(got).all()
    np.testing.assert_allclose(got, direct, rtol=1e-7)


def test_force_direct_bypasses_compression():
    """engine.sweep(force_direct=True) must ignore an enabled
    compression and reproduce the plain direct sweep exactly."""
    psrs = make_synthetic_pta(npsr=2, ntoa=300, ntm=6, seed=18)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=8, gwb_comps=8)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    D = 3
    rng = np.random.default_rng(4)
    phiinvs = [
        c.get_phiinv({
            n: (rng.uniform(2, 6, D) if n.endswith("gamma")
                else rng.uniform(-16, -14, D))
            for n in pta.params
        })
        for c in pta.rn_containers
    ]
    freqs = np.linspace(3e-9, 5e-8, 7)
    eng_plain = FpEngine(psrs, Nvecs, Ts, device="cpu").precompute(freqs)
    want = eng_plain.sweep(phiinvs=phiinvs).numpy()
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu").precompute(freqs)
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    assert all(b.comp is not None for b in eng.blocks)
    got = eng.sweep(phiinvs=phiinvs, force_direct=True).numpy()
    np.testing.assert_array_equal(got, want)


def test_per_draw_margin_vector():
    """compression_margin_per_draw: per-draw mins, +inf when nothing
    compressed, consistent with the scalar margin."""
    psrs = make_synthetic_pta(npsr=2, ntoa=200, ntm=5, seed=19)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=5, gwb_comps=5)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    freqs = np.linspace(3e-9, 5e-8, 4)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu").precompute(freqs)
    D = 4
    rng = np.random.default_rng(6)
    samples = {
        n: (rng.uniform(3, 5, D) if n.endswith("gamma")
            else rng.uniform(-15.5, -14.5, D))
        for n in pta.params
    }
    from fastfp_amd.noise import batch_phiinv

    piv = [p if p.dim() == 2 else p[None]
           for p in batch_phiinv(pta.rn_containers, samples)]
    # nothing compressed yet -> +inf margins
    m0 = eng.compression_margin_per_draw(piv)
    assert m0.shape == (D,) and bool(torch.isinf(m0).all())
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    md = eng.compression_margin_per_draw(piv)
    assert md.shape == (D,) and bool((md > 0).all())
    assert float(md.min()) == pytest.approx(eng.compression_margin(piv))


def test_compression_margin_fallback():
    """Draws with absurdly large phi (phiinv near the jitter floor)
    must disable compression and still produce the exact direct
    answer."""
    psrs = make_synthetic_pta(npsr=2, ntoa=80, ntm=4, seed=15)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=False, rn_comps=4)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    D = 3
    samples = {}
    for n in pta.params:
        # gamma ~ 13, log10_A ~ -6: phi astronomically large
        samples[n] = (np.full(D, 12.9) if n.endswith("gamma")
                      else np.full(D, -6.0))
    nm = NMFp(psrs, pta.rn_containers)
    freqs = np.linspace(4e-9, 5e-8, 4)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(freqs)
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv({k: v[0] for k, v in samples.items()})
         for c in pta.rn_containers],
    )
    # first guard layer: the enable-time accuracy probe already rejects
    # compression at these pathological parameters
    assert all(blk.comp is None for blk in eng.blocks)
    # and the end-to-end sweep (second layer: the per-call margin check)
    # still produces the exact direct answer
    got = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu", engine=None)
    direct = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu",
                      compress=False)
    np.testing.assert_allclose(got, direct, rtol=1e-7)


def test_sweep_bitwise_invariant_to_draw_chunk():
    """Chunking the draw axis must not change results AT ALL (each
    chunk's accumulation order is per-element identical)."""
    psrs = make_synthetic_pta(npsr=2, ntoa=70, ntm=3, seed=16)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=4, gwb_comps=3)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    D = 7
    rng = np.random.default_rng(4)
    samples = {
        n: (rng.uniform(2, 6, D) if n.endswith("gamma")
            else rng.uniform(-16, -14, D))
        for n in pta.params
    }
    nm = NMFp(psrs, pta.rn_containers)
    freqs = np.linspace(4e-9, 5e-8, 5)
    a = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu", draw_chunk=2)
    b = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu", draw_chunk=7)
    np.testing.assert_array_equal(a, b)


def test_disable_compression_clears_stack():
    """disable_draw_compression must clear BOTH the per-pulsar state and
    the stacked-launch cache (the GPU sweep consults the stack first)."""
    psrs = make_synthetic_pta(npsr=2, ntoa=60, ntm=3, seed=18)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=False, rn_comps=3)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(np.linspace(4e-9, 5e-8, 4))
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    eng._comp_stack = object()  # simulate the GPU stacked cache
    eng.disable_draw_compression()
    assert eng._comp_stack is None
    assert all(blk.comp is None for blk in eng.blocks)


def test_nmfp_at_fixed_params_equals_plain_fp():
    """NM-Fp rows evaluated AT the noise-dict parameters must equal the
    plain Fp sweep (cross-validates the two API paths end to end)."""
    psrs = make_synthetic_pta(npsr=3, ntoa=80, ntm=3, seed=19)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.2
        noise[f"{p.name}_red_noise_log10_A"] = -14.3
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=4, gwb_comps=3)
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    freqs = np.linspace(4e-9, 5e-8, 6)
    plain = FastFp(psrs, pta).sweep(freqs, Nvecs, Ts, sigmas, device="cpu")

    D = 3
    samples = {k: np.full(D, float(noise[k])) for k in pta.params}
    nm = NMFp(psrs, pta.rn_containers)
    vals = nm.sweep(freqs, samples, Nvecs, Ts, device="cpu")
    for d in range(D):
        np.testing.assert_allclose(vals[d], plain, rtol=1e-7)


def test_probe_keeps_compression_on_healthy_model():
    """The enable-time accuracy probe must NOT disable compression for
    a healthy (bench-like) model — regression guard for the probe's
    error metric."""
    psrs = make_synthetic_pta(npsr=3, ntoa=400, ntm=8, seed=77)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 13.0 / 3.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=8, gwb_comps=8)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(np.arange(1, 41) / pta.Tspan)
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    assert all(blk.comp is not None for blk in eng.blocks), \
        "probe must keep compression for the healthy benchmark model"


def test_mixed_compression_rank_deficient_pulsar():
    """A pulsar with fewer TOAs than basis columns must stay on the
    exact direct path (Sigma_0 is jitter-supported along the null
    space) while the rest keep the compressed path — and the MIXED
    sweep must match the fully-direct sweep."""
    import torch

    from fastfp_amd.noise import batch_phiinv, check_batch_homogeneous

    psrs = make_synthetic_pta(npsr=2, ntoa=90, ntm=4, seed=23) + \
        make_synthetic_pta(npsr=1, ntoa=10, ntm=4, seed=24, ragged=False)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=5, gwb_comps=4)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    freqs = np.linspace(4e-9, 5e-8, 6)
    eng.precompute(freqs)
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    kept = [blk.comp is not None for blk in eng.blocks]
    assert kept == [True, True, False], kept

    D = 4
    rng = np.random.default_rng(31)
    pool = {
        n: torch.as_tensor(
            rng.uniform(2, 6, D) if n.endswith("gamma")
            else rng.uniform(-16, -14, D), dtype=torch.float64)
        for n in pta.params
    }
    homog = check_batch_homogeneous(pta.rn_containers)
    phiinvs = batch_phiinv(pta.rn_containers, pool, homogeneous=homog)
    mixed = torch.zeros((D, 6), dtype=torch.float64)
    eng.sweep(phiinvs=phiinvs, draw_chunk=4, accumulate_to=mixed)

    eng.disable_draw_compression()
    direct = torch.zeros_like(mixed)
    eng.sweep(phiinvs=phiinvs, draw_chunk=4, accumulate_to=direct)
    np.testing.assert_allclose(mixed.numpy(), direct.numpy(), rtol=1e-7)


def test_precompute_invalidates_stale_compression():
    """Re-running precompute on a NEW frequency grid must drop
    compression state built against the old grid (K's columns are
    per-frequency)."""
    psrs = make_synthetic_pta(npsr=2, ntoa=60, ntm=3, seed=40)
    noise = {"gw_gamma": 13.0 / 3.0, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=3, gwb_comps=2)
    TNTs, Nvecs, Ts = get_mats_nmfp(pta, noise)
    eng = FpEngine(psrs, Nvecs, Ts, device="cpu")
    eng.precompute(np.linspace(4e-9, 5e-8, 4))
    eng.enable_draw_compression(
        [c.var_slice for c in pta.rn_containers],
        [c.get_phiinv(noise) for c in pta.rn_containers],
    )
    assert all(blk.comp is not None for blk in eng.blocks)
    eng.precompute(np.linspace(4e-9, 5e-8, 9))
    assert all(blk.comp is None for blk in eng.blocks)
    assert eng._comp_stack is None


def test_graph_cache_keyed_by_engine_identity():
    """The hipGraph CLI cache closes over one engine (its freq grid),
    so a same-shaped sweep against a DIFFERENT engine must not replay
    a stale capture: the cache key includes the engine's identity.
    Exercised with stub engines that satisfy the graph-path
    preconditions (the capture itself is HIP-only and fails closed)."""
    from fastfp_amd.nmfp import NMFp

    class StubEngine:
        _use_hip = True
        _comp_stack = object()
        device = "cpu"

    nm = NMFp.__new__(NMFp)
    nm.rn_sigs = []
    nm._phi_homog = True
    nm._graphs = {}
    nm._graph_seen = {}
    samples = {"gw_gamma": np.ones(3), "gw_log10_A": np.full(3, -14.5)}

    eng_a, eng_b = StubEngine(), StubEngine()
    # first sighting of (engine A, shape): eager
    assert nm._sweep_graphed(eng_a, samples, 4, True, None, None) is None
    key_a = (id(eng_a), 3, 4, tuple(samples))
    assert nm._graph_seen[key_a] == 1 and key_a not in nm._graphs
    # second sighting: capture attempted; stub engine -> permanent
    # eager fallback CACHED UNDER ENGINE A's KEY
    assert nm._sweep_graphed(eng_a, samples, 4, True, None, None) is None
    assert nm._graphs[key_a] is False
    # engine B, same shapes: must NOT hit engine A's cache entry
    assert nm._sweep_graphed(eng_b, samples, 4, True, None, None) is None
    key_b = (id(eng_b), 3, 4, tuple(samples))
    assert key_b != key_a
    assert nm._graph_seen[key_b] == 1 and key_b not in nm._graphs


def test_duplicate_toas_all_noise_modes():
    """EXACT duplicate TOAs (simultaneous multi-channel observations,
    common in real backends): engine == parity in all three noise
    modes (plain / GP-ECORR / EcorrKernelNoise)."""
    from fastfp_amd.data import PulsarData

    rng = np.random.default_rng(0)
    n = 60
    base = np.sort(rng.uniform(0, 4e8, n // 3))
    toas = np.repeat(base, 3)
    M = np.stack([np.ones(n), toas / toas.max(),
                  (toas / toas.max()) ** 2], axis=1)
    psr = PulsarData(
        name="JDUP", toas=toas, toaerrs=np.full(n, 1e-6),
        residuals=rng.normal(0, 1e-6, n), Mmat=M,
        backend_flags=np.array(["A"] * n, dtype=object),
        pos=np.array([0.3, 0.5, 0.8]),
    )
    noise = {"gw_gamma": 13 / 3, "gw_log10_A": float(np.log10(2e-15)),
             "JDUP_red_noise_gamma": 4.0,
             "JDUP_red_noise_log10_A": -14.5,
             "JDUP_basis_ecorr_A_log10_ecorr": -6.5}
    for ek, ie in ((False, False), (False, True), (True, False)):
        pta = initialize_pta([psr], noise, inc_cp=True, rn_comps=3,
                             gwb_comps=2, ecorr_kernel=ek, inc_ecorr=ie)
        Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
        freqs = np.array([8e-9, 2e-8])
        fp_obj = FastFp([psr])
        want = np.array([fp_obj.calculate_Fp(f, Nvecs, Ts, sigmas)
                         for f in freqs])
        got = fp_obj.sweep(freqs, Nvecs, Ts, sigmas, device="cpu")
        np.testing.assert_allclose(got, want, rtol=1e-6,
                                   err_msg=f"ek={ek} ie={ie}")


def test_degenerate_full_span_timing_model_nonfinite_consistently():
    """A timing model whose basis spans the ENTIRE residual space
    absorbs every filter (M -> 0, N -> 0): Fp is mathematically
    undefined (0/0).  The two resolutions round-off can produce are
    ~0 (solve of round-off by round-off) or non-finite (closed-form
    det underflow) — BOTH are safely self-announcing (the CLIs warn on
    non-finite values; ~0 is an obvious null).  What must never
    happen is a plausible-looking finite statistic; this documents
    that property."""
    psr = make_synthetic_pta(npsr=1, ntoa=12, ntm=3, seed=2,
                             ragged=False)[0]
    rng = np.random.default_rng(1)
    psr.Mmat = np.linalg.qr(rng.normal(size=(12, 12)))[0]  # complete basis
    noise = {"gw_gamma": 13 / 3, "gw_log10_A": float(np.log10(2e-15)),
             f"{psr.name}_red_noise_gamma": 4.0,
             f"{psr.name}_red_noise_log10_A": -14.5}
    pta = initialize_pta([psr], noise, inc_cp=True, rn_comps=2,
                         gwb_comps=2)
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    got = FastFp([psr]).sweep(np.array([1e-8]), Nvecs, Ts, sigmas,
                              device="cpu")
    want = FastFp([psr]).calculate_Fp(1e-8, Nvecs, Ts, sigmas)
    for v in (float(got[0]), float(want)):
        assert (not np.isfinite(v)) or abs(v) < 1e-6, v


def test_zero_residuals_give_zero_fp():
    """No signal, no noise realization -> Fp exactly 0 (N = 0)."""
    psrs = make_synthetic_pta(npsr=2, ntoa=60, ntm=3, seed=1)
    for p in psrs:
        p.residuals = np.zeros_like(p.residuals)
    noise = {"gw_gamma": 13 / 3, "gw_log10_A": float(np.log10(2e-15))}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    pta = initialize_pta(psrs, noise, inc_cp=True, rn_comps=3, gwb_comps=2)
    Nvecs, Ts, sigmas = get_mats_fp(pta, noise)
    got = FastFp(psrs).sweep(np.array([1e-8, 3e-8]), Nvecs, Ts, sigmas,
                             device="cpu")
    np.testing.assert_array_equal(got, 0.0)
