"""The reference's import paths must work with s/fastfp/fastfp_amd/."""

import pickle

import numpy as np


def test_reference_import_paths():
    from fastfp_amd.fastfp import FastFp  # noqa: F401
    from fastfp_amd.nmfp import NMFP, NMFp  # noqa: F401
    from fastfp_amd import (  # noqa: F401
        CURN_container,
        GPEcorr_container,
        RN_container,
    )
    from fastfp_amd.utils import (  # noqa: F401
        get_mats_fp,
        get_mats_nmfp,
        get_xCy,
        initialize_pta,
    )
    from fastfp_amd.constants import fyr, yr, day  # noqa: F401
    assert NMFP is NMFp


class _MockEnterprisePulsar:
    """Duck-typed stand-in for enterprise.pulsar.Pulsar."""

    def __init__(self, n=40, seed=0):
        rng = np.random.default_rng(seed)
        self.name = "J0000+0000"
        self.toas = np.sort(rng.uniform(0, 3e8, n))
        self.toaerrs = np.full(n, 1e-6)
        self.residuals = rng.normal(0, 1e-6, n)
        self.Mmat = rng.normal(size=(n, 4))
        self.backend_flags = np.array(["b1"] * n, dtype=object)


def test_enterprise_style_pickle_loads(tmp_path):
    from fastfp_amd.data import load_pulsars

    path = str(tmp_path / "psrs.pkl")
    with open(path, "wb") as f:
        pickle.dump([_MockEnterprisePulsar()], f)
    psrs = load_pulsars(path)
    assert len(psrs) == 1 and psrs[0].ntoa == 40
    assert psrs[0].name == "J0000+0000"


def test_feather_roundtrip(tmp_path):
    """Per-pulsar feather save/load round-trips exactly (the SURVEY §7
    npz/feather loader plan; fastfp_amd's own self-describing schema)."""
    from fastfp_amd.data import PulsarData, load_pulsars, make_synthetic_pta

    psrs = make_synthetic_pta(npsr=2, ntoa=50, ntm=4, seed=3)
    for p in psrs:
        p.save_feather(str(tmp_path / f"{p.name}.feather"))
    back = load_pulsars(str(tmp_path))
    assert len(back) == 2
    by_name = {p.name: p for p in back}
    for p in psrs:
        q = by_name[p.name]
        np.testing.assert_array_equal(q.toas, p.toas)
        np.testing.assert_array_equal(q.toaerrs, p.toaerrs)
        np.testing.assert_array_equal(q.residuals, p.residuals)
        np.testing.assert_array_equal(q.Mmat, p.Mmat)
        assert list(q.backend_flags) == list(p.backend_flags)
    # single-file load too
    one = load_pulsars(str(tmp_path / f"{psrs[0].name}.feather"))
    assert len(one) == 1 and one[0].ntoa == psrs[0].ntoa
