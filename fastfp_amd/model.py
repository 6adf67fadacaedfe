"""Native PTA model: the enterprise surface the reference actually uses,
rebuilt from scratch.

The reference calls ``pta.get_phiinv / get_TNT / get_ndiag / get_basis``
(``/root/reference/fastfp/utils.py:72-75``) and ``pta.params`` /
``pta.map_params`` (``/root/reference/examples/run_nmfp.py:174-186``) on an
``enterprise.signal_base.PTA``.  :class:`PTAModel` provides exactly that
surface, with the basis/phi conventions of :mod:`fastfp_amd.bases` and
:mod:`fastfp_amd.noise`:

``T_p = [tm_svd | U_ecorr | F_rn]`` and diagonal
``phi = [1e40*ones | ecorr | rn(+curn on the first 2*ngwb bins)]``.

When a common process is included, the CURN shares the red-noise Fourier
basis (both built on the PTA-wide Tspan) and its phi is ADDED onto the
first ``2*gwb_comps`` red-noise bins — the same convention enterprise
reaches by de-duplicating identical bases and that the reference
hard-codes (``/root/reference/fastfp/nmfp.py:247``).  This requires
``gwb_comps <= rn_comps`` (asserted; the reference default is 30/30).
"""

from __future__ import annotations

import numpy as np

from fastfp_amd.bases import (
    create_freqarray,
    ecorr_basis_by_backend,
    fourier_basis,
    timing_model_basis_svd,
)
from fastfp_amd.data import get_tspan
from fastfp_amd.blocknoise import BlockNoise
from fastfp_amd.noise import (
    CURNContainer,
    GPEcorrContainer,
    RNContainer,
    white_noise_nvec,
)


class PTAModel:
    """Holds the assembled per-pulsar bases and noise model.

    Parameters are fixed at construction from the noise dict except the
    sampled red-noise parameters ``{psr}_red_noise_{gamma,log10_A}`` and
    (if ``inc_cp``) ``gw_{gamma,log10_A}``.
    """

    def __init__(
        self,
        psrs: list,
        noise: dict,
        inc_cp: bool = True,
        rn_comps: int = 30,
        gwb_comps: int = 30,
        simple_wn: bool = True,
        inc_ecorr: bool = False,
        select: str = "backend",
        ecorr_kernel: bool = False,
        per_psr_tspan: bool = False,
    ):
        """``ecorr_kernel=True`` models ECORR as BLOCK-DIAGONAL white
        noise (enterprise's EcorrKernelNoise) instead of a basis GP:
        ``get_ndiag`` then returns :class:`BlockNoise` objects and the
        basis/phi carry no ECORR columns.  This is the case the
        reference explicitly does not support
        (``/root/reference/fastfp/utils.py:30-31``, README.md:22)."""
        if inc_ecorr and ecorr_kernel:
            raise ValueError("choose GP ecorr (inc_ecorr) OR kernel ecorr")
        if per_psr_tspan and inc_cp:
            # the reference's Tspan=None + add_curn combination adds the
            # CURN PSD onto per-pulsar-frequency bins, which mixes
            # different physical frequencies per pulsar; we reject it
            # rather than reproduce the quirk (SURVEY.md §2.5 spirit)
            raise ValueError(
                "per-pulsar Tspan red-noise bases are incompatible with a "
                "shared-basis common process; use a PTA-wide Tspan"
            )
        self.ecorr_kernel = ecorr_kernel
        self.per_psr_tspan = per_psr_tspan
        if inc_cp:
            assert gwb_comps <= rn_comps, (
                "shared-basis CURN requires gwb_comps <= rn_comps "
                f"(got {gwb_comps} > {rn_comps})"
            )
        self.psrs = psrs
        self.noise = dict(noise) if noise else {}
        self.inc_cp = inc_cp
        self.rn_comps = rn_comps
        self.gwb_comps = gwb_comps
        self.simple_wn = simple_wn
        self.inc_ecorr = inc_ecorr
        self.select = select

        self.Tspan = get_tspan(psrs)
        #: PTA-wide red-noise grid (or per-pulsar grids, the reference's
        #: setup_fp_model(Tspan=None) mode, run_nmfp.py:94-98)
        self.Ffreqs_rn = create_freqarray(self.Tspan, rn_comps)

        curn = None
        if inc_cp:
            curn = CURNContainer(create_freqarray(self.Tspan, gwb_comps))
        self.curn_container = curn

        self._Ts = []
        self._Nvecs = []
        self.rn_containers = []
        self.tm_slices = []
        self.ecorr_slices = []
        self.rn_slices = []
        for psr in psrs:
            tmU = timing_model_basis_svd(psr.Mmat)
            blocks = [tmU]
            ecorr_obj = None
            n_ec = 0
            if inc_ecorr:
                Uec, weights = ecorr_basis_by_backend(psr)
                ecorr_obj = GPEcorrContainer(psr, weights, fix_wn_vals=self.noise)
                blocks.append(Uec)
                n_ec = Uec.shape[1]
            Ffreqs_p = (
                create_freqarray(psr.Tspan, rn_comps)
                if per_psr_tspan
                else self.Ffreqs_rn
            )
            Frn = fourier_basis(psr.toas, Ffreqs_p)
            blocks.append(Frn)
            T = np.concatenate(blocks, axis=1)
            self._Ts.append(T)
            ntm = tmU.shape[1]
            self.tm_slices.append(slice(0, ntm))
            self.ecorr_slices.append(slice(ntm, ntm + n_ec))
            self.rn_slices.append(slice(ntm + n_ec, T.shape[1]))

            if ecorr_kernel:
                self._Nvecs.append(
                    BlockNoise(
                        psr, self.noise, simple_wn=simple_wn, select=select
                    )
                )
            else:
                self._Nvecs.append(
                    white_noise_nvec(
                        psr, self.noise, simple_wn=simple_wn, select=select
                    )
                )
            self.rn_containers.append(
                RNContainer(
                    psr,
                    Ffreqs=Ffreqs_p,
                    ncomps=rn_comps,
                    gp_ecorr=inc_ecorr,
                    ecorr_container=ecorr_obj,
                    add_curn=inc_cp,
                    curn_container=curn,
                )
            )

        names = []
        for psr in psrs:
            names.append(f"{psr.name}_red_noise_gamma")
            names.append(f"{psr.name}_red_noise_log10_A")
        if inc_cp:
            names.append("gw_gamma")
            names.append("gw_log10_A")
        #: sampled-parameter names, sorted alphabetically as enterprise's
        #: ``PTA.params`` does (chain column convention).
        self.param_names = sorted(names)

    # ------------------------------------------------------------------
    # enterprise-compatible surface
    # ------------------------------------------------------------------
    @property
    def params(self):
        return list(self.param_names)

    def map_params(self, xs) -> dict:
        """Map a 1-D parameter vector (chain row) to a name->value dict."""
        xs = np.asarray(xs)
        return {name: xs[i] for i, name in enumerate(self.param_names)}

    def get_ndiag(self, noise: dict = None) -> list:
        return [
            nv if isinstance(nv, BlockNoise) else nv.copy()
            for nv in self._Nvecs
        ]

    def get_basis(self, noise: dict = None) -> list:
        return [T.copy() for T in self._Ts]

    def get_phiinv(self, noise: dict) -> list:
        """Per-pulsar diagonal phi^-1 at fixed parameter values from the
        noise dict (the plain-Fp path, ``/root/reference/fastfp/utils.py:72``)."""
        out = []
        for cont in self.rn_containers:
            phiinv = cont.get_phiinv(noise)
            out.append(phiinv.cpu().numpy())
        return out

    def get_TNT(self, noise: dict = None) -> list:
        out = []
        for T, nv in zip(self._Ts, self._Nvecs):
            if isinstance(nv, BlockNoise):
                Tp = T[nv.perm, :]
                out.append(Tp.T @ nv.solve(Tp))
            else:
                TN = T / nv[:, None]
                out.append(T.T @ TN)
        return out


def initialize_pta(
    psrs,
    noise,
    inc_cp=True,
    rn_comps=30,
    gwb_comps=30,
    simple_wn=True,
    inc_ecorr=False,
    select="backend",
    ecorr_kernel=False,
    per_psr_tspan=False,
) -> PTAModel:
    """Build the PTA model — signature parity with the reference's
    ``initialize_pta`` (``/root/reference/fastfp/utils.py:104-113``),
    plus ``ecorr_kernel`` for the block-diagonal-N ECORR path the
    reference lacks."""
    return PTAModel(
        psrs,
        noise,
        inc_cp=inc_cp,
        rn_comps=rn_comps,
        gwb_comps=gwb_comps,
        simple_wn=simple_wn,
        inc_ecorr=inc_ecorr,
        select=select,
        ecorr_kernel=ecorr_kernel,
        per_psr_tspan=per_psr_tspan,
    )


def get_mats_fp(pta: PTAModel, noise: dict):
    """(Nvecs, Ts, sigmas) precompute for the plain Fp path — parity with
    ``/root/reference/fastfp/utils.py:57-78``."""
    phiinvs = pta.get_phiinv(noise)
    TNTs = pta.get_TNT(noise)
    Nvecs = pta.get_ndiag(noise)
    Ts = pta.get_basis(noise)
    sigmas = [TNT + np.diag(phiinv) for TNT, phiinv in zip(TNTs, phiinvs)]
    return Nvecs, Ts, sigmas


def get_mats_nmfp(pta: PTAModel, noise: dict = None):
    """(TNTs, Nvecs, Ts) precompute for the NM-Fp path — parity with
    ``/root/reference/fastfp/utils.py:81-101``."""
    TNTs = pta.get_TNT(noise)
    Nvecs = pta.get_ndiag(noise)
    Ts = pta.get_basis(noise)
    return TNTs, Nvecs, Ts
