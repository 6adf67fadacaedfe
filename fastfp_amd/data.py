"""Pulsar data containers, loaders, and a synthetic-PTA generator.

The reference delegates all data handling to ``enterprise.pulsar.Pulsar``
pickles (``/root/reference/examples/run_fp.py:34-35``); the attribute
surface it actually consumes is ``.toas``, ``.residuals``, ``.Mmat``,
``.backend_flags`` and ``.name`` (``/root/reference/fastfp/fastfp.py:44-45``,
``/root/reference/fastfp/nmfp.py:186,444``).  This module owns that surface
natively: an ``npz`` on-disk format, a duck-typed converter for
enterprise-style pickles, and a synthetic NANOGrav/SKA-scale PTA generator
for benchmarks (no network: synthetic data is the only data source here).

Everything is float64 end to end.
"""

from __future__ import annotations

import glob
import os
import pickle
from dataclasses import dataclass, field

import numpy as np

from fastfp_amd.constants import yr, day


@dataclass
class PulsarData:
    """Per-pulsar data container.

    Attributes
    ----------
    name : str
        Pulsar name (e.g. ``J1909-3744``).
    toas : (ntoa,) float64
        Times of arrival, seconds.
    toaerrs : (ntoa,) float64
        TOA measurement uncertainties, seconds.
    residuals : (ntoa,) float64
        Timing residuals, seconds.
    Mmat : (ntoa, ntm) float64
        Timing-model design matrix.
    backend_flags : (ntoa,) object/str array
        Receiver/backend flag per TOA.
    """

    name: str
    toas: np.ndarray
    toaerrs: np.ndarray
    residuals: np.ndarray
    Mmat: np.ndarray
    backend_flags: np.ndarray = field(default=None)
    #: unit 3-vector pointing from the SSB to the pulsar (enterprise's
    #: ``Pulsar.pos``); needed by the sky-coherent Fe statistic.  None
    #: for data sources that carry no astrometry.
    pos: np.ndarray = field(default=None)

    def __post_init__(self):
        self.toas = np.asarray(self.toas, dtype=np.float64)
        self.toaerrs = np.asarray(self.toaerrs, dtype=np.float64)
        self.residuals = np.asarray(self.residuals, dtype=np.float64)
        self.Mmat = np.asarray(self.Mmat, dtype=np.float64)
        if self.backend_flags is None:
            self.backend_flags = np.array(["backend"] * self.ntoa, dtype=object)
        else:
            self.backend_flags = np.asarray(self.backend_flags)
        if self.pos is not None:
            self.pos = np.asarray(self.pos, dtype=np.float64)
            self.pos = self.pos / np.linalg.norm(self.pos)

    # ------------------------------------------------------------------
    @property
    def ntoa(self) -> int:
        return self.toas.shape[0]

    @property
    def ntm(self) -> int:
        return self.Mmat.shape[1]

    @property
    def Tspan(self) -> float:
        return float(np.max(self.toas) - np.min(self.toas))

    # ------------------------------------------------------------------
    def save_npz(self, path: str) -> None:
        arrs = dict(
            name=np.asarray(self.name),
            toas=self.toas,
            toaerrs=self.toaerrs,
            residuals=self.residuals,
            Mmat=self.Mmat,
            backend_flags=np.asarray(self.backend_flags, dtype=str),
        )
        if self.pos is not None:
            arrs["pos"] = self.pos
        np.savez_compressed(path, **arrs)

    @classmethod
    def load_npz(cls, path: str) -> "PulsarData":
        z = np.load(path, allow_pickle=False)
        return cls(
            name=str(z["name"]),
            toas=z["toas"],
            toaerrs=z["toaerrs"],
            residuals=z["residuals"],
            Mmat=z["Mmat"],
            backend_flags=z["backend_flags"].astype(object),
            pos=z["pos"] if "pos" in z.files else None,
        )

    def save_feather(self, path: str) -> None:
        """Write the pulsar as an Arrow/feather table (one row per TOA:
        toas/toaerrs/residuals/backend_flags columns plus the design
        matrix as ``Mmat_j`` columns; pulsar name in the schema
        metadata).  This is fastfp_amd's own self-describing feather
        schema, not enterprise's internal FeatherPulsar layout (whose
        spec is not public); ``load_pulsars`` round-trips it exactly."""
        import pyarrow as pa
        import pyarrow.ipc as paipc

        cols = {
            "toas": self.toas,
            "toaerrs": self.toaerrs,
            "residuals": self.residuals,
            "backend_flags": np.asarray(self.backend_flags, dtype=str),
        }
        for j in range(self.ntm):
            cols[f"Mmat_{j}"] = self.Mmat[:, j]
        meta = {"fastfp_amd.name": self.name, "fastfp_amd.ntm": str(self.ntm)}
        if self.pos is not None:
            meta["fastfp_amd.pos"] = ",".join(repr(float(v)) for v in self.pos)
        table = pa.table(cols).replace_schema_metadata(meta)
        # the Arrow IPC file format IS Feather V2
        with paipc.new_file(path, table.schema) as w:
            w.write_table(table)

    @classmethod
    def load_feather(cls, path: str) -> "PulsarData":
        import pyarrow.ipc as paipc

        with paipc.open_file(path) as r:
            table = r.read_all()
        meta = table.schema.metadata or {}
        name = meta.get(b"fastfp_amd.name", b"unknown").decode()
        ntm = int(meta.get(b"fastfp_amd.ntm", b"0"))
        pos_s = meta.get(b"fastfp_amd.pos", None)
        pos = (
            np.array([float(v) for v in pos_s.decode().split(",")])
            if pos_s else None
        )
        cols = {c: table[c].to_numpy() for c in table.column_names}
        Mmat = np.stack(
            [cols[f"Mmat_{j}"] for j in range(ntm)], axis=1
        ) if ntm else np.zeros((len(cols["toas"]), 0))
        return cls(
            name=name,
            toas=cols["toas"],
            toaerrs=cols["toaerrs"],
            residuals=cols["residuals"],
            Mmat=Mmat,
            backend_flags=cols["backend_flags"].astype(object),
            pos=pos,
        )

    @classmethod
    def from_object(cls, obj) -> "PulsarData":
        """Duck-typed converter from an enterprise-style Pulsar object
        (anything exposing .name/.toas/.toaerrs/.residuals/.Mmat/
        .backend_flags)."""
        toaerrs = getattr(obj, "toaerrs", None)
        if toaerrs is None:
            toaerrs = np.full(len(obj.toas), 1e-6)
        # keep a MISSING backend_flags as None (np.asarray(None) is a
        # 0-d object array, which would bypass the per-TOA default in
        # __post_init__ and break backend masking downstream)
        bflags = getattr(obj, "backend_flags", None)
        pos = getattr(obj, "pos", None)
        return cls(
            name=str(obj.name),
            toas=np.asarray(obj.toas, dtype=np.float64),
            toaerrs=np.asarray(toaerrs, dtype=np.float64),
            residuals=np.asarray(obj.residuals, dtype=np.float64),
            Mmat=np.asarray(obj.Mmat, dtype=np.float64),
            backend_flags=None if bflags is None else np.asarray(bflags),
            pos=None if pos is None else np.asarray(pos, dtype=np.float64),
        )


# ----------------------------------------------------------------------
# loaders
# ----------------------------------------------------------------------
def load_pulsars(path: str) -> list:
    """Load a list of pulsars from one of:

    - a ``.pkl`` pickle of a list of enterprise-style Pulsar objects
      (the reference's input format, ``/root/reference/examples/run_fp.py:34``),
    - a ``.npz`` file saved by :func:`save_pulsars`,
    - a per-pulsar ``.feather`` file (:meth:`PulsarData.save_feather`),
    - a directory of per-pulsar ``.npz`` and/or ``.feather`` files.
    """
    if os.path.isdir(path):
        out = [
            PulsarData.load_npz(f)
            for f in sorted(glob.glob(os.path.join(path, "*.npz")))
        ]
        out += [
            PulsarData.load_feather(f)
            for f in sorted(glob.glob(os.path.join(path, "*.feather")))
        ]
        return out
    if path.endswith(".feather"):
        return [PulsarData.load_feather(path)]
    if path.endswith(".pkl") or path.endswith(".pickle"):
        with open(path, "rb") as f:
            objs = pickle.load(f)
        return [
            o if isinstance(o, PulsarData) else PulsarData.from_object(o)
            for o in objs
        ]
    if path.endswith(".npz"):
        z = np.load(path, allow_pickle=True)
        n = int(z["npsr"])
        out = []
        for i in range(n):
            out.append(
                PulsarData(
                    name=str(z[f"name_{i}"]),
                    toas=z[f"toas_{i}"],
                    toaerrs=z[f"toaerrs_{i}"],
                    residuals=z[f"residuals_{i}"],
                    Mmat=z[f"Mmat_{i}"],
                    backend_flags=z[f"backend_flags_{i}"].astype(object),
                    pos=z[f"pos_{i}"] if f"pos_{i}" in z.files else None,
                )
            )
        return out
    raise ValueError(f"unrecognized pulsar file format: {path}")


def save_pulsars(psrs: list, path: str) -> None:
    """Save a list of PulsarData into a single .npz archive."""
    arrs = {"npsr": np.asarray(len(psrs))}
    for i, p in enumerate(psrs):
        arrs[f"name_{i}"] = np.asarray(p.name)
        arrs[f"toas_{i}"] = p.toas
        arrs[f"toaerrs_{i}"] = p.toaerrs
        arrs[f"residuals_{i}"] = p.residuals
        arrs[f"Mmat_{i}"] = p.Mmat
        arrs[f"backend_flags_{i}"] = np.asarray(p.backend_flags, dtype=str)
        if p.pos is not None:
            arrs[f"pos_{i}"] = p.pos
    np.savez_compressed(path, **arrs)


# ----------------------------------------------------------------------
# synthetic PTA generator
# ----------------------------------------------------------------------
def _design_matrix(toas: np.ndarray, ntm: int, rng: np.random.Generator) -> np.ndarray:
    """Synthetic timing-model design matrix.

    Columns: quadratic spindown (1, t, t^2), annual sin/cos (astrometry
    proxy), then smooth random Fourier-like columns up to ``ntm`` (DM
    model / jumps proxy).  Only the column SPAN matters downstream: the
    engine orthonormalizes via SVD exactly as enterprise's
    ``TimingModel(use_svd=True)`` does (``/root/reference/fastfp/utils.py:146``).
    """
    t0 = toas - toas.mean()
    tspan = toas.max() - toas.min()
    cols = [np.ones_like(t0), t0 / tspan, (t0 / tspan) ** 2]
    cols.append(np.sin(2 * np.pi * t0 / yr))
    cols.append(np.cos(2 * np.pi * t0 / yr))
    k = 1
    while len(cols) < ntm:
        # low-frequency smooth columns (DM/FD/jump proxies).  The
        # half-integer frequencies overlap the red-noise Fourier basis
        # REALISTICALLY (strong correlation) without being exactly
        # degenerate with it.
        cols.append(
            np.sin(2 * np.pi * (k + 0.37) * t0 / tspan + rng.uniform(0, 2 * np.pi))
        )
        k += 1
    return np.stack(cols[:ntm], axis=1)


def make_pulsar(
    name: str,
    ntoa: int = 1000,
    tspan_yr: float = 15.0,
    toaerr: float = 1e-6,
    ntm: int = 10,
    backends=("BE_A", "BE_B"),
    red_amp: float = 0.0,
    red_gamma: float = 13.0 / 3.0,
    red_ncomps: int = 30,
    cw_amp: float = 0.0,
    cw_freq: float = 1e-8,
    rng: np.random.Generator = None,
) -> PulsarData:
    """Generate one synthetic pulsar.

    Residuals are white noise of std ``toaerr`` per TOA plus (optionally)
    a power-law red-noise realization with amplitude ``red_amp`` drawn on
    a ``red_ncomps``-component Fourier basis, plus (optionally) a
    continuous-wave signal of amplitude ``cw_amp`` (seconds) at
    ``cw_freq`` (Hz) with a random phase — for Fp detection validation.
    """
    rng = rng or np.random.default_rng(0)
    # epoch-structured TOAs: real PTA observations come in epochs of
    # several TOAs (frequency channels) within a fraction of a day --
    # this also gives the ECORR quantization buckets real structure.
    toas_per_epoch = 4
    nepoch = max(1, ntoa // toas_per_epoch)
    centers = np.sort(rng.uniform(0.0, tspan_yr * yr, size=nepoch))
    epoch_of = rng.integers(0, nepoch, size=ntoa)
    toas = centers[epoch_of] + rng.uniform(0.0, 0.2 * day, size=ntoa)
    order = np.argsort(toas)
    toas = toas[order]
    epoch_of = epoch_of[order]
    # reference epoch offset so toas look MJD-ish in seconds
    toas = toas + 53000.0 * day
    toaerrs = np.full(ntoa, toaerr)
    resid = rng.normal(0.0, toaerr, size=ntoa)

    if red_amp > 0.0:
        from fastfp_amd.bases import fourier_basis, create_freqarray
        from fastfp_amd.noise import powerlaw_psd

        Tspan = toas.max() - toas.min()
        Ffreqs = create_freqarray(Tspan, red_ncomps)
        F = fourier_basis(toas, Ffreqs)
        phi = powerlaw_psd(Ffreqs, np.log10(red_amp), red_gamma)
        coeffs = rng.normal(0.0, np.sqrt(phi))
        resid = resid + F @ coeffs

    # backend per epoch (one receiver per observation)
    if cw_amp > 0.0:
        phase = rng.uniform(0, 2 * np.pi)
        resid = resid + cw_amp * np.sin(2 * np.pi * cw_freq * toas + phase)

    bflags = np.asarray(
        [backends[int(e) % len(backends)] for e in epoch_of], dtype=object
    )
    Mmat = _design_matrix(toas, ntm, rng)
    # isotropic sky position (unit vector) for the sky-coherent Fe path
    v = rng.normal(size=3)
    return PulsarData(
        name=name,
        toas=toas,
        toaerrs=toaerrs,
        residuals=resid,
        Mmat=Mmat,
        backend_flags=bflags,
        pos=v / np.linalg.norm(v),
    )


def make_synthetic_pta(
    npsr: int = 3,
    ntoa: int = 1000,
    tspan_yr: float = 15.0,
    toaerr: float = 1e-6,
    ntm: int = 10,
    red_amp: float = 0.0,
    red_gamma: float = 13.0 / 3.0,
    cw_amp: float = 0.0,
    cw_freq: float = 1e-8,
    seed: int = 0,
    ragged: bool = True,
) -> list:
    """Generate a synthetic PTA (list of PulsarData).

    With ``ragged=True`` the per-pulsar TOA counts vary +-20% around
    ``ntoa`` — the engine must handle ragged pulsars, so tests and
    benchmarks exercise that by default.
    """
    rng = np.random.default_rng(seed)
    psrs = []
    for i in range(npsr):
        n = ntoa
        if ragged and ntoa >= 10:
            n = int(ntoa * rng.uniform(0.8, 1.2))
        psrs.append(
            make_pulsar(
                name=f"J{i:04d}+{seed:04d}",
                ntoa=n,
                tspan_yr=tspan_yr,
                toaerr=toaerr,
                ntm=ntm,
                red_amp=red_amp,
                red_gamma=red_gamma,
                cw_amp=cw_amp,
                cw_freq=cw_freq,
                rng=rng,
            )
        )
    return psrs


def get_tspan(psrs: list) -> float:
    """Max - min TOA over the whole array (parity with
    ``enterprise_extensions.model_utils.get_tspan``, used at
    ``/root/reference/fastfp/utils.py:145``)."""
    tmin = min(np.min(p.toas) for p in psrs)
    tmax = max(np.max(p.toas) for p in psrs)
    return float(tmax - tmin)
