"""CLI round-trip tests: input loading, output formats (JSON / .npy),
resume behavior — the reference's exact output contracts
(/root/reference/examples/run_fp.py:69-71, run_nmfp.py:270-276)."""

import json
import os

import numpy as np
import pytest

from fastfp_amd.cli import run_fp, run_nmfp
from fastfp_amd.data import make_synthetic_pta, save_pulsars


@pytest.fixture(scope="module")
def inputs(tmp_path_factory):
    tmp = tmp_path_factory.mktemp("cli")
    psrs = make_synthetic_pta(npsr=2, ntoa=60, ntm=3, seed=0)
    psrfile = str(tmp / "psrs.npz")
    save_pulsars(psrs, psrfile)
    noise = {}
    for p in psrs:
        noise[f"{p.name}_red_noise_gamma"] = 4.0
        noise[f"{p.name}_red_noise_log10_A"] = -14.5
    noisefile = str(tmp / "noise.json")
    with open(noisefile, "w") as f:
        json.dump(noise, f)
    # fake chain: params sorted (2 per psr + 2 gw) + 4 bookkeeping cols
    nparams = 2 * len(psrs) + 2
    rng = np.random.default_rng(1)
    chain = np.zeros((40, nparams + 4))
    for i in range(nparams):
        chain[:, i] = rng.uniform(2.0, 6.0, 40) if i % 2 == 0 else rng.uniform(-16, -14, 40)
    # columns alternate by sorted name: gamma cols then log10_A cols per psr;
    # order doesn't matter for smoke purposes as long as values are sane
    chainfile = str(tmp / "chain.txt")
    np.savetxt(chainfile, chain)
    return tmp, psrfile, noisefile, chainfile


def test_run_fp_json_output(inputs):
    tmp, psrfile, noisefile, _ = inputs
    out = str(tmp / "fpout")
    run_fp.main(psrfile, noisefile, out, nfreqs=5, rn_comps=3, gwb_comps=3,
                device="cpu")
    with open(out + ".json") as f:
        res = json.load(f)
    assert len(res) == 5
    freqs = np.array(sorted(float(k) for k in res))
    np.testing.assert_allclose(freqs, np.linspace(2e-9, 3e-7, 5))
    assert all(np.isfinite(v) for v in res.values())


def test_run_nmfp_npy_output(inputs):
    tmp, psrfile, noisefile, chainfile = inputs
    outdir = str(tmp / "res")
    run_nmfp.main(
        psrfile, noisefile, chainfile, "nm",
        inc_cp=True, nrncomps=3, ngwbcomps=3, ncwfreqs=4,
        nsamples=6, batch_size=4, outdir=outdir, device="cpu",
    )
    vals = np.load(os.path.join(outdir, "nm.npy"))
    assert vals.shape == (6, 4)
    assert np.isfinite(vals).all()


def test_run_nmfp_resume(inputs):
    tmp, psrfile, noisefile, chainfile = inputs
    outdir = str(tmp / "res2")
    kwargs = dict(
        inc_cp=False, nrncomps=3, ncwfreqs=3, nsamples=4, batch_size=2,
        outdir=outdir, device="cpu", seed=3,
    )
    run_nmfp.main(psrfile, noisefile, chainfile, "nm", **kwargs)
    a = np.load(os.path.join(outdir, "nm.npy"))
    # resume run must reproduce identical output from the batch shards
    run_nmfp.main(psrfile, noisefile, chainfile, "nm", resume=True, **kwargs)
    b = np.load(os.path.join(outdir, "nm.npy"))
    np.testing.assert_array_equal(a, b)


def test_run_nmfp_resume_invalidates_on_config_change(inputs):
    """A resume with a changed seed must NOT mix stale batch shards into
    the output (the shards are keyed only by rank/offset; the manifest
    fingerprints the run config)."""
    tmp, psrfile, noisefile, chainfile = inputs
    outdir = str(tmp / "res3")
    kwargs = dict(
        inc_cp=False, nrncomps=3, ncwfreqs=3, nsamples=4, batch_size=2,
        outdir=outdir, device="cpu",
    )
    run_nmfp.main(psrfile, noisefile, chainfile, "nm", seed=3, **kwargs)
    # different seed -> different draw selection -> shards are stale
    run_nmfp.main(psrfile, noisefile, chainfile, "nm", seed=4, resume=True,
                  **kwargs)
    b = np.load(os.path.join(outdir, "nm.npy"))
    # fresh non-resumed run with seed=4 is the ground truth
    outdir2 = str(tmp / "res3b")
    kwargs["outdir"] = outdir2
    run_nmfp.main(psrfile, noisefile, chainfile, "nm", seed=4, **kwargs)
    c = np.load(os.path.join(outdir2, "nm.npy"))
    np.testing.assert_array_equal(b, c)


def test_run_nmfp_no_checkpoint(inputs):
    """--no-checkpoint skips shard writes but produces identical output."""
    tmp, psrfile, noisefile, chainfile = inputs
    outdir = str(tmp / "res4")
    kwargs = dict(
        inc_cp=False, nrncomps=3, ncwfreqs=3, nsamples=4, batch_size=2,
        device="cpu", seed=5,
    )
    run_nmfp.main(psrfile, noisefile, chainfile, "nm", outdir=outdir, **kwargs)
    a = np.load(os.path.join(outdir, "nm.npy"))
    outdir2 = str(tmp / "res4b")
    run_nmfp.main(psrfile, noisefile, chainfile, "nm", outdir=outdir2,
                  checkpoint=False, **kwargs)
    b = np.load(os.path.join(outdir2, "nm.npy"))
    np.testing.assert_array_equal(a, b)
    assert not os.path.exists(os.path.join(outdir2, ".nm.batches"))


def test_run_fp_ecorr_kernel(inputs, tmp_path):
    """Fp CLI with --ecorr_kernel: ECORR as block-diagonal white noise
    (the reference's unsupported case) end to end through the CLI."""
    tmp, psrfile, noisefile, _ = inputs
    with open(noisefile) as f:
        noise = json.load(f)
    psrs_noise = dict(noise)
    # per-backend ECORR keys for both synthetic backends of each pulsar
    for name in {k.rsplit("_red_noise", 1)[0] for k in noise
                 if "_red_noise_gamma" in k}:
        for b in ("BE_A", "BE_B"):
            psrs_noise[f"{name}_basis_ecorr_{b}_log10_ecorr"] = -6.5
    nf2 = str(tmp_path / "noise_ec.json")
    with open(nf2, "w") as f:
        json.dump(psrs_noise, f)
    out = str(tmp_path / "fp_ec")
    run_fp.main(psrfile, nf2, out, nfreqs=4, rn_comps=3, gwb_comps=3,
                device="cpu", ecorr_kernel=True)
    with open(out + ".json") as f:
        res = json.load(f)
    assert len(res) == 4
    assert all(np.isfinite(v) for v in res.values())


def test_run_fp_torchrun_ws2_gloo(inputs, tmp_path):
    """The Fp CLI under the production multi-GPU launch shape (2 ranks,
    gloo on CPU): frequency sharding + all-gather, rank-0 JSON output
    identical in contract to the single-process run."""
    import subprocess
    import sys

    tmp, psrfile, noisefile, _ = inputs
    repo = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))
    out = str(tmp_path / "fp_ws2")
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29881", "-m", "fastfp_amd.cli.run_fp",
         psrfile, noisefile, out, "--nfreqs", "5", "--rn_comps", "3",
         "--gwb_comps", "3", "--device", "cpu"],
        cwd=repo, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    with open(out + ".json") as f:
        res = json.load(f)
    assert len(res) == 5
    # bitwise contract: sharded == single process (fp64 determinism)
    out1 = str(tmp_path / "fp_ws1")
    run_fp.main(psrfile, noisefile, out1, nfreqs=5, rn_comps=3,
                gwb_comps=3, device="cpu")
    with open(out1 + ".json") as f:
        res1 = json.load(f)
    assert res == res1


def test_run_fe_json_output(inputs, tmp_path):
    """Fe CLI: (sky x freq) map JSON with the documented keys; single
    sky location mode too."""
    from fastfp_amd.cli import run_fe

    tmp, psrfile, noisefile, _ = inputs
    out = str(tmp_path / "feout")
    run_fe.main(psrfile, noisefile, out, nfreqs=4, nsky=6, rn_comps=3,
                gwb_comps=3, device="cpu")
    with open(out + ".json") as f:
        res = json.load(f)
    assert len(res["freqs"]) == 4 and len(res["sky"]) == 6
    fe = np.asarray(res["fe"])
    assert fe.shape == (6, 4) and np.isfinite(fe).all()
    # single-sky mode
    out1 = str(tmp_path / "feout1")
    run_fe.main(psrfile, noisefile, out1, nfreqs=3, theta=1.0, phi=2.0,
                rn_comps=3, gwb_comps=3, device="cpu")
    with open(out1 + ".json") as f:
        res1 = json.load(f)
    assert np.asarray(res1["fe"]).shape == (1, 3)


def test_fibonacci_sky_coverage():
    from fastfp_amd.cli.run_fe import fibonacci_sky

    sky = fibonacci_sky(100)
    th = np.array([t for t, _ in sky])
    ph = np.array([p for _, p in sky])
    assert ((th >= 0) & (th <= np.pi)).all()
    assert ((ph >= 0) & (ph < 2 * np.pi)).all()
    # equal-area: z = cos(theta) should be ~uniform on [-1, 1]
    z = np.sort(np.cos(th))
    assert np.abs(z - np.linspace(z[0], z[-1], 100)).max() < 0.03


def test_run_nmfe_output(inputs, tmp_path):
    """NM-Fe CLI: (nsamples, nsky, nfreqs) npy + axes metadata."""
    from fastfp_amd.cli import run_nmfe

    tmp, psrfile, noisefile, chainfile = inputs
    run_nmfe.main(psrfile, noisefile, chainfile, "nmfe", inc_cp=True,
                  nrncomps=3, ngwbcomps=3, ncwfreqs=4, nsamples=5,
                  nsky=3, outdir=str(tmp_path), device="cpu",
                  batch_size=2)
    out = np.load(tmp_path / "nmfe.npy")
    assert out.shape == (5, 3, 4)
    assert np.isfinite(out).all()
    with open(tmp_path / "nmfe.meta.json") as f:
        meta = json.load(f)
    assert len(meta["freqs"]) == 4 and len(meta["sky"]) == 3


def test_run_nmfe_torchrun_ws2_gloo(inputs, tmp_path):
    """NM-Fe CLI under the multi-rank launch (2 ranks, gloo): draw
    sharding + gather across the (draws, sky, freqs) output."""
    import subprocess
    import sys

    tmp, psrfile, noisefile, chainfile = inputs
    repo = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29884", "-m", "fastfp_amd.cli.run_nmfe",
         psrfile, noisefile, chainfile, "nmfe2", "--inc_cp",
         "--nrncomps", "3", "--ngwbcomps", "3", "--ncwfreqs", "3",
         "--nsamples", "5", "--nsky", "2", "--outdir", str(tmp_path),
         "--device", "cpu", "--batch_size", "2"],
        cwd=repo, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    out = np.load(tmp_path / "nmfe2.npy")
    assert out.shape == (5, 2, 3) and np.isfinite(out).all()


def test_run_nmfp_more_ranks_than_draws(inputs, tmp_path):
    """world_size 3 with only 2 draws: the zero-draw rank must
    participate in the padded all-gather and the output stays
    (nsamples, nfreqs)."""
    import subprocess
    import sys

    tmp, psrfile, noisefile, chainfile = inputs
    repo = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "3", "--master-addr", "127.0.0.1",
         "--master-port", "29893", "-m", "fastfp_amd.cli.run_nmfp",
         psrfile, noisefile, chainfile, "zd", "--inc_cp",
         "--nrncomps", "3", "--ngwbcomps", "3", "--ncwfreqs", "3",
         "--nsamples", "2", "--outdir", str(tmp_path),
         "--device", "cpu"],
        cwd=repo, capture_output=True, text=True, timeout=600,
    )
    assert r.returncode == 0, r.stderr[-2000:]
    out = np.load(tmp_path / "zd.npy")
    assert out.shape == (2, 3) and np.isfinite(out).all()
