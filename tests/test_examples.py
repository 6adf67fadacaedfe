"""The examples must stay runnable (CPU, small sizes)."""

import os
import subprocess
import sys

REPO = os.path.abspath(os.path.join(os.path.dirname(__file__), ".."))


def test_walkthrough_runs(tmp_path):
    out = subprocess.run(
        [sys.executable, "examples/walkthrough.py", "--npsr", "3",
         "--ntoa", "120", "--nfreqs", "30", "--outdir", str(tmp_path),
         "--device", "cpu"],
        cwd=REPO, capture_output=True, text=True, timeout=300,
    )
    assert out.returncode == 0, out.stderr[-1500:]
    assert (tmp_path / "fp_spectrum.json").exists()
    assert (tmp_path / "nmfp.npy").exists()
    assert "KS test" in out.stdout


def test_walkthrough_notebook_code_runs(tmp_path, monkeypatch):
    """The notebook's code cells execute and the recorded outputs match
    the statistics they print (chi^2 check, shapes)."""
    import contextlib
    import io
    import json
    import os

    nb_path = os.path.join(REPO, "examples", "walkthrough.ipynb")
    with open(nb_path) as f:
        nb = json.load(f)
    src = "\n".join(
        "".join(c["source"]) for c in nb["cells"] if c["cell_type"] == "code"
    )
    monkeypatch.chdir(tmp_path)
    buf = io.StringIO()
    with contextlib.redirect_stdout(buf):
        exec(src, {})
    out = buf.getvalue()
    assert "KS test vs chi2(24)" in out
    assert "vals shape: (32, 100)" in out
    assert "Fe map shape: (6, 200)" in out
