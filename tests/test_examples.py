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
