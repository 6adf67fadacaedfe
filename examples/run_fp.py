"""Reference-layout entry point (the reference ships
``examples/run_fp.py``): thin wrapper over :mod:`fastfp_amd.cli.run_fp`.

    python examples/run_fp.py psrs.npz noise.json out
"""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from fastfp_amd.cli.run_fp import cli  # noqa: E402

if __name__ == "__main__":
    cli()
