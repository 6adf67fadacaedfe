"""Reference-layout entry point (the reference ships
``examples/run_nmfp.py``): thin wrapper over
:mod:`fastfp_amd.cli.run_nmfp`.

    python examples/run_nmfp.py psrs.npz noise.json chain.txt out --inc_cp
"""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from fastfp_amd.cli.run_nmfp import cli  # noqa: E402

if __name__ == "__main__":
    cli()
