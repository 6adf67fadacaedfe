"""Sky-coherent Fe map entry point (no reference counterpart — the
reference's to-do): thin wrapper over :mod:`fastfp_amd.cli.run_fe`.

    python examples/run_fe.py psrs.npz noise.json fe_out --nsky 96
"""

import os
import sys

sys.path.insert(0, os.path.abspath(os.path.join(os.path.dirname(__file__), "..")))

from fastfp_amd.cli.run_fe import cli  # noqa: E402

if __name__ == "__main__":
    cli()
