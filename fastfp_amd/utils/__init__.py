"""Module-path parity with the reference's ``fastfp.utils``
(``/root/reference/fastfp/utils.py``): the same callables importable
from the same place."""

from fastfp_amd.xcy import get_xCy, get_xCy_blockdiag  # noqa: F401
from fastfp_amd.model import (  # noqa: F401
    get_mats_fp,
    get_mats_nmfp,
    initialize_pta,
)
