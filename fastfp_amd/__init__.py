"""fastfp_amd — MI355X-native pulsar-timing-array Fp-statistic engine.

Brand-new implementation of the capabilities of ``gabefreedman/fastfp``
(plain and noise-marginalized Fp statistics) designed MI355X-first:
PyTorch-ROCm fp64 tensors, hand-written HIP/CDNA4 kernels (MFMA fp64
DGEMM, batched Cholesky, fused triangular-solve + 2x2 Fp reduction) and
``torch.distributed``/RCCL draw- and frequency-sharding over xGMI.
No JAX, no Triton, no CUDA-compat shims.

Everything is float64 end to end — the reference enforces the same
policy globally (``/root/reference/fastfp/__init__.py:3``); here fp64 is
explicit per-tensor rather than a global default.
"""

__version__ = "0.1.0"

from fastfp_amd.data import (  # noqa: F401
    PulsarData,
    get_tspan,
    load_pulsars,
    make_synthetic_pta,
    save_pulsars,
)
from fastfp_amd.bases import create_freqarray  # noqa: F401
from fastfp_amd.noise import (  # noqa: F401
    CURNContainer,
    GPEcorrContainer,
    RNContainer,
)
from fastfp_amd.model import (  # noqa: F401
    PTAModel,
    get_mats_fp,
    get_mats_nmfp,
    initialize_pta,
)
from fastfp_amd.xcy import get_xCy  # noqa: F401
from fastfp_amd.fpstat import FastFp, compute_Fp  # noqa: F401
from fastfp_amd.festat import FastFe, NMFe, compute_Fe  # noqa: F401
from fastfp_amd.nmfp import NMFp  # noqa: F401
from fastfp_amd.engine import FpEngine  # noqa: F401

# reference-compatible aliases (the reference exports these exact names)
NMFP = NMFp
RN_container = RNContainer
CURN_container = CURNContainer
GPEcorr_container = GPEcorrContainer
