"""Module-path parity with the reference: ``from fastfp.fastfp import
FastFp`` maps to ``from fastfp_amd.fastfp import FastFp``."""

from fastfp_amd.fpstat import FastFp, compute_Fp  # noqa: F401
