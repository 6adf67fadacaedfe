"""Physical constants used throughout the Fp-statistic engine.

Mirrors the constant surface of the reference implementation
(``/root/reference/fastfp/constants.py:7-9``): Julian year in seconds,
day in seconds, and the reference frequency ``fyr = 1/yr`` used by the
power-law red-noise PSD.
"""

import scipy.constants as sc

#: one Julian year in seconds
yr = sc.Julian_year

#: one day in seconds
day = sc.day

#: reference frequency (1/yr) for power-law PSDs, Hz
fyr = 1.0 / yr
