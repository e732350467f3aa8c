"""Fourier-space machinery: FFTs, spectra, projectors, field init.

Analogue of reference pystella/fourier/.  Single-GPU transforms go
through torch.fft (rocFFT on ROCm); the distributed path is a custom
pencil FFT with RCCL all-to-all transposes (reference used
mpi4py-fft/FFTW on the host: pystella/fourier/dft.py:352-427).
"""

from pystella_amd.fourier.dft import (  # noqa: F401
    DFT, BaseDFT, fftfreq, get_sliced_momenta, pDFT, pyclDFT)
from pystella_amd.fourier.derivs import SpectralCollocator  # noqa: F401
from pystella_amd.fourier.poisson import SpectralPoissonSolver  # noqa: F401
from pystella_amd.fourier.projectors import Projector  # noqa: F401
from pystella_amd.fourier.spectra import PowerSpectra  # noqa: F401
from pystella_amd.fourier.rayleigh import RayleighGenerator  # noqa: F401

import numpy as np


def get_real_dtype_with_matching_prec(dtype):
    dtype = np.dtype(dtype)
    return np.dtype("float32") if dtype.itemsize in (4, 8) and \
        dtype in (np.dtype("float32"), np.dtype("complex64")) \
        else np.dtype("float64")


def get_complex_dtype_with_matching_prec(dtype):
    dtype = np.dtype(dtype)
    if dtype in (np.dtype("float32"), np.dtype("complex64")):
        return np.dtype("complex64")
    return np.dtype("complex128")


def gDFT(*args, **kwargs):
    """Deprecated name kept for API parity (reference
    fourier/dft.py:509 aliases its clFFT backend); here every
    single-rank transform already runs the GPU path."""
    from warnings import warn
    warn("gDFT is deprecated; use DFT (or pyclDFT).",
         DeprecationWarning, stacklevel=2)
    return pyclDFT(*args, **kwargs)


__all__ = [
    "DFT", "BaseDFT", "fftfreq", "get_sliced_momenta", "pDFT",
    "pyclDFT", "gDFT", "RayleighGenerator", "Projector",
    "PowerSpectra", "SpectralCollocator", "SpectralPoissonSolver",
    "get_real_dtype_with_matching_prec",
    "get_complex_dtype_with_matching_prec",
]
