"""Spectral-collocation derivatives: FFT → ik-multiply → iFFT.

Analogue of reference pystella/fourier/derivs.py:28-205.  The k-space
multiplies are trivially bandwidth-bound torch ops fused per call; the
FFTs are rocFFT (GPU) / FFTW-class (CPU) through torch.fft.
"""

from __future__ import annotations

import numpy as np
import torch

__all__ = ["SpectralCollocator"]


class SpectralCollocator:
    """Exact spectral derivatives with the same call interface as
    :class:`~pystella_amd.FiniteDifferencer`."""

    def __init__(self, fft, dk):
        self.fft = fft
        self.grid_size = int(np.prod(fft.grid_shape))

        self.k1 = []   # Nyquist- and zero-zeroed (first derivatives)
        self.k2 = []   # full magnitudes (Laplacian)
        for mu, name in enumerate(("momenta_x", "momenta_y", "momenta_z")):
            kk = fft.sub_k[name].cpu().numpy().astype(int)
            kk_mu = dk[mu] * kk.astype(np.float64)
            self.k2.append(torch.as_tensor(
                kk_mu.copy(), device=fft.fk.device))
            kk_mu = kk_mu.copy()
            kk_mu[np.abs(kk) == fft.grid_shape[mu] // 2] = 0.
            kk_mu[kk == 0] = 0.
            self.k1.append(torch.as_tensor(kk_mu, device=fft.fk.device))

        shape = (-1, 1, 1), (1, -1, 1), (1, 1, -1)
        self.k1 = [k.view(s) for k, s in zip(self.k1, shape)]
        self.k2 = [k.view(s) for k, s in zip(self.k2, shape)]

    def _pd(self, fk, mu):
        return 1j * self.k1[mu] * fk

    def _lap(self, fk):
        kmag_sq = (self.k2[0]**2 + self.k2[1]**2 + self.k2[2]**2)
        return -kmag_sq * fk

    def __call__(self, queue=None, fx=None, *, lap=None, pdx=None, pdy=None,
                 pdz=None, grd=None, allocator=None):
        if fx is None and isinstance(queue, torch.Tensor):
            fx = queue
            queue = None
        if grd is not None:
            if isinstance(grd, (tuple, list)):
                pdx, pdy, pdz = grd
            else:
                pdx = grd[..., 0, :, :, :]
                pdy = grd[..., 1, :, :, :]
                pdz = grd[..., 2, :, :, :]

        from itertools import product
        slices = list(product(*[range(n) for n in fx.shape[:-3]]))
        inv_n = 1.0 / self.grid_size
        for s in slices:
            fk = self.fft.dft(fx[s]).clone() * inv_n
            if lap is not None:
                self.fft.idft(self._lap(fk), lap[s])
            if pdx is not None:
                self.fft.idft(self._pd(fk, 0), pdx[s])
            if pdy is not None:
                self.fft.idft(self._pd(fk, 1), pdy[s])
            if pdz is not None:
                self.fft.idft(self._pd(fk, 2), pdz[s])

    def divergence(self, queue=None, vec=None, div=None, allocator=None):
        if vec is None and isinstance(queue, torch.Tensor):
            vec = queue
            queue = None
        from itertools import product
        slices = list(product(*[range(n) for n in vec.shape[:-4]]))
        inv_n = 1.0 / self.grid_size
        for s in slices:
            div_k = None
            for mu in range(3):
                fk = self.fft.dft(vec[s][mu])
                term = self._pd(fk, mu) * inv_n
                div_k = term if div_k is None else div_k + term
            self.fft.idft(div_k, div[s])
