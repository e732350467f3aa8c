"""Fourier-space Poisson solver: ∇²f − m²f = ρ.

Analogue of reference pystella/fourier/poisson.py:33-125.  The solve is
implemented relative to the eigenvalues of a chosen second-difference
stencil (``effective_k(k, dx)``), so solutions are consistent with the
finite-difference Laplacian of the same order.
"""

from __future__ import annotations

import numpy as np
import torch

__all__ = ["SpectralPoissonSolver"]


class SpectralPoissonSolver:
    def __init__(self, fft, dk, dx, effective_k):
        self.fft = fft
        self.grid_size = int(np.prod(fft.grid_shape))
        dev = fft.fk.device

        ks = []
        for mu, name in enumerate(("momenta_x", "momenta_y", "momenta_z")):
            kk = fft.sub_k[name].cpu().numpy().astype(int)
            kk_mu = effective_k(dk[mu] * kk.astype(np.float64), dx[mu])
            ks.append(torch.as_tensor(np.asarray(kk_mu, dtype=np.float64),
                                      device=dev))
        shape = (-1, 1, 1), (1, -1, 1), (1, 1, -1)
        # eigenvalues of the per-axis second difference (≤ 0); their sum
        # is −k²_eff
        self.minus_k_squared = sum(k.view(s) for k, s in zip(ks, shape))

    def __call__(self, queue=None, fx=None, rho=None, m_squared=0,
                 allocator=None):
        if fx is None and isinstance(queue, torch.Tensor):
            fx = queue
            queue = None
        rhok = self.fft.dft(rho)
        denom = self.minus_k_squared - m_squared
        sol = torch.where(
            self.minus_k_squared < 0,
            rhok * (1.0 / self.grid_size) / denom,
            torch.zeros((), dtype=rhok.dtype, device=rhok.device))
        self.fft.idft(sol, fx)
        return fx
