"""Binned isotropic power spectra and gravitational-wave abundances.

Analogue of reference pystella/fourier/spectra.py:29-419.  Binning
(``bin = round(|k|/Δk)``), r2c double-count weights
(spectra.py:81-86, 113-119), normalization and GW formulas match the
reference.  The per-mode accumulation runs as torch ops (scatter-add on
GPU) followed by one packed all-reduce.
"""

from __future__ import annotations

import numpy as np
import torch

__all__ = ["PowerSpectra"]


class PowerSpectra:
    def __init__(self, decomp, fft, dk, volume, **kwargs):
        self.decomp = decomp
        self.fft = fft
        self.grid_shape = fft.grid_shape
        self.kshape = fft.shape(True)
        self.dk = dk
        self.bin_width = kwargs.pop("bin_width", min(dk))

        d3x = volume / np.prod(self.grid_shape)
        self.norm = (1 / 2 / np.pi**2 / volume) * d3x**2

        sub_k = [fft.sub_k[n].cpu().numpy()
                 for n in ("momenta_x", "momenta_y", "momenta_z")]
        kvecs = np.meshgrid(*sub_k, indexing="ij", sparse=False)
        kmags = np.sqrt(sum((dki * ki)**2
                            for dki, ki in zip(self.dk, kvecs)))

        if fft.is_real:
            counts = 2. * np.ones_like(kmags)
            counts[kvecs[2] == 0] = 1.
            counts[kvecs[2] == self.grid_shape[-1] // 2] = 1.
        else:
            counts = 1. * np.ones_like(kmags)

        max_k = float(self.decomp.allreduce(np.max(kmags), op="max"))
        self.num_bins = int(max_k / self.bin_width + .5) + 1
        bins = np.arange(-.5, self.num_bins + .5) * self.bin_width

        sub_bin_counts = np.histogram(kmags, weights=counts, bins=bins)[0]
        self.bin_counts = self.decomp.allreduce(sub_bin_counts)

        dev = fft.fk.device
        self._kmags = torch.as_tensor(kmags, device=dev)
        self._counts = torch.as_tensor(counts, device=dev)
        self._bin_idx = torch.as_tensor(
            np.round(kmags / self.bin_width).astype(np.int64),
            device=dev).reshape(-1).clamp_(0, self.num_bins - 1)
        self._wbase_cache = {}
        self._bidx32 = None

    def bin_power(self, fk, queue=None, k_power=3, allocator=None):
        """Unnormalized binned |f_k|² k^n with r2c double-count weights
        (reference spectra.py:140-176).

        GPU path: LDS-binned HIP kernel (torch index_add_ serializes on
        the ~500 global bins and is ~500× slower at 512³)."""
        if (isinstance(fk, torch.Tensor) and fk.is_cuda
                and fk.dtype == torch.complex128 and fk.is_contiguous()
                and self.num_bins <= 4096):
            from pystella_amd.backend.hip import spectra_bin
            wb = self._wbase_cache.get(k_power)
            if wb is None:
                wb = (self._counts * self._kmags ** k_power
                      ).reshape(-1).to(torch.float64).contiguous()
                self._wbase_cache[k_power] = wb
                self._bidx32 = self._bin_idx.to(torch.int32).contiguous()
            hist = spectra_bin(fk.reshape(-1), wb, self._bidx32,
                               self.num_bins).cpu().numpy()
        else:
            w = (self._counts * self._kmags ** k_power
                 * torch.abs(fk) ** 2).reshape(-1)
            hist = torch.zeros(self.num_bins, dtype=torch.float64,
                               device=w.device)
            hist.index_add_(0, self._bin_idx, w.to(torch.float64))
            hist = hist.cpu().numpy()
        hist = self.decomp.allreduce(hist)
        return hist / self.bin_counts

    def __call__(self, fx, queue=None, k_power=3, allocator=None):
        """Δ²_f(k): FFT then bin_power per outer component
        (reference spectra.py:177-226)."""
        outer_shape = tuple(fx.shape[:-3])
        from itertools import product
        slices = list(product(*[range(n) for n in outer_shape]))
        result = np.zeros(outer_shape + (self.num_bins,))
        for s in slices:
            fk = self.fft.dft(fx[s])
            result[s] = self.bin_power(fk, k_power=k_power)
        return self.norm * result

    def polarization(self, vector, projector, queue=None, k_power=3,
                     allocator=None):
        cdtype = self.fft.fk.dtype
        dev = self.fft.fk.device
        vec_k = torch.empty((3,) + tuple(self.kshape), dtype=cdtype,
                            device=dev)
        plus, minus = vec_k[0], vec_k[1]
        outer_shape = tuple(vector.shape[:-4])
        from itertools import product
        slices = list(product(*[range(n) for n in outer_shape]))
        result = np.zeros(outer_shape + (2, self.num_bins))
        for s in slices:
            for mu in range(3):
                self.fft.dft(vector[s][mu], vec_k[mu])
            projector.vec_to_pol(plus=plus, minus=minus, vector=vec_k)
            result[s][0] = self.bin_power(plus, k_power=k_power)
            result[s][1] = self.bin_power(minus, k_power=k_power)
        return self.norm * result

    def vector_decomposition(self, vector, projector, queue=None, k_power=3,
                             allocator=None):
        cdtype = self.fft.fk.dtype
        dev = self.fft.fk.device
        vec_k = torch.empty((3,) + tuple(self.kshape), dtype=cdtype,
                            device=dev)
        plus, minus, lng = vec_k[0], vec_k[1], vec_k[2]
        outer_shape = tuple(vector.shape[:-4])
        from itertools import product
        slices = list(product(*[range(n) for n in outer_shape]))
        result = np.zeros(outer_shape + (3, self.num_bins))
        for s in slices:
            for mu in range(3):
                self.fft.dft(vector[s][mu], vec_k[mu])
            projector.decompose_vector(vector=vec_k, plus=plus, minus=minus,
                                       lng=lng, times_abs_k=True)
            result[s][0] = self.bin_power(plus, k_power=k_power)
            result[s][1] = self.bin_power(minus, k_power=k_power)
            result[s][2] = self.bin_power(lng, k_power=k_power)
        return self.norm * result

    def gw(self, hij, projector, hubble, queue=None, k_power=3,
           allocator=None):
        """Δ²_h(k) of transverse-traceless GW (reference
        spectra.py:322-370)."""
        from pystella_amd.sectors import tensor_index as tid
        cdtype = self.fft.fk.dtype
        dev = self.fft.fk.device
        hij_k = torch.empty((6,) + tuple(self.kshape), dtype=cdtype,
                            device=dev)
        for mu in range(6):
            self.fft.dft(hij[mu], hij_k[mu])
        projector.transverse_traceless(hij=hij_k)
        gw_spec = [self.bin_power(hij_k[mu], k_power=k_power)
                   for mu in range(6)]
        gw_tot = sum(gw_spec[tid(i, j)]
                     for i in range(1, 4) for j in range(1, 4))
        return self.norm / 12 / hubble**2 * gw_tot

    def gw_polarization(self, hij, projector, hubble, queue=None, k_power=3,
                        allocator=None):
        cdtype = self.fft.fk.dtype
        dev = self.fft.fk.device
        hij_k = torch.empty((6,) + tuple(self.kshape), dtype=cdtype,
                            device=dev)
        plus, minus = hij_k[0], hij_k[1]
        for mu in range(6):
            self.fft.dft(hij[mu], hij_k[mu])
        projector.tensor_to_pol(plus=plus, minus=minus, hij=hij_k)
        result = np.zeros((2, self.num_bins))
        result[0] = self.bin_power(plus, k_power=k_power)
        result[1] = self.bin_power(minus, k_power=k_power)
        return self.norm / 12 / hubble**2 * result
