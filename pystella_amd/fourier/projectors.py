"""k-space vector/tensor algebra: longitudinal projection, ± polarization
bases, and transverse-traceless projection.

Analogue of reference pystella/fourier/projectors.py:30-464.  All
operations act on momentum-space arrays (shape ``fft.shape(True)``) with
this rank's effective momenta; they are implemented as fused complex
torch expressions (rocm kernels on GPU).  Polarization conventions and
the k_x = k_y = 0 special case match the reference
(projectors.py:123-142).
"""

from __future__ import annotations

import numpy as np
import torch

from pystella_amd.sectors import tensor_index as tid

__all__ = ["Projector"]


class Projector:
    def __init__(self, fft, effective_k, dk, dx):
        self.fft = fft

        if not callable(effective_k):
            if effective_k != 0:
                from pystella_amd.derivs import FirstCenteredDifference
                h = effective_k
                effective_k = FirstCenteredDifference(h).get_eigenvalues
            else:
                def effective_k(k, dx):  # noqa: ARG001
                    return k

        dev = fft.fk.device
        eff = []
        for mu, name in enumerate(("momenta_x", "momenta_y", "momenta_z")):
            kk = fft.sub_k[name].cpu().numpy().astype(int)
            eff_k = np.asarray(
                effective_k(dk[mu] * kk.astype(np.float64), dx[mu]),
                dtype=np.float64).copy()
            eff_k[np.abs(kk) == fft.grid_shape[mu] // 2] = 0.
            eff_k[kk == 0] = 0.
            eff.append(torch.as_tensor(eff_k, device=dev))
        shapes = (-1, 1, 1), (1, -1, 1), (1, 1, -1)
        self.eff_mom = {name: e for name, e in zip(
            ("eff_mom_x", "eff_mom_y", "eff_mom_z"), eff)}
        kx, ky, kz = (e.view(s) for e, s in zip(eff, shapes))
        self.kvec = (kx, ky, kz)

        ksq = kx**2 + ky**2 + kz**2
        self.ksq = ksq
        self.kmag = torch.sqrt(ksq)
        self.kvec_zero = ((kx.abs() < 1e-14) & (ky.abs() < 1e-14)
                          & (kz.abs() < 1e-14))

        # ± polarization basis (reference projectors.py:123-142)
        Kappa = torch.sqrt(kx**2 + ky**2)
        kx_ky_zero = (kx.abs() < 1e-10) & (ky.abs() < 1e-10)
        kz_nonzero = kz.abs() > 1e-10
        kmag_safe = torch.where(self.kmag > 0, self.kmag,
                                torch.ones_like(self.kmag))
        Kappa_safe = torch.where(Kappa > 0, Kappa, torch.ones_like(Kappa))
        s2 = 1 / np.sqrt(2)
        czero = torch.zeros((), dtype=fft.fk.dtype, device=dev)
        eps0 = torch.where(
            kx_ky_zero,
            torch.where(kz_nonzero, (s2 + 0j) * torch.ones_like(czero),
                        czero),
            ((kx * kz / kmag_safe - 1j * ky) / Kappa_safe * s2))
        eps1 = torch.where(
            kx_ky_zero,
            torch.where(kz_nonzero, (1j * s2) * torch.ones_like(czero),
                        czero),
            ((ky * kz / kmag_safe + 1j * kx) / Kappa_safe * s2))
        eps2 = torch.where(kx_ky_zero, czero,
                           (-Kappa / kmag_safe * s2) + 0j)
        self.eps = (eps0, eps1, eps2)

        khat = tuple(k / kmag_safe for k in self.kvec)
        self.khat = khat

        # contiguous per-axis effective momenta for the fused kernels
        self._eff_c = tuple(
            self.eff_mom[n].to(torch.float64).contiguous()
            for n in ("eff_mom_x", "eff_mom_y", "eff_mom_z"))
        self._kshape = tuple(len(e) for e in self._eff_c)

    # ------------------------------------------------------------------
    def _gpu_fast(self, *tensors):
        """All tensors CUDA complex128 contiguous with this projector's
        k-space trailing shape → route to the fused one-launch kernels
        (pystella_amd/backend/hip.py projector_op)."""
        for t in tensors:
            if not (isinstance(t, torch.Tensor) and t.is_cuda
                    and t.dtype == torch.complex128 and t.is_contiguous()
                    and tuple(t.shape[-3:]) == self._kshape):
                return False
        return True

    def _run_op(self, op, tensors, times_abs_k=False):
        from pystella_amd.backend.hip import projector_op
        projector_op(op, self._kshape,
                     [t.data_ptr() for t in tensors], self._eff_c,
                     times_abs_k=times_abs_k)

    # ------------------------------------------------------------------
    def _zero_where_kvec_zero(self, x):
        return torch.where(self.kvec_zero, torch.zeros_like(x), x)

    def transversify(self, queue=None, vector=None, vector_T=None):
        """v_T = v − k (k·v)/k²  (reference projectors.py:238-261)."""
        if isinstance(queue, torch.Tensor):
            vector, vector_T = queue, vector
            queue = None
        out = vector if vector_T is None else vector_T
        if self._gpu_fast(vector, out):
            self._run_op("transversify", [vector, out])
            return out
        div = sum(self.kvec[mu] * vector[mu] for mu in range(3))
        ksq_safe = torch.where(self.ksq > 0, self.ksq,
                               torch.ones_like(self.ksq))
        res = [self._zero_where_kvec_zero(
            vector[mu] - self.kvec[mu] / ksq_safe * div) for mu in range(3)]
        for mu in range(3):
            out[mu].copy_(res[mu])
        return out

    def vec_to_pol(self, queue=None, plus=None, minus=None, vector=None):
        if isinstance(queue, torch.Tensor):
            plus, minus, vector = queue, plus, minus
            queue = None
        if self._gpu_fast(vector, plus, minus):
            self._run_op("vec_to_pol", [vector, plus, minus])
            return plus, minus
        p = sum(vector[mu] * self.eps[mu].conj() for mu in range(3))
        m = sum(vector[mu] * self.eps[mu] for mu in range(3))
        plus.copy_(p)
        minus.copy_(m)
        return plus, minus

    def pol_to_vec(self, queue=None, plus=None, minus=None, vector=None):
        if isinstance(queue, torch.Tensor):
            queue, plus, minus, vector = None, queue, plus, minus
        if self._gpu_fast(plus, minus, vector):
            self._run_op("pol_to_vec", [plus, minus, vector])
            return vector
        res = [plus * self.eps[mu] + minus * self.eps[mu].conj()
               for mu in range(3)]
        for mu in range(3):
            vector[mu].copy_(res[mu])
        return vector

    def decompose_vector(self, queue=None, vector=None, plus=None,
                         minus=None, lng=None, times_abs_k=False):
        """Full helicity decomposition of a vector field
        (reference projectors.py:313-350)."""
        if isinstance(queue, torch.Tensor):
            queue, vector, plus, minus, lng = \
                None, queue, vector, plus, minus
        if self._gpu_fast(vector, plus, minus, lng):
            self._run_op("decompose_vector", [vector, plus, minus, lng],
                         times_abs_k=times_abs_k)
            return plus, minus, lng
        # compute the divergence BEFORE vec_to_pol stores: plus/minus
        # are routinely views of `vector` itself (the reference passes
        # vec_k[0:2] as outputs, spectra.py:300-303, and its
        # one-kernel semantics read every input before writing)
        div = sum(self.kvec[mu] * vector[mu] for mu in range(3))
        self.vec_to_pol(plus=plus, minus=minus, vector=vector)
        ksq_safe = torch.where(self.ksq > 0, self.ksq,
                               torch.ones_like(self.ksq))
        if times_abs_k:
            val = -div / torch.sqrt(ksq_safe) * 1j
        else:
            val = -div / ksq_safe * 1j
        lng.copy_(self._zero_where_kvec_zero(val))
        return plus, minus, lng

    def decomp_to_vec(self, queue=None, plus=None, minus=None, lng=None,
                      vector=None, *, times_abs_k=False):
        if isinstance(queue, torch.Tensor):
            queue, plus, minus, lng, vector = \
                None, queue, plus, minus, lng
        if self._gpu_fast(plus, minus, lng, vector):
            self._run_op("decomp_to_vec", [plus, minus, lng, vector],
                         times_abs_k=times_abs_k)
            return vector
        kmag_safe = torch.where(self.kmag > 0, self.kmag,
                                torch.ones_like(self.kmag))
        res = []
        for mu in range(3):
            v = plus * self.eps[mu] + minus * self.eps[mu].conj()
            if times_abs_k:
                extra = 1j * self.kvec[mu] * lng
            else:
                extra = 1j * self.kvec[mu] / kmag_safe * lng
            res.append(v + self._zero_where_kvec_zero(extra))
        for mu in range(3):
            vector[mu].copy_(res[mu])
        return vector

    def transverse_traceless(self, queue=None, hij=None, hij_TT=None):
        """h_ij → (P_ac P_db − ½ P_ab P_cd) h_cd
        (reference projectors.py:388-411).

        On the GPU the per-site 3×3 contractions run on the CDNA4
        matrix cores (f64 MFMA, csrc/tt_mfma.hip); the torch path below
        is the CPU/oracle implementation."""
        if isinstance(queue, torch.Tensor):
            hij, hij_TT = queue, hij
            queue = None
        out = hij if hij_TT is None else hij_TT
        if (isinstance(hij, torch.Tensor) and hij.is_cuda
                and hij.dtype == torch.complex128
                and hij.is_contiguous() and out.is_contiguous()):
            from pystella_amd.backend.hip import _stream, ext
            kshape = hij.shape[-3:]
            vol = int(np.prod(kshape))
            kx = self.eff_mom["eff_mom_x"].contiguous()
            ky = self.eff_mom["eff_mom_y"].contiguous()
            kz = self.eff_mom["eff_mom_z"].contiguous()
            ext().tt_project(hij.data_ptr(), out.data_ptr(),
                             kx.data_ptr(), ky.data_ptr(), kz.data_ptr(),
                             int(kshape[1]), int(kshape[2]), vol,
                             _stream())
            return out
        khat = self.khat

        def P(a, b):
            delta = 1.0 if a == b else 0.0
            return delta - khat[a - 1] * khat[b - 1]

        res = []
        for a in range(1, 4):
            for b in range(a, 4):
                acc = 0
                for c in range(1, 4):
                    for d in range(1, 4):
                        acc = acc + (P(a, c) * P(d, b)
                                     - P(a, b) * P(c, d) / 2) * hij[tid(c, d)]
                res.append(self._zero_where_kvec_zero(acc))
        for i, r in enumerate(res):
            out[i].copy_(r)
        return out

    def tensor_to_pol(self, queue=None, plus=None, minus=None, hij=None):
        if isinstance(queue, torch.Tensor):
            queue, plus, minus, hij = None, queue, plus, minus
        if self._gpu_fast(hij, plus, minus):
            self._run_op("tensor_to_pol", [hij, plus, minus])
            return plus, minus
        p = 0
        m = 0
        for c in range(1, 4):
            for d in range(1, 4):
                p = p + hij[tid(c, d)] * self.eps[c - 1].conj() \
                    * self.eps[d - 1].conj()
                m = m + hij[tid(c, d)] * self.eps[c - 1] * self.eps[d - 1]
        plus.copy_(p)
        minus.copy_(m)
        return plus, minus

    def pol_to_tensor(self, queue=None, plus=None, minus=None, hij=None):
        if isinstance(queue, torch.Tensor):
            queue, plus, minus, hij = None, queue, plus, minus
        if self._gpu_fast(plus, minus, hij):
            self._run_op("pol_to_tensor", [plus, minus, hij])
            return hij
        res = []
        for a in range(1, 4):
            for b in range(a, 4):
                res.append(plus * self.eps[a - 1] * self.eps[b - 1]
                           + minus * self.eps[a - 1].conj()
                           * self.eps[b - 1].conj())
        for i, r in enumerate(res):
            hij[i].copy_(r)
        return hij
