"""Cross-component kernel fusion (MI355X-specific optimizations).

These components have no analogue in the reference — they exist because
on MI355X the scalar-preheating hot loop is HBM-bandwidth bound, and the
reference's structure (separate stencil pass + separate energy
reduction, examples/scalar_preheating.py:258-271) re-reads f and lap_f
from HBM each stage.  :class:`FusedLaplacianReduction` runs the
Laplacian stencil and the energy reductions in one pass.

Numerics are identical to the unfused path (same stencil coefficients,
same per-site expressions; only the reduction's accumulation grouping
changes, which for fp64 sums is within rounding).  The CPU path simply
composes :class:`~pystella_amd.FiniteDifferencer` +
:class:`~pystella_amd.Reduction` (and is the oracle for the GPU tests).
"""

from __future__ import annotations

import numpy as np
import torch

from pystella_amd.reduction import Reduction

__all__ = ["FusedLaplacianReduction"]


class FusedLaplacianReduction(Reduction):
    """Computes ``lap_f = ∇² f`` (all outer components) *and* the given
    reductions — whose expressions may reference both ``f`` and the
    freshly computed ``lap_f`` — in one fused GPU pass.

    Call semantics combine ``derivs(fx=f, lap=lap_f)`` (including the
    halo exchange) followed by ``Reduction.__call__``.

    :arg derivs: a :class:`~pystella_amd.FiniteDifferencer` (supplies
        dx, halo and the CPU oracle path).
    :arg f_name/lap_name: names of the stencil field and its Laplacian
        in the reduction expressions.
    """

    def __init__(self, decomp, input, derivs, f_name="f", lap_name="lap_f",
                 **kwargs):
        super().__init__(decomp, input, **kwargs)
        self.derivs = derivs
        self.f_name = f_name
        self.lap_name = lap_name
        self._fused_kernel = None

    def __call__(self, queue=None, filter_args=False, **kwargs):
        f = kwargs[self.f_name]
        lap = kwargs[self.lap_name]
        self.decomp.share_halos(f)
        if not (isinstance(f, torch.Tensor) and f.is_cuda):
            # CPU oracle: unfused compose
            self.derivs.decomp = self.decomp
            from itertools import product
            for s in product(*[range(n) for n in f.shape[:-3]]):
                self.derivs._apply_lap_cpu(f[s], lap[s])
            return super().__call__(**kwargs)

        rank_shape = self._infer_shapes(kwargs)
        if self._fused_kernel is None or \
                self._fused_kernel.rank_shape != rank_shape:
            from pystella_amd.backend.hip import get_lap_reduction_kernel
            nf = int(np.prod(f.shape[:-3])) if f.dim() > 3 else 1
            self._fused_kernel = get_lap_reduction_kernel(
                [(expr, op) for _, _, expr, op in self.flat],
                self.field_args, sorted(self.scalar_names),
                self.halo_shape, rank_shape, self.derivs.dx, nf,
                self.f_name, self.lap_name)
        local = self._fused_kernel(kwargs)
        return self._combine(local, rank_shape)
