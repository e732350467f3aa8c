"""Inter-grid transfer operators (restriction / prolongation).

Analogue of reference pystella/multigrid/transfer.py:40-264.  The
tensor-product transfer stencils are implemented as strided tensor ops
(every-other-point slices), which map to single strided-copy kernels on
GPU; the hot multigrid cost is the smoother (see ``relax.py``), which
runs through the fused HIP elementwise path.

Conventions (match the reference): ``f1`` is the *fine*-grid array,
``f2`` the *coarse*-grid array; both are halo-padded.  ``correct=True``
variants apply the FAS corrections ``f2 -= R(f1)`` / ``f1 += I(f2)``.
"""

from __future__ import annotations


__all__ = ["RestrictionBase", "FullWeighting", "Injection",
           "InterpolationBase", "LinearInterpolation",
           "CubicInterpolation"]


class RestrictionBase:
    """Tensor-product restriction: f2_i = Σ_a c_a f1_{2i+a} per axis
    (reference transfer.py:40-103)."""

    def __init__(self, coefs, halo_shape, correct=False, **kwargs):
        self.coefs = dict(coefs)
        self.h = halo_shape
        self.correct = correct

    def __call__(self, queue=None, f1=None, f2=None):
        h = self.h
        n2 = tuple(s - 2 * h for s in f2.shape[-3:])
        # Separable tensor product: one strided pass per axis instead
        # of len(coefs)^3 full-grid passes (the reference's loopy
        # kernel reads each fine point once; the eager equivalent of
        # that is per-axis factorization).  Axis order x, y, z: the
        # stride-2 read only wastes HBM 32B-granule bandwidth on the
        # contiguous (z) axis, so that pass runs last when the
        # intermediate is already 4x smaller.
        x = f1
        items = sorted(self.coefs.items())
        for k in range(3):
            ax = x.ndim - 3 + k
            sl = [slice(None)] * x.ndim
            (a0, c0) = items[0]
            sl[ax] = slice(h + a0, h + a0 + 2 * n2[k], 2)
            acc = x[tuple(sl)].mul(c0)
            for a, c in items[1:]:
                sl[ax] = slice(h + a, h + a + 2 * n2[k], 2)
                acc.add_(x[tuple(sl)], alpha=c)
            x = acc
        interior = (Ellipsis,) + tuple(slice(h, h + n) for n in n2)
        if self.correct:
            f2[interior] -= x
        else:
            f2[interior] = x
        return f2


def FullWeighting(halo_shape=1, correct=False, **kwargs):
    """(¼, ½, ¼) per-axis full weighting
    (reference transfer.py:105-126)."""
    return RestrictionBase({-1: 0.25, 0: 0.5, 1: 0.25}, halo_shape,
                           correct, **kwargs)


def Injection(halo_shape=1, correct=False, **kwargs):
    """Direct injection f2_ijk = f1_{2i,2j,2k}
    (reference transfer.py:128-143)."""
    return RestrictionBase({0: 1.0}, halo_shape, correct, **kwargs)


class InterpolationBase:
    """Tensor-product prolongation with per-parity coefficients
    (reference transfer.py:146-206).

    Fine point m = 2i+p (parity p): even points use ``even_coefs``
    (offsets on the coarse grid), odd points ``odd_coefs`` with
    half-offsets ``(p+a)//2``.
    """

    def __init__(self, even_coefs, odd_coefs, halo_shape, correct=False,
                 **kwargs):
        self.even = dict(even_coefs)
        self.odd = dict(odd_coefs)
        self.h = halo_shape
        self.correct = correct

    def _axis_coefs(self, parity):
        # returns {coarse_offset: coef}
        coefs = self.odd if parity else self.even
        return {(parity + a) // 2: c for a, c in coefs.items()}

    def _axis_pass(self, x, k, h, n2k):
        """Double axis k (coarse interior+halo extent -> fine interior
        2*n2k) by the parity-dependent tensor-product rule."""
        import torch
        ax = x.ndim - 3 + k
        halves = []
        for p in (0, 1):
            items = sorted(self._axis_coefs(p).items())
            sl = [slice(None)] * x.ndim
            (a0, c0) = items[0]
            sl[ax] = slice(h + a0, h + a0 + n2k)
            acc = x[tuple(sl)].mul(c0)
            for a, c in items[1:]:
                sl[ax] = slice(h + a, h + a + n2k)
                acc.add_(x[tuple(sl)], alpha=c)
            halves.append(acc)
        if ax == x.ndim - 1:
            # contiguous axis: interleave via stack+reshape so the
            # write is a contiguous stream (a stride-2 scatter on the
            # fastest axis would double the 32B-granule write traffic)
            out = torch.stack(halves, dim=x.ndim)
            return out.reshape(*x.shape[:-1], 2 * n2k)
        shape = list(halves[0].shape)
        shape[ax] = 2 * n2k
        out = torch.empty(shape, dtype=x.dtype, device=x.device)
        for p in (0, 1):
            sl = [slice(None)] * x.ndim
            sl[ax] = slice(p, 2 * n2k, 2)
            out[tuple(sl)] = halves[p]
        return out

    def __call__(self, queue=None, f1=None, f2=None):
        h = self.h
        n2 = tuple(s - 2 * h for s in f2.shape[-3:])
        # Separable per-axis doubling (3 passes) instead of 8 parity
        # cases x len(coefs)^3 full-grid passes.  The contiguous axis
        # is doubled FIRST, while the intermediate is smallest.
        x = f2
        for k in (2, 1, 0):
            x = self._axis_pass(x, k, h, n2[k])
        out = (Ellipsis,) + tuple(slice(h, h + 2 * n) for n in n2)
        if self.correct:
            f1[out] += x
        else:
            f1[out] = x
        return f1


def LinearInterpolation(halo_shape=1, correct=False, **kwargs):
    """Linear prolongation (reference transfer.py:208-232)."""
    return InterpolationBase({0: 1.0}, {-1: 0.5, 1: 0.5}, halo_shape,
                             correct, **kwargs)


def CubicInterpolation(halo_shape=2, correct=False, **kwargs):
    """Cubic prolongation; needs halo ≥ 2
    (reference transfer.py:234-264)."""
    if halo_shape < 2:
        raise ValueError("CubicInterpolation requires halo_shape >= 2")
    return InterpolationBase(
        {0: 1.0},
        {-3: -1 / 16, -1: 9 / 16, 1: 9 / 16, 3: -1 / 16},
        halo_shape, correct, **kwargs)
