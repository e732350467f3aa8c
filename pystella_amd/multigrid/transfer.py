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

from itertools import product

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
        acc = None
        for (a, ca), (b, cb), (c, cc) in product(
                self.coefs.items(), self.coefs.items(),
                self.coefs.items()):
            w = ca * cb * cc
            sl = (Ellipsis,
                  slice(h + a, h + a + 2 * n2[0], 2),
                  slice(h + b, h + b + 2 * n2[1], 2),
                  slice(h + c, h + c + 2 * n2[2], 2))
            term = w * f1[sl]
            acc = term if acc is None else acc + term
        interior = (Ellipsis,) + tuple(slice(h, h + n) for n in n2)
        if self.correct:
            f2[interior] -= acc
        else:
            f2[interior] = acc
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

    def __call__(self, queue=None, f1=None, f2=None):
        h = self.h
        n2 = tuple(s - 2 * h for s in f2.shape[-3:])
        for px, py, pz in product((0, 1), repeat=3):
            cx = self._axis_coefs(px)
            cy = self._axis_coefs(py)
            cz = self._axis_coefs(pz)
            acc = None
            for (a, ca), (b, cb), (c, cc) in product(
                    cx.items(), cy.items(), cz.items()):
                w = ca * cb * cc
                sl = (Ellipsis,
                      slice(h + a, h + a + n2[0]),
                      slice(h + b, h + b + n2[1]),
                      slice(h + c, h + c + n2[2]))
                term = w * f2[sl]
                acc = term if acc is None else acc + term
            out = (Ellipsis,
                   slice(h + px, h + 2 * n2[0], 2),
                   slice(h + py, h + 2 * n2[1], 2),
                   slice(h + pz, h + 2 * n2[2], 2))
            if self.correct:
                f1[out] += acc
            else:
                f1[out] = acc
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
