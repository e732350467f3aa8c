"""Inter-grid transfer operators (restriction / prolongation).

Analogue of reference pystella/multigrid/transfer.py:40-264.
Implemented as tensor-product stencils; GPU kernels in csrc/transfer.hip.

Status: full implementation arrives with the multigrid milestone.
"""


class RestrictionBase:
    def __init__(self, *a, **kw):
        raise NotImplementedError("multigrid transfers: in progress")


class FullWeighting(RestrictionBase):
    pass


class Injection(RestrictionBase):
    pass


class InterpolationBase:
    def __init__(self, *a, **kw):
        raise NotImplementedError("multigrid transfers: in progress")


class LinearInterpolation(InterpolationBase):
    pass


class CubicInterpolation(InterpolationBase):
    pass
