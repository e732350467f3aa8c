"""Geometric multigrid: FAS and linear MG with Jacobi/Newton smoothers.

Analogue of reference pystella/multigrid/.  Smoothing kernels run
through the hiprtc-specialized CDNA4 elementwise templates; transfer
operators are strided tensor ops (see ``transfer.py``).
"""

from pystella_amd.multigrid.transfer import (  # noqa: F401
    RestrictionBase, FullWeighting, Injection, InterpolationBase,
    LinearInterpolation, CubicInterpolation,
)
from pystella_amd.multigrid.relax import (  # noqa: F401
    RelaxationBase, JacobiIterator, NewtonIterator,
    RedBlackIterator,
)
from pystella_amd.multigrid.solver import (  # noqa: F401
    FullApproximationScheme, MultiGridSolver, mu_cycle, v_cycle, w_cycle,
    f_cycle,
)

__all__ = [
    "Injection", "FullWeighting", "LinearInterpolation",
    "CubicInterpolation", "JacobiIterator", "NewtonIterator",
    "RedBlackIterator",
    "FullApproximationScheme", "MultiGridSolver",
    "v_cycle", "w_cycle", "f_cycle",
]
