"""Geometric multigrid solvers (FAS and linear MG).

Analogue of reference pystella/multigrid/__init__.py:55-493.

Status: implemented in this round incrementally — cycle builders and the
relaxation/transfer kernels live in ``relax.py`` / ``transfer.py``.
"""

from pystella_amd.multigrid.transfer import (  # noqa: F401
    FullWeighting, Injection, LinearInterpolation, CubicInterpolation,
)
from pystella_amd.multigrid.relax import (  # noqa: F401
    JacobiIterator, NewtonIterator,
)


def mu_cycle(level, mu, nu1, nu2, max_levels):
    """Recursive μ-cycle schedule (reference multigrid/__init__.py:55-82):
    returns a list of (level, nu) smoothing events."""
    events = [(level, nu1)]
    if level + 1 < max_levels:
        for _ in range(mu):
            events += mu_cycle(level + 1, mu, nu1, nu2, max_levels)
    events += [(level, nu2)]
    return events


def v_cycle(nu1, nu2, max_levels):
    return mu_cycle(0, 1, nu1, nu2, max_levels)


def w_cycle(nu1, nu2, max_levels):
    return mu_cycle(0, 2, nu1, nu2, max_levels)


def f_cycle(nu1, nu2, max_levels):
    """F-cycle: progressively deeper V-cycles
    (reference multigrid/__init__.py:140-166)."""
    events = []
    for depth in range(max_levels - 1, 0, -1):
        events += [(lvl, nu1) for lvl in range(depth)]
        events += mu_cycle(depth, 1, nu1, nu2, max_levels)[1:]
    return events


from pystella_amd.multigrid.solver import (  # noqa: F401,E402
    FullApproximationScheme, MultiGridSolver,
)
