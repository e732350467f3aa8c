"""Multigrid relaxation (smoothers): damped Jacobi / Newton iteration.

Analogue of reference pystella/multigrid/relax.py:36-373.

Status: full implementation arrives with the multigrid milestone.
"""


class RelaxationBase:
    def __init__(self, *a, **kw):
        raise NotImplementedError("multigrid relaxation: in progress")


class JacobiIterator(RelaxationBase):
    pass


class NewtonIterator(RelaxationBase):
    pass
