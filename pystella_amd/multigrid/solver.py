"""Multigrid drivers: Full Approximation Scheme and linear MG.

Analogue of reference pystella/multigrid/__init__.py:169-493.

Status: full implementation arrives with the multigrid milestone.
"""


class FullApproximationScheme:
    def __init__(self, *a, **kw):
        raise NotImplementedError("FAS multigrid: in progress")


class MultiGridSolver:
    def __init__(self, *a, **kw):
        raise NotImplementedError("linear multigrid: in progress")
