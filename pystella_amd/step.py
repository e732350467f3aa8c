"""Explicit Runge-Kutta time steppers as fused per-stage kernels.

Analogue of reference pystella/step.py:67-853.  Each stage is one fused
elementwise kernel that evaluates the symbolic right-hand sides inline
and applies the tableau update — on GPU this becomes a single
bandwidth-bound CDNA4 kernel over all unknowns (see
``backend/codegen.py``); on CPU it runs through the torch evaluator.

Low-storage (2N) steppers allocate one *unpadded* temporary per unknown
(halo layers carry no k-state), saving HBM capacity and bandwidth
relative to the reference's padded temporaries
(reference step.py:493-518).

Tableau constants are from the published literature cited on each class
(Carpenter & Kennedy 1994; Niegemann, Diehl & Busch 2012; Williamson
1980).
"""

from __future__ import annotations

import math
import numbers

import torch

from pystella_amd.field import Field, Subscript, Variable, var
from pystella_amd.elementwise import ElementWiseMap

__all__ = [
    "Stepper", "RungeKuttaStepper", "LowStorageRKStepper",
    "RungeKutta4", "RungeKutta3SSP", "RungeKutta3Heun", "RungeKutta3Nystrom",
    "RungeKutta3Ralston", "RungeKutta2Midpoint", "RungeKutta2Heun",
    "RungeKutta2Ralston", "LowStorageRK54", "LowStorageRK144",
    "LowStorageRK134", "LowStorageRK124", "LowStorageRK3Williamson",
    "LowStorageRK3Inhomogeneous", "LowStorageRK3SSP", "all_steppers",
]


def _field_of(key):
    if isinstance(key, Field):
        return key, ()
    if isinstance(key, Subscript) and isinstance(key.aggregate, Field):
        return key.aggregate, key.index
    raise ValueError("rhs_dict keys must be Fields or subscripted Fields")


def _prepend_index(expr, q):
    """Prepend outer index ``q`` to every Field access in ``expr``
    (the classical-RK copy axis; reference step.py:202-212)."""
    from pystella_amd.field import map_expr

    def leaf(x):
        if isinstance(x, Field):
            return Subscript(x, (q,))
        if isinstance(x, Subscript) and isinstance(x.aggregate, Field):
            return Subscript(x.aggregate, (q,) + x.index)
        if isinstance(x, Subscript) and \
                isinstance(x.aggregate, Subscript):
            # map_expr rebuilds f[i] as Subscript(leaf(f), (i,)) =
            # Subscript(Subscript(f, (q,)), (i,)); flatten to f[q, i]
            inner = x.aggregate
            if isinstance(inner.aggregate, Field):
                return Subscript(inner.aggregate, inner.index + x.index)
        return x

    return map_expr(expr, leaf)


class Stepper:
    """Base time stepper; consumes a ``rhs_dict`` (or Sector(s)) mapping
    unknowns to their time derivatives (reference step.py:67-171)."""

    num_stages = None
    expected_order = None
    num_copies = None

    def __init__(self, input, dt=None, halo_shape=0, rank_shape=None,
                 **kwargs):
        from pystella_amd.sectors import Sector
        if isinstance(input, Sector):
            self.rhs_dict = dict(input.rhs_dict)
        elif isinstance(input, list):
            self.rhs_dict = {}
            for s in input:
                self.rhs_dict.update(s.rhs_dict)
        elif isinstance(input, dict):
            self.rhs_dict = dict(input)
        else:
            raise TypeError("input must be a dict, Sector, or list thereof")

        self.dt = dt
        self.halo_shape = halo_shape
        self.rank_shape = rank_shape
        self.num_unknowns = len(self.rhs_dict)
        fixed = dict(kwargs.pop("fixed_parameters", {}))
        if dt is not None:
            fixed["dt"] = dt
        self._kwargs = kwargs
        self.steps = self.make_steps(fixed_parameters=fixed, **kwargs)

    def make_steps(self, **kwargs):
        raise NotImplementedError

    def __call__(self, stage, queue=None, filter_args=None, **kwargs):
        self.steps[stage](**kwargs)


class RungeKuttaStepper(Stepper):
    """Classical multi-copy RK: unknown arrays carry a leading
    temporary-storage axis of length :attr:`num_copies`
    (reference step.py:173-239)."""

    def step_statements(self, stage, fq, dt, rhs):
        raise NotImplementedError

    def make_steps(self, fixed_parameters=None, **kwargs):
        dt = var("dt")
        steps = []
        for stage in range(self.num_stages):
            q = 0 if stage == 0 else 1
            tmp = {}
            rk_dict = {}
            for i, (key, rhs_expr) in enumerate(self.rhs_dict.items()):
                rhs_name = var(f"rhs_{i}")
                tmp[rhs_name] = _prepend_index(rhs_expr, q)
                f, outer = _field_of(key)
                fq = [Subscript(f, (c,) + outer)
                      for c in range(max(self.num_copies, 3))]
                for lhs, val in self.step_statements(
                        stage, fq, dt, rhs_name).items():
                    rk_dict[lhs] = val
            steps.append(ElementWiseMap(
                rk_dict, tmp_instructions=tmp, halo_shape=self.halo_shape,
                rank_shape=self.rank_shape, name=f"rk_stage{stage}",
                fixed_parameters=fixed_parameters, **kwargs))
        return steps


class RungeKutta4(RungeKuttaStepper):
    """Classical RK4 (reference step.py:242-265)."""
    num_stages, expected_order, num_copies = 4, 4, 3

    def step_statements(self, stage, fq, dt, rhs):
        if stage == 0:
            return {fq[1]: fq[0] + dt / 2 * rhs,
                    fq[2]: fq[0] + dt / 6 * rhs}
        if stage == 1:
            return {fq[1]: fq[0] + dt / 2 * rhs,
                    fq[2]: fq[2] + dt / 3 * rhs}
        if stage == 2:
            return {fq[1]: fq[0] + dt * rhs,
                    fq[2]: fq[2] + dt / 3 * rhs}
        return {fq[0]: fq[2] + dt / 6 * rhs}


class RungeKutta3Heun(RungeKuttaStepper):
    num_stages, expected_order, num_copies = 3, 3, 3

    def step_statements(self, stage, fq, dt, rhs):
        if stage == 0:
            return {fq[1]: fq[0] + dt / 3 * rhs,
                    fq[2]: fq[0] + dt / 4 * rhs}
        if stage == 1:
            return {fq[1]: fq[0] + dt * 2 / 3 * rhs}
        return {fq[0]: fq[2] + dt * 3 / 4 * rhs}


class RungeKutta3Nystrom(RungeKuttaStepper):
    num_stages, expected_order, num_copies = 3, 3, 3

    def step_statements(self, stage, fq, dt, rhs):
        if stage == 0:
            return {fq[1]: fq[0] + dt * 2 / 3 * rhs,
                    fq[2]: fq[0] + dt * 2 / 8 * rhs}
        if stage == 1:
            return {fq[1]: fq[0] + dt * 2 / 3 * rhs,
                    fq[2]: fq[2] + dt * 3 / 8 * rhs}
        return {fq[0]: fq[2] + dt * 3 / 8 * rhs}


class RungeKutta3Ralston(RungeKuttaStepper):
    num_stages, expected_order, num_copies = 3, 3, 3

    def step_statements(self, stage, fq, dt, rhs):
        if stage == 0:
            return {fq[1]: fq[0] + dt / 2 * rhs,
                    fq[2]: fq[0] + dt * 2 / 9 * rhs}
        if stage == 1:
            return {fq[1]: fq[0] + dt * 3 / 4 * rhs,
                    fq[2]: fq[2] + dt / 3 * rhs}
        return {fq[0]: fq[2] + dt * 4 / 9 * rhs}


class RungeKutta3SSP(RungeKuttaStepper):
    num_stages, expected_order, num_copies = 3, 3, 2

    def step_statements(self, stage, fq, dt, rhs):
        if stage == 0:
            return {fq[1]: fq[0] + dt * rhs}
        if stage == 1:
            return {fq[1]: 3 / 4 * fq[0] + 1 / 4 * fq[1] + dt / 4 * rhs}
        return {fq[0]: 1 / 3 * fq[0] + 2 / 3 * fq[1] + dt * 2 / 3 * rhs}


class RungeKutta2Midpoint(RungeKuttaStepper):
    num_stages, expected_order, num_copies = 2, 2, 2

    def step_statements(self, stage, fq, dt, rhs):
        if stage == 0:
            return {fq[1]: fq[0] + dt / 2 * rhs}
        return {fq[0]: fq[0] + dt * rhs}


class RungeKutta2Heun(RungeKuttaStepper):
    num_stages, expected_order, num_copies = 2, 2, 2

    def step_statements(self, stage, fq, dt, rhs):
        if stage == 0:
            return {fq[1]: fq[0] + dt * rhs,
                    fq[0]: fq[0] + dt / 2 * rhs}
        return {fq[0]: fq[0] + dt / 2 * rhs}


class RungeKutta2Ralston(RungeKuttaStepper):
    num_stages, expected_order, num_copies = 2, 2, 2

    def step_statements(self, stage, fq, dt, rhs):
        if stage == 0:
            return {fq[1]: fq[0] + dt * 2 / 3 * rhs,
                    fq[0]: fq[0] + dt / 4 * rhs}
        return {fq[0]: fq[0] + dt * 3 / 4 * rhs}


class LowStorageRKStepper(Stepper):
    """2N-storage RK: per unknown, one temporary array ``k``;
    per stage: ``k = A_s k + dt rhs;  f = f + B_s k``
    (reference step.py:441-529)."""

    _A: list = []
    _B: list = []
    _C: list = []

    def make_steps(self, fixed_parameters=None, **kwargs):
        dt = var("dt")
        self._unknowns = []
        for key in self.rhs_dict:
            f, outer = _field_of(key)
            self._unknowns.append((f, outer))
        self.dof_names = {f.name for f, _ in self._unknowns}

        steps = []
        for stage in range(self.num_stages):
            tmp = {}
            rk_dict = {}
            for i, (key, rhs_expr) in enumerate(self.rhs_dict.items()):
                f, outer = _field_of(key)
                k = Field(f"{f.name}_tmp", offset=0, shape=f.shape,
                          indices=f.indices, dtype=f.dtype)
                k_acc = k[outer] if outer else k
                rhs_name = var(f"rhs_{i}")
                tmp[rhs_name] = rhs_expr
                rk_dict[k_acc] = self._A[stage] * k_acc + dt * rhs_name
                rk_dict[key] = key + self._B[stage] * k_acc
            steps.append(ElementWiseMap(
                rk_dict, tmp_instructions=tmp, halo_shape=self.halo_shape,
                rank_shape=self.rank_shape, name=f"rk_stage{stage}",
                fixed_parameters=fixed_parameters, **kwargs))
        self.tmp_arrays = {}
        return steps

    def get_tmp_arrays_like(self, **kwargs):
        """Allocate the unpadded k-temporaries matching the passed
        unknown arrays (reference step.py:493-518; unpadded here)."""
        h = ((self.halo_shape,) * 3
             if isinstance(self.halo_shape, numbers.Number)
             else tuple(self.halo_shape))
        tmp_arrays = {}
        for name in self.dof_names:
            f = kwargs[name]
            tmp_name = f"{name}_tmp"
            if isinstance(f, torch.Tensor) and f.dim() >= 3:
                shape = list(f.shape)
                for d in range(3):
                    shape[-3 + d] -= 2 * h[d]
                t = torch.zeros(shape, dtype=f.dtype, device=f.device)
            elif isinstance(f, torch.Tensor):
                t = torch.zeros_like(f)
            else:
                import numpy as np
                t = np.zeros_like(f)
            tmp_arrays[tmp_name] = t
        return tmp_arrays

    def __call__(self, stage, queue=None, filter_args=None, **kwargs):
        if not self.tmp_arrays:
            self.tmp_arrays = self.get_tmp_arrays_like(**kwargs)
        return self.steps[stage](**kwargs, **self.tmp_arrays)


class LowStorageRK54(LowStorageRKStepper):
    """Carpenter & Kennedy (1994) five-stage fourth-order 2N-storage RK
    (reference step.py:531-565)."""

    num_stages, expected_order = 5, 4
    _A = [
        0,
        -567301805773 / 1357537059087,
        -2404267990393 / 2016746695238,
        -3550918686646 / 2091501179385,
        -1275806237668 / 842570457699,
    ]
    _B = [
        1432997174477 / 9575080441755,
        5161836677717 / 13612068292357,
        1720146321549 / 2090206949498,
        3134564353537 / 4481467310338,
        2277821191437 / 14882151754819,
    ]
    _C = [
        0,
        1432997174477 / 9575080441755,
        2526269341429 / 6820363962896,
        2006345519317 / 3224310063776,
        2802321613138 / 2924317926251,
    ]


class LowStorageRK144(LowStorageRKStepper):
    """Niegemann, Diehl & Busch (2012), 14-stage 4th-order, elliptic
    stability region (reference step.py:568-633)."""

    num_stages, expected_order = 14, 4
    _A = [
        0, -0.7188012108672410, -0.7785331173421570, -0.0053282796654044,
        -0.8552979934029281, -3.9564138245774565, -1.5780575380587385,
        -2.0837094552574054, -0.7483334182761610, -0.7032861106563359,
        0.0013917096117681, -0.0932075369637460, -0.9514200470875948,
        -7.1151571693922548,
    ]
    _B = [
        0.0367762454319673, 0.3136296607553959, 0.1531848691869027,
        0.0030097086818182, 0.3326293790646110, 0.2440251405350864,
        0.3718879239592277, 0.6204126221582444, 0.1524043173028741,
        0.0760894927419266, 0.0077604214040978, 0.0024647284755382,
        0.0780348340049386, 5.5059777270269628,
    ]
    _C = [
        0, 0.0367762454319673, 0.1249685262725025, 0.2446177702277698,
        0.2476149531070420, 0.2969311120382472, 0.3978149645802642,
        0.5270854589440328, 0.6981269994175695, 0.8190890835352128,
        0.8527059887098624, 0.8604711817462826, 0.8627060376969976,
        0.8734213127600976,
    ]


class LowStorageRK134(LowStorageRKStepper):
    """Niegemann et al. (2012), 13-stage 4th-order, circular stability
    region (reference step.py:634-696)."""

    num_stages, expected_order = 13, 4
    # note: the A_i of Niegemann et al. are negative; the reference
    # stores them positive (step.py:648-663) but never exercises this
    # stepper in its test matrix (step.py:849 omits it) — with positive
    # A the scheme diverges.  Signs fixed here; 4th order verified in
    # tests/test_step.py.
    _A = [
        0, -0.6160178650170565, -0.4449487060774118, -1.0952033345276178,
        -1.2256030785959187, -0.2740182222332805, -0.0411952089052647,
        -0.179708489915356, -1.1771530652064288, -0.4078831463120878,
        -0.8295636426191777, -4.789597058425229, -0.6606671432964504,
    ]
    _B = [
        0.0271990297818803, 0.1772488819905108, 0.0378528418949694,
        0.6086431830142991, 0.21543139743161, 0.2066152563885843,
        0.0415864076069797, 0.0219891884310925, 0.9893081222650993,
        0.0063199019859826, 0.3749640721105318, 1.6080235151003195,
        0.0961209123818189,
    ]
    _C = [
        0, 0.0271990297818803, 0.0952594339119365, 0.1266450286591127,
        0.1825883045699772, 0.3737511439063931, 0.5301279418422206,
        0.5704177433952291, 0.5885784947099155, 0.6160769826246714,
        0.6223252334314046, 0.6897593128753419, 0.9126827615920843,
    ]


class LowStorageRK124(LowStorageRKStepper):
    """Niegemann et al. (2012), 12-stage 4th-order, inviscid-optimized
    (reference step.py:697-756)."""

    num_stages, expected_order = 12, 4
    # A_i signs fixed relative to the reference, as for LowStorageRK134
    _A = [
        0, -0.0923311242368072, -0.9441056581158819, -4.327127324757639,
        -2.155777132902607, -0.9770727190189062, -0.7581835342571139,
        -1.79775254708255, -2.691566797270077, -4.646679896026814,
        -0.1539613783825189, -0.5943293901830616,
    ]
    _B = [
        0.0650008435125904, 0.0161459902249842, 0.5758627178358159,
        0.1649758848361671, 0.3934619494248182, 0.0443509641602719,
        0.2074504268408778, 0.6914247433015102, 0.3766646883450449,
        0.0757190350155483, 0.2027862031054088, 0.2167029365631842,
    ]
    _C = [
        0, 0.0650008435125904, 0.0796560563081853, 0.1620416710085376,
        0.2248877362907778, 0.2952293985641261, 0.3318332506149405,
        0.4094724050198658, 0.6356954475753369, 0.6806551557645497,
        0.714377371241835, 0.9032588871651854,
    ]


class LowStorageRK3Williamson(LowStorageRKStepper):
    """Williamson (1980) 3-stage 3rd-order (reference step.py:757-775)."""

    num_stages, expected_order = 3, 3
    _A = [0, -5 / 9, -153 / 128]
    _B = [1 / 3, 15 / 16, 8 / 15]
    _C = [0, 4 / 9, 15 / 32]


class LowStorageRK3Inhomogeneous(LowStorageRKStepper):
    num_stages, expected_order = 3, 3
    _A = [0, -17 / 32, -32 / 27]
    _B = [1 / 4, 8 / 9, 3 / 4]
    _C = [0, 15 / 32, 4 / 9]


def _rk3ssp_coefs():
    c2 = .924574
    z1 = math.sqrt(36 * c2**4 + 36 * c2**3 - 135 * c2**2 + 84 * c2 - 12)
    z2 = 2 * c2**2 + c2 - 2
    z3 = 12 * c2**4 - 18 * c2**3 + 18 * c2**2 - 11 * c2 + 2
    z4 = 36 * c2**4 - 36 * c2**3 + 13 * c2**2 - 8 * c2 + 4
    z5 = 69 * c2**3 - 62 * c2**2 + 28 * c2 - 8
    z6 = 34 * c2**4 - 46 * c2**3 + 34 * c2**2 - 13 * c2 + 2
    B1 = c2
    B2 = ((12 * c2 * (c2 - 1) * (3 * z2 - z1) - (3 * z2 - z1)**2)
          / (144 * c2 * (3 * c2 - 2) * (c2 - 1)**2))
    B3 = (-24 * (3 * c2 - 2) * (c2 - 1)**2
          / ((3 * z2 - z1)**2 - 12 * c2 * (c2 - 1) * (3 * z2 - z1)))
    A2 = ((-z1 * (6 * c2**2 - 4 * c2 + 1) + 3 * z3)
          / ((2 * c2 + 1) * z1 - 3 * (c2 + 2) * (2 * c2 - 1)**2))
    A3 = ((-z4 * z1 + 108 * (2 * c2 - 1) * c2**5 - 3 * (2 * c2 - 1) * z5)
          / (24 * z1 * c2 * (c2 - 1)**4 + 72 * c2 * z6
             + 72 * c2**6 * (2 * c2 - 13)))
    return ([0, A2, A3], [B1, B2, B3], [0, B1, B1 + B2 * (A2 + 1)])


_ssp_A, _ssp_B, _ssp_C = _rk3ssp_coefs()


class LowStorageRK3SSP(LowStorageRKStepper):
    """3-stage 3rd-order SSP low-storage RK
    (reference step.py:833-847)."""

    num_stages, expected_order = 3, 3
    _A, _B, _C = _ssp_A, _ssp_B, _ssp_C


all_steppers = [
    RungeKutta4, RungeKutta3SSP, RungeKutta3Heun, RungeKutta3Nystrom,
    RungeKutta3Ralston, RungeKutta2Midpoint, RungeKutta2Ralston,
    LowStorageRK54, LowStorageRK144, LowStorageRK134, LowStorageRK124,
    LowStorageRK3Williamson, LowStorageRK3Inhomogeneous, LowStorageRK3SSP,
]
