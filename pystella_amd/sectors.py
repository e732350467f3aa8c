"""Physics model definitions ("sectors") as symbolic rhs/reducer dicts.

Analogue of reference pystella/sectors.py:42-229.  A Sector supplies

* ``rhs_dict`` — the equations of motion consumed by a Stepper,
* ``reducers`` — volume-averaged quantities (energy components) consumed
  by :class:`~pystella_amd.Reduction`, and
* ``stress_tensor`` — T_{μν} components sourcing tensor perturbations.
"""

from __future__ import annotations

import numpy as np

from pystella_amd.field import DynamicField, Field, diff, var

__all__ = ["Sector", "ScalarSector", "TensorPerturbationSector",
           "tensor_index", "get_rho_and_p"]


class Sector:
    """Abstract base class (reference sectors.py:42-88)."""

    def __init__(self):
        raise NotImplementedError

    @property
    def rhs_dict(self):
        raise NotImplementedError

    @property
    def reducers(self):
        raise NotImplementedError

    def stress_tensor(self, mu, nu, drop_trace=True):
        raise NotImplementedError


class ScalarSector(Sector):
    """Scalar fields in conformal FLRW: Klein-Gordon equations of motion
    with Hubble friction and potential coupling
    (reference sectors.py:92-162).

    :arg nscalars: number of scalar fields.
    :arg f: the :class:`DynamicField`; defaults to
        ``DynamicField("f", offset="h", shape=(nscalars,))``.
    :arg potential: callable mapping the field (subscriptable) to the
        scalar potential V(f).
    """

    def __init__(self, nscalars, **kwargs):
        self.nscalars = nscalars
        self.f = kwargs.pop(
            "f", DynamicField("f", offset="h", shape=(nscalars,)))
        self.potential = kwargs.pop("potential", lambda x: 0)

    @property
    def rhs_dict(self):
        f = self.f
        H = Field("hubble", indices=[])
        a = Field("a", indices=[])
        V = self.potential(f)

        rhs = {}
        for fld in range(self.nscalars):
            rhs[f[fld]] = f.dot[fld]
            rhs[f.dot[fld]] = (f.lap[fld]
                               - 2 * H * f.dot[fld]
                               - a**2 * diff(V, f[fld]))
        return rhs

    @property
    def reducers(self):
        f = self.f
        a = var("a")
        reducers = {}
        reducers["kinetic"] = [f.dot[fld]**2 / 2 / a**2
                               for fld in range(self.nscalars)]
        reducers["potential"] = [self.potential(f)]
        reducers["gradient"] = [-f[fld] * f.lap[fld] / 2 / a**2
                                for fld in range(self.nscalars)]
        return reducers

    def stress_tensor(self, mu, nu, drop_trace=False):
        f = self.f
        a = Field("a", indices=[])

        Tmunu = sum(f.d(fld, mu) * f.d(fld, nu)
                    for fld in range(self.nscalars))
        if drop_trace:
            return Tmunu

        metric_con = np.diag((-1 / a**2, 1 / a**2, 1 / a**2, 1 / a**2))
        lag = (-sum(sum(metric_con[m, n] * f.d(fld, m) * f.d(fld, n)
                        for m in range(4) for n in range(4))
                    for fld in range(self.nscalars)) / 2
               - self.potential(self.f))
        metric_cov = np.diag((-a**2, a**2, a**2, a**2))
        return Tmunu + metric_cov[mu, nu] * lag


def tensor_index(i, j):
    """Symmetric-pair index for h_ij storage, i,j ∈ {1,2,3} → 0..5
    (reference sectors.py:164-167, shifted to 0-based)."""
    a = min(i, j)
    b = max(i, j)
    return (7 - a) * a // 2 - 4 + b


class TensorPerturbationSector(Sector):
    """Tensor metric perturbations h_ij sourced by the (traceless part
    of the) stress tensor of the given sectors
    (reference sectors.py:170-208)."""

    def __init__(self, sectors, components=None, **kwargs):
        ncomp = 6 if components is None else len(components)
        name = kwargs.pop("name", "hij")
        self.hij = kwargs.pop(
            "hij", DynamicField(name, offset="h", shape=(ncomp,)))
        self.sectors = sectors
        # components: subset of the 6 symmetric-pair indices this
        # sector's field carries (the h_ij components are mutually
        # independent in the EOM, so the sector can be split into
        # several smaller stencil families — on MI355X this halves the
        # per-kernel register-ring footprint; see bench.py --gws)
        self.components = (tuple(range(6)) if components is None
                           else tuple(components))

    @property
    def rhs_dict(self):
        hij = self.hij
        H = Field("hubble", indices=[])
        rhs = {}
        for i in range(1, 4):
            for j in range(i, 4):
                pair = tensor_index(i, j)
                if pair not in self.components:
                    continue
                fld = self.components.index(pair)
                Sij = sum(sector.stress_tensor(i, j, drop_trace=True)
                          for sector in self.sectors)
                rhs[hij[fld]] = hij.dot[fld]
                rhs[hij.dot[fld]] = (hij.lap[fld]
                                     - 2 * H * hij.dot[fld]
                                     + 16 * np.pi * Sij)
        return rhs

    @property
    def reducers(self):
        return {}


def get_rho_and_p(energy):
    """Combine energy components into total density and pressure
    (reference sectors.py:211-229)."""
    energy["total"] = sum(sum(e) for e in energy.values())
    energy["pressure"] = 0
    if "kinetic" in energy:
        energy["pressure"] += sum(energy["kinetic"])
    if "gradient" in energy:
        energy["pressure"] += -sum(energy["gradient"]) / 3
    if "potential" in energy:
        energy["pressure"] += -sum(energy["potential"])
    return energy
