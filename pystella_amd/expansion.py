"""FLRW scale-factor evolution (conformal time).

Analogue of reference pystella/expansion.py:28-176.  The reference
generates a C kernel via loopy's ExecutableCTarget for this 0-D ODE; here
it simply runs through the same symbolic Stepper machinery on host numpy
scalars — a 0-D ODE needs no device kernel.
"""

from __future__ import annotations

import numpy as np

from pystella_amd.field import Field, var

__all__ = ["Expansion"]


class Expansion:
    """Background scale factor a(τ) evolution coupled to the
    volume-averaged energy density and pressure via Friedmann's
    equations.
    """

    def __init__(self, energy, Stepper, mpl=1., dtype=np.float64):
        self.mpl = mpl
        from pystella_amd.step import LowStorageRKStepper
        self.is_low_storage = issubclass(Stepper, LowStorageRKStepper)
        num_copies = Stepper.num_copies or 1
        shape = (num_copies,)
        self.a = np.ones(shape, dtype=dtype)
        self.adot = self.adot_friedmann_1(self.a, energy)
        self.hubble = self.adot / self.a

        arg_shape = (1,) if self.is_low_storage else ()
        slc = (0,) if self.is_low_storage else ()
        _a = Field("a", indices=[], shape=arg_shape)[slc]
        _adot = Field("adot", indices=[], shape=arg_shape)[slc]
        _e = var("energy")
        _p = var("pressure")
        rhs_dict = {
            _a: _adot,
            _adot: self.addot_friedmann_2(_a, _e, _p),
        }
        self.stepper = Stepper(rhs_dict, rank_shape=(0, 0, 0), halo_shape=0)

    def adot_friedmann_1(self, a, energy):
        """H² ≡ (∂_τ a / a)² = 8π a²/(3 m_pl²) ρ̄  (conformal)."""
        return np.sqrt(8 * np.pi * a**2 / 3 / self.mpl**2 * energy) * a

    def addot_friedmann_2(self, a, energy, pressure):
        """∂²_τ a / a = 4π a²/(3 m_pl²) (ρ̄ − 3 P̄)."""
        return 4 * np.pi * a**2 / 3 / self.mpl**2 * (energy - 3 * pressure) * a

    def step(self, stage, energy, pressure, dt):
        self.stepper(stage, a=self.a, adot=self.adot, dt=dt,
                     energy=float(energy), pressure=float(pressure))
        self.hubble[()] = self.adot / self.a

    def constraint(self, energy):
        """|adot_friedmann_1(a, ρ)/adot − 1| — the Friedmann-constraint
        violation used as the physics regression metric
        (reference test/test_examples.py:33)."""
        return np.abs(
            self.adot_friedmann_1(self.a[0], energy) / self.adot[0] - 1)
