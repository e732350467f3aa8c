"""Stepper convergence tests: integrate y' = y^n with every stepper and
check the measured global-error order (oracle style of reference
test/test_step.py:66-99)."""

import math

import pytest
import torch

import pystella_amd as ps
from pystella_amd.field import Field


@pytest.mark.parametrize("Stepper", ps.all_steppers)
def test_convergence_order(Stepper):
    # dy/dt = y^2, y(0) = 1  →  y(t) = 1/(1−t)
    y = Field("y", offset=0)
    rhs = {y: y**2}
    t_end = 0.5
    exact = 1 / (1 - t_end)

    errs = []
    dts = [0.05, 0.025]
    for dt in dts:
        stepper = Stepper(rhs, dt=dt, halo_shape=0, rank_shape=(4, 4, 4))
        if Stepper.num_copies is not None:
            arr = torch.ones((Stepper.num_copies, 4, 4, 4),
                             dtype=torch.float64)
        else:
            arr = torch.ones((4, 4, 4), dtype=torch.float64)
            stepper.tmp_arrays = {}
        t = 0
        while t < t_end - 1e-12:
            for s in range(stepper.num_stages):
                stepper(s, y=arr)
            t += dt
        val = arr.reshape(-1)[0].item()
        errs.append(abs(val - exact))

    order = math.log(errs[0] / errs[1]) / math.log(dts[0] / dts[1])
    expected = Stepper.expected_order
    assert errs[1] < 0.1, (Stepper.__name__, errs)
    assert order > 0.9 * expected, (Stepper.__name__, order, errs)


def test_low_storage_tmp_identity():
    # tmp arrays must persist across stages (reference
    # test/test_step.py:121-124)
    y = Field("y", offset=0)
    stepper = ps.LowStorageRK54({y: y}, dt=0.01, halo_shape=0,
                                rank_shape=(4, 4, 4))
    arr = torch.ones((4, 4, 4), dtype=torch.float64)
    stepper(0, y=arr)
    tmp0 = stepper.tmp_arrays["y_tmp"]
    stepper(1, y=arr)
    assert stepper.tmp_arrays["y_tmp"] is tmp0


def test_multi_unknown_system():
    # d(f)/dt = g, d(g)/dt = -f  →  harmonic oscillator
    f = Field("f", offset=0)
    g = Field("g", offset=0)
    rhs = {f: g, g: -1 * f}
    dt = 0.01
    stepper = ps.LowStorageRK54(rhs, dt=dt, halo_shape=0,
                                rank_shape=(2, 2, 2))
    af = torch.ones((2, 2, 2), dtype=torch.float64)
    ag = torch.zeros((2, 2, 2), dtype=torch.float64)
    t = 0.
    while t < 1.0 - 1e-12:
        for s in range(stepper.num_stages):
            stepper(s, f=af, g=ag)
        t += dt
    assert abs(af[0, 0, 0].item() - math.cos(1.0)) < 1e-8
    assert abs(ag[0, 0, 0].item() + math.sin(1.0)) < 1e-8


def test_classical_rk4_with_sector():
    """Classical (multi-copy) RK4 consumes a Sector rhs with outer
    indices — regression for the q-axis prepend on f[i] accesses."""
    import numpy as np
    import pystella_amd as ps

    sector = ps.ScalarSector(1, potential=lambda f: f[0]**4 / 4)
    h, grid_shape, dt = 1, (8, 8, 8), 1e-3
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(3)
    nc = ps.RungeKutta4.num_copies
    f = torch.rand((nc, 1) + pad, dtype=torch.float64)
    d = torch.rand((nc, 1) + pad, dtype=torch.float64)
    lap = torch.rand((nc, 1) + grid_shape, dtype=torch.float64)
    a = np.array([1.0] * nc)
    hub = np.array([0.1] * nc)
    st = ps.RungeKutta4([sector], dt=dt, halo_shape=h,
                        rank_shape=grid_shape)
    f0 = f[0].clone()
    for s in range(st.num_stages):
        st(s, a=a, hubble=hub, f=f, dfdt=d, lap_f=lap)
    assert torch.isfinite(f).all()
    # the update is O(dt): result stays close to the input state
    assert (f[0] - f0).abs().max().item() < 10 * dt
