"""Multigrid tests: transfer-operator oracles, relaxation convergence,
and FAS solves of a manufactured Poisson problem (style of reference
test/test_transfer.py, test/test_relax.py, test/test_multigrid.py)."""

import numpy as np
import pytest
import torch

import pystella_amd as ps
from pystella_amd.field import Field, var
from pystella_amd.multigrid import (
    FullWeighting, Injection, LinearInterpolation, CubicInterpolation,
    JacobiIterator, NewtonIterator, FullApproximationScheme,
    MultiGridSolver, v_cycle,
)


def _padded_random(shape, h, seed=0):
    rng = np.random.default_rng(seed)
    t = torch.zeros(tuple(n + 2 * h for n in shape), dtype=torch.float64)
    t[h:-h, h:-h, h:-h] = torch.as_tensor(rng.random(shape))
    return t


def test_restriction_oracle(h=1, n2=(8, 8, 8)):
    nf = tuple(2 * n for n in n2)
    decomp_f = ps.DomainDecomposition((1, 1, 1), h, rank_shape=nf)
    f1 = _padded_random(nf, h)
    decomp_f.share_halos(f1)
    f2 = torch.zeros(tuple(n + 2 * h for n in n2), dtype=torch.float64)

    FullWeighting(halo_shape=h)(f1=f1, f2=f2)
    # oracle
    w = {-1: .25, 0: .5, 1: .25}
    out = np.zeros(n2)
    f1n = f1.numpy()
    for i in range(n2[0]):
        for a, ca in w.items():
            for b, cb in w.items():
                for c, cc in w.items():
                    out[i] += ca * cb * cc * f1n[
                        h + 2 * i + a, h + 2 * np.arange(n2[1])[:, None] + b,
                        h + 2 * np.arange(n2[2])[None, :] + c]
    assert np.allclose(f2[h:-h, h:-h, h:-h].numpy(), out)

    Injection(halo_shape=h)(f1=f1, f2=f2)
    assert np.allclose(f2[h:-h, h:-h, h:-h].numpy(),
                       f1n[h:-h:2, h:-h:2, h:-h:2])


@pytest.mark.parametrize("Interp", [LinearInterpolation,
                                    CubicInterpolation])
def test_interpolation_exact_for_linear(Interp, n2=(8, 8, 8)):
    """Interpolation must reproduce polynomials of its order on interior
    points; test with a linear function (both schemes exact)."""
    h = 2
    nf = tuple(2 * n for n in n2)
    xs2 = [np.arange(-h, n + h) + 0.0 for n in n2]
    X2, Y2, Z2 = np.meshgrid(*xs2, indexing="ij")
    lin2 = 1.0 + 0.5 * X2 + 0.25 * Y2 - 0.125 * Z2   # values at coarse pts
    f2 = torch.as_tensor(lin2)
    f1 = torch.zeros(tuple(2 * n + 2 * h for n in n2),
                     dtype=torch.float64)
    Interp(halo_shape=h)(f1=f1, f2=f2)

    # fine-grid coordinates: fine index m ↔ coarse coordinate m/2
    xs1 = [np.arange(0, 2 * n) / 2 for n in n2]
    X1, Y1, Z1 = np.meshgrid(*xs1, indexing="ij")
    expect = 1.0 + 0.5 * X1 + 0.25 * Y1 - 0.125 * Z1
    got = f1[h:-h, h:-h, h:-h].numpy()
    # interior fine points that don't touch the coarse boundary stencil
    sl = (slice(2, -2),) * 3
    assert np.allclose(got[sl], expect[sl], atol=1e-12)


def _poisson_setup(n, h=1, L=10.0, seed=1):
    grid_shape = (n, n, n)
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = L / n

    def get_laplacian(f):
        from pystella_amd.derivs import _LAP_COEFS, centered_diff
        lap_coefs = _LAP_COEFS[h]
        return sum(centered_diff(f, lap_coefs, direction=mu, order=2)
                   for mu in range(1, 4)) / var("dx")**2

    f = Field("f", offset="h")
    rho = Field("rho", offset="h")
    problems = {f: (get_laplacian(f), rho)}

    # manufactured solution: f* = sin(2π x/L) sin(2π y/L) sin(2π z/L)
    xs = np.arange(-h, n + h) * dx
    X, Y, Z = np.meshgrid(xs, xs, xs, indexing="ij")
    k = 2 * np.pi / L
    f_exact = np.sin(k * X) * np.sin(k * Y) * np.sin(k * Z)
    # discrete rho = L_h(f*) so the discrete solve is exact
    fe = torch.as_tensor(f_exact)
    derivs = ps.FiniteDifferencer(decomp, h, (dx,) * 3,
                                  rank_shape=grid_shape)
    rho_t = torch.zeros(grid_shape, dtype=torch.float64)
    derivs(fx=fe.clone(), lap=rho_t)
    rho_pad = torch.zeros_like(fe)
    rho_pad[h:-h, h:-h, h:-h] = rho_t
    decomp.share_halos(rho_pad)
    return decomp, dx, problems, fe, rho_pad


@pytest.mark.parametrize("Solver", [JacobiIterator, NewtonIterator])
def test_relax_reduces_error(Solver):
    n, h = 32, 1
    decomp, dx, problems, f_exact, rho = _poisson_setup(n, h)
    solver = Solver(decomp, problems, halo_shape=h,
                    fixed_parameters=dict(omega=0.8))
    f = torch.zeros_like(f_exact)
    tmp = torch.zeros_like(f)
    err0 = solver.get_error(f=f, tmp_f=tmp, r_f=tmp, rho=rho,
                            dx=np.array(dx))["f"]
    solver(decomp, iterations=100, f=f, tmp_f=tmp, rho=rho,
           dx=np.array(dx))
    err1 = solver.get_error(f=f, tmp_f=tmp, r_f=tmp, rho=rho,
                            dx=np.array(dx))["f"]
    assert err1[0] < 0.5 * err0[0], (err0, err1)
    assert err1[1] < 0.5 * err0[1]


@pytest.mark.parametrize("MG", [FullApproximationScheme, MultiGridSolver])
def test_multigrid_solve(MG):
    n, h = 32, 1
    decomp, dx, problems, f_exact, rho = _poisson_setup(n, h)
    solver = NewtonIterator(decomp, problems, halo_shape=h,
                            fixed_parameters=dict(omega=0.8))
    mg = MG(solver, halo_shape=h)
    f = torch.zeros_like(f_exact)

    errs = mg(decomp, dx0=dx, cycle=v_cycle(10, 20, 2),
              f=f, rho=rho)
    # final error on the finest level
    final = [e for lvl, e in errs if lvl == 0][-1]["f"]
    initial = [e for lvl, e in errs if lvl == 0][0]["f"]
    assert final[1] < 0.05 * initial[1], (initial, final)

    # the discrete solution approaches the manufactured solution up to
    # the nullspace (mean); compare mean-removed fields
    got = f[h:-h, h:-h, h:-h]
    want = f_exact[h:-h, h:-h, h:-h]
    got = got - got.mean()
    want = want - want.mean()
    rel = (got - want).abs().max().item() / want.abs().max().item()
    assert rel < 0.05, rel


def test_multigrid_solve_fp32():
    """The whole MG stack also runs in fp32 (the 1024^3-scale
    production configuration; kernels compile with `using real=float`
    on the GPU and the torch path follows the array dtype)."""
    n, h = 32, 1
    decomp, dx, problems, f_exact, rho = _poisson_setup(n, h)
    solver = NewtonIterator(decomp, problems, halo_shape=h,
                            fixed_parameters=dict(omega=0.8))
    mg = FullApproximationScheme(solver, halo_shape=h)
    f = torch.zeros_like(f_exact, dtype=torch.float32)
    rho32 = rho.to(torch.float32)
    errs = mg(decomp, dx0=dx, cycle=v_cycle(10, 20, 2),
              f=f, rho=rho32)
    final = [e for lvl, e in errs if lvl == 0][-1]["f"]
    initial = [e for lvl, e in errs if lvl == 0][0]["f"]
    assert f.dtype == torch.float32
    assert final[1] < 0.05 * initial[1], (initial, final)
    got = f[h:-h, h:-h, h:-h].double()
    want = f_exact[h:-h, h:-h, h:-h]
    got = got - got.mean()
    want = want - want.mean()
    rel = (got - want).abs().max().item() / want.abs().max().item()
    assert rel < 0.05, rel


def test_rbgs_smoother_and_mg():
    """Red-black Gauss-Seidel smoother: converges at least as fast as
    Jacobi per sweep and drives the FAS solver."""
    from pystella_amd.multigrid import RedBlackIterator
    n, h = 32, 1
    decomp, dx, problems, f_exact, rho = _poisson_setup(n, h)
    solver = RedBlackIterator(decomp, problems, halo_shape=h,
                              fixed_parameters=dict(omega=1.0))
    f = torch.zeros_like(f_exact)
    tmp = torch.zeros_like(f)
    err0 = solver.get_error(f=f, tmp_f=tmp, r_f=tmp.clone(), rho=rho,
                            dx=np.array(dx))["f"]
    solver(decomp, iterations=50, f=f, tmp_f=tmp, rho=rho,
           dx=np.array(dx))
    err1 = solver.get_error(f=f, tmp_f=tmp, r_f=tmp.clone(), rho=rho,
                            dx=np.array(dx))["f"]
    assert err1[1] < 0.3 * err0[1], (err0, err1)

    mg = FullApproximationScheme(solver, halo_shape=h)
    f2 = torch.zeros_like(f_exact)
    errs = mg(decomp, dx0=dx, cycle=v_cycle(4, 8, 2), f=f2, rho=rho)
    final = [e for lvl, e in errs if lvl == 0][-1]["f"]
    initial = [e for lvl, e in errs if lvl == 0][0]["f"]
    assert final[1] < 0.1 * initial[1], (initial, final)
