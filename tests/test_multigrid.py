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


def _dist_mg_worker(rank, world_size, proc_shape, use_rbgs=False):
    """FAS V-cycle on a distributed decomposition produces the SAME
    iterate as the single-rank solve (same smoother, same cycle) —
    Jacobi/Newton sweeps are rank-count-independent, so the fields
    must agree to fp roundoff."""
    n, h, L = 32, 1, 10.0

    # single-rank oracle, computed identically on every rank (a
    # (1,1,1) decomposition is local even inside a multi-rank program)
    decomp1, dx, problems1, f_exact, rho_pad = _poisson_setup(n, h, L)
    Smoother = NewtonIterator
    kwargs = dict(fixed_parameters=dict(omega=0.8))
    if use_rbgs:
        from pystella_amd.multigrid import RedBlackIterator
        Smoother = RedBlackIterator
        kwargs = dict(fixed_parameters=dict(omega=1.0))
    solver1 = Smoother(decomp1, problems1, halo_shape=h, **kwargs)
    mg1 = FullApproximationScheme(solver1, halo_shape=h)
    f1 = torch.zeros_like(f_exact)
    errs1 = mg1(decomp1, dx0=dx, cycle=v_cycle(6, 12, 2),
                f=f1, rho=rho_pad)

    # distributed solve of the same problem
    grid_shape = (n, n, n)
    decomp = ps.DomainDecomposition(proc_shape, h, grid_shape=grid_shape)
    rank_shape, start = decomp.get_rank_shape_start(grid_shape)
    pad_sl = tuple(slice(s, s + m + 2 * h)
                   for s, m in zip(start, rank_shape))
    f = torch.zeros(tuple(m + 2 * h for m in rank_shape),
                    dtype=torch.float64)
    rho_loc = rho_pad[pad_sl].contiguous()
    decomp.share_halos(rho_loc)

    solver = Smoother(decomp, problems1, halo_shape=h, **kwargs)
    mg = FullApproximationScheme(solver, halo_shape=h)
    errs = mg(decomp, dx0=dx, cycle=v_cycle(6, 12, 2),
              f=f, rho=rho_loc)

    # error histories agree (they are global reductions)
    for (l1, e1), (l2, e2) in zip(errs1, errs):
        assert l1 == l2
        assert np.allclose(e1["f"], e2["f"], rtol=1e-10), (l1, e1, e2)

    # and the final iterate agrees pointwise with the oracle's slice
    int_sl = tuple(slice(s + h, s + h + m)
                   for s, m in zip(start, rank_shape))
    mine = f[(slice(h, -h),) * 3]
    want = f1[int_sl]
    err = (mine - want).abs().max().item()
    assert err < 1e-11, (rank, err)


def test_distributed_mg_2rank():
    from tests.conftest import run_distributed
    run_distributed(_dist_mg_worker, 2, args=((2, 1, 1),))


def test_distributed_mg_2rank_z():
    from tests.conftest import run_distributed
    run_distributed(_dist_mg_worker, 2, args=((1, 1, 2),))


def test_distributed_mg_222():
    """The driver's N=8 topology: 32^3 over (2,2,2) coarsens to
    4^3+halos per rank at the bottom of the V-cycle — deep coarsening
    across ranks must still match the single-rank solve."""
    from tests.conftest import run_distributed
    run_distributed(_dist_mg_worker, 8, args=((2, 2, 2),))


def test_distributed_mg_rbgs_2rank():
    """Red-black Gauss-Seidel across a rank seam: the global
    checkerboard parity fix (rb_off from the rank's global start) makes
    the distributed sweep identical to the single-rank sweep."""
    from tests.conftest import run_distributed
    run_distributed(_dist_mg_worker, 2, args=((2, 1, 1), True))


def _dist_rbgs_odd_worker(rank, world_size):
    """Standalone RBGS sweeps over an ODD split (30 -> 15+15): rank 1's
    global x-start is odd, so its checkerboard color must flip
    (rb_off=1) to agree with the single-rank sweep."""
    from pystella_amd.multigrid import RedBlackIterator
    n, h, L = 30, 1, 10.0
    decomp1, dx, problems, f_exact, rho_pad = _poisson_setup(n, h, L)
    s1 = RedBlackIterator(decomp1, problems, halo_shape=h,
                          fixed_parameters=dict(omega=1.0))
    f1 = torch.zeros_like(f_exact)
    s1(decomp1, iterations=4, f=f1, tmp_f=torch.zeros_like(f1),
       rho=rho_pad, dx=np.array(dx))

    grid_shape = (n, n, n)
    decomp = ps.DomainDecomposition((2, 1, 1), h, grid_shape=grid_shape)
    rank_shape, start = decomp.get_rank_shape_start(grid_shape)
    pad_sl = tuple(slice(s, s + m + 2 * h)
                   for s, m in zip(start, rank_shape))
    rho_loc = rho_pad[pad_sl].contiguous()
    decomp.share_halos(rho_loc)
    f = torch.zeros(tuple(m + 2 * h for m in rank_shape),
                    dtype=torch.float64)
    s = RedBlackIterator(decomp, problems, halo_shape=h,
                         fixed_parameters=dict(omega=1.0))
    s(decomp, iterations=4, f=f, tmp_f=torch.zeros_like(f),
      rho=rho_loc, dx=np.array(dx))

    int_sl = tuple(slice(st + h, st + h + m)
                   for st, m in zip(start, rank_shape))
    err = (f[(slice(h, -h),) * 3] - f1[int_sl]).abs().max().item()
    assert err < 1e-12, (rank, err)


def test_distributed_rbgs_odd_split():
    from tests.conftest import run_distributed
    run_distributed(_dist_rbgs_odd_worker, 2)


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


def test_constraint_machinery():
    """Integral-constraint solve (reference relax.py:268-320 declares
    this API but disables its implementation): for L = lap f - m^2 f,
    shifting f by s changes the mean residual by -m^2 s, so the solve
    must zero the volume-averaged residual."""
    n, h, L = 16, 1, 10.0
    decomp = ps.DomainDecomposition((1, 1, 1), h,
                                    rank_shape=(n, n, n))
    dx = L / n
    msq = 0.7

    def get_lap(f):
        from pystella_amd.derivs import _LAP_COEFS, centered_diff
        return sum(centered_diff(f, _LAP_COEFS[h], direction=mu, order=2)
                   for mu in range(1, 4)) / var("dx")**2

    f = Field("f", offset="h")
    rho = Field("rho", offset="h")
    problems = {f: (get_lap(f) - msq * f, rho)}
    solver = NewtonIterator(decomp, problems, halo_shape=h,
                            fixed_parameters=dict(omega=0.8))

    rng = np.random.default_rng(3)
    pad = (n + 2 * h,) * 3
    fv = torch.as_tensor(rng.standard_normal(pad))
    decomp.share_halos(fv)
    rhov = torch.as_tensor(rng.standard_normal(pad))
    decomp.share_halos(rhov)
    kw = dict(f=fv, tmp_f=torch.zeros_like(fv), rho=rhov,
              dx=np.array(dx))

    avg0 = solver.eval_constraint(shifts=np.zeros(1), scales=np.ones(1),
                                  **kw)
    shifts = solver.solve_constraint(**kw)
    avg1 = solver.eval_constraint(shifts=np.zeros(1), scales=np.ones(1),
                                  **kw)
    # analytic check: shift = avg0 / (-(-m^2)) = avg0 / -m^2... the
    # mean residual is linear in s with slope -m^2
    assert abs(avg1[0]) < 1e-12 * max(1.0, abs(avg0[0])), (avg0, avg1)
    assert np.allclose(shifts, avg0 / msq, rtol=1e-8), (shifts, avg0)


def test_bench_mg_distributed_cpu(tmp_path):
    """tools/bench_mg.py under torchrun (2 ranks, gloo): BASELINE
    config 5 names multigrid on 8x MI355X — the distributed launch
    path must work end to end."""
    import json
    import os
    import subprocess
    import sys
    import glob
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--standalone", "--local-addr", "127.0.0.1",
           "--nnodes=1", "--nproc-per-node", "2",
           "--redirects", "3", "--log-dir", str(tmp_path / "trlogs"),
           os.path.join(repo, "tools", "bench_mg.py"),
           "--grid", "32", "--cycles", "1", "--depth", "2",
           "--device", "cpu", "--dtype", "float64",
           "--smoother", "newton"]
    from tests.conftest import run_torchrun
    out, logs = run_torchrun(cmd, repo, tmp_path / "trlogs")
    assert out.returncode == 0, (out.stderr[-1500:], logs[-1500:])
    line = [ln for ln in logs.splitlines() if ln.startswith("{")][-1]
    d = json.loads(line)
    assert d["n_ranks"] == 2
    assert d["resid_L2_end"] < d["resid_L2_start"]


def test_restriction_interpolation_noncubic(h=1, n2=(4, 6, 8)):
    """Separable transfer passes on a noncubic grid against the
    direct 27-term / 8-parity oracle."""
    from itertools import product as iproduct
    nf = tuple(2 * n for n in n2)
    decomp_f = ps.DomainDecomposition((1, 1, 1), h, rank_shape=nf)
    f1 = _padded_random(nf, h)
    decomp_f.share_halos(f1)
    f2 = torch.zeros(tuple(n + 2 * h for n in n2), dtype=torch.float64)
    FullWeighting(halo_shape=h)(f1=f1, f2=f2)
    w = {-1: .25, 0: .5, 1: .25}
    f1n = f1.numpy()
    out = np.zeros(n2)
    for (a, ca), (b, cb), (c, cc) in iproduct(
            w.items(), w.items(), w.items()):
        out += ca * cb * cc * f1n[
            h + a:h + a + 2 * n2[0]:2,
            h + b:h + b + 2 * n2[1]:2,
            h + c:h + c + 2 * n2[2]:2]
    assert np.allclose(f2[h:-h, h:-h, h:-h].numpy(), out)

    # interpolation: linear function reproduced exactly off-boundary
    xs2 = [np.arange(-h, n + h) + 0.0 for n in n2]
    X2, Y2, Z2 = np.meshgrid(*xs2, indexing="ij")
    lin2 = 0.3 - 0.5 * X2 + 0.25 * Y2 + 0.125 * Z2
    c2 = torch.as_tensor(lin2)
    fine = torch.zeros(tuple(2 * n + 2 * h for n in n2),
                       dtype=torch.float64)
    LinearInterpolation(halo_shape=h)(f1=fine, f2=c2)
    xs1 = [np.arange(0, 2 * n) / 2 for n in n2]
    X1, Y1, Z1 = np.meshgrid(*xs1, indexing="ij")
    expect = 0.3 - 0.5 * X1 + 0.25 * Y1 + 0.125 * Z1
    got = fine[h:-h, h:-h, h:-h].numpy()
    sl = (slice(1, -1),) * 3
    assert np.allclose(got[sl], expect[sl], atol=1e-12)

    # correct=True variants accumulate/subtract rather than overwrite
    f2b = f2.clone()
    FullWeighting(halo_shape=h, correct=True)(f1=f1, f2=f2b)
    assert np.allclose(f2b[h:-h, h:-h, h:-h].numpy(), 0.0)
    fine2 = fine.clone()
    LinearInterpolation(halo_shape=h, correct=True)(f1=fine2, f2=c2)
    assert np.allclose(fine2[h:-h, h:-h, h:-h].numpy(), 2 * got)
