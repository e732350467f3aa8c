"""GPU numerics tests: every HIP kernel vs the torch-CPU fp64 oracle.

All tests here are marked ``gpu`` and run on a real MI355X
(`pytest -m gpu`).  They verify that the hand-written/JIT'd CDNA4
kernels produce the same numbers as the CPU reference path.
"""

import math

import numpy as np
import pytest
import torch

import pystella_amd as ps
from pystella_amd.field import Field, var

pytestmark = pytest.mark.gpu

requires_gpu = pytest.mark.skipif(not torch.cuda.is_available(),
                                  reason="needs a GPU")


def _to_dev(*tensors):
    return [t.cuda() for t in tensors]


@requires_gpu
@pytest.mark.parametrize("h", [1, 2, 3, 4])
@pytest.mark.parametrize("tdtype", [torch.float64, torch.float32])
def test_gradlap_vs_cpu(h, tdtype, grid_shape=(32, 32, 32)):
    """AOT stencil kernels (csrc/derivs.hip, dtype-templated) vs the
    CPU torch path, fp64 and fp32 (the reference is dtype-generic:
    derivs.py:234)."""
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.1, 0.11, 0.12)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(0)
    f = torch.rand((2,) + pad, dtype=tdtype)
    tol = 1e-12 if tdtype == torch.float64 else 2e-4

    lap_c = torch.zeros((2,) + grid_shape, dtype=tdtype)
    grd_c = torch.zeros((2, 3) + grid_shape, dtype=tdtype)
    derivs(fx=f.clone(), lap=lap_c, grd=grd_c)

    fg = f.clone().cuda()
    lap_g = torch.zeros((2,) + grid_shape, dtype=tdtype,
                        device="cuda")
    grd_g = torch.zeros((2, 3) + grid_shape, dtype=tdtype,
                        device="cuda")
    derivs(fx=fg, lap=lap_g, grd=grd_g)
    torch.cuda.synchronize()

    assert (lap_g.cpu() - lap_c).abs().max().item() < tol * (
        1. + lap_c.abs().max().item())
    assert (grd_g.cpu() - grd_c).abs().max().item() < tol * (
        1. + grd_c.abs().max().item())


@requires_gpu
@pytest.mark.parametrize("h", [1, 2])
def test_lap_only_and_pd(h, grid_shape=(24, 24, 24)):
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.1, 0.1, 0.1)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(1)
    f = torch.rand(pad, dtype=torch.float64)

    lap_c = torch.zeros(grid_shape, dtype=torch.float64)
    derivs(fx=f.clone(), lap=lap_c)
    pdy_c = torch.zeros(grid_shape, dtype=torch.float64)
    derivs(fx=f.clone(), pdy=pdy_c)

    fg = f.clone().cuda()
    lap_g = torch.zeros(grid_shape, dtype=torch.float64, device="cuda")
    derivs(fx=fg, lap=lap_g)
    pdy_g = torch.zeros(grid_shape, dtype=torch.float64, device="cuda")
    derivs(fx=fg, pdy=pdy_g)
    torch.cuda.synchronize()
    assert (lap_g.cpu() - lap_c).abs().max().item() < 1e-12
    assert (pdy_g.cpu() - pdy_c).abs().max().item() < 1e-12


@requires_gpu
def test_divergence_gpu(grid_shape=(24, 24, 24), h=2):
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.1, 0.11, 0.12)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(5)
    vec = torch.rand((3,) + pad, dtype=torch.float64)
    div_c = torch.zeros(grid_shape, dtype=torch.float64)
    derivs.divergence(vec=vec.clone(), div=div_c)
    vg = vec.clone().cuda()
    div_g = torch.zeros(grid_shape, dtype=torch.float64, device="cuda")
    derivs.divergence(vec=vg, div=div_g)
    torch.cuda.synchronize()
    assert (div_g.cpu() - div_c).abs().max().item() < 1e-12


@requires_gpu
def test_stage_kernel_vs_cpu(grid_shape=(16, 16, 16), h=2):
    """Fused RK stage kernel (hiprtc-JIT) vs the torch evaluator."""
    def potential(f):
        phi, chi = f[0], f[1]
        return phi**2 / 2 + phi**2 * chi**2 / 4

    sector = ps.ScalarSector(2, potential=potential)
    dt = 1e-3
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(2)
    f = torch.rand((2,) + pad, dtype=torch.float64)
    dfdt = torch.rand((2,) + pad, dtype=torch.float64)
    lap = torch.rand((2,) + grid_shape, dtype=torch.float64)
    a = np.array([1.1])
    hub = np.array([0.3])

    st_c = ps.LowStorageRK54([sector], dt=dt, halo_shape=h,
                             rank_shape=grid_shape)
    fc, dc = f.clone(), dfdt.clone()
    for s in range(st_c.num_stages):
        st_c(s, a=a, hubble=hub, f=fc, dfdt=dc, lap_f=lap)

    st_g = ps.LowStorageRK54([sector], dt=dt, halo_shape=h,
                             rank_shape=grid_shape)
    fg, dg, lg = _to_dev(f.clone(), dfdt.clone(), lap)
    for s in range(st_g.num_stages):
        st_g(s, a=a, hubble=hub, f=fg, dfdt=dg, lap_f=lg)
    torch.cuda.synchronize()

    assert (fg.cpu() - fc).abs().max().item() < 1e-13
    assert (dg.cpu() - dc).abs().max().item() < 1e-13


@requires_gpu
def test_reduction_vs_cpu(grid_shape=(32, 32, 32), h=1):
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(3)
    f = torch.rand(pad, dtype=torch.float64)
    F = Field("f", offset="h")
    a = var("a")
    red = ps.Reduction(decomp, {
        "mean": [F],
        "kin": [F**2 / 2 / a**2],
        "mx": [(F, "max")],
        "mn": [(F, "min")],
    }, halo_shape=h, grid_size=float(np.prod(grid_shape)))
    out_c = red(f=f, a=np.array([1.7]))
    red2 = ps.Reduction(decomp, {
        "mean": [F],
        "kin": [F**2 / 2 / a**2],
        "mx": [(F, "max")],
        "mn": [(F, "min")],
    }, halo_shape=h, grid_size=float(np.prod(grid_shape)))
    out_g = red2(f=f.cuda(), a=np.array([1.7]))
    for k in out_c:
        assert np.allclose(out_c[k], out_g[k], rtol=1e-12), k


@requires_gpu
def test_histogram_vs_cpu(grid_shape=(32, 32, 32)):
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    torch.manual_seed(4)
    f = torch.rand(grid_shape, dtype=torch.float64)
    F = Field("f", offset=0)
    num_bins = 64
    mk = lambda: ps.Histogrammer(  # noqa: E731
        decomp, {"h": (F * num_bins, 1), "w": (F * num_bins, F)},
        num_bins, np.float64, halo_shape=0)
    out_c = mk()(f=f)
    out_g = mk()(f=f.cuda())
    for k in out_c:
        assert np.allclose(out_c[k], out_g[k], rtol=1e-12, atol=1e-9), k


@requires_gpu
def test_histogram_large_bins_global_atomics(grid_shape=(24, 24, 24)):
    """num_bins=4096 with 3 simultaneous histograms exceeds the LDS
    budget (3*4096 doubles > 64 KB) and must route to the
    global-atomics fallback, matching the CPU result (reference caps
    its workgroup + merges via global atomics: histogram.py:69,
    114-163)."""
    from pystella_amd.backend.hip import _HIST_LDS_DOUBLES
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    torch.manual_seed(5)
    f = torch.rand(grid_shape, dtype=torch.float64)
    F = Field("f", offset=0)
    num_bins = 4096
    assert 3 * num_bins > _HIST_LDS_DOUBLES
    pairs = {"a": (F * num_bins, 1),
             "b": (F * num_bins, F),
             "c": ((1 - F) * num_bins, F * F)}
    mk = lambda: ps.Histogrammer(  # noqa: E731
        decomp, pairs, num_bins, np.float64, halo_shape=0)
    out_c = mk()(f=f)
    out_g = mk()(f=f.cuda())
    for k in out_c:
        assert np.allclose(out_c[k], out_g[k], rtol=1e-12, atol=1e-9), k


@requires_gpu
def test_wave_equation_gpu_matches_cpu():
    import os
    import sys
    sys.path.insert(0, os.path.join(os.path.dirname(
        os.path.dirname(os.path.abspath(__file__))), "examples"))
    import wave_equation
    e_cpu = wave_equation.main(["--grid-shape", "16", "16", "16",
                                "--end-time", "0.2", "--device", "cpu"])
    e_gpu = wave_equation.main(["--grid-shape", "16", "16", "16",
                                "--end-time", "0.2", "--device", "cuda"])
    assert math.isfinite(e_gpu)
    assert abs(e_cpu - e_gpu) < 1e-10 * abs(e_cpu) + 1e-12


@requires_gpu
def test_native_extension_is_loaded():
    """Fail loudly if the HIP extension didn't load — GPU tests must not
    silently run on a torch fallback."""
    from pystella_amd.backend.hip import ext
    e = ext()
    assert e.device_count() >= 1
    assert "gfx95" in e.arch()


@requires_gpu
def test_fused_lap_reduction_vs_unfused(grid_shape=(32, 32, 32), h=2):
    from pystella_amd.fusion import FusedLaplacianReduction
    from pystella_amd.sectors import get_rho_and_p

    def potential(f):
        phi, chi = f[0], f[1]
        return phi**2 / 2 + phi**2 * chi**2 / 4

    sector = ps.ScalarSector(2, potential=potential)
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.1, 0.1, 0.1)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(7)
    f = torch.rand((2,) + pad, dtype=torch.float64).cuda()
    dfdt = torch.rand((2,) + pad, dtype=torch.float64).cuda()
    lap_u = torch.zeros((2,) + grid_shape, dtype=torch.float64,
                        device="cuda")
    lap_fz = torch.zeros_like(lap_u)
    a = np.array([1.3])

    gs = float(np.prod(grid_shape))
    unfused = ps.Reduction(decomp, sector, halo_shape=h, grid_size=gs,
                           callback=get_rho_and_p)
    derivs(fx=f, lap=lap_u)
    out_u = unfused(f=f, dfdt=dfdt, lap_f=lap_u, a=a)

    fused = FusedLaplacianReduction(decomp, sector, derivs, halo_shape=h,
                                    grid_size=gs, callback=get_rho_and_p)
    out_f = fused(f=f, dfdt=dfdt, lap_f=lap_fz, a=a)
    torch.cuda.synchronize()

    scale = lap_u.abs().max().item()
    assert (lap_fz - lap_u).abs().max().item() < 1e-13 * max(scale, 1.),         (lap_fz - lap_u).abs().max().item() / scale
    for k in ("kinetic", "potential", "gradient"):
        assert np.allclose(out_u[k], out_f[k], rtol=1e-11), (
            k, out_u[k], out_f[k])
    assert abs(out_u["total"] - out_f["total"]) < 1e-11 * abs(
        out_u["total"])


@requires_gpu
def test_stencil_stepper_vs_unfused_gpu(grid_shape=(24, 24, 24), h=2):
    """Fully fused hot loop (inline-Laplacian stage kernels + no-lap
    energy reduction) reproduces the reference-structure loop."""
    from pystella_amd.fusion import (
        FusedLaplacianReduction, StencilRKStepper)
    from pystella_amd.sectors import get_rho_and_p

    def potential(f):
        return f[0]**2 / 2 + f[0]**2 * f[1]**2 / 4

    sector = ps.ScalarSector(2, potential=potential)
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.2, 0.2, 0.2)
    dt = 0.01
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    gs = float(np.prod(grid_shape))
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(11)
    f0 = torch.rand((2,) + pad, dtype=torch.float64)
    d0 = torch.rand((2,) + pad, dtype=torch.float64)
    a = np.ones(1)
    hub = 0.1 * np.ones(1)
    cut = (slice(None),) + (slice(h, -h),) * 3

    # reference-structure on GPU
    fu, du = f0.clone().cuda(), d0.clone().cuda()
    lap = torch.zeros((2,) + grid_shape, dtype=torch.float64,
                      device="cuda")
    st = ps.LowStorageRK54([sector], dt=dt, halo_shape=h,
                           rank_shape=grid_shape)
    red_u = FusedLaplacianReduction(decomp, sector, derivs, halo_shape=h,
                                    grid_size=gs, callback=get_rho_and_p,
                                    store_lap=True)
    e_u = red_u(f=fu, dfdt=du, lap_f=lap, a=a)
    for s in range(st.num_stages):
        st(s, a=a, hubble=hub, f=fu, dfdt=du, lap_f=lap)
        e_u = red_u(f=fu, dfdt=du, lap_f=lap, a=a)

    # fused structure on GPU
    ff, df = f0.clone().cuda(), d0.clone().cuda()
    fnx = torch.zeros_like(ff)
    fst = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                           halo_shape=h, rank_shape=grid_shape, dt=dt)
    red_f = FusedLaplacianReduction(decomp, sector, derivs, halo_shape=h,
                                    grid_size=gs, callback=get_rho_and_p,
                                    store_lap=False)
    arrays = {"f": ff, "dfdt": df, "f_next": fnx}
    e_f = red_f(f=arrays["f"], dfdt=arrays["dfdt"], a=a)
    for s in range(fst.num_stages):
        fst(s, a=a, hubble=hub, **arrays)
        arrays["f"], arrays["f_next"] = arrays["f_next"], arrays["f"]
        e_f = red_f(f=arrays["f"], dfdt=arrays["dfdt"], a=a)
    torch.cuda.synchronize()

    assert (arrays["f"][cut] - fu[cut]).abs().max().item() < 1e-12
    assert (arrays["dfdt"][cut] - du[cut]).abs().max().item() < 1e-12
    assert abs(e_f["total"] - e_u["total"]) < 1e-10 * abs(e_u["total"])


@requires_gpu
def test_stage_fused_energy_gpu(grid_shape=(32, 32, 32)):
    """GPU energy-fused RK stage kernel vs the CPU oracle: same energy
    values and same updated fields."""
    from pystella_amd.fusion import StencilRKStepper
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.31, 0.32)
    dt = 0.01
    gsize = float(np.prod(grid_shape))

    def pot(f):
        return f[0]**2 / 2 + f[0]**2 * f[1]**2 / 4

    sector = ps.ScalarSector(2, potential=pot)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(5)
    f0 = torch.rand((2,) + pad, dtype=torch.float64)
    d0 = torch.rand((2,) + pad, dtype=torch.float64)
    a = np.ones(1)
    hub = 0.1 * np.ones(1)
    cut = (slice(None),) + (slice(h, -h),) * 3

    def run(device):
        ff = f0.clone().to(device)
        df = d0.clone().to(device)
        fst = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                               halo_shape=h, rank_shape=grid_shape,
                               dt=dt, reducers=sector, grid_size=gsize,
                               callback=get_rho_and_p)
        arrays = {"f": ff, "dfdt": df, "f_next": torch.zeros_like(ff)}
        decomp.share_halos(arrays["f"])
        energies = []
        for s in range(fst.num_stages):
            energies.append(fst(s, a=a, hubble=hub, **arrays))
            arrays["f"], arrays["f_next"] = \
                arrays["f_next"], arrays["f"]
            decomp.share_halos(arrays["f"])
        return arrays, energies

    arr_c, en_c = run("cpu")
    arr_g, en_g = run("cuda")
    torch.cuda.synchronize()
    for s, (ec, eg) in enumerate(zip(en_c, en_g)):
        for key in ec:
            assert np.allclose(np.asarray(ec[key]), np.asarray(eg[key]),
                               rtol=1e-12), (s, key)
    assert (arr_g["f"].cpu()[cut] - arr_c["f"][cut]).abs().max() < 1e-12
    assert (arr_g["dfdt"].cpu()[cut]
            - arr_c["dfdt"][cut]).abs().max() < 1e-12


@requires_gpu
def test_scalar_preheating_gws_gpu(tmp_path):
    """Full scalar-preheating example with gravitational waves, spectra
    and output on the GPU (fourier stack on rocFFT via torch.fft,
    stencil/stage/reduction kernels on the HIP extension); GPU result
    matches the CPU run."""
    import os
    import sys
    sys.path.insert(0, os.path.join(
        os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        "examples"))
    import scalar_preheating
    os.chdir(tmp_path)
    args = ["--grid-shape", "16", "16", "16", "--end-time", "0.3",
            "--no-output", "--gravitational-waves"]
    expand_g, energy_g = scalar_preheating.main(
        args + ["--device", "cuda"])
    expand_c, energy_c = scalar_preheating.main(
        args + ["--device", "cpu"])
    assert np.isfinite(expand_g.constraint(energy_g["total"]))
    assert np.allclose(energy_g["total"], energy_c["total"], rtol=1e-8)
    assert np.allclose(float(expand_g.a[0]), float(expand_c.a[0]),
                       rtol=1e-10)


@requires_gpu
@pytest.mark.parametrize("periodic", ["0", "1"])
def test_device_friedmann_loop_gpu(periodic, monkeypatch,
                                   grid_shape=(32, 32, 32)):
    """Fully device-resident step (stage kernel + on-device Friedmann)
    matches the host fused loop to fp64 accuracy — with and without
    in-kernel periodic stencil reads."""
    monkeypatch.setenv("PYSTELLA_PERIODIC", periodic)
    from pystella_amd.fusion import (
        DeviceFriedmannLoop, FusedLaplacianReduction, StencilRKStepper)
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.3, 0.3)
    dt = 0.005
    gsize = float(np.prod(grid_shape))

    def pot(f):
        return f[0]**2 / 2 + f[0]**2 * f[1]**2 / 4

    sector = ps.ScalarSector(2, potential=pot)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(11)
    f0 = 0.2 + 0.01 * torch.rand((2,) + pad, dtype=torch.float64)
    d0 = 0.01 * torch.rand((2,) + pad, dtype=torch.float64)

    def make(dev):
        ff, df = f0.clone().to(dev), d0.clone().to(dev)
        st = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                              halo_shape=h, rank_shape=grid_shape,
                              dt=dt, reducers=sector, grid_size=gsize,
                              callback=get_rho_and_p)
        red = FusedLaplacianReduction(
            decomp, sector, derivs, halo_shape=h,
            callback=get_rho_and_p, rank_shape=grid_shape,
            grid_size=gsize, store_lap=False)
        e0 = red(f=ff, dfdt=df, a=np.ones(1))
        ex = ps.Expansion(e0["total"], ps.LowStorageRK54)
        arrays = {"f": ff, "dfdt": df, "f_next": torch.zeros_like(ff)}
        decomp.share_halos(arrays["f"])
        return st, ex, arrays, e0

    nsteps = 3
    # host loop (GPU kernels, host Friedmann)
    st, ex, arrays, energy = make("cuda")
    for _ in range(nsteps):
        for s in range(st.num_stages):
            e_in = st(s, a=ex.a, hubble=ex.hubble, **arrays)
            arrays["f"], arrays["f_next"] = \
                arrays["f_next"], arrays["f"]
            decomp.share_halos(arrays["f"])
            ex.step(s, e_in["total"], e_in["pressure"], dt)
    torch.cuda.synchronize()
    f_host = arrays["f"].cpu().clone()
    a_host = float(ex.a[0])

    # device loop (no host syncs)
    st2, ex2, arrays2, _ = make("cuda")
    dl = DeviceFriedmannLoop(st2, decomp, ex2, gsize, dt)
    for _ in range(nsteps):
        dl.step(arrays2)
    state = dl.read_state()
    f_dev = arrays2["f"].cpu()

    assert abs(state["a"] - a_host) < 1e-13 * abs(a_host), \
        (state["a"], a_host)
    cut = (slice(None),) + (slice(h, -h),) * 3
    assert (f_dev[cut] - f_host[cut]).abs().max().item() < 1e-13


@requires_gpu
@pytest.mark.parametrize("tdtype", [torch.float64, torch.float32])
def test_relax_gpu_matches_cpu(tdtype, n=32, h=1):
    """Jacobi relaxation (multigrid smoother) on GPU vs CPU, fp64 and
    fp32 (`using real = float` kernels)."""
    from pystella_amd.multigrid import JacobiIterator
    from pystella_amd.field import Field, shift_fields

    decomp = ps.DomainDecomposition((1, 1, 1), h,
                                    rank_shape=(n, n, n))
    dx = (2 * np.pi / n,) * 3
    f = Field("f", offset="h")
    rho = Field("rho", offset="h")
    lap = sum(
        (shift_fields(f, tuple(s * int(mu == d) for mu in range(3)))
         - 2 * f
         + shift_fields(f, tuple(-s * int(mu == d) for mu in range(3))))
        for d in range(3) for s in [1]) / var("dx")[0]**2
    problems = {f: (lap, rho)}
    solver = JacobiIterator(decomp, problems, halo_shape=h,
                            fixed_parameters=dict(omega=0.8))

    torch.manual_seed(7)
    pad = (n + 2 * h,) * 3
    rho_t = (torch.rand(pad, dtype=torch.float64) - 0.5).to(tdtype)
    decomp.share_halos(rho_t)

    def run(device):
        ff = torch.zeros(pad, dtype=tdtype, device=device)
        tmp = torch.zeros_like(ff)
        solver(decomp, iterations=20, f=ff, tmp_f=tmp,
               rho=rho_t.to(device), dx=np.array(dx))
        return ff.cpu()

    fc = run("cpu")
    fg = run("cuda")
    torch.cuda.synchronize()
    tol = 1e-12 if tdtype == torch.float64 else 1e-4
    denom = fc.abs().max().item() + 1e-30
    assert (fg - fc).abs().max().item() / denom < tol


@requires_gpu
@pytest.mark.parametrize("grid_shape", [(16, 16, 16), (64, 48, 32)])
def test_tt_projection_mfma_vs_oracle(grid_shape):
    """Transverse-traceless projection on the f64 matrix cores
    (csrc/tt_mfma.hip) vs the torch oracle: transversality, traceless-
    ness and exact agreement."""
    from pystella_amd.fourier import DFT
    decomp = ps.DomainDecomposition((1, 1, 1), 0,
                                    rank_shape=grid_shape)
    fft = DFT(decomp, grid_shape=grid_shape, dtype=np.float64,
              device="cuda")
    dk = (2 * np.pi / 5,) * 3
    dx = (5 / grid_shape[0],) * 3
    proj = ps.Projector(fft, 1, dk, dx)

    kshape = fft.shape(True)
    torch.manual_seed(9)
    hij = (torch.randn((6,) + kshape, dtype=torch.float64)
           + 1j * torch.randn((6,) + kshape, dtype=torch.float64)
           ).to(torch.complex128)

    # oracle: torch path (runs when tensors are CPU)
    want = hij.clone()
    proj_cpu = ps.Projector(
        DFT(decomp, grid_shape=grid_shape, dtype=np.float64,
            device="cpu"), 1, dk, dx)
    proj_cpu.transverse_traceless(want)

    got = hij.clone().cuda().contiguous()
    proj.transverse_traceless(got)
    torch.cuda.synchronize()
    got = got.cpu()

    err = (got - want).abs().max().item()
    scale = want.abs().max().item()
    assert err < 1e-12 * max(scale, 1.0), (err, scale)

    # tracelessness of the MFMA result
    tr = got[0] + got[3] + got[5]
    assert tr.abs().max().item() < 1e-11 * max(scale, 1.0)


@requires_gpu
def test_wrap_star_kernel(grid_shape=(12, 10, 14), h=2):
    """Fused periodic-wrap kernel vs the torch slicing path (face
    halos; star contract)."""
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(13)
    base = torch.rand((3,) + pad, dtype=torch.float64)

    ref = base.clone()
    decomp.share_halos(ref)         # CPU slicing path (with corners)

    got = base.clone().cuda()
    handle = decomp.share_halos_start(got)   # GPU fused wrap kernel
    handle.finish()
    torch.cuda.synchronize()
    got = got.cpu()

    nx, ny, nz = grid_shape
    regions = [
        (slice(h, h + nx), slice(h, h + ny), slice(h, h + nz)),
        (slice(0, h), slice(h, h + ny), slice(h, h + nz)),
        (slice(h + nx, None), slice(h, h + ny), slice(h, h + nz)),
        (slice(h, h + nx), slice(0, h), slice(h, h + nz)),
        (slice(h, h + nx), slice(h + ny, None), slice(h, h + nz)),
        (slice(h, h + nx), slice(h, h + ny), slice(0, h)),
        (slice(h, h + nx), slice(h, h + ny), slice(h + nz, None)),
    ]
    for reg in regions:
        r = (slice(None),) + reg
        assert torch.equal(got[r], ref[r]), reg


@requires_gpu
def test_stage_fused_gw_gpu(grid_shape=(16, 16, 16)):
    """Multi-family ring stage kernels (scalar + GW tensor groups) on
    GPU vs the CPU oracle."""
    from pystella_amd.fusion import StencilRKStepper
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.31, 0.32)
    dt = 0.01
    gsize = float(np.prod(grid_shape))

    def pot(f):
        return f[0]**2 / 2 + f[0]**2 * f[1]**2 / 4

    sector = ps.ScalarSector(2, potential=pot)
    tensor = ps.TensorPerturbationSector([sector])
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(23)
    f0 = torch.rand((2,) + pad, dtype=torch.float64)
    d0 = torch.rand((2,) + pad, dtype=torch.float64)
    h0 = 0.01 * torch.rand((6,) + pad, dtype=torch.float64)
    hd0 = 0.01 * torch.rand((6,) + pad, dtype=torch.float64)
    a = np.ones(1)
    hub = 0.1 * np.ones(1)
    cut = (slice(None),) + (slice(h, -h),) * 3

    def run(device):
        fst = StencilRKStepper(
            ps.LowStorageRK54, [sector, tensor], derivs, halo_shape=h,
            rank_shape=grid_shape, dt=dt, reducers=sector,
            grid_size=gsize, callback=get_rho_and_p)
        arrays = {
            "f": f0.clone().to(device), "dfdt": d0.clone().to(device),
            "hij": h0.clone().to(device),
            "dhijdt": hd0.clone().to(device),
            "dfdx": torch.zeros((2, 3) + grid_shape,
                                dtype=torch.float64, device=device)}
        arrays["f_next"] = torch.zeros_like(arrays["f"])
        arrays["hij_next"] = torch.zeros_like(arrays["hij"])
        decomp.share_halos(arrays["f"])
        decomp.share_halos(arrays["hij"])
        energies = []
        for s in range(fst.num_stages):
            derivs(fx=arrays["f"], grd=arrays["dfdx"])
            energies.append(fst(s, a=a, hubble=hub, **arrays))
            for name in fst.pingpong:
                arrays[name], arrays[f"{name}_next"] = \
                    arrays[f"{name}_next"], arrays[name]
                decomp.share_halos(arrays[name])
        return arrays, energies

    arr_c, en_c = run("cpu")
    arr_g, en_g = run("cuda")
    torch.cuda.synchronize()
    for s, (ec, eg) in enumerate(zip(en_c, en_g)):
        assert np.allclose(ec["total"], eg["total"], rtol=1e-12), s
    for name in ("f", "dfdt", "hij", "dhijdt"):
        err = (arr_g[name].cpu()[cut] - arr_c[name][cut]).abs().max()
        assert err.item() < 1e-12, (name, err)


@requires_gpu
def test_fourier_stack_gpu_matches_cpu(grid_shape=(16, 16, 16)):
    """Power spectra on the GPU (rocFFT via torch.fft) match the CPU
    path; Rayleigh init runs end to end on-device."""
    from pystella_amd.fourier import DFT, PowerSpectra, RayleighGenerator
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    L = 5.0
    dk = (2 * np.pi / L,) * 3
    dx = (L / grid_shape[0],) * 3

    def run(device, fx):
        fft = DFT(decomp, grid_shape=grid_shape, dtype=np.float64,
                  device=device)
        spec = PowerSpectra(decomp, fft, dk, L**3)
        return np.asarray(spec(fx.to(device)))

    # deterministic field => binned spectra must agree exactly
    ax = torch.arange(grid_shape[0], dtype=torch.float64) * dx[0]
    s1 = torch.sin(2 * np.pi * ax / L)
    fx = (s1[:, None, None] * s1[None, :, None]
          + 0.3 * s1[None, None, :])
    s_c = run("cpu", fx)
    s_g = run("cuda", fx)
    torch.cuda.synchronize()
    assert np.allclose(s_c, s_g, rtol=1e-10), (s_c, s_g)

    # Rayleigh init runs on-device end to end (GPU Philox stream)
    fft_g = DFT(decomp, grid_shape=grid_shape, dtype=np.float64,
                device="cuda")
    gen = RayleighGenerator(fft=fft_g, dk=dk, volume=L**3, seed=123)
    fg = torch.zeros(grid_shape, dtype=torch.float64, device="cuda")
    gen.init_field(fg)
    torch.cuda.synchronize()
    assert torch.isfinite(fg).all()
    assert fg.abs().max().item() > 0


@requires_gpu
def test_checkpoint_roundtrip_gpu(tmp_path=None):
    """Checkpoint save/restore with GPU-resident fields."""
    import tempfile
    from pystella_amd.checkpoint import load_checkpoint, save_checkpoint
    h = 2
    grid = (12, 12, 12)
    decomp = ps.DomainDecomposition((1, 1, 1), h, grid_shape=grid)
    pad = tuple(n + 2 * h for n in grid)
    torch.manual_seed(31)
    f = torch.rand((2,) + pad, dtype=torch.float64, device="cuda")
    with tempfile.TemporaryDirectory() as d:
        path = f"{d}/ckpt.pt"
        save_checkpoint(path, decomp, {"f": f}, attrs={"t": 1.5})
        g = torch.zeros_like(f)
        attrs = load_checkpoint(path, decomp, {"f": g})
        torch.cuda.synchronize()
        cut = (slice(None),) + (slice(h, -h),) * 3
        assert torch.equal(g[cut].cpu(), f[cut].cpu())
        assert attrs["t"] == 1.5


@requires_gpu
def test_profiler_reports_gpu():
    """HIP-event Profiler measures a real kernel region."""
    from pystella_amd.profiling import Profiler
    prof = Profiler(enabled=True)
    x = torch.rand(1 << 22, dtype=torch.float64, device="cuda")
    y = torch.empty_like(x)
    nbytes = x.numel() * 8 * 2
    for _ in range(5):
        with prof.region("copy", bytes=nbytes):
            y.copy_(x)
    torch.cuda.synchronize()
    rep = prof.report()
    assert "copy" in rep
    assert prof.calls["copy"] == 5
    assert prof.times["copy"] > 0


@requires_gpu
@pytest.mark.parametrize("tdtype", [torch.float64, torch.float32])
def test_rbgs_gpu_matches_cpu(tdtype, n=32, h=1):
    """Red-black GS smoother on GPU (in-place masked half-sweeps,
    parity intrinsic) vs CPU."""
    from pystella_amd.multigrid import RedBlackIterator
    from pystella_amd.field import Field, shift_fields

    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=(n, n, n))
    dx = (2 * np.pi / n,) * 3
    f = Field("f", offset="h")
    rho = Field("rho", offset="h")
    lap = sum(
        (shift_fields(f, tuple(s * int(m == d) for m in range(3)))
         - 2 * f
         + shift_fields(f, tuple(-s * int(m == d) for m in range(3))))
        for d in range(3) for s in [1]) / var("dx")[0]**2
    solver = RedBlackIterator(decomp, {f: (lap, rho)}, halo_shape=h,
                              fixed_parameters=dict(omega=1.0))
    torch.manual_seed(17)
    pad = (n + 2 * h,) * 3
    rho_t = (torch.rand(pad, dtype=torch.float64) - 0.5).to(tdtype)
    decomp.share_halos(rho_t)

    def run(device):
        ff = torch.zeros(pad, dtype=tdtype, device=device)
        solver(decomp, iterations=15, f=ff, rho=rho_t.to(device),
               dx=np.array(dx))
        return ff.cpu()

    fc = run("cpu")
    fg = run("cuda")
    torch.cuda.synchronize()
    tol = 1e-12 if tdtype == torch.float64 else 1e-4
    denom = fc.abs().max().item() + 1e-30
    assert (fg - fc).abs().max().item() / denom < tol


@requires_gpu
def test_classical_rk4_gpu(grid_shape=(16, 16, 16), h=1):
    """Multi-copy classical RK4 stage kernels (leading q axis) on GPU
    vs CPU."""
    def potential(f):
        return f[0]**4 / 4

    sector = ps.ScalarSector(1, potential=potential)
    dt = 1e-3
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(3)
    nc = ps.RungeKutta4.num_copies
    f = torch.rand((nc, 1) + pad, dtype=torch.float64)
    d = torch.rand((nc, 1) + pad, dtype=torch.float64)
    lap = torch.rand((nc, 1) + grid_shape, dtype=torch.float64)
    a = np.array([1.0] * nc)
    hub = np.array([0.1] * nc)

    def run(device):
        st = ps.RungeKutta4([sector], dt=dt, halo_shape=h,
                            rank_shape=grid_shape)
        ff, dd, ll = (f.clone().to(device), d.clone().to(device),
                      lap.clone().to(device))
        for s in range(st.num_stages):
            st(s, a=a, hubble=hub, f=ff, dfdt=dd, lap_f=ll)
        return ff.cpu(), dd.cpu()

    fc, dc = run("cpu")
    fg, dg = run("cuda")
    torch.cuda.synchronize()
    assert (fg - fc).abs().max().item() < 1e-13
    assert (dg - dc).abs().max().item() < 1e-13


@requires_gpu
@pytest.mark.parametrize("h", [1, 3, 4])
def test_stage_fused_energy_gpu_halo_orders(h, grid_shape=(24, 24, 24)):
    """Energy-fused ring stage kernel at h=1/3/4 (h=2 covered above)."""
    from pystella_amd.fusion import StencilRKStepper
    from pystella_amd.sectors import get_rho_and_p
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.31, 0.32)
    dt = 0.01
    gsize = float(np.prod(grid_shape))
    sector = ps.ScalarSector(2, potential=lambda f: f[0]**2 / 2)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(41 + h)
    f0 = torch.rand((2,) + pad, dtype=torch.float64)
    d0 = torch.rand((2,) + pad, dtype=torch.float64)
    a = np.ones(1)
    hub = 0.1 * np.ones(1)
    cut = (slice(None),) + (slice(h, -h),) * 3

    def run(device):
        fst = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                               halo_shape=h, rank_shape=grid_shape,
                               dt=dt, reducers=sector, grid_size=gsize,
                               callback=get_rho_and_p)
        arrays = {"f": f0.clone().to(device),
                  "dfdt": d0.clone().to(device)}
        arrays["f_next"] = torch.zeros_like(arrays["f"])
        decomp.share_halos(arrays["f"])
        es = []
        for s in range(fst.num_stages):
            es.append(fst(s, a=a, hubble=hub, **arrays))
            arrays["f"], arrays["f_next"] = \
                arrays["f_next"], arrays["f"]
            decomp.share_halos(arrays["f"])
        return arrays, es

    ac, ec = run("cpu")
    ag, eg = run("cuda")
    torch.cuda.synchronize()
    for s in range(len(ec)):
        assert np.allclose(ec[s]["total"], eg[s]["total"],
                           rtol=1e-12), s
    assert (ag["f"].cpu()[cut] - ac["f"][cut]).abs().max() < 1e-12
    assert (ag["dfdt"].cpu()[cut] - ac["dfdt"][cut]).abs().max() < 1e-12


@requires_gpu
def test_multigrid_fas_gpu(n=64, h=1):
    """Full FAS multigrid solve on GPU (fp32) matches the manufactured
    solution."""
    from pystella_amd.field import Field, shift_fields
    from pystella_amd.multigrid import (
        FullApproximationScheme, RedBlackIterator, v_cycle)
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=(n, n, n))
    L = 2 * np.pi
    dx = (L / n,) * 3
    f = Field("f", offset="h")
    rho = Field("rho", offset="h")
    lap = sum(
        (shift_fields(f, tuple(s * int(m == d) for m in range(3)))
         - 2 * f
         + shift_fields(f, tuple(-s * int(m == d) for m in range(3))))
        for d in range(3) for s in [1]) / var("dx")[0]**2
    solver = RedBlackIterator(decomp, {f: (lap, rho)}, halo_shape=h,
                              fixed_parameters=dict(omega=1.0))
    mg = FullApproximationScheme(solver, halo_shape=h)

    ax = torch.arange(n, dtype=torch.float64) * dx[0]
    s1 = torch.sin(ax)
    f_exact = s1[:, None, None] * s1[None, :, None] * s1[None, None, :]
    pad = (n + 2 * h,) * 3
    rho_t = torch.zeros(pad, dtype=torch.float64)
    rho_t[h:-h, h:-h, h:-h] = -3.0 * f_exact
    rho_t = rho_t.to(torch.float32).cuda()
    decomp.share_halos(rho_t)
    ff = torch.zeros(pad, dtype=torch.float32, device="cuda")
    mg(decomp, dx0=dx, cycle=v_cycle(6, 12, 3), f=ff, rho=rho_t)
    torch.cuda.synchronize()
    got = ff[h:-h, h:-h, h:-h].double().cpu()
    got -= got.mean()
    want = f_exact - f_exact.mean()
    rel = (got - want).abs().max().item() / want.abs().max().item()
    assert rel < 0.05, rel


@requires_gpu
def test_device_friedmann_loop_gw_gpu(grid_shape=(16, 16, 16)):
    """Multi-family device-resident loop (scalar + GW tensor) matches
    the host fused loop."""
    from pystella_amd.fusion import (
        DeviceFriedmannLoop, FusedLaplacianReduction, StencilRKStepper)
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.3, 0.3)
    dt = 0.005
    gsize = float(np.prod(grid_shape))
    sector = ps.ScalarSector(2, potential=lambda f: f[0]**2 / 2)
    tensor = ps.TensorPerturbationSector([sector])
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(51)
    f0 = 0.2 + 0.01 * torch.rand((2,) + pad, dtype=torch.float64)
    d0 = 0.01 * torch.rand((2,) + pad, dtype=torch.float64)
    h0 = 0.01 * torch.rand((6,) + pad, dtype=torch.float64)
    hd0 = 0.01 * torch.rand((6,) + pad, dtype=torch.float64)

    def make():
        st = StencilRKStepper(ps.LowStorageRK54, [sector, tensor],
                              derivs, halo_shape=h,
                              rank_shape=grid_shape, dt=dt,
                              reducers=sector, grid_size=gsize,
                              callback=get_rho_and_p, inline_grad=True)
        red = FusedLaplacianReduction(
            decomp, sector, derivs, halo_shape=h,
            callback=get_rho_and_p, rank_shape=grid_shape,
            grid_size=gsize, store_lap=False)
        arrays = {
            "f": f0.clone().cuda(), "dfdt": d0.clone().cuda(),
            "hij": h0.clone().cuda(), "dhijdt": hd0.clone().cuda()}
        arrays["f_next"] = torch.zeros_like(arrays["f"])
        arrays["hij_next"] = torch.zeros_like(arrays["hij"])
        e0 = red(f=arrays["f"], dfdt=arrays["dfdt"], a=np.ones(1))
        ex = ps.Expansion(e0["total"], ps.LowStorageRK54)
        for name in st.pingpong:
            decomp.share_halos(arrays[name])
        return st, ex, arrays

    nsteps = 2
    st, ex, arrays = make()
    for _ in range(nsteps):
        for s in range(st.num_stages):
            e_in = st(s, a=ex.a, hubble=ex.hubble, **arrays)
            for name in st.pingpong:
                arrays[name], arrays[f"{name}_next"] = \
                    arrays[f"{name}_next"], arrays[name]
                decomp.share_halos(arrays[name])
            ex.step(s, e_in["total"], e_in["pressure"], dt)
    torch.cuda.synchronize()
    host = {k: v.cpu().clone() for k, v in arrays.items()}
    a_host = float(ex.a[0])

    st2, ex2, arrays2 = make()
    dl = DeviceFriedmannLoop(st2, decomp, ex2, gsize, dt)
    for _ in range(nsteps):
        dl.step(arrays2)
    state = dl.read_state()
    cut = (slice(None),) + (slice(h, -h),) * 3
    assert abs(state["a"] - a_host) < 1e-12 * abs(a_host)
    for name in ("f", "dfdt", "hij", "dhijdt"):
        err = (arrays2[name].cpu()[cut] - host[name][cut]).abs().max()
        assert err.item() < 1e-12, (name, err)


@requires_gpu
def test_spectral_collocator_gpu(grid_shape=(16, 16, 16)):
    """Spectral derivatives (rocFFT path) on GPU vs CPU."""
    from pystella_amd.fourier import DFT
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    L = 2 * np.pi
    dk = (2 * np.pi / L,) * 3

    def run(device):
        fft = DFT(decomp, grid_shape=grid_shape, dtype=np.float64,
                  device=device)
        coll = ps.SpectralCollocator(fft, dk)
        ax = torch.arange(grid_shape[0], dtype=torch.float64) \
            * (L / grid_shape[0])
        s1 = torch.sin(ax)
        fx = (s1[:, None, None] * s1[None, :, None]
              * s1[None, None, :]).to(device)
        lap = torch.zeros(grid_shape, dtype=torch.float64,
                          device=device)
        grd = torch.zeros((3,) + grid_shape, dtype=torch.float64,
                          device=device)
        coll(fx=fx, lap=lap, grd=grd)
        return lap.cpu(), grd.cpu()

    lap_c, grd_c = run("cpu")
    lap_g, grd_g = run("cuda")
    torch.cuda.synchronize()
    assert (lap_g - lap_c).abs().max().item() < 1e-12
    assert (grd_g - grd_c).abs().max().item() < 1e-12


@requires_gpu
def test_device_loop_region_split_gpu(grid_shape=(32, 32, 32)):
    """Force the interior/boundary-slab split on one GPU (simulating a
    z-decomposed rank, halos supplied by the periodic wrap): the
    multi-launch partials offsets must reproduce the single-launch
    result exactly."""
    from pystella_amd.fusion import (
        DeviceFriedmannLoop, FusedLaplacianReduction, StencilRKStepper)
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.3, 0.3)
    dt = 0.005
    gsize = float(np.prod(grid_shape))
    sector = ps.ScalarSector(2, potential=lambda f: f[0]**2 / 2)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(61)
    f0 = 0.2 + 0.01 * torch.rand((2,) + pad, dtype=torch.float64)
    d0 = 0.01 * torch.rand((2,) + pad, dtype=torch.float64)

    def run(split):
        st = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                              halo_shape=h, rank_shape=grid_shape,
                              dt=dt, reducers=sector, grid_size=gsize,
                              callback=get_rho_and_p)
        red = FusedLaplacianReduction(
            decomp, sector, derivs, halo_shape=h,
            callback=get_rho_and_p, rank_shape=grid_shape,
            grid_size=gsize, store_lap=False)
        arrays = {"f": f0.clone().cuda(), "dfdt": d0.clone().cuda()}
        arrays["f_next"] = torch.zeros_like(arrays["f"])
        e0 = red(f=arrays["f"], dfdt=arrays["dfdt"], a=np.ones(1))
        ex = ps.Expansion(e0["total"], ps.LowStorageRK54)
        decomp.share_halos(arrays["f"])
        dl = DeviceFriedmannLoop(st, decomp, ex, gsize, dt)
        if split:
            nx, ny, nz = grid_shape

            def fake_regions(rank_shape, split_axes=None):
                return ((0, nx, 0, ny, h, nz - h),
                        [(0, nx, 0, ny, 0, h),
                         (0, nx, 0, ny, nz - h, nz)])

            dl._regions = fake_regions
        for _ in range(2):
            dl.step(arrays)
        return arrays["f"].cpu(), dl.read_state()

    f_one, st_one = run(False)
    f_spl, st_spl = run(True)
    cut = (slice(None),) + (slice(h, -h),) * 3
    assert torch.equal(f_one[cut], f_spl[cut])
    assert st_one["a"] == st_spl["a"]
    assert st_one["energy"] == st_spl["energy"]


@requires_gpu
@pytest.mark.gpu
def test_fused_projector_kernels_vs_oracle(grid_shape=(24, 20, 16)):
    """All fused single-launch projector kernels (transversify,
    vec<->pol, decompose/decomp_to_vec, tensor<->pol; backend/hip.py
    projector_op) vs the torch oracle path, including aliased
    outputs (plus/minus as views of the input array, as PowerSpectra
    uses them)."""
    from pystella_amd.fourier import DFT
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    dk = (2 * np.pi / 5,) * 3
    dx = (5 / grid_shape[0],) * 3
    fft_g = DFT(decomp, grid_shape=grid_shape, dtype=np.float64,
                device="cuda")
    fft_c = DFT(decomp, grid_shape=grid_shape, dtype=np.float64,
                device="cpu")
    pg = ps.Projector(fft_g, 1, dk, dx)
    pc = ps.Projector(fft_c, 1, dk, dx)
    kshape = tuple(fft_c.shape(True))
    torch.manual_seed(31)

    def rnd(n):
        return (torch.randn((n,) + kshape, dtype=torch.float64)
                + 1j * torch.randn((n,) + kshape, dtype=torch.float64)
                ).to(torch.complex128)

    def check(name, got, want):
        err = (got.cpu() - want).abs().max().item()
        scale = max(want.abs().max().item(), 1.0)
        assert err < 1e-12 * scale, (name, err, scale)

    # transversify (in place)
    v0 = rnd(3)
    vc = v0.clone()
    pc.transversify(vector=vc)
    vg = v0.clone().cuda().contiguous()
    pg.transversify(vector=vg)
    check("transversify", vg, vc)

    # vec_to_pol / pol_to_vec with separate buffers
    v0 = rnd(3)
    pl_c = torch.empty(kshape, dtype=torch.complex128)
    mi_c = torch.empty_like(pl_c)
    pc.vec_to_pol(plus=pl_c, minus=mi_c, vector=v0.clone())
    pl_g = torch.empty(kshape, dtype=torch.complex128,
                       device="cuda")
    mi_g = torch.empty_like(pl_g)
    pg.vec_to_pol(plus=pl_g, minus=mi_g, vector=v0.clone().cuda())
    check("vec_to_pol+", pl_g, pl_c)
    check("vec_to_pol-", mi_g, mi_c)

    out_c = torch.empty((3,) + kshape, dtype=torch.complex128)
    pc.pol_to_vec(plus=pl_c, minus=mi_c, vector=out_c)
    out_g = torch.empty((3,) + kshape, dtype=torch.complex128,
                        device="cuda")
    pg.pol_to_vec(plus=pl_g, minus=mi_g, vector=out_g)
    check("pol_to_vec", out_g, out_c)

    # decompose_vector with ALIASED outputs (views of the input),
    # exactly as PowerSpectra.vector_decomposition calls it
    for tak in (False, True):
        v0 = rnd(3)
        vc = v0.clone()
        pc.decompose_vector(vector=vc, plus=vc[0], minus=vc[1],
                            lng=vc[2], times_abs_k=tak)
        vg = v0.clone().cuda().contiguous()
        pg.decompose_vector(vector=vg, plus=vg[0], minus=vg[1],
                            lng=vg[2], times_abs_k=tak)
        check(f"decompose_vector tak={tak}", vg, vc)

    # decomp_to_vec
    for tak in (False, True):
        d0 = rnd(3)
        out_c = torch.empty((3,) + kshape, dtype=torch.complex128)
        pc.decomp_to_vec(plus=d0[0].clone(), minus=d0[1].clone(),
                         lng=d0[2].clone(), vector=out_c,
                         times_abs_k=tak)
        dg = d0.clone().cuda().contiguous()
        out_g = torch.empty((3,) + kshape, dtype=torch.complex128,
                            device="cuda")
        pg.decomp_to_vec(plus=dg[0], minus=dg[1], lng=dg[2],
                         vector=out_g, times_abs_k=tak)
        check(f"decomp_to_vec tak={tak}", out_g, out_c)

    # tensor_to_pol with aliased plus/minus (as gw_polarization does)
    h0 = rnd(6)
    hc = h0.clone()
    pc.tensor_to_pol(plus=hc[0], minus=hc[1], hij=hc)
    hg = h0.clone().cuda().contiguous()
    pg.tensor_to_pol(plus=hg[0], minus=hg[1], hij=hg)
    check("tensor_to_pol", hg[:2], hc[:2])

    # pol_to_tensor
    d0 = rnd(2)
    h_c = torch.empty((6,) + kshape, dtype=torch.complex128)
    pc.pol_to_tensor(plus=d0[0].clone(), minus=d0[1].clone(), hij=h_c)
    dg = d0.clone().cuda().contiguous()
    h_g = torch.empty((6,) + kshape, dtype=torch.complex128,
                      device="cuda")
    pg.pol_to_tensor(plus=dg[0], minus=dg[1], hij=h_g)
    check("pol_to_tensor", h_g, h_c)


@requires_gpu
@pytest.mark.parametrize("tdtype", [torch.float64, torch.float32])
def test_lds_stencil_vs_cpu(tdtype, grid_shape=(24, 20, 16)):
    """Generic LDS-staged Stencil kernel (backend/hip.py JitStencil):
    star reads from LDS tiles + x register rings, mixed corner reads
    from global, ping-pong output — vs the CPU torcheval oracle."""
    from pystella_amd.field import Field, shift_fields
    from pystella_amd.stencil import Stencil
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    f = Field("f", offset="h")
    g = Field("g", offset="h")
    out = Field("out", offset=0)
    # star reads of f at two radii, x-ring reads, a corner (mixed)
    # read, a second prefetched field, and a scalar parameter
    expr = (2.0 * f
            + shift_fields(f, (1, 0, 0)) + shift_fields(f, (-2, 0, 0))
            + 0.5 * (shift_fields(f, (0, 1, 0)) + shift_fields(f, (0, -2, 0)))
            + 0.25 * (shift_fields(f, (0, 0, 2)) + shift_fields(f, (0, 0, -1)))
            + 0.125 * shift_fields(f, (1, 1, 0))     # mixed -> global
            + shift_fields(g, (0, 2, -1)) * var("c"))
    st = Stencil({out: expr}, halo_shape=h, rank_shape=grid_shape)

    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(11)
    fv = torch.rand(pad, dtype=tdtype)
    gv = torch.rand(pad, dtype=tdtype)
    out_c = torch.zeros(grid_shape, dtype=tdtype)
    st(f=fv, g=gv, out=out_c, c=1.5)

    out_g = torch.zeros(grid_shape, dtype=tdtype, device="cuda")
    st2 = Stencil({out: expr}, halo_shape=h, rank_shape=grid_shape)
    st2(f=fv.cuda(), g=gv.cuda(), out=out_g, c=1.5)
    torch.cuda.synchronize()
    from pystella_amd.backend.hip import JitStencil
    assert isinstance(st2._hip_kernel, JitStencil), \
        "expected the LDS-staged kernel to be selected"
    tol = 1e-12 if tdtype == torch.float64 else 1e-5
    err = (out_g.cpu() - out_c).abs().max().item()
    assert err < tol * (1 + out_c.abs().max().item()), err


@requires_gpu
def test_device_loop_noncubic_split_gpu(grid_shape=(16, 32, 24)):
    """Non-cubic rank shape with a forced x-axis split — the exact
    geometry of an N=2 (2,1,1) rank (256x512x512 at the bench scale).
    Multi-launch result must equal the single-launch result."""
    from pystella_amd.fusion import (
        DeviceFriedmannLoop, FusedLaplacianReduction, StencilRKStepper)
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.31, 0.29)
    dt = 0.005
    gsize = float(np.prod(grid_shape))
    sector = ps.ScalarSector(2, potential=lambda f: f[0]**2 / 2)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(71)
    f0 = 0.2 + 0.01 * torch.rand((2,) + pad, dtype=torch.float64)
    d0 = 0.01 * torch.rand((2,) + pad, dtype=torch.float64)

    def run(split):
        st = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                              halo_shape=h, rank_shape=grid_shape,
                              dt=dt, reducers=sector, grid_size=gsize,
                              callback=get_rho_and_p)
        red = FusedLaplacianReduction(
            decomp, sector, derivs, halo_shape=h,
            callback=get_rho_and_p, rank_shape=grid_shape,
            grid_size=gsize, store_lap=False)
        arrays = {"f": f0.clone().cuda(), "dfdt": d0.clone().cuda()}
        arrays["f_next"] = torch.zeros_like(arrays["f"])
        e0 = red(f=arrays["f"], dfdt=arrays["dfdt"], a=np.ones(1))
        ex = ps.Expansion(e0["total"], ps.LowStorageRK54)
        decomp.share_halos(arrays["f"])
        dl = DeviceFriedmannLoop(st, decomp, ex, gsize, dt)
        if split:
            nx, ny, nz = grid_shape

            def fake_regions(rank_shape, split_axes=None):
                return ((h, nx - h, 0, ny, 0, nz),
                        [(0, h, 0, ny, 0, nz),
                         (nx - h, nx, 0, ny, 0, nz)])

            dl._regions = fake_regions
        for _ in range(2):
            dl.step(arrays)
        return arrays["f"].cpu(), dl.read_state()

    f_one, st_one = run(False)
    f_spl, st_spl = run(True)
    cut = (slice(None),) + (slice(h, -h),) * 3
    assert torch.equal(f_one[cut], f_spl[cut])
    assert st_one["a"] == st_spl["a"]
    assert st_one["energy"] == st_spl["energy"]


@requires_gpu
def test_device_loop_full_shell_split_gpu(grid_shape=(24, 24, 24)):
    """Force the FULL 6-slab split (the (2,2,2) N=8 rank geometry):
    the multi-box shell kernel's flat-block decode across mixed x/y/z
    slab orientations must reproduce the single-launch result
    bit-exactly."""
    from pystella_amd.fusion import (
        DeviceFriedmannLoop, FusedLaplacianReduction, StencilRKStepper)
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.31, 0.29)
    dt = 0.005
    gsize = float(np.prod(grid_shape))
    sector = ps.ScalarSector(2, potential=lambda f: f[0]**2 / 2)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(81)
    f0 = 0.2 + 0.01 * torch.rand((2,) + pad, dtype=torch.float64)
    d0 = 0.01 * torch.rand((2,) + pad, dtype=torch.float64)

    def run(split):
        st = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                              halo_shape=h, rank_shape=grid_shape,
                              dt=dt, reducers=sector, grid_size=gsize,
                              callback=get_rho_and_p)
        red = FusedLaplacianReduction(
            decomp, sector, derivs, halo_shape=h,
            callback=get_rho_and_p, rank_shape=grid_shape,
            grid_size=gsize, store_lap=False)
        arrays = {"f": f0.clone().cuda(), "dfdt": d0.clone().cuda()}
        arrays["f_next"] = torch.zeros_like(arrays["f"])
        e0 = red(f=arrays["f"], dfdt=arrays["dfdt"], a=np.ones(1))
        ex = ps.Expansion(e0["total"], ps.LowStorageRK54)
        decomp.share_halos(arrays["f"])
        dl = DeviceFriedmannLoop(st, decomp, ex, gsize, dt)
        if split:
            nx, ny, nz = grid_shape

            def fake_regions(rank_shape, split_axes=None):
                interior = (h, nx - h, h, ny - h, h, nz - h)
                slabs = [
                    (0, h, 0, ny, 0, nz),
                    (nx - h, nx, 0, ny, 0, nz),
                    (h, nx - h, 0, h, 0, nz),
                    (h, nx - h, ny - h, ny, 0, nz),
                    (h, nx - h, h, ny - h, 0, h),
                    (h, nx - h, h, ny - h, nz - h, nz),
                ]
                return interior, slabs

            dl._regions = fake_regions
        for _ in range(2):
            dl.step(arrays)
        return arrays["f"].cpu(), dl.read_state()

    f_one, st_one = run(False)
    f_spl, st_spl = run(True)
    cut = (slice(None),) + (slice(h, -h),) * 3
    assert torch.equal(f_one[cut], f_spl[cut])
    assert st_one["a"] == st_spl["a"]
    assert st_one["energy"] == st_spl["energy"]


@requires_gpu
def test_smoother_star_wrap_bit_equality(n=32, h=1):
    """The smoothing loop's fused face-wrap fast path (star operators,
    fully-local decomp) is bit-identical to full corner-propagating
    shares — star kernels never read the corners the fast path leaves
    stale, and the final share is always full."""
    from pystella_amd.field import Field, shift_fields
    from pystella_amd.multigrid import RedBlackIterator

    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=(n, n, n))
    dx = (2 * np.pi / n,) * 3
    f = Field("f", offset="h")
    rho = Field("rho", offset="h")
    lap = sum(
        (shift_fields(f, tuple(s * int(m == d) for m in range(3)))
         - 2 * f
         + shift_fields(f, tuple(-s * int(m == d) for m in range(3))))
        for d in range(3) for s in [1]) / var("dx")[0]**2
    torch.manual_seed(23)
    pad = (n + 2 * h,) * 3
    rho_t = (torch.rand(pad, dtype=torch.float64, device="cuda") - 0.5)
    decomp.share_halos(rho_t)

    def run(force_full):
        solver = RedBlackIterator(decomp, {f: (lap, rho)}, halo_shape=h,
                                  fixed_parameters=dict(omega=1.0))
        assert solver._star_operator
        if force_full:
            solver._star_cached = False
        ff = torch.zeros(pad, dtype=torch.float64, device="cuda")
        solver(decomp, iterations=7, f=ff, rho=rho_t, dx=np.array(dx))
        torch.cuda.synchronize()
        return ff.clone()

    fast = run(force_full=False)
    full = run(force_full=True)
    # interiors + faces identical bit for bit; corners of the halo may
    # legitimately differ mid-loop but the FINAL share refreshes all
    assert torch.equal(fast, full)


@requires_gpu
def test_rbgs_graph_bit_equality(n=24, h=1):
    """The hipGraph-captured RBGS smoothing loop replays bit-identically
    to the eager loop (PYSTELLA_MG_GRAPH=0), across repeated calls
    (graph cache reuse) and after state changes between calls."""
    import os
    from pystella_amd.field import Field, shift_fields
    from pystella_amd.multigrid import RedBlackIterator

    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=(n, n, n))
    dx = (2 * np.pi / n,) * 3
    f = Field("f", offset="h")
    rho = Field("rho", offset="h")
    lap = sum(
        (shift_fields(f, tuple(s * int(m == d) for m in range(3)))
         - 2 * f
         + shift_fields(f, tuple(-s * int(m == d) for m in range(3))))
        for d in range(3) for s in [1]) / var("dx")[0]**2
    torch.manual_seed(29)
    pad = (n + 2 * h,) * 3
    rho_t = (torch.rand(pad, dtype=torch.float64, device="cuda") - 0.5)
    decomp.share_halos(rho_t)

    def run(graph):
        old = os.environ.get("PYSTELLA_MG_GRAPH")
        os.environ["PYSTELLA_MG_GRAPH"] = "1" if graph else "0"
        try:
            solver = RedBlackIterator(
                decomp, {f: (lap, rho)}, halo_shape=h,
                fixed_parameters=dict(omega=1.0))
            ff = torch.zeros(pad, dtype=torch.float64, device="cuda")
            out = []
            for rep in range(3):   # repeated calls: capture then replays
                solver(decomp, iterations=5, f=ff, rho=rho_t,
                       dx=np.array(dx))
                torch.cuda.synchronize()
                out.append(ff.clone())
            if graph:
                assert any(g is not False
                           for g in getattr(solver, "_graphs",
                                            {}).values()), \
                    "graph path did not engage"
            return out
        finally:
            if old is None:
                os.environ.pop("PYSTELLA_MG_GRAPH", None)
            else:
                os.environ["PYSTELLA_MG_GRAPH"] = old

    eager = run(graph=False)
    graphed = run(graph=True)
    for a, b in zip(eager, graphed):
        assert torch.equal(a, b)
