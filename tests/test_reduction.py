"""Reduction / FieldStatistics tests vs direct torch reference."""

import numpy as np
import torch

import pystella_amd as ps
from pystella_amd.field import Field, var


def test_reduction_ops(grid_shape=(16, 16, 16)):
    h = 1
    pad = tuple(n + 2 * h for n in grid_shape)
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    torch.manual_seed(0)
    f = torch.rand(pad, dtype=torch.float64)
    interior = f[h:-h, h:-h, h:-h]

    F = Field("f", offset="h")
    red = ps.Reduction(decomp, {
        "mean": [F],
        "sum_sq": [(F**2, "sum")],
        "mx": [(F, "max")],
        "mn": [(F, "min")],
    }, halo_shape=h, grid_size=float(np.prod(grid_shape)))

    out = red(f=f)
    assert abs(out["mean"][0] - interior.mean().item()) < 1e-12
    assert abs(out["sum_sq"][0] - (interior**2).sum().item()) < 1e-9
    assert out["mx"][0] == interior.max().item()
    assert out["mn"][0] == interior.min().item()


def test_reduction_with_scalar_arg(grid_shape=(8, 8, 8)):
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    f = torch.rand(grid_shape, dtype=torch.float64)
    F = Field("f", offset=0)
    a = var("a")
    red = ps.Reduction(decomp, {"kin": [F**2 / 2 / a**2]}, halo_shape=0,
                       grid_size=float(np.prod(grid_shape)))
    out = red(f=f, a=np.array([2.0]))
    expect = (f**2).mean().item() / 8
    assert abs(out["kin"][0] - expect) < 1e-12


def test_field_statistics(grid_shape=(16, 16, 16)):
    h = 2
    pad = tuple(n + 2 * h for n in grid_shape)
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    stats = ps.FieldStatistics(decomp, h, rank_shape=grid_shape,
                               grid_size=float(np.prod(grid_shape)))
    torch.manual_seed(1)
    f = torch.rand((2,) + pad, dtype=torch.float64)
    out = stats(f)
    for i in range(2):
        interior = f[i, h:-h, h:-h, h:-h]
        assert abs(out["mean"][i] - interior.mean().item()) < 1e-12
        assert abs(out["variance"][i] - interior.var(correction=0).item()) \
            < 1e-10


def test_scalar_sector_energy(grid_shape=(12, 12, 12)):
    """Energy reducers vs a hand-written torch computation
    (analogue of reference test/test_energy.py)."""
    from pystella_amd.sectors import get_rho_and_p
    h = 1
    pad = tuple(n + 2 * h for n in grid_shape)
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    torch.manual_seed(2)
    f = torch.rand((2,) + pad, dtype=torch.float64)
    dfdt = torch.rand((2,) + pad, dtype=torch.float64)
    lap_f = torch.rand((2,) + grid_shape, dtype=torch.float64)
    a = 1.3

    def potential(fld):
        return fld[0]**2 / 2 + fld[0]**2 * fld[1]**2 / 4

    sector = ps.ScalarSector(2, potential=potential)
    red = ps.Reduction(decomp, sector, halo_shape=h,
                       grid_size=float(np.prod(grid_shape)),
                       callback=get_rho_and_p)
    out = red(f=f, dfdt=dfdt, lap_f=lap_f, a=np.array([a]))

    fi = f[:, h:-h, h:-h, h:-h]
    dfi = dfdt[:, h:-h, h:-h, h:-h]
    kin = [(dfi[i]**2 / 2 / a**2).mean().item() for i in range(2)]
    pot = (fi[0]**2 / 2 + fi[0]**2 * fi[1]**2 / 4).mean().item()
    grad = [(-fi[i] * lap_f[i] / 2 / a**2).mean().item() for i in range(2)]
    for i in range(2):
        assert abs(out["kinetic"][i] - kin[i]) < 1e-12
        assert abs(out["gradient"][i] - grad[i]) < 1e-12
    assert abs(out["potential"][0] - pot) < 1e-12
    total = sum(kin) + sum(grad) + pot
    assert abs(out["total"] - total) < 1e-11


def test_fused_lap_reduction_cpu(grid_shape=(16, 16, 16)):
    """CPU path of the fused lap+energy component equals derivs +
    Reduction composition."""
    from pystella_amd.fusion import FusedLaplacianReduction
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    pad = tuple(n + 2 * h for n in grid_shape)
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.2, 0.2, 0.2)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)

    def potential(f):
        return f[0]**2 / 2

    sector = ps.ScalarSector(2, potential=potential)
    torch.manual_seed(9)
    f = torch.rand((2,) + pad, dtype=torch.float64)
    dfdt = torch.rand((2,) + pad, dtype=torch.float64)
    lap1 = torch.zeros((2,) + grid_shape, dtype=torch.float64)
    lap2 = torch.zeros_like(lap1)
    gs = float(np.prod(grid_shape))

    derivs(fx=f, lap=lap1)
    unfused = ps.Reduction(decomp, sector, halo_shape=h, grid_size=gs,
                           callback=get_rho_and_p)
    out_u = unfused(f=f, dfdt=dfdt, lap_f=lap1, a=np.array([1.1]))

    fused = FusedLaplacianReduction(decomp, sector, derivs, halo_shape=h,
                                    grid_size=gs, callback=get_rho_and_p)
    out_f = fused(f=f, dfdt=dfdt, lap_f=lap2, a=np.array([1.1]))

    assert (lap1 - lap2).abs().max().item() < 1e-14
    assert abs(out_u["total"] - out_f["total"]) < 1e-13


def test_stencil_stepper_cpu_equivalence(grid_shape=(12, 12, 12)):
    """Inline-Laplacian (ping-pong) stepper reproduces the
    reference-structure loop bit-for-bit on interiors (CPU)."""
    from pystella_amd.fusion import StencilRKStepper
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.3, 0.3)
    dt = 0.01

    def pot(f):
        return f[0]**2 / 2 + f[0]**2 * f[1]**2 / 4

    sector = ps.ScalarSector(2, potential=pot)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(0)
    f0 = torch.rand((2,) + pad, dtype=torch.float64)
    d0 = torch.rand((2,) + pad, dtype=torch.float64)
    a = np.ones(1)
    hub = 0.1 * np.ones(1)
    cut = (slice(None),) + (slice(h, -h),) * 3

    fu, du = f0.clone(), d0.clone()
    lap = torch.zeros((2,) + grid_shape, dtype=torch.float64)
    st = ps.LowStorageRK54([sector], dt=dt, halo_shape=h,
                           rank_shape=grid_shape)
    for s in range(st.num_stages):
        derivs(fx=fu, lap=lap)
        st(s, a=a, hubble=hub, f=fu, dfdt=du, lap_f=lap)

    ff, df = f0.clone(), d0.clone()
    fnx = torch.zeros_like(ff)
    fst = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                           halo_shape=h, rank_shape=grid_shape, dt=dt)
    arrays = {"f": ff, "dfdt": df, "f_next": fnx}
    decomp.share_halos(arrays["f"])
    for s in range(fst.num_stages):
        fst(s, a=a, hubble=hub, **arrays)
        arrays["f"], arrays["f_next"] = arrays["f_next"], arrays["f"]
        decomp.share_halos(arrays["f"])

    assert (arrays["f"][cut] - fu[cut]).abs().max().item() < 1e-15
    assert (arrays["dfdt"][cut] - du[cut]).abs().max().item() < 1e-15


def test_stage_fused_energy_cpu(grid_shape=(12, 12, 12)):
    """The energy-fused stage kernel returns exactly the reduction
    values of its input state and applies the same update."""
    from pystella_amd.fusion import StencilRKStepper
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.3, 0.3)
    dt = 0.01
    gsize = float(np.prod(grid_shape))

    def pot(f):
        return f[0]**2 / 2 + f[0]**2 * f[1]**2 / 4

    sector = ps.ScalarSector(2, potential=pot)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(3)
    f0 = torch.rand((2,) + pad, dtype=torch.float64)
    d0 = torch.rand((2,) + pad, dtype=torch.float64)
    a = np.ones(1)
    hub = 0.1 * np.ones(1)
    cut = (slice(None),) + (slice(h, -h),) * 3

    # reference: unfused loop + standalone reduction of the input state
    red = ps.Reduction(decomp, sector, halo_shape=h,
                       callback=get_rho_and_p, rank_shape=grid_shape,
                       grid_size=gsize)
    fu, du = f0.clone(), d0.clone()
    lap = torch.zeros((2,) + grid_shape, dtype=torch.float64)
    st = ps.LowStorageRK54([sector], dt=dt, halo_shape=h,
                           rank_shape=grid_shape)
    ref_energies = []
    for s in range(st.num_stages):
        derivs(fx=fu, lap=lap)
        ref_energies.append(red(f=fu, dfdt=du, lap_f=lap, a=a))
        st(s, a=a, hubble=hub, f=fu, dfdt=du, lap_f=lap)

    # fused: stage kernel returns the input-state energy
    ff, df = f0.clone(), d0.clone()
    fst = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                           halo_shape=h, rank_shape=grid_shape, dt=dt,
                           reducers=sector, grid_size=gsize,
                           callback=get_rho_and_p)
    arrays = {"f": ff, "dfdt": df, "f_next": torch.zeros_like(ff)}
    decomp.share_halos(arrays["f"])
    for s in range(fst.num_stages):
        e_in = fst(s, a=a, hubble=hub, **arrays)
        arrays["f"], arrays["f_next"] = arrays["f_next"], arrays["f"]
        decomp.share_halos(arrays["f"])
        for key in ("kinetic", "potential", "gradient"):
            assert np.allclose(e_in[key], ref_energies[s][key],
                               rtol=1e-13), (s, key)
        assert np.allclose(e_in["total"], ref_energies[s]["total"],
                           rtol=1e-13)

    assert (arrays["f"][cut] - fu[cut]).abs().max().item() < 1e-14
    assert (arrays["dfdt"][cut] - du[cut]).abs().max().item() < 1e-14


def test_stage_fused_gw_cpu(grid_shape=(8, 8, 8)):
    """Energy-fused multi-sector (scalar + GW tensor) stepper matches
    the unfused reference loop on CPU."""
    from pystella_amd.fusion import StencilRKStepper
    from pystella_amd.sectors import get_rho_and_p
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = (0.3, 0.3, 0.3)
    dt = 0.01
    gsize = float(np.prod(grid_shape))

    def pot(f):
        return f[0]**2 / 2 + f[0]**2 * f[1]**2 / 4

    sector = ps.ScalarSector(2, potential=pot)
    tensor = ps.TensorPerturbationSector([sector])
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    pad = tuple(n + 2 * h for n in grid_shape)
    torch.manual_seed(21)
    f0 = torch.rand((2,) + pad, dtype=torch.float64)
    d0 = torch.rand((2,) + pad, dtype=torch.float64)
    h0 = 0.01 * torch.rand((6,) + pad, dtype=torch.float64)
    hd0 = 0.01 * torch.rand((6,) + pad, dtype=torch.float64)
    a = np.ones(1)
    hub = 0.1 * np.ones(1)
    cut = (slice(None),) + (slice(h, -h),) * 3

    # reference structure
    fu, du = f0.clone(), d0.clone()
    hu, hdu = h0.clone(), hd0.clone()
    lap = torch.zeros((2,) + grid_shape, dtype=torch.float64)
    lap_h = torch.zeros((6,) + grid_shape, dtype=torch.float64)
    grd = torch.zeros((2, 3) + grid_shape, dtype=torch.float64)
    st = ps.LowStorageRK54([sector, tensor], dt=dt, halo_shape=h,
                           rank_shape=grid_shape)
    for s in range(st.num_stages):
        derivs(fx=fu, lap=lap, grd=grd)
        derivs(fx=hu, lap=lap_h)
        st(s, a=a, hubble=hub, f=fu, dfdt=du, lap_f=lap, dfdx=grd,
           hij=hu, dhijdt=hdu, lap_hij=lap_h)

    # fused (2 ring groups)
    ff, df = f0.clone(), d0.clone()
    hf, hdf = h0.clone(), hd0.clone()
    fst = StencilRKStepper(ps.LowStorageRK54, [sector, tensor], derivs,
                           halo_shape=h, rank_shape=grid_shape, dt=dt,
                           reducers=sector, grid_size=gsize,
                           callback=get_rho_and_p)
    assert fst._stepper.steps[0].ring is not None
    assert len(fst._stepper.steps[0].ring) == 2
    arrays = {"f": ff, "dfdt": df, "f_next": torch.zeros_like(ff),
              "hij": hf, "dhijdt": hdf, "hij_next": torch.zeros_like(hf),
              "dfdx": torch.zeros((2, 3) + grid_shape,
                                  dtype=torch.float64)}
    decomp.share_halos(arrays["f"])
    decomp.share_halos(arrays["hij"])
    for s in range(fst.num_stages):
        derivs(fx=arrays["f"], grd=arrays["dfdx"])
        e_in = fst(s, a=a, hubble=hub, **arrays)
        assert np.isfinite(e_in["total"])
        for name in fst.pingpong:
            arrays[name], arrays[f"{name}_next"] = \
                arrays[f"{name}_next"], arrays[name]
            decomp.share_halos(arrays[name])

    assert (arrays["f"][cut] - fu[cut]).abs().max().item() < 1e-14
    assert (arrays["dfdt"][cut] - du[cut]).abs().max().item() < 1e-14
    assert (arrays["hij"][cut] - hu[cut]).abs().max().item() < 1e-14
    assert (arrays["dhijdt"][cut] - hdu[cut]).abs().max().item() < 1e-14


def test_device_loop_region_partition():
    """The interior/boundary-slab partition used by the overlapped
    device loop covers the rank box exactly once for every proc_shape
    (pure geometry; the GPU path launches one kernel per region)."""
    from pystella_amd.fusion import DeviceFriedmannLoop

    class _D:
        pass

    for proc_shape in [(1, 1, 1), (2, 1, 1), (1, 2, 1), (1, 1, 2),
                       (2, 2, 1), (2, 2, 2), (4, 2, 1)]:
        loop = DeviceFriedmannLoop.__new__(DeviceFriedmannLoop)
        loop.decomp = _D()
        loop.decomp.proc_shape = proc_shape

        class _S:
            pass

        loop.stepper = _S()
        loop.stepper._stepper = _S()
        loop.stepper._stepper.halo_shape = 2
        rank_shape = (16, 12, 20)
        interior, slabs = loop._regions(rank_shape)
        count = np.zeros(rank_shape, dtype=int)
        for (i0, i1, j0, j1, k0, k1) in [interior] + slabs:
            count[i0:i1, j0:j1, k0:k1] += 1
        assert (count == 1).all(), (proc_shape, count.min(), count.max())


def test_field_statistics_min_max(grid_shape=(12, 12, 12)):
    """FieldStatistics with max_min=True (reference reduction.py:284-302)."""
    h = 1
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    stats = ps.FieldStatistics(decomp, h, rank_shape=grid_shape,
                               max_min=True)
    torch.manual_seed(77)
    pad = tuple(n + 2 * h for n in grid_shape)
    f = torch.rand((2,) + pad, dtype=torch.float64) - 0.3
    out = stats(f)
    cut = (slice(None),) + (slice(h, -h),) * 3
    fi = f[cut]
    for i in range(2):
        assert np.isclose(out["mean"][i], fi[i].mean().item())
        assert np.isclose(out["variance"][i], fi[i].var(
            unbiased=False).item(), rtol=1e-10)
        assert np.isclose(out["max"][i], fi[i].max().item())
        assert np.isclose(out["min"][i], fi[i].min().item())
        assert np.isclose(out["abs_max"][i], fi[i].abs().max().item())
        assert np.isclose(out["abs_min"][i], fi[i].abs().min().item())
