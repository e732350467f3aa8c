"""Per-component GPU timings for the scalar-preheating hot loop.

Times each piece of the bench step separately (stage kernel, fused
lap+energy, halo wrap, reduction finish) with HIP events via torch, and
prints achieved HBM bandwidth per kernel.
"""

import sys
import time

import numpy as np
import torch

sys.path.insert(0, ".")
import pystella_amd as ps  # noqa: E402
from pystella_amd.sectors import get_rho_and_p  # noqa: E402
from pystella_amd.fusion import FusedLaplacianReduction  # noqa: E402


def timeit(fn, n=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main(n=512):
    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)
    grid = (n, n, n)
    h = 2
    dx = tuple(5 / g for g in grid)
    dt = 0.1 * min(dx)
    nscalars = 2
    sites = float(np.prod(grid))

    decomp = ps.DomainDecomposition((1, 1, 1), h, grid_shape=grid)
    pad = tuple(g + 2 * h for g in grid)

    def potential(f):
        return (1.2e-6**2 / 2 * f[0]**2
                + 2.5e-7 / 2 * f[0]**2 * f[1]**2) / 1.2e-6**2

    sector = ps.ScalarSector(nscalars, potential=potential)
    stepper = ps.LowStorageRK54([sector], halo_shape=h, rank_shape=grid,
                                dt=dt)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid)
    fused = FusedLaplacianReduction(
        decomp, sector, derivs, halo_shape=h, callback=get_rho_and_p,
        rank_shape=grid, grid_size=sites)
    unfused_red = ps.Reduction(decomp, sector, halo_shape=h,
                               callback=get_rho_and_p, rank_shape=grid,
                               grid_size=sites)

    gen = torch.Generator(device="cpu").manual_seed(1)
    f = (0.19 + 1e-3 * torch.rand((nscalars,) + pad, dtype=torch.float64,
                                  generator=gen)).to(device)
    dfdt = torch.zeros_like(f)
    lap = torch.zeros((nscalars,) + grid, dtype=torch.float64,
                      device=device)
    a = np.ones(1)
    hub = np.zeros(1)

    GB = 1 << 30

    # stage kernel alone
    def stage():
        stepper(0, a=a, hubble=hub, f=f, dfdt=dfdt, lap_f=lap)
    ms = timeit(stage)
    byts = sites * nscalars * 8 * (3 + 2)  # r: f,k,lap; w: f,k
    print(f"stage kernel:      {ms:8.3f} ms   {byts/ms/1e6/GB*1e3:6.2f} "
          f"TB/s effective")

    # halo wrap alone
    def wrap():
        decomp.share_halos(f)
    ms = timeit(wrap)
    print(f"halo wrap:         {ms:8.3f} ms")

    # AOT gradlap (lap only)
    def gl():
        derivs(fx=f, lap=lap)
    ms = timeit(gl)
    byts = sites * nscalars * 8 * 2
    print(f"sharehalos+gradlap:{ms:8.3f} ms   {byts/ms/1e6/GB*1e3:6.2f} "
          f"TB/s algorithmic")

    # unfused reduction
    def red():
        unfused_red(f=f, dfdt=dfdt, lap_f=lap, a=a)
    ms = timeit(red)
    byts = sites * nscalars * 8 * 3
    print(f"energy reduction:  {ms:8.3f} ms   {byts/ms/1e6/GB*1e3:6.2f} "
          f"TB/s algorithmic")

    # fused lap+energy
    def fus():
        fused(f=f, dfdt=dfdt, lap_f=lap, a=a)
    ms = timeit(fus)
    byts = sites * nscalars * 8 * 3  # r: f, dfdt; w: lap
    print(f"fused lap+energy:  {ms:8.3f} ms   {byts/ms/1e6/GB*1e3:6.2f} "
          f"TB/s algorithmic")

    # pure HBM reference: torch copy of one field array
    src = torch.rand((2,) + grid, dtype=torch.float64, device=device)
    dst = torch.empty_like(src)

    def cp():
        dst.copy_(src)
    ms = timeit(cp)
    byts = src.numel() * 8 * 2
    print(f"torch copy_ ref:   {ms:8.3f} ms   {byts/ms/1e6/GB*1e3:6.2f} "
          f"TB/s")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 512)
