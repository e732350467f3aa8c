"""Single-GPU proxy for the N=8 rank-local COMPUTE geometry: force the
interior + 6-boundary-slab region split (as a (2,2,2) rank runs it)
on one GPU at 256^3 and time the device loop.  A/B's the tile-matched
slab kernel variants (PYSTELLA_SLAB_TILES=0/1).
"""

import argparse
import sys
import time

import numpy as np
import torch

sys.path.insert(0, ".")
import pystella_amd as ps  # noqa: E402
from pystella_amd.fusion import (  # noqa: E402
    DeviceFriedmannLoop, FusedLaplacianReduction, StencilRKStepper)
from pystella_amd.sectors import get_rho_and_p  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--grid", type=int, default=256)
    ap.add_argument("--steps", type=int, default=40)
    ap.add_argument("--warmup", type=int, default=10)
    ap.add_argument("--halo", type=int, default=2)
    p = ap.parse_args()

    grid_shape = (p.grid,) * 3
    h = p.halo
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    dx = tuple(5 / n for n in grid_shape)
    dt = min(0.1 * min(dx), 1e-3)
    gsize = float(np.prod(grid_shape))

    def potential(f):
        return (1.2e-6**2 / 2 * f[0]**2
                + 2.5e-7 / 2 * f[0]**2 * f[1]**2) / 1.2e-6**2

    sector = ps.ScalarSector(2, potential=potential)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)
    st = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                          halo_shape=h, rank_shape=grid_shape, dt=dt,
                          reducers=sector, grid_size=gsize,
                          callback=get_rho_and_p)
    red = FusedLaplacianReduction(
        decomp, sector, derivs, halo_shape=h, callback=get_rho_and_p,
        rank_shape=grid_shape, grid_size=gsize, store_lap=False)

    pad = tuple(n + 2 * h for n in grid_shape)
    gen = torch.Generator(device="cpu").manual_seed(7)
    arrays = {
        "f": (0.193 + 1e-3 * torch.rand((2,) + pad, dtype=torch.float64,
                                        generator=gen)).cuda(),
        "dfdt": (-0.142 + 1e-3 * torch.rand((2,) + pad,
                                            dtype=torch.float64,
                                            generator=gen)).cuda(),
    }
    arrays["f_next"] = torch.zeros_like(arrays["f"])
    e0 = red(f=arrays["f"], dfdt=arrays["dfdt"], a=np.ones(1))
    ex = ps.Expansion(e0["total"], ps.LowStorageRK54)
    decomp.share_halos(arrays["f"])
    dl = DeviceFriedmannLoop(st, decomp, ex, gsize, dt)

    # force the full 6-slab split of a (2,2,2)-decomposed rank
    nx, ny, nz = grid_shape

    def fake_regions(rank_shape, split_axes=None):
        interior = (h, nx - h, h, ny - h, h, nz - h)
        slabs = [
            (0, h, 0, ny, 0, nz), (nx - h, nx, 0, ny, 0, nz),
            (h, nx - h, 0, h, 0, nz), (h, nx - h, ny - h, ny, 0, nz),
            (h, nx - h, h, ny - h, 0, h),
            (h, nx - h, h, ny - h, nz - h, nz),
        ]
        return interior, slabs

    dl._regions = fake_regions

    for _ in range(p.warmup):
        dl.step(arrays)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(p.steps):
        dl.step(arrays)
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    msites = gsize * p.steps / el / 1e6
    assert np.isfinite(dl.read_state()["energy"])
    import json
    print(json.dumps({
        "metric": "Msites/s, forced 6-slab split (N=8 rank proxy)",
        "value": msites, "ms_per_step": el / p.steps * 1e3,
        "grid": p.grid}))


if __name__ == "__main__":
    main()
