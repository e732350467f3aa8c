"""fp32 flagship datapoint: the scalar-preheating
hot loop at 512^3 in fp32 through the dtype-generic component stack
(AOT fp32 stencil kernels + fp32 elementwise stage kernels + fp32
reduction).  This is a SIDE datapoint — the headline bench.py metric
stays fp64, matching the reference's default precision.
"""

import argparse
import sys
import time

import numpy as np
import torch

sys.path.insert(0, ".")
import pystella_amd as ps  # noqa: E402
from pystella_amd.sectors import get_rho_and_p  # noqa: E402


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--grid", type=int, default=512)
    ap.add_argument("--steps", type=int, default=10)
    ap.add_argument("--warmup", type=int, default=3)
    ap.add_argument("--dtype", default="float32")
    p = ap.parse_args()

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    dtype = getattr(torch, p.dtype)
    grid = (p.grid,) * 3
    h = 2
    dx = tuple(5 / n for n in grid)
    dt = min(0.1 * min(dx), 1e-3)
    grid_size = float(np.prod(grid))
    mphi, gsq = 1.2e-6, 2.5e-7

    decomp = ps.DomainDecomposition((1, 1, 1), h, grid_shape=grid)
    pad = tuple(n + 2 * h for n in grid)

    def potential(f):
        phi, chi = f[0], f[1]
        return (mphi**2 / 2 * phi**2 + gsq / 2 * phi**2 * chi**2) \
            / mphi**2

    sector = ps.ScalarSector(2, potential=potential)
    stepper = ps.LowStorageRK54([sector], halo_shape=h,
                                rank_shape=grid, dt=dt)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid)
    reduce_energy = ps.Reduction(decomp, sector, halo_shape=h,
                                 callback=get_rho_and_p,
                                 rank_shape=grid, grid_size=grid_size)

    gen = torch.Generator(device="cpu").manual_seed(7)
    f = (0.193 + 1e-3 * torch.rand((2,) + pad, generator=gen)) \
        .to(dtype).to(device)
    dfdt = (-0.142 + 1e-3 * torch.rand((2,) + pad, generator=gen)) \
        .to(dtype).to(device)
    lap_f = torch.zeros((2,) + grid, dtype=dtype, device=device)

    def compute_energy(a):
        decomp.share_halos(f)
        derivs(fx=f, lap=lap_f)
        return reduce_energy(f=f, dfdt=dfdt, lap_f=lap_f, a=np.array(a))

    energy = compute_energy(1.)
    expand = ps.Expansion(energy["total"], ps.LowStorageRK54)

    def step():
        nonlocal energy
        for s in range(stepper.num_stages):
            stepper(s, a=expand.a, hubble=expand.hubble,
                    f=f, dfdt=dfdt, lap_f=lap_f)
            expand.step(s, energy["total"], energy["pressure"], dt)
            energy = compute_energy(expand.a)

    for _ in range(p.warmup):
        step()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(p.steps):
        step()
    torch.cuda.synchronize()
    el = time.perf_counter() - t0
    msites = grid_size * p.steps / el / 1e6
    import json
    print(json.dumps({
        "metric": "Msite-updates/sec, scalar-preheating (fp32 side "
                  "datapoint, reference-structure loop)",
        "value": msites, "unit": "Msites/s",
        "ms_per_step": el / p.steps * 1e3,
        "dtype": p.dtype, "grid": list(grid),
    }))


if __name__ == "__main__":
    main()
