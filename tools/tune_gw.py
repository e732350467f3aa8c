"""Focused sweep: GW tensor-group ring kernel (tile x min-waves)."""
import sys
import time

import numpy as np
import torch

sys.path.insert(0, ".")
import pystella_amd as ps  # noqa: E402
from pystella_amd.backend import hip as H  # noqa: E402
from pystella_amd.fusion import StencilRKStepper  # noqa: E402
from pystella_amd.sectors import get_rho_and_p  # noqa: E402


def timeit(fn, n=5, warmup=2):
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
    dt = 1e-3
    sites = float(np.prod(grid))
    decomp = ps.DomainDecomposition((1, 1, 1), h, grid_shape=grid)
    pad = tuple(g + 2 * h for g in grid)

    sector = ps.ScalarSector(2, potential=lambda f: f[0]**2 / 2)
    tensor = ps.TensorPerturbationSector([sector])
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid)
    fst = StencilRKStepper(ps.LowStorageRK54, [sector, tensor], derivs,
                           halo_shape=h, rank_shape=grid, dt=dt,
                           reducers=sector, grid_size=sites,
                           callback=get_rho_and_p, inline_grad=True)
    sm = fst._stepper.steps[1]
    gi = [i for i, g in enumerate(sm.ring) if g[3] == "hij"][0]
    rk_t, tmp_t, red_t, fname, nf = sm.ring[gi]

    f = torch.rand((2,) + pad, dtype=torch.float64, device=device)
    hij = torch.zeros((6,) + pad, dtype=torch.float64, device=device)
    env = dict(a=np.ones(1), hubble=np.zeros(1), f=f, hij=hij,
               hij_next=torch.zeros_like(hij),
               dhijdt=torch.zeros_like(hij),
               hij_tmp=torch.zeros((6,) + grid, dtype=torch.float64,
                                   device=device),
               dhijdt_tmp=torch.zeros((6,) + grid, dtype=torch.float64,
                                      device=device),
               dt=dt)
    gb = sites * 8 * 50 / 1e9
    results = []
    for minw in (1, 2, 3):
        for tile in [(64, 8, 64), (64, 8, 32), (128, 4, 32),
                     (64, 4, 64)]:
            k = H.JitLapStage(
                rk_t, tmp_t, red_t or [(0.0, "sum")],
                sm._ring_field_args[gi], [], (h,) * 3, grid, dx, nf,
                f_name=fname, lap_name=f"lap_{fname}",
                name=f"tgw{minw}_{tile[0]}_{tile[1]}_{tile[2]}",
                tile=tile, min_waves=minw)
            ms = timeit(lambda: k(env))
            results.append((ms, minw, tile))
            print(f"  minw={minw} tile={tile}:  {ms:7.3f} ms  "
                  f"{gb/ms:5.2f} TB/s", flush=True)
    results.sort()
    print("BEST:", results[0])


if __name__ == "__main__" and "--ballast" not in sys.argv:
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 512)


def stage0_ballast(n=512):
    """Sweep dynamic-LDS occupancy ballast on the GW hij kernels."""
    device = torch.device("cuda", 0)
    torch.cuda.set_device(device)
    grid = (n, n, n)
    h = 2
    dx = tuple(5 / g for g in grid)
    dt = 1e-3
    sites = float(np.prod(grid))
    decomp = ps.DomainDecomposition((1, 1, 1), h, grid_shape=grid)
    pad = tuple(g + 2 * h for g in grid)
    sector = ps.ScalarSector(2, potential=lambda f: f[0]**2 / 2)
    tensor = ps.TensorPerturbationSector([sector])
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid)
    fst = StencilRKStepper(ps.LowStorageRK54, [sector, tensor], derivs,
                           halo_shape=h, rank_shape=grid, dt=dt,
                           reducers=sector, grid_size=sites,
                           callback=get_rho_and_p, inline_grad=True)
    f = torch.rand((2,) + pad, dtype=torch.float64, device=device)
    hij = torch.zeros((6,) + pad, dtype=torch.float64, device=device)
    env = dict(a=np.ones(1), hubble=np.zeros(1), f=f, hij=hij,
               hij_next=torch.zeros_like(hij),
               dhijdt=torch.zeros_like(hij),
               hij_tmp=torch.zeros((6,) + grid, dtype=torch.float64,
                                   device=device),
               dhijdt_tmp=torch.zeros((6,) + grid, dtype=torch.float64,
                                      device=device),
               dt=dt)
    for stage in (0, 1):
        sm = fst._stepper.steps[stage]
        gi = [i for i, g in enumerate(sm.ring) if g[3] == "hij"][0]
        rk_t, tmp_t, red_t, fname, nf = sm.ring[gi]
        for shmem in (0, 24 * 1024, 40 * 1024, 56 * 1024):
            k = H.JitLapStage(
                rk_t, tmp_t, red_t or [(0.0, "sum")],
                sm._ring_field_args[gi], [], (h,) * 3, grid, dx, nf,
                f_name=fname, lap_name=f"lap_{fname}",
                name=f"bal{stage}_{shmem}")
            k.shmem = shmem
            ms = timeit(lambda: k(env))
            print(f"  stage{stage} shmem={shmem:6d}:  {ms:7.3f} ms",
                  flush=True)


if __name__ == "__main__" and "--ballast" in sys.argv:
    stage0_ballast()
