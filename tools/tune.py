"""Tile-shape sweep for the hot kernels on a live MI355X.

Times the RK stage kernel, the fused lap+energy kernel and the AOT
gradlap kernel across (TBZ, TBY, XCHUNK) tiles and prints a ranked
table.  Winning defaults get baked into backend/hip.py / the
PYSTELLA_XCHUNK env default.
"""

import os
import sys
import time

import numpy as np
import torch

sys.path.insert(0, ".")
import pystella_amd as ps  # noqa: E402
from pystella_amd.backend import hip as H  # noqa: E402
from pystella_amd.sectors import get_rho_and_p  # noqa: E402
from pystella_amd.reduction import Reduction  # noqa: E402


def timeit(fn, n=10, warmup=2):
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
    GBms = sites * nscalars * 8 / 1e9  # GB per (field pass) per ms scale

    decomp = ps.DomainDecomposition((1, 1, 1), h, grid_shape=grid)
    pad = tuple(g + 2 * h for g in grid)

    def potential(f):
        return (1.2e-6**2 / 2 * f[0]**2
                + 2.5e-7 / 2 * f[0]**2 * f[1]**2) / 1.2e-6**2

    sector = ps.ScalarSector(nscalars, potential=potential)
    gen = torch.Generator(device="cpu").manual_seed(1)
    f = (0.19 + 1e-3 * torch.rand((nscalars,) + pad, dtype=torch.float64,
                                  generator=gen)).to(device)
    dfdt = torch.zeros_like(f)
    lap = torch.zeros((nscalars,) + grid, dtype=torch.float64,
                      device=device)
    a = np.ones(1)
    hub = np.zeros(1)

    # raw streaming roofline: copy (1R+1W) via torch
    src = torch.rand(512**3 * 2, dtype=torch.float64, device=device)
    dst = torch.empty_like(src)
    ms = timeit(lambda: dst.copy_(src), n=20)
    gb = src.numel() * 8 * 2 / 1e9
    print(f"== torch copy roofline: {ms:.3f} ms  {gb/ms:.2f} TB/s "
          f"(read+write combined)")

    tiles = [(64, 4, 512), (64, 4, 128), (64, 4, 64), (64, 4, 32),
             (64, 8, 64), (64, 8, 32), (128, 2, 64), (64, 2, 64),
             (256, 1, 64), (64, 16, 32)]

    # --- RK stage kernel sweep
    stepper = ps.LowStorageRK54([sector], halo_shape=h, rank_shape=grid,
                                dt=dt)
    step1 = stepper.steps[1]      # stage 1 reads+writes everything
    env = dict(a=a, hubble=hub, f=f, dfdt=dfdt, lap_f=lap,
               f_tmp=torch.zeros((nscalars,) + grid, dtype=torch.float64,
                                 device=device),
               dfdt_tmp=torch.zeros((nscalars,) + grid,
                                    dtype=torch.float64, device=device),
               dt=dt)
    print(f"== rk stage kernel (10 passes ~ {10*GBms:.1f} GB)")
    results = []
    for tile in tiles:
        k = H.JitElementwise(step1.map_dict, step1.tmp_instructions,
                             step1.field_args, [], (h,) * 3, grid,
                             name=f"tune_st_{tile[0]}_{tile[1]}_{tile[2]}",
                             tile=tile)
        ms = timeit(lambda: k(env))
        bw = 10 * GBms / ms
        results.append((ms, tile))
        print(f"  tile={tile}:  {ms:7.3f} ms   {bw:5.2f} TB/s")
    results.sort()
    print("  BEST:", results[0])

    # --- fused lap+energy sweep
    red = Reduction(decomp, sector, halo_shape=h, grid_size=sites,
                    callback=get_rho_and_p, rank_shape=grid)
    entries = [(e, o) for _, _, e, o in red.flat]
    print(f"== fused lap+energy (6 passes ~ {6*GBms:.1f} GB)")
    results = []
    env2 = dict(f=f, dfdt=dfdt, lap_f=lap, a=a)
    for tile in tiles:
        k = H.JitLapReduction(
            entries, red.field_args, [], (h,) * 3, grid, dx, nscalars,
            name=f"tune_lr_{tile[0]}_{tile[1]}_{tile[2]}", tile=tile)
        ms = timeit(lambda: k(env2))
        bw = 6 * GBms / ms
        results.append((ms, tile))
        print(f"  tile={tile}:  {ms:7.3f} ms   {bw:5.2f} TB/s")
    results.sort()
    print("  BEST:", results[0])

    # --- energy-fused RK stage kernel sweep (the current hot-loop
    # kernel: update + input-state energy reduction in one pass)
    from pystella_amd.fusion import StencilRKStepper
    derivs0 = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid)
    fst = StencilRKStepper(ps.LowStorageRK54, [sector], derivs0,
                           halo_shape=h, rank_shape=grid, dt=dt,
                           reducers=sector, grid_size=sites,
                           callback=get_rho_and_p)
    sm = fst._stepper.steps[1]._map
    env3 = dict(a=a, hubble=hub, f=f, f_next=torch.zeros_like(f),
                dfdt=dfdt,
                f_tmp=torch.zeros((nscalars,) + grid,
                                  dtype=torch.float64, device=device),
                dfdt_tmp=torch.zeros((nscalars,) + grid,
                                     dtype=torch.float64, device=device),
                dt=dt)
    red_entries = fst._stepper.steps[1].red_entries
    print(f"== energy-fused rk stage kernel (16 passes ~ "
          f"{16*GBms:.1f} GB)")
    results = []
    for tile in tiles:
        k = H.JitStageReduction(
            sm.map_dict, sm.tmp_instructions, red_entries,
            sm.field_args, [], (h,) * 3, grid,
            name=f"tune_sr_{tile[0]}_{tile[1]}_{tile[2]}", tile=tile)
        ms = timeit(lambda: k(env3))
        bw = 16 * GBms / ms / nscalars
        results.append((ms, tile))
        print(f"  tile={tile}:  {ms:7.3f} ms   {bw:5.2f} TB/s")
    results.sort()
    print("  BEST:", results[0])

    # --- register-ring energy-fused stage kernel sweep (current
    # default hot-loop kernel)
    sm1 = fst._stepper.steps[1]
    rk_o, tmp_o, red_o, f_name, nf = sm1.ring[0]
    print(f"== ring energy-fused rk stage kernel (16 passes ~ "
          f"{16*GBms/nscalars:.1f} GB)")
    results = []
    for nt in (True, False):
        for tile in tiles + [(128, 2, 32), (128, 4, 32), (32, 8, 32),
                             (128, 1, 32), (256, 1, 32), (256, 1, 16),
                             (512, 1, 32), (256, 2, 32)]:
            k = H.JitLapStage(
                rk_o, tmp_o, red_o, sm1._ring_field_args[0], [], (h,) * 3,
                grid, dx, nf, f_name=f_name, nt=nt,
                name=f"tune_ls{int(nt)}_{tile[0]}_{tile[1]}_{tile[2]}",
                tile=tile)
            ms = timeit(lambda: k(env3))
            bw = 16 * GBms / ms / nscalars
            results.append((ms, nt, tile))
            print(f"  nt={int(nt)} tile={tile}:  {ms:7.3f} ms   "
                  f"{bw:5.2f} TB/s")
    results.sort()
    print("  BEST:", results[0])

    # --- GW tensor-group ring kernel sweep
    tensor = ps.TensorPerturbationSector([sector])
    fstg = StencilRKStepper(ps.LowStorageRK54, [sector, tensor],
                            derivs0, halo_shape=h, rank_shape=grid,
                            dt=dt, reducers=sector, grid_size=sites,
                            callback=get_rho_and_p, inline_grad=True)
    smg = fstg._stepper.steps[1]
    gi = [i for i, g in enumerate(smg.ring) if g[3] == "hij"][0]
    rk_t, tmp_t, red_t, fname_t, nf_t = smg.ring[gi]
    hij = torch.zeros((6,) + pad, dtype=torch.float64, device=device)
    env4 = dict(a=a, hubble=hub, f=f, hij=hij,
                hij_next=torch.zeros_like(hij),
                dhijdt=torch.zeros_like(hij),
                hij_tmp=torch.zeros((6,) + grid, dtype=torch.float64,
                                    device=device),
                dhijdt_tmp=torch.zeros((6,) + grid, dtype=torch.float64,
                                       device=device),
                dt=dt)
    print("== GW tensor-group ring kernel (~50 comps "
          f"~ {50*GBms/nscalars:.1f} GB)")
    results = []
    for minw in (1, 2, 3, 4):
        for tile in [(64, 8, 64), (64, 8, 32), (128, 4, 32),
                     (64, 4, 32)]:
            k = H.JitLapStage(
                rk_t, tmp_t, red_t or [(0.0, "sum")],
                smg._ring_field_args[gi], [], (h,) * 3, grid, dx, nf_t,
                f_name=fname_t, lap_name=f"lap_{fname_t}",
                name=f"tune_gw{minw}_{tile[0]}_{tile[1]}_{tile[2]}",
                tile=tile, min_waves=minw)
            ms = timeit(lambda: k(env4), n=5)
            results.append((ms, minw, tile))
            print(f"  minw={minw} tile={tile}:  {ms:7.3f} ms   "
                  f"{50*GBms/nscalars/ms:5.2f} TB/s")
    results.sort()
    print("  BEST:", results[0])

    # --- AOT gradlap XCHUNK sweep (env var)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid)
    print(f"== AOT gradlap lap-only (4 passes ~ {4*GBms:.1f} GB)")
    for xc in (512, 128, 64, 32):
        os.environ["PYSTELLA_XCHUNK"] = str(xc)
        ms = timeit(lambda: H.derivs(f, lap=lap, halo=(h,) * 3, dx=dx,
                                     h=h))
        print(f"  XCHUNK={xc}:  {ms:7.3f} ms   {4*GBms/ms:5.2f} TB/s")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 512)
