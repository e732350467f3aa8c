"""Flagship benchmark: scalar-preheating RK4 step throughput.

Driver contract (see repo instructions): measures the BASELINE.json
metric — Msite-updates/sec of the scalar_preheating hot loop (2 scalar
fields, fp64, halo 2, 512³ global grid, LowStorageRK54 = 4th-order RK)
— on N GPUs of one node, strong scaling (fixed 512³ total grid).

Each timed step is the full reference hot loop
(reference examples/scalar_preheating.py:258-271): per RK stage, the
fused stage kernel, the host Friedmann update, halo exchange +
fused grad/lap stencil, and the ⟨ρ⟩/⟨P⟩ energy reduction + allreduce.
Synthetic random-init fields (no dataset exists for this workload).
"""

import argparse
import json
import os
import time

import numpy as np
import torch

import pystella_amd as ps
from pystella_amd.sectors import get_rho_and_p


def proc_shape_for(n):
    shapes = {1: (1, 1, 1), 2: (2, 1, 1), 4: (2, 2, 1), 8: (2, 2, 2),
              3: (3, 1, 1), 6: (3, 2, 1)}
    if n in shapes:
        return shapes[n]
    return (n, 1, 1)


def main():
    parser = argparse.ArgumentParser()
    parser.add_argument("--gpus", type=int, default=1)
    parser.add_argument("--steps", type=int, default=20)
    parser.add_argument("--warmup", type=int, default=5)
    parser.add_argument("--grid", type=int, default=512)
    parser.add_argument("--halo", type=int, default=2)
    parser.add_argument("--device", default=None)
    parser.add_argument("--stepper", default="LowStorageRK54",
                        help="low-storage stepper class name (any "
                             "2N-storage tableau: LowStorageRK54/144/"
                             "134/124/3Williamson/...)")
    parser.add_argument("--no-fuse", action="store_true",
                        help="reference-structure loop (separate lap "
                             "array + unfused stage kernels)")
    parser.add_argument("--no-fuse-energy", action="store_true",
                        help="keep the energy reduction as a separate "
                             "fused lap+reduce kernel instead of folding "
                             "it into the RK stage kernel")
    parser.add_argument("--gws", action="store_true",
                        help="include the gravitational-wave tensor "
                             "sector (6 h_ij components sourced by the "
                             "scalar stress tensor)")
    parser.add_argument("--gws-split", action="store_true",
                        help="split the GW tensor sector into two "
                             "3-component stencil families (measured "
                             "slower: the stress-source gradients are "
                             "recomputed per family; kept as an A/B "
                             "flag)")
    parser.add_argument("--no-device-friedmann", action="store_true",
                        help="run the Friedmann (a, adot) update on the "
                             "host (one sync per RK stage) instead of "
                             "on-device")
    p = parser.parse_args()

    world_size = int(os.environ.get("WORLD_SIZE", "1"))
    if world_size > 1:
        ps.init_distributed()
    if p.gpus > 1 and world_size != p.gpus:
        raise SystemExit(
            f"--gpus {p.gpus} requires a torchrun launch with "
            f"WORLD_SIZE={p.gpus} (got WORLD_SIZE={world_size}); a "
            f"single-process run would do 1-GPU work but report "
            f"n_gpus={p.gpus}")
    n_gpus = max(p.gpus, world_size)

    if p.device is not None:
        device = torch.device(p.device)
    elif torch.cuda.is_available():
        device = ps.choose_device()
    else:
        device = torch.device("cpu")
    on_gpu = device.type == "cuda"

    grid_shape = (p.grid,) * 3
    grid_size = float(np.prod(grid_shape))
    h = p.halo
    box = (5., 5., 5.)
    dx = tuple(L / N for L, N in zip(box, grid_shape))
    # 0.1 dx is the CFL-ish step at the flagship 512^3 (dt=9.8e-4);
    # additionally cap by the Friedmann timescale (H ~ 30 in these
    # units) so small sanity grids stay stable
    dt = min(0.1 * min(dx), 1e-3)
    nscalars = 2
    mphi, mpl, gsq = 1.2e-6, 1.0, 2.5e-7

    proc_shape = proc_shape_for(world_size if world_size > 1 else 1)
    decomp = ps.DomainDecomposition(proc_shape, h, grid_shape=grid_shape)
    rank_shape = decomp.rank_shape
    pad = tuple(n + 2 * h for n in rank_shape)

    def potential(f):
        phi, chi = f[0], f[1]
        return (mphi**2 / 2 * phi**2 + gsq / 2 * phi**2 * chi**2) / mphi**2

    Stepper = getattr(ps, p.stepper)
    sector = ps.ScalarSector(nscalars, potential=potential)
    sectors = [sector]
    fuse_energy_pre = not (p.no_fuse or p.no_fuse_energy)
    if p.gws and fuse_energy_pre and p.gws_split:
        # split the 6 independent h_ij components into two 3-component
        # stencil families: halves each ring kernel's register
        # footprint (the components are views of one parent array)
        sectors.append(ps.TensorPerturbationSector(
            [sector], components=(0, 1, 2), name="hij_a"))
        sectors.append(ps.TensorPerturbationSector(
            [sector], components=(3, 4, 5), name="hij_b"))
    elif p.gws:
        sectors.append(ps.TensorPerturbationSector([sector]))
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=rank_shape)
    from pystella_amd.fusion import (
        FusedLaplacianReduction, StencilRKStepper)
    fuse_energy = not (p.no_fuse or p.no_fuse_energy)
    if p.no_fuse:
        stepper = Stepper(sectors, halo_shape=h,
                                    rank_shape=rank_shape, dt=dt)
    elif fuse_energy:
        # fully fused MI355X structure: one kernel per RK stage that
        # evaluates the Laplacian inline (ping-pong f), updates the
        # unknowns AND reduces the input-state energy — no separate
        # energy pass in the hot loop at all
        stepper = StencilRKStepper(Stepper, sectors, derivs,
                                   halo_shape=h, rank_shape=rank_shape,
                                   dt=dt, reducers=sector,
                                   grid_size=grid_size,
                                   callback=get_rho_and_p,
                                   inline_grad=p.gws)
    else:
        # stage kernels evaluate the Laplacian inline (ping-pong f);
        # the energy reduction is a separate fused lap+reduce kernel
        stepper = StencilRKStepper(Stepper, sectors, derivs,
                                   halo_shape=h, rank_shape=rank_shape,
                                   dt=dt)
    reduce_energy = FusedLaplacianReduction(
        decomp, sector, derivs, halo_shape=h, callback=get_rho_and_p,
        rank_shape=rank_shape, grid_size=grid_size,
        store_lap=p.no_fuse)

    gen = torch.Generator(device="cpu").manual_seed(7 + decomp.rank)
    f = (0.193 + 1e-3 * torch.rand((nscalars,) + pad, dtype=torch.float64,
                                   generator=gen)).to(device)
    dfdt = (-0.142 + 1e-3 * torch.rand((nscalars,) + pad,
                                       dtype=torch.float64,
                                       generator=gen)).to(device)
    arrays = {"f": f, "dfdt": dfdt}
    if p.gws and fuse_energy and p.gws_split:
        hij = torch.zeros((6,) + pad, dtype=torch.float64, device=device)
        hij_next = torch.zeros_like(hij)
        dhijdt = torch.zeros_like(hij)
        arrays["hij_a"] = hij[0:3]
        arrays["hij_b"] = hij[3:6]
        arrays["hij_a_next"] = hij_next[0:3]
        arrays["hij_b_next"] = hij_next[3:6]
        arrays["dhij_adt"] = dhijdt[0:3]
        arrays["dhij_bdt"] = dhijdt[3:6]
    elif p.gws:
        hij = torch.zeros((6,) + pad, dtype=torch.float64, device=device)
        arrays["hij"] = hij
        arrays["dhijdt"] = torch.zeros_like(hij)
        if fuse_energy:
            arrays["hij_next"] = torch.zeros_like(hij)
        if not fuse_energy:
            # unfused paths read the scalar gradients (stress tensor
            # source) from an array; the fused stage kernels compute
            # them inline (inline_grad)
            arrays["dfdx"] = torch.zeros(
                (nscalars, 3) + tuple(rank_shape),
                dtype=torch.float64, device=device)
    if p.no_fuse:
        arrays["lap_f"] = torch.zeros(
            (nscalars,) + tuple(rank_shape), dtype=torch.float64,
            device=device)
        if p.gws:
            arrays["lap_hij"] = torch.zeros(
                (6,) + tuple(rank_shape), dtype=torch.float64,
                device=device)
    else:
        arrays["f_next"] = torch.zeros_like(f)
        if p.gws and not fuse_energy:
            arrays["hij_next"] = torch.zeros_like(hij)

    energy = None

    def compute_energy(a):
        # fused: halo exchange + inline Laplacian stencil + energy
        # reduction (one kernel)
        kw = {k: v for k, v in arrays.items()
              if not k.endswith("_next")}
        return reduce_energy(a=np.array(a), **kw)

    energy = compute_energy(1.)
    expand = ps.Expansion(energy["total"], Stepper, mpl=mpl)

    device_loop = None
    if fuse_energy and on_gpu and not p.no_device_friedmann:
        # fully device-resident step: stage kernel + partials finish +
        # (RCCL allreduce) + on-device Friedmann ODE — zero host syncs
        from pystella_amd.fusion import DeviceFriedmannLoop
        device_loop = DeviceFriedmannLoop(
            stepper, decomp, expand, grid_size, dt, mpl=mpl)

    prof = None
    if os.environ.get("PYSTELLA_PROFILE"):
        from pystella_amd.profiling import Profiler
        prof = Profiler(enabled=True)
        step_bytes = 16 * grid_size / max(1, decomp.nranks) * 8 \
            * stepper.num_stages

    def step():
        nonlocal energy
        if prof is not None:
            with prof.region("step", bytes=step_bytes):
                _step_inner()
            return
        _step_inner()

    def _step_inner():
        nonlocal energy
        if device_loop is not None:
            device_loop.step(arrays)
            return
        for s in range(stepper.num_stages):
            if fuse_energy:
                # the stage kernel itself returns the input-state
                # energy — identical values to the reference loop's
                # standalone reduction after the previous stage
                # (gradients for the GW source are inlined)
                energy = stepper(s, a=expand.a, hubble=expand.hubble,
                                 **arrays)
                for name in stepper.pingpong:
                    arrays[name], arrays[f"{name}_next"] = \
                        arrays[f"{name}_next"], arrays[name]
                    decomp.share_halos(arrays[name])
                expand.step(s, energy["total"], energy["pressure"], dt)
            else:
                if p.gws:
                    derivs(fx=arrays["f"], grd=arrays["dfdx"])
                    if p.no_fuse:
                        derivs(fx=arrays["hij"],
                               lap=arrays["lap_hij"])
                stepper(s, a=expand.a, hubble=expand.hubble, **arrays)
                if not p.no_fuse:
                    for name in stepper.pingpong:
                        arrays[name], arrays[f"{name}_next"] = \
                            arrays[f"{name}_next"], arrays[name]
                        decomp.share_halos(arrays[name])
                expand.step(s, energy["total"], energy["pressure"], dt)
                energy = compute_energy(expand.a)

    def sync():
        if on_gpu:
            torch.cuda.synchronize()
        decomp.barrier()

    for _ in range(p.warmup):
        step()
    sync()
    t0 = time.perf_counter()
    for _ in range(p.steps):
        step()
    sync()
    elapsed = time.perf_counter() - t0
    # max over ranks
    elapsed = float(decomp.allreduce(elapsed, op="max"))

    # a diverged (NaN) run must not be reported as a valid measurement
    if device_loop is not None:
        final_energy = device_loop.read_state()["energy"]
    else:
        final_energy = float(np.asarray(energy["total"]).reshape(-1)[0])
    assert np.isfinite(final_energy), \
        f"run diverged: energy={final_energy}"

    ms_per_step = elapsed / p.steps * 1e3
    msites = grid_size * p.steps / elapsed / 1e6

    if prof is not None and decomp.rank == 0:
        import sys as _sys
        print(prof.report(), file=_sys.stderr)

    if decomp.rank == 0:
        print(json.dumps({
            "metric": "Msite-updates/sec, scalar-preheating 512^3 RK4",
            "value": msites,
            "unit": "Msites/s",
            "n_gpus": n_gpus,
            "steps": p.steps,
            "warmup": p.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "strong",
            "vs_baseline": None,
            "dtype": "fp64",
            "data": "synthetic random-init fields",
            "config": {
                "model": ("scalar_preheating+gw" if p.gws
                          else "scalar_preheating"),
                "grid_shape": list(grid_shape),
                "halo": h,
                "nscalars": nscalars,
                "stepper": f"{p.stepper} "
                           f"({stepper.num_stages} stages)",
                "global_batch": None,
                "seq_len": None,
                "parallelism": f"decomp3d{list(proc_shape)}",
            },
        }), flush=True)


if __name__ == "__main__":
    main()
