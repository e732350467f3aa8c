"""Poisson geometric-multigrid benchmark: 1024^3 fp32 on MI355X
(BASELINE.json config #5).

Manufactured-solution Poisson problem, FAS V-cycles with a Newton
smoother; reports seconds per V-cycle and residual-reduction per cycle.
"""

import argparse
import sys
import time

import numpy as np
import torch

sys.path.insert(0, ".")
import pystella_amd as ps  # noqa: E402
from pystella_amd.field import Field, shift_fields, var  # noqa: E402
from pystella_amd.multigrid import (  # noqa: E402
    FullApproximationScheme, NewtonIterator, RedBlackIterator, v_cycle)


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--n", "--grid", dest="n", type=int,
                    default=1024)
    ap.add_argument("--dtype", default="float32")
    ap.add_argument("--cycles", type=int, default=4)
    ap.add_argument("--depth", type=int, default=5)
    ap.add_argument("--device", default=None)
    ap.add_argument("--smoother", default="rbgs",
                    choices=("rbgs", "newton"))
    p = ap.parse_args()

    dtype = getattr(torch, p.dtype)
    device = (torch.device(p.device) if p.device
              else torch.device("cuda" if torch.cuda.is_available()
                                else "cpu"))
    n, h = p.n, 1
    grid = (n, n, n)
    # distributed: launch under torchrun (one rank per GPU); the
    # decomposition follows WORLD_SIZE like bench.py
    import os
    world = int(os.environ.get("WORLD_SIZE", "1"))
    if world > 1:
        ps.init_distributed()
        if device.type == "cuda":
            device = ps.choose_device()
    shapes = {1: (1, 1, 1), 2: (2, 1, 1), 4: (2, 2, 1), 8: (2, 2, 2)}
    proc_shape = shapes.get(world, (world, 1, 1))
    decomp = ps.DomainDecomposition(proc_shape, h, grid_shape=grid)
    L = 2 * np.pi
    dx = (L / n,) * 3

    f = Field("f", offset="h")
    rho = Field("rho", offset="h")
    lap = sum(
        (shift_fields(f, tuple(s * int(mu == d) for mu in range(3)))
         - 2 * f
         + shift_fields(f, tuple(-s * int(mu == d) for mu in range(3))))
        for d in range(3) for s in [1]) / var("dx")[0]**2
    problems = {f: (lap, rho)}
    if p.smoother == "rbgs":
        solver = RedBlackIterator(decomp, problems, halo_shape=h,
                                  fixed_parameters=dict(omega=1.0))
    else:
        solver = NewtonIterator(decomp, problems, halo_shape=h,
                                fixed_parameters=dict(omega=0.8))
    mg = FullApproximationScheme(solver, halo_shape=h)

    # manufactured solution: f* = sin(x)sin(y)sin(z), rho = -3 f*
    # (built directly on the DEVICE in the target dtype — no full-grid
    # fp64 host arrays; at 2048^3 those alone are 69 GB each and can
    # OOM the host)
    rank_shape, start = decomp.get_rank_shape_start(grid)
    axes = [(torch.arange(s, s + m, dtype=torch.float64,
                          device=device) * d).to(dtype)
            for s, m, d in zip(start, rank_shape, dx)]
    sins = [torch.sin(a) for a in axes]

    def f_exact_on(dev=None):
        return (sins[0][:, None, None] * sins[1][None, :, None]
                * sins[2][None, None, :])

    pad = tuple(m + 2 * h for m in rank_shape)
    rho_t = torch.zeros(pad, dtype=dtype, device=device)
    rho_t[h:-h, h:-h, h:-h] = -3.0 * f_exact_on()
    decomp.share_halos(rho_t)

    ff = torch.zeros(pad, dtype=dtype, device=device)
    cyc = v_cycle(10, 20, p.depth)

    def one_cycle():
        return mg(decomp, dx0=dx, cycle=cyc, f=ff, rho=rho_t)

    errs = one_cycle()   # warmup (includes setup + JIT)
    if device.type == "cuda":
        torch.cuda.synchronize()
    decomp.barrier()
    t0 = time.perf_counter()
    for _ in range(p.cycles):
        errs = one_cycle()
    if device.type == "cuda":
        torch.cuda.synchronize()
    decomp.barrier()
    dtime = (time.perf_counter() - t0) / p.cycles
    dtime = float(decomp.allreduce(dtime, op="max"))

    final = [e for lvl, e in errs if lvl == 0][-1]["f"]
    initial = [e for lvl, e in errs if lvl == 0][0]["f"]
    # error vs the exact solution without materializing extra
    # full-grid temporaries: stream x-plane by x-plane
    want = f_exact_on()
    got = ff[h:-h, h:-h, h:-h]
    gmean = (got.double().mean() if got.numel() < 2**30
             else sum(got[i].double().mean()
                      for i in range(got.shape[0])) / got.shape[0])
    wmean = want.double().mean()
    num, den = 0.0, 0.0
    for i in range(got.shape[0]):
        d_ = (got[i].double() - gmean) - (want[i].double() - wmean)
        num = max(num, d_.abs().max().item())
        den = max(den, (want[i].double() - wmean).abs().max().item())
    rel = num / den
    import json
    if decomp.rank != 0:
        return
    print(json.dumps({
        "n_ranks": decomp.nranks,
        "metric": "seconds per FAS V-cycle, Poisson",
        "value": dtime, "unit": "s", "higher_is_better": False,
        "grid": list(grid), "dtype": p.dtype, "depth": p.depth,
        "smoother": p.smoother,
        "resid_L2_start": float(initial[1]),
        "resid_L2_end": float(final[1]),
        "rel_err_vs_exact": rel,
    }))


if __name__ == "__main__":
    main()
