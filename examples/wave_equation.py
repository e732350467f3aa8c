"""Minimal wave-equation demo (analogue of reference
examples/wave_equation.py).

Runs on CPU by default; pass --device cuda on a GPU machine.
"""

import argparse

import torch

import pystella_amd as ps

parser = argparse.ArgumentParser()
parser.add_argument("--grid-shape", type=int, nargs=3, default=(32, 32, 32))
parser.add_argument("--proc-shape", type=int, nargs=3, default=(1, 1, 1))
parser.add_argument("--halo-shape", type=int, default=1)
parser.add_argument("--device", default="cpu")
parser.add_argument("--end-time", type=float, default=1.)


def main(args=None):
    p = parser.parse_args(args)
    grid_shape = tuple(p.grid_shape)
    h = p.halo_shape
    dx = tuple(10 / Ni for Ni in grid_shape)
    dt = min(dx) / 10

    ps.init_distributed()
    device = ps.choose_device() if p.device != "cpu" else torch.device("cpu")
    decomp = ps.DomainDecomposition(p.proc_shape, h, grid_shape=grid_shape)
    rank_shape = decomp.rank_shape
    pad = tuple(n + 2 * h for n in rank_shape)

    gen = torch.Generator(device="cpu").manual_seed(42 + decomp.rank)
    f = torch.rand(pad, dtype=torch.float64, generator=gen).to(device)
    dfdt = torch.rand(pad, dtype=torch.float64, generator=gen).to(device)
    lap_f = torch.zeros(rank_shape, dtype=torch.float64, device=device)

    f_ = ps.DynamicField("f", offset="h")
    rhs_dict = {f_: f_.dot, f_.dot: f_.lap}

    stepper = ps.LowStorageRK54(rhs_dict, dt=dt, halo_shape=h,
                                rank_shape=rank_shape)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=rank_shape)

    t = 0.
    while t < p.end_time - 1e-12:
        for s in range(stepper.num_stages):
            derivs(fx=f, lap=lap_f)
            stepper(s, f=f, dfdt=dfdt, lap_f=lap_f)
        t += dt

    energy = (dfdt[..., h:-h, h:-h, h:-h] ** 2).mean().item()
    if decomp.rank == 0:
        print(f"wave_equation done: t={t:.3f} <dfdt^2>={energy:.6e}")
    return energy


if __name__ == "__main__":
    main()
