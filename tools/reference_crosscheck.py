"""Run the REFERENCE (zachjweiner/pystella) from pinned initial data
and print its end state, for cross-code validation against this repo.

This image ships no pyopencl/loopy, so this script cannot run here —
it is the offline half of the cross-code experiment (the in-image
half is tests/test_crosscheck.py, which validates this repo's
integration against an independent numpy transcription of the same
physics):

1. here:   python examples/scalar_preheating.py --grid-shape 32 32 32 \
               --end-time 1 --no-output --save-init init.npz
           -> prints final a and Friedmann constraint
2. offline (env with pyopencl+loopy+pocl and the reference repo):
           python tools/reference_crosscheck.py init.npz
           -> prints the reference's final a and constraint from the
              SAME initial data
3. compare: both runs integrate identical data with the same scheme
           (LowStorageRK54, h=2 stencils, per-stage Friedmann), so
           the end states must agree to integration tolerance.

Usage: python tools/reference_crosscheck.py init.npz [end_time]
"""

import sys

import numpy as np


def main():
    try:
        import pyopencl as cl
        import pystella as ps
    except ImportError as e:
        raise SystemExit(
            f"needs the reference's environment (pyopencl/loopy): {e}")

    npz = np.load(sys.argv[1])
    end_time = float(sys.argv[2]) if len(sys.argv) > 2 else 1.0
    f_init = npz["f"]          # (nscalars, Nx, Ny, Nz)
    df_init = npz["dfdt"]
    nscalars, *grid_shape = f_init.shape
    grid_shape = tuple(grid_shape)

    mphi, mpl, gsq = 1.2e-6, 1.0, 2.5e-7
    box = (5., 5., 5.)
    dx = tuple(L / N for L, N in zip(box, grid_shape))
    dt = 1 / 10 * min(dx)
    h = 2
    proc_shape = (1, 1, 1)

    ctx = ps.choose_device_and_make_context()
    queue = cl.CommandQueue(ctx)
    decomp = ps.DomainDecomposition(proc_shape, h, grid_shape)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=grid_shape)

    def potential(f):
        phi, chi = f[0], f[1]
        return (mphi**2 / 2 * phi**2
                + gsq / 2 * phi**2 * chi**2) / mphi**2

    sector = ps.ScalarSector(nscalars, potential=potential)
    stepper = ps.LowStorageRK54([sector], halo_shape=h,
                                rank_shape=grid_shape, dt=dt)
    from pystella.sectors import get_rho_and_p
    reduce_energy = ps.Reduction(
        decomp, sector, halo_shape=h, callback=get_rho_and_p,
        rank_shape=grid_shape, grid_size=float(np.prod(grid_shape)))

    import pyopencl.array as cla
    pad = tuple(n + 2 * h for n in grid_shape)
    f = cla.zeros(queue, (nscalars,) + pad, np.float64)
    dfdt = cla.zeros(queue, (nscalars,) + pad, np.float64)
    lap_f = cla.zeros(queue, (nscalars,) + grid_shape, np.float64)
    for i in range(nscalars):
        fi = np.zeros(pad)
        fi[h:-h, h:-h, h:-h] = f_init[i]
        f[i] = fi
        di = np.zeros(pad)
        di[h:-h, h:-h, h:-h] = df_init[i]
        dfdt[i] = di

    def compute_energy(a):
        derivs(queue, fx=f, lap=lap_f)
        return reduce_energy(queue, f=f, dfdt=dfdt, lap_f=lap_f,
                             a=np.array(a))

    energy = compute_energy(1.)
    expand = ps.Expansion(energy["total"], ps.LowStorageRK54, mpl=mpl)

    t = 0.
    while t < end_time:
        for s in range(stepper.num_stages):
            stepper(s, queue=queue, a=expand.a, hubble=expand.hubble,
                    f=f, dfdt=dfdt, lap_f=lap_f, filter_args=True)
            expand.step(s, energy["total"], energy["pressure"], dt)
            energy = compute_energy(expand.a)
        t += dt

    constraint = expand.constraint(energy["total"])
    print(f"reference end state: a={float(expand.a[0]):.16g} "
          f"constraint={float(constraint):.16g} "
          f"energy={float(energy['total']):.16g}")


if __name__ == "__main__":
    main()
