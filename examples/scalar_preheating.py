"""Scalar-field preheating simulation with optional gravitational waves.

MI355X-native analogue of reference examples/scalar_preheating.py:
two coupled scalar fields evolve through an FLRW background whose
expansion is driven by their volume-averaged energy density; output
includes energy components, field statistics, histograms and power
spectra, and optionally the sourced tensor-perturbation (GW) sector.
"""

import argparse
import time

import numpy as np
import torch

import pystella_amd as ps
from pystella_amd.sectors import get_rho_and_p

parser = argparse.ArgumentParser()
parser.add_argument("--grid-shape", "-grid", type=int, nargs=3,
                    default=(128, 128, 128))
parser.add_argument("--proc-shape", "-proc", type=int, nargs=3,
                    default=(1, 1, 1))
parser.add_argument("--halo-shape", type=int, default=2)
parser.add_argument("--box-dim", "-box", type=float, nargs=3,
                    default=(5, 5, 5))
parser.add_argument("--kappa", type=float, default=1 / 10)
parser.add_argument("--mpl", type=float, default=1)
parser.add_argument("--mphi", type=float, default=1.20e-6)
# per-scalar couplings: pass one value per coupled scalar chi_i
# (reference CLI surface: scalar_preheating.py:52-58 nargs="*"; this
# implementation actually supports multiple chis — the reference
# declares the lists but hardcodes nscalars=2)
parser.add_argument("--mchi", type=float, nargs="*", default=0.,
                    help="the mass(es) of coupled scalars")
parser.add_argument("--gsq", type=float, nargs="*", default=2.5e-7,
                    help="the 2-2 coupling of phi to other scalars")
parser.add_argument("--sigma", type=float, nargs="*", default=0.,
                    help="the trilinear coupling of phi to other scalars")
parser.add_argument("--lambda4", type=float, nargs="*", default=0.,
                    help="the quartic self-coupling of other scalars")
parser.add_argument("--dtype", default="float64",
                    choices=["float64", "float32"])
parser.add_argument("--end-time", "-end-t", type=float, default=20)
parser.add_argument("--end-scale-factor", "-end-a", type=float, default=20)
parser.add_argument("--gravitational-waves", "-gws", action="store_true")
parser.add_argument("--device", default=None)
parser.add_argument("--outfile", default=None)
parser.add_argument("--no-output", action="store_true")
# cross-code validation hooks: dump the initial (f, dfdt) realization
# to an NPZ, or load one instead of drawing a Rayleigh realization —
# lets this code and the reference run from IDENTICAL initial data
# (see tools/reference_crosscheck.py)
parser.add_argument("--save-init", default=None, metavar="FILE.npz")
parser.add_argument("--load-init", default=None, metavar="FILE.npz")


def main(args=None):
    p = parser.parse_args(args)
    p.grid_shape = tuple(p.grid_shape)
    p.proc_shape = tuple(p.proc_shape)
    grid_size = float(np.prod(p.grid_shape))
    p.box_dim = tuple(p.box_dim)
    volume = float(np.prod(p.box_dim))
    dx = tuple(L / N for L, N in zip(p.box_dim, p.grid_shape))
    dk = tuple(2 * np.pi / L for L in p.box_dim)
    dt = p.kappa * min(dx)
    h = p.halo_shape

    def _coupling_list(v):
        if isinstance(v, (int, float)):
            return [float(v)]
        return [float(x) for x in v] or [0.0]

    mchi, gsq, sigma, lambda4 = map(
        _coupling_list, (p.mchi, p.gsq, p.sigma, p.lambda4))
    nchi = max(map(len, (mchi, gsq, sigma, lambda4)))
    for lst in (mchi, gsq, sigma, lambda4):
        if len(lst) == 1:
            lst *= nchi
        elif len(lst) != nchi:
            raise SystemExit("coupling lists must have equal lengths")

    nscalars = 1 + nchi
    f0 = [.193 * p.mpl] + [0] * nchi
    df0 = [-.142231 * p.mpl] + [0] * nchi
    np_dtype = np.dtype(p.dtype)
    torch_dtype = (torch.float64 if np_dtype == np.float64
                   else torch.float32)
    Stepper = ps.LowStorageRK54

    ps.init_distributed()
    if p.device is None:
        device = ps.choose_device()
    else:
        device = torch.device(p.device)

    decomp = ps.DomainDecomposition(p.proc_shape, h,
                                    grid_shape=p.grid_shape)
    rank_shape = decomp.rank_shape
    pad = tuple(n + 2 * h for n in rank_shape)

    fft = ps.DFT(decomp, grid_shape=p.grid_shape, dtype=np_dtype,
                 device=device)
    derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=rank_shape)

    def potential(f):
        phi = f[0]
        unscaled = p.mphi**2 / 2 * phi**2
        for i in range(nchi):
            chi = f[1 + i]
            unscaled = (unscaled
                        + mchi[i]**2 / 2 * chi**2
                        + gsq[i] / 2 * phi**2 * chi**2
                        + sigma[i] / 2 * phi * chi**2
                        + lambda4[i] / 4 * chi**4)
        return unscaled / p.mphi**2

    scalar_sector = ps.ScalarSector(nscalars, potential=potential)
    sectors = [scalar_sector]
    if p.gravitational_waves:
        gw_sector = ps.TensorPerturbationSector([scalar_sector])
        sectors += [gw_sector]

    stepper = Stepper(sectors, halo_shape=h, rank_shape=rank_shape, dt=dt)

    reduce_energy = ps.Reduction(
        decomp, scalar_sector, halo_shape=h, callback=get_rho_and_p,
        rank_shape=rank_shape, grid_size=grid_size)

    def compute_energy(f, dfdt, lap_f, dfdx, a):
        if p.gravitational_waves:
            derivs(fx=f, lap=lap_f, grd=dfdx)
        else:
            derivs(fx=f, lap=lap_f)
        return reduce_energy(f=f, dfdt=dfdt, lap_f=lap_f,
                             a=np.array(a))

    # observables
    out = None
    if decomp.rank == 0 and not p.no_output:
        out = ps.OutputFile(name=p.outfile, runfile=__file__)
    statistics = ps.FieldStatistics(decomp, h, rank_shape=rank_shape,
                                    grid_size=grid_size)
    spectra = ps.PowerSpectra(decomp, fft, dk, volume)
    projector = ps.Projector(fft, h, dk, dx)
    hist = ps.FieldHistogrammer(decomp, 1000, np_dtype,
                                rank_shape=rank_shape)

    a_sq_rho = (3 * p.mpl**2 * ps.Field("hubble", indices=[])**2
                / 8 / np.pi)
    rho_dict = {ps.Field("rho", offset=0):
                scalar_sector.stress_tensor(0, 0) / a_sq_rho}
    compute_rho = ps.ElementWiseMap(rho_dict, halo_shape=h,
                                    rank_shape=rank_shape)

    def output(step_count, t, energy, expand,
               f, dfdt, lap_f, dfdx, hij, dhijdt, lap_hij):
        if p.no_output:
            return
        if step_count % 4 == 0:
            f_stats = statistics(f)
            if out is not None:
                out.output(
                    "energy", t=t, a=expand.a[0],
                    adot=expand.adot[0] / expand.a[0],
                    hubble=expand.hubble[0] / expand.a[0],
                    **{k: np.asarray(v) for k, v in energy.items()},
                    eos=energy["pressure"] / energy["total"],
                    constraint=expand.constraint(energy["total"]))
                out.output("statistics/f", t=t, a=expand.a[0], **f_stats)

        if expand.a[0] / output.a_last_spec >= 1.05:
            output.a_last_spec = expand.a[0]
            if not p.gravitational_waves:
                derivs(fx=f, grd=dfdx)
            tmp = torch.empty(rank_shape, dtype=torch_dtype,
                              device=device)
            compute_rho(a=expand.a, hubble=expand.hubble, rho=tmp,
                        f=f, dfdt=dfdt, dfdx=dfdx)
            rho_hist = hist(tmp)
            spec_out = {"scalar": spectra(f), "rho": spectra(tmp)}
            if p.gravitational_waves:
                spec_out["gw"] = spectra.gw(dhijdt, projector,
                                            expand.hubble[0])
            if out is not None:
                out.output("rho_histogram", t=t, a=expand.a[0], **rho_hist)
                out.output("spectra", t=t, a=expand.a[0], **spec_out)

    output.a_last_spec = .1

    # field allocation & init
    f = torch.empty((nscalars,) + pad, dtype=torch_dtype, device=device)
    dfdt = torch.empty_like(f)
    dfdx = torch.empty((nscalars, 3) + tuple(rank_shape),
                       dtype=torch_dtype, device=device)
    lap_f = torch.empty((nscalars,) + tuple(rank_shape),
                        dtype=torch_dtype, device=device)
    if p.gravitational_waves:
        hij = torch.zeros((6,) + pad, dtype=torch_dtype, device=device)
        dhijdt = torch.zeros_like(hij)
        lap_hij = torch.zeros((6,) + tuple(rank_shape),
                              dtype=torch_dtype, device=device)
    else:
        hij = dhijdt = lap_hij = None

    for i in range(nscalars):
        f[i] = f0[i]
        dfdt[i] = df0[i]

    energy = compute_energy(f, dfdt, lap_f, dfdx, 1.)
    expand = ps.Expansion(energy["total"], Stepper, mpl=p.mpl)

    # effective masses incl. Hubble correction
    addot = expand.addot_friedmann_2(expand.a, energy["total"],
                                     energy["pressure"])
    hubble_correction = float((-addot / expand.a)[0])
    fields = [ps.var("f0")[i] for i in range(nscalars)]
    d2Vd2f = [ps.diff(potential(fields), field, field) for field in fields]
    from pystella_amd.backend.torcheval import eval_expr, EvalContext
    ctx = EvalContext(0, (1, 1, 1))
    eff_mass = [
        float(eval_expr(x, {"f0": np.array(f0)}, ctx)) + hubble_correction
        for x in d2Vd2f]

    modes = ps.RayleighGenerator(fft=fft, dk=dk, volume=volume,
                                 seed=49279 * (decomp.rank + 1))
    for fld in range(nscalars):
        modes.init_WKB_fields(
            f[fld], dfdt[fld], norm=p.mphi**2,
            omega_k=lambda k, fld=fld: np.sqrt(k**2 + eff_mass[fld]),
            hubble=float(expand.hubble[0]))
    for i in range(nscalars):
        f[i] += f0[i]
        dfdt[i] += df0[i]

    if p.load_init:
        data = np.load(p.load_init) if decomp.rank == 0 else None
        for name, t in (("f", f), ("dfdt", dfdt)):
            src = (torch.as_tensor(np.ascontiguousarray(data[name]))
                   .to(device=device, dtype=torch_dtype)
                   if decomp.rank == 0 else t)
            piece = decomp.scatter_array(src)
            decomp.restore_halos(t, piece.to(device))
            decomp.share_halos(t)
    if p.save_init:
        full_f = decomp.gather_array(decomp.remove_halos(f))
        full_df = decomp.gather_array(decomp.remove_halos(dfdt))
        if decomp.rank == 0:
            np.savez(p.save_init, f=full_f.cpu().numpy(),
                     dfdt=full_df.cpu().numpy())

    energy = compute_energy(f, dfdt, lap_f, dfdx, float(expand.a[0]))
    expand = ps.Expansion(energy["total"], Stepper, mpl=p.mpl)

    t = 0.
    step_count = 0
    output(step_count, t, energy, expand, f=f, dfdt=dfdt, lap_f=lap_f,
           dfdx=dfdx, hij=hij, dhijdt=dhijdt, lap_hij=lap_hij)

    if decomp.rank == 0:
        print("time\tscale factor\tms/step\tsteps/s", flush=True)
    start = time.time()
    last_out = start

    while t < p.end_time and expand.a[0] < p.end_scale_factor:
        for s in range(stepper.num_stages):
            stepper(s, a=expand.a, hubble=expand.hubble,
                    f=f, dfdt=dfdt, dfdx=dfdx, lap_f=lap_f,
                    hij=hij, dhijdt=dhijdt, lap_hij=lap_hij,
                    filter_args=True)
            expand.step(s, energy["total"], energy["pressure"], dt)
            energy = compute_energy(f, dfdt, lap_f, dfdx, expand.a)
            if p.gravitational_waves:
                derivs(fx=hij, lap=lap_hij)
        t += dt
        step_count += 1
        output(step_count, t, energy, expand, f=f, dfdt=dfdt, lap_f=lap_f,
               dfdx=dfdx, hij=hij, dhijdt=dhijdt, lap_hij=lap_hij)
        if time.time() - last_out > 30 and decomp.rank == 0:
            last_out = time.time()
            ms = (last_out - start) * 1e3 / step_count
            print(f"{t:<12.3f}{expand.a[0]:<12.3f}{ms:<12.3f}"
                  f"{1e3 / ms:<12.3f}", flush=True)

    if decomp.rank == 0:
        print("simulation complete")
    return expand, energy


if __name__ == "__main__":
    main()
