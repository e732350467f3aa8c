"""Emit the GW hij ring-stage kernel source (flat and sectioned forms)
and cross-compile each with hipcc --offload-arch=gfx950 to validate
syntax and report the register allocation (no GPU needed).

Usage: python tools/emit_gw_kernel.py [--sections N] [--no-compile]
"""

import argparse
import os
import subprocess
import sys
import tempfile

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np  # noqa: E402
import torch  # noqa: E402


def build_sources(sections):
    """Return {kernel_name: source} for the bench --gws stage kernels,
    built exactly as DeviceFriedmannLoop would build them."""
    import pystella_amd as ps
    from pystella_amd.fusion import StencilRKStepper
    from pystella_amd.sectors import get_rho_and_p
    import pystella_amd.backend.hip as hip

    # stub out compilation: capture sources instead
    captured = {}

    class _StubExt:
        def jit_compile(self, src, name):
            captured[name] = src
            return name

        def jit_launch(self, *a, **k):
            raise RuntimeError("launch not available here")

    orig_ext = hip.ext
    hip.ext = lambda: _StubExt()
    try:
        grid = (512, 512, 512)
        h = 2
        dx = tuple(5 / n for n in grid)
        dt = 1e-3
        decomp = ps.DomainDecomposition((1, 1, 1), h, grid_shape=grid)
        rank_shape = decomp.rank_shape

        def potential(f):
            return (1.2e-6**2 / 2 * f[0]**2
                    + 2.5e-7 / 2 * f[0]**2 * f[1]**2) / 1.2e-6**2

        sector = ps.ScalarSector(2, potential=potential)
        sectors = [sector, ps.TensorPerturbationSector([sector])]
        derivs = ps.FiniteDifferencer(decomp, h, dx, rank_shape=rank_shape)
        stepper = StencilRKStepper(
            ps.LowStorageRK54, sectors, derivs, halo_shape=h,
            rank_shape=rank_shape, dt=dt, reducers=sector,
            grid_size=float(np.prod(grid)), callback=get_rho_and_p,
            inline_grad=True)

        smap = stepper._stepper.steps[1]
        pad = tuple(n + 2 * h for n in rank_shape)
        env = {
            "f": torch.zeros((2,) + pad), "f_next": torch.zeros((2,) + pad),
            "dfdt": torch.zeros((2,) + pad),
            "f_tmp": torch.zeros((2,) + pad),
            "dfdt_tmp": torch.zeros((2,) + pad),
            "hij": torch.zeros((6,) + pad),
            "hij_next": torch.zeros((6,) + pad),
            "dhijdt": torch.zeros((6,) + pad),
            "hij_tmp": torch.zeros((6,) + pad),
            "dhijdt_tmp": torch.zeros((6,) + pad),
            "a": np.ones(1), "hubble": np.zeros(1), "dt": dt,
        }
        if not stepper._stepper.tmp_arrays:
            stepper._stepper.tmp_arrays = \
                stepper._stepper.get_tmp_arrays_like(**env)
        env.update(stepper._stepper.tmp_arrays)

        m = smap._map
        rs = m._infer_rank_shape(env)
        if sections is not None:
            os.environ["PYSTELLA_SECTIONS"] = str(sections)
        from pystella_amd.backend.hip import get_lap_stage_kernel
        for (rk_o, tmp_o, red_o, f_name, nf), fargs in zip(
                smap.ring, smap._ring_field_args):
            get_lap_stage_kernel(
                rk_o, tmp_o, red_o or [(0.0, "sum")], fargs, [],
                m.halo_shape, rs, smap.derivs.dx, nf,
                f_name=f_name, lap_name=f"lap_{f_name}",
                name=f"stage1_{f_name}",
                state_map={"a": 0, "hubble": 4})
    finally:
        hip.ext = orig_ext
        os.environ.pop("PYSTELLA_SECTIONS", None)
    return captured


def compile_one(name, src, keep=None):
    with tempfile.TemporaryDirectory() as td:
        path = os.path.join(td, f"{name}.hip")
        with open(path, "w") as f:
            f.write("#include <hip/hip_runtime.h>\n")
            f.write(src)
        out = os.path.join(td, f"{name}.o")
        cmd = ["hipcc", "--offload-arch=gfx950", "-O3", "-c", path,
               "-o", out, "-Rpass-analysis=kernel-resource-usage"]
        r = subprocess.run(cmd, capture_output=True, text=True)
        if r.returncode != 0:
            print(f"--- {name}: COMPILE FAILED ---")
            print(r.stderr[-3000:])
            if keep:
                with open(keep, "w") as f:
                    f.write(src)
                print(f"source kept at {keep}")
            return False
        usage = [ln for ln in r.stderr.splitlines()
                 if "SGPRs" in ln or "VGPRs" in ln or "Occupancy" in ln
                 or "kernel-resource-usage" in ln or "Spill" in ln]
        print(f"--- {name}: OK ---")
        for ln in usage:
            print("   ", ln.strip())
        return True


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--sections", type=int, default=None)
    ap.add_argument("--no-compile", action="store_true")
    ap.add_argument("--dump", default=None,
                    help="write each source to DIR/<name>.hip")
    p = ap.parse_args()

    srcs = build_sources(p.sections)
    print("kernels built:", sorted(srcs))
    for name, src in sorted(srcs.items()):
        if p.dump:
            os.makedirs(p.dump, exist_ok=True)
            with open(os.path.join(p.dump, f"{name}.hip"), "w") as f:
                f.write(src)
        if not p.no_compile:
            ok = compile_one(name, src,
                             keep=f"/tmp/{name}_failed.hip")
            if not ok:
                sys.exit(1)


if __name__ == "__main__":
    main()
