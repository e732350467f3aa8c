"""Observables-path benchmark at 512^3 on 1 GPU: r2c FFT, power
spectrum binning, TT projection (f64 MFMA) and Rayleigh init."""

import sys
import time

import numpy as np
import torch

sys.path.insert(0, ".")
import pystella_amd as ps  # noqa: E402
from pystella_amd.fourier import DFT  # noqa: E402


def timeit(fn, n=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main(n=512):
    dev = torch.device("cuda", 0)
    torch.cuda.set_device(dev)
    grid = (n, n, n)
    L = 5.0
    dk = (2 * np.pi / L,) * 3
    dx = (L / n,) * 3
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid)
    fft = DFT(decomp, grid_shape=grid, dtype=np.float64, device=dev)
    spec = ps.PowerSpectra(decomp, fft, dk, L**3)
    proj = ps.Projector(fft, 2, dk, dx)
    gen = ps.RayleighGenerator(fft=fft, dk=dk, volume=L**3, seed=1)

    fx = torch.rand(grid, dtype=torch.float64, device=dev)
    kshape = fft.shape(True)
    print(f"== observables @ {n}^3 fp64, 1x MI355X")
    ms = timeit(lambda: fft.dft(fx))
    print(f"r2c FFT               {ms:8.3f} ms")
    fk = fft.dft(fx)
    ms = timeit(lambda: spec.bin_power(fk))
    print(f"spectrum binning      {ms:8.3f} ms")
    ms = timeit(lambda: spec(fx))
    print(f"full PowerSpectra     {ms:8.3f} ms")
    hij = (torch.randn((6,) + kshape, dtype=torch.float64, device=dev)
           + 1j * torch.randn((6,) + kshape, dtype=torch.float64,
                              device=dev)).to(torch.complex128)
    ms = timeit(lambda: proj.transverse_traceless(hij))
    gb = hij.numel() * 16 * 2 / 1e9
    print(f"TT projection (MFMA)  {ms:8.3f} ms   {gb/ms:5.2f} TB/s")

    # fused one-launch projector family (round 2)
    vec = hij[:3].clone().contiguous()
    gbv = vec.numel() * 16 * 2 / 1e9
    ms = timeit(lambda: proj.transversify(vector=vec))
    print(f"transversify (fused)  {ms:8.3f} ms   {gbv/ms:5.2f} TB/s")
    plus, minus = vec[0], vec[1]
    ms = timeit(lambda: proj.vec_to_pol(plus=plus, minus=minus,
                                        vector=vec))
    gb2 = vec.numel() * 16 / 1e9 + 2 * plus.numel() * 16 / 1e9
    print(f"vec_to_pol (fused)    {ms:8.3f} ms   {gb2/ms:5.2f} TB/s")
    ms = timeit(lambda: proj.decompose_vector(
        vector=vec, plus=vec[0], minus=vec[1], lng=vec[2],
        times_abs_k=True))
    print(f"decompose_vec (fused) {ms:8.3f} ms   {gb2/ms:5.2f} TB/s")
    ms = timeit(lambda: proj.tensor_to_pol(plus=hij[0], minus=hij[1],
                                           hij=hij))
    gb3 = hij.numel() * 16 / 1e9 + 2 * plus.numel() * 16 / 1e9
    print(f"tensor_to_pol (fused) {ms:8.3f} ms   {gb3/ms:5.2f} TB/s")
    vx = torch.rand((3,) + grid, dtype=torch.float64, device=dev)
    ms = timeit(lambda: spec.polarization(vx, proj), n=5)
    print(f"full polarization     {ms:8.3f} ms")
    ms = timeit(lambda: spec.vector_decomposition(vx, proj), n=5)
    print(f"full vec_decomp       {ms:8.3f} ms")
    ms = timeit(lambda: gen.init_field(fx), n=5)
    print(f"Rayleigh init_field   {ms:8.3f} ms")


if __name__ == "__main__":
    main(int(sys.argv[1]) if len(sys.argv) > 1 else 512)
