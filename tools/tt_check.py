import sys
sys.path.insert(0, ".")
import numpy as np
import torch
import pystella_amd as ps
from pystella_amd.fourier import DFT

for n in (64, 192, 320):
    grid_shape = (n, n, n)
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    L = 5.0
    dk = (2 * np.pi / L,) * 3
    dx = (L / n,) * 3
    fft_g = DFT(decomp, grid_shape=grid_shape, dtype=np.float64, device="cuda")
    proj_g = ps.Projector(fft_g, 1, dk, dx)
    fft_c = DFT(decomp, grid_shape=grid_shape, dtype=np.float64, device="cpu")
    proj_c = ps.Projector(fft_c, 1, dk, dx)
    kshape = fft_g.shape(True)
    torch.manual_seed(1)
    hij = (torch.randn((6,) + kshape, dtype=torch.float64)
           + 1j * torch.randn((6,) + kshape, dtype=torch.float64)).to(torch.complex128)
    want = hij.clone()
    proj_c.transverse_traceless(want)
    got = hij.clone().cuda().contiguous()
    proj_g.transverse_traceless(got)
    torch.cuda.synchronize()
    err = (got.cpu() - want).abs().max().item()
    scale = want.abs().max().item()
    print(f"n={n} kshape={tuple(kshape)} err={err:.3e} scale={scale:.1f}", flush=True)

# large-volume timing sanity (the uint32 work-item overflow regression)
if "--big" in sys.argv:
    import time
    n = 1024
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=(n,)*3)
    L = 5.0
    dk = (2 * np.pi / L,) * 3
    dx = (L / n,) * 3
    fft_g = DFT(decomp, grid_shape=(n,)*3, dtype=np.float64,
                device="cuda")
    proj_g = ps.Projector(fft_g, 1, dk, dx)
    kshape = fft_g.shape(True)
    hij = torch.randn((6,) + kshape, dtype=torch.float64,
                      device="cuda").to(torch.complex128)
    for _ in range(2):
        proj_g.transverse_traceless(hij)
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(5):
        proj_g.transverse_traceless(hij)
    torch.cuda.synchronize()
    ms = (time.perf_counter() - t0) / 5 * 1e3
    gb = hij.numel() * 16 * 2 / 1e9
    print(f"TT 1024^3: {ms:.2f} ms  {gb/ms:.2f} TB/s", flush=True)
    # sanity: a real pass over 103 GB cannot take < 10 ms
    assert ms > 10.0, ms
