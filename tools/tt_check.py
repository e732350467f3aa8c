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
