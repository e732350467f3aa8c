"""DFT tests: single-rank vs numpy, unnormalized-roundtrip convention,
halo strip/restore, and the distributed pencil FFT vs the single-rank
result (gloo, world_size 2)."""

import numpy as np
import torch

import pystella_amd as ps
from tests.conftest import run_distributed


def test_single_rank_r2c(grid_shape=(16, 12, 8)):
    decomp = ps.DomainDecomposition((1, 1, 1), 0, rank_shape=grid_shape)
    fft = ps.DFT(decomp, grid_shape=grid_shape, dtype=np.float64)
    rng = np.random.default_rng(0)
    fx = rng.random(grid_shape)
    fk = fft.dft(torch.as_tensor(fx)).clone()
    expect = np.fft.rfftn(fx)
    assert np.allclose(fk.numpy(), expect, atol=1e-10)

    # unnormalized roundtrip: idft(dft(x)) == N x
    out = torch.empty(grid_shape, dtype=torch.float64)
    fft.idft(fk, out)
    assert np.allclose(out.numpy(), fx * np.prod(grid_shape), atol=1e-8)


def test_halo_strip_restore(grid_shape=(8, 8, 8)):
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    fft = ps.DFT(decomp, grid_shape=grid_shape, dtype=np.float64)
    rng = np.random.default_rng(1)
    pad = tuple(n + 2 * h for n in grid_shape)
    fx = torch.zeros(pad, dtype=torch.float64)
    interior = rng.random(grid_shape)
    fx[h:-h, h:-h, h:-h] = torch.as_tensor(interior)
    fk = fft.dft(fx)
    assert np.allclose(fk.numpy(), np.fft.rfftn(interior), atol=1e-10)
    out = torch.zeros(pad, dtype=torch.float64)
    fft.idft(fk, out)
    assert np.allclose(out[h:-h, h:-h, h:-h].numpy(),
                       interior * np.prod(grid_shape), atol=1e-8)


def _pencil_worker(rank, world_size, proc_shape, grid_shape):
    decomp = ps.DomainDecomposition(proc_shape, 0, grid_shape=grid_shape)
    fft = ps.DFT(decomp, grid_shape=grid_shape, dtype=np.float64)
    rank_shape, start = decomp.get_rank_shape_start(grid_shape)

    rng = np.random.default_rng(5)
    full = rng.random(grid_shape)
    sl = tuple(slice(s, s + n) for s, n in zip(start, rank_shape))
    fx = torch.as_tensor(full[sl]).contiguous()

    fk = fft.dft(fx).clone()
    expect_full = np.fft.rfftn(full)
    # k-layout: kx full, ky split over px (rx), kz split over the
    # py*pz row dimension at position q = ry*pz + rz
    from pystella_amd.decomp import get_size_start
    px, py, pz = proc_shape
    Ny = grid_shape[1]
    NKz = grid_shape[2] // 2 + 1
    q = decomp.ry * pz + decomp.rz
    ny2, y0 = get_size_start(Ny, px, decomp.rx)
    nkz, z0 = get_size_start(NKz, py * pz, q)
    expect = expect_full[:, y0:y0 + ny2, z0:z0 + nkz]
    assert np.allclose(fk.numpy(), expect, atol=1e-8), \
        f"rank {rank} fk mismatch"

    # sub_k matches the slice actually held
    kx = np.fft.fftfreq(grid_shape[0], 1 / grid_shape[0])
    if grid_shape[0] % 2 == 0:
        kx[grid_shape[0] // 2] = abs(kx[grid_shape[0] // 2])
    ky = np.fft.fftfreq(Ny, 1 / Ny)
    if Ny % 2 == 0:
        ky[Ny // 2] = abs(ky[Ny // 2])
    kz = np.fft.rfftfreq(grid_shape[2], 1 / grid_shape[2])
    assert np.allclose(fft.sub_k["momenta_x"].numpy(), kx)
    assert np.allclose(fft.sub_k["momenta_y"].numpy(), ky[y0:y0 + ny2])
    assert np.allclose(fft.sub_k["momenta_z"].numpy(), kz[z0:z0 + nkz])

    out = torch.empty(rank_shape, dtype=torch.float64)
    fft.idft(fk, out)
    assert np.allclose(out.numpy(), full[sl] * np.prod(grid_shape),
                       atol=1e-6)


def test_pencil_fft_slab_x():
    run_distributed(_pencil_worker, 2, args=((2, 1, 1), (8, 8, 8)))


def test_pencil_fft_slab_y():
    run_distributed(_pencil_worker, 2, args=((1, 2, 1), (8, 8, 8)))


def test_pencil_fft_slab_z():
    run_distributed(_pencil_worker, 2, args=((1, 1, 2), (8, 8, 8)))


def test_pencil_fft_uneven():
    run_distributed(_pencil_worker, 2, args=((2, 1, 1), (10, 6, 8)))


def test_pencil_fft_uneven_z():
    run_distributed(_pencil_worker, 2, args=((1, 1, 2), (8, 10, 6)))


def test_pencil_fft_2d_yz():
    run_distributed(_pencil_worker, 4, args=((1, 2, 2), (8, 8, 8)))


def test_pencil_fft_3d_222():
    """The driver's natural N=8 topology (2,2,2): full 3-D pencil."""
    run_distributed(_pencil_worker, 8, args=((2, 2, 2), (8, 8, 8)))


def test_pencil_fft_3d_222_uneven():
    run_distributed(_pencil_worker, 8, args=((2, 2, 2), (10, 6, 12)))


def _pencil_c2c_worker(rank, world_size, proc_shape, grid_shape):
    decomp = ps.DomainDecomposition(proc_shape, 0, grid_shape=grid_shape)
    fft = ps.DFT(decomp, grid_shape=grid_shape, dtype=np.complex128)
    rank_shape, start = decomp.get_rank_shape_start(grid_shape)

    rng = np.random.default_rng(6)
    full = rng.random(grid_shape) + 1j * rng.random(grid_shape)
    sl = tuple(slice(s, s + n) for s, n in zip(start, rank_shape))
    fx = torch.as_tensor(full[sl]).contiguous()

    fk = fft.dft(fx).clone()
    expect_full = np.fft.fftn(full)
    from pystella_amd.decomp import get_size_start
    px, py, pz = proc_shape
    q = decomp.ry * pz + decomp.rz
    ny2, y0 = get_size_start(grid_shape[1], px, decomp.rx)
    nkz, z0 = get_size_start(grid_shape[2], py * pz, q)
    assert np.allclose(fk.numpy(),
                       expect_full[:, y0:y0 + ny2, z0:z0 + nkz],
                       atol=1e-8)
    out = torch.empty(rank_shape, dtype=torch.complex128)
    fft.idft(fk, out)
    assert np.allclose(out.numpy(), full[sl] * np.prod(grid_shape),
                       atol=1e-6)


def test_pencil_fft_c2c_222():
    run_distributed(_pencil_c2c_worker, 8, args=((2, 2, 2), (8, 8, 8)))
