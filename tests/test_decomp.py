"""DomainDecomposition tests: single-rank periodic wrap, and
multi-process (gloo, world_size 2) halo exchange / collectives /
gather-scatter vs a periodified numpy oracle (style of reference
test/test_decomp.py:62-89)."""

import numpy as np
import torch

import pystella_amd as ps
from tests.conftest import run_distributed


def periodic_pad(a, h):
    return np.pad(a, h, mode="wrap")


def test_single_rank_wrap(grid_shape=(8, 10, 12)):
    h = 2
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    rng = np.random.default_rng(0)
    interior = rng.random(grid_shape)
    fx = torch.zeros(tuple(n + 2 * h for n in grid_shape),
                     dtype=torch.float64)
    fx[h:-h, h:-h, h:-h] = torch.as_tensor(interior)
    decomp.share_halos(fx)
    assert np.allclose(fx.numpy(), periodic_pad(interior, h))


def test_single_rank_wrap_outer_axes(grid_shape=(6, 6, 6)):
    h = 1
    decomp = ps.DomainDecomposition((1, 1, 1), h, rank_shape=grid_shape)
    rng = np.random.default_rng(1)
    interior = rng.random((2,) + grid_shape)
    fx = torch.zeros((2,) + tuple(n + 2 * h for n in grid_shape),
                     dtype=torch.float64)
    fx[:, h:-h, h:-h, h:-h] = torch.as_tensor(interior)
    decomp.share_halos(fx)
    for i in range(2):
        assert np.allclose(fx[i].numpy(), periodic_pad(interior[i], h))


def _halo_worker(rank, world_size, proc_shape, grid_shape, h):
    decomp = ps.DomainDecomposition(proc_shape, h, grid_shape=grid_shape)
    rank_shape, start = decomp.get_rank_shape_start(grid_shape)

    rng = np.random.default_rng(42)
    full = rng.random(grid_shape)
    padded_full = periodic_pad(full, h)

    fx = torch.zeros(tuple(n + 2 * h for n in rank_shape),
                     dtype=torch.float64)
    sl = tuple(slice(s, s + n) for s, n in zip(start, rank_shape))
    fx[h:-h, h:-h, h:-h] = torch.as_tensor(full[sl])
    decomp.share_halos(fx)

    expect = padded_full[tuple(slice(s, s + n + 2 * h)
                               for s, n in zip(start, rank_shape))]
    assert np.allclose(fx.numpy(), expect), \
        f"rank {rank} halo mismatch (proc_shape={proc_shape})"


def test_halo_exchange_x():
    run_distributed(_halo_worker, 2, args=((2, 1, 1), (8, 8, 8), 2))


def test_halo_exchange_y():
    run_distributed(_halo_worker, 2, args=((1, 2, 1), (8, 8, 8), 1))


def test_halo_exchange_z():
    # the reference raises NotImplementedError for z decomposition
    # (decomp.py:129-130); we support it
    run_distributed(_halo_worker, 2, args=((1, 1, 2), (8, 8, 8), 2))


def _collective_worker(rank, world_size):
    decomp = ps.DomainDecomposition((world_size, 1, 1), 0,
                                    rank_shape=(4, 4, 4))
    assert decomp.allreduce(1.0) == world_size
    vec = np.array([rank + 1.0, 2.0])
    out = decomp.allreduce(vec)
    assert np.allclose(out, [sum(r + 1.0 for r in range(world_size)),
                             2.0 * world_size])
    assert decomp.allreduce(float(rank), op="max") == world_size - 1
    assert decomp.bcast(rank if rank == 0 else None, root=0) == 0


def test_collectives():
    run_distributed(_collective_worker, 2)


def _gather_scatter_worker(rank, world_size, grid_shape):
    decomp = ps.DomainDecomposition((world_size, 1, 1), 0,
                                    grid_shape=grid_shape)
    rank_shape, start = decomp.get_rank_shape_start(grid_shape)
    local = torch.full(rank_shape, float(rank), dtype=torch.float64)
    full = decomp.gather_array(local, root=0)
    if rank == 0:
        for r in range(world_size):
            shp, st = decomp.get_rank_shape_start(
                grid_shape, (r, 0, 0))
            sl = tuple(slice(s, s + n) for s, n in zip(st, shp))
            assert torch.all(full[sl] == r)
    back = decomp.scatter_array(full if rank == 0 else local, root=0)
    assert torch.all(back == rank)


def test_gather_scatter():
    run_distributed(_gather_scatter_worker, 2, args=((9, 4, 4),))


def test_uneven_split():
    n, s = ps.decomp.get_size_start(10, 3, 0)
    assert (n, s) == (4, 0)
    n, s = ps.decomp.get_size_start(10, 3, 1)
    assert (n, s) == (3, 4)
    n, s = ps.decomp.get_size_start(10, 3, 2)
    assert (n, s) == (3, 7)


def _checkpoint_worker(rank, world_size, mode):
    import torch
    from pystella_amd.checkpoint import save_checkpoint, load_checkpoint
    import tempfile
    import os
    h = 1
    grid = (8, 8, 8)
    decomp = ps.DomainDecomposition((world_size, 1, 1), h, grid_shape=grid)
    rank_shape, _ = decomp.get_rank_shape_start(grid)
    pad = tuple(n + 2 * h for n in rank_shape)
    torch.manual_seed(rank)
    f = torch.rand((2,) + pad, dtype=torch.float64)
    decomp.share_halos(f)
    path = os.path.join(tempfile.gettempdir(),
                        f"ckpt_test_{mode}_{world_size}.pt")
    save_checkpoint(path, decomp, {"f": f}, attrs={"t": 1.5}, mode=mode)
    f2 = torch.zeros_like(f)
    attrs = load_checkpoint(path, decomp, {"f": f2})
    assert attrs["t"] == 1.5
    assert torch.allclose(f, f2), f"rank {rank} mismatch"


def test_checkpoint_gather():
    run_distributed(_checkpoint_worker, 2, args=("gather",))


def test_checkpoint_shard():
    run_distributed(_checkpoint_worker, 2, args=("shard",))


def _overlap_halo_worker(rank, world_size, proc_shape, grid_shape, h):
    """share_halos_start/finish fills all FACE halos correctly (star
    stencil contract; edge/corner halos are excluded by design)."""
    decomp = ps.DomainDecomposition(proc_shape, h, grid_shape=grid_shape)
    rank_shape, start = decomp.get_rank_shape_start(grid_shape)

    rng = np.random.default_rng(42)
    full = rng.random(grid_shape)
    padded_full = periodic_pad(full, h)

    fx = torch.zeros(tuple(n + 2 * h for n in rank_shape),
                     dtype=torch.float64)
    sl = tuple(slice(s, s + n) for s, n in zip(start, rank_shape))
    fx[h:-h, h:-h, h:-h] = torch.as_tensor(full[sl])
    handle = decomp.share_halos_start(fx)
    handle.finish()

    expect = padded_full[tuple(slice(s, s + n + 2 * h)
                               for s, n in zip(start, rank_shape))]
    nx, ny, nz = rank_shape
    got = fx.numpy()
    # check the 6 face-halo regions + the interior
    regions = [
        (slice(h, h + nx), slice(h, h + ny), slice(h, h + nz)),
        (slice(0, h), slice(h, h + ny), slice(h, h + nz)),
        (slice(h + nx, None), slice(h, h + ny), slice(h, h + nz)),
        (slice(h, h + nx), slice(0, h), slice(h, h + nz)),
        (slice(h, h + nx), slice(h + ny, None), slice(h, h + nz)),
        (slice(h, h + nx), slice(h, h + ny), slice(0, h)),
        (slice(h, h + nx), slice(h, h + ny), slice(h + nz, None)),
    ]
    for reg in regions:
        assert np.allclose(got[reg], expect[reg]), \
            f"rank {rank} face-halo mismatch {reg}"


def test_overlap_halo_exchange_single():
    _overlap_halo_worker(0, 1, (1, 1, 1), (8, 8, 8), 2)


def test_overlap_halo_exchange_x():
    run_distributed(_overlap_halo_worker, 2, args=((2, 1, 1), (8, 8, 8), 2))


def test_overlap_halo_exchange_z():
    run_distributed(_overlap_halo_worker, 2, args=((1, 1, 2), (8, 8, 8), 2))


def test_halo_exchange_2d_pencil():
    """4 ranks, (2,2,1) pencil — the reference CI's mpirun -np 4
    --proc_shape 2,2,1 configuration (ci.yml:97-99)."""
    run_distributed(_halo_worker, 4, args=((2, 2, 1), (8, 8, 8), 2))


def test_overlap_halo_exchange_222():
    """8 ranks, (2,2,2): ALL axes remote — one batched group carries
    12 concurrent ops per rank, exactly the pattern the N=8 bench's
    overlapped path posts over RCCL."""
    run_distributed(_overlap_halo_worker, 8, args=((2, 2, 2), (8, 8, 8), 2))


def test_overlap_halo_exchange_2d():
    run_distributed(_overlap_halo_worker, 4, args=((2, 2, 1), (8, 8, 8), 1))


def test_halo_exchange_3d():
    """8 ranks, full 3-D (2,2,2) decomposition (beyond the reference,
    which caps at 2-D)."""
    run_distributed(_halo_worker, 8, args=((2, 2, 2), (8, 8, 8), 1))
