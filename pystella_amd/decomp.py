"""Domain decomposition and communication over torch.distributed.

MI355X-native replacement for the reference's mpi4py-based
``DomainDecomposition`` (reference: pystella/decomp.py:32-725).  Design
differences, deliberate:

* One process per GPU; collectives and halo point-to-point go through
  ``torch.distributed`` — the ``"nccl"`` backend IS RCCL on ROCm, so halo
  faces move GPU-direct over xGMI with no host staging (the reference
  stages through the host: decomp.py:402-419).  On CPU (tests) the
  ``"gloo"`` backend runs the identical code path.
* Full 3-D decomposition is supported (the reference raises
  ``NotImplementedError`` for ``proc_shape[2] != 1``, decomp.py:129-130).
* Halo exchange packs *all* outer field components into one contiguous
  buffer per direction — fewer, larger RCCL messages, sized for xGMI
  (7 p2p links × ≈153 GB/s), instead of one message per component.
* Rank layout matches the reference: ``rank = rz + pz*(ry + py*rx)``
  (decomp.py:137-139), and uneven division follows mpi4py-fft pencil
  semantics (decomp.py:323-337).
"""

from __future__ import annotations

import logging
import numbers
import os

import numpy as np
import torch

__all__ = ["DomainDecomposition", "init_distributed"]

logger = logging.getLogger(__name__)

# Poor-man's distributed trace to localize hangs, mirroring the
# reference's debug-level collective tracing with serializing barriers
# (reference decomp.py:355-363): export PYSTELLA_DEBUG_COLLECTIVES=1
# (and set logging to DEBUG) to log + barrier around every collective.
_DEBUG_COLLECTIVES = bool(os.environ.get("PYSTELLA_DEBUG_COLLECTIVES"))


def _trace(decomp, what):
    if _DEBUG_COLLECTIVES:
        logger.debug("rank %d: entering %s", decomp.rank, what)
        if decomp.nranks > 1:
            _dist().barrier()
        logger.debug("rank %d: %s barrier passed", decomp.rank, what)


def _dist():
    import torch.distributed as dist
    return dist


def init_distributed(backend=None):
    """Initialize torch.distributed from torchrun-style env vars if a
    multi-process launch is detected and no process group exists yet.

    Returns True if a process group is (now) initialized.
    """
    import os
    dist = _dist()
    if dist.is_initialized():
        return True
    if "RANK" not in os.environ or "WORLD_SIZE" not in os.environ:
        return False
    if int(os.environ.get("WORLD_SIZE", "1")) <= 1 and backend is None:
        return False
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if backend == "nccl":
        torch.cuda.set_device(int(os.environ.get("LOCAL_RANK", "0")))
    dist.init_process_group(backend=backend)
    return True


def get_size_start(N, size, rank):
    """Uneven-division pencil split, matching mpi4py-fft / reference
    pystella/decomp.py:323-329."""
    q, r = divmod(N, size)
    n = q + (1 if r > rank else 0)
    start = rank * q + min(rank, r)
    return n, start


class _HaloHandle:
    """Pending overlapped halo exchange (see share_halos_start)."""

    def __init__(self, works, fills, release=None):
        self.works = works
        self.fills = fills
        self._release = release
        self._done = False

    def finish(self):
        if self._done:
            return
        for w in self.works:
            w.wait()
        for dst, src in self.fills:
            dst.copy_(src)
        if self._release is not None:
            self._release()
        self._done = True


class DomainDecomposition:
    """Pencil/slab/3-D domain decomposition with halo exchange.

    :arg proc_shape: 3-tuple processor grid (px, py, pz).
    :arg halo_shape: int or 3-tuple of halo layers per axis.
    :arg rank_shape: this rank's interior grid shape (optional).
    :arg grid_shape: the global grid shape (optional alternative).
    """

    def __init__(self, proc_shape=(1, 1, 1), halo_shape=0, rank_shape=None,
                 grid_shape=None):
        self.proc_shape = tuple(proc_shape)
        self.halo_shape = ((halo_shape,) * 3 if isinstance(halo_shape, int)
                           else tuple(halo_shape))

        dist = _dist()
        px, py, pz = self.proc_shape
        if px * py * pz == 1:
            # a (1,1,1) decomposition is always LOCAL — self-contained
            # even inside a multi-rank program (single-rank oracles in
            # distributed tests, per-rank side computations, coarse
            # gather-to-one-rank solves)
            self.rank = 0
            self.nranks = 1
        elif dist.is_initialized():
            self.rank = dist.get_rank()
            self.nranks = dist.get_world_size()
        else:
            self.rank = 0
            self.nranks = 1

        if px * py * pz != self.nranks:
            raise ValueError(
                f"{proc_shape} is an invalid decomposition for "
                f"{self.nranks} ranks")

        # rank = rz + pz*(ry + py*rx)  (reference decomp.py:137-139)
        self.rz = self.rank % pz
        self.ry = (self.rank // pz) % py
        self.rx = self.rank // (pz * py)

        if grid_shape is not None:
            if rank_shape is not None:
                raise ValueError("pass only one of rank_shape or grid_shape")
            rank_shape, _ = self.get_rank_shape_start(grid_shape)
        self.rank_shape = tuple(rank_shape) if rank_shape is not None else None
        self.grid_shape = tuple(grid_shape) if grid_shape is not None else None

        # persistent communication buffer pool, keyed by
        # (shape, dtype, device): avoids a contiguous()+empty_like
        # allocator round trip per face per field per RK stage
        # (the reference caches its comm buffers too, decomp.py:339-349)
        self._buf_pool = {}

    def _acquire_bufs(self, shape, dtype, device, n=4):
        """Check out ``n`` persistent contiguous buffers of the given
        face shape (returned to the pool via :meth:`_release_bufs`)."""
        key = (tuple(shape), dtype, str(device))
        pool = self._buf_pool.setdefault(key, [])
        out = []
        for _ in range(n):
            if pool:
                out.append(pool.pop())
            else:
                out.append(torch.empty(shape, dtype=dtype, device=device))
        return out

    def _release_bufs(self, shape, dtype, device, bufs):
        key = (tuple(shape), dtype, str(device))
        self._buf_pool.setdefault(key, []).extend(bufs)

    # -- topology -----------------------------------------------------------

    @property
    def rank_tuple(self):
        return (self.rx, self.ry, self.rz)

    def rankID(self, rx, ry, rz):
        px, py, pz = self.proc_shape
        return (rz % pz) + pz * ((ry % py) + py * (rx % px))

    def get_rank_shape_start(self, grid_shape, rank_tuple=None):
        rank_tuple = rank_tuple or self.rank_tuple
        shape, start = [], []
        for N, size, r in zip(grid_shape, self.proc_shape, rank_tuple):
            n, s = get_size_start(N, size, r)
            shape.append(n)
            start.append(s)
        return tuple(shape), tuple(start)

    # -- halo exchange ------------------------------------------------------

    def _wrap_axis(self, fx, axis, h):
        """Periodic wrap for a non-decomposed axis (in-place)."""
        n = fx.shape[axis] - 2 * h
        lo = [slice(None)] * fx.dim()
        hi = [slice(None)] * fx.dim()
        src_lo = [slice(None)] * fx.dim()
        src_hi = [slice(None)] * fx.dim()
        lo[axis] = slice(0, h)
        src_lo[axis] = slice(n, n + h)
        hi[axis] = slice(n + h, n + 2 * h)
        src_hi[axis] = slice(h, 2 * h)
        fx[tuple(lo)] = fx[tuple(src_lo)]
        fx[tuple(hi)] = fx[tuple(src_hi)]

    def _exchange_axis(self, fx, axis, h, neighbors):
        """Exchange h-deep faces along `axis` with the two neighbor ranks.

        Faces span the full (padded) extents of the other axes, so doing
        axes sequentially propagates edge/corner halos correctly.
        """
        dist = _dist()
        dim = fx.dim()
        n = fx.shape[axis] - 2 * h

        def face(lo, extent):
            sl = [slice(None)] * dim
            sl[axis] = slice(lo, lo + extent)
            return fx[tuple(sl)]

        fshape = tuple(face(h, h).shape)
        send_lo, send_hi, recv_lo, recv_hi = self._acquire_bufs(
            fshape, fx.dtype, fx.device)
        send_lo.copy_(face(h, h))                  # my low interior face
        send_hi.copy_(face(n, h))                  # my high interior face
        lo_rank, hi_rank = neighbors

        # Pairing convention (also correct when lo_rank == hi_rank, e.g.
        # two ranks with periodic wrap): phase A moves low faces downward
        # (my high halo ← hi_rank's low face), phase B moves high faces
        # upward.  Every rank posts A-ops before B-ops so same-peer
        # send/recv pairs match by order.
        ops = [
            dist.P2POp(dist.irecv, recv_hi, hi_rank),
            dist.P2POp(dist.isend, send_lo, lo_rank),
            dist.P2POp(dist.irecv, recv_lo, lo_rank),
            dist.P2POp(dist.isend, send_hi, hi_rank),
        ]
        for work in dist.batch_isend_irecv(ops):
            work.wait()

        face(0, h).copy_(recv_lo)
        face(n + h, h).copy_(recv_hi)
        self._release_bufs(fshape, fx.dtype, fx.device,
                           (send_lo, send_hi, recv_lo, recv_hi))

    def share_halos(self, fx):
        """Impose periodic boundary conditions on the halo padding of
        ``fx`` (shape ``outer + (nx+2hx, ny+2hy, nz+2hz)``), exchanging
        faces with neighbor ranks along decomposed axes.
        """
        _trace(self, "share_halos")
        hx, hy, hz = self.halo_shape
        px, py, pz = self.proc_shape
        dim = fx.dim()
        for ax_rel, (h, p) in enumerate(zip((hx, hy, hz), (px, py, pz))):
            if h == 0:
                continue
            axis = dim - 3 + ax_rel
            if p == 1:
                self._wrap_axis(fx, axis, h)
            else:
                delta = [0, 0, 0]
                delta[ax_rel] = 1
                lo_rank = self.rankID(self.rx - delta[0], self.ry - delta[1],
                                      self.rz - delta[2])
                hi_rank = self.rankID(self.rx + delta[0], self.ry + delta[1],
                                      self.rz + delta[2])
                self._exchange_axis(fx, axis, h, (lo_rank, hi_rank))

    def share_halos_start(self, fx, skip_wrap=False, wrap_axes=None):
        """Overlap-friendly halo exchange for STAR stencils: wraps
        single-rank axes in place immediately (stream-ordered) and posts
        ALL remote-axis face exchanges in one batched non-blocking
        group, so interior compute can run while xGMI transfers are in
        flight.  Unlike :meth:`share_halos`, edge/corner halo values are
        NOT propagated (the faces are sent concurrently) — valid only
        for axis-aligned (star) stencil reads, e.g. the Laplacian hot
        loop.  Returns a handle; call ``handle.finish()`` before any
        kernel that reads the halos.

        :arg wrap_axes: explicit subset of single-rank axes to wrap
            (axes whose periodicity the consuming kernel handles
            in-register pass a reduced set); default = all single-rank
            axes, or none with ``skip_wrap``.
        """
        dist = _dist()
        hx, hy, hz = self.halo_shape
        px, py, pz = self.proc_shape
        dim = fx.dim()
        ops = []
        fills = []
        checked_out = []
        if wrap_axes is None:
            wrap_axes = [] if skip_wrap else [
                ax for ax, (h, p) in enumerate(
                    zip((hx, hy, hz), (px, py, pz))) if h > 0 and p == 1]
        else:
            skip_wrap = True     # caller controls non-wrapped axes
        wrapped_fused = skip_wrap
        if wrap_axes and isinstance(fx, torch.Tensor) and fx.is_cuda:
            from pystella_amd.backend.hip import wrap_star
            wrap_star(fx, self.halo_shape, wrap_axes)
            wrapped_fused = True
        elif wrap_axes and skip_wrap:
            # explicit subset on the CPU path
            for ax in wrap_axes:
                self._wrap_axis(fx, dim - 3 + ax,
                                self.halo_shape[ax])
            wrapped_fused = True
        for ax_rel, (h, p) in enumerate(zip((hx, hy, hz), (px, py, pz))):
            if h == 0:
                continue
            axis = dim - 3 + ax_rel
            if p == 1:
                if not wrapped_fused:
                    self._wrap_axis(fx, axis, h)
                continue
            n = fx.shape[axis] - 2 * h

            def face(lo, extent, axis=axis):
                sl = [slice(None)] * dim
                sl[axis] = slice(lo, lo + extent)
                return fx[tuple(sl)]

            fshape = tuple(face(h, h).shape)
            send_lo, send_hi, recv_lo, recv_hi = self._acquire_bufs(
                fshape, fx.dtype, fx.device)
            send_lo.copy_(face(h, h))
            send_hi.copy_(face(n, h))
            checked_out.append((fshape, fx.dtype, fx.device,
                                (send_lo, send_hi, recv_lo, recv_hi)))
            delta = [0, 0, 0]
            delta[ax_rel] = 1
            lo_rank = self.rankID(self.rx - delta[0], self.ry - delta[1],
                                  self.rz - delta[2])
            hi_rank = self.rankID(self.rx + delta[0], self.ry + delta[1],
                                  self.rz + delta[2])
            ops += [
                dist.P2POp(dist.irecv, recv_hi, hi_rank),
                dist.P2POp(dist.isend, send_lo, lo_rank),
                dist.P2POp(dist.irecv, recv_lo, lo_rank),
                dist.P2POp(dist.isend, send_hi, hi_rank),
            ]
            fills.append((face(0, h), recv_lo))
            fills.append((face(n + h, h), recv_hi))

        works = dist.batch_isend_irecv(ops) if ops else []

        def release():
            for shape, dtype, device, bufs in checked_out:
                self._release_bufs(shape, dtype, device, bufs)

        return _HaloHandle(works, fills, release=release)

    # -- collectives --------------------------------------------------------

    def _comm_backend_is_nccl(self):
        dist = _dist()
        try:
            return "nccl" in str(dist.get_backend())
        except (RuntimeError, ValueError):
            return False

    def allreduce(self, rank_reduction, op="sum"):
        """All-reduce a scalar, numpy array, or torch tensor.

        RCCL (the ``"nccl"`` backend) only reduces device tensors, so
        host-side scalars/arrays are staged through the GPU when that
        is the communicator backend.
        """
        _trace(self, f"allreduce({op})")
        dist = _dist()
        if self.nranks == 1:
            return rank_reduction

        red_op = {"sum": dist.ReduceOp.SUM, "max": dist.ReduceOp.MAX,
                  "min": dist.ReduceOp.MIN, "prod": dist.ReduceOp.PRODUCT,
                  }[op]

        def _reduce(t):
            if not t.is_cuda and self._comm_backend_is_nccl():
                d = t.cuda()
                dist.all_reduce(d, op=red_op)
                return d.cpu()
            dist.all_reduce(t, op=red_op)
            return t

        if isinstance(rank_reduction, torch.Tensor):
            return _reduce(rank_reduction.clone())
        if isinstance(rank_reduction, np.ndarray):
            return _reduce(torch.from_numpy(rank_reduction.copy())).numpy()
        if isinstance(rank_reduction, numbers.Number):
            t = torch.tensor([rank_reduction], dtype=torch.float64)
            return _reduce(t).item()
        raise TypeError(f"cannot allreduce {type(rank_reduction)}")

    def bcast(self, x, root=0):
        dist = _dist()
        if self.nranks == 1:
            return x
        obj = [x]
        dist.broadcast_object_list(obj, src=root)
        return obj[0]

    def barrier(self):
        if self.nranks > 1:
            _dist().barrier()

    Barrier = barrier

    # -- gather / scatter (I/O path) ---------------------------------------

    def gather_array(self, arr, root=0, grid_shape=None):
        """Gather rank-local interior pencils into a global array on
        ``root`` (returns None elsewhere).  ``arr`` is the *unpadded*
        rank-local array (outer axes allowed, grid axes last).
        """
        dist = _dist()
        grid_shape = grid_shape or self.grid_shape
        if self.nranks == 1:
            return arr.clone() if isinstance(arr, torch.Tensor) else arr.copy()
        if grid_shape is None:
            raise ValueError("grid_shape needed for gather_array")
        outer = tuple(arr.shape[:-3])
        if self.rank == root:
            full = torch.empty(outer + tuple(grid_shape), dtype=arr.dtype,
                               device=arr.device)
            for r in range(self.nranks):
                rz = r % self.proc_shape[2]
                ry = (r // self.proc_shape[2]) % self.proc_shape[1]
                rx = r // (self.proc_shape[2] * self.proc_shape[1])
                shape, start = self.get_rank_shape_start(
                    grid_shape, (rx, ry, rz))
                sl = (Ellipsis,) + tuple(
                    slice(s, s + n) for s, n in zip(start, shape))
                if r == root:
                    full[sl] = arr
                else:
                    buf = torch.empty(outer + shape, dtype=arr.dtype,
                                      device=arr.device)
                    dist.recv(buf, src=r)
                    full[sl] = buf
            return full
        else:
            dist.send(arr.contiguous(), dst=root)
            return None

    def scatter_array(self, full, root=0, grid_shape=None):
        """Scatter a global array on ``root`` into rank-local interior
        pencils (inverse of :meth:`gather_array`)."""
        dist = _dist()
        grid_shape = grid_shape or self.grid_shape
        if self.nranks == 1:
            return full.clone()
        if grid_shape is None:
            raise ValueError("grid_shape needed for scatter_array")
        shape, start = self.get_rank_shape_start(grid_shape)
        if self.rank == root:
            outer = tuple(full.shape[:-3])
            out = None
            for r in range(self.nranks):
                rz = r % self.proc_shape[2]
                ry = (r // self.proc_shape[2]) % self.proc_shape[1]
                rx = r // (self.proc_shape[2] * self.proc_shape[1])
                shp, st = self.get_rank_shape_start(grid_shape, (rx, ry, rz))
                sl = (Ellipsis,) + tuple(
                    slice(s, s + n) for s, n in zip(st, shp))
                piece = full[sl].contiguous()
                if r == root:
                    out = piece
                else:
                    dist.send(piece, dst=r)
            return out
        else:
            # non-root: outer shape must be communicated implicitly; the
            # caller passes a same-outer-shape dummy or None full
            outer = tuple(full.shape[:-3]) if full is not None else ()
            buf = torch.empty(outer + shape,
                              dtype=full.dtype if full is not None
                              else torch.float64,
                              device=full.device if full is not None
                              else "cpu")
            dist.recv(buf, src=root)
            return buf

    # -- padding helpers ----------------------------------------------------

    def remove_halos(self, in_array, out_array=None):
        h = self.halo_shape
        sl = (Ellipsis,) + tuple(
            slice(hi, in_array.shape[d - 3] - hi) if hi else slice(None)
            for d, hi in enumerate(h))
        interior = in_array[(Ellipsis,) + tuple(
            slice(hi, in_array.shape[in_array.dim() - 3 + d] - hi)
            for d, hi in enumerate(h))]
        if out_array is None:
            return interior.contiguous()
        out_array.copy_(interior)
        return out_array

    def restore_halos(self, out_array, in_array):
        h = self.halo_shape
        interior = out_array[(Ellipsis,) + tuple(
            slice(hi, out_array.shape[out_array.dim() - 3 + d] - hi)
            for d, hi in enumerate(h))]
        interior.copy_(in_array)
        return out_array
