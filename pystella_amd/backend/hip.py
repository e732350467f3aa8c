"""GPU execution: AOT CDNA4 kernels + hiprtc-JIT'd kernel templates.

This module is the only place that touches the native extension
(``pystella_amd._C``).  Policy: on a GPU box the HIP path is the one
that runs — if the extension is missing or fails to load, GPU calls
raise immediately (no silent torch fallback; see repo instructions on
native-code loading).
"""

from __future__ import annotations

import math
import os

import numpy as np
import torch

from pystella_amd.backend.codegen import (
    Codegen, PREAMBLE, geometry_defines,
)

_EXT = None


def ext():
    global _EXT
    if _EXT is None:
        from pystella_amd.backend import build as _build
        if _build.needs_rebuild():
            # sources newer than the shipped .so (e.g. edited after the
            # last build) — rebuild in place rather than run stale code
            _build.build_extension()
        try:
            from pystella_amd import _C
        except ImportError as e:
            raise ImportError(
                "pystella_amd._C native extension not built; run "
                "`python -m pystella_amd.backend.build` (hipcc, gfx950). "
                f"Underlying error: {e}") from e
        cache = os.environ.get(
            "PYSTELLA_JIT_CACHE",
            os.path.join(os.path.dirname(os.path.dirname(
                os.path.abspath(__file__))), ".hiprtc_cache"))
        os.makedirs(cache, exist_ok=True)
        _C.set_cache_dir(cache)
        _EXT = _C
    return _EXT


def _stream():
    return torch.cuda.current_stream().cuda_stream


def _check_tensor(name, t):
    if not (isinstance(t, torch.Tensor) and t.is_cuda):
        raise TypeError(f"argument {name} must be a CUDA tensor, got "
                        f"{type(t)}")
    if not t.is_contiguous():
        raise ValueError(f"argument {name} must be contiguous")
    return t


def _resolve_scalar(env, key):
    if isinstance(key, tuple):
        name, idx = key
        v = env[name]
        if isinstance(v, torch.Tensor):
            return float(v.reshape(-1)[np.ravel_multi_index(
                idx, v.shape)] if v.numel() > 1 else v.item())
        v = np.asarray(v)
        if v.ndim == 0 or v.size == 1:
            return float(v.reshape(-1)[0])
        return float(v[idx])
    v = env[key]
    if isinstance(v, torch.Tensor):
        return float(v.item())
    return float(np.asarray(v).reshape(-1)[0])


# ---------------------------------------------------------------------------
# JIT'd elementwise map (fused RK stage kernels etc.)

ELEMENTWISE_TEMPLATE = """{defines}
{preamble}
extern "C" __global__ __launch_bounds__(TBZ * TBY) void {name}(
    {params})
{{
    const int k = blockIdx.x * TBZ + (threadIdx.x % TBZ);
    const int j = blockIdx.y * TBY + (threadIdx.x / TBZ);
    if (k >= NZ || j >= NY) return;
    const int i0 = blockIdx.z * XCHUNK;
    const int i1 = (i0 + XCHUNK < NX) ? i0 + XCHUNK : NX;
    for (int i = i0; i < i1; ++i) {{
        {body}
    }}
}}
"""


def _tile_defines(tile, rank_shape):
    tbz, tby, xchunk = tile
    return (f"#define TBZ {tbz}\n#define TBY {tby}\n"
            f"#define XCHUNK {xchunk}\n")


def _tile_grid(tile, rank_shape):
    tbz, tby, xchunk = tile
    nx, ny, nz = rank_shape
    return ((nz + tbz - 1) // tbz, (ny + tby - 1) // tby,
            (nx + xchunk - 1) // xchunk)


_RTYPE = {torch.float64: "double", torch.float32: "float"}


class JitElementwise:
    """Compiled fused per-site map over the interior grid.  Supports
    fp64 and fp32 arrays (``using real = double|float`` baked into the
    kernel; run-time scalars stay double and are cast on use)."""

    def __init__(self, map_dict, tmp_instructions, field_args, scalar_names,
                 halo, rank_shape, name="ew_map", tile=(64, 4, 64),
                 dtype=torch.float64):
        self.rank_shape = tuple(rank_shape)
        self.tile = tile
        self.dtype = dtype
        self.field_args = [fa for fa in field_args if fa.spatial]
        cg = Codegen(field_args, halo, rank_shape)
        body = cg.emit_statements(map_dict, tmp_instructions)
        ptr_params = ", ".join(
            f"real* __restrict__ {fa.name}" for fa in self.field_args)
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (ptr_params, dbl_params) if x)
        src = ELEMENTWISE_TEMPLATE.format(
            defines=geometry_defines(halo, rank_shape,
                                     rtype=_RTYPE[dtype])
            + _tile_defines(tile, rank_shape),
            preamble=PREAMBLE, name=name, params=params, body=body)
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)
        self.grid = _tile_grid(tile, rank_shape)
        self.block = tile[0] * tile[1]

    def __call__(self, env):
        ptrs = []
        for fa in self.field_args:
            t = _check_tensor(fa.name, env[fa.name])
            if t.dtype != self.dtype:
                raise TypeError(
                    f"argument {fa.name} has dtype {t.dtype}, kernel "
                    f"compiled for {self.dtype}")
            ptrs.append(t.data_ptr())
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, self.grid[0], self.grid[1],
                         self.grid[2], self.block, 1, 1, 0, _stream(),
                         ptrs, [], doubles)


def get_elementwise_kernel(map_dict, tmp_instructions, field_args,
                           scalar_names, halo, rank_shape, name="ew_map",
                           tile=(64, 4, 64), dtype=torch.float64):
    return JitElementwise(map_dict, tmp_instructions, field_args,
                          scalar_names, halo, rank_shape, name=name,
                          tile=tile, dtype=dtype)


# ---------------------------------------------------------------------------
# Generic LDS-staged stencil kernel: x-marching blocks stage each
# neighbor-read field's current x-plane tile (with the ±H ghost rim)
# into LDS and serve pure-x shifts from a per-thread register ring —
# the generic-expression analogue of csrc/derivs.hip's
# gradlap_lds_knl (measured 17 % faster than plain L1 reuse for the
# isolated Laplacian, profiles/r01_lds_vs_ring.txt).  This is the
# CDNA4 realization of the reference's ``Stencil``/``StreamingStencil``
# workgroup-prefetch kernels (reference stencil.py:36-141).

STENCIL_TEMPLATE = """{defines}
{preamble}
extern "C" __global__ __launch_bounds__(SBZ * SBY) void {name}(
    {params})
{{
{lds_decls}
    const int lz = (int)(threadIdx.x % SBZ);
    const int ly = (int)(threadIdx.x / SBZ);
    const int k = blockIdx.x * SBZ + lz;
    const int j = blockIdx.y * SBY + ly;
    const int i0 = blockIdx.z * SXCHUNK;
    const int i1 = (i0 + SXCHUNK < NX) ? i0 + SXCHUNK : NX;
    const bool active = (k < NZ) && (j < NY);
    const int jc = j < NY ? j : NY - 1;
    const int kc = k < NZ ? k : NZ - 1;
    (void)jc; (void)kc;
{ring_init}
    for (int i = i0; i < i1; ++i) {{
{ring_load}
        __syncthreads();
        for (int t = (int)threadIdx.x; t < (SBY + 2*H) * (SBZ + 2*H);
             t += SBZ * SBY) {{
            const int tz = t % (SBZ + 2*H);
            const int ty = t / (SBZ + 2*H);
            int gj = blockIdx.y * SBY + ty;
            int gk = blockIdx.x * SBZ + tz;
            if (gj > NY + 2*H - 1) gj = NY + 2*H - 1;
            if (gk > NZ + 2*H - 1) gk = NZ + 2*H - 1;
            const long goff = (long)(i + H) * PSY * PSZ
                              + (long)gj * PSZ + gk;
{stage}
        }}
        __syncthreads();
        if (active) {{
            {body}
        }}
{ring_shift}
    }}
}}
"""


class _StencilCodegen(Codegen):
    """Codegen redirecting reads of prefetched components: (sy,sz)
    shifts -> LDS tile, pure-x shifts -> register ring."""

    def __init__(self, field_args, halo, rank_shape, tiles, rings):
        super().__init__(field_args, halo, rank_shape)
        self.tiles = tiles      # set of (name, lin)
        self.rings = rings      # set of (name, lin)
        self.store_ctx = False

    def field_access(self, f, outer_idx):
        if (not self.store_ctx and f.is_spatial and f.is_padded
                and len(outer_idx) <= 1):
            lin = int(outer_idx[0]) if outer_idx else 0
            sx, sy, sz = f.shift
            if sx == 0 and (f.name, lin) in self.tiles:
                return (f"t_{f.name}_{lin}[(ly + H + ({sy}))"
                        f" * (SBZ + 2*H) + (lz + H + ({sz}))]")
            if sy == 0 and sz == 0 and (f.name, lin) in self.rings:
                return f"r_{f.name}_{lin}[H + ({sx})]"
        return super().field_access(f, outer_idx)


class JitStencil:
    """Compiled LDS-staged stencil map (see STENCIL_TEMPLATE)."""

    SBZ, SBY = 32, 8

    def __init__(self, map_dict, tmp_instructions, field_args,
                 scalar_names, halo, rank_shape, name="stencil_map",
                 xchunk=32, dtype=torch.float64):
        from pystella_amd.field import (
            Field, Subscript, iter_exprs, walk_expr)
        self.rank_shape = tuple(rank_shape)
        self.dtype = dtype
        self.field_args = [fa for fa in field_args if fa.spatial]
        h = max(halo) if isinstance(halo, (tuple, list)) else halo

        # classify padded-field reads by shift pattern
        padded = {fa.name for fa in self.field_args if fa.padded}
        read_yz = set()      # (name, lin) with a (sy|sz)!=0 read
        read_x = set()       # (name, lin) with a pure-x != 0 read
        stores = set()       # components written (never prefetch)

        def scan(x, into_yz=read_yz, into_x=read_x):
            f = None
            lin = 0
            if isinstance(x, Subscript) and isinstance(x.aggregate, Field):
                f = x.aggregate
                if len(x.index) == 1 and isinstance(x.index[0], int):
                    lin = int(x.index[0])
                elif x.index:
                    return
            elif isinstance(x, Field):
                f = x
            if f is None or f.name not in padded or not f.is_spatial:
                return
            sx, sy, sz = f.shift
            if sy or sz:
                into_yz.add((f.name, lin))
            elif sx:
                into_x.add((f.name, lin))

        exprs = list((tmp_instructions or {}).values()) \
            + list(map_dict.values())
        for e in iter_exprs(exprs):
            walk_expr(e, scan)
        for lhs in map_dict:
            f = lhs.aggregate if isinstance(lhs, Subscript) else lhs
            lin = (int(lhs.index[0]) if isinstance(lhs, Subscript)
                   and lhs.index and isinstance(lhs.index[0], int) else 0)
            if isinstance(f, Field):
                stores.add((f.name, lin))
        # a stored component cannot be served from stale LDS/ring
        tiles = read_yz - stores
        rings = read_x - stores

        sbz, sby = self.SBZ, self.SBY
        tile_doubles = (sby + 2 * h) * (sbz + 2 * h)
        esize = 8 if dtype == torch.float64 else 4
        self.lds_bytes = len(tiles) * tile_doubles * esize
        if self.lds_bytes > 48 * 1024:
            raise ValueError("stencil LDS tiles exceed budget")

        cg = _StencilCodegen(field_args, halo, rank_shape, tiles, rings)

        lds_decls, stage, ring_init, ring_load, ring_shift = \
            [], [], [], [], []
        for nm, lin in sorted(tiles):
            lds_decls.append(
                f"    __shared__ real t_{nm}_{lin}"
                f"[(SBY + 2*H) * (SBZ + 2*H)];")
            base = f"{nm} + {lin}L * PVOL" if lin else nm
            stage.append(
                f"            t_{nm}_{lin}[t] = ({base})[goff];")
        for nm, lin in sorted(rings):
            base = f"{nm} + {lin}L * PVOL" if lin else nm
            ring_init.append(
                f"    real r_{nm}_{lin}[2*H + 1];\n"
                f"    for (int p = 0; p < 2*H; ++p)\n"
                f"        r_{nm}_{lin}[p] = ({base})["
                f"(long)(i0 + p) * PSY * PSZ"
                f" + (long)(jc + H) * PSZ + (kc + H)];")
            ring_load.append(
                f"        r_{nm}_{lin}[2*H] = ({base})["
                f"(long)(i + 2*H) * PSY * PSZ"
                f" + (long)(jc + H) * PSZ + (kc + H)];")
            ring_shift.append(
                f"        for (int p = 0; p < 2*H; ++p)\n"
                f"            r_{nm}_{lin}[p] = r_{nm}_{lin}[p + 1];")

        body = cg.emit_statements(map_dict, tmp_instructions)
        ptr_params = ", ".join(
            f"real* __restrict__ {fa.name}" for fa in self.field_args)
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (ptr_params, dbl_params) if x)
        defines = geometry_defines(halo, rank_shape, rtype=_RTYPE[dtype])
        defines += (f"#define SBZ {sbz}\n#define SBY {sby}\n"
                    f"#define SXCHUNK {xchunk}\n")
        src = STENCIL_TEMPLATE.format(
            defines=defines, preamble=PREAMBLE, name=name, params=params,
            lds_decls="\n".join(lds_decls),
            stage="\n".join(stage),
            ring_init="\n".join(ring_init),
            ring_load="\n".join(ring_load),
            ring_shift="\n".join(ring_shift),
            body=body)
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)
        nx, ny, nz = rank_shape
        self.grid = ((nz + sbz - 1) // sbz, (ny + sby - 1) // sby,
                     (nx + xchunk - 1) // xchunk)
        self.block = sbz * sby

    def __call__(self, env):
        ptrs = []
        for fa in self.field_args:
            t = _check_tensor(fa.name, env[fa.name])
            if t.dtype != self.dtype:
                raise TypeError(
                    f"argument {fa.name} has dtype {t.dtype}, kernel "
                    f"compiled for {self.dtype}")
            ptrs.append(t.data_ptr())
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, self.grid[0], self.grid[1],
                         self.grid[2], self.block, 1, 1, 0, _stream(),
                         ptrs, [], doubles)


def get_stencil_kernel(map_dict, tmp_instructions, field_args,
                       scalar_names, halo, rank_shape,
                       name="stencil_map", dtype=torch.float64):
    """LDS-staged stencil kernel, falling back to the plain elementwise
    form when no prefetchable reads exist or LDS would overflow."""
    try:
        k = JitStencil(map_dict, tmp_instructions, field_args,
                       scalar_names, halo, rank_shape, name=name,
                       dtype=dtype)
        if k.lds_bytes > 0:
            return k
    except ValueError:
        pass
    return JitElementwise(map_dict, tmp_instructions, field_args,
                          scalar_names, halo, rank_shape, name=name,
                          dtype=dtype)


# ---------------------------------------------------------------------------
# JIT'd fused elementwise map + simultaneous reductions over the INPUT
# state (MI355X traffic optimization: the RK stage kernel reads f, dfdt
# and computes lap f inline anyway — accumulating the energy reducers in
# the same pass removes the standalone lap+reduction kernel from the hot
# loop entirely; see fusion.StencilRKStepper).

STAGERED_TEMPLATE = """{defines}
{preamble}
#define NRED {nred}
extern "C" __global__ __launch_bounds__(TBZ * TBY) void {name}(
    {params})
{{
    double acc[NRED];
    {init}
    const int k = blockIdx.x * TBZ + (threadIdx.x % TBZ);
    const int j = blockIdx.y * TBY + (threadIdx.x / TBZ);
    const int i0 = blockIdx.z * XCHUNK;
    const int i1 = (i0 + XCHUNK < NX) ? i0 + XCHUNK : NX;
    if (k < NZ && j < NY) {{
        for (int i = i0; i < i1; ++i) {{
            {body}
        }}
    }}
"""


class JitStageReduction:
    """Fused per-site map + multi-quantity reduction of the pre-update
    state.  Emission order per site: temporaries (which include the
    inline Laplacian), reduction accumulation (reads input values), then
    the update stores — so the reducers see the input state even for
    in-place unknowns."""

    def __init__(self, map_dict, tmp_instructions, entries, field_args,
                 scalar_names, halo, rank_shape, name="rk_stage_red",
                 tile=(64, 4, 64)):
        self.rank_shape = tuple(rank_shape)
        self.tile = tile
        self.dtype = torch.float64      # fp64-only kernel (double* params)
        self.entries = entries
        self.field_args = [fa for fa in field_args if fa.spatial]
        nred = len(entries)
        cg = Codegen(field_args, halo, rank_shape)

        lines = []
        for lhs, rhs in (tmp_instructions or {}).items():
            tname = lhs.name if hasattr(lhs, "name") else str(lhs)
            cg.tmp_names.add(tname)
            lines.append(f"const double {tname} = {cg.emit(rhs)};")
        init_lines, combine_cases = [], []
        for r, (expr, op) in enumerate(entries):
            init_lines.append(f"acc[{r}] = {_OP_INIT[op]};")
            comb = _OP_COMBINE[op]
            val = cg.emit(expr)
            lines.append(
                "{ const double a = acc[%d]; const double b = %s; "
                "acc[%d] = %s; }" % (r, val, r, comb))
            combine_cases.append(f"(r == {r}) ? {comb} : ")
        for lhs, rhs in map_dict.items():
            lines.append(f"{cg.emit(lhs)} = {cg.emit(rhs)};")
        combine = "".join(combine_cases) + "0.0"

        ptr_params = ", ".join(
            f"double* __restrict__ {fa.name}" for fa in self.field_args)
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (
            ptr_params, "double* __restrict__ partials", dbl_params) if x)

        defines = geometry_defines(halo, rank_shape)
        defines += _tile_defines(tile, rank_shape)
        defines += "#define COMBINE(r, a, b) (" + combine + ")\n"
        src = (STAGERED_TEMPLATE + REDUCTION_TAIL).format(
            defines=defines, preamble=PREAMBLE, nred=nred, name=name,
            params=params, init="\n    ".join(init_lines),
            body="\n            ".join(lines))
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)
        self.grid = _tile_grid(tile, rank_shape)
        self.block = tile[0] * tile[1]
        self.nblk = self.grid[0] * self.grid[1] * self.grid[2]
        self._partials = None

    def __call__(self, env):
        dev = None
        ptrs = []
        want = getattr(self, "dtype", None)
        for fa in self.field_args:
            t = _check_tensor(fa.name, env[fa.name])
            if want is None:
                want = t.dtype
            elif t.dtype != want:
                raise TypeError(
                    f"reduction argument '{fa.name}' has dtype "
                    f"{t.dtype}, kernel expects {want}")
            dev = t.device
            ptrs.append(t.data_ptr())
        nred = len(self.entries)
        if (self._partials is None
                or self._partials.device != dev
                or self._partials.shape[1] != self.nblk):
            self._partials = torch.empty((nred, self.nblk),
                                         dtype=torch.float64, device=dev)
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, self.grid[0], self.grid[1],
                         self.grid[2], self.block, 1, 1, 0, _stream(),
                         ptrs + [self._partials.data_ptr()], [], doubles)
        return self._finish(dev)


def get_stage_reduction_kernel(map_dict, tmp_instructions, entries,
                               field_args, scalar_names, halo, rank_shape,
                               name="rk_stage_red", tile=(64, 4, 64)):
    return JitStageReduction(map_dict, tmp_instructions, entries,
                             field_args, scalar_names, halo, rank_shape,
                             name=name, tile=tile)


# ---------------------------------------------------------------------------
# Fused periodic-wrap kernel: all single-rank axes' halo faces in ONE
# launch (the torch slicing path costs ~6 kernel launches per field per
# stage).  Axes are wrapped concurrently, so edge/corner halos are NOT
# propagated — star-stencil contract, same as share_halos_start.

WRAP_TEMPLATE = """{defines}
extern "C" __global__ __launch_bounds__(256) void {name}(
    real* __restrict__ f)
{{
    const long tid = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
#if WRAPX
    for (long t = tid; t < (long)NF * 2 * H * PSY * PSZ; t += stride) {{
        const int c = (int)(t % PSZ);
        long r = t / PSZ;
        const int j = (int)(r % PSY);
        r /= PSY;
        const int p = (int)(r % (2 * H));
        const long base = (r / (2 * H)) * PVOL;
        const int xd = p < H ? p : NX + H + (p - H);
        const int xs = p < H ? p + NX : p;
        f[base + ((long)xd * PSY + j) * PSZ + c] =
            f[base + ((long)xs * PSY + j) * PSZ + c];
    }}
#endif
#if WRAPY
    for (long t = tid; t < (long)NF * 2 * H * PSX * PSZ; t += stride) {{
        const int c = (int)(t % PSZ);
        long r = t / PSZ;
        const int i = (int)(r % PSX);
        r /= PSX;
        const int p = (int)(r % (2 * H));
        const long base = (r / (2 * H)) * PVOL;
        const int yd = p < H ? p : NY + H + (p - H);
        const int ys = p < H ? p + NY : p;
        f[base + ((long)i * PSY + yd) * PSZ + c] =
            f[base + ((long)i * PSY + ys) * PSZ + c];
    }}
#endif
#if WRAPZ
    for (long t = tid; t < (long)NF * 2 * H * PSX * PSY; t += stride) {{
        long r = t;
        const int j = (int)(r % PSY);
        r /= PSY;
        const int i = (int)(r % PSX);
        r /= PSX;
        const int p = (int)(r % (2 * H));
        const long base = (r / (2 * H)) * PVOL;
        const int zd = p < H ? p : NZ + H + (p - H);
        const int zs = p < H ? p + NZ : p;
        f[base + ((long)i * PSY + j) * PSZ + zd] =
            f[base + ((long)i * PSY + j) * PSZ + zs];
    }}
#endif
}}
"""

_wrap_cache = {}


def wrap_star(fx, halo, wrap_axes):
    """Periodic wrap of all ``wrap_axes`` halo faces of ``fx`` in one
    kernel launch (star-stencil contract; see WRAP_TEMPLATE)."""
    h = max(halo) if isinstance(halo, (tuple, list)) else halo
    nxp, nyp, nzp = fx.shape[-3:]
    rank_shape = (nxp - 2 * h, nyp - 2 * h, nzp - 2 * h)
    nf = 1
    for n in fx.shape[:-3]:
        nf *= n
    key = (rank_shape, h, nf, tuple(sorted(wrap_axes)), fx.dtype)
    k = _wrap_cache.get(key)
    if k is None:
        defines = geometry_defines(h, rank_shape, rtype=_RTYPE[fx.dtype])
        defines += f"#define NF {nf}\n"
        for ax, nm in enumerate("XYZ"):
            defines += f"#define WRAP{nm} {1 if ax in wrap_axes else 0}\n"
        name = f"wrap_star_{h}_{nf}_" + "".join(
            str(int(ax in wrap_axes)) for ax in range(3))
        src = WRAP_TEMPLATE.format(defines=defines, name=name)
        key_id = ext().jit_compile(src, name)
        cells = max((nxp + 2 * h) * (nyp + 2 * h), 1) * 2 * h * nf
        grid = min(4096, (cells + 255) // 256)
        _wrap_cache[key] = k = (key_id, grid)
    _check_tensor("f", fx)
    ext().jit_launch(k[0], k[1], 1, 1, 256, 1, 1, 0, _stream(),
                     [fx.data_ptr()], [], [])


# ---------------------------------------------------------------------------
# JIT'd simultaneous reductions

REDUCTION_TAIL = """
    __shared__ double sd[TBZ * TBY];
    const int nblk = gridDim.x * gridDim.y * gridDim.z;
    const int bid = (blockIdx.z * gridDim.y + blockIdx.y) * gridDim.x
                    + blockIdx.x;
    for (int r = 0; r < NRED; ++r) {{
        sd[threadIdx.x] = acc[r];
        __syncthreads();
        for (int s = (TBZ * TBY) / 2; s > 0; s >>= 1) {{
            if ((int)threadIdx.x < s)
                sd[threadIdx.x] = COMBINE(r, sd[threadIdx.x],
                                          sd[threadIdx.x + s]);
            __syncthreads();
        }}
        if (threadIdx.x == 0)
            partials[(long)r * nblk + bid] = sd[0];
        __syncthreads();
    }}
}}
"""

REDUCTION_TEMPLATE = """{defines}
{preamble}
#define NRED {nred}
extern "C" __global__ __launch_bounds__(TBZ * TBY) void {name}(
    {params})
{{
    double acc[NRED];
    {init}
    const int k = blockIdx.x * TBZ + (threadIdx.x % TBZ);
    const int j = blockIdx.y * TBY + (threadIdx.x / TBZ);
    const int i0 = blockIdx.z * XCHUNK;
    const int i1 = (i0 + XCHUNK < NX) ? i0 + XCHUNK : NX;
    if (k < NZ && j < NY) {{
        for (int i = i0; i < i1; ++i) {{
            {body}
        }}
    }}
""" + REDUCTION_TAIL

_OP_INIT = {"sum": "0.0", "avg": "0.0", "prod": "1.0",
            "max": "-1.0e308", "min": "1.0e308"}
_OP_COMBINE = {"sum": "(a + b)", "avg": "(a + b)", "prod": "(a * b)",
               "max": "fmax(a, b)", "min": "fmin(a, b)"}


class JitReduction:
    """Fused multi-quantity grid reduction → per-block partials,
    finished with torch ops + one packed allreduce by the caller."""

    def __init__(self, entries, field_args, scalar_names, halo, rank_shape,
                 name="reduce_map", tile=(64, 4, 64), dtype=None):
        import torch as _t
        dtype = dtype if dtype is not None else _t.float64
        self.rank_shape = tuple(rank_shape)
        self.tile = tile
        self.dtype = dtype
        self.entries = entries
        self.field_args = [fa for fa in field_args if fa.spatial]
        nred = len(entries)
        cg = Codegen(field_args, halo, rank_shape)

        init_lines = []
        body_lines = []
        combine_cases = []
        for r, (expr, op) in enumerate(entries):
            init_lines.append(f"acc[{r}] = {_OP_INIT[op]};")
            comb = _OP_COMBINE[op]
            val = cg.emit(expr)
            body_lines.append(
                "{ const double a = acc[%d]; const double b = %s; "
                "acc[%d] = %s; }" % (r, val, r, comb))
            combine_cases.append(f"(r == {r}) ? {comb} : ")
        combine = "".join(combine_cases) + "0.0"

        ptr_params = ", ".join(
            f"const real* __restrict__ {fa.name}"
            for fa in self.field_args)
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (
            ptr_params, "double* __restrict__ partials", dbl_params) if x)

        defines = geometry_defines(halo, rank_shape, rtype=_RTYPE[dtype])
        defines += _tile_defines(tile, rank_shape)
        defines += ("#define COMBINE(r, a, b) (" + combine + ")\n")
        src = REDUCTION_TEMPLATE.format(
            defines=defines, preamble=PREAMBLE, nred=nred, name=name,
            params=params,
            init="\n    ".join(init_lines),
            body="\n        ".join(body_lines))
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)
        self.grid = _tile_grid(tile, rank_shape)
        self.block = tile[0] * tile[1]
        self.nblk = self.grid[0] * self.grid[1] * self.grid[2]
        self._partials = None

    def _finish(self, dev):
        """Combine per-block partials on-device, one packed D2H copy."""
        vals = torch.empty(len(self.entries), dtype=torch.float64,
                           device=dev)
        for r, (_, op) in enumerate(self.entries):
            row = self._partials[r]
            if op in ("sum", "avg"):
                vals[r] = row.sum()
            elif op == "prod":
                vals[r] = row.prod()
            elif op == "max":
                vals[r] = row.max()
            else:
                vals[r] = row.min()
        return vals.cpu().tolist()

    def __call__(self, env):
        dev = None
        ptrs = []
        want = getattr(self, "dtype", None)
        for fa in self.field_args:
            t = _check_tensor(fa.name, env[fa.name])
            if want is None:
                want = t.dtype
            elif t.dtype != want:
                raise TypeError(
                    f"reduction argument '{fa.name}' has dtype "
                    f"{t.dtype}, kernel expects {want}")
            dev = t.device
            ptrs.append(t.data_ptr())
        nred = len(self.entries)
        if (self._partials is None
                or self._partials.device != dev
                or self._partials.shape[1] != self.nblk):
            self._partials = torch.empty((nred, self.nblk),
                                         dtype=torch.float64, device=dev)
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, self.grid[0], self.grid[1],
                         self.grid[2], self.block, 1, 1, 0, _stream(),
                         ptrs + [self._partials.data_ptr()], [], doubles)
        return self._finish(dev)


JitStageReduction._finish = JitReduction._finish


def get_reduction_kernel(entries, field_args, scalar_names, halo,
                         rank_shape, tile=(64, 4, 64), dtype=None):
    return JitReduction(entries, field_args, scalar_names, halo,
                        rank_shape, tile=tile, dtype=dtype)


# ---------------------------------------------------------------------------
# JIT'd histogrammer: LDS bins + device-scope atomic merge

HISTOGRAM_TEMPLATE = """{defines}
{preamble}
#define NHIST {nhist}
#define NBINS {nbins}
extern "C" __global__ __launch_bounds__(256) void {name}(
    {params})
{{
    __shared__ double lh[NHIST * NBINS];
    for (int b = threadIdx.x; b < NHIST * NBINS; b += blockDim.x)
        lh[b] = 0.0;
    __syncthreads();

    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long total = (long)NX * NY * NZ;
    const long stride = (long)gridDim.x * blockDim.x;
    for (; idx < total; idx += stride) {{
        const int k = (int)(idx % NZ);
        const long t = idx / NZ;
        const int j = (int)(t % NY);
        const int i = (int)(t / NY);
        {body}
    }}
    __syncthreads();
    for (int b = threadIdx.x; b < NHIST * NBINS; b += blockDim.x)
        atomicAdd(&hist[b], lh[b]);
}}
"""


# Large-bin-count fallback: accumulate straight into global memory
# (device-scope atomics) when NHIST*NBINS doubles exceed the LDS
# budget of one workgroup (the reference caps the workgroup and merges
# through global atomics too: histogram.py:69,114-163).
HISTOGRAM_GLOBAL_TEMPLATE = """{defines}
{preamble}
#define NHIST {nhist}
#define NBINS {nbins}
extern "C" __global__ __launch_bounds__(256) void {name}(
    {params})
{{
    double* lh = hist;
    long idx = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long total = (long)NX * NY * NZ;
    const long stride = (long)gridDim.x * blockDim.x;
    for (; idx < total; idx += stride) {{
        const int k = (int)(idx % NZ);
        const long t = idx / NZ;
        const int j = (int)(t % NY);
        const int i = (int)(t / NY);
        {body}
    }}
}}
"""

# one workgroup may use at most 64 KB LDS on gfx9xx
_HIST_LDS_DOUBLES = 8192


class JitHistogram:
    def __init__(self, pairs, num_bins, field_args, scalar_names, halo,
                 rank_shape, name="hist_map"):
        self.rank_shape = tuple(rank_shape)
        self.num_bins = num_bins
        self.nhist = len(pairs)
        self.field_args = [fa for fa in field_args if fa.spatial]
        cg = Codegen(field_args, halo, rank_shape)
        body = []
        for hh, (bin_expr, weight_expr) in enumerate(pairs):
            b = cg.emit(bin_expr)
            w = cg.emit(weight_expr)
            body.append(
                "{ int bb = (int)(%s); bb = bb < 0 ? 0 : "
                "(bb >= NBINS ? NBINS - 1 : bb); "
                "atomicAdd(&lh[%d * NBINS + bb], (double)(%s)); }"
                % (b, hh, w))
        ptr_params = ", ".join(
            f"const double* __restrict__ {fa.name}"
            for fa in self.field_args)
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (
            ptr_params, "double* __restrict__ hist", dbl_params) if x)
        template = (HISTOGRAM_TEMPLATE
                    if self.nhist * num_bins <= _HIST_LDS_DOUBLES
                    else HISTOGRAM_GLOBAL_TEMPLATE)
        src = template.format(
            defines=geometry_defines(halo, rank_shape), preamble=PREAMBLE,
            nhist=self.nhist, nbins=num_bins, name=name, params=params,
            body="\n        ".join(body))
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)
        total = int(np.prod(rank_shape))
        self.grid = min((total + 255) // 256, 1024)

    def __call__(self, env):
        ptrs = []
        dev = None
        for fa in self.field_args:
            t = _check_tensor(fa.name, env[fa.name])
            dev = t.device
            ptrs.append(t.data_ptr())
        hist = torch.zeros((self.nhist, self.num_bins),
                           dtype=torch.float64, device=dev)
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, self.grid, 1, 1, 256, 1, 1, 0,
                         _stream(), ptrs + [hist.data_ptr()], [], doubles)
        return hist.cpu().numpy()


def get_histogram_kernel(pairs, num_bins, field_args, scalar_names, halo,
                         rank_shape):
    return JitHistogram(pairs, num_bins, field_args, scalar_names, halo,
                        rank_shape)


# ---------------------------------------------------------------------------
# Spectra binning: LDS per-block bins + one global merge.  torch's
# index_add_ funnels every site's atomic into ~500 global bins and
# takes seconds at 512^3; this kernel does it in milliseconds
# (reference K11, spectra.py:103-138).

SPECTRA_BIN_SRC = """
#define NBINS {nbins}
extern "C" __global__ __launch_bounds__(256) void spectra_bin(
    const double* __restrict__ fk, const double* __restrict__ wbase,
    const int* __restrict__ bidx, double* __restrict__ hist, long n)
{{
    __shared__ double lh[NBINS];
    for (int b = threadIdx.x; b < NBINS; b += 256) lh[b] = 0.0;
    __syncthreads();
    long i = (long)blockIdx.x * blockDim.x + threadIdx.x;
    const long stride = (long)gridDim.x * blockDim.x;
    for (; i < n; i += stride) {{
        const double re = fk[2 * i];
        const double im = fk[2 * i + 1];
        atomicAdd(&lh[bidx[i]], wbase[i] * (re * re + im * im));
    }}
    __syncthreads();
    for (int b = threadIdx.x; b < NBINS; b += 256)
        atomicAdd(&hist[b], lh[b]);
}}
"""

_spectra_bin_cache = {}


def spectra_bin(fk, wbase, bidx, num_bins):
    """|f_k|²-weighted LDS-binned histogram; fk complex128, wbase fp64,
    bidx int32 (all flat, same length).  Returns fp64 hist[num_bins]."""
    _check_tensor("fk", fk)
    n = fk.numel()
    key = num_bins
    k = _spectra_bin_cache.get(key)
    if k is None:
        src = SPECTRA_BIN_SRC.format(nbins=num_bins)
        kid = ext().jit_compile(src, "spectra_bin")
        _spectra_bin_cache[key] = k = kid
    hist = torch.zeros(num_bins, dtype=torch.float64, device=fk.device)
    grid = min(4096, (n + 255) // 256)
    ext().jit_launch(k, grid, 1, 1, 256, 1, 1, 0, _stream(),
                     [fk.data_ptr(), wbase.data_ptr(),
                      bidx.data_ptr(), hist.data_ptr()], [n], [])
    return hist


# ---------------------------------------------------------------------------
# fused k-space projector kernels: one launch per operation, ε-basis
# built in registers (reference pystella/fourier/projectors.py:108-236
# runs each of these as ONE generated kernel too; round 1 shipped them
# as multi-launch torch chains — these close K10)

PROJECTOR_PREAMBLE = """
struct cplx { double x; double y; };
__device__ inline cplx cmul(cplx a, cplx b)
{ return {a.x*b.x - a.y*b.y, a.x*b.y + a.y*b.x}; }
__device__ inline cplx cadd(cplx a, cplx b)
{ return {a.x + b.x, a.y + b.y}; }
__device__ inline cplx conjg(cplx a) { return {a.x, -a.y}; }
__device__ inline cplx cscale(double s, cplx a)
{ return {s * a.x, s * a.y}; }
#define LDC(p, c) cplx{ (p)[2*((long)(c)*VOLK + idx)], \\
                        (p)[2*((long)(c)*VOLK + idx) + 1] }
#define STC(p, c, v) { (p)[2*((long)(c)*VOLK + idx)] = (v).x; \\
                       (p)[2*((long)(c)*VOLK + idx) + 1] = (v).y; }
"""

PROJECTOR_KERNEL = """
extern "C" __global__ __launch_bounds__(256) void {name}(
    {params},
    const double* __restrict__ eff_x,
    const double* __restrict__ eff_y,
    const double* __restrict__ eff_z)
{{
    const long idx = (long)blockIdx.x * 256 + threadIdx.x;
    if (idx >= VOLK) return;
    const int kk = (int)(idx % NKZ);
    const int jj = (int)((idx / NKZ) % NKY);
    const int ii = (int)(idx / ((long)NKZ * NKY));
    const double kx = eff_x[ii], ky = eff_y[jj], kz = eff_z[kk];
    const double ksq = kx*kx + ky*ky + kz*kz;
    const bool kzero = (fabs(kx) < 1e-14) && (fabs(ky) < 1e-14)
                        && (fabs(kz) < 1e-14);
    (void)ksq; (void)kzero;
    {eps}
    {body}
}}
"""

# ε-basis construction, matching the torch construction in
# pystella_amd/fourier/projectors.py:57-78 (reference
# projectors.py:123-142 incl. the k_x = k_y = 0 special case)
PROJECTOR_EPS = """
    const double Kappa = sqrt(kx*kx + ky*ky);
    const double kmag = sqrt(ksq);
    const double kmag_s = (kmag > 0.) ? kmag : 1.0;
    const double Kappa_s = (Kappa > 0.) ? Kappa : 1.0;
    const bool kxy0 = (fabs(kx) < 1e-10) && (fabs(ky) < 1e-10);
    const bool kznz = fabs(kz) > 1e-10;
    const double S2 = 0.7071067811865476;
    cplx eps0, eps1, eps2;
    if (kxy0) {
        eps0 = kznz ? cplx{S2, 0.} : cplx{0., 0.};
        eps1 = kznz ? cplx{0., S2} : cplx{0., 0.};
        eps2 = cplx{0., 0.};
    } else {
        eps0 = cplx{ kx*kz/kmag_s/Kappa_s*S2, -ky/Kappa_s*S2 };
        eps1 = cplx{ ky*kz/kmag_s/Kappa_s*S2,  kx/Kappa_s*S2 };
        eps2 = cplx{ -Kappa/kmag_s*S2, 0. };
    }
    (void)eps0; (void)eps1; (void)eps2;
"""


def _proj_sym(c, d):
    """tid(c,d) for 1-based (c,d) -> symmetric 6-component index."""
    a, b = min(c, d), max(c, d)
    return {(1, 1): 0, (1, 2): 1, (1, 3): 2,
            (2, 2): 3, (2, 3): 4, (3, 3): 5}[(a, b)]


def _projector_bodies():
    eps = ["eps0", "eps1", "eps2"]
    bodies = {}

    bodies["transversify"] = ("""
    cplx v0 = LDC(vec,0), v1 = LDC(vec,1), v2 = LDC(vec,2);
    const double iksq = (ksq > 0.) ? 1.0 / ksq : 0.0;
    cplx div = cadd(cadd(cscale(kx,v0), cscale(ky,v1)), cscale(kz,v2));
    cplx r0 = {v0.x - kx*iksq*div.x, v0.y - kx*iksq*div.y};
    cplx r1 = {v1.x - ky*iksq*div.x, v1.y - ky*iksq*div.y};
    cplx r2 = {v2.x - kz*iksq*div.x, v2.y - kz*iksq*div.y};
    if (kzero) { r0 = {0.,0.}; r1 = {0.,0.}; r2 = {0.,0.}; }
    STC(out,0,r0) STC(out,1,r1) STC(out,2,r2)
""", ["vec", "out"], False)

    pm_sum = "".join(
        f"""
    p = cadd(p, cmul(v{m}, conjg({eps[m]})));
    mi = cadd(mi, cmul(v{m}, {eps[m]}));""" for m in range(3))
    bodies["vec_to_pol"] = ("""
    cplx v0 = LDC(vec,0), v1 = LDC(vec,1), v2 = LDC(vec,2);
    cplx p = {0.,0.}, mi = {0.,0.};""" + pm_sum + """
    STC(plus,0,p) STC(minus,0,mi)
""", ["vec", "plus", "minus"], True)

    p2v = "".join(
        f"""
    cplx r{m} = cadd(cmul(p, {eps[m]}), cmul(mi, conjg({eps[m]})));"""
        for m in range(3))
    stores = " ".join(f"STC(out,{m},r{m})" for m in range(3))
    bodies["pol_to_vec"] = ("""
    cplx p = LDC(plus,0), mi = LDC(minus,0);""" + p2v + f"""
    {stores}
""", ["plus", "minus", "out"], True)

    bodies["decompose_vector"] = ("""
    cplx v0 = LDC(vec,0), v1 = LDC(vec,1), v2 = LDC(vec,2);
    cplx p = {0.,0.}, mi = {0.,0.};""" + pm_sum + """
    cplx div = cadd(cadd(cscale(kx,v0), cscale(ky,v1)), cscale(kz,v2));
    const double denom = (ksq > 0.) ? (TIMES_ABS_K ? sqrt(ksq) : ksq)
                                     : 1.0;
    cplx lng = { div.y / denom, -div.x / denom };
    if (kzero) lng = {0., 0.};
    STC(plus,0,p) STC(minus,0,mi) STC(lngp,0,lng)
""", ["vec", "plus", "minus", "lngp"], True)

    d2v = "".join(f"""
    cplx r{m} = cadd(cmul(p, {eps[m]}), cmul(mi, conjg({eps[m]})));
    {{ const double km = {"kx" if m == 0 else ("ky" if m == 1 else "kz")};
       const double fac = TIMES_ABS_K ? km : km / kmag_s;
       cplx extra = {{ -fac * lng.y, fac * lng.x }};
       if (!kzero) r{m} = cadd(r{m}, extra); }}""" for m in range(3))
    bodies["decomp_to_vec"] = ("""
    cplx p = LDC(plus,0), mi = LDC(minus,0), lng = LDC(lngp,0);"""
        + d2v + f"""
    {stores}
""", ["plus", "minus", "lngp", "out"], True)

    t2p_terms = "".join(
        f"""
    p = cadd(p, cmul(h{_proj_sym(c, d)},
                     cmul(conjg({eps[c-1]}), conjg({eps[d-1]}))));
    mi = cadd(mi, cmul(h{_proj_sym(c, d)},
                       cmul({eps[c-1]}, {eps[d-1]})));"""
        for c in range(1, 4) for d in range(1, 4))
    loads6 = " ".join(f"cplx h{m} = LDC(hij,{m});" for m in range(6))
    bodies["tensor_to_pol"] = (f"""
    {loads6}
    cplx p = {{0.,0.}}, mi = {{0.,0.}};""" + t2p_terms + """
    STC(plus,0,p) STC(minus,0,mi)
""", ["hij", "plus", "minus"], True)

    p2t = "".join(
        f"""
    cplx r{_proj_sym(a, b)} = cadd(
        cmul(p, cmul({eps[a-1]}, {eps[b-1]})),
        cmul(mi, cmul(conjg({eps[a-1]}), conjg({eps[b-1]}))));"""
        for a in range(1, 4) for b in range(a, 4))
    stores6 = " ".join(f"STC(hij,{m},r{m})" for m in range(6))
    bodies["pol_to_tensor"] = ("""
    cplx p = LDC(plus,0), mi = LDC(minus,0);""" + p2t + f"""
    {stores6}
""", ["plus", "minus", "hij"], True)

    return bodies


_PROJ_BODIES = _projector_bodies()
_proj_cache = {}


def projector_source(op, kshape, times_abs_k=False):
    body, names, needs_eps = _PROJ_BODIES[op]
    params = ",\n    ".join(f"double* __restrict__ {n}" for n in names)
    head = (f"#define TIMES_ABS_K {1 if times_abs_k else 0}\n"
            f"#define NKX {kshape[0]}\n"
            f"#define NKY {kshape[1]}\n"
            f"#define NKZ {kshape[2]}\n"
            "#define VOLK ((long)NKX * NKY * NKZ)\n")
    return head + PROJECTOR_PREAMBLE + PROJECTOR_KERNEL.format(
        name=f"proj_{op}", params=params,
        eps=PROJECTOR_EPS if needs_eps else "", body=body)


def projector_op(op, kshape, ptrs, eff, times_abs_k=False):
    """Launch fused projector kernel ``op`` over k-space ``kshape``.

    :arg ptrs: list of data_ptrs in the op's parameter order.
    :arg eff: (eff_x, eff_y, eff_z) contiguous fp64 device tensors.
    """
    body, names, needs_eps = _PROJ_BODIES[op]
    key = (op, tuple(kshape), bool(times_abs_k))
    kid = _proj_cache.get(key)
    if kid is None:
        src = projector_source(op, kshape, times_abs_k)
        kid = ext().jit_compile(src, f"proj_{op}")
        _proj_cache[key] = kid
    vol = int(np.prod(kshape))
    grid = (vol + 255) // 256
    ext().jit_launch(kid, grid, 1, 1, 256, 1, 1, 0, _stream(),
                     list(ptrs) + [e.data_ptr() for e in eff], [], [])


# ---------------------------------------------------------------------------
# AOT stencil kernels (csrc/derivs.hip)

def _flat_fields(t, ndim_grid=3):
    """Collapse outer axes; returns (tensor_view, nf)."""
    outer = t.shape[:-ndim_grid]
    nf = int(np.prod(outer)) if outer else 1
    return t, nf


_DERIV_DTYPES = {torch.float64: 0, torch.float32: 1}


def _deriv_dtype(name, t):
    code = _DERIV_DTYPES.get(t.dtype)
    if code is None:
        raise TypeError(f"{name}: stencil kernels support fp64/fp32, "
                        f"got {t.dtype}")
    return code


def derivs(fx, lap=None, pdx=None, pdy=None, pdz=None, grd=None, halo=None,
           dx=None, h=None, stream=True):
    if len(set(halo)) != 1:
        raise NotImplementedError("GPU stencils require isotropic halo")
    _check_tensor("fx", fx)
    dtype = _deriv_dtype("fx", fx)
    esize = fx.element_size()
    nxp, nyp, nzp = fx.shape[-3:]
    nx, ny, nz = nxp - 2 * h, nyp - 2 * h, nzp - 2 * h
    _, nf = _flat_fields(fx)
    uvol = nx * ny * nz

    def ptr(t):
        if t is None:
            return 0
        _check_tensor("out", t)
        if t.dtype != fx.dtype:
            raise TypeError(f"output dtype {t.dtype} != input {fx.dtype}")
        return t.data_ptr()

    e = ext()
    # gradient outputs: either a packed (..., 3, nx, ny, nz) grd array
    # (per-field component stride 3*uvol) or three standalone arrays
    if grd is not None and isinstance(grd, torch.Tensor):
        _check_tensor("grd", grd)
        gp = ptr(grd)
        px, py, pz_ = gp, gp + esize * uvol, gp + 2 * esize * uvol
        g_fstride = 3 * uvol
        want_grad = True
    elif pdx is not None and pdy is not None and pdz is not None:
        px, py, pz_ = ptr(pdx), ptr(pdy), ptr(pdz)
        g_fstride = uvol
        want_grad = True
    else:
        px = py = pz_ = 0
        g_fstride = 0
        want_grad = False

    if lap is not None or want_grad:
        e.gradlap(fx.data_ptr(), ptr(lap), px, py, pz_, g_fstride,
                  h, nx, ny, nz, nf, dx[0], dx[1], dx[2], dtype,
                  _stream())
        return
    # single-axis derivatives
    for axis, out in enumerate((pdx, pdy, pdz)):
        if out is not None:
            e.pd(fx.data_ptr(), ptr(out), h, axis, 0, nx, ny, nz, nf,
                 dx[axis], dtype, _stream())


def divergence(vec, div, halo=None, dx=None, h=None):
    _check_tensor("vec", vec)
    _check_tensor("div", div)
    dtype = _deriv_dtype("vec", vec)
    if div.dtype != vec.dtype:
        raise TypeError(f"div dtype {div.dtype} != vec {vec.dtype}")
    nxp, nyp, nzp = vec.shape[-3:]
    nx, ny, nz = nxp - 2 * h, nyp - 2 * h, nzp - 2 * h
    outer = vec.shape[:-4]
    e = ext()
    from itertools import product
    for s in product(*[range(n) for n in outer]):
        e.pd(vec[s][0].data_ptr(), div[s].data_ptr(), h, 0, 0,
             nx, ny, nz, 1, dx[0], dtype, _stream())
        e.pd(vec[s][1].data_ptr(), div[s].data_ptr(), h, 1, 1,
             nx, ny, nz, 1, dx[1], dtype, _stream())
        e.pd(vec[s][2].data_ptr(), div[s].data_ptr(), h, 2, 1,
             nx, ny, nz, 1, dx[2], dtype, _stream())


# ---------------------------------------------------------------------------
# Fused Laplacian + reduction kernel (MI355X-specific optimization).
#
# The reference hot loop runs the stencil pass and the energy reduction
# as separate kernels (reference examples/scalar_preheating.py:258-271 →
# derivs.py:339-429 then reduction.py:206); that re-reads f and lap_f
# from HBM.  Here one x-marching pass computes lap (register ring +
# current-plane neighbor loads), stores it, and accumulates the energy
# reductions with the freshly computed Laplacian still in registers:
# HBM traffic per site drops from (f, lap w, f, dfdt, lap r) to
# (f, dfdt, lap w).

def _lap_stencil_pieces(h, dx, periodic):
    """Shared codegen for the inline-Laplacian ring kernels: stencil
    term strings, the center coefficient, and the x-ring load/init
    forms.  periodic=(px,py,pz) wraps that axis's stencil reads
    in-kernel (star stencil) so non-decomposed axes need NO halo fill.
    """
    from pystella_amd.derivs import _LAP_COEFS
    inv2 = [1.0 / d / d for d in dx]
    coefs = _LAP_COEFS[h]
    px_, py_, pz_ = periodic
    wrap_decls = []
    lap_terms = []
    for s in range(1, h + 1):
        c = coefs[s]
        if py_:
            wrap_decls.append(
                f"const long ypo{s} = ((j + {s} < NY) ? {s}L "
                f": {s}L - NY) * PSZ;")
            wrap_decls.append(
                f"const long ymo{s} = ((j >= {s}) ? -{s}L "
                f": NY - {s}L) * PSZ;")
            yp, ym = f"ypo{s}", f"ymo{s}"
        else:
            yp, ym = f"{s}*PSZ", f"-{s}*PSZ"
        if pz_:
            wrap_decls.append(
                f"const int zpo{s} = (k + {s} < NZ) ? {s} "
                f": {s} - NZ;")
            wrap_decls.append(
                f"const int zmo{s} = (k >= {s}) ? -{s} "
                f": NZ - {s};")
            zp, zm = f"zpo{s}", f"zmo{s}"
        else:
            zp, zm = f"{s}", f"-{s}"
        lap_terms.append(
            f"la += {c!r} * ((ring[fld][H+{s}] + ring[fld][H-{s}])"
            f"*{inv2[0]!r} + (cp[{yp}] + cp[{ym}])*{inv2[1]!r}"
            f" + (cp[{zp}] + cp[{zm}])*{inv2[2]!r});")
    lapc0 = coefs[0] * (inv2[0] + inv2[1] + inv2[2])
    if px_:
        x_off = ("int xq = i + 2 * H; "
                 "if (xq >= NX + H) xq -= NX;")
        ring_load = "ring[fld][2 * H] = fp[(long)xq * sx];"
        ring_init = ("int xq0 = i0 + p; "
                     "if (xq0 < H) xq0 += NX; "
                     "if (xq0 >= NX + H) xq0 -= NX; "
                     "ring[fld][p] = fp[(long)xq0 * sx];")
    else:
        x_off = ""
        ring_load = "ring[fld][2 * H] = fp[(long)(i + 2 * H) * sx];"
        ring_init = "ring[fld][p] = fp[(long)(i0 + p) * sx];"
    return wrap_decls, lap_terms, lapc0, x_off, ring_load, ring_init



LAPRED_TEMPLATE = """{defines}
{preamble}
#define NRED {nred}
#define NF {nf}
extern "C" __global__ __launch_bounds__(TBZ * TBY) void {name}(
    {params})
{{
    double acc[NRED];
    {init}
    const int k = blockIdx.x * TBZ + (threadIdx.x % TBZ);
    const int j = blockIdx.y * TBY + (threadIdx.x / TBZ);
    const int i0 = blockIdx.z * XCHUNK;
    const int i1 = (i0 + XCHUNK < NX) ? i0 + XCHUNK : NX;
    if (k < NZ && j < NY) {{
        const long sx = PSY * PSZ;
        double ring[NF][2 * H + 1];
        #pragma unroll
        for (int fld = 0; fld < NF; ++fld) {{
            const double* fp = {fname} + (long)fld * PVOL
                               + (long)(j + H) * PSZ + (k + H);
            #pragma unroll
            for (int p = 0; p < 2 * H; ++p)
                ring[fld][p] = fp[(long)(i0 + p) * sx];
        }}
        for (int i = i0; i < i1; ++i) {{
            double lapv[NF];
            #pragma unroll
            for (int fld = 0; fld < NF; ++fld) {{
                const double* fp = {fname} + (long)fld * PVOL
                                   + (long)(j + H) * PSZ + (k + H);
                ring[fld][2 * H] = fp[(long)(i + 2 * H) * sx];
                const double* cp = fp + (long)(i + H) * sx;
                double la = ring[fld][H] * LAPC0;
                {lap_terms}
                lapv[fld] = la;
#if STORE_LAP
                {lapname}[(long)fld * UVOL + (((long)i * NY + j) * NZ + k)]
                    = la;
#endif
            }}
            {body}
            #pragma unroll
            for (int fld = 0; fld < NF; ++fld)
                #pragma unroll
                for (int p = 0; p < 2 * H; ++p)
                    ring[fld][p] = ring[fld][p + 1];
        }}
    }}
""" + REDUCTION_TAIL


class _LapCodegen(Codegen):
    """Codegen that maps accesses to the stencil field's center value and
    the freshly computed Laplacian onto kernel registers."""

    def __init__(self, field_args, halo, rank_shape, f_name, lap_name):
        super().__init__(field_args, halo, rank_shape)
        self.f_name = f_name
        self.lap_name = lap_name

    def field_access(self, f, outer_idx):
        if f.is_spatial and not any(f.shift) and len(outer_idx) <= 1:
            lin = int(outer_idx[0]) if outer_idx else 0
            if f.name == self.lap_name:
                return f"lapv[{lin}]"
            if f.name == self.f_name:
                return f"ring[{lin}][H]"
        return super().field_access(f, outer_idx)


class JitLapReduction:
    """Fused lap-stencil + multi-quantity reduction (see module note)."""

    def __init__(self, entries, field_args, scalar_names, halo, rank_shape,
                 dx, nf, f_name="f", lap_name="lap_f", name="lapred_map",
                 tile=(64, 4, 64), store_lap=True,
                 periodic=(False, False, False)):
        from pystella_amd.derivs import _LAP_COEFS
        self.rank_shape = tuple(rank_shape)
        self.tile = tile
        self.store_lap = store_lap
        self.entries = entries
        h = max(halo) if isinstance(halo, (tuple, list)) else halo
        self.nf = nf
        cg = _LapCodegen(field_args, halo, rank_shape, f_name, lap_name)

        init_lines, body_lines, combine_cases = [], [], []
        for r, (expr, op) in enumerate(entries):
            init_lines.append(f"acc[{r}] = {_OP_INIT[op]};")
            comb = _OP_COMBINE[op]
            val = cg.emit(expr)
            body_lines.append(
                "{ const double a = acc[%d]; const double b = %s; "
                "acc[%d] = %s; }" % (r, val, r, comb))
            combine_cases.append(f"(r == {r}) ? {comb} : ")
        combine = "".join(combine_cases) + "0.0"

        # Laplacian stencil terms with dx baked in
        self.periodic = tuple(periodic)
        (wrap_decls, lap_terms, lapc0, x_off, ring_load,
         ring_init) = _lap_stencil_pieces(h, dx, periodic)

        # pointer params: stencil field first, then lap (if stored),
        # then the rest
        self.ptr_names = ([f_name, lap_name] if store_lap
                          else [f_name]) + sorted(
            fa.name for fa in field_args
            if fa.spatial and fa.name not in (f_name, lap_name))
        by_name = {fa.name: fa for fa in field_args}
        self.field_args = [by_name[n] for n in self.ptr_names if n in by_name]
        ptr_params = ", ".join(
            f"double* __restrict__ {n}" for n in self.ptr_names)
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (
            ptr_params, "double* __restrict__ partials", dbl_params) if x)

        defines = geometry_defines(halo, rank_shape)
        defines += _tile_defines(tile, rank_shape)
        defines += f"#define COMBINE(r, a, b) ({combine})\n"
        defines += f"#define LAPC0 ({lapc0!r})\n"
        defines += f"#define STORE_LAP {1 if store_lap else 0}\n"
        src = LAPRED_TEMPLATE.format(
            defines=defines, preamble=PREAMBLE, nred=len(entries), nf=nf,
            name=name, params=params, fname=f_name, lapname=lap_name,
            init="\n    ".join(init_lines),
            lap_terms="\n                ".join(lap_terms),
            body="\n            ".join(body_lines))
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)
        self.grid = _tile_grid(tile, rank_shape)
        self.block = tile[0] * tile[1]
        self.nblk = self.grid[0] * self.grid[1] * self.grid[2]
        self._partials = None

    _finish = JitReduction._finish

    def __call__(self, env):
        dev = None
        ptrs = []
        for n in self.ptr_names:
            t = _check_tensor(n, env[n])
            dev = t.device
            ptrs.append(t.data_ptr())
        nred = len(self.entries)
        if (self._partials is None
                or self._partials.device != dev
                or self._partials.shape[1] != self.nblk):
            self._partials = torch.empty((nred, self.nblk),
                                         dtype=torch.float64, device=dev)
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, self.grid[0], self.grid[1],
                         self.grid[2], self.block, 1, 1, 0, _stream(),
                         ptrs + [self._partials.data_ptr()], [], doubles)
        return self._finish(dev)


# Reduction tail with runtime partials geometry (sub-box launches of
# the same kernel write disjoint slices of one shared partials buffer).
REDUCTION_TAIL_BOX = """
    __shared__ double sd[TBZ * TBY];
    const int bid = bid0 + (blockIdx.z * gridDim.y + blockIdx.y)
                    * gridDim.x + blockIdx.x;
    for (int r = 0; r < NRED; ++r) {{
        sd[threadIdx.x] = acc[r];
        __syncthreads();
        for (int s = (TBZ * TBY) / 2; s > 0; s >>= 1) {{
            if ((int)threadIdx.x < s)
                sd[threadIdx.x] = COMBINE(r, sd[threadIdx.x],
                                          sd[threadIdx.x + s]);
            __syncthreads();
        }}
        if (threadIdx.x == 0)
            partials[(long)r * nblkT + bid] = sd[0];
        __syncthreads();
    }}
}}
"""

LAPSTAGE_TEMPLATE = """{defines}
{preamble}
#define NRED {nred}
#define NF {nf}
extern "C" __global__ __launch_bounds__(TBZ * TBY, MINW) void {name}(
    {params})
{{
    double acc[NRED];
    {init}
    const int k = k0b + blockIdx.x * TBZ + (threadIdx.x % TBZ);
    const int j = j0b + blockIdx.y * TBY + (threadIdx.x / TBZ);
    const int i0 = i0b + blockIdx.z * XCHUNK;
    const int i1 = (i0 + XCHUNK < i1b) ? i0 + XCHUNK : i1b;
    if (k < k1b && j < j1b) {{
        const long sx = PSY * PSZ;
        {wrap_decls}
        double ring[NF][2 * H + 1];
        #pragma unroll
        for (int fld = 0; fld < NF; ++fld) {{
            const double* fp = {fname} + (long)fld * PVOL
                               + (long)(j + H) * PSZ + (k + H);
            #pragma unroll
            for (int p = 0; p < 2 * H; ++p) {{
                {ring_init}
            }}
        }}
        for (int i = i0; i < i1; ++i) {{
            double lapv[NF];
            {x_off}
            #pragma unroll
            for (int fld = 0; fld < NF; ++fld) {{
                const double* fp = {fname} + (long)fld * PVOL
                                   + (long)(j + H) * PSZ + (k + H);
                {ring_load}
                const double* cp = fp + (long)(i + H) * sx;
                double la = ring[fld][H] * LAPC0;
                {lap_terms}
                lapv[fld] = la;
            }}
            {body}
            #pragma unroll
            for (int fld = 0; fld < NF; ++fld)
                #pragma unroll
                for (int p = 0; p < 2 * H; ++p)
                    ring[fld][p] = ring[fld][p + 1];
        }}
    }}
""" + REDUCTION_TAIL_BOX


class _NTLapCodegen(_LapCodegen):
    """_LapCodegen plus read-redirection to nontemporal preloads and
    device-state scalars (scalars read from a device buffer instead of
    being passed by value — lets the whole RK step run without host
    synchronization, see JitFriedmann)."""

    def __init__(self, *args, state_map=None, **kwargs):
        super().__init__(*args, **kwargs)
        self.preload = {}       # (name, lin) -> register var
        self.store_ctx = False
        self.state_map = state_map or {}
        self.sectioned_lapv = False

    def scalar_param(self, name, idx=()):
        # state scalars are hoisted into per-thread registers once at
        # kernel entry (st_<name>, see JitLapStage state prologue)
        if name in self.state_map and not idx:
            return f"st_{name}"
        return super().scalar_param(name, idx)

    def field_access(self, f, outer_idx):
        if (not self.store_ctx and f.is_spatial
                and len(outer_idx) <= 1 and not any(f.shift)):
            lin = int(outer_idx[0]) if outer_idx else 0
            reg = self.preload.get((f.name, lin))
            if reg is not None:
                return reg
        if (self.sectioned_lapv and f.is_spatial and not any(f.shift)
                and len(outer_idx) <= 1 and f.name == self.lap_name):
            lin = int(outer_idx[0]) if outer_idx else 0
            return f"lapv_{lin}"
        return super().field_access(f, outer_idx)


# Multi-box "shell" form: ONE launch covers all boundary slabs of a
# rank (separate thin-slab launches are latency-bound at ~1 wave/CU
# each and serialize).  Box bounds and per-box block-grid tables are
# baked as constant arrays; blocks decode their (box, tile) from the
# flat blockIdx.x, and the reduction tail indexes partials by the flat
# block id.
LAPSTAGE_SHELL_TEMPLATE = """{defines}
{preamble}
#define NRED {nred}
#define NF {nf}
#define NBOX {nbox}
__constant__ int sh_k0[NBOX] = {{ {k0s} }};
__constant__ int sh_k1[NBOX] = {{ {k1s} }};
__constant__ int sh_j0[NBOX] = {{ {j0s} }};
__constant__ int sh_j1[NBOX] = {{ {j1s} }};
__constant__ int sh_i0[NBOX] = {{ {i0s} }};
__constant__ int sh_i1[NBOX] = {{ {i1s} }};
__constant__ int sh_gz[NBOX] = {{ {gzs} }};
__constant__ int sh_gy[NBOX] = {{ {gys} }};
__constant__ int sh_blk0[NBOX] = {{ {blk0s} }};
extern "C" __global__ __launch_bounds__(TBZ * TBY, MINW) void {name}(
    {params})
{{
    double acc[NRED];
    {init}
    const int bflat = (int)blockIdx.x;
    int box = 0;
    #pragma unroll
    for (int b = 1; b < NBOX; ++b) box += (bflat >= sh_blk0[b]);
    const int lb = bflat - sh_blk0[box];
    const int bz = lb % sh_gz[box];
    const int by = (lb / sh_gz[box]) % sh_gy[box];
    const int bx = lb / (sh_gz[box] * sh_gy[box]);
    const int k = sh_k0[box] + bz * TBZ + (int)(threadIdx.x % TBZ);
    const int j = sh_j0[box] + by * TBY + (int)(threadIdx.x / TBZ);
    const int i0 = sh_i0[box] + bx * XCHUNK;
    const int i1 = (i0 + XCHUNK < sh_i1[box]) ? i0 + XCHUNK
                                              : sh_i1[box];
    if (k < sh_k1[box] && j < sh_j1[box]) {{
        const long sx = PSY * PSZ;
        {wrap_decls}
        double lapv[NF];
        double ring[NF][2 * H + 1];
        #pragma unroll
        for (int fld = 0; fld < NF; ++fld) {{
            const double* fp = {fname} + (long)fld * PVOL
                               + (long)(j + H) * PSZ + (k + H);
            #pragma unroll
            for (int p = 0; p < 2 * H; ++p) {{
                {ring_init}
            }}
        }}
        for (int i = i0; i < i1; ++i) {{
            {x_off}
            #pragma unroll
            for (int fld = 0; fld < NF; ++fld) {{
                const double* fp = {fname} + (long)fld * PVOL
                                   + (long)(j + H) * PSZ + (k + H);
                {ring_load}
                const double* cp = fp + (long)(i + H) * sx;
                double la = ring[fld][H] * LAPC0;
                {lap_terms}
                lapv[fld] = la;
            }}
            {body}
            #pragma unroll
            for (int fld = 0; fld < NF; ++fld)
                #pragma unroll
                for (int p = 0; p < 2 * H; ++p)
                    ring[fld][p] = ring[fld][p + 1];
        }}
    }}
    __shared__ double sd[TBZ * TBY];
    const int bid = bid0 + bflat;
    for (int r = 0; r < NRED; ++r) {{
        sd[threadIdx.x] = acc[r];
        __syncthreads();
        for (int s = (TBZ * TBY) / 2; s > 0; s >>= 1) {{
            if ((int)threadIdx.x < s)
                sd[threadIdx.x] = COMBINE(r, sd[threadIdx.x],
                                          sd[threadIdx.x + s]);
            __syncthreads();
        }}
        if (threadIdx.x == 0)
            partials[(long)r * nblkT + bid] = sd[0];
        __syncthreads();
    }}
}}
"""


LAPSTAGE_SEC_TEMPLATE = """{defines}
{preamble}
#define NRED {nred}
#define NF {nf}
extern "C" __global__ __launch_bounds__(TBZ * TBY, MINW) void {name}(
    {params})
{{
    double acc[NRED];
    {init}
    const int k = k0b + blockIdx.x * TBZ + (threadIdx.x % TBZ);
    const int j = j0b + blockIdx.y * TBY + (threadIdx.x / TBZ);
    const int i0 = i0b + blockIdx.z * XCHUNK;
    const int i1 = (i0 + XCHUNK < i1b) ? i0 + XCHUNK : i1b;
    if (k < k1b && j < j1b) {{
        const long sx = PSY * PSZ;
        {wrap_decls}
        double ring[NF][2 * H + 1];
        #pragma unroll
        for (int fld = 0; fld < NF; ++fld) {{
            const double* fp = {fname} + (long)fld * PVOL
                               + (long)(j + H) * PSZ + (k + H);
            #pragma unroll
            for (int p = 0; p < 2 * H; ++p) {{
                {ring_init}
            }}
        }}
        for (int i = i0; i < i1; ++i) {{
            {x_off}
            {site_body}
            #pragma unroll
            for (int fld = 0; fld < NF; ++fld)
                #pragma unroll
                for (int p = 0; p < 2 * H; ++p)
                    ring[fld][p] = ring[fld][p + 1];
        }}
    }}
""" + REDUCTION_TAIL_BOX


class JitLapStage:
    """Ring-buffer RK stage kernel fused with input-state reductions:
    the stencil field marches along x through a per-thread register ring
    (one new load per site instead of 4h+1), the Laplacian is formed in
    registers, the reducers accumulate input-state values, and the
    2N-storage update stores last.  The k-arrays are read once via
    nontemporal loads and every store is nontemporal (streaming data
    never pollutes the L2, which stays available for stencil-neighbor
    reuse).  This is the hot loop's only kernel
    (see fusion.StencilRKStepper).

    For large stencil families (the 6-component GW h_ij sector), the
    flat per-site emission keeps all 6 Laplacians + 12 k-preloads +
    ring live simultaneously — 256 VGPRs, 2 waves/SIMD, ~4.2 TB/s
    (measured round 1).  ``section_size`` restructures the per-site
    body into per-component SECTIONS (ring load → Laplacian → RHS →
    k update → stores for one component, then the next) separated by
    ``__builtin_amdgcn_sched_barrier(0)`` so the register allocator
    sees short, non-overlapping live ranges; occupancy goes up at the
    price of less in-thread ILP (hidden instead by the extra waves).
    Set ``PYSTELLA_SECTIONS=<n>`` to force a section size (0 = flat
    emission)."""

    # tile candidates, best-first for large grids; smaller tiles win on
    # small per-rank grids (multi-GPU strong scaling) where the big
    # tile cannot fill 256 CUs with enough workgroups
    TILE_CANDIDATES = ((64, 8, 64), (64, 4, 32), (64, 2, 16),
                       (32, 2, 8))

    @classmethod
    def pick_tile(cls, rank_shape):
        import os as _os
        env = _os.environ.get("PYSTELLA_TILE")
        if env:
            return tuple(int(x) for x in env.split(","))
        for tile in cls.TILE_CANDIDATES:
            g = _tile_grid(tile, rank_shape)
            if g[0] * g[1] * g[2] >= 2048:
                return tile
        return max(cls.TILE_CANDIDATES,
                   key=lambda t: int(np.prod(_tile_grid(t, rank_shape))))

    def __init__(self, map_dict, tmp_instructions, entries, field_args,
                 scalar_names, halo, rank_shape, dx, nf, f_name="f",
                 lap_name="lap_f", name="rk_lapstage", tile=None,
                 nt=True, state_map=None, min_waves=1,
                 periodic=(False, False, False), section_size=None,
                 guard_stores=frozenset(), guard_scalar=None):
        import os as _os
        from pystella_amd.field import (
            Field, Subscript, Variable, is_number, iter_exprs, walk_expr)
        if tile is None:
            tile = self.pick_tile(rank_shape)
        self.rank_shape = tuple(rank_shape)
        # dynamic-LDS ballast at launch: caps waves/CU without kernel
        # changes (occupancy throttling for cache-thrash regimes)
        self.shmem = 0
        self.tile = tile
        self.entries = entries
        h = max(halo) if isinstance(halo, (tuple, list)) else halo
        self.nf = nf
        self.state_map = state_map
        cg = _NTLapCodegen(field_args, halo, rank_shape, f_name, lap_name,
                           state_map=state_map)

        tmp_items = list((tmp_instructions or {}).items())
        store_items = list(map_dict.items())
        tmp_set = set()
        for lhs, _ in tmp_items:
            tname = lhs.name if hasattr(lhs, "name") else str(lhs)
            tmp_set.add(tname)
            cg.tmp_names.add(tname)

        # ---- section-size selection -----------------------------------
        env = _os.environ.get("PYSTELLA_SECTIONS")
        if env is not None and env != "":
            section_size = int(env)
        entries_trivial = all(is_number(e) for e, _ in entries)
        if section_size is None:
            section_size = 0    # default pending A/B measurement

        def _store_comp(lhs):
            if isinstance(lhs, Subscript) and lhs.index \
                    and isinstance(lhs.index[0], (int, np.integer)):
                return int(lhs.index[0])
            return None

        comps = [_store_comp(lhs) for lhs, _ in store_items]
        sectioned = bool(
            section_size and section_size < nf and entries_trivial
            and all(c is not None and 0 <= c < nf for c in comps))

        # ---- NT preload discovery --------------------------------------
        nt_used = set()
        if nt:
            # preload every read component of the unpadded streaming
            # arrays (the RK k-arrays) with a nontemporal load
            nt_names = {fa.name for fa in field_args
                        if fa.spatial and not fa.padded
                        and fa.name != lap_name
                        and len(fa.outer_shape) <= 1}

            def visit(x):
                if isinstance(x, Subscript) and \
                        isinstance(x.aggregate, Field) and \
                        x.aggregate.name in nt_names and \
                        len(x.index) == 1 and \
                        isinstance(x.index[0], int):
                    nt_used.add((x.aggregate.name, int(x.index[0])))
                elif isinstance(x, Field) and x.name in nt_names \
                        and not x.shape:
                    nt_used.add((x.name, 0))

            for e in iter_exprs([[r for _, r in tmp_items],
                                 [e for e, _ in entries]]):
                walk_expr(e, visit)
            for nm, lin in sorted(nt_used):
                cg.preload[(nm, lin)] = f"pl_{nm}_{lin}"

        def _preload_decl(nm, lin):
            off = "(((long)i*NY + j)*NZ + k)"
            if lin:
                off = f"({lin}L*UVOL + {off})"
            return (f"const double pl_{nm}_{lin} = "
                    f"__builtin_nontemporal_load(&{nm}[{off}]);")

        # ---- shared emission helpers -----------------------------------
        self.periodic = tuple(periodic)
        (wrap_decls, lap_terms, lapc0, x_off, ring_load,
         ring_init) = _lap_stencil_pieces(h, dx, periodic)

        def _emit_tmps(items, out):
            for lhs, rhs in items:
                tname = lhs.name if hasattr(lhs, "name") else str(lhs)
                out.append(f"const double {tname} = {cg.emit(rhs)};")

        def _emit_stores(items, out):
            for lhs, rhs in items:
                val = cg.emit(rhs)
                cg.store_ctx = True
                dst = cg.emit(lhs)
                cg.store_ctx = False
                if nt:
                    out.append(
                        f"__builtin_nontemporal_store({val}, &{dst});")
                else:
                    out.append(f"{dst} = {val};")

        def _fp_decl(lin):
            return (f"    const double* fp = {f_name} + (long)fld * PVOL"
                    f" + (long)(j + H) * PSZ + (k + H);")

        def _stencil_block(lin, out):
            out.append(f"double lapv_{lin};")
            out.append("{")
            out.append(f"    const int fld = {lin};")
            out.append(_fp_decl(lin))
            out.append("    " + ring_load)
            out.append("    const double* cp = fp + (long)(i + H) * sx;")
            out.append("    double la = ring[fld][H] * LAPC0;")
            for t in lap_terms:
                out.append("    " + t)
            out.append(f"    lapv_{lin} = la;")
            out.append("}")

        def _ring_load_block(lin, out):
            out.append("{")
            out.append(f"    const int fld = {lin};")
            out.append(_fp_decl(lin))
            out.append("    " + ring_load)
            out.append("}")

        init_lines, combine_cases = [], []
        for nm, sidx in sorted((state_map or {}).items()):
            init_lines.append(f"const double st_{nm} = state[{sidx}];")
        for r, (expr, op) in enumerate(entries):
            init_lines.append(f"acc[{r}] = {_OP_INIT[op]};")
            combine_cases.append(f"(r == {r}) ? {_OP_COMBINE[op]} : ")
        combine = "".join(combine_cases) + "0.0"

        def _entry_lines(out):
            for r, (expr, op) in enumerate(entries):
                val = cg.emit(expr)
                out.append(
                    "{ const double a = acc[%d]; const double b = %s; "
                    "acc[%d] = %s; }" % (r, val, r, _OP_COMBINE[op]))

        def _vars_of(expr):
            out = set()

            def v(x):
                if isinstance(x, Variable):
                    out.add(x.name)

            walk_expr(expr, v)
            return out

        def _lap_lins(exprs):
            lins = set()

            def v(x):
                if isinstance(x, Subscript) \
                        and isinstance(x.aggregate, Field) \
                        and x.aggregate.name == lap_name and x.index \
                        and isinstance(x.index[0], (int, np.integer)):
                    lins.add(int(x.index[0]))
                elif isinstance(x, Field) and x.name == lap_name:
                    lins.add(0)

            for e in exprs:
                walk_expr(e, v)
            return lins

        def _nt_in(exprs):
            found = set()

            def v(x):
                if isinstance(x, Subscript) \
                        and isinstance(x.aggregate, Field) \
                        and len(x.index) == 1 \
                        and isinstance(x.index[0], int):
                    key = (x.aggregate.name, int(x.index[0]))
                    if key in cg.preload:
                        found.add(key)
                elif isinstance(x, Field) and not x.shape \
                        and (x.name, 0) in cg.preload:
                    found.add((x.name, 0))

            for e in exprs:
                walk_expr(e, v)
            return found

        if sectioned:
            # per-component sections: ring load + Laplacian + tmps +
            # stores for one component at a time, sched_barrier-fenced
            cg.sectioned_lapv = True
            n_sec = (nf + section_size - 1) // section_size
            sec_stores = [[] for _ in range(n_sec)]
            for (lhs, rhs), c in zip(store_items, comps):
                sec_stores[c // section_size].append((lhs, rhs))

            deps = {}
            for lhs, rhs in tmp_items:
                tname = lhs.name if hasattr(lhs, "name") else str(lhs)
                deps[tname] = _vars_of(rhs) & tmp_set
            need = {t: set() for t in tmp_set}
            for s in range(n_sec):
                frontier = set()
                for _, rhs in sec_stores[s]:
                    frontier |= _vars_of(rhs) & tmp_set
                seen = set()
                while frontier:
                    t = frontier.pop()
                    if t in seen:
                        continue
                    seen.add(t)
                    need[t].add(s)
                    frontier |= deps.get(t, set())
            pro_tmps = []
            sec_tmps = [[] for _ in range(n_sec)]
            for lhs, rhs in tmp_items:
                tname = lhs.name if hasattr(lhs, "name") else str(lhs)
                regs = need.get(tname, set())
                if len(regs) == 1:
                    sec_tmps[next(iter(regs))].append((lhs, rhs))
                else:
                    pro_tmps.append((lhs, rhs))

            region_exprs = [
                [r for _, r in sec_tmps[s]] + [r for _, r in sec_stores[s]]
                for s in range(n_sec)]
            pro_exprs = ([r for _, r in pro_tmps]
                         + [e for e, _ in entries])
            used_by = {lin: set() for lin in range(nf)}
            for s in range(n_sec):
                for lin in _lap_lins(region_exprs[s]):
                    used_by.setdefault(lin, set()).add(s)
            for lin in _lap_lins(pro_exprs):
                used_by.setdefault(lin, set()).add(-1)
            lin_place = {}
            for lin in range(nf):
                regs = used_by.get(lin, set())
                if len(regs) == 1 and -1 not in regs:
                    lin_place[lin] = next(iter(regs))
                elif regs:
                    lin_place[lin] = -1
                else:
                    lin_place[lin] = None    # ring-load only

            nt_region = {key: set() for key in nt_used}
            for s in range(n_sec):
                for key in _nt_in(region_exprs[s]):
                    nt_region[key].add(s)
            for key in _nt_in(pro_exprs):
                nt_region[key].add(-1)

            # PYSTELLA_SECTION_MASK: instruction categories allowed to
            # cross the section fences (sched_barrier mask; e.g. 0x20
            # lets VMEM reads prefetch across sections while keeping
            # arithmetic sectioned)
            mask = int(_os.environ.get("PYSTELLA_SECTION_MASK", "0"), 0)
            fence = f"__builtin_amdgcn_sched_barrier({mask});"
            body = []
            for lin in range(nf):
                if lin_place[lin] is None:
                    _ring_load_block(lin, body)
                elif lin_place[lin] == -1:
                    _stencil_block(lin, body)
            for key in sorted(k for k, v in nt_region.items()
                              if len(v) != 1 or -1 in v):
                body.append(_preload_decl(*key))
            _emit_tmps(pro_tmps, body)
            _entry_lines(body)
            for s in range(n_sec):
                body.append(fence)
                for lin in range(nf):
                    if lin_place[lin] == s:
                        _stencil_block(lin, body)
                for key in sorted(k for k, v in nt_region.items()
                                  if v == {s}):
                    body.append(_preload_decl(*key))
                _emit_tmps(sec_tmps[s], body)
                _emit_stores(sec_stores[s], body)
            body.append(fence)
            lines = body
        else:
            lines = []
            for nm, lin in sorted(nt_used):
                lines.append(_preload_decl(nm, lin))
            _emit_tmps(tmp_items, lines)
            _entry_lines(lines)

            def _lhs_name(lhs):
                f = lhs.aggregate if isinstance(lhs, Subscript) else lhs
                return getattr(f, "name", None)

            if guard_stores and guard_scalar is not None:
                # runtime-skippable stores (dead last-stage k writes,
                # see fusion.py): identical compiled form across
                # stages, one uniform scalar branch around the block
                normal = [(l, r) for l, r in store_items
                          if _lhs_name(l) not in guard_stores]
                guarded = [(l, r) for l, r in store_items
                           if _lhs_name(l) in guard_stores]
                _emit_stores(normal, lines)
                if guarded:
                    cond = cg.scalar_param(guard_scalar)
                    lines.append(f"if ({cond} != 0.0) {{")
                    _emit_stores(guarded, lines)
                    lines.append("}")
            else:
                _emit_stores(store_items, lines)
        self.sectioned = sectioned

        # pointer params: stencil field first, then every other spatial
        # field referenced by the statements or reducers, then the
        # optional device-state scalar buffer
        self.ptr_names = [f_name] + sorted(
            fa.name for fa in field_args
            if fa.spatial and fa.name not in (f_name, lap_name))
        by_name = {fa.name: fa for fa in field_args}
        self.field_args = [by_name[n] for n in self.ptr_names
                           if n in by_name]
        ptr_params = ", ".join(
            f"double* __restrict__ {n}" for n in self.ptr_names)
        if state_map:
            self.ptr_names.append("state")
            ptr_params += ", const double* __restrict__ state"
        int_params = ("const int k0b, const int k1b, const int j0b, "
                      "const int j1b, const int i0b, const int i1b, "
                      "const int nblkT, const int bid0")
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (
            ptr_params, "double* __restrict__ partials", int_params,
            dbl_params) if x)

        defines = geometry_defines(halo, rank_shape)
        defines += _tile_defines(tile, rank_shape)
        defines += f"#define MINW {min_waves}\n"
        defines += f"#define COMBINE(r, a, b) ({combine})\n"
        defines += f"#define LAPC0 ({lapc0!r})\n"
        if sectioned:
            src = LAPSTAGE_SEC_TEMPLATE.format(
                defines=defines, preamble=PREAMBLE, nred=len(entries),
                nf=nf, name=name, params=params, fname=f_name,
                init="\n    ".join(init_lines),
                wrap_decls="\n        ".join(wrap_decls),
                x_off=x_off, ring_init=ring_init,
                site_body="\n            ".join(lines))
            self._shell_parts = None
        else:
            fmt = dict(
                defines=defines, preamble=PREAMBLE, nred=len(entries),
                nf=nf, name=name, params=params, fname=f_name,
                init="\n    ".join(init_lines),
                wrap_decls="\n        ".join(wrap_decls),
                x_off=x_off, ring_load=ring_load, ring_init=ring_init,
                lap_terms="\n                ".join(lap_terms),
                body="\n            ".join(lines))
            src = LAPSTAGE_TEMPLATE.format(**fmt)
            self._shell_parts = fmt
        self._shell_cache = {}
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)
        self.grid = _tile_grid(tile, rank_shape)
        self.block = tile[0] * tile[1]
        self.nblk = self.grid[0] * self.grid[1] * self.grid[2]
        self._partials = None

    _finish = JitReduction._finish

    def _box_geometry(self, box):
        """(grid, ints-prefix) for a sub-box launch; box =
        (i0, i1, j0, j1, k0, k1) in interior coordinates."""
        tbz, tby, xchunk = self.tile
        i0, i1, j0, j1, k0, k1 = box
        grid = ((k1 - k0 + tbz - 1) // tbz,
                (j1 - j0 + tby - 1) // tby,
                (i1 - i0 + xchunk - 1) // xchunk)
        return grid, [k0, k1, j0, j1, i0, i1]

    def box_nblk(self, box):
        grid, _ = self._box_geometry(box)
        return grid[0] * grid[1] * grid[2]

    # -- multi-box shell launch (all boundary slabs in ONE kernel) ------
    def _get_shell(self, boxes):
        cached = self._shell_cache.get(boxes)
        if cached is not None:
            return cached
        if self._shell_parts is None:
            raise RuntimeError("shell form unavailable (sectioned)")
        grids = [self._box_geometry(b)[0] for b in boxes]
        nbs = [g[0] * g[1] * g[2] for g in grids]
        blk0 = [0]
        for n in nbs[:-1]:
            blk0.append(blk0[-1] + n)
        fmt = dict(self._shell_parts)
        fmt["name"] = fmt["name"] + "_shell"
        fmt["nbox"] = len(boxes)

        def ints(vals):
            return ", ".join(str(int(v)) for v in vals)

        src = LAPSTAGE_SHELL_TEMPLATE.format(
            k0s=ints(b[4] for b in boxes), k1s=ints(b[5] for b in boxes),
            j0s=ints(b[2] for b in boxes), j1s=ints(b[3] for b in boxes),
            i0s=ints(b[0] for b in boxes), i1s=ints(b[1] for b in boxes),
            gzs=ints(g[0] for g in grids), gys=ints(g[1] for g in grids),
            blk0s=ints(blk0), **fmt)
        kid = ext().jit_compile(src, fmt["name"])
        cached = (kid, sum(nbs))
        self._shell_cache[boxes] = cached
        return cached

    def shell_nblk(self, boxes):
        return self._get_shell(tuple(boxes))[1]

    def launch_shell(self, env, boxes, partials, bid0, nblk_tot):
        """ONE launch covering every box in ``boxes`` (the rank's
        boundary slabs); per-block partials land flat at ``bid0``."""
        kid, total = self._get_shell(tuple(boxes))
        ptrs = []
        for n in self.ptr_names:
            t = _check_tensor(n, env[n])
            ptrs.append(t.data_ptr())
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(kid, total, 1, 1, self.block, 1, 1,
                         self.shmem, _stream(),
                         ptrs + [partials.data_ptr()],
                         [0, 0, 0, 0, 0, 0, nblk_tot, bid0], doubles)

    def launch_box(self, env, box, partials, bid0, nblk_tot):
        """Launch over a sub-box, writing this launch's per-block
        partials at column offset ``bid0`` of the shared ``partials``
        buffer [nred, nblk_tot].  Stream-ordered, no host sync."""
        ptrs = []
        for n in self.ptr_names:
            t = _check_tensor(n, env[n])
            ptrs.append(t.data_ptr())
        grid, ints = self._box_geometry(box)
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, grid[0], grid[1], grid[2],
                         self.block, 1, 1, self.shmem, _stream(),
                         ptrs + [partials.data_ptr()],
                         ints + [nblk_tot, bid0], doubles)

    def launch_only(self, env):
        """Full-grid launch without finishing the reduction; returns the
        raw per-block partials tensor [nred, nblk] (stream-ordered, no
        host synchronization)."""
        dev = None
        for n in self.ptr_names:
            dev = _check_tensor(n, env[n]).device
        nred = len(self.entries)
        if (self._partials is None
                or self._partials.device != dev
                or self._partials.shape[1] != self.nblk):
            self._partials = torch.empty((nred, self.nblk),
                                         dtype=torch.float64, device=dev)
        nx, ny, nz = self.rank_shape
        self.launch_box(env, (0, nx, 0, ny, 0, nz), self._partials,
                        0, self.nblk)
        return self._partials

    def __call__(self, env):
        dev = None
        for n in self.ptr_names:
            dev = _check_tensor(n, env[n]).device
        self.launch_only(env)
        return self._finish(dev)


FRIEDMANN_TEMPLATE = """
#define NRED {nred}
#define NBLK {nblk}
extern "C" __global__ __launch_bounds__(256) void {name}_sums(
    const double* __restrict__ partials, double* __restrict__ sums)
{{
    __shared__ double sd[256];
    const int r = blockIdx.x;
    double acc = 0.0;
    for (int c = threadIdx.x; c < NBLK; c += 256)
        acc += partials[(long)r * NBLK + c];
    sd[threadIdx.x] = acc;
    __syncthreads();
    for (int s = 128; s > 0; s >>= 1) {{
        if ((int)threadIdx.x < s)
            sd[threadIdx.x] += sd[threadIdx.x + s];
        __syncthreads();
    }}
    if (threadIdx.x == 0) sums[r] = sd[0];
}}

extern "C" __global__ void {name}_step(
    const double* __restrict__ sums, double* __restrict__ state,
    double A, double B, double dt)
{{
    const double wt[NRED] = {{ {wt} }};
    const double wp[NRED] = {{ {wp} }};
    double E = 0.0, P = 0.0;
    #pragma unroll
    for (int r = 0; r < NRED; ++r) {{
        const double avg = sums[r] * {inv_grid!r};
        E += wt[r] * avg;
        P += wp[r] * avg;
    }}
    const double a = state[0];
    const double adot = state[1];
    const double rhs_a = adot;
    const double rhs_adot = {frw_coef!r} * a * a * (E - 3.0 * P) * a;
    const double ka = A * state[2] + dt * rhs_a;
    const double kd = A * state[3] + dt * rhs_adot;
    const double an = a + B * ka;
    const double adn = adot + B * kd;
    state[0] = an;
    state[1] = adn;
    state[2] = ka;
    state[3] = kd;
    state[4] = adn / an;
    state[5] = E;
    state[6] = P;
}}
"""


class JitFriedmann:
    """Device-resident Friedmann update: finishes the stage kernel's
    per-block energy partials into E and P and advances the 2N-storage
    (a, adot) ODE — all on the stream, no host round trip.  The state
    buffer layout is [a, adot, k_a, k_adot, hubble, energy, pressure]
    (the stage kernels read a and hubble straight from it).

    Host analogue: :class:`pystella_amd.Expansion` (which mirrors
    reference pystella/expansion.py:28-176)."""

    STATE_A, STATE_ADOT, STATE_HUBBLE = 0, 1, 4
    STATE_ENERGY, STATE_PRESSURE = 5, 6

    def __init__(self, nred, nblk, wt, wp, grid_size, mpl=1.,
                 name="friedmann"):
        import math as _math
        self.nred = nred
        self.nblk = nblk
        frw_coef = 4 * _math.pi / 3 / mpl**2
        src = FRIEDMANN_TEMPLATE.format(
            nred=nred, nblk=nblk, name=name,
            wt=", ".join(repr(float(w)) for w in wt),
            wp=", ".join(repr(float(w)) for w in wp),
            inv_grid=1.0 / grid_size, frw_coef=frw_coef)
        self.source = src
        self.key_sums = ext().jit_compile(src, name + "_sums")
        self.key_step = ext().jit_compile(src, name + "_step")

    def finish_sums(self, partials, sums):
        ext().jit_launch(self.key_sums, self.nred, 1, 1, 256, 1, 1, 0,
                         _stream(),
                         [partials.data_ptr(), sums.data_ptr()], [], [])

    def step(self, sums, state, A, B, dt):
        ext().jit_launch(self.key_step, 1, 1, 1, 1, 1, 1, 0, _stream(),
                         [sums.data_ptr(), state.data_ptr()], [],
                         [float(A), float(B), float(dt)])


def get_lap_stage_kernel(map_dict, tmp_instructions, entries, field_args,
                         scalar_names, halo, rank_shape, dx, nf,
                         f_name="f", lap_name="lap_f",
                         name="rk_lapstage", tile=None, nt=True,
                         state_map=None, periodic=(False, False, False),
                         guard_stores=frozenset(), guard_scalar=None):
    return JitLapStage(map_dict, tmp_instructions, entries, field_args,
                       scalar_names, halo, rank_shape, dx, nf,
                       f_name=f_name, lap_name=lap_name, name=name,
                       tile=tile, nt=nt, state_map=state_map,
                       periodic=periodic, guard_stores=guard_stores,
                       guard_scalar=guard_scalar)


def get_lap_reduction_kernel(entries, field_args, scalar_names, halo,
                             rank_shape, dx, nf, f_name="f",
                             lap_name="lap_f", tile=(64, 4, 64),
                             store_lap=True):
    return JitLapReduction(entries, field_args, scalar_names, halo,
                           rank_shape, dx, nf, f_name, lap_name, tile=tile,
                           store_lap=store_lap)
