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
extern "C" __global__ __launch_bounds__(256) void {name}(
    {params})
{{
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


class JitElementwise:
    """Compiled fused per-site map over the interior grid."""

    def __init__(self, map_dict, tmp_instructions, field_args, scalar_names,
                 halo, rank_shape, name="ew_map"):
        self.rank_shape = tuple(rank_shape)
        self.field_args = [fa for fa in field_args if fa.spatial]
        cg = Codegen(field_args, halo, rank_shape)
        body = cg.emit_statements(map_dict, tmp_instructions)
        ptr_params = ", ".join(
            f"double* __restrict__ {fa.name}" for fa in self.field_args)
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (ptr_params, dbl_params) if x)
        src = ELEMENTWISE_TEMPLATE.format(
            defines=geometry_defines(halo, rank_shape), preamble=PREAMBLE,
            name=name, params=params, body=body)
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)

        total = int(np.prod(rank_shape))
        self.grid = min((total + 255) // 256, 4096)

    def __call__(self, env):
        ptrs = []
        for fa in self.field_args:
            t = _check_tensor(fa.name, env[fa.name])
            ptrs.append(t.data_ptr())
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, self.grid, 1, 1, 256, 1, 1, 0,
                         _stream(), ptrs, [], doubles)


def get_elementwise_kernel(map_dict, tmp_instructions, field_args,
                           scalar_names, halo, rank_shape):
    return JitElementwise(map_dict, tmp_instructions, field_args,
                          scalar_names, halo, rank_shape)


# ---------------------------------------------------------------------------
# JIT'd simultaneous reductions

REDUCTION_TEMPLATE = """{defines}
{preamble}
#define NRED {nred}
extern "C" __global__ __launch_bounds__(256) void {name}(
    {params})
{{
    double acc[NRED];
    {init}
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
    __shared__ double sd[256];
    for (int r = 0; r < NRED; ++r) {{
        sd[threadIdx.x] = acc[r];
        __syncthreads();
        for (int s = 128; s > 0; s >>= 1) {{
            if (threadIdx.x < s)
                sd[threadIdx.x] = COMBINE(r, sd[threadIdx.x],
                                          sd[threadIdx.x + s]);
            __syncthreads();
        }}
        if (threadIdx.x == 0)
            partials[(long)r * gridDim.x + blockIdx.x] = sd[0];
        __syncthreads();
    }}
}}
"""

_OP_INIT = {"sum": "0.0", "avg": "0.0", "prod": "1.0",
            "max": "-1.0e308", "min": "1.0e308"}
_OP_COMBINE = {"sum": "(a + b)", "avg": "(a + b)", "prod": "(a * b)",
               "max": "fmax(a, b)", "min": "fmin(a, b)"}


class JitReduction:
    """Fused multi-quantity grid reduction → per-block partials,
    finished with torch ops + one packed allreduce by the caller."""

    def __init__(self, entries, field_args, scalar_names, halo, rank_shape,
                 name="reduce_map"):
        self.rank_shape = tuple(rank_shape)
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
            f"const double* __restrict__ {fa.name}"
            for fa in self.field_args)
        dbl_params = ", ".join(f"double {c}" for c, _ in cg.scalars)
        params = ", ".join(x for x in (
            ptr_params, "double* __restrict__ partials", dbl_params) if x)

        defines = geometry_defines(halo, rank_shape)
        defines += ("#define COMBINE(r, a, b) (" + combine + ")\n")
        src = REDUCTION_TEMPLATE.format(
            defines=defines, preamble=PREAMBLE, nred=nred, name=name,
            params=params,
            init="\n    ".join(init_lines),
            body="\n        ".join(body_lines))
        self.source = src
        self.scalar_keys = [k for _, k in cg.scalars]
        self.key = ext().jit_compile(src, name)

        total = int(np.prod(rank_shape))
        self.grid = min((total + 255) // 256, 2048)
        self._partials = None

    def __call__(self, env):
        dev = None
        ptrs = []
        for fa in self.field_args:
            t = _check_tensor(fa.name, env[fa.name])
            dev = t.device
            ptrs.append(t.data_ptr())
        nred = len(self.entries)
        if (self._partials is None
                or self._partials.device != dev
                or self._partials.shape[1] != self.grid):
            self._partials = torch.empty((nred, self.grid),
                                         dtype=torch.float64, device=dev)
        doubles = [_resolve_scalar(env, k) for k in self.scalar_keys]
        ext().jit_launch(self.key, self.grid, 1, 1, 256, 1, 1, 0,
                         _stream(), ptrs + [self._partials.data_ptr()],
                         [], doubles)
        out = []
        for r, (_, op) in enumerate(self.entries):
            row = self._partials[r]
            if op in ("sum", "avg"):
                out.append(row.sum().item())
            elif op == "prod":
                out.append(row.prod().item())
            elif op == "max":
                out.append(row.max().item())
            else:
                out.append(row.min().item())
        return out


def get_reduction_kernel(entries, field_args, scalar_names, halo,
                         rank_shape):
    return JitReduction(entries, field_args, scalar_names, halo, rank_shape)


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
        src = HISTOGRAM_TEMPLATE.format(
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
# AOT stencil kernels (csrc/derivs.hip)

def _flat_fields(t, ndim_grid=3):
    """Collapse outer axes; returns (tensor_view, nf)."""
    outer = t.shape[:-ndim_grid]
    nf = int(np.prod(outer)) if outer else 1
    return t, nf


def derivs(fx, lap=None, pdx=None, pdy=None, pdz=None, halo=None, dx=None,
           h=None, stream=True):
    if len(set(halo)) != 1:
        raise NotImplementedError("GPU stencils require isotropic halo")
    _check_tensor("fx", fx)
    nxp, nyp, nzp = fx.shape[-3:]
    nx, ny, nz = nxp - 2 * h, nyp - 2 * h, nzp - 2 * h
    _, nf = _flat_fields(fx)

    def ptr(t):
        if t is None:
            return 0
        _check_tensor("out", t)
        return t.data_ptr()

    e = ext()
    want_grad = pdx is not None and pdy is not None and pdz is not None
    if lap is not None or want_grad:
        if lap is not None and not want_grad:
            e.gradlap(fx.data_ptr(), ptr(lap), 0, 0, 0, h, nx, ny, nz, nf,
                      dx[0], dx[1], dx[2], _stream())
        elif want_grad and lap is None:
            e.gradlap(fx.data_ptr(), 0, ptr(pdx), ptr(pdy), ptr(pdz), h,
                      nx, ny, nz, nf, dx[0], dx[1], dx[2], _stream())
        else:
            e.gradlap(fx.data_ptr(), ptr(lap), ptr(pdx), ptr(pdy),
                      ptr(pdz), h, nx, ny, nz, nf, dx[0], dx[1], dx[2],
                      _stream())
        return
    # single-axis derivatives
    for axis, out in enumerate((pdx, pdy, pdz)):
        if out is not None:
            e.pd(fx.data_ptr(), ptr(out), h, axis, 0, nx, ny, nz, nf,
                 dx[axis], _stream())


def divergence(vec, div, halo=None, dx=None, h=None):
    _check_tensor("vec", vec)
    _check_tensor("div", div)
    nxp, nyp, nzp = vec.shape[-3:]
    nx, ny, nz = nxp - 2 * h, nyp - 2 * h, nzp - 2 * h
    outer = vec.shape[:-4]
    e = ext()
    from itertools import product
    for s in product(*[range(n) for n in outer]):
        e.pd(vec[s][0].data_ptr(), div[s].data_ptr(), h, 0, 0,
             nx, ny, nz, 1, dx[0], _stream())
        e.pd(vec[s][1].data_ptr(), div[s].data_ptr(), h, 1, 1,
             nx, ny, nz, 1, dx[1], _stream())
        e.pd(vec[s][2].data_ptr(), div[s].data_ptr(), h, 2, 1,
             nx, ny, nz, 1, dx[2], _stream())
