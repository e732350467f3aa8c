"""Fused elementwise map kernels over the rank-local grid.

Analogue of the reference's ``ElementWiseMap`` (pystella/elementwise.py:81),
re-designed for MI355X: instead of loopy→OpenCL codegen, the statement
dict is either

* evaluated with torch tensor ops (CPU path — the test oracle), or
* lowered to HIP C++ by ``backend/codegen.py`` and spliced into a
  hand-written grid-stride CDNA4 kernel template, JIT-compiled once with
  hiprtc and cached (GPU path).

Statements execute in dict order with per-site sequential semantics;
temporaries (``tmp_instructions``) are computed first.
"""

from __future__ import annotations

import numbers

import torch

from pystella_amd.field import (
    Field, Variable, Subscript, collect_fields, get_field_args,
)
from pystella_amd.backend.torcheval import EvalContext, eval_statements

__all__ = ["ElementWiseMap"]


class ElementWiseMap:
    """Maps ``{lhs: rhs}`` statement dicts to a fused per-site kernel.

    :arg map_dict: dict of statements whose keys are :class:`Field`\\ s
        (or Subscripts thereof) and values are expressions.
    :arg tmp_instructions: dict of per-site temporaries (keys are
        :class:`Variable`\\ s with unique names) computed before the main
        statements.
    :arg halo_shape: halo padding of padded array arguments.
    :arg rank_shape: optional fixed interior shape (inferred per call
        otherwise).
    """

    def __init__(self, map_dict, tmp_instructions=None, halo_shape=0,
                 rank_shape=None, args=None, name="ew_map", **kwargs):
        self.name = name
        self.map_dict = dict(map_dict)
        self.tmp_instructions = dict(tmp_instructions or {})
        self.halo_shape = ((halo_shape,) * 3
                           if isinstance(halo_shape, numbers.Number)
                           else tuple(halo_shape))
        self.rank_shape = tuple(rank_shape) if rank_shape else None
        self.fixed_parameters = dict(kwargs.pop("fixed_parameters", {}))

        all_exprs = {**self.tmp_instructions, **self.map_dict}
        self.fields = collect_fields(all_exprs)
        self.field_args = get_field_args(all_exprs)
        self.arg_names = {f.name for f in self.fields}
        # non-field variables (scalars like dt) that need values at call
        self.scalar_names = set()
        from pystella_amd.field import walk_expr, iter_exprs
        tmp_names = {k.name for k in self.tmp_instructions}

        def visit(x):
            if isinstance(x, Variable) and not isinstance(x, Field):
                if x.name not in tmp_names:
                    self.scalar_names.add(x.name)

        for e in iter_exprs(all_exprs):
            walk_expr(e, visit)

        self._hip_kernel = None

    # ------------------------------------------------------------------
    def _infer_rank_shape(self, env):
        if not any(fa.spatial for fa in self.field_args):
            return (0, 0, 0)
        if self.rank_shape is not None:
            return self.rank_shape
        h = self.halo_shape
        for fa in self.field_args:
            if not fa.spatial:
                continue
            t = env.get(fa.name)
            if t is None or not isinstance(t, torch.Tensor):
                continue
            nx, ny, nz = t.shape[-3:]
            if fa.padded:
                return (nx - 2 * h[0], ny - 2 * h[1], nz - 2 * h[2])
            return (nx, ny, nz)
        raise ValueError("could not infer rank_shape from arguments")

    def _build_env(self, kwargs):
        env = dict(self.fixed_parameters)
        env.update(kwargs)
        missing = [n for n in (self.arg_names | self.scalar_names)
                   if n not in env]
        if missing:
            raise TypeError(f"missing kernel arguments: {sorted(missing)}")
        return env

    def __call__(self, queue=None, **kwargs):
        env = self._build_env(kwargs)
        rank_shape = self._infer_rank_shape(env)
        on_gpu = any(isinstance(v, torch.Tensor) and v.is_cuda
                     for v in env.values())
        if on_gpu:
            self._call_hip(env, rank_shape)
        else:
            ctx = EvalContext(self.halo_shape, rank_shape)
            eval_statements(self.map_dict, env, ctx,
                            tmp_statements=self.tmp_instructions)

    # ------------------------------------------------------------------
    def _call_hip(self, env, rank_shape):
        from pystella_amd.backend.hip import get_elementwise_kernel
        dtype = None
        for fa in self.field_args:
            t = env.get(fa.name)
            if isinstance(t, torch.Tensor) and fa.spatial:
                dtype = t.dtype
                break
        if dtype is None:
            dtype = torch.float64
        if self._hip_kernel is None or \
                self._hip_kernel.rank_shape != rank_shape or \
                self._hip_kernel.dtype != dtype:
            self._hip_kernel = get_elementwise_kernel(
                self.map_dict, self.tmp_instructions, self.field_args,
                sorted(self.scalar_names), self.halo_shape, rank_shape,
                name=self.name, dtype=dtype)
        self._hip_kernel(env)
