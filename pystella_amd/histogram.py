"""On-device weighted histograms.

Analogue of reference pystella/histogram.py:33-350.  The GPU path is a
hand-written CDNA4 kernel: per-workgroup LDS bin accumulation with LDS
atomics, merged with device-scope atomics into the global histogram —
two launches (zero + accumulate) instead of the reference's in-kernel
global barrier (histogram.py:129).  Bin/weight expressions are spliced
in via hiprtc.  CPU path: torch ``index_add_`` (oracle).
"""

from __future__ import annotations

import numbers

import numpy as np
import torch

from pystella_amd.backend.torcheval import EvalContext, eval_expr
from pystella_amd.field import (
    Call, Field, Variable, collect_fields, get_field_args, iter_exprs,
    walk_expr, fabs, log,
)
from pystella_amd.reduction import Reduction

__all__ = ["Histogrammer", "FieldHistogrammer"]


def _round(x):
    return Call("round", (x,))


def _clip(expr, hi):
    return Call("max", (Call("min", (expr, hi)), 0))


class Histogrammer:
    """Computes simultaneous weighted histograms of expressions.

    :arg decomp: a :class:`DomainDecomposition`.
    :arg histograms: dict name → ``(bin_expr, weight_expr)``; the bin
        value is truncated to int (wrap in ``round`` to round).
    :arg num_bins: number of bins.
    """

    def __init__(self, decomp, histograms, num_bins, dtype=np.float64,
                 halo_shape=0, rank_shape=None, **kwargs):
        self.decomp = decomp
        self.histograms = dict(histograms)
        self.num_bins = num_bins
        self.dtype = dtype
        self.halo_shape = ((halo_shape,) * 3
                           if isinstance(halo_shape, numbers.Number)
                           else tuple(halo_shape))
        self.rank_shape = tuple(rank_shape) if rank_shape else None

        exprs = [e for pair in self.histograms.values() for e in pair]
        self.field_args = get_field_args(exprs)
        self.scalar_names = set()
        tmp = collect_fields(exprs)
        field_names = {f.name for f in tmp}

        def visit(x):
            if isinstance(x, Variable) and not isinstance(x, Field):
                if x.name not in field_names:
                    self.scalar_names.add(x.name)

        for e in iter_exprs(exprs):
            walk_expr(e, visit)
        self._hip_kernel = None

    def _infer_shapes(self, env):
        h = self.halo_shape
        for fa in self.field_args:
            if not fa.spatial:
                continue
            t = env.get(fa.name)
            if not isinstance(t, torch.Tensor):
                continue
            nx, ny, nz = t.shape[-3:]
            if fa.padded:
                return (nx - 2 * h[0], ny - 2 * h[1], nz - 2 * h[2])
            return (nx, ny, nz)
        if self.rank_shape:
            return self.rank_shape
        raise ValueError("could not infer rank_shape")

    def _local_torch(self, env, rank_shape):
        ctx = EvalContext(self.halo_shape, rank_shape)
        out = np.zeros((len(self.histograms), self.num_bins))
        for j, (name, (bin_expr, weight_expr)) in enumerate(
                self.histograms.items()):
            bins = eval_expr(bin_expr, env, ctx)
            w = eval_expr(weight_expr, env, ctx)
            if not isinstance(bins, torch.Tensor):
                bins = torch.as_tensor(float(bins)).expand(rank_shape)
            b = bins.reshape(-1).long().clamp_(0, self.num_bins - 1)
            if isinstance(w, torch.Tensor):
                wv = w.reshape(-1).to(torch.float64)
            else:
                wv = torch.full((b.numel(),), float(w), dtype=torch.float64)
            hist = torch.zeros(self.num_bins, dtype=torch.float64)
            hist.index_add_(0, b, wv)
            out[j] = hist.numpy()
        return out

    def _local_hip(self, env, rank_shape):
        from pystella_amd.backend.hip import get_histogram_kernel
        if self._hip_kernel is None or \
                self._hip_kernel.rank_shape != rank_shape:
            self._hip_kernel = get_histogram_kernel(
                list(self.histograms.values()), self.num_bins,
                self.field_args, sorted(self.scalar_names),
                self.halo_shape, rank_shape)
        return self._hip_kernel(env)

    def __call__(self, queue=None, filter_args=False, **kwargs):
        env = dict(kwargs)
        rank_shape = self._infer_shapes(env)
        on_gpu = any(isinstance(v, torch.Tensor) and v.is_cuda
                     for v in env.values())
        if on_gpu:
            hist = self._local_hip(env, rank_shape)
        else:
            hist = self._local_torch(env, rank_shape)
        full = self.decomp.allreduce(np.ascontiguousarray(hist))
        return {name: full[j]
                for j, name in enumerate(self.histograms.keys())}


class FieldHistogrammer(Histogrammer):
    """Linear- and log-binned histograms of field values with
    automatically computed bounds (reference histogram.py:210-350)."""

    def __init__(self, decomp, num_bins, dtype=np.float64, **kwargs):
        from pystella_amd.field import var
        halo_shape = kwargs.pop("halo_shape", 0)
        f = Field("f", offset="h" if halo_shape else 0)

        max_f, min_f = var("max_f"), var("min_f")
        max_log_f, min_log_f = var("max_log_f"), var("min_log_f")

        linear_bin = (f - min_f) / (max_f - min_f)
        log_bin = (log(fabs(f)) - min_log_f) / (max_log_f - min_log_f)
        histograms = {
            "linear": (_clip(linear_bin * num_bins, num_bins - 1), 1),
            "log": (_clip(log_bin * num_bins, num_bins - 1), 1),
        }
        super().__init__(decomp, histograms, num_bins, dtype,
                         halo_shape=halo_shape, **kwargs)

        reducers = {
            "max_f": [(f, "max")],
            "min_f": [(f, "min")],
            "max_log_f": [(log(fabs(f)), "max")],
            "min_log_f": [(log(fabs(f)), "min")],
        }
        self.get_min_max = Reduction(decomp, reducers,
                                     halo_shape=halo_shape, **kwargs)

    def __call__(self, f, queue=None, **kwargs):
        from itertools import product
        outer_shape = tuple(f.shape[:-3])
        slices = list(product(*[range(n) for n in outer_shape]))

        min_max_keys = set(self.get_min_max.reducers.keys())
        bounds_passed = min_max_keys.issubset(set(kwargs.keys()))

        out = {}
        for key in ("linear", "log"):
            out[key] = np.zeros(outer_shape + (self.num_bins,))
            out[key + "_bins"] = np.zeros(outer_shape + (self.num_bins + 1,))

        for s in slices:
            if not bounds_passed:
                bounds = self.get_min_max(f=f[s])
                bounds = {k: v[0] for k, v in bounds.items()}
            else:
                bounds = {k: kwargs[k][s] for k in min_max_keys}

            hists = super().__call__(f=f[s], **bounds)
            for key, val in hists.items():
                out[key][s] = val

            out["linear_bins"][s] = np.linspace(
                bounds["min_f"], bounds["max_f"], self.num_bins + 1)
            out["log_bins"][s] = np.exp(np.linspace(
                bounds["min_log_f"], bounds["max_log_f"], self.num_bins + 1))
        return out
