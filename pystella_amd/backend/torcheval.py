"""Evaluate pystella_amd symbolic statements with torch tensor ops.

This is the reference/oracle execution path: every HIP kernel's numerics
test compares against this evaluator on CPU fp64 tensors.  It implements
the same per-site statement semantics as the fused GPU kernels:
statements execute in dict order, each left-hand side becoming visible
to subsequent statements (the low-storage RK update relies on this).
"""

from __future__ import annotations

import numbers

import torch

from pystella_amd.field import (
    Field, Variable, Subscript, Sum, Product, Quotient, Power, Call,
    Comparison, If, is_number,
)

_CALLS = {
    "sin": torch.sin, "cos": torch.cos, "tan": torch.tan,
    "exp": torch.exp, "log": torch.log, "sqrt": torch.sqrt,
    "tanh": torch.tanh, "sinh": torch.sinh, "cosh": torch.cosh,
    "fabs": torch.abs,
}


class EvalContext:
    """Grid geometry needed to slice interior views of padded arrays."""

    def __init__(self, halo, rank_shape):
        self.halo = (halo,) * 3 if isinstance(halo, int) else tuple(halo)
        self.rank_shape = tuple(rank_shape)

    def interior(self, tensor, shift=(0, 0, 0)):
        h = self.halo
        n = self.rank_shape
        sl = tuple(
            slice(h[d] + shift[d], h[d] + shift[d] + n[d]) for d in range(3))
        return tensor[(Ellipsis,) + sl]


def _as_scalar(x):
    if isinstance(x, torch.Tensor):
        return x
    try:
        import numpy as np
        if isinstance(x, np.ndarray):
            return float(x.reshape(-1)[0]) if x.size == 1 else x
    except ImportError:
        pass
    return x


def eval_field(f, outer_idx, env, ctx):
    t = env[f.name]
    if not f.is_spatial:
        t = _as_scalar(t)
        if outer_idx:
            if isinstance(t, numbers.Number):
                if outer_idx == (0,) * len(outer_idx):
                    return t
                raise IndexError(f"scalar {f.name} indexed with {outer_idx}")
            return t[outer_idx]
        return t
    if outer_idx:
        t = t[outer_idx]
    if f.is_padded:
        return ctx.interior(t, f.shift if any(f.shift) else (0, 0, 0))
    if any(f.shift):
        raise ValueError(
            f"stencil shift on unpadded field {f.name}: {f.shift}")
    return t


def eval_expr(expr, env, ctx):
    if is_number(expr):
        return expr
    if isinstance(expr, Field):
        return eval_field(expr, (), env, ctx)
    if isinstance(expr, Subscript):
        agg = expr.aggregate
        idx = tuple(int(eval_expr(i, env, ctx)) for i in expr.index)
        if isinstance(agg, Field):
            return eval_field(agg, idx, env, ctx)
        if isinstance(agg, Variable):
            return _as_scalar(env[agg.name])[idx]
        raise TypeError(f"cannot subscript {type(agg)}")
    if isinstance(expr, Variable):
        return _as_scalar(env[expr.name])
    if isinstance(expr, Sum):
        out = 0
        for c in expr.children:
            out = out + eval_expr(c, env, ctx)
        return out
    if isinstance(expr, Product):
        out = 1
        for c in expr.children:
            out = out * eval_expr(c, env, ctx)
        return out
    if isinstance(expr, Quotient):
        return eval_expr(expr.num, env, ctx) / eval_expr(expr.den, env, ctx)
    if isinstance(expr, Power):
        base = eval_expr(expr.base, env, ctx)
        expo = eval_expr(expr.exponent, env, ctx)
        return base ** expo
    if isinstance(expr, Call):
        if expr.func == "parity":
            base = getattr(ctx, "_parity", None)
            if base is None:
                n = ctx.rank_shape
                ii = torch.arange(n[0]).view(-1, 1, 1)
                jj = torch.arange(n[1]).view(1, -1, 1)
                kk = torch.arange(n[2]).view(1, 1, -1)
                base = (ii + jj + kk).to(torch.float64)
                ctx._parity = base
            off = int(eval_expr(expr.args[0], env, ctx)) if expr.args \
                else 0
            return (base + off) % 2
        args = [eval_expr(a, env, ctx) for a in expr.args]
        if all(isinstance(a, numbers.Number) for a in args):
            import math
            if expr.func == "fabs":
                return abs(args[0])
            if expr.func in ("fmin", "min"):
                return min(args)
            if expr.func in ("fmax", "max"):
                return max(args)
            if expr.func == "round":
                return float(round(args[0]))
            return getattr(math, expr.func)(*args)
        if expr.func in ("fmin", "fmax", "min", "max"):
            fn = (torch.minimum if expr.func in ("fmin", "min")
                  else torch.maximum)
            return fn(*(torch.as_tensor(x, dtype=torch.float64)
                        if isinstance(x, numbers.Number) else x
                        for x in args))
        if expr.func == "round":
            return torch.round(args[0])
        return _CALLS[expr.func](*(torch.as_tensor(a) for a in args))
    if isinstance(expr, Comparison):
        left = eval_expr(expr.left, env, ctx)
        right = eval_expr(expr.right, env, ctx)
        ops = {"<": lambda a, b: a < b, "<=": lambda a, b: a <= b,
               ">": lambda a, b: a > b, ">=": lambda a, b: a >= b,
               "==": lambda a, b: a == b, "!=": lambda a, b: a != b}
        return ops[expr.op](left, right)
    if isinstance(expr, If):
        cond = eval_expr(expr.condition, env, ctx)
        then = eval_expr(expr.then, env, ctx)
        els = eval_expr(expr.else_, env, ctx)
        if isinstance(cond, torch.Tensor):
            return torch.where(cond, torch.as_tensor(then),
                               torch.as_tensor(els))
        return then if cond else els
    raise TypeError(f"unhandled node {type(expr)}")


def _store(lhs, value, env, ctx):
    if isinstance(lhs, Subscript):
        f = lhs.aggregate
        idx = tuple(int(eval_expr(i, env, ctx)) for i in lhs.index)
    else:
        f = lhs
        idx = ()
    if not isinstance(f, Field):
        if idx:
            raise TypeError("subscripted non-Field statement targets are "
                            "not supported; use uniquely named temporaries")
        env[f.name] = value
        return
    view = eval_field(f, idx, env, ctx)
    if isinstance(view, torch.Tensor):
        if isinstance(value, torch.Tensor):
            view.copy_(value)
        else:
            view.fill_(float(value))
    else:
        # scalar field target: write back into backing array
        t = env[f.name]
        import numpy as np
        if isinstance(t, np.ndarray):
            t.reshape(-1)[0 if not idx else idx] = value
        else:
            env[f.name] = value


def eval_statements(statements, env, ctx, tmp_statements=None):
    """Execute ``tmp_statements`` (into fresh temporaries) then
    ``statements`` (into field arrays), in order.

    ``env`` maps names to tensors/scalars; temporaries are added to a
    scratch copy so the caller's env is not polluted.
    """
    scratch = dict(env)
    if tmp_statements:
        for lhs, rhs in tmp_statements.items():
            val = eval_expr(rhs, scratch, ctx)
            name = lhs.name if hasattr(lhs, "name") else str(lhs)
            scratch[name] = val
    for lhs, rhs in statements.items():
        val = eval_expr(rhs, scratch, ctx)
        _store(lhs, val, scratch, ctx)
