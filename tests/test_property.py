"""Property-based tests (hypothesis): random expression trees evaluate
identically through the torch evaluator before/after sympy round trips
and substitution identities."""

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from pystella_amd.backend.torcheval import EvalContext, eval_expr
from pystella_amd.field import Field, substitute, var
from pystella_amd.field.sympy import from_sympy, to_sympy

F = Field("f", offset=0, shape=(2,))
G = Field("g", offset=0)
A = var("a")
LEAVES = [F[0], F[1], G, A, 1.5, 2, 0.25]


def expr_strategy(depth=3):
    if depth == 0:
        return st.sampled_from(LEAVES)
    sub = expr_strategy(depth - 1)
    return st.one_of(
        st.sampled_from(LEAVES),
        st.tuples(sub, sub).map(lambda t: t[0] + t[1]),
        st.tuples(sub, sub).map(lambda t: t[0] * t[1]),
        st.tuples(sub, sub).map(lambda t: t[0] - t[1]),
        sub.map(lambda e: e ** 2),
    )


CTX = EvalContext(0, (4, 4, 4))


def _env():
    g = torch.Generator().manual_seed(0)
    return {
        "f": 0.5 + torch.rand((2, 4, 4, 4), dtype=torch.float64,
                              generator=g),
        "g": 0.5 + torch.rand((4, 4, 4), dtype=torch.float64,
                              generator=g),
        "a": 1.3,
    }


def _val(x):
    if isinstance(x, torch.Tensor):
        return x
    return torch.full((4, 4, 4), float(x), dtype=torch.float64)


@settings(max_examples=60, deadline=None)
@given(expr_strategy())
def test_sympy_roundtrip_preserves_value(e):
    env = _env()
    v0 = _val(eval_expr(e, env, CTX))
    rt = from_sympy(to_sympy(e))
    v1 = _val(eval_expr(rt, env, CTX))
    scale = v0.abs().max().item() + 1.0
    assert (v0 - v1).abs().max().item() < 1e-9 * scale


@settings(max_examples=60, deadline=None)
@given(expr_strategy())
def test_substitute_identity(e):
    # substituting a fresh variable that does not occur is a no-op
    env = _env()
    v0 = _val(eval_expr(e, env, CTX))
    e2 = substitute(e, {var("zzz"): 123.0})
    v1 = _val(eval_expr(e2, env, CTX))
    # rebuild may reassociate constant factors: allow fp round-off
    scale = v0.abs().max().item() + 1.0
    assert (v0 - v1).abs().max().item() < 1e-12 * scale


@settings(max_examples=40, deadline=None)
@given(expr_strategy(2), expr_strategy(2))
def test_arithmetic_consistency(x, y):
    # (x+y) and (y+x) evaluate equal; (x*y) and (y*x) too
    env = _env()
    s1 = _val(eval_expr(x + y, env, CTX))
    s2 = _val(eval_expr(y + x, env, CTX))
    assert (s1 - s2).abs().max().item() < 1e-12 * \
        (s1.abs().max().item() + 1)
    p1 = _val(eval_expr(x * y, env, CTX))
    p2 = _val(eval_expr(y * x, env, CTX))
    assert (p1 - p2).abs().max().item() < 1e-12 * \
        (p1.abs().max().item() + 1)
