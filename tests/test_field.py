"""Symbolic layer tests (expression algebra, differentiation,
substitution, shifting, argument inference).  Mirrors the oracle style
of reference test/test_field.py."""

import numpy as np
import torch

import pystella_amd as ps
from pystella_amd.field import (
    Field, DynamicField, Sum, Product, collect_fields, get_field_args,
    shift_fields, substitute, diff, var,
)
from pystella_amd.backend.torcheval import EvalContext, eval_expr


def test_expr_algebra():
    x, y = var("x"), var("y")
    e = 2 * x + y * x - x / 2
    env = {"x": 3.0, "y": 4.0}
    ctx = EvalContext(0, (1, 1, 1))
    assert abs(eval_expr(e, env, ctx) - (6 + 12 - 1.5)) < 1e-14
    assert eval_expr(x**3, env, ctx) == 27
    assert eval_expr((x + 1) ** 0, env, ctx) == 1


def test_structural_equality_and_hash():
    f = Field("f", offset="h")
    g = Field("f", offset="h")
    assert f == g and hash(f) == hash(g)
    assert f[0] == g[0]
    d = {f[0]: 1, f[1]: 2}
    assert d[g[0]] == 1 and d[g[1]] == 2
    assert Field("f") != Field("g")


def test_diff_basic():
    x = var("x")
    ctx = EvalContext(0, (1, 1, 1))
    env = {"x": 2.0}
    assert eval_expr(diff(x**3, x), env, ctx) == 12
    assert eval_expr(diff(x * x * x, x), env, ctx) == 12
    e = diff(ps.field.sin(x) * x, x)
    expect = np.sin(2.) + 2 * np.cos(2.)
    assert abs(eval_expr(e, env, ctx) - expect) < 1e-14
    # second derivative
    assert eval_expr(diff(x**4, x, x), env, ctx) == 48
    # derivative wrt a subscripted leaf
    f = Field("f", offset="h", shape=(2,))
    V = f[0]**2 * f[1]
    assert str(diff(V, f[0])) != str(0)
    assert diff(f[0], f[1]) == 0


def test_dynamic_field():
    f = DynamicField("f", offset="h", shape=(2,))
    assert f.dot.name == "dfdt"
    assert f.lap.name == "lap_f"
    assert f.pd.name == "dfdx"
    assert f.d(1, 0) == f.dot[1]
    assert f.d(1, 2) == f.pd[1, 1]


def test_shift_fields_and_eval():
    f = Field("f", offset="h")
    e = shift_fields(f, (1, 0, -1))
    n = (4, 4, 4)
    h = 1
    t = torch.arange(6 ** 3, dtype=torch.float64).reshape(6, 6, 6)
    ctx = EvalContext(h, n)
    shifted = eval_expr(e, {"f": t}, ctx)
    base = eval_expr(f, {"f": t}, ctx)
    assert torch.equal(shifted, t[2:6, 1:5, 0:4])
    assert torch.equal(base, t[1:5, 1:5, 1:5])


def test_substitute():
    x, y = var("x"), var("y")
    e = x**2 + y
    e2 = substitute(e, {x: y})
    ctx = EvalContext(0, (1, 1, 1))
    assert eval_expr(e2, {"y": 3.0}, ctx) == 12
    e3 = substitute(e, x=2)
    assert eval_expr(e3, {"y": 1.0}, ctx) == 5


def test_get_field_args():
    f = DynamicField("f", offset="h", shape=(3,))
    rhs = {f[0]: f.dot[0], f.dot[0]: f.lap[0]}
    args = {a.name: a for a in get_field_args(rhs)}
    assert args["f"].padded and args["f"].outer_shape == (3,)
    assert not args["lap_f"].padded
    fields = collect_fields(rhs)
    assert {x.name for x in fields} == {"f", "dfdt", "lap_f"}


def test_flatten():
    x = var("x")
    s = x + (x + x)
    assert isinstance(s, Sum) and len(s.children) == 3
    p = x * (2 * x)
    assert isinstance(p, Product)
    assert 0 * x == 0
    assert 1 * x is x


def test_stringify():
    f = Field("f", offset="h")
    assert "f" in str(f + 1)
    assert str(var("a") * var("b"))


def test_stencil_map():
    """Stencil kernels: shifted reads of padded fields
    (analogue of reference test/test_stencil.py)."""
    import pystella_amd as ps
    from pystella_amd.field import shift_fields
    h = 1
    n = (8, 8, 8)
    f = Field("f", offset="h")
    out = Field("out", offset=0)
    # 6-point neighbor sum
    expr = sum(shift_fields(f, tuple(d * s for d in dirn))
               for dirn in ((1, 0, 0), (0, 1, 0), (0, 0, 1))
               for s in (1, -1))
    knl = ps.Stencil({out: expr}, halo_shape=h, rank_shape=n)
    t = torch.arange((8 + 2)**3, dtype=torch.float64).reshape(10, 10, 10)
    o = torch.zeros(n, dtype=torch.float64)
    knl(f=t, out=o)
    expect = (t[2:, 1:-1, 1:-1] + t[:-2, 1:-1, 1:-1]
              + t[1:-1, 2:, 1:-1] + t[1:-1, :-2, 1:-1]
              + t[1:-1, 1:-1, 2:] + t[1:-1, 1:-1, :-2])
    assert torch.allclose(o, expect)


def test_profiler_cpu():
    from pystella_amd.profiling import Profiler
    prof = Profiler(enabled=True, use_events=False)
    with prof.region("work", bytes=8):
        sum(range(1000))
    rep = prof.report()
    assert "work" in rep


def test_index_tuple():
    """Field.index_tuple: fully-expanded subscripts (reference
    field/__init__.py:178-185)."""
    import pystella_amd as ps
    from pystella_amd.field import shift_fields
    f = ps.Field("f", offset="h")
    assert [str(t) for t in f.index_tuple] == \
        ["i + h", "j + h", "k + h"]
    g = ps.Field("g", offset=0)
    assert [str(t) for t in g.index_tuple] == ["i", "j", "k"]
    s = shift_fields(f, (1, 0, -2))
    strs = [str(t) for t in s.index_tuple]
    assert "1" in strs[0] and "-2" in strs[2].replace("(-2)", "-2")
