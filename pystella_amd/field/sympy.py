"""Sympy interop for pystella_amd expressions.

Analogue of reference pystella/field/sympy.py:40-176 (``SympyField``,
``pymbolic_to_sympy``, ``sympy_to_pymbolic``, ``simplify``): round-trip
:mod:`pystella_amd.field` expression trees to :mod:`sympy` so that
physics-model RHS expressions can be simplified symbolically before
being lowered to kernels.

Fields (and subscripted fields) are carried through sympy as
:class:`SympyField` — a ``sympy.Symbol`` subclass that remembers the
original expression leaf — so the round trip is exact for array
accesses.
"""

from __future__ import annotations

import sympy as sp

from pystella_amd.field.expr import (
    Call, Comparison, Expr, If, Power, Product, Quotient, Subscript, Sum,
    Variable, is_number,
)

__all__ = ["SympyField", "to_sympy", "from_sympy", "simplify",
           "pymbolic_to_sympy", "sympy_to_pymbolic"]


class SympyField(sp.Symbol):
    """A sympy Symbol that wraps a pystella_amd expression leaf
    (a :class:`~pystella_amd.field.Field`, a subscripted field, or a
    :class:`~pystella_amd.field.expr.Variable`), so that
    :func:`from_sympy` can restore the exact original leaf.
    (reference pystella/field/sympy.py:40-60)
    """

    def __new__(cls, leaf, name):
        obj = sp.Symbol.__new__(cls, name)
        obj.leaf = leaf
        return obj

    def __getnewargs_ex__(self):
        return (self.leaf, self.name), {}


_FUNCS = {
    "sin": sp.sin, "cos": sp.cos, "tan": sp.tan, "exp": sp.exp,
    "log": sp.log, "sqrt": sp.sqrt, "tanh": sp.tanh, "sinh": sp.sinh,
    "cosh": sp.cosh, "fabs": sp.Abs, "fmin": sp.Min, "fmax": sp.Max,
}
_FUNCS_INV = {
    sp.sin: "sin", sp.cos: "cos", sp.tan: "tan", sp.exp: "exp",
    sp.log: "log", sp.tanh: "tanh", sp.sinh: "sinh", sp.cosh: "cosh",
    sp.Abs: "fabs", sp.Min: "fmin", sp.Max: "fmax",
}

_COMPARISON = {"<": sp.Lt, "<=": sp.Le, ">": sp.Gt, ">=": sp.Ge,
               "==": sp.Eq, "!=": sp.Ne}


def _leaf_name(leaf):
    # sympy caches Symbol instances by (class, name): two structurally
    # different leaves (e.g. a padded and an unpadded Field both called
    # "f") must not collide on a cached SympyField, so the structural
    # hash is folded into the name.  Identical leaves share a symbol,
    # which is what lets sympy collect terms.
    if isinstance(leaf, Subscript):
        idx = "_".join(str(i) for i in leaf.index)
        base = f"{leaf.aggregate.name}__{idx}"
    else:
        base = leaf.name
    if isinstance(leaf, Variable):
        return base
    return f"{base}_{hash(leaf) & 0xffffffffffff:012x}"


def to_sympy(expr):
    """Convert a pystella_amd expression to a sympy expression.

    Field and Variable leaves (and field subscripts like ``f[0]``)
    become :class:`SympyField` symbols that remember the original leaf.
    """
    if is_number(expr):
        return sp.sympify(expr)
    if isinstance(expr, Subscript) or isinstance(expr, Variable):
        return SympyField(expr, _leaf_name(expr))
    if isinstance(expr, Sum):
        return sp.Add(*[to_sympy(c) for c in expr.children])
    if isinstance(expr, Product):
        return sp.Mul(*[to_sympy(c) for c in expr.children])
    if isinstance(expr, Quotient):
        return to_sympy(expr.num) / to_sympy(expr.den)
    if isinstance(expr, Power):
        return to_sympy(expr.base) ** to_sympy(expr.exponent)
    if isinstance(expr, Call):
        fn = _FUNCS.get(expr.func)
        if fn is None:
            fn = sp.Function(expr.func)
        return fn(*[to_sympy(a) for a in expr.args])
    if isinstance(expr, Comparison):
        return _COMPARISON[expr.op](to_sympy(expr.left), to_sympy(expr.right))
    if isinstance(expr, If):
        return sp.Piecewise((to_sympy(expr.then), to_sympy(expr.condition)),
                            (to_sympy(expr.else_), True))
    # Field subclasses Expr but not Subscript; it is a leaf too.
    if isinstance(expr, Expr):
        return SympyField(expr, _leaf_name(expr))
    raise TypeError(f"cannot convert {type(expr)} to sympy")


def from_sympy(expr):
    """Convert a sympy expression back to a pystella_amd expression.
    (reference pystella/field/sympy.py:131-147)
    """
    from pystella_amd.field.expr import flattened_product, flattened_sum

    if isinstance(expr, SympyField):
        return expr.leaf
    if isinstance(expr, sp.Symbol):
        return Variable(expr.name)
    if isinstance(expr, sp.Integer):
        return int(expr)
    if isinstance(expr, sp.Rational):
        return float(expr)
    if expr is sp.pi:
        return float(sp.pi)
    if isinstance(expr, sp.Float) or expr.is_number:
        if expr.is_real is False:
            return complex(expr)
        return float(expr)
    if isinstance(expr, sp.Add):
        return flattened_sum(tuple(from_sympy(a) for a in expr.args))
    if isinstance(expr, sp.Mul):
        return flattened_product(tuple(from_sympy(a) for a in expr.args))
    if isinstance(expr, sp.Pow):
        base = from_sympy(expr.base)
        exponent = from_sympy(expr.exp)
        if is_number(exponent) and exponent == -1:
            return Quotient(1, base)
        if (is_number(exponent) and float(exponent) == int(exponent)
                and abs(exponent) < 2**31):
            exponent = int(exponent)
            if exponent < 0:
                return Quotient(1, Power(base, -exponent))
        return Power(base, exponent)
    if isinstance(expr, sp.Piecewise):
        if len(expr.args) == 2 and expr.args[1].cond == sp.true:
            return If(from_sympy(expr.args[0].cond),
                      from_sympy(expr.args[0].expr),
                      from_sympy(expr.args[1].expr))
        raise ValueError(f"cannot convert Piecewise {expr}")
    if isinstance(expr, sp.Rel):
        op = {sp.Lt: "<", sp.Le: "<=", sp.Gt: ">", sp.Ge: ">=",
              sp.Eq: "==", sp.Ne: "!="}[type(expr)]
        return Comparison(from_sympy(expr.lhs), op, from_sympy(expr.rhs))
    func = type(expr)
    if func in _FUNCS_INV:
        return Call(_FUNCS_INV[func],
                    tuple(from_sympy(a) for a in expr.args))
    if func is sp.sqrt or (isinstance(expr, sp.Function)):
        return Call(str(expr.func),
                    tuple(from_sympy(a) for a in expr.args))
    raise TypeError(f"cannot convert sympy {type(expr)} back")


def simplify(expr, sympy_out=False):
    """Simplify a pystella_amd expression via sympy.
    (reference pystella/field/sympy.py:150-176)
    """
    result = sp.simplify(to_sympy(expr))
    if sympy_out:
        return result
    return from_sympy(result)


# Names matching the reference API (pystella/field/sympy.py:131-150).
pymbolic_to_sympy = to_sympy
sympy_to_pymbolic = from_sympy
