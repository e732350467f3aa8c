"""Symbolic differentiation for pystella_amd expressions.

Analogue of the reference's FieldDifferentiationMapper / diff
(reference: pystella/field/diff.py:29-95), rebuilt on the local
expression core.  ``diff(expr, x)`` is the plain partial derivative of
``expr`` with respect to the leaf ``x`` (a Field, Variable, or a
Subscript thereof); extra arguments request successive derivatives.
"""

from __future__ import annotations

from pystella_amd.field.expr import (
    Expr, Variable, Subscript, Sum, Product, Quotient, Power, Call,
    flattened_sum, flattened_product, is_number,
)

__all__ = ["diff"]


_DERIVS = {
    "sin": lambda x: Call("cos", (x,)),
    "cos": lambda x: -1 * Call("sin", (x,)),
    "tan": lambda x: 1 / Call("cos", (x,)) ** 2,
    "exp": lambda x: Call("exp", (x,)),
    "log": lambda x: Quotient(1, x),
    "sqrt": lambda x: Quotient(1, 2 * Call("sqrt", (x,))),
    "tanh": lambda x: 1 - Call("tanh", (x,)) ** 2,
    "sinh": lambda x: Call("cosh", (x,)),
    "cosh": lambda x: Call("sinh", (x,)),
}


def _leaves_equal(a, b):
    return a == b


def _diff(expr, x):
    if is_number(expr):
        return 0
    if _leaves_equal(expr, x):
        return 1
    from pystella_amd.field import Field
    if isinstance(expr, (Variable, Subscript, Field)):
        # distinct leaf
        return 0
    if isinstance(expr, Sum):
        return flattened_sum(tuple(_diff(c, x) for c in expr.children
                                   if not is_number(c)))
    if isinstance(expr, Product):
        terms = []
        ch = expr.children
        for i, c in enumerate(ch):
            dc = _diff(c, x)
            if is_number(dc) and dc == 0:
                continue
            rest = ch[:i] + ch[i + 1:]
            terms.append(flattened_product((dc,) + rest))
        return flattened_sum(tuple(terms))
    if isinstance(expr, Quotient):
        dn = _diff(expr.num, x)
        dd = _diff(expr.den, x)
        if is_number(dd) and dd == 0:
            if is_number(dn) and dn == 0:
                return 0
            return Quotient(dn, expr.den)
        return Quotient(dn * expr.den - expr.num * dd, Power(expr.den, 2))
    if isinstance(expr, Power):
        db = _diff(expr.base, x)
        de = _diff(expr.exponent, x)
        terms = 0
        if not (is_number(db) and db == 0):
            terms = terms + expr.exponent * Power(expr.base,
                                                  expr.exponent - 1) * db
        if not (is_number(de) and de == 0):
            terms = terms + expr * Call("log", (expr.base,)) * de
        return terms
    if isinstance(expr, Call):
        if len(expr.args) != 1 or expr.func not in _DERIVS:
            raise NotImplementedError(
                f"derivative of {expr.func} not implemented")
        inner = expr.args[0]
        di = _diff(inner, x)
        if is_number(di) and di == 0:
            return 0
        return _DERIVS[expr.func](inner) * di
    raise TypeError(f"cannot differentiate {type(expr)}")


def diff(expr, *variables):
    """Differentiate ``expr`` successively with respect to each of
    ``variables`` (reference pystella/field/diff.py:57)."""
    result = expr
    for v in variables:
        if not isinstance(v, Expr):
            raise TypeError("differentiation variable must be an expression")
        result = _diff(result, v)
    return result
