"""Human-readable stringification of pystella_amd expressions."""

from __future__ import annotations

from pystella_amd.field.expr import (
    Variable, Subscript, Sum, Product, Quotient, Power, Call, Comparison, If,
    is_number,
)

# precedence levels
_P_SUM, _P_PROD, _P_POW, _P_UNARY, _P_ATOM = 1, 2, 3, 4, 5


def _paren(s, prec, outer):
    return f"({s})" if prec < outer else s


def stringify(expr, outer=0):
    if is_number(expr):
        s = repr(expr)
        return f"({s})" if (isinstance(expr, (int, float)) and expr < 0
                            and outer > _P_SUM) else s
    # Field subclasses Variable-like leaves
    from pystella_amd.field import Field
    if isinstance(expr, Field):
        base = expr.name
        if any(expr.shift):
            base += "{" + ",".join(map(str, expr.shift)) + "}"
        return base
    if isinstance(expr, Variable):
        return expr.name
    if isinstance(expr, Subscript):
        idx = ", ".join(stringify(i) for i in expr.index)
        return f"{stringify(expr.aggregate, _P_ATOM)}[{idx}]"
    if isinstance(expr, Sum):
        s = " + ".join(stringify(c, _P_SUM + 1) for c in expr.children)
        return _paren(s, _P_SUM, outer)
    if isinstance(expr, Product):
        s = "*".join(stringify(c, _P_PROD + 1) for c in expr.children)
        return _paren(s, _P_PROD, outer)
    if isinstance(expr, Quotient):
        s = (f"{stringify(expr.num, _P_PROD + 1)} / "
             f"{stringify(expr.den, _P_PROD + 1)}")
        return _paren(s, _P_PROD, outer)
    if isinstance(expr, Power):
        s = (f"{stringify(expr.base, _P_POW + 1)}**"
             f"{stringify(expr.exponent, _P_POW + 1)}")
        return _paren(s, _P_POW, outer)
    if isinstance(expr, Call):
        args = ", ".join(stringify(a) for a in expr.args)
        return f"{expr.func}({args})"
    if isinstance(expr, Comparison):
        return f"({stringify(expr.left)} {expr.op} {stringify(expr.right)})"
    if isinstance(expr, If):
        return (f"({stringify(expr.condition)} ? {stringify(expr.then)}"
                f" : {stringify(expr.else_)})")
    return repr(expr)
