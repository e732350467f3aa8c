"""Minimal immutable symbolic-expression core for pystella_amd.

This replaces the reference's dependency on :mod:`pymbolic`
(reference: pystella/field/__init__.py) with a small, self-contained
expression tree designed for two consumers:

* ``pystella_amd.backend.torcheval`` — evaluation over torch tensors
  (the CPU oracle path and the reference implementation for tests), and
* ``pystella_amd.backend.codegen`` — emission of HIP C++ scalar code that
  is spliced into hand-written CDNA4 kernel templates and JIT-compiled
  with hiprtc on the GPU.

Expression trees may contain plain Python numbers (int/float/complex) as
children, exactly like pymbolic.  All nodes are immutable and
structurally hashable so they can key ``rhs_dict``-style dictionaries.
"""

from __future__ import annotations

import math
import numbers

__all__ = [
    "Expr", "Variable", "Subscript", "Sum", "Product", "Quotient", "Power",
    "Call", "Comparison", "If", "var", "is_zero", "is_number", "flattened_sum",
    "flattened_product",
]


def is_number(x):
    return isinstance(x, numbers.Number)


def is_zero(x):
    return is_number(x) and x == 0


def _wrapped(x):
    if isinstance(x, Expr) or is_number(x):
        return x
    if isinstance(x, str):
        return Variable(x)
    raise TypeError(f"cannot use {type(x)} in an expression")


class Expr:
    """Base class for all expression nodes.

    Subclasses define ``init_args`` (tuple of constructor attribute names)
    used for structural equality/hash and for generic reconstruction.
    """

    __slots__ = ("_hash",)
    init_args: tuple[str, ...] = ()

    def _key(self):
        return (type(self),) + tuple(getattr(self, a) for a in self.init_args)

    def __eq__(self, other):
        if self is other:
            return True
        if not isinstance(other, Expr):
            return NotImplemented
        return self._key() == other._key()

    def __ne__(self, other):
        r = self.__eq__(other)
        return NotImplemented if r is NotImplemented else not r

    def __hash__(self):
        try:
            return self._hash
        except AttributeError:
            object.__setattr__(self, "_hash", hash(self._key()))
            return self._hash

    # -- arithmetic ---------------------------------------------------------
    def __add__(self, other):
        if is_zero(other):
            return self
        return flattened_sum((self, _wrapped(other)))

    def __radd__(self, other):
        if is_zero(other):
            return self
        return flattened_sum((_wrapped(other), self))

    def __sub__(self, other):
        return self + (-_wrapped(other) if isinstance(other, Expr) else -other)

    def __rsub__(self, other):
        return _wrapped(other) + (-self)

    def __neg__(self):
        return Product((-1, self))

    def __pos__(self):
        return self

    def __mul__(self, other):
        other = _wrapped(other)
        if is_number(other):
            if other == 1:
                return self
            if other == 0:
                return 0
        return flattened_product((self, other))

    def __rmul__(self, other):
        other = _wrapped(other)
        if is_number(other):
            if other == 1:
                return self
            if other == 0:
                return 0
        return flattened_product((other, self))

    def __truediv__(self, other):
        other = _wrapped(other)
        if is_number(other) and other == 1:
            return self
        return Quotient(self, other)

    def __rtruediv__(self, other):
        return Quotient(_wrapped(other), self)

    def __pow__(self, other):
        other = _wrapped(other)
        if is_number(other):
            if other == 1:
                return self
            if other == 0:
                return 1
        return Power(self, other)

    def __rpow__(self, other):
        return Power(_wrapped(other), self)

    def __getitem__(self, index):
        if index == ():
            return self
        if not isinstance(index, tuple):
            index = (index,)
        return Subscript(self, index)

    def __lt__(self, other):
        return Comparison(self, "<", _wrapped(other))

    def __le__(self, other):
        return Comparison(self, "<=", _wrapped(other))

    def __gt__(self, other):
        return Comparison(self, ">", _wrapped(other))

    def __ge__(self, other):
        return Comparison(self, ">=", _wrapped(other))

    def eq(self, other):
        return Comparison(self, "==", _wrapped(other))

    def __str__(self):
        from pystella_amd.field.stringify import stringify
        return stringify(self)

    def __repr__(self):
        return f"{type(self).__name__}({self!s})"

    def __bool__(self):
        raise TypeError(
            "symbolic expressions have no truth value; "
            "use .eq() / comparisons to build conditionals")

    # numpy interop: make np_scalar * Expr produce an Expr, not an ndarray
    __array_priority__ = 100
    __array_ufunc__ = None


class Variable(Expr):
    __slots__ = ("name",)
    init_args = ("name",)

    def __init__(self, name):
        object.__setattr__(self, "name", name)

    def __setattr__(self, k, v):
        raise AttributeError("immutable")


def var(name):
    return Variable(name)


class _Node(Expr):
    __slots__ = ()

    def __init__(self, *args):
        for name, val in zip(self.init_args, args):
            object.__setattr__(self, name, val)

    def __setattr__(self, k, v):
        raise AttributeError("immutable")


class Subscript(_Node):
    """``aggregate[index]`` with ``index`` a tuple of ints/exprs."""
    __slots__ = ("aggregate", "index")
    init_args = ("aggregate", "index")

    def __init__(self, aggregate, index):
        if not isinstance(index, tuple):
            index = (index,)
        super().__init__(aggregate, index)

    @property
    def name(self):
        return self.aggregate.name


class Sum(_Node):
    __slots__ = ("children",)
    init_args = ("children",)


class Product(_Node):
    __slots__ = ("children",)
    init_args = ("children",)


class Quotient(_Node):
    __slots__ = ("num", "den")
    init_args = ("num", "den")


class Power(_Node):
    __slots__ = ("base", "exponent")
    init_args = ("base", "exponent")


class Call(_Node):
    """Intrinsic function call; ``func`` is a string like ``"sin"``."""
    __slots__ = ("func", "args")
    init_args = ("func", "args")

    def __init__(self, func, args):
        if not isinstance(args, tuple):
            args = (args,)
        super().__init__(func, args)


class Comparison(_Node):
    __slots__ = ("left", "op", "right")
    init_args = ("left", "op", "right")


class If(_Node):
    __slots__ = ("condition", "then", "else_")
    init_args = ("condition", "then", "else_")


def flattened_sum(children):
    new = []
    const = 0
    for c in children:
        if is_number(c):
            const += c
        elif isinstance(c, Sum):
            new.extend(c.children)
        else:
            new.append(c)
    if const != 0:
        new.append(const)
    if not new:
        return 0
    if len(new) == 1:
        return new[0]
    return Sum(tuple(new))


def flattened_product(children):
    new = []
    const = 1
    for c in children:
        if is_number(c):
            const *= c
        elif isinstance(c, Product):
            new.extend(c.children)
        else:
            new.append(c)
    if const == 0:
        return 0
    if const != 1:
        new.insert(0, const)
    if not new:
        return 1
    if len(new) == 1:
        return new[0]
    return Product(tuple(new))


# -- intrinsic math functions on expressions --------------------------------

def _make_fn(name, pyfn):
    def fn(x):
        if is_number(x):
            return pyfn(x)
        return Call(name, (x,))
    fn.__name__ = name
    return fn


sin = _make_fn("sin", math.sin)
cos = _make_fn("cos", math.cos)
tan = _make_fn("tan", math.tan)
exp = _make_fn("exp", math.exp)
log = _make_fn("log", math.log)
sqrt = _make_fn("sqrt", math.sqrt)
tanh = _make_fn("tanh", math.tanh)
sinh = _make_fn("sinh", math.sinh)
cosh = _make_fn("cosh", math.cosh)
fabs = _make_fn("fabs", abs)


def fmin(a, b):
    if is_number(a) and is_number(b):
        return min(a, b)
    return Call("fmin", (a, b))


def fmax(a, b):
    if is_number(a) and is_number(b):
        return max(a, b)
    return Call("fmax", (a, b))
