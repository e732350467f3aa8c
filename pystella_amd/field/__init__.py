"""Array-aware symbolic fields for pystella_amd.

MI355X-native re-design of the reference's symbolic layer
(reference: pystella/field/__init__.py:52-652).  :class:`Field` is a leaf
expression denoting a (possibly halo-padded) rank-local grid array;
:class:`DynamicField` bundles the companion arrays (``dot``, ``lap``,
``pd``) that time steppers and finite-difference kernels exchange.

Unlike the reference — which lowers Fields to loopy subscript expressions
for OpenCL codegen — these Fields are consumed directly by

* the torch evaluator (CPU oracle / test path), which maps a Field access
  with grid shift ``(sx, sy, sz)`` to a shifted interior view of the
  padded tensor, and
* the HIP codegen (``backend/codegen.py``), which maps it to a strided
  load in a hand-written CDNA4 kernel template.
"""

from __future__ import annotations

from dataclasses import dataclass

from pystella_amd.field.expr import (  # noqa: F401
    Expr, Variable, Subscript, Sum, Product, Quotient, Power, Call,
    Comparison, If, var, is_number, is_zero,
    sin, cos, tan, exp, log, sqrt, tanh, sinh, cosh, fabs, fmin, fmax,
)

__all__ = [
    "Field", "DynamicField", "shift_fields", "substitute", "collect_fields",
    "get_field_args", "FieldArg", "diff", "Expr", "Variable", "Subscript",
    "Sum", "Product", "Quotient", "Power", "Call", "Comparison", "If", "var",
    "index_fields", "collect_field_indices", "indices_to_domain",
    "infer_field_domains",
]


def _offset_tuple(offset, nspatial):
    if isinstance(offset, (tuple, list)):
        return tuple(offset)
    return (offset,) * nspatial


class Field(Expr):
    """A symbolic leaf denoting an array over the rank-local grid.

    :arg name: The array name.
    :arg offset: ``"h"`` if the array is halo-padded (stencil reads
        allowed), ``0`` for an unpadded interior-only array.  Mirrors the
        reference's index-offset convention (pystella/field/__init__.py:145).
    :arg shape: The *outer* (non-grid) shape, e.g. ``(nscalars,)``.
    :arg indices: Spatial index names; ``()`` or ``[]`` declares a
        grid-constant scalar (kernel scalar argument).
    :arg shift: Static grid shift applied by stencil expansion
        (see :func:`shift_fields`).
    """

    __slots__ = ("name", "offset", "shape", "indices", "shift", "dtype")
    init_args = ("name", "offset", "shape", "indices", "shift", "dtype")

    def __init__(self, name, offset=0, shape=tuple(), indices=("i", "j", "k"),
                 shift=None, dtype=None):
        indices = tuple(indices)
        object.__setattr__(self, "name", name)
        object.__setattr__(self, "offset", _offset_tuple(offset, len(indices)))
        object.__setattr__(self, "shape", tuple(shape))
        object.__setattr__(self, "indices", indices)
        object.__setattr__(self, "shift",
                           tuple(shift) if shift else (0,) * len(indices))
        object.__setattr__(self, "dtype", dtype)

    def __setattr__(self, k, v):
        raise AttributeError("immutable")

    @property
    def is_spatial(self):
        return len(self.indices) > 0

    @property
    def is_padded(self):
        return any(o == "h" for o in self.offset)

    @property
    def index_tuple(self):
        """The fully-expanded subscript: each spatial index name offset
        by the field's halo offset and static shift (reference
        pystella/field/__init__.py:178-185)."""
        out = []
        for idx, off, sh in zip(self.indices, self.offset, self.shift):
            term = Variable(idx) if isinstance(idx, str) else idx
            if off == "h":
                term = term + Variable("h")
            elif off:
                term = term + off
            if sh:
                term = term + sh
            out.append(term)
        return tuple(out)

    def copy(self, **kwargs):
        init = {a: getattr(self, a) for a in Field.init_args}
        init.update(kwargs)
        return Field(**init)


class DynamicField(Field):
    """A :class:`Field` with companion arrays for its time derivative
    (``dot``), Laplacian (``lap``) and spatial gradient (``pd``).

    Mirrors reference pystella/field/__init__.py:204-298.
    """

    __slots__ = ("dot", "lap", "pd")
    init_args = Field.init_args + ("dot", "lap", "pd")

    def __init__(self, name, offset=0, shape=tuple(), indices=("i", "j", "k"),
                 shift=None, dtype=None, dot=None, lap=None, pd=None):
        super().__init__(name, offset=offset, shape=shape, indices=indices,
                         shift=shift, dtype=dtype)
        object.__setattr__(self, "dot", dot if dot is not None else Field(
            f"d{name}dt", offset=offset, shape=shape, indices=indices,
            dtype=dtype))
        object.__setattr__(self, "lap", lap if lap is not None else Field(
            f"lap_{name}", offset=0, shape=shape, indices=indices,
            dtype=dtype))
        object.__setattr__(self, "pd", pd if pd is not None else Field(
            f"d{name}dx", offset=0, shape=tuple(shape) + (3,), indices=indices,
            dtype=dtype))

    def d(self, *args):
        """``f.d(outer..., mu)``: the mu-th spacetime derivative; ``mu=0``
        maps to ``f.dot[outer]``, ``mu=i`` to ``f.pd[outer, i-1]``.
        (reference pystella/field/__init__.py:264-298)
        """
        mu = args[-1]
        outer = args[:-1]
        if mu == 0:
            return self.dot[outer] if outer else self.dot
        return self.pd[outer + (mu - 1,)]

    def copy(self, **kwargs):
        init = {a: getattr(self, a) for a in Field.init_args}
        init.update(kwargs)
        return DynamicField(**init)


# -- generic tree walking ---------------------------------------------------

def map_expr(expr, leaf_fn):
    """Rebuild ``expr`` bottom-up; ``leaf_fn`` is applied to leaves
    (Field/Variable/Subscript-aggregate untouched numbers pass through)."""
    if is_number(expr):
        return expr
    if isinstance(expr, Field) or isinstance(expr, Variable):
        return leaf_fn(expr)
    if isinstance(expr, Subscript):
        agg = map_expr(expr.aggregate, leaf_fn)
        idx = tuple(map_expr(i, leaf_fn) for i in expr.index)
        return leaf_fn(Subscript(agg, idx))
    if isinstance(expr, Sum):
        out = 0
        for c in expr.children:
            out = out + map_expr(c, leaf_fn)
        return out
    if isinstance(expr, Product):
        out = 1
        for c in expr.children:
            out = out * map_expr(c, leaf_fn)
        return out
    if isinstance(expr, Quotient):
        num = map_expr(expr.num, leaf_fn)
        den = map_expr(expr.den, leaf_fn)
        if is_number(num) and is_number(den):
            return num / den
        return Quotient(num, den)
    if isinstance(expr, Power):
        return Power(map_expr(expr.base, leaf_fn),
                     map_expr(expr.exponent, leaf_fn))
    if isinstance(expr, Call):
        return Call(expr.func, tuple(map_expr(a, leaf_fn) for a in expr.args))
    if isinstance(expr, Comparison):
        return Comparison(map_expr(expr.left, leaf_fn), expr.op,
                          map_expr(expr.right, leaf_fn))
    if isinstance(expr, If):
        return If(map_expr(expr.condition, leaf_fn),
                  map_expr(expr.then, leaf_fn),
                  map_expr(expr.else_, leaf_fn))
    raise TypeError(f"unhandled node {type(expr)}")


def walk_expr(expr, visit):
    """Call ``visit`` on every node (pre-order)."""
    if is_number(expr):
        return
    visit(expr)
    if isinstance(expr, Subscript):
        walk_expr(expr.aggregate, visit)
        for i in expr.index:
            walk_expr(i, visit)
    elif isinstance(expr, (Sum, Product)):
        for c in expr.children:
            walk_expr(c, visit)
    elif isinstance(expr, Quotient):
        walk_expr(expr.num, visit)
        walk_expr(expr.den, visit)
    elif isinstance(expr, Power):
        walk_expr(expr.base, visit)
        walk_expr(expr.exponent, visit)
    elif isinstance(expr, Call):
        for a in expr.args:
            walk_expr(a, visit)
    elif isinstance(expr, Comparison):
        walk_expr(expr.left, visit)
        walk_expr(expr.right, visit)
    elif isinstance(expr, If):
        walk_expr(expr.condition, visit)
        walk_expr(expr.then, visit)
        walk_expr(expr.else_, visit)


def iter_exprs(expressions):
    """Yield every expression contained in dicts/lists/tuples of exprs."""
    if isinstance(expressions, dict):
        for k, v in expressions.items():
            yield from iter_exprs(k)
            yield from iter_exprs(v)
    elif isinstance(expressions, (list, tuple)):
        for x in expressions:
            yield from iter_exprs(x)
    else:
        yield expressions


def shift_fields(expr, shift):
    """Shift every spatial :class:`Field` access in ``expr`` by ``shift``
    grid points (reference pystella/field/__init__.py:479)."""
    shift = tuple(shift)

    def leaf(x):
        if isinstance(x, Field) and x.is_spatial:
            return x.copy(shift=tuple(s + d for s, d in zip(x.shift, shift)))
        return x

    return map_expr(expr, leaf)


def substitute(expression, variable_assignments=None, **kwargs):
    """Substitute expressions or plain values for leaves.

    Keys may be :class:`Field`/:class:`Variable`/:class:`Subscript`
    instances or names (strings).  (reference pystella/field/__init__.py:519)
    """
    assignments = dict(variable_assignments or {})
    assignments.update(kwargs)
    by_name = {k: v for k, v in assignments.items() if isinstance(k, str)}

    def leaf(x):
        if x in assignments:
            return assignments[x]
        name = getattr(x, "name", None)
        if name is not None and name in by_name:
            return by_name[name]
        return x

    def rec(e):
        if isinstance(e, (list, tuple)):
            return type(e)(rec(x) for x in e)
        if isinstance(e, dict):
            return {rec(k): rec(v) for k, v in e.items()}
        return map_expr(e, leaf)

    return rec(expression)


def collect_fields(expressions):
    """Return the set of :class:`Field` instances in ``expressions``
    (a single expr or any nesting of dict/list/tuple)."""
    found = set()

    def visit(x):
        if isinstance(x, Field):
            found.add(x)

    for e in iter_exprs(expressions):
        walk_expr(e, visit)
    return found


@dataclass(frozen=True)
class FieldArg:
    """Inferred kernel argument descriptor for a field array.

    ``outer_shape`` holds the non-grid axes; ``padded`` says whether the
    grid axes carry a halo; ``spatial`` is False for grid-constant scalars.
    (analogue of reference get_field_args, pystella/field/__init__.py:536)
    """
    name: str
    outer_shape: tuple
    padded: bool
    spatial: bool
    dtype: object = None


def get_field_args(expressions, prepend_with=None):
    """Infer a unique :class:`FieldArg` per field name used in
    ``expressions``.  ``prepend_with`` prepends extra outer axes
    (used by multi-copy classical RK steppers)."""
    args = {}
    for f in collect_fields(expressions):
        outer = tuple(prepend_with or ()) + f.shape
        arg = FieldArg(name=f.name, outer_shape=outer,
                       padded=f.is_padded, spatial=f.is_spatial,
                       dtype=f.dtype)
        prev = args.get(f.name)
        if prev is not None and prev != arg:
            raise ValueError(
                f"inconsistent field specs for {f.name}: {prev} vs {arg}")
        args[f.name] = arg
    return sorted(args.values(), key=lambda a: a.name)


def index_fields(expr, prepend_with=None):
    """Compatibility shim for the reference API: returns the expression
    unchanged (pystella_amd consumes Fields directly; there is no
    subscript-lowering step)."""
    return expr


def collect_field_indices(expressions):
    """All spatial index names appearing in the expressions' Fields
    (reference field/__init__.py:collect_field_indices; there the
    result feeds loopy ISL domains — here geometry is compile-time, so
    this is introspection only)."""
    indices = set()
    for f in collect_fields(expressions):
        indices |= set(f.indices)
    return tuple(sorted(indices))


def indices_to_domain(indices):
    """ISL-style domain string for the given index names (reference
    analogue; informational — kernels bake geometry at compile time)."""
    names = ", ".join(indices)
    bounds = " and ".join(f"0 <= {i} < N{i}" for i in indices)
    return f"{{[{names}]: {bounds}}}" if indices else "{[]}"


def infer_field_domains(expressions):
    """Reference-API analogue (field/__init__.py:633): domain string
    inferred from the expressions' field indices."""
    return indices_to_domain(collect_field_indices(expressions))


from pystella_amd.field.diff import diff  # noqa: E402  (cycle-free)
