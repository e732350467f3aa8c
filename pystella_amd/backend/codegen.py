"""Lower pystella_amd symbolic statements to HIP C++ source.

The emitted scalar code is spliced into hand-written CDNA4 kernel
templates (grid-stride elementwise map, hierarchical reduction, LDS
histogram — see ``backend/hip.py``) and compiled with hiprtc by the
native runtime (``csrc/module.cpp``).  This replaces the reference's
loopy→OpenCL codegen (reference: pystella/elementwise.py:164-298) with
direct CDNA4 source generation: grid geometry, halos and outer shapes
are baked in as compile-time constants; only array pointers and runtime
scalars are kernel arguments.
"""

from __future__ import annotations

import numbers

from pystella_amd.field import (
    Field, Variable, Subscript, Sum, Product, Quotient, Power, Call,
    Comparison, If, is_number,
)

_C_FUNCS = {
    "sin": "sin", "cos": "cos", "tan": "tan", "exp": "exp", "log": "log",
    "sqrt": "sqrt", "tanh": "tanh", "sinh": "sinh", "cosh": "cosh",
    "fabs": "fabs", "fmin": "fmin", "fmax": "fmax", "min": "fmin",
    "max": "fmax", "round": "round",
}


def _c_double(x):
    # literal as a functional cast so that float-mode kernels
    # (``using real = float``) do not silently promote to double
    return f"real({float(x)!r})"


class Codegen:
    """Expression → C emitter with field/scalar argument discovery.

    ``field_args``: list of FieldArg (declaration order = pointer
    argument order).  Scalar parameters (plain Variables and non-spatial
    Field accesses) are collected into ``self.scalars`` in first-use
    order; each entry is ``(c_name, value_key)`` where ``value_key`` is
    ``name`` or ``(name, idx_tuple)``.
    """

    def __init__(self, field_args, halo, rank_shape, tmp_names=()):
        self.field_specs = {fa.name: fa for fa in field_args}
        self.halo = halo
        self.rank_shape = rank_shape
        self.scalars = []           # [(c_name, value_key)]
        self._scalar_index = {}
        self.tmp_names = set(tmp_names)

    def scalar_param(self, name, idx=()):
        key = (name, tuple(idx)) if idx else name
        if key in self._scalar_index:
            return self._scalar_index[key]
        c_name = name if not idx else name + "_" + "_".join(map(str, idx))
        c_name = "s_" + c_name
        # cache the real()-wrapped form so repeated references evaluate
        # in the kernel's real type consistently (scalars are passed as
        # double args; in float kernels an unwrapped later use would
        # silently promote the expression to double)
        self._scalar_index[key] = f"real({c_name})"
        self.scalars.append((c_name, key))
        return self._scalar_index[key]

    # ------------------------------------------------------------------
    def field_access(self, f: Field, outer_idx):
        if not f.is_spatial:
            return self.scalar_param(f.name, outer_idx)
        spec = self.field_specs[f.name]
        # linearize outer index over the field's outer shape
        outer_lin = 0
        for n, ix in zip(spec.outer_shape, outer_idx):
            outer_lin = outer_lin * n + int(ix)
        sx, sy, sz = f.shift
        if spec.padded:
            off = (f"((((long)(i+H+({sx})))*PSY + (j+H+({sy})))*PSZ"
                   f" + (k+H+({sz})))")
            if outer_lin:
                off = f"({outer_lin}L*PVOL + {off})"
        else:
            if any(f.shift):
                raise ValueError(
                    f"stencil shift on unpadded field {f.name}")
            off = "(((long)i*NY + j)*NZ + k)"
            if outer_lin:
                off = f"({outer_lin}L*UVOL + {off})"
        return f"{f.name}[{off}]"

    def emit(self, expr):
        if is_number(expr):
            if isinstance(expr, complex):
                raise NotImplementedError("complex JIT expressions")
            return _c_double(expr)
        if isinstance(expr, Field):
            return self.field_access(expr, ())
        if isinstance(expr, Subscript):
            agg = expr.aggregate
            idx = tuple(int(i) if is_number(i) else i for i in expr.index)
            if isinstance(agg, Field):
                return self.field_access(agg, idx)
            if isinstance(agg, Variable):
                return self.scalar_param(agg.name, idx)
            raise TypeError(f"cannot subscript {type(agg)}")
        if isinstance(expr, Variable):
            if expr.name in self.tmp_names:
                return expr.name
            return self.scalar_param(expr.name)
        if isinstance(expr, Sum):
            return "(" + " + ".join(self.emit(c) for c in expr.children) \
                + ")"
        if isinstance(expr, Product):
            return "(" + "*".join(self.emit(c) for c in expr.children) + ")"
        if isinstance(expr, Quotient):
            return f"({self.emit(expr.num)} / {self.emit(expr.den)})"
        if isinstance(expr, Power):
            base = self.emit(expr.base)
            if is_number(expr.exponent) and \
                    isinstance(expr.exponent, numbers.Integral):
                n = int(expr.exponent)
                if n == 0:
                    return "1.0"
                if 1 <= n <= 8:
                    return "ps_pow" + str(n) + f"({base})"
                if -8 <= n < 0:
                    return f"(1.0/ps_pow{-n}({base}))"
            return f"pow({base}, {self.emit(expr.exponent)})"
        if isinstance(expr, Call):
            if expr.func == "parity":
                # checkerboard color of the GLOBAL site: (i+j+k+off)&1,
                # off = parity of the rank's global start offset
                off = self.emit(expr.args[0]) if expr.args else "0"
                return f"real(((i + j + k + (int)({off})) & 1))"
            fn = _C_FUNCS.get(expr.func)
            if fn is None:
                raise NotImplementedError(f"function {expr.func}")
            return fn + "(" + ", ".join(self.emit(a)
                                        for a in expr.args) + ")"
        if isinstance(expr, Comparison):
            return (f"({self.emit(expr.left)} {expr.op} "
                    f"{self.emit(expr.right)})")
        if isinstance(expr, If):
            return (f"({self.emit(expr.condition)} ? "
                    f"{self.emit(expr.then)} : {self.emit(expr.else_)})")
        raise TypeError(f"unhandled node {type(expr)}")

    def emit_statements(self, statements, tmp_statements=None):
        """Emit tmp defs then in-order stores; returns the body string."""
        lines = []
        for lhs, rhs in (tmp_statements or {}).items():
            name = lhs.name if hasattr(lhs, "name") else str(lhs)
            self.tmp_names.add(name)
            lines.append(f"const real {name} = {self.emit(rhs)};")
        for lhs, rhs in statements.items():
            lines.append(f"{self.emit(lhs)} = {self.emit(rhs)};")
        return "\n        ".join(lines)


PREAMBLE = """
#define ps_pow1(x) (x)
template <class T> __device__ inline T ps_pow2(T x) { return x*x; }
template <class T> __device__ inline T ps_pow3(T x) { return x*x*x; }
template <class T> __device__ inline T ps_pow4(T x)
{ T y = x*x; return y*y; }
template <class T> __device__ inline T ps_pow5(T x)
{ T y = x*x; return y*y*x; }
template <class T> __device__ inline T ps_pow6(T x)
{ T y = x*x; return y*y*y; }
template <class T> __device__ inline T ps_pow7(T x)
{ T y = x*x; return y*y*y*x; }
template <class T> __device__ inline T ps_pow8(T x)
{ T y = x*x; y = y*y; return y*y; }
"""


def geometry_defines(halo, rank_shape, max_h=None, rtype="double"):
    h = max(halo) if isinstance(halo, (tuple, list)) else halo
    nx, ny, nz = rank_shape
    return f"""
using real = {rtype};
#define H {h}
#define NX {nx}
#define NY {ny}
#define NZ {nz}
#define PSX (NX + 2*H)
#define PSY ((long)(NY + 2*H))
#define PSZ ((long)(NZ + 2*H))
#define PVOL ((long)PSX*PSY*PSZ)
#define UVOL ((long)NX*NY*NZ)
"""
