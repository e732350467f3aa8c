"""Multigrid relaxation (smoothers): damped Jacobi / Newton iteration.

Analogue of reference pystella/multigrid/relax.py:36-373.  The
relaxation step, residual and FAS lhs-correction are fused elementwise
kernels built from the user's symbolic operator ``L(f)`` — on GPU these
go through the same hiprtc-specialized CDNA4 templates as the RK stage
kernels; the diagonal ``D = ∂L/∂f`` is derived symbolically.
"""

from __future__ import annotations

import numbers

import numpy as np

from pystella_amd.elementwise import ElementWiseMap
from pystella_amd.stencil import Stencil
from pystella_amd.field import Field, diff, fabs, var
from pystella_amd.reduction import Reduction

__all__ = ["RelaxationBase", "JacobiIterator", "NewtonIterator",
           "RedBlackIterator"]


class RelaxationBase:
    """Solves L(f) = ρ by relaxation sweeps with ping-pong temporaries
    (reference relax.py:36-320).

    :arg lhs_dict: dict mapping unknown :class:`Field`\\ s to tuples
        ``(lhs, rho)`` — the operator applied to the unknown and the
        right-hand-side Field.
    """

    def __init__(self, decomp, queue=None, lhs_dict=None, halo_shape=0,
                 dtype=np.float64, **kwargs):
        if lhs_dict is None and isinstance(queue, dict):
            lhs_dict = queue
            queue = None
        self.decomp = decomp
        self.lhs_dict = dict(lhs_dict)
        self.halo_shape = halo_shape
        h3 = ((halo_shape,) * 3 if isinstance(halo_shape, numbers.Number)
              else tuple(halo_shape))
        self._h3 = h3
        kwargs.pop("dtype", None)
        self.fixed_parameters = dict(kwargs.pop("fixed_parameters", {}))
        rank_shape = kwargs.pop("rank_shape", None)

        self.unknown_names = [f.name for f in self.lhs_dict]
        self.f_to_rho_dict = {
            f.name: rho.name for f, (_lhs, rho) in self.lhs_dict.items()}

        common = dict(halo_shape=halo_shape, rank_shape=rank_shape,
                      fixed_parameters=self.fixed_parameters)

        # relaxation step: tmp_f = step_operator(f)
        step_dict = {}
        for f, (lhs, rho) in self.lhs_dict.items():
            tmp = Field("tmp_" + f.name, offset=f.offset)
            step_dict[tmp] = self.step_operator(f, lhs, rho)
        # measured on MI355X (profiles/r02_ab2.txt): for these h=1
        # smoother kernels the plain elementwise form beats the
        # LDS-staged Stencil by ~26 % (L1 covers the 6-point reuse;
        # the per-x-plane staging + syncthreads overhead dominates)
        # - keep the fast path here, Stencil remains available
        self.stepper = ElementWiseMap(step_dict, **common)

        # residual: r_f = rho - L(f)
        residual_dict = {}
        for f, (lhs, rho) in self.lhs_dict.items():
            resid = Field("r_" + f.name, offset="h")
            residual_dict[resid] = rho - lhs
        self.residual = ElementWiseMap(residual_dict, **common)

        # FAS lhs correction: rho = r + L(f)   (on the coarse level)
        tmp_dict = {}
        lhs_corr = {}
        for i, (f, (lhs, rho)) in enumerate(self.lhs_dict.items()):
            t = var(f"tmp_lhs_{i}")
            tmp_dict[t] = lhs
            resid = Field("r_" + f.name, offset="h")
            lhs_corr[rho] = resid + t
        self.lhs_correction = ElementWiseMap(
            lhs_corr, tmp_instructions=tmp_dict, **common)

        # residual statistics (L_inf and L2)
        reducers = {}
        for name in self.unknown_names:
            resid = Field("r_" + name, offset="h")
            reducers[name] = [(fabs(resid), "max"), (resid**2, "avg")]
        self.resid_stats = Reduction(decomp, reducers,
                                     halo_shape=halo_shape, **kwargs)

    def step_operator(self, f, lhs, rho):
        raise NotImplementedError

    def step(self, queue=None, **kwargs):
        self.stepper(**kwargs)

    @property
    def _star_operator(self):
        """True when every stencil read in the operators is star-shaped
        (at most one nonzero shift component) — then the smoothing
        loop's intermediate halo shares only need faces, not corners."""
        star = getattr(self, "_star_cached", None)
        if star is None:
            from pystella_amd.field import collect_fields
            exprs = [lhs for (lhs, _rho) in self.lhs_dict.values()]
            star = all(
                sum(1 for s in fld.shift if s) <= 1
                for fld in collect_fields(exprs))
            self._star_cached = star
        return star

    def _smoother_share(self, decomp, arr, final=False):
        """Halo share inside the smoothing loop.  Intermediate shares
        of a star operator on a fully-local decomposition go through
        the single-launch fused face wrap (backend/hip.py wrap_star, as
        in the fused RK loop) — on the coarse levels the sequential
        12-copy wrap is pure launch latency.  The FINAL share is always
        a full (corner-propagating) share_halos: downstream restriction
        reads 27-point corner halos."""
        import torch
        if (not final and self._star_operator
                and all(p == 1 for p in decomp.proc_shape)
                and len(set(self._h3)) == 1 and self._h3[0] > 0
                and isinstance(arr, torch.Tensor) and arr.is_cuda):
            from pystella_amd.backend.hip import wrap_star
            wrap_star(arr, self._h3[0], (0, 1, 2))
        else:
            decomp.share_halos(arr)

    def __call__(self, decomp, queue=None, iterations=100, **kwargs):
        """Run ``iterations`` (rounded up to even) relaxation sweeps,
        ping-ponging each unknown with its ``tmp_`` array and sharing
        halos after every sweep (reference relax.py:164-200)."""
        kwargs.pop("solve_constraint", None)
        even_iterations = iterations + (iterations % 2)
        for it in range(even_iterations):
            final = it == even_iterations - 1
            self.stepper(**kwargs)
            for name in self.unknown_names:
                kwargs[name], kwargs["tmp_" + name] = \
                    kwargs["tmp_" + name], kwargs[name]
                self._smoother_share(decomp, kwargs[name], final=final)

    def get_error(self, queue=None, **kwargs):
        """L∞ and L2 norms of the residual per unknown
        (reference relax.py:225-266)."""
        self.residual(**kwargs)
        f0 = kwargs[self.unknown_names[0]]
        h3 = self._h3
        rank_shape = tuple(n - 2 * hh
                           for n, hh in zip(f0.shape[-3:], h3))
        grid_size = float(np.prod(self.decomp.proc_shape)
                          * np.prod(rank_shape))
        self.resid_stats.grid_size = grid_size
        errs = self.resid_stats(**kwargs)
        for k, v in errs.items():
            errs[k][1] = v[1] ** 0.5
        return errs

    # -- integral-constraint machinery ---------------------------------
    # (reference relax.py:268-320; the reference's solve_constraint
    # raises NotImplementedError("constraint solving untested") — this
    # implementation actually solves the shift)

    def _ensure_constraint_kernels(self):
        if hasattr(self, "_shifter"):
            return
        common = dict(halo_shape=self.halo_shape,
                      fixed_parameters=self.fixed_parameters)
        f = Field("f", offset="h")
        tmp = Field("tmp", offset="h")
        self._shifter = ElementWiseMap(
            {tmp: var("scale") * f + var("shift")}, **common)
        from pystella_amd.field import map_expr
        avg_reducers = {}
        for fld, (lhs, rho) in self.lhs_dict.items():

            def rename(x, name=fld.name):
                # shift-preserving rename: stencil reads f(±s) must
                # become tmp_f(±s), not the unshifted center value
                if isinstance(x, Field) and x.name == name:
                    return x.copy(name="tmp_" + name)
                return x

            avg_reducers[fld.name] = [(map_expr(lhs, rename) - rho,
                                       "avg")]
        self._avg_resid = Reduction(self.decomp, avg_reducers,
                                    halo_shape=self.halo_shape)

    def eval_constraint(self, queue=None, shifts=None, scales=None,
                        **kwargs):
        """⟨L(scale·f + shift) − ρ⟩ per unknown, evaluated on shifted
        copies in the ``tmp_`` arrays (reference relax.py:280-298)."""
        if shifts is None and queue is not None:
            shifts = queue
        self._ensure_constraint_kernels()
        for name, shift, scale in zip(self.unknown_names,
                                      np.atleast_1d(shifts),
                                      np.atleast_1d(scales)):
            self._shifter(f=kwargs[name], tmp=kwargs["tmp_" + name],
                          shift=float(shift), scale=float(scale))
            self.decomp.share_halos(kwargs["tmp_" + name])
        f0 = kwargs[self.unknown_names[0]]
        rank_shape = tuple(n - 2 * hh
                           for n, hh in zip(f0.shape[-3:], self._h3))
        self._avg_resid.grid_size = float(
            np.prod(self.decomp.proc_shape) * np.prod(rank_shape))
        out = self._avg_resid(**kwargs)
        return np.array([out[n][0] for n in self.unknown_names])

    def solve_constraint(self, queue=None, **kwargs):
        """Find per-unknown constant shifts so the volume-averaged
        residual vanishes (⟨L(f + s) − ρ⟩ = 0) and apply them in
        place.  Fixes the additive nullspace of pure-Neumann/periodic
        problems (the reference declares this API but its
        implementation is disabled, relax.py:301-320)."""
        from scipy.optimize import root
        self._ensure_constraint_kernels()
        n = len(self.unknown_names)

        def F(shifts):
            return self.eval_constraint(shifts=shifts,
                                        scales=np.ones(n), **kwargs)

        sol = root(F, np.zeros(n), method="hybr", tol=1e-14)
        if not sol.success:
            raise RuntimeError(f"constraint solve failed: {sol.message}")
        for name, shift in zip(self.unknown_names,
                               np.atleast_1d(sol.x)):
            kwargs[name] += float(shift)
        return np.atleast_1d(sol.x)


class JacobiIterator(RelaxationBase):
    """Damped Jacobi: f ← (1−ω) f + ω D⁻¹ (ρ − (L−D) f)
    (reference relax.py:323-349)."""

    def step_operator(self, f, lhs, rho):
        D = diff(lhs, f)
        R_y = lhs - D * f  # valid for linear operators
        omega = var("omega")
        return (1 - omega) * f + omega * (rho - R_y) / D


class NewtonIterator(RelaxationBase):
    """Newton iteration: f ← f − ω (L(f) − ρ) / (∂L/∂f)
    (reference relax.py:352-373)."""

    def step_operator(self, f, lhs, rho):
        D = diff(lhs, f)
        omega = var("omega")
        return f - omega * (lhs - rho) / D


class RedBlackIterator(RelaxationBase):
    """Red-black Gauss–Seidel smoother: two masked in-place half-sweeps
    per iteration, updating one checkerboard color from the freshly
    updated other color.  Converges roughly twice as fast per sweep as
    damped Jacobi and needs no ping-pong temporary.

    Valid for odd-offset (h=1) star stencils only — same-color
    neighbors would race the in-place update otherwise.  No analogue in
    the reference (its smoothers are Jacobi/Newton, relax.py:323-373);
    this is an MI355X-motivated addition (in-place halves the HBM
    write+swap traffic of the smoothing loop).
    """

    def __init__(self, decomp, queue=None, lhs_dict=None, halo_shape=0,
                 **kwargs):
        super().__init__(decomp, queue=queue, lhs_dict=lhs_dict,
                         halo_shape=halo_shape, **kwargs)
        h3 = self._h3
        if max(h3) != 1:
            raise ValueError("RedBlackIterator requires halo_shape=1 "
                             "(odd-offset star stencils)")
        from pystella_amd.field import Call, Comparison, If
        common = dict(halo_shape=halo_shape,
                      rank_shape=getattr(self, "rank_shape", None),
                      fixed_parameters=self.fixed_parameters)
        off = var("rb_off")
        self.color_steppers = []
        for color in (0, 1):
            step_dict = {}
            for f, (lhs, rho) in self.lhs_dict.items():
                gs = self.step_operator(f, lhs, rho)
                cond = Comparison(Call("parity", (off,)), "==",
                                  float(color))
                step_dict[f] = If(cond, gs, f)
            self.color_steppers.append(
                ElementWiseMap(step_dict, name=f"rbgs_color{color}",
                               **common))

    def step_operator(self, f, lhs, rho):
        D = diff(lhs, f)
        omega = var("omega")
        return f - omega * (lhs - rho) / D

    def __call__(self, decomp, queue=None, iterations=100, **kwargs):
        kwargs.pop("solve_constraint", None)
        if "rb_off" not in kwargs:
            # checkerboard color is of the GLOBAL site: offset the
            # rank-local (i+j+k) parity by the parity of this rank's
            # global start so colors agree across rank seams
            rb_off = 0.0
            if decomp is not None and decomp.grid_shape is not None:
                _, start = decomp.get_rank_shape_start(decomp.grid_shape)
                rb_off = float(sum(start) % 2)
            kwargs["rb_off"] = rb_off
        if self._graph_call(decomp, iterations, kwargs):
            return
        for it in range(iterations):
            for ci, stepper in enumerate(self.color_steppers):
                final = (it == iterations - 1
                         and ci == len(self.color_steppers) - 1)
                stepper(**{k: v for k, v in kwargs.items()
                           if not k.startswith("tmp_")})
                for name in self.unknown_names:
                    self._smoother_share(decomp, kwargs[name],
                                         final=final)

    # -- hipGraph fast path --------------------------------------------
    # The in-place two-color loop is launch-latency bound on coarse
    # levels (each sweep = 2 stepper kernels + 2 fused wraps, all
    # in-place with by-value scalars — no host work between launches).
    # Capture the whole nu-iteration loop once per (level arrays,
    # iterations) into a hipGraph and replay it as ONE launch.  The
    # final corner-propagating share runs outside the graph.
    # Measured NEUTRAL at 1024^3 fp32 once the fused wrap landed
    # (0.337 vs 0.335 s/V-cycle, profiles/r02_mg_graph_ab.txt) - the
    # wrap fusion already removed the launch-latency tail - so this
    # stays opt-in: PYSTELLA_MG_GRAPH=1.
    def _graph_call(self, decomp, iterations, kwargs):
        import os
        import torch
        if (os.environ.get("PYSTELLA_MG_GRAPH", "0") != "1"
                or not self._star_operator
                or not all(p == 1 for p in decomp.proc_shape)
                or len(set(self._h3)) != 1 or self._h3[0] <= 0):
            return False
        tensors = {k: v for k, v in kwargs.items()
                   if isinstance(v, torch.Tensor)}
        if not tensors or not all(v.is_cuda for v in tensors.values()):
            return False
        scalars = tuple(sorted(
            (k, tuple(np.asarray(v, dtype=float).reshape(-1).tolist()))
            for k, v in kwargs.items()
            if isinstance(v, (int, float, np.floating, np.ndarray))
            and np.asarray(v).size >= 1))
        key = (iterations, scalars, tuple(sorted(
            (k, v.data_ptr(), tuple(v.shape))
            for k, v in tensors.items())))
        cache = getattr(self, "_graphs", None)
        if cache is None:
            cache = self._graphs = {}
        graph = cache.get(key)
        if graph is None:
            from pystella_amd.backend.hip import wrap_star
            step_kwargs = {k: v for k, v in kwargs.items()
                           if not k.startswith("tmp_")}

            def body():
                for _ in range(iterations):
                    for stepper in self.color_steppers:
                        stepper(**step_kwargs)
                        for name in self.unknown_names:
                            wrap_star(kwargs[name], self._h3[0],
                                      (0, 1, 2))
            try:
                # warm up on a side stream with CLONED state (compiles
                # and caches every kernel so capture sees only
                # launches, without mutating the real arrays)
                warm_kwargs = {
                    k: (v.clone() if isinstance(v, torch.Tensor) else v)
                    for k, v in step_kwargs.items()}
                s = torch.cuda.Stream()
                s.wait_stream(torch.cuda.current_stream())
                with torch.cuda.stream(s):
                    for stepper in self.color_steppers:
                        stepper(**warm_kwargs)
                    for name in self.unknown_names:
                        wrap_star(warm_kwargs[name], self._h3[0],
                                  (0, 1, 2))
                torch.cuda.current_stream().wait_stream(s)
                torch.cuda.synchronize()
                del warm_kwargs
                graph = torch.cuda.CUDAGraph()
                with torch.cuda.graph(graph):
                    body()
            except Exception:
                cache[key] = False
                return False
            cache[key] = graph
        elif graph is False:
            return False
        graph.replay()
        for name in self.unknown_names:
            decomp.share_halos(kwargs[name])
        return True
