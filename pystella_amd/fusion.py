"""Cross-component kernel fusion (MI355X-specific optimizations).

These components have no analogue in the reference — they exist because
on MI355X the scalar-preheating hot loop is HBM-bandwidth bound, and the
reference's structure (separate stencil pass + separate energy
reduction, examples/scalar_preheating.py:258-271) re-reads f and lap_f
from HBM each stage.  :class:`FusedLaplacianReduction` runs the
Laplacian stencil and the energy reductions in one pass.

Numerics are identical to the unfused path (same stencil coefficients,
same per-site expressions; only the reduction's accumulation grouping
changes, which for fp64 sums is within rounding).  The CPU path simply
composes :class:`~pystella_amd.FiniteDifferencer` +
:class:`~pystella_amd.Reduction` (and is the oracle for the GPU tests).
"""

from __future__ import annotations

import os

import numpy as np
import torch

from pystella_amd.reduction import Reduction

__all__ = ["FusedLaplacianReduction", "StencilRKStepper",
           "DeviceFriedmannLoop"]


class FusedLaplacianReduction(Reduction):
    """Computes ``lap_f = ∇² f`` (all outer components) *and* the given
    reductions — whose expressions may reference both ``f`` and the
    freshly computed ``lap_f`` — in one fused GPU pass.

    Call semantics combine ``derivs(fx=f, lap=lap_f)`` (including the
    halo exchange) followed by ``Reduction.__call__``.

    :arg derivs: a :class:`~pystella_amd.FiniteDifferencer` (supplies
        dx, halo and the CPU oracle path).
    :arg f_name/lap_name: names of the stencil field and its Laplacian
        in the reduction expressions.
    """

    def __init__(self, decomp, input, derivs, f_name="f", lap_name="lap_f",
                 store_lap=True, share_halos=True, **kwargs):
        super().__init__(decomp, input, **kwargs)
        self.derivs = derivs
        self.f_name = f_name
        self.lap_name = lap_name
        self.store_lap = store_lap
        self.share = share_halos
        self._fused_kernel = None

    def __call__(self, queue=None, filter_args=False, **kwargs):
        f = kwargs[self.f_name]
        if self.share:
            self.decomp.share_halos(f)
        if not (isinstance(f, torch.Tensor) and f.is_cuda):
            # CPU oracle: unfused compose (lap into a scratch array if
            # the caller does not keep one)
            lap = kwargs.get(self.lap_name)
            if lap is None:
                shape = f.shape[:-3] + tuple(
                    n - 2 * hh for n, hh in zip(f.shape[-3:],
                                                self.halo_shape))
                lap = torch.zeros(shape, dtype=f.dtype, device=f.device)
                kwargs[self.lap_name] = lap
            from itertools import product
            for s in product(*[range(n) for n in f.shape[:-3]]):
                self.derivs._apply_lap_cpu(f[s], lap[s])
            return super().__call__(**kwargs)

        rank_shape = self._infer_shapes(kwargs)
        if self._fused_kernel is None or \
                self._fused_kernel.rank_shape != rank_shape:
            from pystella_amd.backend.hip import get_lap_reduction_kernel
            nf = int(np.prod(f.shape[:-3])) if f.dim() > 3 else 1
            self._fused_kernel = get_lap_reduction_kernel(
                [(expr, op) for _, _, expr, op in self.flat],
                self.field_args, sorted(self.scalar_names),
                self.halo_shape, rank_shape, self.derivs.dx, nf,
                self.f_name, self.lap_name, store_lap=self.store_lap)
        local = self._fused_kernel(kwargs)
        return self._combine(local, rank_shape)


class _StageRedMap:
    """One RK stage kernel fused with input-state reductions.

    GPU: a single JIT kernel (``backend.hip.JitStageReduction``) whose
    per-site order is temporaries (incl. the inline Laplacian) →
    reduction accumulation → update stores, so the reducers see the
    stage's input state.  CPU oracle: the unfused compose — Laplacian
    scratch via the FiniteDifferencer, torch reduction, then the stage
    map.  Returns ``(local_reduction_values, rank_shape)``.
    """

    def __init__(self, map_dict, tmp_instructions=None, red_entries=(),
                 reduction=None, derivs=None, lap_names=(),
                 ring=None, guard_kstores=frozenset(), **kwargs):
        from pystella_amd.elementwise import ElementWiseMap
        self.guard_kstores = frozenset(guard_kstores)
        self._map = ElementWiseMap(map_dict, tmp_instructions, **kwargs)
        self.red_entries = list(red_entries)
        self.reduction = reduction
        self.derivs = derivs
        self.lap_names = list(lap_names)
        # ring = list of (rk_orig, tmp_orig, red_entries_orig, f_name,
        # nf), one per stencil family: the pre-substitution statements
        # (lap accesses intact) for the register-ring GPU kernels
        self.ring = ring
        if ring is not None:
            from pystella_amd.field import get_field_args
            self._ring_field_args = [
                get_field_args([tmp_o, rk_o, [e for e, _ in red_o]])
                for rk_o, tmp_o, red_o, _, _ in ring]
        # extend argument discovery with the reducer expressions
        from pystella_amd.field import (
            Field, Variable, collect_fields, get_field_args, iter_exprs,
            walk_expr)
        exprs = [e for e, _ in self.red_entries]
        everything = [self._map.tmp_instructions, self._map.map_dict,
                      exprs]
        self._map.fields = collect_fields(everything)
        self._map.field_args = get_field_args(everything)
        self._map.arg_names = {f.name for f in self._map.fields}
        tmp_names = {k.name for k in self._map.tmp_instructions}
        scal = set(self._map.scalar_names)

        def visit(x):
            if isinstance(x, Variable) and not isinstance(x, Field):
                if x.name not in tmp_names:
                    scal.add(x.name)

        for e in iter_exprs(exprs):
            walk_expr(e, visit)
        self._map.scalar_names = scal
        self._hip_kernel = None
        self._kernel_shape = None

    def __call__(self, queue=None, **kwargs):
        import torch as _torch
        m = self._map
        env = m._build_env(kwargs)
        rank_shape = m._infer_rank_shape(env)
        on_gpu = any(isinstance(v, _torch.Tensor) and v.is_cuda
                     for v in env.values())
        if on_gpu:
            if self._hip_kernel is None or \
                    self._kernel_shape != rank_shape:
                self._kernel_shape = rank_shape
                if self.ring is not None:
                    from pystella_amd.backend.hip import (
                        get_lap_stage_kernel)
                    self._hip_kernel = [
                        get_lap_stage_kernel(
                            rk_o, tmp_o,
                            red_o or [(0.0, "sum")],
                            fargs, [], m.halo_shape, rank_shape,
                            self.derivs.dx, nf, f_name=f_name,
                            lap_name=f"lap_{f_name}",
                            name=f"{m.name}_{f_name}",
                            guard_stores=self.guard_kstores,
                            guard_scalar="store_k")
                        for (rk_o, tmp_o, red_o, f_name, nf), fargs
                        in zip(self.ring, self._ring_field_args)]
                else:
                    from pystella_amd.backend.hip import (
                        get_stage_reduction_kernel)
                    self._hip_kernel = get_stage_reduction_kernel(
                        m.map_dict, m.tmp_instructions, self.red_entries,
                        m.field_args, sorted(m.scalar_names),
                        m.halo_shape, rank_shape, name=m.name)
            if self.ring is not None:
                local = None
                for kern, (_, _, red_o, _, _) in zip(self._hip_kernel,
                                                     self.ring):
                    out = kern(env)
                    if red_o:
                        local = out
                return local, rank_shape
            local = self._hip_kernel(env)
            return local, rank_shape
        # CPU oracle: reduction of the input state with a lap scratch,
        # then the stage update
        local = None
        if self.reduction is not None:
            from itertools import product
            env2 = dict(env)
            for lap_name in self.lap_names:
                f_name = lap_name[len("lap_"):]
                f = env[f_name]
                shape = f.shape[:-3] + tuple(rank_shape)
                lap = torch.zeros(shape, dtype=f.dtype, device=f.device)
                for s in product(*[range(n) for n in f.shape[:-3]]):
                    self.derivs._apply_lap_cpu(f[s], lap[s])
                env2[lap_name] = lap
            local = self.reduction._local_torch(env2, rank_shape)
        from pystella_amd.backend.torcheval import (
            EvalContext, eval_statements)
        ctx = EvalContext(m.halo_shape, rank_shape)
        eval_statements(m.map_dict, env, ctx,
                        tmp_statements=m.tmp_instructions)
        return local, rank_shape


class StencilRKStepper:
    """Low-storage RK stepper whose stage kernels evaluate the Laplacian
    *inline* from the finite-difference stencil instead of reading a
    precomputed ``lap_f`` array (MI355X traffic optimization: the lap
    array never exists in HBM — per stage this removes one full
    write+read pass per unknown vs the reference structure,
    examples/scalar_preheating.py:258-271).

    Fields whose ``.lap`` appears in the equations of motion are
    double-buffered (the stencil reads ``f``; the update writes
    ``f_next``) to avoid intra-kernel races; call :meth:`swap` names
    after each stage.  All other unknowns update in place.

    Usage::

        stepper = StencilRKStepper(ps.LowStorageRK54, [sector], derivs,
                                   halo_shape=h, rank_shape=shape, dt=dt)
        arrays = {"f": f, "f_next": f_next, "dfdt": dfdt}
        for s in range(stepper.num_stages):
            stepper(s, a=a, hubble=hub, **arrays)
            arrays["f"], arrays["f_next"] = arrays["f_next"], arrays["f"]
            decomp.share_halos(arrays["f"])
            # ... energy reduction on arrays["f"] ...
    """

    def __init__(self, Stepper, input, derivs, halo_shape=0,
                 rank_shape=None, dt=None, reducers=None, grid_size=None,
                 callback=None, inline_grad=False, **kwargs):
        from pystella_amd.field import (
            DynamicField, Field, Subscript, substitute, collect_fields,
            var)
        from pystella_amd.derivs import _LAP_COEFS, centered_diff
        from pystella_amd.step import LowStorageRKStepper, _field_of
        from pystella_amd.sectors import Sector

        if not issubclass(Stepper, LowStorageRKStepper):
            raise TypeError("StencilRKStepper requires a low-storage "
                            "Stepper")
        if isinstance(input, Sector):
            rhs_dict = dict(input.rhs_dict)
        elif isinstance(input, list):
            rhs_dict = {}
            for s in input:
                rhs_dict.update(s.rhs_dict)
        else:
            rhs_dict = dict(input)

        h = max(derivs.halo_shape)
        dx = derivs.dx
        coefs = _LAP_COEFS[h]

        # find DynamicFields whose .lap is referenced; compute each
        # Laplacian component once per site into a named temporary, and
        # substitute that temporary for lap accesses in the equations of
        # motion AND (below) the energy reducers; build the ping-pong
        # name map
        fields = collect_fields(list(rhs_dict.values()))
        if reducers is not None:
            red_input = (reducers.reducers
                         if isinstance(reducers, Sector) else reducers)
            red_exprs = []
            for v in red_input.values():
                red_exprs.extend(v if isinstance(v, list) else [v])
            fields |= collect_fields(
                [e[0] if isinstance(e, tuple) else e for e in red_exprs])
        lap_names = {f.name for f in fields if f.name.startswith("lap_")}
        self.pingpong = []
        subs = {}
        lap_tmps = {}
        for key in rhs_dict:
            f, outer = _field_of(key)
            lap_name = f"lap_{f.name}"
            if lap_name in lap_names:
                self.pingpong.append(f.name)
                lap_f = Field(lap_name, offset=0, shape=f.shape,
                              indices=f.indices)
                for fld in range(f.shape[0] if f.shape else 1):
                    access = f[fld] if f.shape else f
                    lap_expr = sum(
                        centered_diff(access, coefs, direction=mu + 1,
                                      order=2) * (1.0 / dx[mu] ** 2)
                        for mu in range(3))
                    lap_acc = lap_f[fld] if f.shape else lap_f
                    lv = var(f"lapv_{f.name}_{fld}")
                    lap_tmps[lv] = lap_expr
                    subs[lap_acc] = lv
        self.pingpong = sorted(set(self.pingpong))

        # optionally inline spatial gradients of stencil unknowns
        # (e.g. the GW stress-tensor source ∂_i f ∂_j f): the pd
        # companion array never exists in HBM and the per-stage
        # gradient pass disappears.  Same stencil coefficients as
        # FiniteDifferencer (derivs.py _GRAD_COEFS).
        grad_tmps = {}
        subs_grad = {}
        self.inline_grad = bool(inline_grad)
        if inline_grad:
            from pystella_amd.derivs import _GRAD_COEFS
            gcoefs = _GRAD_COEFS[h]
            referenced = {f.name for f in fields}
            seen = set()
            for key in rhs_dict:
                kf, _ = _field_of(key)
                if not isinstance(kf, DynamicField) or kf.name in seen:
                    continue
                seen.add(kf.name)
                if kf.pd.name not in referenced:
                    continue
                for fld in range(kf.shape[0] if kf.shape else 1):
                    access = kf[fld] if kf.shape else kf
                    for mu in range(3):
                        gexpr = centered_diff(
                            access, gcoefs, direction=mu + 1,
                            order=1) * (1.0 / dx[mu])
                        gv = var(f"gradv_{kf.name}_{fld}_{mu}")
                        grad_tmps[gv] = gexpr
                        pd_acc = (kf.pd[fld, mu] if kf.shape
                                  else kf.pd[mu])
                        subs_grad[pd_acc] = gv
            if subs_grad:
                # the group ("orig") statements keep lap accesses but
                # use the inlined gradients
                rhs_dict = {k: substitute(v, subs_grad)
                            for k, v in rhs_dict.items()}
        self._grad_tmps = grad_tmps

        new_rhs = {k: substitute(v, subs) for k, v in rhs_dict.items()}

        # fused input-state reducers (energy each RK stage without a
        # separate lap+reduction pass)
        self._reduction = None
        red_entries = []
        if reducers is not None:
            self._reduction = Reduction(
                derivs.decomp, reducers, halo_shape=halo_shape,
                rank_shape=rank_shape, grid_size=grid_size,
                callback=callback)
            red_entries = [(substitute(expr, subs), op)
                           for _, _, expr, op in self._reduction.flat]
        self._derivs = derivs
        self._lap_names = sorted(lap_names)

        # redirect writes of ping-ponged fields to NAME_next
        reduction = self._reduction
        lap_name_list = self._lap_names
        derivs_ref = derivs
        rhs_dict_orig = dict(rhs_dict)
        red_entries_orig = ([(expr, op)
                             for _, _, expr, op in reduction.flat]
                            if reduction is not None else [])

        # register-ring kernel grouping: partition the unknowns by
        # stencil family (a DynamicField and its .dot companion), one
        # ring kernel per family per stage; the energy reducers ride
        # with the family whose lap they reference
        ring_groups = None     # [(f_name, nf, dot_name)]
        key_group = {}         # key field name -> group index
        red_group = 0
        grad_groups = set()    # group indices that use inline grads
        if lap_name_list:
            by_name = {f.name: f for f in fields}
            # the unknown itself may appear only as a KEY (e.g. the
            # wave equation's h_ij has no bare-field RHS term)
            for key in rhs_dict:
                kf, _ = _field_of(key)
                by_name.setdefault(kf.name, kf)
            ring_groups = []
            for lap_name in lap_name_list:
                fname = lap_name[len("lap_"):]
                F = by_name.get(fname)
                if F is None or not isinstance(F, DynamicField):
                    ring_groups = None
                    break
                nf = F.shape[0] if F.shape else 1
                gi = len(ring_groups)
                ring_groups.append((fname, nf))
                key_group[fname] = gi
                key_group[F.dot.name] = gi
            if ring_groups is not None:
                for key in rhs_dict:
                    kf, _ = _field_of(key)
                    if kf.name not in key_group:
                        ring_groups = None    # unknown outside families
                        break
            if ring_groups is not None:
                # each group's ORIGINAL rhs may reference only its own
                # lap (cross-group lap reads would need the other ring)
                for key, expr in rhs_dict_orig.items():
                    kf, _ = _field_of(key)
                    gi = key_group[kf.name]
                    own_lap = f"lap_{ring_groups[gi][0]}"
                    for fld in collect_fields([expr]):
                        if fld.name.startswith("lap_") and \
                                fld.name != own_lap:
                            ring_groups = None
                            break
                    if ring_groups is None:
                        break
            if ring_groups is not None and grad_tmps:
                # which groups' statements consume the inlined
                # gradients (their kernels must define the tmps)
                from pystella_amd.field import (
                    Variable as _Var, walk_expr as _walk)
                grad_groups.clear()
                for key, expr in rhs_dict_orig.items():
                    kf, _ = _field_of(key)
                    gi = key_group[kf.name]
                    hits = []

                    def _v(x, hits=hits):
                        if isinstance(x, _Var) and \
                                x.name.startswith("gradv_"):
                            hits.append(x.name)

                    _walk(expr, _v)
                    if hits:
                        grad_groups.add(gi)
            if ring_groups is not None and red_entries_orig:
                red_laps = {f.name for f in collect_fields(
                    [e for e, _ in red_entries_orig])
                    if f.name.startswith("lap_")}
                for gi, (fname, _) in enumerate(ring_groups):
                    if f"lap_{fname}" in red_laps:
                        red_group = gi
                        break

        class _Fused(Stepper):
            pingpong = set(self.pingpong)

            def make_steps(self_inner, fixed_parameters=None, **kw):
                from pystella_amd.field import var
                from pystella_amd.elementwise import ElementWiseMap
                dtv = var("dt")
                self_inner._unknowns = []
                for key in self_inner.rhs_dict:
                    ff, outer = _field_of(key)
                    self_inner._unknowns.append((ff, outer))
                self_inner.dof_names = {ff.name for ff, _ in
                                        self_inner._unknowns}
                steps = []
                for stage in range(self_inner.num_stages):
                    tmp = {**grad_tmps, **lap_tmps}
                    rk = {}
                    guard_names = set()
                    ngroups = len(ring_groups) if ring_groups else 1
                    tmp_g = [dict(grad_tmps) if gi in grad_groups
                             else {} for gi in range(ngroups)]
                    rk_g = [{} for _ in range(ngroups)]
                    for i, (key, rhs_expr) in enumerate(
                            self_inner.rhs_dict.items()):
                        ff, outer = _field_of(key)
                        gi = key_group.get(ff.name, 0)
                        k = Field(f"{ff.name}_tmp", offset=0,
                                  shape=ff.shape, indices=ff.indices)
                        k_acc = k[outer] if outer else k
                        rhs_name = var(f"rhs_{i}")
                        tmp[rhs_name] = rhs_expr
                        tmp_g[gi][rhs_name] = rhs_dict_orig[key]
                        # keep the updated k in a register: one load and
                        # one store of the k array per site, and stores
                        # are never read back (safe for nontemporal).
                        # For large families (GW hij), stage 0's elided
                        # A*k term produces a measurably SLOWER kernel
                        # (18 vs 13 ms at 512^3) — keep the term with a
                        # runtime-zero coefficient so the compiled form
                        # (incl. the latency-hiding k preloads) matches
                        # the other stages.
                        coefA = self_inner._A[stage]
                        if (stage == 0 and ring_groups is not None
                                and ring_groups[gi][1] >= 4):
                            coefA = var("rk_a0")
                        k_new = var(f"knew_{i}")
                        tmp[k_new] = coefA * k_acc + dtv * rhs_name
                        tmp_g[gi][k_new] = tmp[k_new]
                        # the LAST stage's k stores are dead: every 2N
                        # tableau has A_0 = 0 (Williamson form), so the
                        # next step's stage 0 multiplies the array by
                        # zero before reading anything else.  Eliding
                        # them removes one full k write pass per step
                        # for SMALL families: measured +4.4 % flagship
                        # (profiles/r02_laststage_elision_ab.txt).
                        # Large (nf>=4) families keep unconditional
                        # stores: BOTH alternatives measured worse for
                        # the GW hij kernel — the store-free stage-4
                        # form recompiles slower (−5 %), and runtime-
                        # guarded stores (`store_k` branch, uniform
                        # form) collapse it to a 150-VGPR serialized
                        # form at −40 % (profiles/r02_guard_ab.txt).
                        # PYSTELLA_KEEP_LASTK=1 restores all stores.
                        keep_env = os.environ.get(
                            "PYSTELLA_KEEP_LASTK") == "1"
                        small_family = (ring_groups is None
                                        or ring_groups[gi][1] < 4)
                        last = (stage == self_inner.num_stages - 1
                                and float(self_inner._A[0]) == 0.0
                                and not keep_env)
                        if not (small_family and last):
                            rk[k_acc] = k_new
                            rk_g[gi][k_acc] = k_new
                        if ff.name in _Fused.pingpong:
                            out_f = Field(f"{ff.name}_next",
                                          offset=ff.offset,
                                          shape=ff.shape,
                                          indices=ff.indices)
                            out_acc = out_f[outer] if outer else out_f
                        else:
                            out_acc = key
                        rk[out_acc] = key + self_inner._B[stage] * k_new
                        rk_g[gi][out_acc] = rk[out_acc]
                    if reduction is not None or \
                            ring_groups is not None:
                        ring = None
                        if ring_groups is not None:
                            ring = [
                                (rk_g[gi], tmp_g[gi],
                                 red_entries_orig if gi == red_group
                                 and reduction is not None
                                 else [], fname, nf)
                                for gi, (fname, nf)
                                in enumerate(ring_groups)]
                        fp = dict(fixed_parameters or {})
                        fp.setdefault("rk_a0", 0.0)
                        is_last = (stage == self_inner.num_stages - 1
                                   and float(self_inner._A[0]) == 0.0)
                        fp["store_k"] = 0.0 if (guard_names and is_last) \
                            else 1.0
                        steps.append(_StageRedMap(
                            rk, tmp_instructions=tmp,
                            red_entries=red_entries,
                            reduction=reduction, derivs=derivs_ref,
                            lap_names=lap_name_list, ring=ring,
                            guard_kstores=frozenset(guard_names),
                            halo_shape=self_inner.halo_shape,
                            rank_shape=self_inner.rank_shape,
                            name=f"rk_stage_red{stage}",
                            fixed_parameters=fp, **kw))
                    else:
                        steps.append(ElementWiseMap(
                            rk, tmp_instructions=tmp,
                            halo_shape=self_inner.halo_shape,
                            rank_shape=self_inner.rank_shape,
                            name=f"rk_stencil_stage{stage}",
                            fixed_parameters=fixed_parameters, **kw))
                self_inner.tmp_arrays = {}
                return steps

        self._stepper = _Fused(new_rhs, dt=dt, halo_shape=halo_shape,
                               rank_shape=rank_shape, **kwargs)
        self.num_stages = self._stepper.num_stages
        self.expected_order = self._stepper.expected_order

    def __call__(self, stage, queue=None, **kwargs):
        """Runs stage ``stage``; when ``reducers`` were given, returns
        the reduced quantities of the stage's INPUT state (the same
        values the reference loop obtains from its standalone energy
        reduction after the previous stage)."""
        result = self._stepper(stage, **kwargs)
        if self._reduction is not None and result is not None:
            local, rank_shape = result
            return self._reduction._combine(local, rank_shape)
        return None

    @property
    def tmp_arrays(self):
        return self._stepper.tmp_arrays


class DeviceFriedmannLoop:
    """Fully device-resident scalar-preheating RK step: per stage, the
    energy-fused ring stage kernel (reading a/H from a device state
    buffer), the partials finish, an optional RCCL all-reduce, and the
    Friedmann (a, ȧ) ODE update all execute on the stream with ZERO host
    synchronization.  The host only reads the state buffer when asked
    (``read_state``).

    Numerics are identical to the host loop (``Expansion.step`` +
    stage-returned energies); a GPU test asserts this.

    :arg stepper: an energy-fused :class:`StencilRKStepper`.
    :arg expand: a host :class:`~pystella_amd.Expansion` providing the
        initial (a, ȧ) state (its Stepper must be the same low-storage
        tableau).
    """

    STATE_LEN = 8

    def __init__(self, stepper, decomp, expand, grid_size, dt,
                 mpl=1.0):
        import copy

        self.stepper = stepper
        self.decomp = decomp
        self.dt = dt
        self.grid_size = float(grid_size)
        self.mpl = mpl
        red = stepper._reduction
        if red is None:
            raise ValueError("stepper must be built with reducers")
        if any(op not in ("avg", "sum") for _, _, _, op in red.flat):
            raise NotImplementedError(
                "device Friedmann loop needs avg/sum reducers")
        self._red = red
        self._A = stepper._stepper._A
        self._B = stepper._stepper._B
        self.num_stages = stepper.num_stages

        # linear weights of each reduction entry in (total, pressure):
        # probe the callback with basis vectors (get_rho_and_p is linear)
        nred = len(red.flat)
        self.wt = np.zeros(nred)
        self.wp = np.zeros(nred)
        for r in range(nred):
            vals = {key: np.zeros(len(entries))
                    for key, entries in red.reducers.items()}
            key_r, i_r, _, _ = red.flat[r]
            vals[key_r][i_r] = 1.0
            out = red.callback(copy.deepcopy(vals))
            self.wt[r] = float(np.asarray(out["total"]).reshape(-1)[0])
            self.wp[r] = float(
                np.asarray(out["pressure"]).reshape(-1)[0])
            # JitFriedmann divides EVERY entry by grid_size (avg
            # semantics); 'sum' reducers must not be averaged, so bake
            # the compensating factor into the weights (matches the
            # host path, Reduction._combine, which only divides 'avg')
            if red.flat[r][3] == "sum":
                self.wt[r] *= self.grid_size
                self.wp[r] *= self.grid_size

        # z-only periodic default above the all-axes volume threshold:
        # pending A/B measurement (see _periodic_axes)
        self._z_only_default = False
        self._side_stream = None    # periodic-wrap overlap stream
        self._slab_streams = None   # concurrent boundary-slab streams
        self._shell_stream = None   # shell-kernel overlap stream

        # device state [a, adot, k_a, k_adot, hubble, energy, pressure]
        self.state = None
        self._expand0 = expand
        self._fk = None
        self._sums = None
        self._boxes = None
        self._partials = None
        self._nblks = None
        self._nblk_tot = 0

    def _ensure_state(self, device):
        if self.state is None:
            e = self._expand0
            st = torch.zeros(self.STATE_LEN, dtype=torch.float64)
            st[0] = float(e.a[0])
            st[1] = float(e.adot[0])
            st[4] = float(e.hubble[0])
            self.state = st.to(device)
            self._sums = torch.zeros(len(self._red.flat),
                                     dtype=torch.float64, device=device)

    def _regions(self, rank_shape, split_axes=None):
        """Partition of the rank box into an interior (stencil-safe
        without fresh halos along the given axes) plus up to 6
        boundary slabs.  Box format: (i0, i1, j0, j1, k0, k1).
        Default split axes: the remote (p>1) axes; the wrap-overlap
        path also splits wrapped single-rank axes so the periodic-wrap
        kernel can run on a side stream under the interior launch."""
        nx, ny, nz = rank_shape
        h = max(self.stepper._stepper.halo_shape) \
            if not isinstance(self.stepper._stepper.halo_shape,
                              int) else self.stepper._stepper.halo_shape
        px, py, pz = self.decomp.proc_shape
        if split_axes is None:
            split_axes = (px > 1, py > 1, pz > 1)
        rx, ry, rz = split_axes
        ix = (h if rx else 0, nx - h if rx else nx)
        jy = (h if ry else 0, ny - h if ry else ny)
        kz = (h if rz else 0, nz - h if rz else nz)
        interior = (ix[0], ix[1], jy[0], jy[1], kz[0], kz[1])
        slabs = []
        if rx:
            slabs.append((0, h, 0, ny, 0, nz))
            slabs.append((nx - h, nx, 0, ny, 0, nz))
        if ry:
            slabs.append((ix[0], ix[1], 0, h, 0, nz))
            slabs.append((ix[0], ix[1], ny - h, ny, 0, nz))
        if rz:
            slabs.append((ix[0], ix[1], jy[0], jy[1], 0, h))
            slabs.append((ix[0], ix[1], jy[0], jy[1], nz - h, nz))
        return interior, slabs

    def step(self, arrays, extra_scalars=None):
        """One full RK step (num_stages stages); swaps the ping-pong
        f/f_next entries of ``arrays`` in place.

        Per stage: post the (star-stencil) halo exchange of f, launch
        the interior stage kernel so compute overlaps the xGMI
        transfers, then the boundary slabs, then finish the energy
        partials + RCCL all-reduce + on-device Friedmann update — all
        stream-ordered with no host synchronization."""
        import torch.distributed as dist

        f = arrays[next(iter(self.stepper.pingpong))]
        self._ensure_state(f.device)
        env = dict(arrays)
        env["state"] = self.state
        env["dt"] = self.dt
        env["rk_a0"] = 0.0
        if extra_scalars:
            env.update(extra_scalars)
        for s in range(self.num_stages):
            smap = self.stepper._stepper.steps[s]
            if not self.stepper._stepper.tmp_arrays:
                self.stepper._stepper.tmp_arrays = \
                    self.stepper._stepper.get_tmp_arrays_like(**arrays)
            env.update(self.stepper._stepper.tmp_arrays)
            env["store_k"] = smap._map.fixed_parameters.get(
                "store_k", 1.0)
            kerns = self._stage_kernels(smap, env)

            periodic = kerns[0][0].periodic
            # wrap only the single-rank axes the kernels do NOT read
            # periodically in-register
            wrap_axes = [ax for ax, (h_, p_, per) in enumerate(zip(
                self.decomp.halo_shape, self.decomp.proc_shape,
                periodic)) if h_ > 0 and p_ == 1 and not per]
            # PYSTELLA_NO_OVERLAP=1: safety valve for real-xGMI bring-up
            # — sequential per-axis share_halos (no concurrent batched
            # group, corners propagated) instead of the overlapped path
            wrap_event = None
            if os.environ.get("PYSTELLA_NO_OVERLAP") == "1":
                # halos are fully fresh here, so the region split is
                # pointless — ONE full-box launch (this also makes
                # NO_OVERLAP the true no-split alternative strategy:
                # serial comm + peak-efficiency compute)
                for name in self.stepper.pingpong:
                    self.decomp.share_halos(arrays[name])
                handles = []
                split_axes = (False, False, False)
            else:
                # OPT-IN (PYSTELLA_WRAP_OVERLAP=1): periodic-wrap
                # kernels on a side stream under the interior launch.
                # Measured −29 % at 512³ (profiles/
                # r02_wrapoverlap_ab.txt): the thin z boundary slabs
                # run at ~1/32 lane efficiency under the (64z,8y)
                # tile, costing far more than the 0.6 ms/step of
                # serial wrap time the overlap reclaims.  Kept as a
                # knob; a per-slab tile shape would be needed to make
                # it pay.
                if wrap_axes and f.is_cuda and os.environ.get(
                        "PYSTELLA_WRAP_OVERLAP") == "1":
                    import torch as _t
                    from pystella_amd.backend.hip import wrap_star
                    if self._side_stream is None:
                        self._side_stream = _t.cuda.Stream()
                    ev0 = _t.cuda.Event()
                    ev0.record()
                    with _t.cuda.stream(self._side_stream):
                        self._side_stream.wait_event(ev0)
                        for name in self.stepper.pingpong:
                            wrap_star(arrays[name],
                                      self.decomp.halo_shape, wrap_axes)
                        wrap_event = _t.cuda.Event()
                        wrap_event.record()
                    wrapped_on_side = True
                else:
                    wrapped_on_side = False
                handles = [self.decomp.share_halos_start(
                               arrays[name],
                               wrap_axes=[] if wrapped_on_side
                               else wrap_axes)
                           for name in self.stepper.pingpong]
                px_, py_, pz_ = self.decomp.proc_shape
                split_axes = None
                if wrap_event is not None:
                    split_axes = tuple(
                        p_ > 1 or ax in wrap_axes
                        for ax, p_ in enumerate((px_, py_, pz_)))
            interior, slabs = self._regions(kerns[0][0].rank_shape,
                                            split_axes)
            # default: ONE "shell" launch covers all boundary slabs
            # (separate thin-slab launches are latency-bound at ~1
            # wave/CU each and serialize — measured 2.3 ms/step of
            # slab time at the 256^3 N=8-rank proxy).
            # PYSTELLA_SHELL=0 falls back to per-slab launches.
            use_shell = (bool(slabs) and f.is_cuda
                         and os.environ.get("PYSTELLA_SHELL") != "0"
                         and getattr(kerns[0][0], "_shell_parts", None)
                         is not None)
            slab_kerns = ([] if use_shell
                          else [self._slab_kerns(smap, kerns, b)
                                for b in slabs])
            if self._partials is None or \
                    self._boxes != (interior, tuple(slabs)):
                self._boxes = (interior, tuple(slabs))
                if use_shell:
                    nblks = [kerns[0][0].box_nblk(interior),
                             kerns[0][0].shell_nblk(slabs)]
                else:
                    nblks = [kerns[0][0].box_nblk(interior)] + [
                        bk[0][0].box_nblk(b)
                        for bk, b in zip(slab_kerns, slabs)]
                self._nblks = nblks
                self._nblk_tot = sum(nblks)
                # one partials buffer per kernel family; only the
                # reducer family's is finished into sums
                self._partials = [
                    torch.empty(
                        (len(self._red.flat) if has_red else 1,
                         self._nblk_tot),
                        dtype=torch.float64, device=f.device)
                    for _, has_red in kerns]

            red_partials = None
            for (kern, has_red), partials in zip(kerns,
                                                 self._partials):
                kern.launch_box(env, interior, partials, 0,
                                self._nblk_tot)
                if has_red:
                    red_partials = partials
            for h in handles:
                h.finish()
            if wrap_event is not None:
                import torch as _t
                _t.cuda.current_stream().wait_event(wrap_event)
            # Boundary slabs are latency-bound (each launch has ~1
            # wave/CU); run them on CONCURRENT side streams so their
            # waves co-reside (and overlap the interior tail) instead
            # of executing sequentially — measured 2.3 ms/step of
            # serial slab time at the 256^3 N=8-rank proxy.
            # PYSTELLA_SLAB_STREAMS=0 falls back to in-order launches.
            shell_ev = None
            if use_shell:
                import torch as _t
                if os.environ.get("PYSTELLA_SHELL") != "stream":
                    # in-order launch (default): at these shapes the
                    # interior saturates the GPU, so the side-stream
                    # form measured no better and costs two events
                    # (profiles/r02_shell_evolution.txt)
                    for (kern, _), partials in zip(kerns,
                                                   self._partials):
                        kern.launch_shell(env, slabs, partials,
                                          self._nblks[0],
                                          self._nblk_tot)
                else:
                    # the shell depends only on the fresh halos, not
                    # the interior launch — run it on a side stream so
                    # it overlaps the interior's tail
                    if self._shell_stream is None:
                        self._shell_stream = _t.cuda.Stream()
                    ev_ready = _t.cuda.Event()
                    ev_ready.record()
                    with _t.cuda.stream(self._shell_stream):
                        self._shell_stream.wait_event(ev_ready)
                        for (kern, _), partials in zip(
                                kerns, self._partials):
                            kern.launch_shell(env, slabs, partials,
                                              self._nblks[0],
                                              self._nblk_tot)
                        shell_ev = _t.cuda.Event()
                        shell_ev.record()
                slabs = []      # handled; skip the per-slab paths
            use_streams = (slabs and f.is_cuda and os.environ.get(
                "PYSTELLA_SLAB_STREAMS") != "0")
            if use_streams:
                import torch as _t
                if self._slab_streams is None:
                    self._slab_streams = [_t.cuda.Stream()
                                          for _ in range(6)]
                ev_ready = _t.cuda.Event()
                ev_ready.record()
                slab_evs = []
            bid0 = self._nblks[0]
            for idx, (slab, bkerns, nb) in enumerate(
                    zip(slabs, slab_kerns, self._nblks[1:])):
                if use_streams:
                    stream = self._slab_streams[
                        idx % len(self._slab_streams)]
                    with _t.cuda.stream(stream):
                        stream.wait_event(ev_ready)
                        for (kern, _), partials in zip(
                                bkerns, self._partials):
                            kern.launch_box(env, slab, partials, bid0,
                                            self._nblk_tot)
                        e = _t.cuda.Event()
                        e.record()
                        slab_evs.append(e)
                else:
                    for (kern, _), partials in zip(bkerns,
                                                   self._partials):
                        kern.launch_box(env, slab, partials, bid0,
                                        self._nblk_tot)
                bid0 += nb
            if use_streams:
                cur = _t.cuda.current_stream()
                for e in slab_evs:
                    cur.wait_event(e)
            if shell_ev is not None:
                import torch as _t
                _t.cuda.current_stream().wait_event(shell_ev)

            if self._fk is None:
                from pystella_amd.backend.hip import JitFriedmann
                self._fk = JitFriedmann(
                    red_partials.shape[0], self._nblk_tot, self.wt,
                    self.wp, self.grid_size, mpl=self.mpl)
            self._fk.finish_sums(red_partials, self._sums)
            if self.decomp.nranks > 1:
                dist.all_reduce(self._sums)
            self._fk.step(self._sums, self.state, self._A[s],
                          self._B[s], self.dt)
            for name in self.stepper.pingpong:
                arrays[name], arrays[f"{name}_next"] = \
                    arrays[f"{name}_next"], arrays[name]
                env[name] = arrays[name]
                env[f"{name}_next"] = arrays[f"{name}_next"]

    def _periodic_axes(self, smap, rank_shape):
        """Axes whose stencil reads can wrap in-kernel: non-decomposed
        axes, and only when no group reads shifted padded fields
        through the generic codegen (e.g. inlined gradients) — those
        would still need filled halos."""
        from pystella_amd.field import collect_fields
        for rk_o, tmp_o, red_o, _, _ in smap.ring:
            exprs = (list(tmp_o.values()) + list(rk_o.values())
                     + [e for e, _ in red_o])
            for fld in collect_fields(exprs):
                if fld.is_padded and any(fld.shift):
                    return (False, False, False)
        # measured on MI355X (r01 + r02 strong-scaling proxy,
        # profiles/r02_strong_proxy.txt): in-kernel periodic reads on
        # non-decomposed axes win +14 % at 128^3 and +3.3 % at 256^3
        # per-rank (the N=8/N=64 strong-scaling shapes), are neutral at
        # 512^3 scalar+GW.  Default: all axes up to 256^3-per-rank
        # volumes; above that, z only (the z-wrap launch is the
        # scattered, expensive one — r01 notes).  PYSTELLA_PERIODIC
        # forces: 1 = all, 0 = none, or an axis subset like "z"/"yz".
        import os
        px, py, pz = self.decomp.proc_shape
        force = os.environ.get("PYSTELLA_PERIODIC")
        if force == "0":
            return (False, False, False)
        if force and force != "1":
            return (px == 1 and "x" in force, py == 1 and "y" in force,
                    pz == 1 and "z" in force)
        if force != "1" and int(np.prod(rank_shape)) > 17_000_000:
            return (False, False, pz == 1 and self._z_only_default)
        return (px == 1, py == 1, pz == 1)

    def _slab_kerns(self, smap, kerns, box):
        """Tile-matched kernel variants for thin boundary slabs
        (OPT-IN, PYSTELLA_SLAB_TILES=1): re-tiles so every lane maps
        to a real site.  Measured SLOWER than the main tile at the
        256^3 proxy (3447 vs 3982; scattered access beats idle lanes
        here) — the slab cost is latency, addressed by the concurrent
        slab streams below instead."""
        if os.environ.get("PYSTELLA_SLAB_TILES") != "1":
            return kerns
        builder = getattr(smap, "_ring_builder", None)
        if builder is None:
            return kerns
        i0, i1, j0, j1, k0, k1 = box
        tbz, tby, xch = kerns[0][0].tile
        kz, jy = k1 - k0, j1 - j0
        if kz < tbz and kz <= 8 and kz <= jy:
            # next power of two >= kz keeps the block a multiple of 64
            tz = 1 << (max(1, kz) - 1).bit_length()
            tile = (tz, max(1, 256 // tz), xch)
            key = ("z", tile)
        elif jy < tby and jy <= 8:
            ty = 1 << (max(1, jy) - 1).bit_length()
            tile = (tbz, ty, xch)
            key = ("y", tile)
        else:
            return kerns
        variants = getattr(smap, "_slab_variants", None)
        if variants is None:
            smap._slab_variants = variants = {}
        kv = variants.get(key)
        if kv is None:
            built = builder(tile=tile, suffix=f"_{key[0]}slab")
            has_red = [bool(r[2]) for r in smap.ring]
            kv = list(zip(built, has_red))
            variants[key] = kv
        return kv

    def _stage_kernels(self, smap, env):
        """List of (ring kernel, has_reducers) for this stage, compiled
        with the device-state scalar map."""
        if smap.ring is None:
            raise NotImplementedError(
                "DeviceFriedmannLoop requires ring-eligible steppers")
        m = smap._map
        rank_shape = m._infer_rank_shape(env)
        kerns = smap._hip_kernel
        ok = (isinstance(kerns, list) and kerns
              and getattr(kerns[0], "state_map", None) is not None
              and kerns[0].rank_shape == rank_shape)
        if not ok:
            from pystella_amd.backend.hip import get_lap_stage_kernel
            periodic = self._periodic_axes(smap, rank_shape)

            def build(tile=None, suffix=""):
                return [
                    get_lap_stage_kernel(
                        rk_o, tmp_o, red_o or [(0.0, "sum")], fargs, [],
                        m.halo_shape, rank_shape, smap.derivs.dx, nf,
                        f_name=f_name, lap_name=f"lap_{f_name}",
                        name=f"{m.name}_{f_name}{suffix}",
                        state_map={"a": 0, "hubble": 4},
                        periodic=periodic, tile=tile,
                        guard_stores=smap.guard_kstores,
                        guard_scalar="store_k")
                    for (rk_o, tmp_o, red_o, f_name, nf), fargs
                    in zip(smap.ring, smap._ring_field_args)]

            kerns = build()
            smap._hip_kernel = kerns
            smap._kernel_shape = rank_shape
            smap._ring_builder = build
            smap._slab_variants = {}
        has_red = [bool(r[2]) for r in smap.ring]
        return list(zip(kerns, has_red))

    def read_state(self):
        """Host-side snapshot {a, adot, hubble, energy, pressure} (one
        sync)."""
        st = self.state.cpu().numpy()
        return {"a": st[0], "adot": st[1], "hubble": st[4],
                "energy": st[5], "pressure": st[6]}
