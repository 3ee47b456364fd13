"""HipOps — the product ops backend over the gfx950 C-ABI kernel library.

Every primitive is a torch.autograd.Function whose forward AND backward are
hand-written HIP kernels (include/distmlip_hip.h); backward scatters run
over the builder's permutation CSRs, so gradients are deterministic (no
atomics).  The dense GEMMs of the gated MLPs stay in rocBLAS via
torch.matmul — this module covers the irregular ops the reference
delegates to DGL.

Fails loudly if the extension is missing or tensors are not fp32 on a HIP
device — there is no fallback.
"""
from __future__ import annotations

import ctypes
from ctypes import POINTER, c_char_p, c_float, c_int32, c_int64, c_uint64

import torch

from distmlip_amd.capi import _load

_hip_lib = None


def hip_lib() -> ctypes.CDLL:
    global _hip_lib
    if _hip_lib is None:
        lib = _load("libdistmlip_hip.so")
        fp = POINTER(c_float)
        ip = POINTER(c_int32)
        lib.dm_gather_rows_f32.restype = c_int32
        lib.dm_gather_rows_f32.argtypes = [fp, ip, fp, c_int64, c_int64, c_uint64]
        lib.dm_gather_add3_f32.restype = c_int32
        lib.dm_gather_add3_f32.argtypes = [fp, fp, fp, ip, ip, fp, fp,
                                           c_int64, c_int64, c_uint64]
        lib.dm_gather_add4_f32.restype = c_int32
        lib.dm_gather_add4_f32.argtypes = [fp, fp, fp, fp, ip, ip, ip, fp,
                                           fp, c_int64, c_int64, c_uint64]
        lib.dm_edge_mlp3_f32.restype = c_int32
        lib.dm_edge_mlp3_f32.argtypes = [fp, fp, fp, fp, fp, ip, ip, fp, fp,
                                         c_int64, c_int64, c_int64, c_uint64]
        lib.dm_edge_mlp4_f32.restype = c_int32
        lib.dm_edge_mlp4_f32.argtypes = [fp, fp, fp, fp, fp, fp, ip, ip, ip,
                                         fp, fp, c_int64, c_int64, c_int64,
                                         c_uint64]
        lib.dm_silu_bwd_f32.restype = c_int32
        lib.dm_silu_bwd_f32.argtypes = [fp, fp, fp, fp, c_int64, c_uint64]
        lib.dm_seg_sum_f32.restype = c_int32
        lib.dm_seg_sum_f32.argtypes = [fp, ip, fp, fp, c_int64, c_int64,
                                       c_uint64]
        lib.dm_seg_sum_gather_f32.restype = c_int32
        lib.dm_seg_sum_gather_f32.argtypes = [fp, ip, ip, fp, fp, c_int64,
                                              c_int64, c_uint64]
        lib.dm_gated_combine_fwd_f32.restype = c_int32
        lib.dm_gated_combine_fwd_f32.argtypes = [fp, fp, fp, fp, fp, c_int64,
                                                 c_uint64]
        lib.dm_gated_combine_bwd_f32.restype = c_int32
        lib.dm_gated_combine_bwd_f32.argtypes = [fp, fp, fp, fp, fp, fp, fp,
                                                 c_int64, c_uint64]
        lib.dm_mace_tp_fwd_f32.restype = c_int32
        lib.dm_mace_tp_fwd_f32.argtypes = [fp, fp, fp, fp, ip, fp, c_int32,
                                           fp, fp, fp, fp, c_int64, c_int32,
                                           c_int32, c_int32, c_uint64]
        lib.dm_rot_gather_f32.restype = c_int32
        lib.dm_rot_gather_f32.argtypes = [fp, ip, fp, c_int32, fp,
                                          c_int64, c_int32, c_uint64]
        lib.dm_rot_scatter_f32.restype = c_int32
        lib.dm_rot_scatter_f32.argtypes = [fp, fp, c_int32, ip, ip, fp,
                                           fp, c_int64, c_int32, c_uint64]
        lib.dm_rot_dD_f32.restype = c_int32
        lib.dm_rot_dD_f32.argtypes = [fp, fp, ip, c_int32, fp, c_int64,
                                      c_int32, c_uint64]
        lib.dm_mace_symc_fwd_f32.restype = c_int32
        lib.dm_mace_symc_fwd_f32.argtypes = [fp, ip, fp, ip, fp, c_int32,
                                             fp, c_int64, c_int32, c_int32,
                                             c_int32, c_uint64]
        lib.dm_mace_symc_bwd_f32.restype = c_int32
        lib.dm_mace_symc_bwd_f32.argtypes = [fp, fp, ip, fp, ip, fp,
                                             c_int32, fp, c_int64, c_int32,
                                             c_int32, c_int32, c_uint64]
        lib.dm_mace_tp_bwd_f32.restype = c_int32
        lib.dm_mace_tp_bwd_f32.argtypes = [fp, fp, fp, fp, fp, fp, fp, fp,
                                           ip, fp, c_int32, fp, fp, fp, fp,
                                           c_int64, c_int32, c_int32,
                                           c_int32, c_uint64]
        cf = ctypes.c_float
        lib.dm_edge_geom_rbf_fwd_f32.restype = c_int32
        lib.dm_edge_geom_rbf_fwd_f32.argtypes = [fp, ip, ip, fp, fp, cf,
                                                 c_int32, c_int32, fp, fp, fp,
                                                 c_int64, c_uint64]
        lib.dm_edge_geom_rbf_bwd_f32.restype = c_int32
        lib.dm_edge_geom_rbf_bwd_f32.argtypes = [fp, fp, fp, fp, fp, fp, cf,
                                                 c_int32, c_int32, fp,
                                                 c_int64, c_uint64]
        lib.dm_rbf_env_fwd_f32.restype = c_int32
        lib.dm_rbf_env_fwd_f32.argtypes = [fp, fp, cf, c_int32, c_int32, fp,
                                           c_int64, c_uint64]
        lib.dm_rbf_env_bwd_f32.restype = c_int32
        lib.dm_rbf_env_bwd_f32.argtypes = [fp, fp, fp, cf, c_int32, c_int32,
                                           fp, c_int64, c_uint64]
        lib.dm_hip_last_error.restype = c_char_p
        _hip_lib = lib
    return _hip_lib


def _fp(t):
    return ctypes.cast(t.data_ptr(), POINTER(c_float))


def _ip(t):
    return ctypes.cast(t.data_ptr(), POINTER(c_int32))


def _stream():
    return c_uint64(torch.cuda.current_stream().cuda_stream)


def _check(rc: int, name: str):
    if rc != 0:
        raise RuntimeError(f"{name} failed: {hip_lib().dm_hip_last_error().decode()}")


def _chk_f32(*ts):
    for t in ts:
        if t is None:
            continue
        if t.device.type != "cuda":
            raise RuntimeError("HipOps requires device tensors (got CPU); "
                               "the product path has no CPU fallback")
        assert t.dtype == torch.float32 and t.is_contiguous()


def raw_gather(x: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    _chk_f32(x)
    out = torch.empty((idx.shape[0],) + tuple(x.shape[1:]), dtype=x.dtype,
                      device=x.device)
    _check(hip_lib().dm_gather_rows_f32(
        _fp(x), _ip(idx), _fp(out), out.shape[0], x.shape[1], _stream()),
        "dm_gather_rows_f32")
    return out


def raw_seg_sum(msg, row_ptr, n_rows, base=None):
    _chk_f32(msg, base)
    out = torch.empty((n_rows,) + tuple(msg.shape[1:]), dtype=msg.dtype,
                      device=msg.device)
    _check(hip_lib().dm_seg_sum_f32(
        _fp(msg), _ip(row_ptr), _fp(base) if base is not None else None,
        _fp(out), n_rows, msg.shape[1], _stream()), "dm_seg_sum_f32")
    return out


def raw_seg_sum_gather(msg, perm, row_ptr, n_rows, base=None):
    _chk_f32(msg, base)
    out = torch.empty((n_rows,) + tuple(msg.shape[1:]), dtype=msg.dtype,
                      device=msg.device)
    _check(hip_lib().dm_seg_sum_gather_f32(
        _fp(msg), _ip(perm), _ip(row_ptr),
        _fp(base) if base is not None else None, _fp(out), n_rows,
        msg.shape[1], _stream()), "dm_seg_sum_gather_f32")
    return out


class _Gather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, idx, perm, row_ptr):
        ctx.perm = perm          # None when idx is the sorted direction
        ctx.row_ptr = row_ptr
        ctx.n_rows = x.shape[0]
        return raw_gather(x, idx)

    @staticmethod
    def backward(ctx, grad):
        grad = grad.contiguous()
        if ctx.perm is None:
            gx = raw_seg_sum(grad, ctx.row_ptr, ctx.n_rows)
        else:
            gx = raw_seg_sum_gather(grad, ctx.perm, ctx.row_ptr, ctx.n_rows)
        return gx, None, None, None


def raw_silu_bwd(go_h, go_z, z):
    """dz = (go_z or 0) + go_h * silu'(z), one fused pass."""
    dz = torch.empty_like(z)
    _check(hip_lib().dm_silu_bwd_f32(
        _fp(go_h), _fp(go_z) if go_z is not None else None, _fp(z), _fp(dz),
        z.numel(), _stream()), "dm_silu_bwd_f32")
    return dz


class _GatherAdd3(torch.autograd.Function):
    """Emits (z, silu(z)) in one kernel; z doubles as the saved activation
    for the fused silu backward."""

    @staticmethod
    def forward(ctx, zs, zd, ze, src, dst, src_perm, src_row_ptr, row_ptr):
        _chk_f32(zs, zd, ze)
        # the z output is usually discarded (only h flows on): without this,
        # autograd materializes a full [E,2d] ZERO go_z every backward (a
        # zero-fill write + an extra read inside silu_bwd, ~0.8 ms/call at
        # li100k, rocprof run 17)
        ctx.set_materialize_grads(False)
        ctx.n_nodes = zs.shape[0]
        z = torch.empty_like(ze)
        # grad mode is always OFF inside Function.forward, so test the
        # inputs: when none requires grad (checkpoint outer pass /
        # inference) no backward will run and silu can go in place over z —
        # the kernel writes z then silu(z) per element, so a single buffer
        # is safe and halves the transient footprint
        if zs.requires_grad or zd.requires_grad or ze.requires_grad:
            h = torch.empty_like(ze)
        else:
            h = z
        _check(hip_lib().dm_gather_add3_f32(
            _fp(zs), _fp(zd), _fp(ze), _ip(src), _ip(dst), _fp(z), _fp(h),
            ze.shape[0], ze.shape[1], _stream()), "dm_gather_add3_f32")
        ctx.save_for_backward(src_perm, src_row_ptr, row_ptr, z)
        return z, h

    @staticmethod
    def backward(ctx, go_z, go_h):
        src_perm, src_row_ptr, row_ptr, z = ctx.saved_tensors
        if go_h is not None:
            dz = raw_silu_bwd(go_h.contiguous(),
                              go_z.contiguous() if go_z is not None else None,
                              z)
        else:
            dz = go_z.contiguous()
        gzs = raw_seg_sum_gather(dz, src_perm, src_row_ptr, ctx.n_nodes)
        gzd = raw_seg_sum(dz, row_ptr, ctx.n_nodes)
        return gzs, gzd, dz, None, None, None, None, None


class _EdgeMlp3(torch.autograd.Function):
    """Fused first-layer edge MLP over the atom graph:
    (z, silu(z)) with z = erow @ WT + bias + zs[src] + zd[dst], the per-edge
    GEMM computed INSIDE the gather kernel (LDS-resident WT) so the [E,2d]
    GEMM output is never materialized.  WT/bias must be frozen (inference
    path) — their grads are not produced."""

    @staticmethod
    def forward(ctx, erow, wt, bias, zs, zd, src, dst, src_perm, src_row_ptr,
                row_ptr):
        _chk_f32(erow, wt, bias, zs, zd)
        assert not wt.requires_grad and not bias.requires_grad
        ctx.set_materialize_grads(False)
        ctx.n_nodes = zs.shape[0]
        E, dout = erow.shape[0], wt.shape[1]
        h = torch.empty(E, dout, dtype=erow.dtype, device=erow.device)
        needs_z = erow.requires_grad or zs.requires_grad or zd.requires_grad
        z = torch.empty_like(h) if needs_z else h[:0]  # no bwd: skip z write
        _check(hip_lib().dm_edge_mlp3_f32(
            _fp(erow), _fp(wt), _fp(bias), _fp(zs), _fp(zd), _ip(src),
            _ip(dst), _fp(z) if needs_z else None, _fp(h), E, wt.shape[0],
            dout, _stream()), "dm_edge_mlp3_f32")
        ctx.save_for_backward(src_perm, src_row_ptr, row_ptr, z, wt)
        return z, h

    @staticmethod
    def backward(ctx, go_z, go_h):
        src_perm, src_row_ptr, row_ptr, z, wt = ctx.saved_tensors
        if go_h is not None:
            dz = raw_silu_bwd(go_h.contiguous(),
                              go_z.contiguous() if go_z is not None else None,
                              z)
        else:
            dz = go_z.contiguous()
        de = dz @ wt.t()
        gzs = raw_seg_sum_gather(dz, src_perm, src_row_ptr, ctx.n_nodes)
        gzd = raw_seg_sum(dz, row_ptr, ctx.n_nodes)
        return de, None, None, gzs, gzd, None, None, None, None, None


class _EdgeMlp4(torch.autograd.Function):
    """Fused first-layer bond MLP over the line graph (3-gather form):
    z = arow @ WT + bias + z1[l_src] + z2[l_dst] + zv[center]."""

    @staticmethod
    def forward(ctx, arow, wt, bias, z1, z2, zv, pd):
        _chk_f32(arow, wt, bias, z1, z2, zv)
        assert not wt.requires_grad and not bias.requires_grad
        ctx.set_materialize_grads(False)
        ctx.pd = pd
        ctx.n_bonds = z1.shape[0]
        ctx.n_nodes = zv.shape[0]
        L, dout = arow.shape[0], wt.shape[1]
        h = torch.empty(L, dout, dtype=arow.dtype, device=arow.device)
        needs_z = (arow.requires_grad or z1.requires_grad or z2.requires_grad
                   or zv.requires_grad)
        z = torch.empty_like(h) if needs_z else h[:0]
        _check(hip_lib().dm_edge_mlp4_f32(
            _fp(arow), _fp(wt), _fp(bias), _fp(z1), _fp(z2), _fp(zv),
            _ip(pd.l_src), _ip(pd.l_dst), _ip(pd.center),
            _fp(z) if needs_z else None, _fp(h), L, wt.shape[0], dout,
            _stream()), "dm_edge_mlp4_f32")
        ctx.save_for_backward(z, wt)
        return z, h

    @staticmethod
    def backward(ctx, go_z, go_h):
        pd = ctx.pd
        z, wt = ctx.saved_tensors
        if go_h is not None:
            dz = raw_silu_bwd(go_h.contiguous(),
                              go_z.contiguous() if go_z is not None else None,
                              z)
        else:
            dz = go_z.contiguous()
        da = dz @ wt.t()
        gz1 = raw_seg_sum_gather(dz, pd.line_src_perm, pd.line_src_row_ptr,
                                 ctx.n_bonds)
        gz2 = raw_seg_sum(dz, pd.line_row_ptr, ctx.n_bonds)
        gzv = raw_seg_sum_gather(dz, pd.center_perm, pd.center_row_ptr,
                                 ctx.n_nodes)
        return da, None, None, gz1, gz2, gzv, None


class _GatherAdd4(torch.autograd.Function):
    """Emits (z, silu(z)) over the line-graph 3-gather-add."""

    @staticmethod
    def forward(ctx, z1, z2, za, zv, pd):
        _chk_f32(z1, z2, za, zv)
        ctx.set_materialize_grads(False)   # see _GatherAdd3.forward
        ctx.pd = pd
        ctx.n_bonds = z1.shape[0]
        ctx.n_nodes = zv.shape[0]
        z = torch.empty_like(za)
        # see _GatherAdd3.forward: in-place silu only when no backward runs
        needs_h = (z1.requires_grad or z2.requires_grad or za.requires_grad
                   or zv.requires_grad)
        h = torch.empty_like(za) if needs_h else z
        _check(hip_lib().dm_gather_add4_f32(
            _fp(z1), _fp(z2), _fp(za), _fp(zv), _ip(pd.l_src), _ip(pd.l_dst),
            _ip(pd.center), _fp(z), _fp(h), za.shape[0], za.shape[1],
            _stream()), "dm_gather_add4_f32")
        ctx.save_for_backward(z)
        return z, h

    @staticmethod
    def backward(ctx, go_z, go_h):
        pd = ctx.pd
        (z,) = ctx.saved_tensors
        if go_h is not None:
            dz = raw_silu_bwd(go_h.contiguous(),
                              go_z.contiguous() if go_z is not None else None,
                              z)
        else:
            dz = go_z.contiguous()
        gz1 = raw_seg_sum_gather(dz, pd.line_src_perm, pd.line_src_row_ptr,
                                 ctx.n_bonds)
        gz2 = raw_seg_sum(dz, pd.line_row_ptr, ctx.n_bonds)
        gzv = raw_seg_sum_gather(dz, pd.center_perm, pd.center_row_ptr,
                                 ctx.n_nodes)
        return gz1, gz2, dz, gzv, None


class _SegSum(torch.autograd.Function):
    """out[n] = base[n] + sum of idx-sorted msg rows in [rp[n], rp[n+1])."""

    @staticmethod
    def forward(ctx, msg, idx, row_ptr, n_rows, base):
        ctx.save_for_backward(idx)
        ctx.has_base = base is not None
        return raw_seg_sum(msg.contiguous(), row_ptr, n_rows,
                           base.contiguous() if base is not None else None)

    @staticmethod
    def backward(ctx, grad):
        (idx,) = ctx.saved_tensors
        gmsg = raw_gather(grad.contiguous(), idx)
        gbase = grad if ctx.has_base else None
        return gmsg, None, None, None, gbase


class _GatedCombine(torch.autograd.Function):
    """out = base + silu(c) * sigmoid(g) * w — fused gated-MLP epilogue."""

    @staticmethod
    def forward(ctx, c, g, w, base):
        _chk_f32(c, g, w, base)
        ctx.save_for_backward(c, g) if w is None else \
            ctx.save_for_backward(c, g, w)
        ctx.has_w = w is not None
        ctx.has_base = base is not None
        out = torch.empty_like(c)
        _check(hip_lib().dm_gated_combine_fwd_f32(
            _fp(c), _fp(g), _fp(w) if w is not None else None,
            _fp(base) if base is not None else None, _fp(out), c.numel(),
            _stream()), "dm_gated_combine_fwd_f32")
        return out

    @staticmethod
    def backward(ctx, go):
        if ctx.has_w:
            c, g, w = ctx.saved_tensors
        else:
            c, g = ctx.saved_tensors
            w = None
        go = go.contiguous()
        dc = torch.empty_like(c)
        dg = torch.empty_like(g)
        dw = torch.empty_like(w) if ctx.has_w else None
        _check(hip_lib().dm_gated_combine_bwd_f32(
            _fp(go), _fp(c), _fp(g), _fp(w) if w is not None else None,
            _fp(dc), _fp(dg), _fp(dw) if dw is not None else None,
            c.numel(), _stream()), "dm_gated_combine_bwd_f32")
        return dc, dg, dw, (go if ctx.has_base else None)


class _EdgeGeomRbf(torch.autograd.Function):
    """Fused bond_vec/bond_dist/RBF*envelope (chgnet.py:96-124 analog)."""

    @staticmethod
    def forward(ctx, pos, offshift, freqs, cutoff, pexp, pd):
        _chk_f32(pos, offshift, freqs)
        E = offshift.shape[0]
        nrbf = freqs.shape[0]
        bv = torch.empty(E, 3, dtype=pos.dtype, device=pos.device)
        bd = torch.empty(E, dtype=pos.dtype, device=pos.device)
        exp_out = torch.empty(E, nrbf, dtype=pos.dtype, device=pos.device)
        _check(hip_lib().dm_edge_geom_rbf_fwd_f32(
            _fp(pos), _ip(pd.src), _ip(pd.dst), _fp(offshift), _fp(freqs),
            float(cutoff), int(pexp), nrbf, _fp(bv), _fp(bd), _fp(exp_out),
            E, _stream()), "dm_edge_geom_rbf_fwd_f32")
        ctx.save_for_backward(bv, bd, freqs)
        ctx.set_materialize_grads(False)   # bv/bd grads are often absent
        ctx.pd = pd
        ctx.cutoff, ctx.pexp, ctx.nrbf = float(cutoff), int(pexp), nrbf
        ctx.n_nodes = pos.shape[0]
        return bv, bd, exp_out

    @staticmethod
    def backward(ctx, go_bv, go_bd, go_exp):
        bv, bd, freqs = ctx.saved_tensors
        pd = ctx.pd
        E = bv.shape[0]
        gbv = torch.empty_like(bv)
        go_exp = go_exp.contiguous() if go_exp is not None else \
            torch.zeros(E, ctx.nrbf, dtype=bv.dtype, device=bv.device)
        _check(hip_lib().dm_edge_geom_rbf_bwd_f32(
            _fp(go_bv.contiguous()) if go_bv is not None else None,
            _fp(go_bd.contiguous()) if go_bd is not None else None,
            _fp(go_exp), _fp(bv), _fp(bd), _fp(freqs), ctx.cutoff, ctx.pexp,
            ctx.nrbf, _fp(gbv), E, _stream()), "dm_edge_geom_rbf_bwd_f32")
        g_pos = raw_seg_sum(gbv, pd.row_ptr, ctx.n_nodes) - \
            raw_seg_sum_gather(gbv, pd.src_perm, pd.src_row_ptr, ctx.n_nodes)
        # freqs gradient not implemented (inference engine; forces only)
        return g_pos, gbv, None, None, None, None


class _RbfEnv(torch.autograd.Function):
    @staticmethod
    def forward(ctx, d, freqs, cutoff, pexp):
        _chk_f32(d, freqs)
        M, nrbf = d.shape[0], freqs.shape[0]
        out = torch.empty(M, nrbf, dtype=d.dtype, device=d.device)
        _check(hip_lib().dm_rbf_env_fwd_f32(
            _fp(d), _fp(freqs), float(cutoff), int(pexp), nrbf, _fp(out), M,
            _stream()), "dm_rbf_env_fwd_f32")
        ctx.save_for_backward(d, freqs)
        ctx.cutoff, ctx.pexp, ctx.nrbf = float(cutoff), int(pexp), nrbf
        return out

    @staticmethod
    def backward(ctx, go):
        d, freqs = ctx.saved_tensors
        gd = torch.empty_like(d)
        _check(hip_lib().dm_rbf_env_bwd_f32(
            _fp(go.contiguous()), _fp(d), _fp(freqs), ctx.cutoff, ctx.pexp,
            ctx.nrbf, _fp(gd), d.shape[0], _stream()), "dm_rbf_env_bwd_f32")
        return gd, None, None, None


class _GatedCombinePacked(torch.autograd.Function):
    """gated_combine over PACKED cg [2,E,D] (cg[0]=c, cg[1]=g): avoids the
    h-slice / c-g-select gradient zero+copy+add passes entirely."""

    @staticmethod
    def forward(ctx, cg, w, base):
        _chk_f32(cg, w, base)
        half = cg.shape[1] * cg.shape[2]
        c_ptr = ctypes.cast(cg.data_ptr(), POINTER(c_float))
        g_ptr = ctypes.cast(cg.data_ptr() + 4 * half, POINTER(c_float))
        out = torch.empty(cg.shape[1], cg.shape[2], dtype=cg.dtype,
                          device=cg.device)
        _check(hip_lib().dm_gated_combine_fwd_f32(
            c_ptr, g_ptr, _fp(w) if w is not None else None,
            _fp(base) if base is not None else None, _fp(out), half,
            _stream()), "dm_gated_combine_fwd_f32")
        ctx.save_for_backward(cg, w) if w is not None else             ctx.save_for_backward(cg)
        ctx.has_w = w is not None
        ctx.has_base = base is not None
        return out

    @staticmethod
    def backward(ctx, go):
        if ctx.has_w:
            cg, w = ctx.saved_tensors
        else:
            (cg,) = ctx.saved_tensors
            w = None
        go = go.contiguous()
        half = cg.shape[1] * cg.shape[2]
        dcg = torch.empty_like(cg)
        c_ptr = ctypes.cast(cg.data_ptr(), POINTER(c_float))
        g_ptr = ctypes.cast(cg.data_ptr() + 4 * half, POINTER(c_float))
        dc_ptr = ctypes.cast(dcg.data_ptr(), POINTER(c_float))
        dg_ptr = ctypes.cast(dcg.data_ptr() + 4 * half, POINTER(c_float))
        dw = torch.empty_like(w) if ctx.has_w else None
        _check(hip_lib().dm_gated_combine_bwd_f32(
            _fp(go), c_ptr, g_ptr, _fp(w) if w is not None else None,
            dc_ptr, dg_ptr, _fp(dw) if dw is not None else None, half,
            _stream()), "dm_gated_combine_bwd_f32")
        return dcg, dw, (go if ctx.has_base else None)


class HipOps:
    """Product ops backend (see ops_base.OpsBackend)."""

    is_reference = False

    def gather(self, x, idx, csr=None):
        if csr is None:
            raise RuntimeError("HipOps.gather needs a (perm, row_ptr) CSR "
                               "for its deterministic backward")
        return _Gather.apply(x.contiguous(), idx, csr[0], csr[1])

    def gather_add3_act(self, zs, zd, ze, pd):
        """silu(zs[src] + zd[dst] + ze), silu fused into the gather kernel."""
        _z, h = _GatherAdd3.apply(zs.contiguous(), zd.contiguous(),
                                  ze.contiguous(), pd.src, pd.dst,
                                  pd.src_perm, pd.src_row_ptr, pd.row_ptr)
        return h

    def edge_mlp3_act(self, erow, wt, bias, zs, zd, pd):
        """silu(erow @ wt + bias + zs[src] + zd[dst]) fused in one kernel
        (wt = first-layer weight.T, [64,128] only)."""
        _z, h = _EdgeMlp3.apply(erow.contiguous(), wt, bias, zs.contiguous(),
                                zd.contiguous(), pd.src, pd.dst, pd.src_perm,
                                pd.src_row_ptr, pd.row_ptr)
        return h

    def edge_mlp4_act(self, arow, wt, bias, z1, z2, zv, pd):
        _z, h = _EdgeMlp4.apply(arow.contiguous(), wt, bias, z1.contiguous(),
                                z2.contiguous(), zv.contiguous(), pd)
        return h

    def gather_add4_act(self, z1, z2, za, zv, pd):
        _z, h = _GatherAdd4.apply(z1.contiguous(), z2.contiguous(),
                                  za.contiguous(), zv.contiguous(), pd)
        return h

    def scatter_edges(self, msg, pd, base=None):
        return _SegSum.apply(msg, pd.dst, pd.row_ptr, pd.n_atoms, base)

    def scatter_rows(self, msg, dst_rel, row_ptr_rel, n_rows):
        """Segment-sum of dst-sorted msg rows over an arbitrary local CSR
        (node-range-chunked message passes; dst_rel/row_ptr_rel are
        relative to the range start)."""
        return _SegSum.apply(msg, dst_rel, row_ptr_rel, n_rows, None)

    # -- raw (non-differentiable) primitives for the hand-sequenced conv
    #    backward (distmlip_amd.conv); see ops_base docstring ------------

    def r_gather_add3(self, zs, zd, ze, pd):
        _chk_f32(zs, zd, ze)
        z = torch.empty_like(ze)
        h = torch.empty_like(ze)
        _check(hip_lib().dm_gather_add3_f32(
            _fp(zs), _fp(zd), _fp(ze), _ip(pd.src), _ip(pd.dst), _fp(z),
            _fp(h), ze.shape[0], ze.shape[1], _stream()),
            "dm_gather_add3_f32")
        return z, h

    def _cg_ptrs(self, cg):
        half = cg.shape[1] * cg.shape[2]
        return (ctypes.cast(cg.data_ptr(), POINTER(c_float)),
                ctypes.cast(cg.data_ptr() + 4 * half, POINTER(c_float)),
                half)

    def r_combine_fwd(self, cg, w, base):
        _chk_f32(cg, w, base)
        c_ptr, g_ptr, half = self._cg_ptrs(cg)
        out = torch.empty(cg.shape[1], cg.shape[2], dtype=cg.dtype,
                          device=cg.device)
        _check(hip_lib().dm_gated_combine_fwd_f32(
            c_ptr, g_ptr, _fp(w) if w is not None else None,
            _fp(base) if base is not None else None, _fp(out), half,
            _stream()), "dm_gated_combine_fwd_f32")
        return out

    def r_combine_bwd(self, go, cg, w):
        _chk_f32(go, cg, w)
        c_ptr, g_ptr, half = self._cg_ptrs(cg)
        dcg = torch.empty_like(cg)
        dc_ptr = ctypes.cast(dcg.data_ptr(), POINTER(c_float))
        dg_ptr = ctypes.cast(dcg.data_ptr() + 4 * half, POINTER(c_float))
        dw = torch.empty_like(w) if w is not None else None
        _check(hip_lib().dm_gated_combine_bwd_f32(
            _fp(go), c_ptr, g_ptr, _fp(w) if w is not None else None,
            dc_ptr, dg_ptr, _fp(dw) if dw is not None else None, half,
            _stream()), "dm_gated_combine_bwd_f32")
        return dcg, dw

    def r_silu_bwd(self, go_h, z):
        return raw_silu_bwd(go_h.contiguous(), None, z)

    def r_gather_dst(self, x, pd):
        return raw_gather(x.contiguous(), pd.dst)

    def r_seg_dst(self, msg, pd, base=None):
        return raw_seg_sum(msg, pd.row_ptr, pd.n_atoms, base)

    def r_seg_src(self, msg, pd):
        return raw_seg_sum_gather(msg, pd.src_perm, pd.src_row_ptr,
                                  pd.n_atoms)

    # line-graph raw primitives (bond-conv hand-sequenced reverse)

    def r_gather_add4(self, z1, z2, za, zv, pd):
        _chk_f32(z1, z2, za, zv)
        z = torch.empty_like(za)
        h = torch.empty_like(za)
        _check(hip_lib().dm_gather_add4_f32(
            _fp(z1), _fp(z2), _fp(za), _fp(zv), _ip(pd.l_src), _ip(pd.l_dst),
            _ip(pd.center), _fp(z), _fp(h), za.shape[0], za.shape[1],
            _stream()), "dm_gather_add4_f32")
        return z, h

    def r_gather_lsrc(self, x, pd):
        return raw_gather(x.contiguous(), pd.l_src)

    def r_gather_ldst(self, x, pd):
        return raw_gather(x.contiguous(), pd.l_dst)

    def r_seg_ldst(self, msg, pd, base=None):
        return raw_seg_sum(msg, pd.line_row_ptr, pd.n_bonds, base)

    def r_seg_lsrc(self, msg, pd):
        return raw_seg_sum_gather(msg, pd.line_src_perm, pd.line_src_row_ptr,
                                  pd.n_bonds)

    def r_seg_center(self, msg, pd):
        return raw_seg_sum_gather(msg, pd.center_perm, pd.center_row_ptr,
                                  pd.n_atoms)

    def scatter_lines(self, msg, pd, base=None):
        return _SegSum.apply(msg, pd.l_dst, pd.line_row_ptr, pd.n_bonds, base)

    def gated_combine(self, c, g, w=None, base=None):
        return _GatedCombine.apply(
            c.contiguous(), g.contiguous(),
            w.contiguous() if w is not None else None,
            base.contiguous() if base is not None else None)

    def gated_combine_packed(self, cg, w=None, base=None):
        return _GatedCombinePacked.apply(
            cg.contiguous(),
            w.contiguous() if w is not None else None,
            base.contiguous() if base is not None else None)

    def edge_geom_rbf(self, pos, offshift, freqs, cutoff, pexp, pd):
        return _EdgeGeomRbf.apply(pos.contiguous(), offshift.contiguous(),
                                  freqs.contiguous(), cutoff, pexp, pd)

    def rbf_env(self, d, freqs, cutoff, pexp):
        return _RbfEnv.apply(d.contiguous(), freqs.contiguous(), cutoff, pexp)
