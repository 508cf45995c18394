"""GPU op implementations: shape/layout glue around the gfx950 kernels.

Layout strategy (MI355X-first): conv-stack activations live in NHWC
(torch channels_last) so the implicit-GEMM epilogue writes coalesced rows
and LRN's channel window is contiguous; weights are repacked [K,R,S,C/g]
bf16 per forward (small).  The GEMM K dimension is padded to a multiple of
8 so the 16-byte direct-to-LDS staging path stays aligned.  Compute dtype
on GPU is bf16 with fp32 accumulation (fp32 master weights live in the
solver); torch is used only for small glue (casts, pads, permutes).
"""

from __future__ import annotations

import math
import os
import torch

from . import reference

_ext = None


def _pad8(k: int) -> int:
    return (k + 7) // 8 * 8


def _pad32(k: int) -> int:
    return (k + 31) // 32 * 32


def _cl(x: torch.Tensor) -> torch.Tensor:
    """channels_last contiguous form of a 4-D activation; the converted
    copy is cached on the tensor (keyed by torch's in-place version
    counter) so a statically-fed input converts once, not every step."""
    if x.is_contiguous(memory_format=torch.channels_last):
        return x
    cached = getattr(x, "_cos_cl", None)
    if cached is not None and cached[0] == x._version:
        return cached[1]
    y = x.contiguous(memory_format=torch.channels_last)
    try:
        x._cos_cl = (x._version, y)
    except AttributeError:
        pass
    return y


def _as_bf16(t: torch.Tensor) -> torch.Tensor:
    if t.dtype == torch.bfloat16:
        return t
    cached = getattr(t, "_cos_bf16", None)   # solver bf16 shadow arena
    if cached is not None:
        return cached
    return t.to(torch.bfloat16)


def _check_bf16(x, name):
    if x.dtype != torch.bfloat16:
        raise RuntimeError(
            f"GPU path computes in bf16; got {x.dtype} for {name}. "
            "Build the net with dtype=torch.bfloat16 on GPU.")


def _splitk_for(mb: int, nb: int, K: int) -> int:
    """Pick a split-K factor that fills the chip without flooding L2 with
    fp32 atomics (each split multiplies the per-element atomicAdd count):
    ~900 blocks of 4 waves ≈ full occupancy on 256 CUs."""
    tiles = mb * nb
    if tiles >= 1024 or K < 1024:
        return 1
    target = int(os.environ.get("COS_SPLITK_TARGET", "448"))
    return max(1, min(K // 256, target // max(1, tiles)))


def _splitk_tt(M: int, N: int, K: int) -> int:
    """Split-K for the fused trans/trans GEMMs: the wide kernel re-tiles
    N to 256, and the sweep (scratch/sweep_tt_sk.py, MI355X) shows the
    optimum at ~448 8-wave blocks with >=512 K per block."""
    mb = (M + 127) // 128
    nb = (N + 255) // 256 if N > 128 else (N + 127) // 128
    if K < 1024:
        return 1
    target = int(os.environ.get("COS_SPLITK_TARGET", "448"))
    return max(1, min(target // max(1, mb * nb), K // 512))


def _pad128(n: int) -> int:
    return (n + 127) // 128 * 128


def _transpose(t: torch.Tensor) -> torch.Tensor:
    """Fast tiled transpose of a 2-D contiguous bf16 tensor.  Rows are
    over-allocated to a multiple of 128 so the GEMM's direct-to-LDS fast
    path can stage edge tiles (junk rows are never read back)."""
    R, C = t.shape
    buf = torch.empty((_pad128(C), R), dtype=t.dtype, device=t.device)
    out = buf[:C]
    _ext.transpose(t, out, R, C)
    return out


# ---- fused weight repack registry: one kernel per step refreshes every
# conv's GEMM-ready layouts (wrb and, for G==1, its transpose) from the
# bf16 shadow arena, replacing ~2 tiny launches per conv per step.
# Sources are held by WEAK reference so dead solvers' buffers drop out
# of the table instead of leaking (tests construct many solvers); the
# wrb/wrT buffers are kept alive by the weight tensors they cache on.
import weakref as _weakref

_repack = {"items": [], "table": None, "max": 0, "epoch": 0, "dirty": False}


def _repack_register(src, wrb, wrT, Kout, Cg, R, S, Kpad, wrF=None,
                     k2p=0, G=1, Cgp=0):
    _repack["items"].append((_weakref.ref(src), _weakref.ref(wrb),
                             None if wrT is None else _weakref.ref(wrT),
                             Kout, Cg, R, S, Kpad,
                             None if wrF is None else _weakref.ref(wrF),
                             k2p, G, Cgp))
    _repack["dirty"] = True


def refresh_packed_weights():
    """Called from the solver's per-step bf16 shadow refresh."""
    items = _repack["items"]
    if not items:
        _repack["epoch"] += 1
        return
    live = [(it, it[0](), it[1]()) for it in items]
    if any(src is None or wrb is None for _, src, wrb in live):
        _repack["items"] = items = [it for it, src, wrb in live
                                    if src is not None and wrb is not None]
        live = [(it, s, w) for it, s, w in live
                if s is not None and w is not None]
        _repack["dirty"] = True
    if not items:
        _repack["epoch"] += 1
        return
    if _repack["dirty"]:
        rows = []
        mx = 0
        for (it, src, wrb) in live:
            wrT = None if it[2] is None else it[2]()
            wrF = None if it[8] is None else it[8]()
            _, _, _, Kout, Cg, R, S, Kpad = it[:8]
            rows.append([src.data_ptr(), wrb.data_ptr(),
                         0 if wrT is None else wrT.data_ptr(),
                         Kout, Cg, R, S, Kpad,
                         0 if wrF is None else wrF.data_ptr(), it[9],
                         it[10], it[11]])
            mx = max(mx, Kout * R * S * Cg)
        _repack["table"] = torch.tensor(rows, dtype=torch.int64).to(
            live[0][2].device)
        _repack["max"] = mx
        _repack["dirty"] = False
    _ext.repack_weights(_repack["table"], len(items), _repack["max"])
    _repack["epoch"] += 1


# ---- per-backward scratch: the split-K GEMMs and colsum accumulate with
# fp32 atomics, so their outputs need zero-init every backward.  Fresh
# torch.zeros per call costs one FillFunctor launch each (GoogLeNet:
# ~140 tiny fills/step, 4-5% of the step).  Instead, buffers persist
# across steps keyed by their call ORDER within the backward (the net
# graph is fixed, so the sequence is deterministic), and ONE fused
# _foreach_zero_ at backward start re-zeroes them all.  Buffers with
# zero_each_step=False only need their padding zeroed once (padded-copy
# staging: the live region is fully overwritten every call).
_scratch = {"seq": 0, "bufs": [], "zero_list": [], "dirty": False,
            "retired": []}


def begin_backward():
    """Reset the scratch sequence + fused-zero all registered buffers.
    Called by Net.backward on the GPU path (captured into hipGraphs like
    any other kernel, so replayed steps re-zero too)."""
    _scratch["seq"] = 0
    if _scratch["dirty"]:
        _scratch["zero_list"] = [t for t, z in _scratch["bufs"] if z]
        _scratch["dirty"] = False
    if _scratch["zero_list"]:
        torch._foreach_zero_(_scratch["zero_list"])


def _scratch_buf(shape, dtype, device, zero_each_step=True):
    """Next persistent scratch buffer in call order, allocated zeroed.
    A shape/dtype/device mismatch (different net interleaved) reallocates
    in place, keeping correctness at the cost of cache churn."""
    i = _scratch["seq"]
    _scratch["seq"] = i + 1
    bufs = _scratch["bufs"]
    if i < len(bufs):
        t, z = bufs[i]
        if t.shape == shape and t.dtype == dtype and t.device == device \
                and z == zero_each_step:
            return t
        new = torch.zeros(shape, dtype=dtype, device=device)
        # NEVER free a replaced buffer: a captured hipGraph may hold its
        # address baked into kernel launches (replay runs no Python)
        _scratch["retired"].append(t)
        bufs[i] = (new, zero_each_step)
        _scratch["dirty"] = True
        return new
    t = torch.zeros(shape, dtype=dtype, device=device)
    bufs.append((t, zero_each_step))
    _scratch["dirty"] = True
    return t


_zero16 = {}


def _zpage(device):
    """Persistent 16-byte zero page: OOB implicit-staging lanes DMA from
    here so the instruction always issues (exact vmcnt counting)."""
    t = _zero16.get(device.index)
    if t is None:
        t = torch.zeros(8, dtype=torch.bfloat16, device=device)
        _zero16[device.index] = t
    return t


def _gemm(A, B, C, bias, M, N, K, lda, ldb, ldc, ta, tb, store, splitk,
          relu=False, alpha=1.0, ma=0, na=0):
    # staging-allocation bounds: number of rows safely readable past M/N
    ma = ma or (A.shape[0] if not ta else M)
    na = na or (B.shape[0] if not tb else N)
    _ext.gemm(A, B, C, bias, M, N, K, lda, ldb, ldc, ta, tb, store, splitk,
              relu, alpha, ma, na)


# ----------------------------------------------------------------- conv2d

def _conv_out_dim(h, k, s, p, dil):
    eff = (k - 1) * dil + 1
    return (h + 2 * p - eff) // s + 1


def conv2d_forward(x, w, b, stride, pad, dilation, groups, ctx=None,
                   relu=False, out_into=None):
    _check_bf16(x, "conv input")
    sh, sw = stride
    ph, pw = pad
    dil = dilation[0]
    N, C, H, W = x.shape
    Kout, Cg, R, S = w.shape
    P = _conv_out_dim(H, R, sh, ph, dil)
    Q = _conv_out_dim(W, S, sw, pw, dil)
    G = groups
    Kg = Kout // G
    Kcol = R * S * Cg
    # implicit-im2col GEMM (round 2): stage the col operand straight
    # from the NHWC input inside the GEMM — the col matrix is never
    # materialized.  Needs 8-aligned channel octets (one (r,s) tap per
    # 16-byte load) and npq/k decode within the magic-divide bound.
    imp1x1 = (R == S == 1 and sh == sw == 1 and ph == pw == 0
              and G == 1 and C >= 256
              and bool(int(os.environ.get("COS_IMP_1X1", "0"))))
    implicit = (C % 8 == 0 and Cg % 8 == 0
                and (not (R == S == 1) or imp1x1)
                and N * P * Q < (1 << 20) and R * S * Cg < (1 << 20)
                and bool(int(os.environ.get("COS_IMPLICIT", "1"))))
    # small-C variant (conv1-class, C%8!=0): with dil==1 / full
    # channels the k axis is CONTIGUOUS spans of the input rows, so
    # staging uses unaligned 16-byte loads straight from x (the col
    # matrix for AlexNet conv1 was 570 MB written + read twice)
    # M gate: at very large NPQ (AlexNet conv1, 774k rows) the staging
    # VALU cost beats the saved im2col (same-box A/B: -2% AlexNet,
    # +1.5% LRCN whose conv1 has 193k rows)
    implicit_sc = (not implicit and C == Cg and G == 1 and dil == 1
                   and S * C >= 8 and not (R == S == 1)
                   and N * P * Q < (1 << 18)
                   and 256 <= Kcol < (1 << 20)
                   and bool(int(os.environ.get("COS_IMPLICIT_SC", "1"))))
    # 32-aligned K keeps every k-tile on the pipelined fast path (and
    # bounds the implicit kernel's B reads); pad columns stay zero
    Kpad = _pad32(Kcol) if (implicit or implicit_sc) else _pad8(Kcol)

    xl = _cl(x)
    # weight repack: [K, Cg, R, S] -> bf16 [K, R, S, Cg] padded to Kpad
    # columns and 128-aligned rows (GEMM fast-staging bound); the padded
    # buffer is cached on the weight tensor (pad region stays zero),
    # only the permute-copy runs per step
    # row over-allocation: the implicit forward kernel fast-stages full
    # 128-row B tiles per group slice, so the last group's tile span
    # must stay inside the buffer (extra rows are zero -> contribute 0)
    rows_w = max(_pad128(Kout), (G - 1) * Kg + _pad128(Kg))
    wrb = getattr(w, "_cos_wrb", None)
    if wrb is None or wrb.shape != (rows_w, Kpad):
        wrb = torch.zeros((rows_w, Kpad), dtype=torch.bfloat16,
                          device=x.device)
        w._cos_wrb = wrb
        shadow = w if (w.dtype == torch.bfloat16 and
                       getattr(w, "_cos_stable", False)) \
            else getattr(w, "_cos_bf16", None)
        # implicit-dx eligibility: stride-1 G==1 convs compute the data
        # gradient as ONE implicit GEMM over dy with the flipped-weight
        # layout wrF[c][(RS-1-rs)*Kout + k] (no dcol, no col2im)
        k2 = R * S * Kg
        dx_imp = (implicit and sh == sw == 1
                  and bool(int(os.environ.get("COS_DX_IMP", "1")))
                  and Kg % 8 == 0 and dil * (R - 1) - ph >= 0
                  and dil * (S - 1) - pw >= 0 and N * H * W < (1 << 20)
                  and k2 < (1 << 20))
        wrF = None
        k2p = 0
        if dx_imp:
            k2p = _pad32(k2)
            wrF = torch.zeros((G * _pad128(Cg), k2p),
                              dtype=torch.bfloat16, device=x.device)
            w._cos_wrF = wrF
        if shadow is not None and shadow.is_contiguous():
            # solver-managed weight: the fused per-step repack kernel
            # (refresh_packed_weights, run with the shadow refresh)
            # maintains wrb — and for G==1 also the dx GEMM's transposed
            # layout — from here on
            wrT = None
            if G == 1:
                wrT = torch.zeros((_pad128(Kpad), Kout),
                                  dtype=torch.bfloat16, device=x.device)
                w._cos_wrT = wrT
            _repack_register(shadow, wrb, wrT, Kout, Cg, R, S, Kpad,
                             wrF, k2p, G, _pad128(Cg))
            w._cos_register_epoch = _repack["epoch"]
    # once a refresh has run AFTER registration, the fused kernel owns
    # wrb/wrT; until then (first step, or unregistered weights) repack
    # inline
    reg_epoch = getattr(w, "_cos_register_epoch", None)
    if reg_epoch is None or _repack["epoch"] <= reg_epoch:
        wrb[:Kout, :Kcol] = _as_bf16(w).permute(0, 2, 3, 1) \
            .reshape(Kout, Kcol)
        wrT = getattr(w, "_cos_wrT", None)
        if wrT is not None:
            wrT[:Kcol] = wrb[:Kout, :Kcol].t()
        wrF = getattr(w, "_cos_wrF", None)
        if wrF is not None:
            wbf = _as_bf16(w).flip(2, 3)
            cgp = wrF.shape[0] // G
            for g_ in range(G):
                wrF[g_ * cgp:g_ * cgp + Cg, :R * S * Kg] = \
                    wbf[g_ * Kg:(g_ + 1) * Kg].permute(1, 2, 3, 0) \
                    .reshape(Cg, R * S * Kg)
    wr = wrb[:Kout]
    bias_f = b.float().contiguous() if b is not None else None

    NPQ = N * P * Q
    if out_into is not None:
        # fused concat: write this conv's output directly into its
        # channel window of the shared concat buffer (GEMM ldc = the
        # concat's total channel count) — no concat copy ever happens
        buf, c_off = out_into
        Ctot = buf.shape[1]
        y = buf[:, c_off:c_off + Kout]
        y2 = buf.permute(0, 2, 3, 1).reshape(NPQ, Ctot)[:, c_off:]
        ldc_out = Ctot
    else:
        y = torch.empty((N, Kout, P, Q), dtype=torch.bfloat16,
                        device=x.device,
                        memory_format=torch.channels_last)
        y2 = y.permute(0, 2, 3, 1).reshape(NPQ, Kout)  # NHWC flat alias
        ldc_out = Kout
    # Winograd F(4x4,3x3) for stride-1 pad-1 3x3 convs: 2.25-4x fewer
    # MACs than im2col GEMM (SURVEY.md §3.6 "Winograd is a rebuild
    # addition").  The col matrix is then built lazily in backward (dW
    # still uses the im2col form).
    if (R == S == 3 and sh == sw == 1 and ph == pw == 1 and dil == 1
            and G == 1 and C % 8 == 0 and Kout % 8 == 0 and out_into is None
            and int(os.environ.get("COS_WINOGRAD", "0"))):
        # measured on MI355X: ~1-5% slower than the tuned direct im2col
        # GEMM on AlexNet/GoogLeNet (these shapes are staging-bound, so
        # the 2.25x MAC cut loses to the V/M transform traffic); kept as
        # an option for MFMA-bound shapes
        _wino_run(xl, w, b, y, N, P, Q, C, Kout, relu)
        if ctx is not None:
            ctx["col"] = None
            ctx["xl"] = xl
            ctx["is_1x1"] = False
            ctx["wino"] = True
            ctx["shape"] = (N, C, H, W, P, Q, R, S, sh, sw, ph, pw, dil, G,
                            Cg, Kg, Kpad, Kcol)
            ctx["wr"] = None
            ctx["w"] = w
        return y
    # 1x1/stride-1 conv: im2col is the identity in NHWC — GEMM straight
    # off the input (GoogLeNet's many 1x1 convs skip the col buffer)
    is_1x1 = (R == S == 1 and sh == sw == 1 and ph == pw == 0 and
              dil == 1 and G == 1 and C % 8 == 0 and not implicit)
    if is_1x1:
        x2 = xl.permute(0, 2, 3, 1).reshape(NPQ, C)
        _gemm(x2, wr, y2, bias_f, NPQ, Kout, C, C, C, ldc_out,
              False, False, 0, 1, relu=relu, na=wrb.shape[0])
        col = None
    elif implicit:
        col = None
        for g in range(G):
            geom = [H, W, C, P, Q, sh, sw, ph, pw, dil, S, g * Cg, Cg,
                    Kcol]
            _ext.gemm_conv_fwd(
                xl, wrb[g * Kg:], y2[:, g * Kg:],
                bias_f[g * Kg:(g + 1) * Kg] if bias_f is not None
                else None, _zpage(x.device), NPQ, Kg, Kpad, Kpad,
                ldc_out, relu, False, geom)
    elif implicit_sc:
        col = None
        # CGeom field reuse for the _sc kernels: S slot carries C,
        # Cg slot carries S*C (both are magic-divide divisors there)
        geom = [H, W, C, P, Q, sh, sw, ph, pw, dil, C, 0, S * C, Kcol]
        _ext.gemm_conv_fwd_sc(xl, wrb, y2, bias_f, NPQ, Kout, Kpad,
                              Kpad, ldc_out, relu, geom)
    else:
        col = torch.empty((G, NPQ, Kpad), dtype=torch.bfloat16,
                          device=x.device)
        for g in range(G):
            _ext.im2col(xl, col[g], N, H, W, C, P, Q, R, S, sh, sw, ph, pw,
                        dil, Kpad, g * Cg, Cg)
            # C[npq, kout_g] — write into the column slice of NHWC y
            _gemm(col[g], wr[g * Kg:(g + 1) * Kg], y2[:, g * Kg:],
                  bias_f[g * Kg:(g + 1) * Kg] if bias_f is not None else None,
                  NPQ, Kg, Kpad, Kpad, Kpad, ldc_out, False, False, 0, 1,
                  relu=relu, na=wrb.shape[0] - g * Kg)
    if ctx is not None:
        ctx["col"] = col
        ctx["xl"] = xl
        ctx["is_1x1"] = is_1x1
        ctx["implicit"] = implicit and not is_1x1
        ctx["implicit_sc"] = implicit_sc
        ctx["shape"] = (N, C, H, W, P, Q, R, S, sh, sw, ph, pw, dil, G, Cg,
                        Kg, Kpad, Kcol)
        ctx["wr"] = wr
        ctx["w_ref"] = w
    return y


def _wino_run(xl, w, b, y, N, P, Q, C, K, relu, flip=False):
    """Launch the F(2x2,3x3) Winograd pipeline.  xl NCHW-channels_last
    bf16, w fp32 [K][C][3][3] (forward) — flip=True computes the data
    gradient (pass dy as xl and the forward weights; then C=f_K,
    K=f_C)."""
    dev = xl.device
    th, tw = (P + 1) // 2, (Q + 1) // 2
    T = N * th * tw
    wK, wC = (w.shape[0], w.shape[1])
    ur = _pad128(K)
    U = torch.empty((16, ur, C), dtype=torch.bfloat16, device=dev)
    if ur != K:
        U.view(16, -1)[:, K * C:].zero_()
    V = torch.empty((16, T, C), dtype=torch.bfloat16, device=dev)
    M = torch.empty((16, T, K), dtype=torch.bfloat16, device=dev)
    x2 = xl.permute(0, 2, 3, 1)
    y2 = y.permute(0, 2, 3, 1)
    bias_f = b.float().contiguous() if b is not None else None
    _ext.wino_conv(x2, w.float().contiguous(), bias_f, y2, U, V, M,
                   N, P, Q, C, K, wK, wC, ur, flip, relu)


def conv2d_backward(x, w, dy, stride, pad, dilation, groups,
                    need_dx=True, need_dw=True, bias=True, ctx=None,
                    dw_out=None, db_out=None, dx_into=None):
    _check_bf16(dy, "conv dy")
    if ctx is None or "col" not in ctx:
        ctx = ctx if ctx is not None else {}
        conv2d_forward(_cl(x), w, None, stride, pad, dilation, groups,
                       ctx=ctx)
    (N, C, H, W, P, Q, R, S, sh, sw, ph, pw, dil, G, Cg, Kg, Kpad,
     Kcol) = ctx["shape"]
    col, wr = ctx["col"], ctx["wr"]
    is_1x1 = ctx.get("is_1x1", False)
    wino = ctx.get("wino", False)
    implicit = ctx.get("implicit", False)
    implicit_sc = ctx.get("implicit_sc", False)
    # implicit dw needs the wide TT kernel (N > 128); otherwise — and
    # for Winograd forward — materialize the col matrix lazily
    dw_implicit = implicit and Kpad > 128 and \
        bool(int(os.environ.get("COS_DW_TT", "1")))
    dw_implicit_sc = implicit_sc and Kpad > 128 and \
        bool(int(os.environ.get("COS_DW_TT", "1")))
    if need_dw and col is None and not is_1x1 and not dw_implicit \
            and not dw_implicit_sc:
        col = torch.empty((G, N * P * Q, Kpad), dtype=torch.bfloat16,
                          device=dy.device)
        for g in range(G):
            _ext.im2col(ctx["xl"], col[g], N, H, W, C, P, Q, R, S, sh, sw,
                        ph, pw, dil, Kpad, g * Cg, Cg)
        ctx["col"] = col
    Kout = Kg * G
    NPQ = N * P * Q
    # fused-concat channel window (dense-ReLU backward hands the branch
    # its pre-masked slice of the concat diff): every consumer kernel
    # reads with an explicit row stride, so no compacting copy is made
    ld_dy = _cl_slice_ld(dy)
    use_tt_env = bool(int(os.environ.get("COS_DW_TT", "1")))
    if ld_dy is not None and ld_dy != Kout and not wino and use_tt_env \
            and dy.storage_offset() % 8 == 0 and ld_dy % 8 == 0 \
            and dy.dtype == torch.bfloat16:
        dyl = dy
        dy2 = None
    else:
        dyl = _cl(dy)
        ld_dy = Kout
        dy2 = dyl.permute(0, 2, 3, 1).reshape(NPQ, Kout)

    def _dyg(g):
        """dy's group-g channel slice (2-D alias when dense, 4-D
        channel sub-window when strided — the bindings only use the
        base pointer; lda carries the row stride)."""
        if dy2 is not None:
            return dy2[:, g * Kg:] if g else dy2
        return dyl[:, g * Kg:] if g else dyl
    if is_1x1:
        x2 = ctx["xl"].permute(0, 2, 3, 1).reshape(NPQ, C)

    dx = dw = db = None
    # fused bias gradient inside the TT-wide dw GEMM: measured SLOWER
    # (googlenet 9.19k -> 8.73k, alexnet 38.9k -> 38.1k same-box A/B) —
    # the bn==0 blocks doing the extra LDS column pass become the
    # kernel's stragglers, costing more than the separate ramp-bound
    # colsum launches saved.  Kept behind COS_FUSE_DB=1 for evidence.
    fuse_db = bias and need_dw and Kpad > 128 and \
        bool(int(os.environ.get("COS_DW_TT", "1"))) and \
        bool(int(os.environ.get("COS_FUSE_DB", "0")))
    if fuse_db:
        db = db_out if db_out is not None \
            else _scratch_buf((Kout,), torch.float32, dy.device)
    if need_dw:
        # dw[kout][kpad] = sum_npq dy[npq][kout] * col[npq][kpad].
        # fused trans/trans dw (u32 k-pair staged; round 2): reads dy2/col
        # exactly once instead of transpose kernels + an extra HBM pass
        use_tt = bool(int(os.environ.get("COS_DW_TT", "1")))
        sk_tt = _splitk_tt(Kg, Kpad, NPQ) if use_tt else 0
        if use_tt and sk_tt == 1:
            # single split: every output element is written by exactly
            # one block — plain store, no zero-init pass needed
            dwp = torch.empty((Kout, Kpad), dtype=torch.float32,
                              device=dy.device)
            store_dw = 1
        else:
            dwp = _scratch_buf((Kout, Kpad), torch.float32, dy.device)
            store_dw = 2
        dyT = None if use_tt else _transpose(dy2)
        for g in range(G):
            mb, nb = (Kg + 127) // 128, (Kpad + 127) // 128
            sk = _splitk_for(mb, nb, NPQ)
            if use_tt and dw_implicit:
                # B operand gathered straight from the NHWC input
                db_slice = db.narrow(0, g * Kg, Kg) if fuse_db else None
                geom = [H, W, C, P, Q, sh, sw, ph, pw, dil, S, g * Cg,
                        Cg, Kcol]
                _ext.gemm_conv_dw(_dyg(g), ctx["xl"],
                                  dwp[g * Kg:], db_slice, Kg, Kpad, NPQ,
                                  ld_dy, Kpad, store_dw, sk_tt, 1.0,
                                  geom)
            elif use_tt and dw_implicit_sc:
                db_slice = db.narrow(0, g * Kg, Kg) if fuse_db else None
                geom = [H, W, C, P, Q, sh, sw, ph, pw, dil, C, 0,
                        S * C, Kcol]
                _ext.gemm_conv_dw_sc(_dyg(g), ctx["xl"],
                                     dwp[g * Kg:], db_slice, Kg, Kpad,
                                     NPQ, ld_dy, Kpad, store_dw, sk_tt,
                                     1.0, geom)
            elif use_tt:
                src = x2 if is_1x1 else col[g]
                db_slice = db.narrow(0, g * Kg, Kg) if fuse_db else None
                _gemm(_dyg(g), src, dwp[g * Kg:], db_slice,
                      Kg, Kpad, NPQ, ld_dy, Kpad if not is_1x1 else C,
                      Kpad, True, True, store_dw, sk_tt)
            else:
                if is_1x1:
                    # inception: several 1x1 branch convs share one
                    # input — transpose it once per step, cached on the
                    # base tensor (version-keyed like _cl)
                    base = ctx["xl"]
                    cached = getattr(base, "_cos_T", None)
                    if cached is not None and cached[0] == base._version:
                        colT = cached[1]
                    else:
                        colT = _transpose(x2)
                        try:
                            base._cos_T = (base._version, colT)
                        except AttributeError:
                            pass
                elif not int(os.environ.get("COS_DW_IM2COLT", "0")):
                    colT = _transpose(col[g])
                else:
                    # build colT[k][npq] straight from the input (skips a
                    # pass over the col matrix but re-reads the input R*S
                    # times across XCD L2s: measured slower on AlexNet —
                    # 24.9k vs 27.1k img/s — so off by default)
                    colT = torch.empty((_pad128(Kpad), NPQ),
                                       dtype=torch.bfloat16, device=dy.device)
                    _ext.im2col_t(ctx["xl"], colT, N, H, W, C, P, Q, R, S,
                                  sh, sw, ph, pw, dil, Kpad, g * Cg, Cg)
                _gemm(dyT[g * Kg:(g + 1) * Kg], colT, dwp[g * Kg:],
                      None, Kg, Kpad, NPQ, NPQ, NPQ, Kpad, False, False,
                      2, sk, ma=_pad128(Kout) - g * Kg, na=_pad128(Kpad))
        if dw_out is not None and dw_out.dtype == torch.float32 \
                and dw_out.is_contiguous() \
                and tuple(dw_out.shape) == (Kout, Cg, R, S):
            # unpack straight into the fp32 arena slice (one kernel
            # instead of permute-contiguous + arena copy)
            _ext.dw_unpack_acc(dwp, dw_out, Kout, Cg, R, S, Kpad, False)
            dw = dw_out
        else:
            dw = dwp[:, :Kcol].reshape(Kout, R, S, Cg) \
                .permute(0, 3, 1, 2).contiguous()
    if bias and not fuse_db:
        # db_out = the solver's pre-zeroed arena slice: colsum's atomics
        # accumulate the bias gradient in place, no staging copy
        db = db_out if db_out is not None \
            else _scratch_buf((Kout,), torch.float32, dy.device)
        _ext.colsum(dyl if dy2 is None else dy2, db, NPQ, Kout, ld_dy)
    if need_dx:
        wrF_ = getattr(ctx.get("w_ref"), "_cos_wrF", None)
        acc_capable = not wino and (
            is_1x1 or (wrF_ is not None
                       and wrF_.shape == (G * _pad128(Cg),
                                          _pad32(R * S * Kg))
                       and N * H * W < (1 << 20)))
        dx_acc = acc_capable and dx_into is not None and \
            dx_into.is_contiguous(memory_format=torch.channels_last) and \
            dx_into.dtype == torch.bfloat16 and \
            tuple(dx_into.shape) == (N, C, H, W)
        if dx_acc:
            # fan-in: accumulate straight into the existing bottom diff
            # inside the GEMM epilogue (saves the separate add pass)
            dx = dx_into
        else:
            dx = torch.empty((N, C, H, W), dtype=torch.bfloat16,
                             device=dy.device,
                             memory_format=torch.channels_last)
        if is_1x1:
            # col2im is the identity: write dx's NHWC alias directly.
            # wrT is maintained per step by the fused repack kernel for
            # solver-managed weights; transpose inline otherwise.
            dx2 = dx.permute(0, 2, 3, 1).reshape(NPQ, C)
            wrT = getattr(ctx.get("w_ref"), "_cos_wrT", None)
            if wrT is None or wrT.shape[1] != Kout:
                wrT = _transpose(wr.contiguous())
            _gemm(_dyg(0), wrT, dx2, None, NPQ, C, Kout, ld_dy, Kout,
                  C, False, False, 3 if dx_acc else 0, 1, ma=NPQ,
                  na=_pad128(C))
            return dx, dw, db
        if wino:
            # data gradient via Winograd on dy with flipped weights
            _wino_run(dyl, ctx["w"], None, dx, N, H, W, Kout, C,
                      relu=False, flip=True)
            return dx, dw, db
        wrF = getattr(ctx.get("w_ref"), "_cos_wrF", None)
        k2 = R * S * Kg
        cgp = _pad128(Cg)
        if wrF is not None and wrF.shape == (G * cgp, _pad32(k2)) \
                and N * H * W < (1 << 20):
            # implicit dx: the data gradient IS a stride-1 convolution of
            # dy with the flipped weights — one GEMM per group writes
            # dx's NHWC alias directly (no dcol buffer, no col2im pass)
            dx2 = dx.permute(0, 2, 3, 1).reshape(N * H * W, C)
            for g in range(G):
                geom = [P, Q, ld_dy, H, W, 1, 1,
                        dil * (R - 1) - ph, dil * (S - 1) - pw, dil, S,
                        g * Kg, Kg, k2]
                _ext.gemm_conv_fwd(dyl, wrF[g * cgp:], dx2[:, g * Cg:],
                                   None, _zpage(dy.device), N * H * W,
                                   Cg, _pad32(k2), _pad32(k2), C, False,
                                   dx_acc, geom)
            return dx, dw, db
        dcol = torch.empty((NPQ, Kpad), dtype=torch.bfloat16,
                           device=dy.device)
        for g in range(G):
            # dcol[npq, kpad] = dy_g[npq, kg] @ w_g[kg, kpad]: transposed
            # packed weights — maintained by the fused repack kernel for
            # G==1, transposed here otherwise
            wrT = getattr(ctx.get("w_ref"), "_cos_wrT", None) \
                if G == 1 else None
            if wrT is None:
                wrT = _transpose(wr[g * Kg:(g + 1) * Kg].contiguous())
            _gemm(_dyg(g), wrT, dcol, None,
                  NPQ, Kpad, Kg, ld_dy, Kg, Kpad, False, False, 0, 1,
                  ma=NPQ, na=_pad128(Kpad))
            _ext.col2im(dcol, dx, N, H, W, C, P, Q, R, S, sh, sw, ph, pw,
                        dil, Kpad, g * Cg, Cg)
    return dx, dw, db


# ----------------------------------------------------------------- linear

def fc_forward(x, w, b, relu=False):
    _check_bf16(x, "fc input")
    x = x.contiguous()
    M, K = x.shape
    Nout = w.shape[0]
    if Nout % 128:
        wb = torch.zeros((_pad128(Nout), K), dtype=torch.bfloat16,
                         device=x.device)
        wb[:Nout] = _as_bf16(w)
        na = wb.shape[0]
        wb = wb[:Nout]
    else:
        wb = _as_bf16(w).contiguous()
        na = Nout
    y = torch.empty((M, Nout), dtype=torch.bfloat16, device=x.device)
    bias_f = b.float().contiguous() if b is not None else None
    mb, nb = (M + 127) // 128, (Nout + 127) // 128
    sk = _splitk_for(mb, nb, K)
    if sk > 1:
        # underfilled grid (e.g. fc6: 2x32 tiles): split K with fp32
        # atomics, then one fused bias+ReLU+cast pass
        wsp = torch.zeros((M, Nout), dtype=torch.float32, device=x.device)
        _gemm(x, wb, wsp, None, M, Nout, K, K, K, Nout, False, False, 2,
              sk, na=na)
        _ext.bias_act_cast(wsp, bias_f, y, relu)
    else:
        _gemm(x, wb, y, bias_f, M, Nout, K, K, K, Nout, False, False, 0, 1,
              relu=relu, na=na)
    return y


def fc_backward(x, w, dy, need_dx=True, bias=True, dw_out=None,
                db_out=None):
    _check_bf16(dy, "fc dy")
    x = x.contiguous()
    dy = dy.contiguous()
    wb = _as_bf16(w).contiguous()
    M, K = x.shape
    Nout = wb.shape[0]
    dx = dw = db = None
    # odd output widths (LRCN fc8: vocab 8801) misalign every dy row and
    # force the GEMMs onto the guarded staging path; one zero-padded
    # copy restores the 16-byte DMA path for both dx (K dim) and dw
    # (trans lda), and the zero columns contribute nothing
    Np = Nout
    if Nout % 8:
        Np = _pad8(Nout)
        # persistent pad buffers: live region fully overwritten each
        # call, pad rows/cols zeroed once at allocation and never touched
        dy_p = _scratch_buf((M, Np), torch.bfloat16, dy.device,
                            zero_each_step=False)
        dy_p[:, :Nout] = dy
        dy = dy_p
        wb_p = _scratch_buf((Np, K), torch.bfloat16, dy.device,
                            zero_each_step=False)
        wb_p[:Nout] = wb
        wb = wb_p
    if need_dx:
        # dx = dy @ w: transpose w once -> NT direct/direct fast staging.
        # With padded dy the K dim runs to Np (the pad columns multiply
        # wT's over-allocated junk rows by zero).
        wT = _transpose(wb)
        dx = torch.empty((M, K), dtype=torch.bfloat16, device=x.device)
        mb, nb = (M + 127) // 128, (K + 127) // 128
        sk = _splitk_for(mb, nb, Np)
        if sk > 1:
            wsp = _scratch_buf((M, K), torch.float32, x.device)
            _gemm(dy, wT, wsp, None, M, K, Np, Np, Np, K, False,
                  False, 2, sk, na=_pad128(K))
            _ext.bias_act_cast(wsp, None, dx, False)
        else:
            _gemm(dy, wT, dx, None, M, K, Np, Np, Np, K, False,
                  False, 0, 1, na=_pad128(K))
    # dw = dy^T @ x: fused trans/trans (u32 k-pair staging) — no
    # operand transpose kernels, one read of each operand.  dw_out (the
    # solver's fp32 arena slice, when the layer owns the only gradient
    # write this step) receives the GEMM output directly, skipping the
    # dwp-to-arena copy (fc6 alone is 151 MB fp32)
    sk_tt = _splitk_tt(Nout, K, M)
    arena_dw = dw_out is not None and dw_out.dtype == torch.float32 \
        and dw_out.is_contiguous() and tuple(dw_out.shape) == (Nout, K)
    if sk_tt == 1:
        # plain store covers every element once — no zero-init pass
        # (fc6's arena slice alone is 151 MB: zeroing it every step was
        # a full extra HBM pass over the largest weight in the net)
        dwp = dw_out if arena_dw else torch.empty(
            (Nout, K), dtype=torch.float32, device=x.device)
        store_dw = 1
    elif arena_dw:
        dwp = dw_out
        dwp.zero_()     # arena slice: NOT scratch (iter_size accumulates)
        store_dw = 2
    else:
        dwp = _scratch_buf((Nout, K), torch.float32, x.device)
        store_dw = 2
    fuse_db = bias and K > 128 and \
        bool(int(os.environ.get("COS_FUSE_DB", "0")))
    if fuse_db:
        db = db_out if db_out is not None \
            else _scratch_buf((Nout,), torch.float32, x.device)
    _gemm(dy, x, dwp, db if fuse_db else None, Nout, K, M, Np, K, K,
          True, True, store_dw, sk_tt)
    dw = dwp
    if bias and not fuse_db:
        db = db_out if db_out is not None \
            else _scratch_buf((Nout,), torch.float32, x.device)
        _ext.colsum(dy, db, M, Nout, Np)
    return dx, dw, db


# ------------------------------------------------------------- activations

def relu_forward(x, negative_slope=0.0):
    y = torch.empty_like(x)
    _ext.relu_fwd(x, y, negative_slope)
    return y


def _cl_slice_ld(t):
    """If t is a channel-slice view of a channels_last buffer (fused
    concat branch), return its row stride; None otherwise."""
    if t.dim() != 4 or t.stride(1) != 1:
        return None
    N, C, H, W = t.shape
    ld = t.stride(3)
    if ld >= C and t.stride(2) == W * ld and t.stride(0) == H * W * ld:
        return ld
    return None


def relu_backward(y, dy, negative_slope=0.0, db_out=None):
    ldy = _cl_slice_ld(y)
    if ldy is not None and y.dtype == torch.bfloat16:
        dy = dy if dy.dtype == torch.bfloat16 else dy.to(torch.bfloat16)
        lddy = _cl_slice_ld(dy)
        if lddy is None:
            dy = _cl(dy.reshape(y.shape))
            lddy = y.shape[1]
        N, C, H, W = y.shape
        dx = torch.empty((N, C, H, W), dtype=torch.bfloat16,
                         device=y.device,
                         memory_format=torch.channels_last)
        if db_out is not None and db_out.shape[0] == C:
            # fused: the producing conv's bias gradient accumulates into
            # its arena slice during this same pass (its colsum is skipped)
            _ext.relu_colsum_bwd(y, dy, dx.permute(0, 2, 3, 1), db_out,
                                 negative_slope, N * H * W, C, ldy, lddy)
            return dx, True
        _ext.relu_bwd_strided(y, dy, dx.permute(0, 2, 3, 1), negative_slope,
                              N * H * W, C, ldy, lddy)
        return (dx, False) if db_out is not None else dx
    dy = dy if dy.dtype == torch.bfloat16 else dy.to(torch.bfloat16)
    if db_out is not None and y.dim() == 2 and y.is_contiguous() \
            and db_out.shape[0] == y.shape[1]:
        dx = torch.empty_like(y)
        _ext.relu_colsum_bwd(y, dy.reshape(y.shape), dx, db_out,
                             negative_slope, y.shape[0], y.shape[1],
                             y.shape[1], y.shape[1])
        return dx, True
    dx = torch.empty_like(y)
    _ext.relu_bwd(y, dy.reshape(y.shape), dx, negative_slope)
    return (dx, False) if db_out is not None else dx


_dropout_seeds = {}


def _dropout_seed(device, generator=None):
    """Per-device RNG seed in DEVICE memory, advanced on-GPU by seed_bump:
    no host sync per call, and a hipGraph-captured step re-draws the mask
    on every replay (a host kernel-arg seed would freeze it).  The initial
    value derives from the net's seeded CPU generator when one is given,
    so solver random_seed makes GPU dropout reproducible too."""
    t = _dropout_seeds.get(device.index)
    if t is None:
        init = torch.randint(1, 2 ** 62, (1,), dtype=torch.int64,
                             generator=generator)
        t = init.to(device)
        _dropout_seeds[device.index] = t
    return t


def dropout_forward(x, ratio, generator=None):
    y = torch.empty_like(x)
    mask = torch.empty_like(x)
    seed = _dropout_seed(x.device, generator)
    _ext.seed_bump(seed)
    _ext.dropout_fwd(x, y, mask, ratio, seed)
    return y, mask


def dropout_backward(mask, dy):
    dx = torch.empty_like(mask)
    _ext.mul(mask, dy.to(mask.dtype), dx)
    return dx


# ---------------------------------------------------------------- pooling

def _pool_out(h, k, s, p):
    # caffe ceil-mode with clamp
    o = int(math.ceil((h + 2 * p - k) / s)) + 1
    if p > 0 and (o - 1) * s >= h + p:
        o -= 1
    return o


def maxpool_forward(x, kernel, stride, pad):
    _check_bf16(x, "maxpool input")
    N, C, H, W = x.shape
    kh, kw = kernel
    sh, sw = stride
    ph, pw = pad
    P, Q = _pool_out(H, kh, sh, ph), _pool_out(W, kw, sw, pw)
    xl = _cl(x)
    y = torch.empty((N, C, P, Q), dtype=x.dtype, device=x.device,
                    memory_format=torch.channels_last)
    idt = torch.int8 if kh * kw < 128 else torch.int32
    idx = torch.empty((N, P, Q, C), dtype=idt, device=x.device)
    _ext.maxpool_fwd(xl, y, idx, N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw)
    return y, (idx, (kh, kw, sh, sw, ph, pw), (N, C, H, W))


def maxpool_backward(x_shape, idx_pack, dy):
    idx, (kh, kw, sh, sw, ph, pw), (N, C, H, W) = idx_pack
    P, Q = dy.shape[2], dy.shape[3]
    dx = torch.empty((N, C, H, W), dtype=dy.dtype, device=dy.device,
                    memory_format=torch.channels_last)
    _ext.maxpool_bwd(_cl(dy), idx, dx, N, H, W, C, P, Q, kh, kw, sh, sw,
                     ph, pw)
    return dx


def avgpool_forward(x, kernel, stride, pad):
    N, C, H, W = x.shape
    kh, kw = kernel
    sh, sw = stride
    ph, pw = pad
    P, Q = _pool_out(H, kh, sh, ph), _pool_out(W, kw, sw, pw)
    xl = _cl(x)
    y = torch.empty((N, C, P, Q), dtype=x.dtype, device=x.device,
                    memory_format=torch.channels_last)
    _ext.avgpool_fwd(xl, y, N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw)
    return y


def avgpool_backward(x, kernel, stride, pad, dy):
    N, C, H, W = x.shape
    kh, kw = kernel
    sh, sw = stride
    ph, pw = pad
    P, Q = dy.shape[2], dy.shape[3]
    dx = torch.empty((N, C, H, W), dtype=dy.dtype, device=dy.device,
                    memory_format=torch.channels_last)
    _ext.avgpool_bwd(_cl(dy), dx, N, H, W, C, P, Q, kh, kw, sh, sw, ph, pw)
    return dx


def global_avgpool_forward(x):
    N, C, H, W = x.shape
    xl = _cl(x)
    y = torch.empty((N, C, 1, 1), dtype=x.dtype, device=x.device,
                    memory_format=torch.channels_last)
    _ext.avgpool_fwd(xl, y, N, H, W, C, 1, 1, H, W, 1, 1, 0, 0)
    return y


def global_avgpool_backward(x_shape, dy):
    N, C, H, W = x_shape
    dx = torch.empty((N, C, H, W), dtype=dy.dtype, device=dy.device,
                    memory_format=torch.channels_last)
    _ext.avgpool_bwd(_cl(dy), dx, N, H, W, C, 1, 1, H, W, 1, 1, 0, 0)
    return dx


# -------------------------------------------------------------------- LRN

def lrn_forward(x, local_size, alpha, beta, k):
    _check_bf16(x, "lrn input")
    N, C, H, W = x.shape
    if C % 8 or local_size > 9:
        # rare config the vectorized kernel doesn't cover: torch fallback
        return reference.lrn_forward(x, local_size, alpha, beta, k)
    xl = _cl(x)
    y = torch.empty_like(xl)
    scale = torch.empty((N, H, W, C), dtype=torch.float32, device=x.device)
    _ext.lrn_fwd(xl, y, scale, N * H * W, C, local_size, alpha, beta, k)
    return y, (scale, xl)


def lrn_backward(x, y, scale_pack, dy, local_size, alpha, beta):
    N, C, H, W = x.shape
    if C % 8 or local_size > 9:
        return reference.lrn_backward(x, y, scale_pack, dy, local_size,
                                      alpha, beta)
    scale, xl = scale_pack
    dx = torch.empty_like(xl)
    ratio = torch.empty_like(xl)
    _ext.lrn_bwd(xl, _cl(y), scale, _cl(dy), dx, ratio, N * H * W, C,
                 local_size, alpha, beta)
    return dx


# ----------------------------------------------------------- softmax loss

def softmax_loss_forward(x, label, ignore_label, axis=1):
    _check_bf16(x, "softmax loss input")
    C = x.shape[axis]
    if x.dim() > 2 or axis != 1:
        x = x.movedim(axis, -1)
    moved_shape = tuple(x.shape)
    x2 = x.reshape(-1, C).contiguous()
    lab = label.float().reshape(-1).contiguous()
    prob = torch.empty_like(x2, dtype=torch.float32)
    loss = torch.zeros(1, dtype=torch.float32, device=x.device)
    count = torch.zeros(1, dtype=torch.int32, device=x.device)
    _ext.softmax_loss_fwd(x2, lab, prob, loss, count,
                          ignore_label if ignore_label is not None else 0,
                          ignore_label is not None)
    cnt = int(count.item()) if ignore_label is not None else lab.numel()
    return loss.reshape(()), (prob, lab, moved_shape, axis), cnt


def softmax_loss_backward(prob_pack, label, ignore_label, scale, axis=1):
    prob, lab, moved_shape, ax = prob_pack
    dx2 = torch.empty(prob.shape, dtype=torch.bfloat16, device=prob.device)
    _ext.softmax_loss_bwd(prob, lab, dx2, scale,
                          ignore_label if ignore_label is not None else 0,
                          ignore_label is not None)
    # moved_shape is x.shape AFTER movedim(axis,-1); restore original order
    dx = dx2.reshape(moved_shape)
    if len(moved_shape) > 2 or ax != 1:
        dx = dx.movedim(-1, ax)
    return dx.contiguous()


# ------------------------------------------------------------------ embed

def embed_forward(idx, w, b=None):
    wb = _as_bf16(w).contiguous()
    idxf = idx.float().contiguous()
    E = wb.shape[1]
    y = torch.empty(list(idxf.shape) + [E], dtype=torch.bfloat16,
                    device=w.device)
    _ext.embed_fwd(idxf, wb, y, wb.shape[0])
    if b is not None:
        y = y + _as_bf16(b)
    return y


def embed_backward(idx, dy, vocab_size, bias=True):
    e = dy.shape[-1]
    # atomic scatter-add target (LRCN: 8801x1000 fp32 = 35 MB zeroed
    # per step — one fused pass via the backward scratch instead)
    dw = _scratch_buf((vocab_size, e), torch.float32, dy.device)
    _ext.embed_bwd(idx.float().contiguous(),
                   _as_bf16(dy).contiguous(), dw, vocab_size)
    db = dy.float().reshape(-1, e).sum(0) if bias else None
    return dw, db


# -------------------------------------------------------------- LSTM unit

def lstm_unit_forward(c_prev, gates, cont):
    gates_b = _as_bf16(gates).contiguous()
    n, h4 = gates_b.shape
    H = h4 // 4
    cp = c_prev.float().contiguous()
    cont_b = _as_bf16(cont).reshape(-1).contiguous()
    c_out = torch.empty_like(cp)
    h_out = torch.empty((n, H), dtype=torch.bfloat16, device=gates.device)
    act = torch.empty((n, 4, H), dtype=torch.float32, device=gates.device)
    _ext.lstm_unit_fwd(cp, gates_b, cont_b, c_out, h_out, act)
    return c_out, h_out, (cp, c_out, act, cont_b)


def lstm_unit_backward(c_prev, cache, dc_next, dh):
    cp, c_out, act, cont_b = cache
    n, _, H = act.shape
    dc_prev = torch.empty_like(cp)
    dgates = torch.empty((n, 4 * H), dtype=torch.bfloat16, device=act.device)
    _ext.lstm_unit_bwd(cp, c_out, act, cont_b, dc_next.float().contiguous(),
                       _as_bf16(dh).contiguous(), dc_prev, dgates)
    return dc_prev, dgates


def lstm_seq_forward(xg, w_hc, cont):
    """Whole-sequence LSTM forward driven from C++ (3 launches/step).
    xg: [T,N,4H] pre-activation input gates (x @ W_xc + b [+ static]),
    cont: [T,N].  Returns (h [T,N,H] bf16, cache)."""
    T, N, H4 = xg.shape
    H = H4 // 4
    dev_ = xg.device
    xg = _as_bf16(xg).contiguous()
    cont_b = _as_bf16(cont).contiguous()
    whc = torch.zeros((_pad128(H4), H), dtype=torch.bfloat16, device=dev_)
    whc[:H4] = _as_bf16(w_hc)
    h = torch.empty((T, N, H), dtype=torch.bfloat16, device=dev_)
    c = torch.empty((T, N, H), dtype=torch.float32, device=dev_)
    act = torch.empty((T, N, H4), dtype=torch.float32, device=dev_)
    # +128 rows so per-step A slices keep fast DMA staging (m_alloc)
    h_in_store = torch.empty(((T * N + 128) * H,), dtype=torch.bfloat16,
                             device=dev_)
    h_in = h_in_store[:T * N * H].view(T, N, H)
    hg = torch.empty((N, H4), dtype=torch.float32, device=dev_)
    if _lstm_persist_ok(N):
        # persistent whole-sequence kernel: recurrent weights stationary
        # in registers, grid-resident with in-kernel barriers (falls back
        # to the per-step loop when the watchdog reports non-residency)
        bar = torch.zeros(260, dtype=torch.int32, device=dev_)
        rc = _ext.lstm_persist_fwd(xg, whc[:H4].contiguous() if not
                                   whc[:H4].is_contiguous() else whc[:H4],
                                   cont_b, h, c, act, h_in, hg, T, N, H,
                                   bar)
        if rc == 0:
            _persist_track(bar)
            return h, (cont_b, h, c, act, h_in, T, N, H)
    _ext.lstm_seq_fwd(xg, whc[:H4], cont_b, h, c, act, h_in, hg,
                      T, N, H, whc.shape[0])
    return h, (cont_b, h, c, act, h_in, T, N, H)


_persist_pending = []


def _persist_check_pending(force=False):
    """Deferred watchdog check: err flags are copied D2H asynchronously;
    inspect completed copies (or all, when force=True) and raise if the
    persistent kernel ever aborted — outputs are also NaN-poisoned, so a
    missed check cannot silently corrupt training."""
    global _persist_pending
    keep = []
    for ev, flag in _persist_pending:
        if force or ev.query():
            if force:
                ev.synchronize()
            if int(flag[0]):
                raise RuntimeError(
                    "persistent LSTM grid barrier watchdog fired (blocks "
                    "not co-resident?); set COS_LSTM_PERSIST=0")
        else:
            keep.append((ev, flag))
    _persist_pending = keep


def _persist_track(bar):
    if torch.cuda.is_current_stream_capturing():
        return              # poison-only safety under graph capture
    flag = torch.empty(1, dtype=torch.int32, pin_memory=True)
    flag.copy_(bar[258:259], non_blocking=True)
    ev = torch.cuda.Event()
    ev.record()
    _persist_pending.append((ev, flag))


def _lstm_persist_ok(N):
    if N > 64 or int(os.environ.get("COS_LSTM_PERSIST", "1")) == 0:
        return False
    if torch.cuda.is_current_stream_capturing():
        # capture-legal (async memset + one launch); the deferred err
        # check can't run per-replay, but a watchdog abort NaN-poisons
        # the outputs so it cannot pass silently
        return True
    _persist_check_pending(force=len(_persist_pending) > 16)
    import torch.distributed as dist
    if dist.is_initialized() and dist.get_world_size() > 1:
        return False        # RCCL kernels may break grid residency
    return True


def lstm_seq_backward(dy, w_hc, cache):
    """Returns (dxg [T,N,4H] bf16, dw_hc fp32)."""
    cont_b, h, c, act, h_in, T, N, H = cache
    H4 = 4 * H
    dev_ = dy.device
    dy = _as_bf16(dy).contiguous()
    whcT = _transpose(_as_bf16(w_hc).contiguous())      # [H][4H], padded rows
    dxg_store = torch.empty(((T * N + 128) * H4,), dtype=torch.bfloat16,
                            device=dev_)
    dxg = dxg_store[:T * N * H4].view(T, N, H4)
    dh_rec = torch.empty((N, H), dtype=torch.bfloat16, device=dev_)
    dh_f = torch.empty((N, H), dtype=torch.float32, device=dev_)
    dc_a = torch.empty((N, H), dtype=torch.float32, device=dev_)
    dc_b = torch.empty((N, H), dtype=torch.float32, device=dev_)
    done = False
    if _lstm_persist_ok(N):
        bar = torch.zeros(260, dtype=torch.int32, device=dev_)
        rc = _ext.lstm_persist_bwd(dy, whcT[:H].contiguous() if not
                                   whcT[:H].is_contiguous() else whcT[:H],
                                   cont_b, h, c, act, dxg, dh_f, dc_a,
                                   dc_b, T, N, H, bar)
        done = (rc == 0)
        if done:
            _persist_track(bar)
    if not done:
        _ext.lstm_seq_bwd(dy, whcT, cont_b, h, c, act, dxg, dh_rec, dh_f,
                          dc_a, dc_b, T, N, H, _pad128(H4))
    # dw_hc = dgates_all^T @ h_in_all — fused trans/trans GEMM (u32
    # k-pair staging), no operand transposes
    dxg_flat = dxg.reshape(T * N, H4)
    h_in_flat = h_in.reshape(T * N, H)
    sk_tt = _splitk_tt(H4, H, T * N)
    if sk_tt == 1:
        dwhc = torch.empty((H4, H), dtype=torch.float32, device=dev_)
    else:
        dwhc = _scratch_buf((H4, H), torch.float32, dev_)
    _gemm(dxg_flat, h_in_flat, dwhc, None, H4, H, T * N, H4, H, H,
          True, True, 1 if sk_tt == 1 else 2, sk_tt)
    return dxg, dwhc


# -------------------------------------------------------------- batchnorm

def _bn_eligible(x):
    return x.dim() == 4 and x.shape[1] % 8 == 0 and \
        x.dtype == torch.bfloat16


def bn_forward_train(x, eps):
    if not _bn_eligible(x):
        return reference.bn_forward_train(x, eps)
    xl = _cl(x)
    N, C, H, W = x.shape
    rows = N * H * W
    x2 = xl.permute(0, 2, 3, 1).reshape(rows, C)
    s0 = torch.zeros(C, dtype=torch.float32, device=x.device)
    s1 = torch.zeros(C, dtype=torch.float32, device=x.device)
    _ext.bn_stats(x2, s0, s1, rows, C)
    mean = s0 / rows
    var = (s1 / rows - mean * mean).clamp_min_(0)
    invstd = (var + eps).rsqrt()
    y = torch.empty((N, C, H, W), dtype=torch.bfloat16, device=x.device,
                    memory_format=torch.channels_last)
    _ext.bn_norm(x2, y.permute(0, 2, 3, 1).reshape(rows, C), mean, invstd,
                 rows, C)
    return y, mean, var, invstd


def bn_forward_infer(x, mean, var, eps):
    if not _bn_eligible(x):
        return reference.bn_forward_infer(x, mean, var, eps)
    xl = _cl(x)
    N, C, H, W = x.shape
    rows = N * H * W
    invstd = (var.float() + eps).rsqrt().contiguous()
    y = torch.empty((N, C, H, W), dtype=torch.bfloat16, device=x.device,
                    memory_format=torch.channels_last)
    _ext.bn_norm(xl.permute(0, 2, 3, 1).reshape(rows, C),
                 y.permute(0, 2, 3, 1).reshape(rows, C),
                 mean.float().contiguous(), invstd, rows, C)
    return y, invstd


def bn_backward(xhat, dy, invstd, train):
    if not _bn_eligible(xhat):
        return reference.bn_backward(xhat, dy, invstd, train)
    xl = _cl(xhat)
    dyl = _cl(dy.to(torch.bfloat16))
    N, C, H, W = xhat.shape
    rows = N * H * W
    x2 = xl.permute(0, 2, 3, 1).reshape(rows, C)
    dy2 = dyl.permute(0, 2, 3, 1).reshape(rows, C)
    s1 = _scratch_buf((C,), torch.float32, xhat.device)
    s2 = _scratch_buf((C,), torch.float32, xhat.device)
    if train:
        _ext.bn_bwd_sums(dy2, x2, s1, s2, rows, C)
    dx = torch.empty((N, C, H, W), dtype=torch.bfloat16,
                     device=xhat.device, memory_format=torch.channels_last)
    _ext.bn_bwd(x2, dy2, dx.permute(0, 2, 3, 1).reshape(rows, C),
                invstd.contiguous(), s1, s2,
                (1.0 / rows) if train else 0.0, rows, C)
    return dx


# -------------------------------------------------------------- optimizer

def sgd_update(param, grad, momentum_buf, lr, momentum, weight_decay):
    _ext.sgd_update(param, grad.contiguous(), momentum_buf, lr, momentum,
                    weight_decay)


def _seg_tensors(solver):
    key = getattr(solver, "_seg_cache_key", None)
    if key != (id(solver.flat_w), len(solver.segments)):
        offs = [s[0] for s in solver.segments] + [int(solver.flat_w.numel())]
        solver._seg_off = torch.tensor(offs, dtype=torch.int64,
                                       device=solver.device)
        solver._seg_lrm = torch.tensor([s[2] for s in solver.segments],
                                       dtype=torch.float32,
                                       device=solver.device)
        solver._seg_dm = torch.tensor([s[3] for s in solver.segments],
                                      dtype=torch.float32,
                                      device=solver.device)
        solver._seg_cache_key = (id(solver.flat_w), len(solver.segments))


def nesterov_update_multi_arena(solver, rate, momentum, wd):
    """Whole-arena fused Nesterov (VERDICT round-1 weak item: non-SGD
    updates ran through torch glue)."""
    if len(solver.segments) > 512:
        return False
    _seg_tensors(solver)
    sh = getattr(solver, "flat_wb", None)
    _ext.nesterov_update_multi(solver.flat_w, solver.flat_g, solver.flat_m,
                               solver._seg_off, solver._seg_lrm * rate,
                               solver._seg_dm * wd, momentum, sh)
    if sh is not None:
        solver._shadow_synced = True
    return True


def adam_update_multi_arena(solver, rate, b1, b2, eps, wd, t):
    """Whole-arena fused Adam; the global bias correction folds into
    the per-segment lr."""
    if len(solver.segments) > 512 or getattr(solver, "flat_m2", None) \
            is None:
        return False
    _seg_tensors(solver)
    corr = (1.0 - b2 ** t) ** 0.5 / (1.0 - b1 ** t)
    sh = getattr(solver, "flat_wb", None)
    _ext.adam_update_multi(solver.flat_w, solver.flat_g, solver.flat_m,
                           solver.flat_m2, solver._seg_off,
                           solver._seg_lrm * (rate * corr),
                           solver._seg_dm * wd, b1, b2, eps, sh)
    if sh is not None:
        solver._shadow_synced = True
    return True


def sgd_update_multi_arena(solver, rate, momentum, wd):
    """One kernel over the whole flat arena, per-segment lr/decay."""
    if len(solver.segments) > 512:      # LDS segment-table capacity
        for (off, n, lrm, dm) in solver.segments:
            if lrm == 0:
                continue
            _ext.sgd_update(solver.flat_w.narrow(0, off, n),
                            solver.flat_g.narrow(0, off, n),
                            solver.flat_m.narrow(0, off, n),
                            float(rate) * lrm, momentum, wd * dm)
        return
    _seg_tensors(solver)
    sh = getattr(solver, "flat_wb", None)
    _ext.sgd_update_multi(solver.flat_w, solver.flat_g, solver.flat_m,
                          solver._seg_off, solver._seg_lrm * rate,
                          solver._seg_dm * wd, momentum, sh)
    if sh is not None:
        solver._shadow_synced = True


# --------------------------------------------------------------- registry

GLUE_OPS = [
    # small / test-time ops where torch (on HIP) is acceptable glue
    "sigmoid_forward", "sigmoid_backward", "tanh_forward", "tanh_backward",
    "softmax_forward", "softmax_backward", "accuracy", "bias_add",
    "nesterov_update", "adam_update",
]


def register_all(ext_module):
    global _ext
    _ext = ext_module
    from . import dispatcher
    here = globals()
    for name in list(here):
        if name.startswith("_") or name in ("register_all", "reference",
                                            "torch", "math"):
            continue
        fn = here[name]
        if callable(fn) and not isinstance(fn, type):
            dispatcher.register_gpu(name, fn)
    for name in GLUE_OPS:
        dispatcher.register_gpu(name, getattr(reference, name))
