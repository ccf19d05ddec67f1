"""Autograd wiring for the HIP kernel path.

Each Function routes forward AND backward through the hand-written gfx950
kernels (zaremba_amd/csrc/). Parameters enter as fp32 masters (autograd
edges / checkpoint format); compute uses the bf16 shadow copies held by
the HipModel runtime, and backward returns fp32 grads that accumulate
into the masters' .grad (which the fused clip+SGD kernels consume).

Workspace-aliasing contract: the per-layer LSTM workspaces (h_all, c_all,
gates, dG, ...) are persistent and re-used every step — valid because the
training loop is strictly fwd -> bwd -> step (one outstanding autograd
graph), which is the reference's own structure (main.py:108-117).
"""

from __future__ import annotations

import torch

from .. import _C


def ext():
    return _C.ext()


class EmbeddingFn(torch.autograd.Function):
    """K1: gather fwd / fp32 scatter-add bwd (reference model.py:6-17)."""

    @staticmethod
    def forward(ctx, W_master, idx, shadow_W, out=None):
        # out: None or a 1-list [buf] (see DropoutFn)
        N = idx.numel()
        H = W_master.size(1)
        out = (torch.empty(N, H, dtype=torch.bfloat16, device=idx.device)
               if out is None else out[0])
        ext().embedding_fwd(shadow_W, idx, out)
        ctx.save_for_backward(idx)
        ctx.V = W_master.size(0)
        return out

    @staticmethod
    def backward(ctx, dY):
        (idx,) = ctx.saved_tensors
        dY = dY.contiguous()
        dW = torch.zeros(ctx.V, dY.size(1), dtype=torch.float32,
                         device=dY.device)
        ext().embedding_bwd(dY, idx, dW)
        return dW, None, None, None


class DropoutFn(torch.autograd.Function):
    """K5: philox inverted dropout; the mask is regenerated in backward
    from (seed, offset) — no mask tensor stored. The philox offset is a
    HOST-side counter (a plain int advanced by the caller): dropout is
    never hipGraph-captured, so the round-1 device-counter + tick-kernel
    machinery was pure overhead (a tick launch + a 1-elem alloc per
    call) and was removed."""

    @staticmethod
    def forward(ctx, x, p, seed, offset, out=None):
        # out: None or a 1-list [buf] (a non-Tensor holder, so autograd
        # does not treat the reused buffer as an aliased input)
        x = x.contiguous()
        y = torch.empty_like(x) if out is None else out[0].view_as(x)
        ext().dropout_fwd(x, y, p, seed, offset)
        ctx.p = p
        ctx.seed = seed
        ctx.offset = offset
        return y

    @staticmethod
    def backward(ctx, dy):
        dy = dy.contiguous()
        dx = torch.empty_like(dy)
        ext().dropout_bwd(dy, dx, ctx.p, ctx.seed, ctx.offset)
        return dx, None, None, None, None


class LinearFn(torch.autograd.Function):
    """K6: output projection via the NT MFMA GEMM; scores fp32
    (reference model.py:57-71)."""

    @staticmethod
    def forward(ctx, x, W_master, b_master, fc_rt):
        # x: [N, H] bf16; fc_rt holds shadows W (bf16 [V,H]) and WT
        x = x.contiguous()
        N, H = x.shape
        V = W_master.size(0)
        scores = torch.empty(N, V, dtype=torch.float32, device=x.device)
        ext().gemm(x, fc_rt.W, scores, b_master, False, False)
        ctx.save_for_backward(x)
        ctx.fc_rt = fc_rt
        return scores

    @staticmethod
    def backward(ctx, dscores):
        (x,) = ctx.saved_tensors
        fc_rt = ctx.fc_rt
        e = ext()
        N, H = x.shape
        V = dscores.size(1)
        if fc_rt.dsc_buf is None or fc_rt.dsc_buf.shape != (N, V):
            from .hip_model import _slacked
            fc_rt.dsc_buf = _slacked((N, V), torch.bfloat16, x.device,
                                     fc_rt.slack_ptrs)
        dsc = fc_rt.dsc_buf
        dsc.copy_(dscores)
        dx = torch.empty(N, H, dtype=torch.bfloat16, device=x.device)
        # dx = dsc @ W: ~1 block/CU at 64-tile -> N-way split-K (co-
        # resident blocks interleave the latency chains; measured
        # near-free grid growth), f32 partials combined with one rounding
        if fc_rt.Vp >= 128:
            if fc_rt.dx_part is None or fc_rt.dx_part[0].shape != (N, H):
                z = torch.empty(N, H, dtype=torch.float32, device=x.device)
                fc_rt.dx_part = (z, torch.empty(3, N, H,
                                                dtype=torch.float32,
                                                device=x.device))
            p1, pex = fc_rt.dx_part
            nz = e.gemm_splitk(dsc, fc_rt.WT, p1, pex, None,
                               fc_rt.Vp if fc_rt.Vp != V else 0)
            e.addn_f32_bf16(p1, pex, dx, nz)
        else:
            e.gemm(dsc, fc_rt.WT, dx, None, False, False,
                   fc_rt.Vp if fc_rt.Vp != V else 0)
        # dW = dsc^T @ x via explicit transposes + the fast NT kernel (the
        # TN staging path is register-starved; measured 2-3x slower).
        # Transposed temps are zero-padded in K so the GEMM has no K-tail
        # tile (a tail tile measured +13.5 us). Off the critical path ->
        # side stream (overlaps the LSTM layers' fused backward trains;
        # see LstmLayerFn.backward).
        Np = ((N + 63) // 64) * 64
        if fc_rt.dscT is None or fc_rt.dscT.size(1) != Np:
            fc_rt.dscT = torch.zeros(V, Np, dtype=torch.bfloat16,
                                     device=x.device)
            fc_rt.xT = torch.zeros(H, Np, dtype=torch.bfloat16,
                                   device=x.device)
        dscores_c = dscores.contiguous()

        def dw_family():
            e.transpose_bf16(dsc, fc_rt.dscT)
            e.transpose_bf16(x, fc_rt.xT)
            dW = torch.empty(V, H, dtype=torch.float32, device=x.device)
            e.gemm(fc_rt.dscT, fc_rt.xT, dW, None, False, False)
            db = torch.zeros(V, dtype=torch.float32, device=x.device)
            e.colsum_f32(dscores_c, db)
            return dW, db

        hm = getattr(fc_rt, "owner", None)
        side = hm.side_stream if hm is not None else None
        if side is not None:
            ev0 = torch.cuda.Event()
            ev0.record()  # dsc / x / dscores ready on the main stream
            side.wait_event(ev0)
            with torch.cuda.stream(side):
                dW, db = dw_family()
                ev1 = torch.cuda.Event()
                ev1.record()
            hm.side_events.append(ev1)
        else:
            dW, db = dw_family()
        return dx, dW, db, None


class NllLossFn(torch.autograd.Function):
    """K7: fused stable log-softmax + batch_size-scaled NLL
    (reference main.py:77-84 semantics)."""

    @staticmethod
    def forward(ctx, scores, y, batch_size):
        scores = scores.contiguous()
        N, V = scores.shape
        yflat = y.reshape(-1).contiguous()
        lse = torch.empty(N, dtype=torch.float32, device=scores.device)
        accum = torch.zeros(1, dtype=torch.float32, device=scores.device)
        ext().lsm_nll_fwd(scores, yflat, lse, accum)
        ctx.save_for_backward(scores, lse, yflat)
        ctx.scale = float(batch_size) / N
        return (accum * ctx.scale).reshape(())

    @staticmethod
    def backward(ctx, grad_out):
        scores, lse, yflat = ctx.saved_tensors
        N, V = scores.shape
        up = grad_out.to(torch.float32).reshape(1).contiguous()
        dscores = torch.empty_like(scores)
        ext().lsm_nll_bwd(scores, lse, yflat, up, ctx.scale, dscores)
        return dscores, None, None


def nll_loss_hip(scores, y, batch_size):
    return NllLossFn.apply(scores, y, batch_size)


class LstmLayerFn(torch.autograd.Function):
    """K2-K4 + K8: one full LSTM layer unroll on the fused cell kernels.

    forward: hoisted input gate GEMM (x@W_x^T + b_x + b_h, MFMA NT), then
    the hipGraph-captured T-step fused-cell sequence.
    backward: graph-captured reverse unroll (dgate elementwise + skinny
    recurrent GEMM per step), then batched dW/dx GEMMs over the stacked
    [T*B, .] buffers.
    """

    @staticmethod
    def forward(ctx, x, h0, c0, Wx_m, Wh_m, bx_m, bh_m, rt):
        # x: [T, B, H_in] bf16; rt: _LayerRuntime with shadows + workspaces
        T, B, Hin = x.shape
        H = rt.H
        x2 = x.contiguous()
        e = ext()
        # input gate GEMM for the whole unroll: [T*B, 4H]
        e.gemm(x2.view(T * B, Hin), rt.Wx, rt.ws.gx.view(T * B, 4 * H),
               rt.bias_sum, False, False)
        rt.ws.h_all[0].copy_(h0.to(torch.bfloat16))
        rt.ws.c_all[0].copy_(c0.to(torch.float32))
        e.lstm_seq_fwd(rt.ws.gx, rt.Wh, rt.WhP, rt.ws.h_all, rt.ws.h_pack,
                       rt.ws.c_all, rt.ws.gates, rt.ws.rec, rt.ws.hgran,
                       rt.ws.abort)
        ctx.save_for_backward(x2)
        ctx.rt = rt
        out = rt.ws.h_all[1:]          # [T, B, H] bf16 view (aliases ws)
        hT = rt.ws.h_all[T]
        cT = rt.ws.c_all[T]
        ctx.mark_non_differentiable(hT, cT)
        return out, hT, cT

    @staticmethod
    def backward(ctx, dY, dhT, dcT):
        (x2,) = ctx.saved_tensors
        rt = ctx.rt
        e = ext()
        ws = rt.ws
        T, B, H = ws.dY.shape
        Hin = x2.size(2)
        # The graph-captured per-step path needs the stable ws.dY pointer;
        # the fused path launches eagerly, so a contiguous bf16 upstream
        # grad can be passed straight through (saves a 2x4.6 us copy).
        if (dY.dtype == torch.bfloat16 and dY.is_contiguous()
                and e.fused_bwd_active(B, H) and T >= 2):
            dyt = dY
        else:
            ws.dY.copy_(dY.to(torch.bfloat16))
            dyt = ws.dY
        e.lstm_seq_bwd(dyt, ws.gates, ws.rec, ws.c_all, rt.WhT, rt.WhTP,
                       ws.dG, ws.dG_pack, ws.dh_rec, ws.dc, ws.hgran,
                       ws.abort)
        TB = T * B
        dG2 = ws.dG.view(TB, 4 * H)

        # dW_h = dG^T @ h_prev_stack ; dW_x = dG^T @ x ; dx = dG @ W_x —
        # weight grads via explicit transposes + the fast NT kernel.
        # K (=T*B) is zero-padded in the persistent transposed temps so
        # no dW GEMM runs a K-tail tile. The whole dW family is OFF the
        # critical path (consumed only by clip+SGD), so it runs on the
        # HipModel side stream and overlaps the next layer's fused
        # backward train; clip_and_sgd joins the events.
        hm = getattr(rt, "owner", None)
        side = hm.side_stream if hm is not None else None
        if rt.xT_buf is None or rt.xT_buf.shape != (Hin, ws.TBp):
            rt.xT_buf = torch.zeros(Hin, ws.TBp, dtype=torch.bfloat16,
                                    device=x2.device)
        xT = rt.xT_buf

        def dw_family():
            dGT = ws.dGT
            e.transpose_bf16(dG2, dGT)
            hpT = ws.hpT
            e.transpose_bf16(ws.h_all[:T].reshape(TB, H), hpT)
            dWh = torch.empty(4 * H, H, dtype=torch.float32,
                              device=x2.device)
            e.gemm(dGT, hpT, dWh, None, False, False)
            e.transpose_bf16(x2.view(TB, Hin), xT)
            dWx = torch.empty(4 * H, Hin, dtype=torch.float32,
                              device=x2.device)
            e.gemm(dGT, xT, dWx, None, False, False)
            db = torch.zeros(4 * H, dtype=torch.float32, device=x2.device)
            e.colsum_bf16(dG2, db)  # grads of b_x and b_h are identical
            return dWh, dWx, db, db.clone()

        if side is not None:
            ev0 = torch.cuda.Event()
            ev0.record()  # dG / h_all / x2 are ready on the main stream
            side.wait_event(ev0)
            with torch.cuda.stream(side):
                dWh, dWx, db, db2 = dw_family()
                ev1 = torch.cuda.Event()
                ev1.record()
            hm.side_events.append(ev1)
        else:
            dWh, dWx, db, db2 = dw_family()

        dx = torch.empty(TB, Hin, dtype=torch.bfloat16, device=x2.device)
        # dG is always slack-provisioned workspace; WxT carries zero pad
        # columns up to H4p from its strided transpose refresh. N-way
        # split-K (see LinearFn.backward) on the ~1-block/CU shape.
        if rt.H4p >= 128:
            if rt.dx_part is None or rt.dx_part[0].shape != (TB, Hin):
                z = torch.empty(TB, Hin, dtype=torch.float32,
                                device=x2.device)
                rt.dx_part = (z, torch.empty(3, TB, Hin,
                                             dtype=torch.float32,
                                             device=x2.device))
            p1, pex = rt.dx_part
            nz = e.gemm_splitk(dG2, rt.WxT, p1, pex, None,
                               rt.H4p if rt.H4p != 4 * H else 0)
            e.addn_f32_bf16(p1, pex, dx, nz)
        else:
            e.gemm(dG2, rt.WxT, dx, None, False, False,
                   rt.H4p if rt.H4p != 4 * H else 0)
        return (dx.view(T, B, Hin), None, None, dWx, dWh, db, db2,
                None)
