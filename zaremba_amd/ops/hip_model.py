"""HipModel — the MI355X execution runtime attached to a Model.

Owns, per model instance:
  * bf16 shadow weights (+ pre-transposed shadows for the backward data
    GEMMs, and the folded b_x+b_h bias vector the input GEMM consumes),
    kept in sync with the fp32 masters by the fused clip+SGD kernels,
  * persistent per-layer workspaces (gx, h_all, c_all, gates, dG, ...)
    sized on first use so the C++ LSTM sequence drivers can hipGraph-
    capture the per-timestep launch trains against stable pointers,
  * the philox dropout offset counter (host-side: dropout is never
    hipGraph-captured),
  * the fused clip+SGD step (grad-norm^2 reduce -> per-param update
    rewriting master fp32 + bf16 shadow in one pass -> transposed-shadow
    refresh kernels).
"""

from __future__ import annotations

import os
from typing import Optional

import torch

from .. import _C  # noqa
from .hip_ops import DropoutFn, EmbeddingFn, LinearFn, LstmLayerFn


def _ks(k):
    return (k + 31) // 32


def _pad64(k):
    return ((k + 63) // 64) * 64


def _slacked(shape, dtype, device, slack_ptrs=None):
    """Allocate a zeroed tensor with 64 extra zeroed elements of storage
    slack after it. Tensors used as the A operand of a k_pad GEMM (see
    ext_bind.hip:gemm) over-read up to 128 bytes past their last row;
    the slack keeps that read in-bounds and finite (zero)."""
    n = 1
    for s in shape:
        n *= s
    flat = torch.zeros(n + 64, dtype=dtype, device=device)
    if slack_ptrs is not None:
        slack_ptrs.add(flat.untyped_storage().data_ptr())
    return flat[:n].view(*shape)


class _LayerWorkspace:
    def __init__(self, T: int, B: int, H: int, device, slack_ptrs=None):
        bf, f32 = torch.bfloat16, torch.float32
        self.T, self.B, self.H = T, B, H
        self.gx = torch.zeros(T, B, 4 * H, dtype=bf, device=device)
        # h_all / dG are slack-provisioned: in eval mode h_all[1:] is the
        # next consumer GEMM's A operand; dG is the dx GEMM's A operand.
        self.h_all = _slacked((T + 1, B, H), bf, device, slack_ptrs)
        self.c_all = torch.zeros(T + 1, B, H, dtype=f32, device=device)
        self.gates = torch.zeros(T, B, 4 * H, dtype=bf, device=device)
        self.dY = torch.zeros(T, B, H, dtype=bf, device=device)
        self.dG = _slacked((T, B, 4 * H), bf, device, slack_ptrs)
        # K-split hop partials: up to 4 slices (ZAREMBA_AMD_BWD_KSPLIT);
        # the per-step fallback pair uses slices 0/1
        self.dh_rec = torch.zeros(4, B, H, dtype=f32, device=device)
        self.dc = torch.zeros(B, H, dtype=f32, device=device)
        # fragment-packed workspaces (zero-prefilled: pad rows/K-tails
        # must read as 0.0 in the packed cell kernels)
        self.h_pack = torch.zeros(T + 1, _ks(H) * 2 * 64 * 8, dtype=bf,
                                  device=device)
        # per-t packed dgates: the persistent backward's in-launch
        # exchange buffers (fresh slot per step); the per-step fallback
        # reuses slot 0
        self.dG_pack = torch.zeros(T, _ks(4 * H) * 2 * 64 * 8, dtype=bf,
                                   device=device)
        # zero-padded transposed temporaries for the dW GEMMs: K (=T*B)
        # padded to a BK multiple so no GEMM runs a K-tail tile; pad
        # columns stay zero (transpose writes only [:, :T*B])
        TBp = ((T * B + 63) // 64) * 64
        self.TBp = TBp
        self.dGT = torch.zeros(4 * H, TBp, dtype=bf, device=device)
        self.hpT = torch.zeros(H, TBp, dtype=bf, device=device)
        # persistent-kernel state: block records [T][NB][B][6][HS],
        # barrier words (re-zeroed per call by the driver) + abort flag
        hs = _C.ext().persistent_hs(H)
        nb = (H + hs - 1) // hs
        self.rec = torch.zeros(T * nb * B * 6 * hs, dtype=bf, device=device)
        # hgran: fwd padded-barrier words (uint32[0, 513)) + fused-bwd
        # pair counters (uint32[544 .. 544+ceil(H/16)))
        self.hgran = torch.zeros(768, dtype=torch.int64, device=device)
        self.abort = torch.zeros(1, dtype=torch.int32, device=device)


class _LayerRuntime:
    """Shadows + workspace for one LSTM layer."""

    def __init__(self, layer, device):
        self.H = layer.hidden_size
        self.Hin = layer.input_size
        H = self.H
        bf = torch.bfloat16
        self.Wx = torch.empty_like(layer.W_x, dtype=bf, device=device)
        self.Wh = torch.empty_like(layer.W_h, dtype=bf, device=device)
        # K-padded dx-GEMM operand: WxT gets zero pad columns up to H4p
        # for free from transpose_bf16's dst-stride, so the dx GEMM runs
        # tail-free (measured ~2 us/launch on the 64-tile shapes; padding
        # the forward GEMMs' Wx/fc.W was measured a net LOSS — the
        # per-step padded-shadow copies cost ~15 us each, more than the
        # 5.7 us tail they remove — and is not done).
        self.H4p = _pad64(4 * H)
        self.WxT = torch.zeros(self.Hin, self.H4p, dtype=bf, device=device)
        self.WhT = torch.empty(H, 4 * H, dtype=bf, device=device)
        # fragment-packed W_h (forward cell) and W_h^T (backward hop)
        nb = (H + 15) // 16
        self.WhP = torch.empty(nb * 4 * _ks(H) * 64 * 8, dtype=bf,
                               device=device)
        self.WhTP = torch.empty(nb * _ks(4 * H) * 64 * 8, dtype=bf,
                                device=device)
        self.bias_sum = torch.empty(4 * H, dtype=torch.float32,
                                    device=device)
        self.ws: Optional[_LayerWorkspace] = None
        self.slack_ptrs = None  # shared set, attached by HipModel
        self.dx_part = None  # f32 split-K partial pair, sized on first bwd
        self.xT_buf = None   # persistent [Hin, TBp] transposed-x temp
        self.owner = None    # back-ref to the HipModel (side stream)

    @torch.no_grad()
    def refresh(self, layer, e):
        self.Wx.copy_(layer.W_x)
        self.Wh.copy_(layer.W_h)
        e.transpose_bf16(self.Wx, self.WxT)
        e.transpose_bf16(self.Wh, self.WhT)
        e.pack_gated_w(self.Wh, self.WhP, self.H, 4, self.H)
        e.pack_gated_w(self.WhT, self.WhTP, self.H, 1, 4 * self.H)
        torch.add(layer.b_x, layer.b_h, out=self.bias_sum)

    def ensure_ws(self, T, B, device, slack_ptrs=None):
        if self.ws is None or self.ws.T != T or self.ws.B != B:
            self.ws = _LayerWorkspace(T, B, self.H, device, slack_ptrs)
        return self.ws


class _FcRuntime:
    def __init__(self, fc, device):
        bf = torch.bfloat16
        V, H = fc.W.shape
        self.W = torch.empty(V, H, dtype=bf, device=device)
        # K-padded backward-dx operand (see _LayerRuntime): WT carries
        # zero pad columns up to Vp from its strided transpose refresh.
        self.Vp = _pad64(V)
        self.WT = torch.zeros(H, self.Vp, dtype=bf, device=device)
        self.dscT = None  # [V, TBp] zero-padded, sized on first backward
        self.xT = None    # [H, TBp]
        self.dsc_buf = None  # slacked [N, V] bf16, sized on first backward
        self.dx_part = None  # f32 split-K partial pair, sized on first bwd
        self.slack_ptrs = None  # shared set, attached by HipModel
        self.owner = None    # back-ref to the HipModel (side stream)

    @torch.no_grad()
    def refresh(self, fc, e):
        self.W.copy_(fc.W)
        e.transpose_bf16(self.W, self.WT)


class HipModel:
    def __init__(self, model):
        self.model = model
        self.device = next(model.parameters()).device
        if self.device.type != "cuda":
            raise RuntimeError("HipModel requires a ROCm GPU device")
        self.e = _C.ext()
        if os.environ.get("ZAREMBA_AMD_GRAPHS", "1") == "0":
            self.e.set_use_graphs(False)
        if os.environ.get("ZAREMBA_AMD_PERSISTENT", "1") == "0":
            self.e.set_use_persistent(False)
        if os.environ.get("ZAREMBA_AMD_FUSED_BWD", "1") == "0":
            self.e.set_use_fused_bwd(False)
        self.e.set_bwd_ksplit(
            int(os.environ.get("ZAREMBA_AMD_BWD_KSPLIT", "2")))
        self.e.set_bwd_threads(
            int(os.environ.get("ZAREMBA_AMD_BWD_WAVES", "16")) * 64)
        if os.environ.get("ZAREMBA_AMD_BWD_BATCH2", "0") == "1":
            self.e.set_bwd_batch2(True)
        self.compute_dtype = torch.bfloat16
        dev = self.device
        self.emb_W = torch.empty_like(model.embed.W, dtype=torch.bfloat16,
                                      device=dev)
        # Storages with >=128 B of zeroed slack past the tensor end: only
        # these may be the A operand of a k_pad GEMM (ext_bind.hip:gemm).
        self.slack_ptrs = set()
        self._slack_bufs = {}
        self.layers = [_LayerRuntime(l, dev) for l in model.rnns]
        for rt in self.layers:
            rt.slack_ptrs = self.slack_ptrs
        self.fc = _FcRuntime(model.fc, dev)
        self.fc.slack_ptrs = self.slack_ptrs
        # Derive the philox seed from the torch generator so --seed
        # controls the HIP dropout masks too (round-2 fix: secrets-based
        # seeding made same-seed GPU runs draw different masks). Resume
        # restores the torch RNG, so a rebuilt model draws a
        # deterministic seed there as well.
        self.dropout_seed = int(torch.randint(
            0, (1 << 62) - 1, (1,), dtype=torch.int64).item())
        self.dropout_counter = 0  # host-side philox quad offset
        self.norm2 = torch.zeros(1, dtype=torch.float32, device=dev)
        self._shadows_fresh = False
        # Side stream for off-critical-path backward work (weight-grad
        # transposes/GEMMs/colsums), overlapping the next layer's fused
        # backward train; clip+SGD joins the recorded events. MEASURED
        # AND REJECTED as the default (ZAREMBA_AMD_SIDE_STREAM=1 to
        # re-enable): 302K vs 316K tokens/s OFF on the same box — the
        # concurrent GEMM blocks contend with the co-resident spin
        # kernels (slower pair partners extend every spin) and the
        # event plumbing adds host latency to a launch-bound train.
        # Also disabled under DP (the bucketer's post-accumulate hooks
        # fire on the main stream and would race). NOTE when enabled:
        # code reading .grad directly after backward() (instead of
        # through sgd_step) must torch.cuda.synchronize() first.
        self.side_stream = None
        self.side_events: list = []
        if os.environ.get("ZAREMBA_AMD_SIDE_STREAM", "0") == "1":
            import torch.distributed as td
            if not (td.is_available() and td.is_initialized()
                    and td.get_world_size() > 1):
                self.side_stream = torch.cuda.Stream(device=dev)
        for rt in self.layers:
            rt.owner = self
        self.fc.owner = self

    def _bump_dropout(self, numel: int) -> int:
        """Reserve a philox quad range for one dropout call and return
        its starting offset (host-side stream bookkeeping)."""
        off = self.dropout_counter
        self.dropout_counter += (numel + 3) // 4
        return off

    def set_compute_dtype(self, dtype):
        if dtype != torch.bfloat16:
            raise ValueError("HIP engine currently runs bf16 compute "
                             "(fp32 master weights); use --engine eager for "
                             "full-fp32 debugging")
        self.compute_dtype = dtype

    # ------------------------------------------------------------------
    @torch.no_grad()
    def refresh_shadows(self):
        m = self.model
        self.emb_W.copy_(m.embed.W)
        for rt, layer in zip(self.layers, m.rnns):
            rt.refresh(layer, self.e)
        self.fc.refresh(m.fc, self.e)
        self._shadows_fresh = True

    def invalidate_shadows(self):
        self._shadows_fresh = False

    def check_aborts(self):
        """Raise if any grid-synchronized kernel hit its bounded-spin
        timeout (the abort flag makes blocks EXIT instead of hanging the
        GPU, but the step's outputs are then garbage). Called from the
        trainer's log steps, which synchronize anyway — failure detection
        at zero steady-state cost."""
        for rt in self.layers:
            if rt.ws is not None and int(rt.ws.abort.item()) != 0:
                raise RuntimeError(
                    "persistent/fused LSTM kernel aborted (bounded-spin "
                    "timeout): a co-residency or synchronization failure; "
                    "results of the affected step are invalid. Set "
                    "ZAREMBA_AMD_PERSISTENT=0 / ZAREMBA_AMD_FUSED_BWD=0 to "
                    "fall back to per-step kernels.")

    def slack_buf(self, key, shape, dtype):
        """Per-call-site cached activation buffer with zeroed storage
        slack (valid to reuse every step: the training loop is strictly
        fwd -> bwd -> step, the module-docstring aliasing contract)."""
        k = (key, tuple(shape), dtype)
        buf = self._slack_bufs.get(k)
        if buf is None:
            buf = _slacked(shape, dtype, self.device, self.slack_ptrs)
            self._slack_bufs[k] = buf
        return buf

    # ------------------------------------------------------------------
    def forward(self, x, states, training: bool):
        if not self._shadows_fresh:
            self.refresh_shadows()
        m = self.model
        T, B = x.shape
        if B > 32:
            # Every cell kernel (persistent, fused backward, per-step
            # fallback) tiles the batch into the 32-row MFMA pair;
            # larger per-GPU batches are out of contract — fail loudly
            # instead of corrupting memory.
            raise RuntimeError(
                f"HIP engine supports batch_size <= 32 per GPU (32-row "
                f"MFMA tiling); got {B}. Use --engine eager, or shard "
                f"the batch across GPUs with data parallelism.")
        p = m.dropout_p
        idx = x.reshape(-1)
        H = m.hidden_size
        bf = torch.bfloat16
        emb = EmbeddingFn.apply(m.embed.W, idx, self.emb_W,
                                [self.slack_buf("emb", (T * B, H), bf)])
        cur = emb.view(T, B, H)
        if training and p > 0:
            cur = DropoutFn.apply(cur, p, self.dropout_seed,
                                  self._bump_dropout(cur.numel()),
                                  [self.slack_buf(("drop", 0), cur.shape, bf)])
        new_states = list(states)
        for i, (rt, layer) in enumerate(zip(self.layers, m.rnns)):
            rt.ensure_ws(T, B, self.device, self.slack_ptrs)
            h0, c0 = states[i]
            out, hT, cT = LstmLayerFn.apply(cur, h0, c0, layer.W_x,
                                            layer.W_h, layer.b_x, layer.b_h,
                                            rt)
            new_states[i] = (hT, cT)
            cur = out
            if training and p > 0:
                cur = DropoutFn.apply(cur, p, self.dropout_seed,
                                      self._bump_dropout(cur.numel()),
                                      [self.slack_buf(("drop", i + 1),
                                                      cur.shape, bf)])
        scores = LinearFn.apply(cur.reshape(T * B, m.hidden_size), m.fc.W,
                                m.fc.b, self.fc)
        for i in range(len(new_states)):
            states[i] = new_states[i]
        return scores, states

    # ------------------------------------------------------------------
    @torch.no_grad()
    def clip_and_sgd(self, lr: float, max_norm: float,
                     grad_scale: float = 1.0):
        """Fused global grad-norm clip + SGD (reference main.py:115-117).

        norm^2 accumulated across every param grad on device; each
        parameter's update also rewrites its bf16 shadow; transposed
        shadows + folded bias refreshed afterwards. Returns the pre-clip
        grad norm as a 0-d device tensor (sync only when printed).
        """
        m = self.model
        e = self.e
        # join the side-stream weight-grad work before any grad is read
        cur = torch.cuda.current_stream()
        if self.side_events:
            for ev in self.side_events:
                cur.wait_event(ev)
            self.side_events.clear()
        self.norm2.zero_()
        params = [p for p in m.parameters() if p.grad is not None]
        if self.side_stream is not None:
            # caching-allocator cross-stream rule: grads allocated on the
            # side stream are consumed here on the main stream
            for p in params:
                p.grad.record_stream(cur)
        shadow_of = {id(m.embed.W): self.emb_W, id(m.fc.W): self.fc.W}
        for rt, layer in zip(self.layers, m.rnns):
            shadow_of[id(layer.W_x)] = rt.Wx
            shadow_of[id(layer.W_h)] = rt.Wh
        # Chunked multi-tensor descriptors: one norm launch + one update
        # launch for all params. Grad tensors are fresh per step, but the
        # caching allocator cycles through a small set of addresses, so
        # descriptor tensors are memoized per pointer signature.
        key = tuple(p.grad.data_ptr() for p in params)
        cache = getattr(self, "_mt_cache", None)
        if cache is None:
            cache = self._mt_cache = {}
        descs = cache.get(key)
        if descs is None:
            # 64K chunks for BOTH: half-size norm chunks measured a
            # 0.3-0.6% END-TO-END loss in the same-box A/B matrix
            # (profiles/s17_ab.txt) despite the latency-bound-looking
            # norm2 counters; ZAREMBA_AMD_NORM_CHUNK for A/B.
            CH = 65536
            CHN = int(os.environ.get("ZAREMBA_AMD_NORM_CHUNK", "65536"))
            nd, sd = [], []
            for p in params:
                mp, gp = p.data.data_ptr(), p.grad.data_ptr()
                sh = shadow_of.get(id(p))
                sp = sh.data_ptr() if sh is not None else 0
                n = p.numel()
                for off in range(0, n, CHN):
                    nd.append((gp + off * 4, min(CHN, n - off)))
                for off in range(0, n, CH):
                    ln = min(CH, n - off)
                    sd.append((mp + off * 4, gp + off * 4,
                               sp + off * 2 if sp else 0, ln))
            descs = (
                torch.tensor(nd, dtype=torch.int64).to(self.device),
                torch.tensor(sd, dtype=torch.int64).to(self.device),
            )
            if len(cache) > 32:
                cache.clear()
            cache[key] = descs
        e.norm2_mt(descs[0], self.norm2)
        e.sgd_mt(descs[1], self.norm2, max_norm, lr, grad_scale)
        # refresh the derived shadows (transposes + packs + folded biases)
        for rt, layer in zip(self.layers, m.rnns):
            e.transpose_bf16(rt.Wx, rt.WxT)
            e.transpose_bf16(rt.Wh, rt.WhT)
            e.pack_gated_w(rt.Wh, rt.WhP, rt.H, 4, rt.H)
            e.pack_gated_w(rt.WhT, rt.WhTP, rt.H, 1, 4 * rt.H)
            torch.add(layer.b_x, layer.b_h, out=rt.bias_sum)
        e.transpose_bf16(self.fc.W, self.fc.WT)
        return (self.norm2.sqrt() * grad_scale).reshape(())
