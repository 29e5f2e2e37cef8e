"""Dispatching op API: HIP autograd.Functions on GPU, reference on CPU.

Each public function mirrors one row of SURVEY.md §2.6's kernel table
(K2 softlexicon, K3/K8 attention, K4 BiLSTM, K5/K6 CRF, K9 TENER,
K10 LayerNorm, K11 bias-GELU, K13 masked CE, K14 dice).
"""
from __future__ import annotations

import math
import os
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

from . import reference as ref
from . import hip_enabled, get_ext


def _pad_last(t: torch.Tensor, target: int) -> torch.Tensor:
    d = t.shape[-1]
    return t if d == target else F.pad(t, (0, target - d))


# ----------------------------------------------------------- layer norm
class _LayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, eps):
        y, mean, rstd = get_ext().layernorm_fwd(x, weight, bias, eps)
        ctx.save_for_backward(x, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = get_ext().layernorm_bwd(dy.contiguous(), x, weight, mean, rstd)
        return dx, dw, db, None


def layernorm(x, weight, bias, eps: float = 1e-12):
    if hip_enabled(x):
        shape = x.shape
        y = _LayerNormFn.apply(x.reshape(-1, shape[-1]).contiguous(), weight, bias, eps)
        return y.reshape(shape)
    return ref.layernorm(x, weight, bias, eps)


class _AddLayerNormFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, residual, weight, bias, eps):
        y, s, mean, rstd = get_ext().add_layernorm_fwd(x, residual, weight, bias, eps)
        ctx.save_for_backward(s, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, dy):
        s, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = get_ext().layernorm_bwd(dy.contiguous(), s, weight, mean, rstd)
        return dx, dx, dw, db, None


_rng_counter = None


def _rng_counter_for(device):
    global _rng_counter
    if _rng_counter is None or _rng_counter.device != device:
        seed = torch.initial_seed() & 0x7FFFFFFF
        _rng_counter = torch.tensor([seed], dtype=torch.int64, device=device)
    return _rng_counter


class _DropoutAddLNFn(torch.autograd.Function):
    """Fused dropout(x) + residual + LayerNorm (BERT post-LN); counter-
    based RNG bumps a device scalar so hipGraph replays draw fresh
    masks."""

    @staticmethod
    def forward(ctx, x, res, weight, bias, eps, keep):
        y, s, mask, mean, rstd = get_ext().dropout_add_ln_fwd(
            x, res, weight, bias, eps, keep,
            _rng_counter_for(x.device))
        ctx.save_for_backward(s, weight, mask, mean, rstd)
        ctx.keep = keep
        return y

    @staticmethod
    def backward(ctx, dy):
        s, weight, mask, mean, rstd = ctx.saved_tensors
        dsum, dw, db = get_ext().layernorm_bwd(dy.contiguous(), s, weight,
                                               mean, rstd)
        dx = get_ext().mask_scale(dsum, mask, 1.0 / ctx.keep)
        return dx, dsum, dw, db, None, None


def dropout_add_layernorm(x, residual, weight, bias, eps: float = 1e-12,
                          p: float = 0.1, training: bool = True):
    """LayerNorm(dropout(x) + residual) in one kernel (plus the shared LN
    backward) — replaces the separate dropout fwd + mask-mul bwd per
    encoder sublayer."""
    if (training and p > 0 and hip_enabled(x) and x.dtype == torch.bfloat16
            and os.environ.get("CHINESENER_NO_FUSED_DROPOUT") != "1"):
        shape = x.shape
        y = _DropoutAddLNFn.apply(
            x.reshape(-1, shape[-1]).contiguous(),
            residual.to(x.dtype).reshape(-1, shape[-1]).contiguous(),
            weight, bias, eps, 1.0 - p)
        return y.reshape(shape)
    if training and p > 0:
        x = F.dropout(x, p, training)
    return add_layernorm(x, residual, weight, bias, eps)


def add_layernorm(x, residual, weight, bias, eps: float = 1e-12):
    """LayerNorm(x + residual) — the BERT post-LN residual pattern."""
    if hip_enabled(x):
        shape = x.shape
        y = _AddLayerNormFn.apply(
            x.reshape(-1, shape[-1]).contiguous(),
            residual.to(x.dtype).reshape(-1, shape[-1]).contiguous(),
            weight, bias, eps)
        return y.reshape(shape)
    return ref.add_layernorm(x, residual, weight, bias, eps)


# ---------------------------------------------------------- fused linear
_GEMM_NT_CHOICE: dict = {}   # shape key -> bool (use in-tree kernel)
_GEMM_NT_CALLS: dict = {}    # shape key -> times seen
_PICK_MIN_RECUR = 6          # only time shapes that actually recur:
                             # ragged inputs (MRC's padded batches) make
                             # every step a fresh shape, and a timing
                             # race per step is slower than any GEMM


def _gemm_nt_mode() -> str:
    """'auto' (default): measure in-tree NT MFMA GEMM vs hipBLASLt once
    per shape and keep the winner; 'force': always in-tree; 'off'."""
    return os.environ.get("CHINESENER_GEMM_NT", "auto")


def _gemm_nt_ok(x2, w):
    return (x2.shape[0] % 128 == 0 and w.shape[0] % 128 == 0
            and w.shape[1] % 64 == 0 and x2.dtype == torch.bfloat16
            and w.dtype == torch.bfloat16)


def _tuning_active() -> bool:
    """hipBLASLt TunableOp tuning in progress: library timings are 2-3x
    inflated, so dispatch decisions must wait (bench.py freezes tuning
    after its warmup steps)."""
    try:
        import torch.cuda.tunable as tun
        return tun.is_enabled() and tun.tuning_is_enabled()
    except Exception:
        return False


def _med_time(fn):
    """Median kernel time: 3 batches of 8 launches per synchronize (a
    per-call sync adds a constant ~10-30us that swamps 20-80us kernels)."""
    import time as _time
    fn(); fn()
    torch.cuda.synchronize()
    ts = []
    for _ in range(3):
        t0 = _time.perf_counter()
        for _i in range(8):
            fn()
        torch.cuda.synchronize()
        ts.append((_time.perf_counter() - t0) / 8)
    return sorted(ts)[1]


def _pick2(key, fn_ours, fn_lib) -> bool:
    """Measured dispatch between two GEMM paths: once the library's
    TunableOp tuning is done, time both (median of 5) and cache the
    winner."""
    hit = _GEMM_NT_CHOICE.get(key)
    if hit is not None:
        return hit
    if torch.cuda.is_current_stream_capturing() or _tuning_active():
        return False               # never measure now; decide later
    n = _GEMM_NT_CALLS.get(key, 0) + 1
    _GEMM_NT_CALLS[key] = n
    if n < _PICK_MIN_RECUR:
        return False               # shape hasn't proven it recurs

    t_ours, t_lib = _med_time(fn_ours), _med_time(fn_lib)
    _GEMM_NT_CHOICE[key] = bool(t_ours <= t_lib)
    if os.environ.get("CHINESENER_GEMM_DEBUG") == "1":
        import sys
        print(f"[gemm-dispatch] {key}: ours {t_ours*1e6:.1f}us "
              f"lib {t_lib*1e6:.1f}us -> "
              f"{'ours' if _GEMM_NT_CHOICE[key] else 'lib'}",
              file=sys.stderr, flush=True)
    return _GEMM_NT_CHOICE[key]


def _pick_gemm_nt(x2, w, bf) -> bool:
    """Measured dispatch: first time a (M,N,K,bias) shape shows up, time
    both paths (median of 5 after warmup) and cache the winner. The
    hand-written kernel carries every shape it wins on; hipBLASLt keeps
    the rest (scripts/bench_gemm_nt.py has the per-shape table)."""
    key = (x2.shape[0], w.shape[0], w.shape[1], bf is not None)
    hit = _GEMM_NT_CHOICE.get(key)
    if hit is not None:
        return hit
    if torch.cuda.is_current_stream_capturing() or _tuning_active():
        return False
    n = _GEMM_NT_CALLS.get(key, 0) + 1
    _GEMM_NT_CALLS[key] = n
    if n < _PICK_MIN_RECUR:
        return False
    ext = get_ext()

    bb = bf.to(torch.bfloat16) if bf is not None else None
    t_ours = _med_time(lambda: ext.gemm_nt(x2, w, bf, False))
    t_lib = _med_time(lambda: F.linear(x2, w, bb))
    _GEMM_NT_CHOICE[key] = bool(t_ours <= t_lib)
    if os.environ.get("CHINESENER_GEMM_DEBUG") == "1":
        import sys
        print(f"[gemm-dispatch] fwd {key}: ours {t_ours*1e6:.1f}us "
              f"lib {t_lib*1e6:.1f}us -> "
              f"{'ours' if _GEMM_NT_CHOICE[key] else 'lib'}",
              file=sys.stderr, flush=True)
    return _GEMM_NT_CHOICE[key]


class _LinearFn(torch.autograd.Function):
    """nn.Linear math with a custom column-sum bias grad (torch's generic
    reduce is ~4.5x off memory-bound for [tokens, features] dbias)."""

    @staticmethod
    def forward(ctx, x, w, b):
        ctx.save_for_backward(x, w)
        ctx.has_bias = b is not None
        mode = _gemm_nt_mode()
        if mode != "off":
            x2 = x.reshape(-1, x.shape[-1])
            if _gemm_nt_ok(x2, w):
                x2 = x2.contiguous()
                wc = w.contiguous()
                bf = b.float() if b is not None else None
                if mode == "force" or _pick_gemm_nt(x2, wc, bf):
                    y = get_ext().gemm_nt(x2, wc, bf, False)
                    return y.reshape(*x.shape[:-1], w.shape[0])
        return F.linear(x, w, b)

    @staticmethod
    def backward(ctx, dy):
        x, w = ctx.saved_tensors
        K = x.shape[-1]
        N = w.shape[0]
        dyf = dy.reshape(-1, N).contiguous()
        x2 = x.reshape(-1, K)
        M = dyf.shape[0]
        mode = _gemm_nt_mode()
        bf16 = dy.dtype == torch.bfloat16 and x.dtype == torch.bfloat16

        # dgrad: dx[M,K] = dy @ w == gemm_nt(dy[M,N], w^T[K,N]) — the
        # transpose of the small weight (few MB) is part of the timed
        # candidate, so the measured dispatch accounts for it
        def dx_ours():
            return get_ext().gemm_nt(dyf, w.t().contiguous(), None, False)

        def dx_lib():
            return dyf @ w.to(dy.dtype)

        if (mode != "off" and bf16 and M % 128 == 0 and K % 128 == 0
                and N % 64 == 0
                and (mode == "force"
                     or _pick2(("dx", M, N, K), dx_ours, dx_lib))):
            dx = dx_ours().reshape(x.shape)
        else:
            dx = dx_lib().reshape(x.shape)

        # wgrad: dW[N,K] = dy^T @ x == gemm_nt(dy^T[N,M], x^T[K,M])
        def dw_ours():
            return get_ext().gemm_nt(dyf.t().contiguous(),
                                     x2.t().contiguous(), None, False)

        def dw_lib():
            return dyf.T @ x2

        if (mode != "off" and bf16 and N % 128 == 0 and K % 128 == 0
                and M % 64 == 0
                and (mode == "force"
                     or _pick2(("dw", M, N, K), dw_ours, dw_lib))):
            dw = dw_ours().to(w.dtype)
        else:
            dw = dw_lib().to(w.dtype)
        db = get_ext().colsum(dyf) if ctx.has_bias else None
        return dx, dw, db


def linear(x, weight, bias=None):
    """Linear with measured GEMM dispatch (in-tree NT MFMA kernel vs
    hipBLASLt, see _pick_gemm_nt) and custom dbias.

    Used on the uniform-dtype pure-bf16 path; under autocast (mixed
    param/activation dtypes) or off-GPU it falls back to F.linear."""
    if (hip_enabled(x) and x.dtype == weight.dtype
            and not torch.is_autocast_enabled()):
        return _LinearFn.apply(x, weight,
                               bias.to(x.dtype) if bias is not None else None)
    return F.linear(x, weight, bias)


# ---------------------------------------------------- embedding gather
class _Embed3Fn(torch.autograd.Function):
    """Fused word+position+token_type gather-sum: one
    kernel instead of three gathers + two adds. Backward keeps the
    standard scatter/sum torch ops (same work nn.Embedding does)."""

    @staticmethod
    def forward(ctx, word_w, pos_w, tok_w, token_ids, segment_ids):
        ctx.save_for_backward(token_ids, segment_ids)
        ctx.shapes = (word_w.shape, pos_w.shape, tok_w.shape)
        return get_ext().embed3_fwd(word_w, pos_w, tok_w,
                                    token_ids.contiguous(),
                                    segment_ids.contiguous())

    @staticmethod
    def backward(ctx, dy):
        token_ids, segment_ids = ctx.saved_tensors
        (vs, ps, ts) = ctx.shapes
        B, L, H = dy.shape
        # fp32 scatter accumulation: thousands of bf16 adds per heavy row
        # (PAD, segment 0) would otherwise lose mass to rounding
        dyf = dy.reshape(-1, H).float()
        dw = torch.zeros(vs, dtype=torch.float32, device=dy.device)
        dw.index_add_(0, token_ids.reshape(-1), dyf)
        dw[0].zero_()          # padding_idx=0 semantics (nn.Embedding)
        dp = torch.zeros(ps, dtype=torch.float32, device=dy.device)
        dp[:L] = dy.float().sum(0)
        dt = torch.zeros(ts, dtype=torch.float32, device=dy.device)
        dt.index_add_(0, segment_ids.reshape(-1), dyf)
        return (dw.to(dy.dtype), dp.to(dy.dtype), dt.to(dy.dtype),
                None, None)


def embed3(word_w, pos_w, tok_w, token_ids, segment_ids):
    """out[b,l] = word_w[token_ids[b,l]] + pos_w[l] + tok_w[seg[b,l]]."""
    if (os.environ.get("CHINESENER_NO_EMBED3") != "1"
            and hip_enabled(word_w) and word_w.dtype == torch.bfloat16
            and word_w.shape[1] % 8 == 0
            and token_ids.shape[1] <= pos_w.shape[0]):
        return _Embed3Fn.apply(word_w, pos_w, tok_w, token_ids, segment_ids)
    pos = torch.arange(token_ids.shape[1], device=token_ids.device)
    return (F.embedding(token_ids, word_w, padding_idx=0) + pos_w[pos]
            + F.embedding(segment_ids, tok_w))


# ------------------------------------------------------------ bias gelu
class _BiasGeluFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, bias):
        y = get_ext().bias_gelu_fwd(x, bias)
        ctx.save_for_backward(x, bias)
        return y

    @staticmethod
    def backward(ctx, dy):
        x, bias = ctx.saved_tensors
        dx, dbias = get_ext().bias_gelu_bwd(dy.contiguous(), x, bias)
        return dx, dbias


def bias_gelu(x, bias):
    if bias is not None and hip_enabled(x):
        shape = x.shape
        y = _BiasGeluFn.apply(x.reshape(-1, shape[-1]).contiguous(), bias)
        return y.reshape(shape)
    return ref.bias_gelu(x, bias)


# ------------------------------------------------------------ attention
class _AttentionFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, lens, scale):
        out, lse = get_ext().attn_fwd(q, k, v, lens, scale, 1.0, None)
        ctx.save_for_backward(q, k, v, out, lse, lens)
        ctx.scale = scale
        return out

    @staticmethod
    def backward(ctx, dout):
        q, k, v, out, lse, lens = ctx.saved_tensors
        dq, dk, dv = get_ext().attn_bwd(dout.contiguous(), q, k, v, out, lse,
                                        lens, ctx.scale, 1.0, None)
        return dq, dk, dv, None, None


def attention(q, k, v, mask: Optional[torch.Tensor] = None,
              scale: Optional[float] = None, lens: Optional[torch.Tensor] = None):
    """Fused attention. q,k,v [B,H,L,D]; mask [B,L] prefix mask or lens [B].

    HIP path: bf16, head dim padded to 32/64, padded L <= 176 i.e.
    L <= 160 (the kernel pads L to a multiple of 32; covers every
    reference NER config — L=150). Longer sequences (MRC's ragged
    batches can pad past 160) fall back to the torch path."""
    D = q.shape[-1]
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    if (hip_enabled(q) and q.dtype == torch.bfloat16 and D <= 64
            and ((q.shape[2] + 31) // 32) * 32 <= 176):
        Dp = 32 if D <= 32 else 64
        if lens is None:
            lens = (mask.long().sum(1) if mask is not None
                    else torch.full((q.shape[0],), q.shape[2],
                                    dtype=torch.long, device=q.device))
        out = _AttentionFn.apply(_pad_last(q, Dp).contiguous(),
                                 _pad_last(k, Dp).contiguous(),
                                 _pad_last(v, Dp).contiguous(),
                                 lens.to(torch.int32), float(scale))
        return out[..., :D] if Dp != D else out
    return ref.attention(q, k, v, mask, scale)


class _AttentionQkvFn(torch.autograd.Function):
    """Packed-layout path: qkv [B,L,3,H,D] -> out [B,L,H,D]; zero
    transpose/copies around the kernel (strided kernel I/O). Optional
    attention-prob dropout: the mask is a counter-hash regenerated in
    backward from the saved seed snapshot (flash-attn style)."""

    @staticmethod
    def forward(ctx, qkv, lens, scale, keep):
        seed = None
        if keep < 1.0:
            ctr = _rng_counter_for(qkv.device)
            seed = ctr.clone()
            get_ext().bump_counter(ctr)
        out, lse = get_ext().attn_fwd_qkv(qkv, lens, scale, keep, seed)
        ctx.save_for_backward(qkv, out, lse, lens,
                              seed if seed is not None
                              else torch.zeros(0))
        ctx.scale = scale
        ctx.keep = keep
        return out

    @staticmethod
    def backward(ctx, dout):
        qkv, out, lse, lens, seed = ctx.saved_tensors
        (dqkv,) = get_ext().attn_bwd_qkv(dout.contiguous(), qkv, out, lse,
                                         lens, ctx.scale, ctx.keep,
                                         seed if seed.numel() else None)
        return dqkv, None, None, None


def attention_qkv(qkv, mask: Optional[torch.Tensor] = None,
                  lens: Optional[torch.Tensor] = None,
                  scale: Optional[float] = None,
                  p_drop: float = 0.0, training: bool = False):
    """Fused attention on the packed QKV projection output.

    qkv: [B, L, 3, H, D] (the natural reshape of the fused QKV GEMM) ->
    [B, L, H, D]. HIP path avoids every layout copy; fallback unpacks.
    p_drop applies BERT's attention-prob dropout inside the kernel."""
    B, L, _, H, D = qkv.shape
    if scale is None:
        scale = 1.0 / math.sqrt(D)
    keep = 1.0 - p_drop if (training and p_drop > 0) else 1.0
    if os.environ.get("CHINESENER_NO_ATTN_DROP") == "1":
        keep = 1.0
    if (hip_enabled(qkv) and qkv.dtype == torch.bfloat16
            and D in (32, 64) and ((L + 31) // 32) * 32 <= 176):
        if lens is None:
            lens = (mask.long().sum(1) if mask is not None
                    else torch.full((B,), L, dtype=torch.long,
                                    device=qkv.device))
        return _AttentionQkvFn.apply(qkv.contiguous(), lens.to(torch.int32),
                                     float(scale), float(keep))
    q, k, v = (qkv[:, :, i].transpose(1, 2) for i in range(3))  # [B,H,L,D]
    if keep < 1.0:
        out = ref.attention(q, k, v, mask, scale, p_drop=p_drop,
                            training=training)
    else:
        out = attention(q, k, v, mask=mask, scale=scale, lens=lens)
    return out.transpose(1, 2)


class _TenerAttentionFn(torch.autograd.Function):
    """Kernel contract: qu = q+u, qv = q+v precomputed (u/v grads are
    reductions of dqu/dqv handled by autograd outside this Function)."""

    @staticmethod
    def forward(ctx, qu, qv, k, v, rel, lens):
        out, lse = get_ext().tener_attn_fwd(qu, qv, k, v, rel, lens)
        ctx.save_for_backward(qu, qv, k, v, rel, out, lse, lens)
        return out

    @staticmethod
    def backward(ctx, dout):
        qu, qv, k, v, rel, out, lse, lens = ctx.saved_tensors
        dqu, dqv, dk, dv = get_ext().tener_attn_bwd(
            dout.contiguous(), qu, qv, k, v, rel, out, lse, lens)
        return dqu, dqv, dk, dv, None, None


def tener_attention(q, k, v, u, vb, rel, mask: Optional[torch.Tensor] = None):
    """TENER relative attention (unscaled, no key projection upstream).

    HIP path: bf16, head dim padded to 32, L <= 160."""
    D = q.shape[-1]
    if (hip_enabled(q) and q.dtype == torch.bfloat16 and D <= 32
            and q.shape[2] <= 160):
        lens = (mask.long().sum(1) if mask is not None
                else torch.full((q.shape[0],), q.shape[2], dtype=torch.long,
                                device=q.device))
        qu = _pad_last(q + u[None, :, None, :].to(q.dtype), 32)
        qv = _pad_last(q + vb[None, :, None, :].to(q.dtype), 32)
        out = _TenerAttentionFn.apply(
            qu.contiguous(), qv.contiguous(),
            _pad_last(k, 32).to(torch.bfloat16).contiguous(),
            _pad_last(v, 32).to(torch.bfloat16).contiguous(),
            _pad_last(rel, 32).to(torch.bfloat16).contiguous(),
            lens.to(torch.int32))
        return out[..., :D] if D != 32 else out
    return ref.tener_attention(q, k, v, u, vb, rel, mask)


# ------------------------------------------------------------------ crf
class _CrfNllFn(torch.autograd.Function):
    """log-likelihood [B]; kernel computes grads (marginals) in the same
    forward-backward pass."""

    @staticmethod
    def forward(ctx, emissions, tags, lens, transitions):
        ll, demis, dtrans = get_ext().crf_fwd(emissions, tags, lens, transitions)
        ctx.save_for_backward(demis, dtrans)
        return ll

    @staticmethod
    def backward(ctx, dll):
        demis, dtrans = ctx.saved_tensors
        # d(ll_b)/d(emissions) stored as +(gold - expected); chain rule with dll.
        return (demis * dll[:, None, None],
                None, None,
                (dtrans * dll[:, None, None]).sum(0))


def crf_nll(emissions, tags, mask, transitions):
    """Per-sequence log-likelihood [B] (caller negates/averages)."""
    if hip_enabled(emissions):
        lens = mask.long().sum(1).to(torch.int32)
        return _CrfNllFn.apply(emissions.float().contiguous(),
                               tags.to(torch.int32).contiguous(), lens,
                               transitions.float().contiguous())
    return ref.crf_log_likelihood(emissions, tags, mask, transitions)


def crf_viterbi(emissions, mask, transitions):
    if hip_enabled(emissions):
        lens = mask.long().sum(1).to(torch.int32)
        return get_ext().crf_viterbi(emissions.float().contiguous(), lens,
                                     transitions.float().contiguous()).long()
    with torch.no_grad():
        return ref.crf_decode(emissions, mask, transitions)


# ----------------------------------------------------------------- lstm
class _LstmDirFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gates_x, w_hh, lens, reverse, act_relu):
        hs, cs, gates = get_ext().lstm_fwd(gates_x, w_hh, lens, reverse, act_relu)
        ctx.save_for_backward(hs, cs, gates, w_hh, lens)
        ctx.flags = (reverse, act_relu)
        return hs

    @staticmethod
    def backward(ctx, dhs):
        hs, cs, gates, w_hh, lens = ctx.saved_tensors
        reverse, act_relu = ctx.flags
        (dgates_x,) = get_ext().lstm_bwd(dhs.contiguous(), hs, cs, gates,
                                         w_hh, lens, reverse, act_relu)
        # dW_hh = sum_t h_{t-1}^T dgates_t — one library GEMM. hs holds the
        # carried state at valid steps and 0 elsewhere; dgates is 0 at
        # invalid steps, so the shifted product is exact.
        B, L, h = hs.shape
        if reverse:
            h_prev = torch.cat([hs[:, 1:], torch.zeros_like(hs[:, :1])], dim=1)
        else:
            h_prev = torch.cat([torch.zeros_like(hs[:, :1]), hs[:, :-1]], dim=1)
        dw_hh = h_prev.reshape(-1, h).T @ dgates_x.reshape(-1, 4 * h)
        return dgates_x, dw_hh.to(w_hh.dtype), None, None, None


def _lstm_dir(x, w_ih, w_hh, b, lens, reverse, activation):
    # x-projection as one library GEMM; the HIP kernel owns the recurrence.
    gates_x = (x @ w_ih.to(x.dtype) + b.to(x.dtype))
    return _LstmDirFn.apply(gates_x.contiguous(),
                            w_hh.to(gates_x.dtype).contiguous(),
                            lens.to(torch.int32), bool(reverse),
                            activation == "relu")


class _BiLstmFn(torch.autograd.Function):
    """Both directions in ONE kernel launch (blockIdx.y = direction):
    gates_x [B,L,8h] (fw|bw halves), hidden out [B,L,2h] — the BiLSTM
    concat is produced directly by the strided kernel."""

    @staticmethod
    def forward(ctx, gates_x, w_hh_f, w_hh_b, lens, act_relu, cell_clip):
        w_hh_t2 = torch.stack([w_hh_f.t(), w_hh_b.t()]) \
            .to(torch.bfloat16).contiguous()
        hs, cs, gates = get_ext().bilstm_fwd(gates_x, w_hh_t2, lens, act_relu,
                                             cell_clip)
        ctx.save_for_backward(hs, cs, gates, w_hh_f, w_hh_b, lens)
        ctx.relu = act_relu
        ctx.cell_clip = cell_clip
        return hs

    @staticmethod
    def backward(ctx, dhs):
        hs, cs, gates, w_hh_f, w_hh_b, lens = ctx.saved_tensors
        w_hh2 = torch.stack([w_hh_f, w_hh_b]).to(torch.bfloat16).contiguous()
        dgates_x = get_ext().bilstm_bwd(dhs.contiguous(), cs, gates, w_hh2,
                                        lens, ctx.relu, ctx.cell_clip)
        h = hs.shape[-1] // 2
        # dW_hh = sum_t h_{t-1}^T dgates_t per direction — library GEMMs.
        # hs holds the carried state at valid steps and 0 elsewhere; dgates
        # is 0 at invalid steps, so the shifted product is exact.
        fw, bw = hs[..., :h], hs[..., h:]
        h_prev_f = torch.cat([torch.zeros_like(fw[:, :1]), fw[:, :-1]], 1)
        h_prev_b = torch.cat([bw[:, 1:], torch.zeros_like(bw[:, :1])], 1)
        dg_f = dgates_x[..., :4 * h]
        dg_b = dgates_x[..., 4 * h:]
        dw_f = h_prev_f.reshape(-1, h).T @ dg_f.reshape(-1, 4 * h)
        dw_b = h_prev_b.reshape(-1, h).T @ dg_b.reshape(-1, 4 * h)
        return (dgates_x, dw_f.to(w_hh_f.dtype), dw_b.to(w_hh_b.dtype),
                None, None, None)


def _pad_gates(w, h, hp):
    """Zero-pad each i/f/g/o gate segment of a [..., 4h] tensor to hp."""
    if h == hp:
        return w
    shape = list(w.shape[:-1])
    wv = w.reshape(*shape, 4, h)
    return F.pad(wv, (0, hp - h)).reshape(*shape, 4 * hp)


def bilstm(x, w_ih_f, w_hh_f, b_f, w_ih_b, w_hh_b, b_b, lens,
           activation: str = "tanh", state_dropout=None,
           cell_clip: float = 0.0):
    """BiLSTM over padded [B,L,E] -> [B,L,2h].

    HIP path: hidden <= 256 (W_hh LDS-resident up to 128, L2-streamed
    above); non-multiple-of-32 hidden sizes are zero-padded — padded
    units stay exactly 0 through the recurrence (sigmoid(0)*act(0)
    structure), so sliced outputs and grads are exact. Larger sizes run
    the torch recurrence (slow path, logged once).
    cell_clip > 0 bounds the cell state (TF LSTMCell cell_clip) — the
    relu recurrence needs it to stay off the exponential-growth regime."""
    h = w_hh_f.shape[0]
    hp = ((h + 31) // 32) * 32
    if hip_enabled(x) and hp <= 256:
        if hp != h:
            w_ih_f = _pad_gates(w_ih_f, h, hp)
            w_ih_b = _pad_gates(w_ih_b, h, hp)
            b_f = _pad_gates(b_f, h, hp)
            b_b = _pad_gates(b_b, h, hp)
            # W_hh: pad input rows, then each gate segment
            w_hh_f = _pad_gates(F.pad(w_hh_f, (0, 0, 0, hp - h)), h, hp)
            w_hh_b = _pad_gates(F.pad(w_hh_b, (0, 0, 0, hp - h)), h, hp)
        # single fused x-projection GEMM for both directions
        w_ih2 = torch.cat([w_ih_f, w_ih_b], dim=1).to(x.dtype)
        b2 = torch.cat([b_f, b_b]).to(x.dtype)
        gates_x = (x @ w_ih2 + b2).contiguous()
        hs = _BiLstmFn.apply(gates_x, w_hh_f, w_hh_b,
                             lens.to(torch.int32), activation == "relu",
                             float(cell_clip))
        if hp != h:
            hs = torch.cat([hs[..., :h], hs[..., hp:hp + h]], dim=-1)
        return hs
    if x.is_cuda:
        global _LSTM_FALLBACK_WARNED
        if not _LSTM_FALLBACK_WARNED:
            import logging
            logging.getLogger("chinesener_amd").warning(
                "BiLSTM hidden=%d not kernel-eligible (kernel covers "
                "hidden <= 256 via zero-padding); using torch recurrence", h)
            _LSTM_FALLBACK_WARNED = True
    return ref.bilstm_forward(x, w_ih_f, w_hh_f, b_f, w_ih_b, w_hh_b, b_b,
                              lens, activation, state_dropout, cell_clip)


_LSTM_FALLBACK_WARNED = False


# ----------------------------------------------------------- softlexicon
class _SoftLexiconFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, table, ids, weights):
        out = get_ext().softlexicon_fwd(table, ids, weights)
        ctx.save_for_backward(table, ids, weights)
        return out

    @staticmethod
    def backward(ctx, dout):
        table, ids, weights = ctx.saved_tensors
        dtable, dweights = get_ext().softlexicon_bwd(dout.contiguous(), table,
                                                     ids, weights)
        return dtable, None, dweights


def softlexicon_fuse(table, ids, weights):
    """Fused gather-scale-reduce: [V,E] x [B,L,40] -> [B,L,4E].

    Kernel computes in fp32; bf16 tables/weights (pure-bf16 mode) go
    through differentiable casts so grads flow back in the param dtype."""
    if hip_enabled(table):
        out_dtype = table.dtype
        out = _SoftLexiconFn.apply(table.float().contiguous(),
                                   ids.to(torch.int32).contiguous(),
                                   weights.float().contiguous())
        return out.to(out_dtype)
    return ref.softlexicon_fuse(table, ids, weights)


# ---------------------------------------------------------------- losses
class _MaskedCeFn(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, labels, mask):
        loss, probs, nvalid = get_ext().masked_ce_fwd(logits, labels, mask)
        ctx.save_for_backward(probs, labels, mask, nvalid)
        return loss

    @staticmethod
    def backward(ctx, dloss):
        probs, labels, mask, nvalid = ctx.saved_tensors
        dlogits = get_ext().masked_ce_bwd(dloss, probs, labels, mask, nvalid)
        return dlogits, None, None


def masked_cross_entropy(logits, labels, mask):
    if hip_enabled(logits):
        T = logits.shape[-1]
        return _MaskedCeFn.apply(logits.reshape(-1, T).float().contiguous(),
                                 labels.reshape(-1).to(torch.int32).contiguous(),
                                 mask.reshape(-1).to(torch.int32).contiguous())
    return ref.masked_cross_entropy(logits, labels, mask)


def dice_loss(logits, labels, mask, idx_skip: Tuple[int, ...],
              alpha: float = 0.1, gamma: float = 1.0):
    # reduction over tags, not perf-critical (eval-style loss) — reference
    # implementation runs on both devices.
    return ref.dice_loss(logits, labels, mask, idx_skip, alpha, gamma)
