"""Pure-torch fp32 reference implementations of every custom op.

These are (a) the CPU execution path, (b) the ground truth the HIP
kernels are tested against (numerics tests compare the gfx950 kernels to
these in fp32 — SURVEY.md §4 implication (a)). All are differentiable so
autograd provides backward on the reference path.

Shapes follow SURVEY.md §2.6: B=batch, L=seq len, H=hidden, T=labels.
"""
from __future__ import annotations

import math
from typing import Optional, Tuple

import torch
import torch.nn.functional as F

NEG_INF = -1e30


# ------------------------------------------------------------- layernorm
def layernorm(x: torch.Tensor, weight: torch.Tensor, bias: torch.Tensor,
              eps: float = 1e-12) -> torch.Tensor:
    return F.layer_norm(x, (x.shape[-1],), weight, bias, eps)


def add_layernorm(x: torch.Tensor, residual: torch.Tensor, weight: torch.Tensor,
                  bias: torch.Tensor, eps: float = 1e-12) -> torch.Tensor:
    """Fused residual-add + LayerNorm (post-LN, as BERT and the reference's
    add_and_norm_layer, tools/transformer/modules.py:58-65)."""
    return F.layer_norm(x + residual, (x.shape[-1],), weight, bias, eps)


# ------------------------------------------------------------- bias gelu
def bias_gelu(x: torch.Tensor, bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """BERT's gelu after a bias add — tanh approximation, the form
    google-research BERT / the reference's bert_base actually computes."""
    if bias is not None:
        x = x + bias
    return F.gelu(x, approximate="tanh")


# ------------------------------------------------------------- attention
def attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
              mask: Optional[torch.Tensor] = None,
              scale: Optional[float] = None,
              p_drop: float = 0.0, training: bool = False) -> torch.Tensor:
    """Scaled dot-product attention. q,k,v: [B,H,L,D]; mask: [B,L] (1=keep).

    Matches BERT attention and the reference's
    scaled_dot_product_attention + normalize_attention
    (tools/transformer/modules.py:101-130): additive large-negative mask
    on padded KEY positions, softmax over the key axis.
    """
    if scale is None:
        scale = 1.0 / math.sqrt(q.shape[-1])
    scores = torch.matmul(q, k.transpose(-1, -2)) * scale
    if mask is not None:
        key_mask = mask[:, None, None, :].to(scores.dtype)
        scores = scores + (1.0 - key_mask) * NEG_INF
    probs = torch.softmax(scores.float(), dim=-1).to(q.dtype)
    if training and p_drop > 0:
        probs = F.dropout(probs, p_drop, training)
    return torch.matmul(probs, v)


def tener_attention(q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
                    u: torch.Tensor, vb: torch.Tensor, rel: torch.Tensor,
                    mask: Optional[torch.Tensor] = None) -> torch.Tensor:
    """TENER relative attention (reference tools/transformer/tener.py:12-74).

    q,k,v: [B,H,L,D]; u,vb: per-head biases [H,D]; rel: sinusoidal table
    [2L, D] covering offsets [-L, L). Unscaled scores; the shift trick is
    replaced by direct indexing R[L-1+(j-i)].
    attn[b,h,i,j] = (q+u)·k + (q+vb)·R[j-i]   (no key projection upstream)
    """
    B, H, L, D = q.shape
    ac = torch.matmul(q + u[None, :, None, :], k.transpose(-1, -2))   # [B,H,L,L]
    # BD term: (q+vb) @ rel^T gives [B,H,L,2L]; gather offset j-i+L-1
    bd_full = torch.matmul(q + vb[None, :, None, :], rel.transpose(0, 1))  # [B,H,L,2L]
    idx = (torch.arange(L, device=q.device)[None, :]
           - torch.arange(L, device=q.device)[:, None]) + (L - 1)     # [L,L]
    bd = bd_full.gather(-1, idx.expand(B, H, L, L))
    scores = ac + bd
    if mask is not None:
        key_mask = mask[:, None, None, :].to(scores.dtype)
        scores = scores + (1.0 - key_mask) * NEG_INF
    probs = torch.softmax(scores.float(), dim=-1).to(q.dtype)
    return torch.matmul(probs, v)


def sinusoidal_table(length: int, dim: int, device=None,
                     dtype=torch.float32) -> torch.Tensor:
    """Sinusoidal position table [length, dim] (reference modules.py:177-197)."""
    pos = torch.arange(length, device=device, dtype=torch.float32)[:, None]
    i = torch.arange(dim, device=device, dtype=torch.float32)[None, :]
    angle = pos / torch.pow(10000.0, (2 * (i // 2)) / dim)
    table = torch.where(i.long() % 2 == 0, torch.sin(angle), torch.cos(angle))
    return table.to(dtype)


def relative_table(seq_len: int, dim: int, device=None,
                   dtype=torch.float32) -> torch.Tensor:
    """Relative-position table over offsets [-L, L) as TENER uses
    (tener.py sinusoid over [-L, L))."""
    pos = torch.arange(-seq_len, seq_len, device=device, dtype=torch.float32)[:, None]
    i = torch.arange(dim, device=device, dtype=torch.float32)[None, :]
    angle = pos / torch.pow(10000.0, (2 * (i // 2)) / dim)
    table = torch.where(i.long() % 2 == 0, torch.sin(angle), torch.cos(angle))
    return table.to(dtype)


# ------------------------------------------------------------------- crf
def crf_log_likelihood(emissions: torch.Tensor, tags: torch.Tensor,
                       mask: torch.Tensor, transitions: torch.Tensor
                       ) -> torch.Tensor:
    """Per-sequence log-likelihood [B]. emissions [B,L,T] (fp32), tags [B,L],
    mask [B,L] (1 on real incl CLS/SEP as reference counts them,
    tools/layer.py:121), transitions [T,T] trans[i,j] = score(i -> j)."""
    B, L, T = emissions.shape
    emissions = emissions.float()
    mask = mask.to(torch.bool)
    score = emissions[:, 0].gather(1, tags[:, :1]).squeeze(1)
    alpha = emissions[:, 0]                                   # [B,T]
    for t in range(1, L):
        m = mask[:, t]
        # gold path score
        step = (transitions[tags[:, t - 1], tags[:, t]]
                + emissions[:, t].gather(1, tags[:, t:t + 1]).squeeze(1))
        score = score + step * m.to(score.dtype)
        # partition forward
        next_alpha = torch.logsumexp(
            alpha[:, :, None] + transitions[None, :, :], dim=1) + emissions[:, t]
        alpha = torch.where(m[:, None], next_alpha, alpha)
    log_z = torch.logsumexp(alpha, dim=1)
    return score - log_z


def crf_decode(emissions: torch.Tensor, mask: torch.Tensor,
               transitions: torch.Tensor) -> torch.Tensor:
    """Viterbi decode -> [B,L] best tag ids (padded positions keep tag of
    last real step, then zeroed)."""
    B, L, T = emissions.shape
    emissions = emissions.float()
    mask = mask.to(torch.bool)
    history = []
    alpha = emissions[:, 0]
    for t in range(1, L):
        scores = alpha[:, :, None] + transitions[None, :, :]    # [B,T,T]
        best, idx = scores.max(dim=1)
        next_alpha = best + emissions[:, t]
        keep = mask[:, t][:, None]
        alpha = torch.where(keep, next_alpha, alpha)
        history.append((idx, mask[:, t]))
    lens = mask.long().sum(1)                                   # [B]
    best_last = alpha.argmax(dim=1)                             # [B]
    out = torch.zeros(B, L, dtype=torch.long, device=emissions.device)
    cur = best_last.clone()
    out[:, L - 1] = cur
    batch_idx = torch.arange(B, device=emissions.device)
    for t in range(L - 2, -1, -1):
        idx, _ = history[t]                 # transition t -> t+1 backpointers
        within = (t + 1) < lens             # step back only inside the sequence
        prev = idx[batch_idx, cur]
        cur = torch.where(within, prev, cur)
        out[:, t] = cur
    # positions beyond len -> 0 ([PAD])
    pos = torch.arange(L, device=emissions.device)[None, :]
    return out * (pos < lens[:, None]).long()


# ------------------------------------------------------------------ lstm
def lstm_forward(x: torch.Tensor, w_ih: torch.Tensor, w_hh: torch.Tensor,
                 b: torch.Tensor, lens: torch.Tensor, reverse: bool = False,
                 activation: str = "tanh",
                 state_dropout: Optional[torch.Tensor] = None,
                 cell_clip: float = 0.0) -> torch.Tensor:
    """Single-direction LSTM over padded [B,L,E] -> [B,L,h].

    Gate order i,f,g,o (torch convention). `activation` applies to the
    cell candidate and output squash (reference supports tanh/relu,
    model/bilstm_crf.py:52 vs bert_bilstm_crf.py:42). Outputs at t>=len
    are zero (like dynamic_rnn with seq_len). state_dropout: [B,h]
    keep-mask/scale applied to h between steps (DropoutWrapper state
    keep-prob semantics, tools/layer.py:20-24).
    """
    B, L, E = x.shape
    h4 = w_hh.shape[1]
    h = h4 // 4
    act = torch.tanh if activation == "tanh" else torch.relu
    gates_x = x @ w_ih + b                                      # [B,L,4h]
    hs = x.new_zeros(B, L, h)
    ht = x.new_zeros(B, h)
    ct = x.new_zeros(B, h)
    steps = range(L - 1, -1, -1) if reverse else range(L)
    for t in steps:
        g = gates_x[:, t] + ht @ w_hh
        i, f, gc, o = g.split(h, dim=-1)
        i, f, o = torch.sigmoid(i), torch.sigmoid(f), torch.sigmoid(o)
        c_new = f * ct + i * act(gc)
        if cell_clip > 0:
            # TF LSTMCell cell_clip: bounds the relu recurrence
            c_new = c_new.clamp(-cell_clip, cell_clip)
        h_new = o * act(c_new)
        valid = (t < lens).to(x.dtype)[:, None]                 # [B,1]
        ct = valid * c_new + (1 - valid) * ct
        h_out = h_new * valid
        if state_dropout is not None:
            h_new = h_new * state_dropout
        ht = valid * h_new + (1 - valid) * ht
        hs[:, t] = h_out
    return hs


def bilstm_forward(x, w_ih_f, w_hh_f, b_f, w_ih_b, w_hh_b, b_b, lens,
                   activation="tanh", state_dropout=None,
                   cell_clip: float = 0.0) -> torch.Tensor:
    fw = lstm_forward(x, w_ih_f, w_hh_f, b_f, lens, False, activation,
                      state_dropout, cell_clip)
    bw = lstm_forward(x, w_ih_b, w_hh_b, b_b, lens, True, activation,
                      state_dropout, cell_clip)
    return torch.cat([fw, bw], dim=-1)


# ----------------------------------------------------------- softlexicon
def softlexicon_fuse(table: torch.Tensor, ids: torch.Tensor,
                     weights: torch.Tensor) -> torch.Tensor:
    """SoftLexicon gather-scale-reduce: table [V,E],
    ids/weights [B,L,40] (4 roles x 10 slots) -> [B,L,4*E]
    (weighted sum within each role group, concat groups)."""
    B, L, K = ids.shape
    R, S = 4, K // 4
    E = table.shape[1]
    emb = F.embedding(ids.long(), table)                       # [B,L,40,E]
    w = weights.to(emb.dtype).unsqueeze(-1)                    # [B,L,40,1]
    fused = (emb * w).view(B, L, R, S, E).sum(dim=3)           # [B,L,4,E]
    return fused.reshape(B, L, R * E)


# ---------------------------------------------------------------- losses
def masked_cross_entropy(logits: torch.Tensor, labels: torch.Tensor,
                         mask: torch.Tensor) -> torch.Tensor:
    """Mean CE over real tokens (reference tools/loss.py:5-16)."""
    T = logits.shape[-1]
    loss = F.cross_entropy(logits.reshape(-1, T).float(), labels.reshape(-1),
                           reduction="none")
    m = mask.reshape(-1).to(loss.dtype)
    return (loss * m).sum() / m.sum().clamp(min=1.0)


def dice_loss(logits: torch.Tensor, labels: torch.Tensor, mask: torch.Tensor,
              idx_skip: Tuple[int, ...], alpha: float = 0.1,
              gamma: float = 1.0) -> torch.Tensor:
    """Dice/DSC loss (reference tools/loss.py:19-46): per-tag soft dice over
    real tokens, summed over tags excluding O/[PAD]/[CLS]/[SEP] (idx_skip),
    with focal-style (1-p)^alpha down-weight and gamma smoothing."""
    T = logits.shape[-1]
    probs = torch.softmax(logits.float(), dim=-1)
    m = mask.to(probs.dtype).reshape(-1, 1)
    probs = probs.reshape(-1, T)
    y = F.one_hot(labels.reshape(-1), T).to(probs.dtype)
    # vectorized over tags (the per-tag python loop launched ~50 tiny
    # kernels per call and cost 2.6 ms fwd+bwd on GPU; this form is a
    # handful of [R,T] passes, same math)
    p = probs * (1 - probs) ** alpha * m
    g = y * m
    num = 2 * (p * g).sum(0) + gamma
    den = p.sum(0) + g.sum(0) + gamma
    dsc = 1 - num / den                     # [T]
    keep = torch.ones(T, dtype=torch.bool, device=logits.device)
    for t in idx_skip:
        keep[t] = False
    return dsc[keep].sum()


def crf_partition_scan(emissions: torch.Tensor, mask: torch.Tensor,
                       transitions: torch.Tensor) -> torch.Tensor:
    """log Z via an ASSOCIATIVE parallel scan (Blelchel-style prefix
    product in the log semiring) — the round-3 blueprint for a
    depth-log(L) CRF forward kernel (docs/ROADMAP.md; the sequential
    scan kernel is latency-bound at ~6.8k cycles/step).

    Formulation: define per-step matrices
        M_t[i, j] = transitions[i, j] + emissions[t, j]   (t >= 1)
    masked steps use the identity of the (max,+/logsumexp) semiring
    (diag 0, off-diag -inf). Then
        alpha_L = alpha_0 (*) M_1 (*) ... (*) M_{L-1}
    with (A (*) B)[i, k] = logsumexp_j(A[i, j] + B[j, k]), which is
    associative — the matrix chain reduces pairwise in log2(L) rounds.
    Returns log Z [B] (bitwise-equivalent math to the sequential
    forward up to fp reduction order; tested against
    crf_log_likelihood)."""
    B, L, T = emissions.shape
    em = emissions.float()
    neg = torch.finfo(torch.float32).min / 4
    eye = torch.full((T, T), neg, device=em.device)
    eye.fill_diagonal_(0.0)
    if L == 1:
        return torch.logsumexp(em[:, 0], dim=1)
    # M[t] for t = 1..L-1, shape [B, L-1, T, T]
    m = mask[:, 1:].to(torch.bool)
    mats = transitions[None, None] + em[:, 1:, None, :]
    mats = torch.where(m[:, :, None, None], mats, eye[None, None])

    def combine(a, b):
        # [*, T, T] (*) [*, T, T] in the log semiring
        return torch.logsumexp(a[..., :, :, None] + b[..., None, :, :],
                               dim=-2)

    chain = mats
    while chain.shape[1] > 1:
        n = chain.shape[1]
        even = chain[:, 0:n - 1:2]
        odd = chain[:, 1:n:2]
        merged = combine(even, odd)
        if n % 2 == 1:                      # carry the unpaired tail
            merged = torch.cat([merged, chain[:, -1:]], dim=1)
        chain = merged
    total = chain[:, 0]                     # [B, T, T]
    alpha0 = em[:, 0]
    return torch.logsumexp(
        torch.logsumexp(alpha0[:, :, None] + total, dim=1), dim=1)
