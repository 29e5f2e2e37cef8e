"""Shared layer library (parity with reference tools/layer.py).

BiLSTM (:10-41), multi-kernel CNN (:44-60), CRF layer (:112-149) live
here as nn.Modules composed from the ops layer; the BERT encoder is in
bert.py, transformer encoders in transformer.py.
"""
from __future__ import annotations

from typing import List, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from .. import ops


class BiLSTM(nn.Module):
    """Bidirectional LSTM with the reference's DropoutWrapper semantics
    (output keep-prob; activation tanh/relu selectable —
    tools/layer.py:10-41, model/bilstm_crf.py:48-52)."""

    def __init__(self, input_size: int, hidden_size: int, activation: str = "tanh",
                 keep_prob: float = 0.8, cell_clip: float = None):
        super().__init__()
        self.hidden_size = hidden_size
        self.activation = activation
        # relu cells are unbounded; clip the cell state like TF's
        # LSTMCell(cell_clip=...) to keep the recurrence stable
        self.cell_clip = (10.0 if activation == "relu" else 0.0) \
            if cell_clip is None else cell_clip
        self.dropout = nn.Dropout(1.0 - keep_prob)

        # TF LSTMCell defaults the reference builds on: glorot-uniform
        # kernels, zero bias with forget-gate bias 1.0 — keeps the (relu)
        # recurrence away from the exponential-growth regime over L=150
        # steps (a uniform 1/sqrt(h) init sits near the stability cliff).
        def w(*shape):
            t = torch.empty(*shape)
            nn.init.xavier_uniform_(t)
            return nn.Parameter(t)

        def b():
            t = torch.zeros(4 * hidden_size)
            t[hidden_size:2 * hidden_size] = 1.0  # forget gate (i,f,g,o)
            return nn.Parameter(t)
        self.w_ih_f, self.w_hh_f, self.b_f = w(input_size, 4 * hidden_size), \
            w(hidden_size, 4 * hidden_size), b()
        self.w_ih_b, self.w_hh_b, self.b_b = w(input_size, 4 * hidden_size), \
            w(hidden_size, 4 * hidden_size), b()

    def forward(self, x: torch.Tensor, lens: torch.Tensor) -> torch.Tensor:
        out = ops.bilstm(x, self.w_ih_f, self.w_hh_f, self.b_f,
                         self.w_ih_b, self.w_hh_b, self.b_b, lens,
                         self.activation, cell_clip=self.cell_clip)
        return self.dropout(out)


class MultiKernelCNN(nn.Module):
    """Per-kernel conv1d SAME + dropout, concat filters
    (reference cnn_layer, tools/layer.py:44-60).

    torch ``padding='same'`` is bit-identical to TF SAME at stride 1 for
    every kernel size incl. even ones — both put the extra zero on the
    RIGHT (total k-1, left (k-1)//2) — verified by
    tests/test_ops_reference.py::test_conv1d_same_matches_tf_same."""

    def __init__(self, input_size: int, filters: int = 128,
                 kernel_sizes: List[int] = (2, 3, 4), keep_prob: float = 0.8):
        super().__init__()
        self.convs = nn.ModuleList([
            nn.Conv1d(input_size, filters, k, padding="same")
            for k in kernel_sizes])
        self.dropout = nn.Dropout(1.0 - keep_prob)
        self.output_size = filters * len(kernel_sizes)

    def forward(self, x: torch.Tensor) -> torch.Tensor:   # [B,L,E]
        if x.is_cuda and x.dtype == torch.bfloat16:
            # SAME conv1d as pad + unfold + linear: the unfolded GEMM is
            # K-major NT, so it rides the in-tree MFMA GEMM / hipBLASLt
            # measured dispatch like every other linear
            outs = []
            for c in self.convs:
                k = c.kernel_size[0]
                left = (k - 1) // 2
                xp = F.pad(x, (0, 0, left, (k - 1) - left))     # [B,L+k-1,E]
                u = xp.unfold(1, k, 1)                          # [B,L,E,k]
                u = u.transpose(2, 3).reshape(x.shape[0], x.shape[1], -1)
                w = c.weight.permute(0, 2, 1).reshape(c.weight.shape[0], -1)
                y = ops.linear(u.contiguous(), w.contiguous(), c.bias)
                outs.append(self.dropout(torch.relu(y)))
            return torch.cat(outs, dim=-1)
        xt = x.transpose(1, 2)
        outs = [self.dropout(torch.relu(c(xt))) for c in self.convs]
        return torch.cat(outs, dim=1).transpose(1, 2)


class CRF(nn.Module):
    """Linear-chain CRF head (reference crf_layer/crf_decode,
    tools/layer.py:112-149): loss over the full padded length including
    CLS/SEP (seq_len counts them, comment :121)."""

    def __init__(self, num_tags: int):
        super().__init__()
        self.num_tags = num_tags
        # near-zero init: with the reference's crf diff-LR x500, a
        # strong random transition prior locks borderline seeds into the
        # all-O basin before the encoder learns anything
        self.transitions = nn.Parameter(
            torch.empty(num_tags, num_tags).uniform_(-0.01, 0.01))

    def neg_log_likelihood(self, emissions, tags, mask) -> torch.Tensor:
        """Sum of -log p(tags | emissions) over the batch."""
        ll = ops.crf_nll(emissions.float(), tags, mask, self.transitions)
        return -ll.sum()

    def decode(self, emissions, mask) -> torch.Tensor:
        return ops.crf_viterbi(emissions.float(), mask, self.transitions)


class TokenEmbedding(nn.Module):
    """Char/word embedding, optionally initialized from a pretrained matrix
    carried in data params (reference base_preprocess.py:223-226) and
    optionally frozen (bichar pretrained constant, bilstm_crf_bichar.py)."""

    def __init__(self, vocab_size: int, dim: int,
                 pretrained: Optional[torch.Tensor] = None, freeze: bool = False,
                 padding_idx: int = 0):
        super().__init__()
        self.emb = nn.Embedding(vocab_size, dim, padding_idx=padding_idx)
        if pretrained is not None:
            with torch.no_grad():
                self.emb.weight.copy_(torch.as_tensor(pretrained))
        if freeze:
            self.emb.weight.requires_grad_(False)

    def forward(self, ids):
        return self.emb(ids)
