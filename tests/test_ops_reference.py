"""Reference-op correctness: CRF vs exhaustive enumeration, Viterbi vs
brute force, LSTM vs step-by-step math, attention vs naive softmax,
TENER vs its definition, softlexicon vs a python loop (SURVEY.md §4
implication (a): these are the ground truths the HIP kernels are later
compared against)."""
import itertools
import math

import pytest
import torch

from chinesener_amd.ops import reference as ref


def _brute_crf(emissions, mask, transitions):
    """Enumerate all paths: returns (logZ, best_path, best_score) per batch."""
    B, L, T = emissions.shape
    lens = mask.long().sum(1)
    logZs, bests, bscores = [], [], []
    for b in range(B):
        n = int(lens[b])
        scores = {}
        for path in itertools.product(range(T), repeat=n):
            s = emissions[b, 0, path[0]].item()
            for t in range(1, n):
                s += transitions[path[t - 1], path[t]].item() + \
                    emissions[b, t, path[t]].item()
            scores[path] = s
        mx = max(scores.values())
        logZ = mx + math.log(sum(math.exp(v - mx) for v in scores.values()))
        best = max(scores, key=scores.get)
        logZs.append(logZ)
        bests.append(list(best) + [0] * (L - n))
        bscores.append(scores[best])
    return torch.tensor(logZs), torch.tensor(bests), torch.tensor(bscores)


@pytest.mark.parametrize("T,L", [(3, 4), (4, 5)])
def test_crf_loglik_vs_enumeration(T, L):
    torch.manual_seed(0)
    B = 3
    em = torch.randn(B, L, T)
    trans = torch.randn(T, T)
    lens = torch.tensor([L, L - 1, 2])
    mask = (torch.arange(L)[None, :] < lens[:, None]).long()
    tags = torch.randint(0, T, (B, L)) * mask
    ll = ref.crf_log_likelihood(em, tags, mask, trans)
    logZ, _, _ = _brute_crf(em, mask, trans)
    # gold path score by hand
    for b in range(B):
        n = int(lens[b])
        s = em[b, 0, tags[b, 0]].item()
        for t in range(1, n):
            s += trans[tags[b, t - 1], tags[b, t]].item() + em[b, t, tags[b, t]].item()
        assert abs(ll[b].item() - (s - logZ[b].item())) < 1e-4


def test_crf_viterbi_vs_enumeration():
    torch.manual_seed(1)
    B, L, T = 4, 5, 3
    em = torch.randn(B, L, T)
    trans = torch.randn(T, T)
    lens = torch.tensor([5, 4, 3, 1])
    mask = (torch.arange(L)[None, :] < lens[:, None]).long()
    pred = ref.crf_decode(em, mask, trans)
    _, best, _ = _brute_crf(em, mask, trans)
    assert torch.equal(pred, best)


def test_crf_grad_is_marginal_gap():
    """d ll / d emissions = onehot(gold) - marginals; check via autograd
    against finite difference."""
    torch.manual_seed(2)
    B, L, T = 2, 4, 3
    em = torch.randn(B, L, T, requires_grad=True)
    trans = torch.randn(T, T)
    mask = torch.ones(B, L, dtype=torch.long)
    tags = torch.randint(0, T, (B, L))
    ll = ref.crf_log_likelihood(em, tags, mask, trans).sum()
    ll.backward()
    eps = 1e-4
    with torch.no_grad():
        e2 = em.detach().clone()
        e2[0, 1, 2] += eps
        l2 = ref.crf_log_likelihood(e2, tags, mask, trans).sum()
        fd = (l2 - ll.detach()) / eps
    assert abs(fd.item() - em.grad[0, 1, 2].item()) < 1e-2


def test_attention_matches_naive():
    torch.manual_seed(3)
    B, H, L, D = 2, 2, 8, 4
    q, k, v = (torch.randn(B, H, L, D) for _ in range(3))
    lens = torch.tensor([8, 5])
    mask = (torch.arange(L)[None, :] < lens[:, None]).long()
    out = ref.attention(q, k, v, mask)
    # naive per-row
    scale = 1 / math.sqrt(D)
    for b in range(B):
        for h in range(H):
            s = (q[b, h] @ k[b, h].T) * scale
            s[:, lens[b]:] = -1e30
            p = torch.softmax(s, -1)
            torch.testing.assert_close(out[b, h], p @ v[b, h], atol=1e-5, rtol=1e-4)


def test_tener_attention_definition():
    torch.manual_seed(4)
    B, H, L, D = 1, 2, 6, 4
    q, k, v = (torch.randn(B, H, L, D) for _ in range(3))
    u, vb = torch.randn(H, D), torch.randn(H, D)
    rel = ref.relative_table(L, D)
    mask = torch.ones(B, L, dtype=torch.long)
    out = ref.tener_attention(q, k, v, u, vb, rel, mask)
    for h in range(H):
        s = torch.zeros(L, L)
        for i in range(L):
            for j in range(L):
                s[i, j] = ((q[0, h, i] + u[h]) @ k[0, h, j]
                           + (q[0, h, i] + vb[h]) @ rel[j - i + L - 1])
        p = torch.softmax(s, -1)
        torch.testing.assert_close(out[0, h], p @ v[0, h], atol=1e-5, rtol=1e-4)


def test_tener_shift_fixture():
    """The reference validates its shift trick on a literal matrix
    (tools/transformer/tener.py:122-128): shift maps column j of the
    [L, 2L] rel-score matrix to offset j-i. Direct indexing must agree."""
    L = 3
    bd_full = torch.arange(-L, L).float().repeat(L, 1)      # rows identical
    idx = (torch.arange(L)[None, :] - torch.arange(L)[:, None]) + (L - 1)
    bd = bd_full.gather(-1, idx)
    # row i, col j should hold offset (j - i) value = (j-i+L-1) - L = j-i-1
    for i in range(L):
        for j in range(L):
            assert bd[i, j].item() == (j - i + L - 1) - L


def test_lstm_matches_manual():
    torch.manual_seed(5)
    B, L, E, h = 2, 5, 3, 4
    x = torch.randn(B, L, E)
    w_ih, w_hh = torch.randn(E, 4 * h) * 0.3, torch.randn(h, 4 * h) * 0.3
    b = torch.randn(4 * h) * 0.1
    lens = torch.tensor([5, 3])
    out = ref.lstm_forward(x, w_ih, w_hh, b, lens)
    ht = torch.zeros(B, h)
    ct = torch.zeros(B, h)
    for t in range(L):
        g = x[:, t] @ w_ih + b + ht @ w_hh
        i, f, gc, o = g.split(h, -1)
        c_new = torch.sigmoid(f) * ct + torch.sigmoid(i) * torch.tanh(gc)
        h_new = torch.sigmoid(o) * torch.tanh(c_new)
        for bb in range(B):
            if t < lens[bb]:
                ct[bb], ht[bb] = c_new[bb], h_new[bb]
                torch.testing.assert_close(out[bb, t], h_new[bb], atol=1e-5,
                                           rtol=1e-4)
            else:
                assert out[bb, t].abs().sum() == 0


def test_lstm_reverse_consistency():
    torch.manual_seed(6)
    B, L, E, h = 1, 4, 3, 2
    x = torch.randn(B, L, E)
    w_ih, w_hh, b = torch.randn(E, 4 * h), torch.randn(h, 4 * h), torch.zeros(4 * h)
    lens = torch.tensor([L])
    fwd_on_flip = ref.lstm_forward(x.flip(1), w_ih, w_hh, b, lens)
    bwd = ref.lstm_forward(x, w_ih, w_hh, b, lens, reverse=True)
    torch.testing.assert_close(bwd, fwd_on_flip.flip(1), atol=1e-5, rtol=1e-4)


def test_softlexicon_fuse_matches_loop():
    torch.manual_seed(7)
    V, E, B, L = 20, 4, 2, 3
    table = torch.randn(V, E)
    ids = torch.randint(0, V, (B, L, 40))
    w = torch.rand(B, L, 40)
    out = ref.softlexicon_fuse(table, ids, w)
    assert out.shape == (B, L, 4 * E)
    for b in range(B):
        for l in range(L):
            for r in range(4):
                acc = torch.zeros(E)
                for s in range(10):
                    k = r * 10 + s
                    acc += table[ids[b, l, k]] * w[b, l, k]
                torch.testing.assert_close(out[b, l, r * E:(r + 1) * E], acc,
                                           atol=1e-5, rtol=1e-4)


def test_masked_ce():
    torch.manual_seed(8)
    logits = torch.randn(2, 5, 4)
    labels = torch.randint(0, 4, (2, 5))
    mask = torch.tensor([[1, 1, 1, 0, 0], [1, 1, 1, 1, 1]])
    loss = ref.masked_cross_entropy(logits, labels, mask)
    manual = 0.0
    for b in range(2):
        for t in range(5):
            if mask[b, t]:
                manual += torch.nn.functional.cross_entropy(
                    logits[b, t][None], labels[b, t][None]).item()
    assert abs(loss.item() - manual / 8) < 1e-5


def test_dice_loss_runs_and_differentiable():
    logits = torch.randn(2, 5, 4, requires_grad=True)
    labels = torch.randint(0, 4, (2, 5))
    mask = torch.ones(2, 5)
    loss = ref.dice_loss(logits, labels, mask, idx_skip=(0,))
    loss.backward()
    assert torch.isfinite(loss) and torch.isfinite(logits.grad).all()


def test_conv1d_same_matches_tf_same():
    """bert_cnn_crf uses even kernels (2,4) with padding='same'
    (models/layers.py MultiKernelCNN); TF SAME at stride 1 pads
    total=k-1 with the extra zero on the RIGHT (left = (k-1)//2,
    reference tools/layer.py:48-55 tf.layers.conv1d SAME). torch 'same'
    must match bit-exactly for every kernel size."""
    import torch
    import torch.nn.functional as F
    torch.manual_seed(0)
    for k in (2, 3, 4, 5):
        x = torch.randn(2, 8, 50)
        w = torch.randn(16, 8, k)
        b = torch.randn(16)
        y_same = F.conv1d(x, w, b, padding="same")
        total = k - 1
        left = total // 2
        y_tf = F.conv1d(F.pad(x, (left, total - left)), w, b)
        assert torch.equal(y_same, y_tf), f"kernel {k} SAME mismatch"


def test_crf_partition_scan_matches_sequential():
    """The associative-scan partition (round-3 kernel blueprint) matches
    the sequential forward's log Z for random lens, all T/L parities."""
    import torch
    from chinesener_amd.ops import reference as ref
    torch.manual_seed(3)
    for B, L, T in [(4, 7, 5), (3, 1, 4), (5, 16, 10), (2, 33, 7)]:
        em = torch.randn(B, L, T)
        trans = torch.randn(T, T)
        lens = torch.randint(1, L + 1, (B,))
        mask = (torch.arange(L)[None] < lens[:, None]).long()
        tags = torch.randint(0, T, (B, L)) * mask
        ll = ref.crf_log_likelihood(em, tags, mask, trans)
        # recover sequential log Z: score - ll
        score = em[:, 0].gather(1, tags[:, :1]).squeeze(1)
        for t in range(1, L):
            m = mask[:, t].bool()
            step = (trans[tags[:, t - 1], tags[:, t]]
                    + em[:, t].gather(1, tags[:, t:t + 1]).squeeze(1))
            score = score + step * m.float()
        logz_seq = score - ll
        logz_scan = ref.crf_partition_scan(em, mask, trans)
        torch.testing.assert_close(logz_scan, logz_seq, atol=1e-4,
                                   rtol=1e-5)
