"""Property-based fuzz tests (hypothesis) for the algorithmic cores:
CRF reference vs brute-force enumeration on random inputs, strict-span
extraction invariants, softlexicon fuse vs a plain numpy re-derivation,
and Viterbi optimality. Complements the fixed fixtures in
tests/test_ops_reference.py."""
import itertools

import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from chinesener_amd.eval.entity_eval import extract_spans
from chinesener_amd.ops import reference as ref

TYPES = ["LOC", "PER", "ORG"]


@st.composite
def tag_sequences(draw):
    n = draw(st.integers(1, 12))
    tags = []
    for _ in range(n):
        kind = draw(st.sampled_from(["O", "B", "I"]))
        if kind == "O":
            tags.append("O")
        else:
            tags.append(f"{kind}-{draw(st.sampled_from(TYPES))}")
    return tags


@given(tag_sequences())
@settings(max_examples=200, deadline=None)
def test_extract_spans_invariants(tags):
    spans = extract_spans(tags)
    seen_positions = set()
    for typ, s, e in spans:
        # well-formed, in-range, typed
        assert 0 <= s < e <= len(tags)
        assert typ in TYPES
        # spans never overlap
        assert not (set(range(s, e)) & seen_positions)
        seen_positions.update(range(s, e))
        # first position of a span is B- or a type-switching I-
        assert tags[s][2:] == typ
    # every B- tag opens exactly one span
    n_b = sum(1 for t in tags if t.startswith("B-"))
    # I- after O or after a different type also opens one (seqeval default)
    n_orphan_i = sum(
        1 for i, t in enumerate(tags)
        if t.startswith("I-") and (i == 0 or tags[i - 1] == "O"
                                   or tags[i - 1][2:] != t[2:]))
    assert len(spans) == n_b + n_orphan_i


@given(st.integers(2, 4), st.integers(1, 5), st.randoms())
@settings(max_examples=60, deadline=None)
def test_crf_loglik_matches_bruteforce(T, L, rnd):
    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    em = torch.randn(1, L, T, generator=g)
    trans = torch.randn(T, T, generator=g)
    tags = torch.randint(0, T, (1, L), generator=g)
    mask = torch.ones(1, L, dtype=torch.long)
    ll = ref.crf_log_likelihood(em, tags, mask, trans)[0]

    def score(path):
        s = sum(float(em[0, t, p]) for t, p in enumerate(path))
        s += sum(float(trans[path[t], path[t + 1]]) for t in range(L - 1))
        return s

    scores = [score(p) for p in itertools.product(range(T), repeat=L)]
    logZ = float(torch.logsumexp(torch.tensor(scores), 0))
    expect = score([int(x) for x in tags[0]]) - logZ
    assert abs(float(ll) - expect) < 1e-4


@given(st.integers(2, 4), st.integers(1, 5), st.randoms())
@settings(max_examples=60, deadline=None)
def test_crf_viterbi_is_optimal(T, L, rnd):
    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    em = torch.randn(1, L, T, generator=g)
    trans = torch.randn(T, T, generator=g)
    mask = torch.ones(1, L, dtype=torch.long)
    pred = ref.crf_decode(em, mask, trans)[0, :L].tolist()

    def score(path):
        s = sum(float(em[0, t, p]) for t, p in enumerate(path))
        s += sum(float(trans[path[t], path[t + 1]]) for t in range(L - 1))
        return s

    best = max(itertools.product(range(T), repeat=L), key=score)
    assert abs(score(pred) - score(list(best))) < 1e-5


@given(st.integers(1, 3), st.integers(1, 6), st.integers(2, 9),
       st.integers(1, 8), st.randoms())
@settings(max_examples=60, deadline=None)
def test_softlexicon_fuse_matches_numpy(B, L, V, E, rnd):
    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    table = torch.randn(V, E, generator=g)
    ids = torch.randint(0, V, (B, L, 40), generator=g)
    weights = torch.rand(B, L, 40, generator=g)
    out = ref.softlexicon_fuse(table, ids, weights).numpy()
    t, i, w = table.numpy(), ids.numpy(), weights.numpy()
    expect = np.zeros((B, L, 4 * E), dtype=np.float32)
    for b in range(B):
        for l in range(L):
            for r in range(4):
                acc = np.zeros(E, dtype=np.float32)
                for s in range(10):
                    k = r * 10 + s
                    acc += w[b, l, k] * t[i[b, l, k]]
                expect[b, l, r * E:(r + 1) * E] = acc
    np.testing.assert_allclose(out, expect, atol=1e-5)


@given(st.integers(2, 5), st.integers(2, 6), st.randoms())
@settings(max_examples=40, deadline=None)
def test_crf_masked_suffix_ignored(T, L, rnd):
    """Positions beyond the mask must not affect the log-likelihood."""
    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    em = torch.randn(1, L, T, generator=g)
    trans = torch.randn(T, T, generator=g)
    tags = torch.randint(0, T, (1, L), generator=g)
    n_valid = 1 + rnd.randint(0, L - 1)
    mask = torch.zeros(1, L, dtype=torch.long)
    mask[0, :n_valid] = 1
    ll = ref.crf_log_likelihood(em, tags, mask, trans)
    em2 = em.clone()
    em2[0, n_valid:] = 999.0          # garbage in the padded tail
    tags2 = tags.clone()
    tags2[0, n_valid:] = (tags2[0, n_valid:] + 1) % T
    ll2 = ref.crf_log_likelihood(em2, tags2, mask, trans)
    assert abs(float(ll) - float(ll2)) < 1e-4


@given(st.integers(1, 3), st.integers(1, 8), st.integers(1, 3),
       st.sampled_from(["tanh", "relu"]), st.randoms())
@settings(max_examples=30, deadline=None)
def test_bilstm_reference_matches_cell_math(B, L, h, act, rnd):
    """Reference BiLSTM vs explicit per-step LSTM cell recursion."""
    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    E = 4
    x = torch.randn(B, L, E, generator=g)
    mk = lambda *sh: torch.randn(*sh, generator=g) * 0.3
    w_ih_f, w_hh_f, b_f = mk(E, 4 * h), mk(h, 4 * h), mk(4 * h)
    w_ih_b, w_hh_b, b_b = mk(E, 4 * h), mk(h, 4 * h), mk(4 * h)
    lens = torch.tensor([1 + rnd.randint(0, L - 1) for _ in range(B)])
    out = ref.bilstm_forward(x, w_ih_f, w_hh_f, b_f, w_ih_b, w_hh_b, b_b,
                             lens, act)

    actf = torch.tanh if act == "tanh" else torch.relu

    def run_dir(w_ih, w_hh, b, reverse):
        hs = torch.zeros(B, L, h)
        hstate = torch.zeros(B, h)
        cstate = torch.zeros(B, h)
        steps = range(L - 1, -1, -1) if reverse else range(L)
        for t in steps:
            gates = x[:, t] @ w_ih + hstate @ w_hh + b
            i, f, gg, o = gates.split(h, dim=1)
            c_new = torch.sigmoid(f) * cstate + torch.sigmoid(i) * actf(gg)
            h_new = torch.sigmoid(o) * actf(c_new)
            valid = (lens > t).float()[:, None]
            cstate = valid * c_new + (1 - valid) * cstate
            hstate = valid * h_new + (1 - valid) * hstate
            hs[:, t] = valid * h_new
        return hs

    expect = torch.cat([run_dir(w_ih_f, w_hh_f, b_f, False),
                        run_dir(w_ih_b, w_hh_b, b_b, True)], dim=-1)
    torch.testing.assert_close(out, expect, atol=2e-5, rtol=1e-4)


@st.composite
def cjk_sentences(draw):
    n = draw(st.integers(1, 12))
    return "".join(chr(0x4E00 + draw(st.integers(0, 99))) for _ in range(n))


@given(cjk_sentences())
@settings(max_examples=100, deadline=None)
def test_word_enhance_shape_invariants(sentence):
    """softword/ex_softword/softlexicon outputs always align 1:1 with the
    characters (the reference asserts this during preprocessing,
    data/word_enhance.py:118,225-227)."""
    from chinesener_amd.data.word_enhance import (Lexicon, build_ex_softword,
                                                  build_soft_lexicon,
                                                  build_softword)
    lex = Lexicon.synthetic([chr(0x4E00 + i) for i in range(100)],
                            n_words=200, seed=7)
    sw = build_softword(sentence, lex)
    assert len(sw) == len(sentence)
    assert all(0 <= t < 5 for t in sw)
    ex = build_ex_softword(sentence, lex)
    assert len(ex) == len(sentence)
    assert all(len(r) == 5 and set(r) <= {0, 1} for r in ex)
    ids, weights = build_soft_lexicon(sentence, lex)
    assert ids.shape == (len(sentence), 40)
    assert weights.shape == (len(sentence), 40)
    assert (ids >= 0).all() and (ids < len(lex)).all()
    # weights are a distribution over the 40 slots (or all-zero rows when
    # only <None> entries carry zero frequency)
    sums = weights.sum(1)
    assert ((np.abs(sums - 1.0) < 1e-5) | (sums == 0)).all()


@given(st.integers(1, 4), st.integers(2, 10), st.randoms())
@settings(max_examples=50, deadline=None)
def test_tag_metrics_matches_sklearn(B, L, rnd):
    """TagMetrics confusion math vs sklearn on random unmasked tokens."""
    from sklearn.metrics import precision_recall_fscore_support
    from chinesener_amd.train.metrics import TagMetrics
    idx2tag = {0: "[PAD]", 1: "O", 2: "B-LOC", 3: "I-LOC"}
    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    labels = torch.randint(1, 4, (B, L), generator=g)
    preds = torch.randint(1, 4, (B, L), generator=g)
    mask = torch.ones(B, L, dtype=torch.long)
    m = TagMetrics(4, idx2tag)
    m.update(preds, labels, mask)
    out = m.compute()
    y, p = labels.reshape(-1).numpy(), preds.reshape(-1).numpy()
    acc = (y == p).mean()
    assert abs(out["accuracy"] - acc) < 1e-9
    # micro over entity tags (2, 3), sklearn labels= restricted micro
    prec, rec, f1, _ = precision_recall_fscore_support(
        y, p, labels=[2, 3], average="micro", zero_division=0)
    if "micro_f1" in out:
        assert abs(out["micro_f1"] - f1) < 1e-9


@given(st.integers(1, 3), st.integers(1, 6), st.integers(2, 5), st.randoms())
@settings(max_examples=50, deadline=None)
def test_masked_ce_matches_manual(B, L, T, rnd):
    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    logits = torch.randn(B, L, T, generator=g)
    labels = torch.randint(0, T, (B, L), generator=g)
    mask = (torch.rand(B, L, generator=g) > 0.3).long()
    if mask.sum() == 0:
        mask[0, 0] = 1
    loss = ref.masked_cross_entropy(logits, labels, mask)
    logp = torch.log_softmax(logits, -1)
    nll = -logp.gather(-1, labels[..., None]).squeeze(-1)
    expect = (nll * mask).sum() / mask.sum()
    assert abs(float(loss) - float(expect)) < 1e-5


@given(tag_sequences())
@settings(max_examples=100, deadline=None)
def test_chunk_roundtrip_preserves_sentence(tags):
    """augment.chunk_by_tag -> chunks_to_bio reproduces the text and a
    normalized tag sequence (I- continuations after O become B- starts,
    matching strict-span semantics)."""
    from chinesener_amd.data.augment import chunk_by_tag, chunks_to_bio
    sent = [chr(0x4E00 + i) for i in range(len(tags))]
    s2, t2 = chunks_to_bio(chunk_by_tag(sent, tags))
    assert s2 == sent
    # spans survive the round trip exactly
    assert extract_spans(t2) == extract_spans(tags)


@given(st.integers(1, 3), st.integers(2, 8), st.randoms())
@settings(max_examples=50, deadline=None)
def test_mrc_span_labels_roundtrip(B, L, rnd):
    """mrc.span_model.make_span_labels start/end/span tensors are mutually
    consistent with the BIO input."""
    from chinesener_amd.mrc.span_model import make_span_labels
    g = torch.Generator().manual_seed(rnd.randint(0, 2**31))
    # random BIO over {0,1,2} with I only continuing something
    labels = torch.zeros(B, L, dtype=torch.long)
    for b in range(B):
        t = 0
        while t < L:
            if rnd.random() < 0.4:
                length = 1 + rnd.randint(0, min(3, L - t - 1))
                labels[b, t] = 1
                labels[b, t + 1:t + length] = 2
                t += length
            else:
                t += 1
    start, end, span = make_span_labels(labels)
    assert (start.sum(1) == end.sum(1)).all()        # every span closes
    assert (span.sum((1, 2)) == start.sum(1)).all()  # one cell per span
    for b in range(B):
        for s, e in span[b].nonzero().tolist():
            assert s <= e
            assert labels[b, s] == 1
            assert (labels[b, s + 1:e + 1] == 2).all()
            if e + 1 < L:
                assert labels[b, e + 1] != 2


@given(st.integers(1, 5), st.integers(1, 7),
       st.sampled_from(["int64", "int32", "float32", "float64"]),
       st.randoms())
@settings(max_examples=60, deadline=None)
def test_rpc_msgpack_ndarray_roundtrip(a, b, dtype, rnd):
    """serve.rpc framing round-trips arbitrary arrays bit-exactly."""
    from chinesener_amd.serve import rpc
    arr = (np.random.default_rng(rnd.randint(0, 2**31))
           .standard_normal((a, b)) * 100).astype(dtype)
    msg = {"inputs": {"x": arr}, "model_spec": {"name": "m", "version": 3}}
    out = rpc.loads(rpc.dumps(msg))
    assert out["model_spec"] == {"name": "m", "version": 3}
    got = out["inputs"]["x"]
    assert got.dtype == arr.dtype and got.shape == arr.shape
    assert (got == arr).all()


@given(st.integers(10, 600), st.floats(0.01, 0.5), st.floats(1e-6, 1e-2))
@settings(max_examples=25, deadline=None)
def test_lr_schedule_bert_shape_properties(total, wr, base):
    """bert schedule: rises monotonically through warmup, peaks at the
    warmup boundary, decays monotonically to ~0, never negative."""
    from chinesener_amd.train.optimizers import LrSchedule
    s = LrSchedule("bert", base, num_train_steps=total, warmup_ratio=wr)
    lrs = [s.lr_at(t) for t in range(1, total + 1)]
    assert all(lr >= 0 for lr in lrs)
    w = s.warmup
    for a, b in zip(lrs[:w - 1], lrs[1:w]):
        assert b >= a                       # warmup non-decreasing
    for a, b in zip(lrs[w:], lrs[w + 1:]):
        assert b <= a + 1e-12               # decay non-increasing
    assert abs(max(lrs) - base) <= base * (1.0 / max(1, w)) + 1e-12
    assert lrs[-1] <= base * 0.02 + 1e-9    # ends near zero


@given(cjk_sentences())
@settings(max_examples=80, deadline=None)
def test_bert_feature_invariants(sentence):
    """build_seq_feature: [CLS] ... [SEP] framing, mask/seq_len agree,
    label row aligned and padded (reference base_preprocess invariants,
    :183-201)."""
    from chinesener_amd.data.preprocess import get_instance
    from chinesener_amd.data.tokenizer import Vocab
    tag2idx = {"[PAD]": 0, "O": 1, "B-LOC": 2, "I-LOC": 3,
               "[CLS]": 4, "[SEP]": 5}
    proc = get_instance("bert", 16, tag2idx, vocab=Vocab.synthetic())
    tok = proc.tokenizer
    tags = ["O"] * len(sentence)
    feat = proc.build_seq_feature(sentence, tags)
    ids, mask, labels = feat["token_ids"], feat["mask"], feat["label_ids"]
    n = int(feat["seq_len"])
    assert len(ids) == len(mask) == len(labels) == 16
    assert mask.sum() == n
    assert (mask[:n] == 1).all() and (mask[n:] == 0).all()
    assert ids[0] == tok.vocab.stoi["[CLS]"]
    assert ids[n - 1] == tok.vocab.stoi["[SEP]"]
    assert labels[0] == tag2idx["[CLS]"] and labels[n - 1] == tag2idx["[SEP]"]
    assert (labels[n:] == tag2idx["[PAD]"]).all()


@given(cjk_sentences(), st.sampled_from(["LOC", "PER", "ORG"]))
@settings(max_examples=60, deadline=None)
def test_mrc_feature_invariants(text, tag_type):
    """MRC [CLS]+query+[SEP]+text framing: segment/text-mask boundaries,
    label region restricted to the text, length accounting."""
    from chinesener_amd.data.tokenizer import Vocab, WordpieceTokenizer
    from chinesener_amd.mrc.dataset import build_single_feature
    tok = WordpieceTokenizer(Vocab.synthetic())
    tags = ["O"] * len(text)
    if text:
        tags[0] = f"B-{tag_type}"
        for i in range(1, min(2, len(text))):
            tags[i] = f"I-{tag_type}"
    f = build_single_feature(tok, "find entities", text, tags, tag_type,
                             max_seq_len=64)
    q, t = int(f["query_len"]), int(f["text_len"])
    n = len(f["token_ids"])
    assert n == q + t <= 64
    assert (f["segment_ids"][:q] == 0).all()
    assert (f["segment_ids"][q:] == 1).all()
    assert (f["text_mask"][:q] == 0).all()
    assert (f["text_mask"][q:] == 1).all()
    # labels only inside the text region
    assert (f["label_ids"][:q] == 0).all()
    if text:
        assert f["label_ids"][q] == 1      # the B we planted


@given(cjk_sentences())
@settings(max_examples=100, deadline=None)
def test_tokenizer_id_roundtrip(text):
    """convert_tokens_to_ids stays in-vocab and maps known chars back."""
    from chinesener_amd.data.tokenizer import Vocab, WordpieceTokenizer
    tok = WordpieceTokenizer(Vocab.synthetic())
    toks = tok.tokenize(text)
    ids = tok.convert_tokens_to_ids(toks)
    assert len(ids) == len(toks)
    itos = tok.vocab.itos
    assert all(0 <= i < len(itos) for i in ids)
    for t, i in zip(toks, ids):
        if t in tok.vocab.stoi:
            assert itos[i] == t
        else:
            assert itos[i] == "[UNK]"


@given(st.lists(st.integers(1, 10000), min_size=1, max_size=12,
                unique=True), st.integers(1, 5))
@settings(max_examples=40, deadline=None)
def test_checkpoint_manager_invariants(steps, keep):
    """Random save sequences: keep_max enforced, latest() is the max
    step, restore returns it."""
    import tempfile
    import torch as _t
    from chinesener_amd.train.checkpoints import CheckpointManager
    d = tempfile.mkdtemp(prefix="ckfuzz_")
    cm = CheckpointManager(str(d), keep_max=keep)
    model = _t.nn.Linear(2, 2)
    for s_ in steps:
        cm.save(s_, model)
    kept = [s_ for s_, _ in cm._paths()]
    assert kept == sorted(steps)[-keep:]
    assert cm.latest().endswith(f"ckpt-{max(steps)}.pt")
    assert cm.restore(_t.nn.Linear(2, 2)) == max(steps)


@given(
    st.lists(
        st.tuples(
            st.text(alphabet="abcdefgh/_0123456789", min_size=1, max_size=24),
            st.lists(st.integers(1, 5), min_size=0, max_size=3),
            st.sampled_from(["float32", "int64", "int32", "float16"]),
        ),
        min_size=1, max_size=8, unique_by=lambda t: t[0]),
    st.integers(0, 2**31 - 1),
)
@settings(max_examples=30, deadline=None)
def test_tf_bundle_roundtrip_fuzz(specs, seed):
    """TF v2 tensor-bundle writer->reader round-trip over random
    variable names, shapes (incl. scalars) and dtypes."""
    import tempfile

    import numpy as np

    from chinesener_amd.models.tf_checkpoint import (read_tf_checkpoint,
                                                     write_tf_checkpoint)
    rng = np.random.default_rng(seed)
    tensors = {}
    for name, shape, dt in specs:
        if dt.startswith("int"):
            arr = rng.integers(-100, 100, size=shape).astype(dt)
        else:
            arr = rng.standard_normal(size=shape).astype(dt)
        tensors[name] = arr
    with tempfile.TemporaryDirectory() as d:
        import os as _os
        prefix = _os.path.join(d, "m.ckpt")
        write_tf_checkpoint(prefix, tensors)
        back = read_tf_checkpoint(prefix)
    assert set(back) == set(tensors)
    for k, v in tensors.items():
        got = back[k]
        assert got.dtype == v.dtype, (k, got.dtype, v.dtype)
        assert got.shape == v.shape, (k, got.shape, v.shape)
        np.testing.assert_array_equal(np.asarray(got), v, err_msg=k)
