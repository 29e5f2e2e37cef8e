"""Data-layer tests: tokenizers, trie, word-enhance features, preprocess
shapes, loaders (reference test strategy upgraded per SURVEY.md §4)."""
import numpy as np
import pytest
import torch

from chinesener_amd.data import word_enhance as we
from chinesener_amd.data.datasets import get_spec, synthetic_corpus
from chinesener_amd.data.loader import MultiDataset, NerDataset, make_synthetic_batch
from chinesener_amd.data.preprocess import extract_prefix_surfix, get_instance
from chinesener_amd.data.tokenizer import (CharTokenizer, Vocab,
                                           WordpieceTokenizer, full_to_half,
                                           get_tokenizer)
from chinesener_amd.data.trie import Trie
from chinesener_amd.data.word_enhance import Lexicon


def test_full_to_half():
    assert full_to_half("Ａｂ１！") == "Ab1!"
    assert full_to_half("　") == " "


def test_trie_match():
    t = Trie(["南京", "南京市", "市长", "长江", "长江大桥", "大桥"])
    assert "南京" in t and "南京市" in t and "北京" not in t
    assert t.prefixes("南京市长江大桥", 0) == ["南京", "南京市"]
    spans = t.max_match_segment("南京市长江大桥")
    words = ["南京市长江大桥"[s:e] for s, e in spans]
    assert words == ["南京市", "长江大桥"]


def test_char_tokenizer_roundtrip():
    v = Vocab.synthetic(500)
    tok = CharTokenizer(v)
    ids = tok.convert_tokens_to_ids(tok.tokenize(v.itos[10] + v.itos[20]))
    assert ids == [10, 20]


def test_wordpiece_unknown():
    v = Vocab.synthetic(200)
    tok = WordpieceTokenizer(v)
    toks = tok.tokenize("X" + v.itos[9])   # X not in CJK synthetic vocab
    assert toks[0] == "[UNK]" and toks[1] == v.itos[9]


def _lexicon():
    chars = [chr(0x4E00 + i) for i in range(50)]
    return Lexicon.synthetic(chars, n_words=100, dim=8, seed=7)


def test_softword_bmes():
    lex = _lexicon()
    w = next(w for w in lex.words[3:] if len(w) == 3)
    ids = we.build_softword(w, lex)
    assert ids == [we.SOFT2IDX["B"], we.SOFT2IDX["M"], we.SOFT2IDX["E"]]


def test_ex_softword_multihot():
    lex = _lexicon()
    w = next(w for w in lex.words[3:] if len(w) == 2)
    hot = we.build_ex_softword(w, lex)
    assert hot[0][we.SOFT2IDX["B"]] == 1
    assert hot[1][we.SOFT2IDX["E"]] == 1
    # char with no match gets None
    hot2 = we.build_ex_softword("龠", lex)   # char outside lexicon range
    assert hot2[0][we.SOFT2IDX["None"]] == 1


def test_softlexicon_shapes_and_weights():
    lex = _lexicon()
    sent = lex.words[5] + lex.words[6]
    ids, wts = we.build_soft_lexicon(sent, lex)
    n = len(sent)
    assert ids.shape == (n, 40) and wts.shape == (n, 40)
    # weights normalized per char
    np.testing.assert_allclose(wts.sum(1), np.ones(n), rtol=1e-5)


def test_preprocess_bert_frame():
    spec = get_spec("msra")
    proc = get_instance("bert", 32, spec.tag2idx)
    feat = proc.build_seq_feature("中国人民银行", ["B-ORG"] + ["I-ORG"] * 5)
    assert feat["token_ids"].shape == (32,)
    assert feat["seq_len"] == 8          # 6 chars + CLS/SEP
    assert feat["label_ids"][0] == spec.tag2idx["[CLS]"]
    assert feat["label_ids"][7] == spec.tag2idx["[SEP]"]
    assert feat["mask"][:8].sum() == 8 and feat["mask"][8:].sum() == 0


def test_preprocess_softlexicon_alignment():
    spec = get_spec("msra")
    proc = get_instance("bert", 32, spec.tag2idx, "softlexicon")
    sent = "".join(proc.lexicon.words[5])
    feat = proc.build_seq_feature(sent, ["O"] * len(sent))
    # row 0 is CLS -> all-zero enhance row; row 1 aligns with sentence[0]
    assert feat["softlexicon_ids"].shape == (32, 40)
    assert feat["softlexicon_ids"][0].sum() == 0
    assert feat["softlexicon_ids"][1].sum() > 0


def test_extract_prefix_surfix():
    assert extract_prefix_surfix("bert_bilstm_crf") == (None, "bert")
    assert extract_prefix_surfix("bilstm_crf_softlexicon") == ("softlexicon", "char")
    assert extract_prefix_surfix("bert_bilstm_crf_softlexicon") == ("softlexicon", "bert")
    assert extract_prefix_surfix("transformer_crf_bichar") == ("bichar", "char")


def test_synthetic_corpus_tags_valid():
    spec = get_spec("msra")
    sents, tags = synthetic_corpus(spec, "valid", n=20)
    for s, t in zip(sents, tags):
        assert len(s) == len(t)
        assert all(tag in spec.tag2idx for tag in t)


def test_ner_dataset_batches(tmp_path):
    pipe = NerDataset(str(tmp_path), "people_daily", 4, 1, "bilstm_crf")
    p = pipe.params
    assert p["label_size"] == get_spec("people_daily").label_size
    assert p["step_per_epoch"] > 0
    batch = next(iter(pipe.iter_batches("train")))
    assert batch["token_ids"].shape == (4, 150)
    assert batch["label_ids"].dtype == torch.int64


def test_multidataset_mix(tmp_path):
    md = MultiDataset(str(tmp_path), ["msra", "msr"], 8, 1, "bert_bilstm_crf_mtl")
    p = md.params
    assert p["msra"]["label_size"] == 10 and p["msr"]["label_size"] == 7
    batch = next(iter(md.iter_batches("train")))
    task = batch["task_ids"][:, 0]
    # strict alternation -> exactly half each
    assert int((task == 0).sum()) == 4 and int((task == 1).sum()) == 4


def test_make_synthetic_batch_softlexicon():
    b = make_synthetic_batch(2, 16, word_enhance="softlexicon")
    assert b["softlexicon_ids"].shape == (2, 16, 40)
    s = b["softlexicon_weights"].sum(-1)
    assert torch.allclose(s, torch.ones_like(s), atol=1e-5)


def test_text_embedding_loader(tmp_path):
    # glove format (no header)
    p = tmp_path / "glove.txt"
    p.write_text("你 0.1 0.2 0.3\n好 0.4 0.5 0.6\n", encoding="utf-8")
    from chinesener_amd.data.embeddings import (add_special_tokens,
                                                combine_embeddings,
                                                load_text_embeddings,
                                                normalize_rows)
    vocab, mat = load_text_embeddings(str(p))
    assert vocab == ["你", "好"] and mat.shape == (2, 3)
    # word2vec format (header)
    p2 = tmp_path / "w2v.txt"
    p2.write_text("2 3\n早 1 0 0\n安 0 1 0\n", encoding="utf-8")
    v2, m2 = load_text_embeddings(str(p2))
    assert v2 == ["早", "安"]
    n = normalize_rows(m2)
    assert abs(float((n[0] ** 2).sum()) - 1.0) < 1e-5
    sv, sm = add_special_tokens(vocab, mat)
    assert sv[:3] == ["<None>", "<PAD>", "<eos>"]
    assert (sm[1] == 0).all()  # PAD row zero
    cv, cm = combine_embeddings(vocab, mat, ["你", "你好"],
                                np.array([[9, 9, 9], [7, 7, 7]], np.float32))
    assert cv == ["你", "好", "你好"]
    assert cm.shape == (3, 3) and (cm[2] == 7).all()


def test_device_prefetcher_cpu_passthrough():
    from chinesener_amd.data.loader import DevicePrefetcher
    import torch
    src = [{"x": torch.full((2, 2), float(i))} for i in range(5)]
    out = list(DevicePrefetcher(iter(src), "cpu"))
    assert len(out) == 5
    for i, b in enumerate(out):
        assert torch.equal(b["x"], src[i]["x"])
    # empty iterator
    assert list(DevicePrefetcher(iter([]), "cpu")) == []


def test_real_file_loaders(tmp_path):
    """Every dataset adapter parses its real on-disk format (reference
    data/{msra,people_daily,cluener,msr,weibo}/preprocess.py layouts) —
    not just the synthetic fallback."""
    import json as _json
    from chinesener_amd.data.datasets import load_data

    # msra: {split}/sentences.txt + tags.txt, space-separated
    msra = tmp_path / "msra" / "train"
    msra.mkdir(parents=True)
    (msra / "sentences.txt").write_text(
        "\u5317 \u4eac \u5f88 \u597d\n\u6211 \u7231 \u4e2d \u56fd\n",
        encoding="utf-8")
    (msra / "tags.txt").write_text(
        "B-LOC I-LOC O O\nO O B-LOC I-LOC\n", encoding="utf-8")
    sents, tags = load_data("msra", str(tmp_path / "msra"), "train")
    assert sents == ["\u5317\u4eac\u5f88\u597d",
                     "\u6211\u7231\u4e2d\u56fd"]
    assert tags[0] == ["B-LOC", "I-LOC", "O", "O"]

    # people_daily: CoNLL example.{split}
    pd = tmp_path / "people_daily"
    pd.mkdir()
    (pd / "example.valid").write_text(
        "\u5317 B-LOC\n\u4eac I-LOC\n\n\u597d O\n", encoding="utf-8")
    sents, tags = load_data("people_daily", str(pd), "valid")
    assert sents == ["\u5317\u4eac", "\u597d"]
    assert tags == [["B-LOC", "I-LOC"], ["O"]]

    # cluener: jsonl with char-span labels
    cl = tmp_path / "cluener"
    cl.mkdir()
    rec = {"text": "\u5f20\u4e09\u5728\u5317\u4eac",
           "label": {"name": {"\u5f20\u4e09": [[0, 1]]},
                     "address": {"\u5317\u4eac": [[3, 4]]},
                     "game": {"x": [[2, 2]]}}}    # unmapped type ignored
    (cl / "train.json").write_text(_json.dumps(rec, ensure_ascii=False) + "\n",
                                   encoding="utf-8")
    sents, tags = load_data("cluener", str(cl), "train")
    assert tags[0] == ["B-PER", "I-PER", "O", "B-LOC", "I-LOC"]

    # msr CWS: conll with BIES tags
    msr = tmp_path / "msr"
    msr.mkdir()
    (msr / "train.txt").write_text(
        "\u4e2d B\n\u56fd E\n\u597d S\n\n", encoding="utf-8")
    sents, tags = load_data("msr", str(msr), "train")
    assert tags == [["B", "E", "S"]]

    # weibo: conll {split}.txt
    wb = tmp_path / "weibo"
    wb.mkdir()
    (wb / "train.txt").write_text("\u5317 B-GPE\n\u4eac I-GPE\n\n",
                                  encoding="utf-8")
    sents, tags = load_data("weibo", str(wb), "train")
    assert tags == [["B-GPE", "I-GPE"]]

    # missing files -> synthetic fallback still works
    sents, tags = load_data("weibo", str(tmp_path / "nowhere"), "train")
    assert len(sents) > 0 and len(sents) == len(tags)


def test_max_prob_segment_vs_max_match():
    """The maxprob segmenter follows jieba's DAG + max-log-prob route:
    where forward max-match greedily grabs the longest prefix, maxprob
    picks the higher-frequency path (reference jieba.cut semantics,
    word_enhance.py:244)."""
    import numpy as np
    from chinesener_amd.data.word_enhance import Lexicon, build_softword

    words = ["<PAD>", "<None>", "<eos>", "ab", "abc", "cd"]
    freq = np.array([0, 0, 0, 1000.0, 2.0, 1000.0])
    emb = np.ones((len(words), 4), dtype=np.float32)
    lex = Lexicon(words, freq, emb)

    # FMM grabs 'abc' then 'd'; maxprob prefers ab|cd (2 common words)
    assert lex.trie.max_match_segment("abcd") == [(0, 3), (3, 4)]
    assert lex.max_prob_segment("abcd") == [(0, 2), (2, 4)]

    # softword ids follow the chosen segmenter
    assert build_softword("abcd", lex, segmenter="maxmatch") == [1, 2, 3, 4]
    assert build_softword("abcd", lex, segmenter="maxprob") == [1, 3, 1, 3]

    # tie on score prefers the longer word (jieba max((score, end)))
    words2 = ["<PAD>", "<None>", "<eos>", "xy", "xyz"]
    freq2 = np.array([0, 0, 0, 5.0, 5.0])
    lex2 = Lexicon(words2, freq2, np.ones((5, 4), np.float32))
    assert lex2.max_prob_segment("xyz") == [(0, 3)]

    # OOV text degrades to single chars (freq-1 convention)
    assert lex.max_prob_segment("zz") == [(0, 1), (1, 2)]


def test_multidataset_reshuffles_each_pass(tmp_path):
    """The shorter task's corpus repeats within one mtl epoch; each
    repeat pass must use a fresh shuffle order (the seed was
    constant across passes)."""
    from chinesener_amd.data.loader import MultiDataset
    md = MultiDataset(str(tmp_path), ["msra", "msr"], batch_size=4,
                      epochs=1, model_name="bert_bilstm_crf_mtl")
    # drive enough batches that task streams wrap their corpora
    orders = {0: [], 1: []}
    n_batches = 0
    for batch in md.iter_batches("train"):
        n_batches += 1
        tid = int(batch["task_ids"][0, 0])
        orders[tid].append(batch["token_ids"][0].numpy().tobytes())
        if n_batches >= 400:
            break
    # find a wrap in task-1's stream: the same sample set must not come
    # back in the same order (first-k fingerprints differ across passes)
    seq = orders[1] if len(orders[1]) > 10 else orders[0]
    seen = {}
    repeat_pairs = 0
    same_successor = 0
    for i, fp in enumerate(seq[:-1]):
        if fp in seen:
            repeat_pairs += 1
            j = seen[fp]
            if j + 1 < len(seq) and seq[j + 1] == seq[i + 1]:
                same_successor += 1
        else:
            seen[fp] = i
    if repeat_pairs >= 5:
        # identical successor for every repeated sample would mean the
        # pass order repeated verbatim
        assert same_successor < repeat_pairs, (
            f"{same_successor}/{repeat_pairs} repeated samples kept the "
            "same successor — shuffle order repeated across passes")
