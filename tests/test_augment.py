"""Augmentation suite (reference data/people_daily_augment/): chunkers,
entity replace, synonym replace, sentence shuffle, MLM refill."""
import os
import pickle

import numpy as np
import pytest

from chinesener_amd.data.augment import (EntityReplace, SentenceShuffle,
                                         SynonymReplace, augment,
                                         build_entity_dict, chunk_by_tag,
                                         chunks_to_bio)


SENT = list("张三去了北京大学")
TAGS = ["B-PER", "I-PER", "O", "O", "B-ORG", "I-ORG", "I-ORG", "I-ORG"]


def test_chunk_roundtrip():
    chunks = chunk_by_tag(SENT, TAGS)
    assert chunks == [("张三", "PER"), ("去了", None), ("北京大学", "ORG")]
    s, t = chunks_to_bio(chunks)
    assert s == SENT and t == TAGS


def test_entity_replace_preserves_tags():
    ent = EntityReplace({"PER": ["李四"], "ORG": ["清华大学"]}, p=1.0)
    s, t = chunks_to_bio(ent(chunk_by_tag(SENT, TAGS)))
    assert "".join(s) == "李四去了清华大学"
    assert t[:2] == ["B-PER", "I-PER"]
    assert t[4:] == ["B-ORG", "I-ORG", "I-ORG", "I-ORG"]


def test_entity_replace_variable_length():
    ent = EntityReplace({"PER": ["欧阳锋"]}, p=1.0)
    s, t = chunks_to_bio(ent(chunk_by_tag(SENT, TAGS)))
    assert t[:3] == ["B-PER", "I-PER", "I-PER"]
    assert len(s) == len(t)


def test_synonym_replace_skips_entities():
    vocab = ["去", "了", "到", "过", "走"]
    emb = np.eye(5, dtype=np.float32) + 0.5
    syn = SynonymReplace(vocab, emb, topn=2, p=1.0)
    chunks = syn(chunk_by_tag(SENT, TAGS))
    assert chunks[0] == ("张三", "PER")       # entity untouched
    assert chunks[2] == ("北京大学", "ORG")
    assert len(chunks[1][0]) == 2             # same length, maybe replaced


def test_sentence_shuffle_swaps_clauses():
    sent = list("今天下雨，我在家，他出门")
    tags = ["O"] * len(sent)
    shuf = SentenceShuffle(seed=3)
    s, t = shuf(sent, tags)
    assert sorted("".join(s)) == sorted("".join(sent))
    assert len(t) == len(tags)


def test_build_entity_dict_and_augment(tmp_path):
    d = build_entity_dict(["msra"], str(tmp_path))
    assert set(d) <= {"PER", "LOC", "ORG"}
    assert all(len(v) > 0 for v in d.values())
    path = augment("msra", str(tmp_path), entity_dict=d,
                   methods=["entity_replace", "sentence_shuffle"])
    assert os.path.exists(path)
    with open(path, "rb") as f:
        blob = pickle.load(f)
    assert len(blob["sentences"]) > 0
    assert all(len(s) == len(t)
               for s, t in zip(blob["sentences"], blob["tags"]))


def test_mlm_augment_preserves_entities():
    from chinesener_amd.data.augment_mlm import MlmSR
    mlm = MlmSR(mask_prob=0.5, seed=7)
    s, t = mlm(SENT, TAGS)
    assert len(s) == len(SENT) and t == TAGS
    # entity chars never rewritten
    assert s[0] == "张" and s[1] == "三"
    assert "".join(s[4:]) == "北京大学"


def test_mlm_head_shapes():
    import torch
    from chinesener_amd.models.bert import BertConfig, BertMlmHead, BertModel
    cfg = BertConfig(vocab_size=300, hidden_size=32, num_hidden_layers=1,
                     num_attention_heads=2, intermediate_size=64)
    bert = BertModel(cfg)
    head = BertMlmHead(bert)
    ids = torch.randint(1, 300, (2, 10))
    logits = head(bert(ids, torch.ones_like(ids)))
    assert logits.shape == (2, 10, 300)
    # tied embedding: grads flow to the word embedding through the head
    logits.sum().backward()
    assert bert.embeddings.word.weight.grad is not None


def test_people_daily_augment_dataset(tmp_path):
    """Virtual dataset: train split comes from train_augment.pkl
    (reference data/people_daily_augment/preprocess.py:8-20)."""
    from chinesener_amd.data.datasets import load_data
    d = build_entity_dict(["people_daily"], str(tmp_path))
    aug_dir = tmp_path / "people_daily_augment"
    path = augment("people_daily", str(tmp_path), entity_dict=d,
                   methods=["entity_replace"])
    # move the dump into the virtual dataset dir
    aug_dir.mkdir(exist_ok=True)
    os.replace(path, aug_dir / "train_augment.pkl")
    s, t = load_data("people_daily_augment", str(aug_dir), "train")
    assert len(s) > 0 and len(s) == len(t)
    # valid split falls back (synthetic here)
    s2, t2 = load_data("people_daily_augment", str(aug_dir), "valid")
    assert len(s2) > 0
