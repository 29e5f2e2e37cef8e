"""Direct tests for TagMetrics (reference tools/train_utils.py:105-142
semantics) and CheckpointManager (reference tools/utils.py:49-61 layout,
keep_checkpoint_max pruning)."""
import torch

from chinesener_amd.train.checkpoints import (CheckpointManager, ckpt_dir,
                                              clear_model)
from chinesener_amd.train.metrics import TagMetrics

IDX2TAG = {0: "[PAD]", 1: "[CLS]", 2: "[SEP]", 3: "O", 4: "B-LOC", 5: "I-LOC"}


def test_tag_metrics_hand_case():
    m = TagMetrics(6, IDX2TAG)
    #             O  B  I  O          (label)   mask last off
    labels = torch.tensor([[3, 4, 5, 3, 3]])
    preds = torch.tensor([[3, 4, 3, 3, 4]])   # I-LOC missed; last masked out
    mask = torch.tensor([[1, 1, 1, 1, 0]])
    m.update(preds, labels, mask)
    out = m.compute()
    # 4 kept tokens, 3 correct
    assert abs(out["accuracy"] - 3 / 4) < 1e-9
    # entity tags only: TP(B-LOC)=1; predicted B/I total 1; support B+I = 2
    # micro P = 1/1, R = 1/2 -> F1 = 2/3
    assert abs(out["micro_f1"] - 2 / 3) < 1e-9
    assert abs(out["B-LOC_f1"] - 1.0) < 1e-9
    assert out["I-LOC_recall"] == 0.0


def test_tag_metrics_special_tokens_excluded():
    m = TagMetrics(6, IDX2TAG)
    labels = torch.tensor([[1, 3, 2]])   # CLS, O, SEP
    preds = torch.tensor([[1, 3, 2]])
    mask = torch.ones(1, 3, dtype=torch.long)
    m.update(preds, labels, mask)
    out = m.compute()
    # only the O token counts
    assert int(m.conf.sum()) == 1
    assert out["accuracy"] == 1.0


def test_tag_metrics_accumulates_across_batches():
    m = TagMetrics(6, IDX2TAG)
    for _ in range(3):
        m.update(torch.tensor([[4]]), torch.tensor([[4]]),
                 torch.tensor([[1]]))
    assert int(m.conf[4, 4]) == 3


def test_ckpt_dir_layout():
    assert ckpt_dir("msra", "bert_bilstm_crf", "/tmp/x") == \
        "/tmp/x/ner_msra_bert_bilstm_crf"


def test_checkpoint_save_restore_prune(tmp_path):
    model = torch.nn.Linear(4, 2)
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    cm = CheckpointManager(str(tmp_path), keep_max=3)
    assert cm.latest() is None
    # a step so optimizer state exists
    model(torch.randn(2, 4)).sum().backward()
    opt.step()
    for step in (100, 200, 300, 400, 500):
        cm.save(step, model, opt)
    paths = cm._paths()
    assert [s for s, _ in paths] == [300, 400, 500]   # pruned to keep_max
    assert cm.latest().endswith("ckpt-500.pt")

    model2 = torch.nn.Linear(4, 2)
    opt2 = torch.optim.Adam(model2.parameters(), lr=1e-3)
    step = cm.restore(model2, opt2)
    assert step == 500
    for a, b in zip(model.parameters(), model2.parameters()):
        assert torch.equal(a, b)
    assert len(opt2.state) == len(opt.state)


def test_clear_model(tmp_path):
    d = tmp_path / "ner_msra_m"
    d.mkdir()
    (d / "ckpt-1.pt").write_bytes(b"x")
    clear_model(str(d))
    assert not d.exists()
    clear_model(str(d))   # idempotent on missing dir
