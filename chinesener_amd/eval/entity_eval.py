"""Entity-level (strict span) and tag-level evaluation reports.

Native re-implementation of what the reference gets from seqeval's
`classification_report(scheme=strict)` + sklearn's tag report
(reference evaluation.py:38-55); seqeval is not in this image, so the
strict BIO/BIES span matcher is implemented here and unit-tested
against hand-computed fixtures."""
from __future__ import annotations

from typing import Dict, List, Sequence, Tuple

Span = Tuple[str, int, int]  # (type, start, end-exclusive)


def extract_spans(tags: Sequence[str]) -> List[Span]:
    """Strict BIO span extraction: a span is B-X followed by I-X* ; an
    I-X without a preceding B-X/I-X of the same type starts a new span
    (seqeval default behaviour). Also accepts BIES (B/M/E/S, used by the
    msr CWS adapter, reference data/msr/preprocess.py)."""
    spans: List[Span] = []
    start, cur = None, None

    def flush(end):
        nonlocal start, cur
        if start is not None:
            spans.append((cur, start, end))
        start, cur = None, None

    for i, tag in enumerate(tags):
        if tag.startswith("B-"):
            flush(i)
            start, cur = i, tag[2:]
        elif tag.startswith("I-"):
            if cur != tag[2:]:
                flush(i)
                start, cur = i, tag[2:]
        elif tag in ("B", "S"):  # BIES segmentation
            flush(i)
            start, cur = i, "SEG"
            if tag == "S":
                flush(i + 1)
        elif tag in ("M", "E"):
            if cur != "SEG":
                flush(i)
                start, cur = i, "SEG"
            if tag == "E":
                flush(i + 1)
        else:
            flush(i)
    flush(len(tags))
    return spans


def _prf(tp: int, pred: int, true: int) -> Tuple[float, float, float]:
    p = tp / pred if pred else 0.0
    r = tp / true if true else 0.0
    f = 2 * p * r / (p + r) if p + r else 0.0
    return p, r, f


def entity_report(y_true: Sequence[Sequence[str]],
                  y_pred: Sequence[Sequence[str]]) -> Dict[str, Dict[str, float]]:
    """Per-type + micro/macro/weighted strict span P/R/F1 (the numbers
    the reference reads off seqeval's report, evaluation.py:48-55)."""
    tp: Dict[str, int] = {}
    n_pred: Dict[str, int] = {}
    n_true: Dict[str, int] = {}
    for t_tags, p_tags in zip(y_true, y_pred):
        ts, ps = set(extract_spans(t_tags)), set(extract_spans(p_tags))
        for typ, *_ in ts:
            n_true[typ] = n_true.get(typ, 0) + 1
        for typ, *_ in ps:
            n_pred[typ] = n_pred.get(typ, 0) + 1
        for span in ts & ps:
            tp[span[0]] = tp.get(span[0], 0) + 1
    types = sorted(set(n_true) | set(n_pred))
    report: Dict[str, Dict[str, float]] = {}
    for typ in types:
        p, r, f = _prf(tp.get(typ, 0), n_pred.get(typ, 0), n_true.get(typ, 0))
        report[typ] = {"precision": p, "recall": r, "f1": f,
                       "support": n_true.get(typ, 0)}
    total_tp = sum(tp.values())
    total_pred = sum(n_pred.values())
    total_true = sum(n_true.values())
    p, r, f = _prf(total_tp, total_pred, total_true)
    report["micro avg"] = {"precision": p, "recall": r, "f1": f,
                           "support": total_true}
    if types:
        report["macro avg"] = {
            "precision": sum(report[t]["precision"] for t in types) / len(types),
            "recall": sum(report[t]["recall"] for t in types) / len(types),
            "f1": sum(report[t]["f1"] for t in types) / len(types),
            "support": total_true}
        wsum = sum(n_true.get(t, 0) for t in types) or 1
        report["weighted avg"] = {
            k: sum(report[t][k] * n_true.get(t, 0) for t in types) / wsum
            for k in ("precision", "recall", "f1")}
        report["weighted avg"]["support"] = total_true
    return report


def tag_report(y_true: Sequence[Sequence[str]],
               y_pred: Sequence[Sequence[str]]) -> Dict[str, Dict[str, float]]:
    """Token-level per-tag P/R/F1 (the reference's sklearn
    classification_report on flattened tags, evaluation.py:38-46)."""
    tp: Dict[str, int] = {}
    n_pred: Dict[str, int] = {}
    n_true: Dict[str, int] = {}
    for t_tags, p_tags in zip(y_true, y_pred):
        for t, p in zip(t_tags, p_tags):
            n_true[t] = n_true.get(t, 0) + 1
            n_pred[p] = n_pred.get(p, 0) + 1
            if t == p:
                tp[t] = tp.get(t, 0) + 1
    report = {}
    for tag in sorted(set(n_true) | set(n_pred)):
        p, r, f = _prf(tp.get(tag, 0), n_pred.get(tag, 0), n_true.get(tag, 0))
        report[tag] = {"precision": p, "recall": r, "f1": f,
                       "support": n_true.get(tag, 0)}
    return report


def report_to_text(report: Dict[str, Dict[str, float]], title: str = "") -> str:
    lines = []
    if title:
        lines.append(title)
    lines.append(f"{'':>14} {'precision':>9} {'recall':>9} {'f1':>9} {'support':>9}")
    for name, row in report.items():
        lines.append(f"{name:>14} {row['precision']:9.4f} {row['recall']:9.4f} "
                     f"{row['f1']:9.4f} {int(row.get('support', 0)):9d}")
    return "\n".join(lines)
