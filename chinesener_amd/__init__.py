"""chinesener_amd — MI355X-native Chinese NER training/serving framework.

A from-scratch rebuild of the capabilities of DSXiangLi/ChineseNER
(reference surveyed in SURVEY.md) designed MI355X-first:

* PyTorch-ROCm host, hand-written CDNA4 (gfx950) HIP kernels for the hot
  ops (fused attention incl. TENER relative-position variant, LayerNorm,
  bias-GELU, BiLSTM recurrence, CRF forward-backward + Viterbi,
  SoftLexicon fusion, multi-tensor Adam) — see ``csrc/``.
* RCCL over xGMI data parallelism with bucketed gradient all-reduce
  overlapped with backward — see ``chinesener_amd/dist``.
* hipGraph-captured inference for serving — see ``chinesener_amd/serve``.

Layer map mirrors the reference: data/ (L0-L1),
data/loader.py (L2), models/ (L3), train/ (L4), eval/ (L5), serve/ (L6).
"""

__version__ = "0.1.0"
