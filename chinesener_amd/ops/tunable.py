"""hipBLASLt GEMM algorithm selection via PyTorch TunableOp.

profiles/tunableop_gfx950.csv holds the offline-tuned winners for the
bench/training GEMM shapes on MI355X (tuned once with
PYTORCH_TUNABLEOP_TUNING=1; ~14% end-to-end step time on
bert_bilstm_crf). load_tuned_gemm_table() activates them read-only —
unknown shapes fall back to the default heuristic."""
from __future__ import annotations

import logging
import os

log = logging.getLogger("chinesener_amd")

_DEFAULT = os.path.join(os.path.dirname(os.path.dirname(
    os.path.dirname(os.path.abspath(__file__)))), "profiles",
    "tunableop_gfx950.csv")


def load_tuned_gemm_table(path: str | None = None) -> bool:
    path = path or _DEFAULT
    if not os.path.exists(path):
        return False
    try:
        import torch
        if not torch.cuda.is_available():
            return False
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        tunable.tuning_enable(False)
        tunable.read_file(path)
        log.info("TunableOp GEMM table loaded from %s", path)
        return True
    except Exception as e:  # pragma: no cover
        log.warning("TunableOp load failed: %s", e)
        return False
