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


def load_tuned_gemm_table(path: str | None = None, tune: bool = False) -> bool:
    """Enable TunableOp GEMM selection.

    tune=True keeps tuning ON: unseen shapes are tuned on first
    encounter (do this during warmup, then call freeze() before the
    timed/production region). The committed table warm-starts so only
    novel shapes pay the tuning cost."""
    if os.environ.get("CHINESENER_NO_TUNABLE") == "1":
        return False
    if "PYTORCH_TUNABLEOP_ENABLED" in os.environ:
        return True  # env-configured at process start — authoritative
    path = path or _DEFAULT
    try:
        import torch
        if not torch.cuda.is_available():
            return False
        import torch.cuda.tunable as tunable
        tunable.enable(True)
        if os.path.exists(path):
            tunable.read_file(path)
            log.info("TunableOp GEMM table loaded from %s", path)
        tunable.tuning_enable(bool(tune))
        return True
    except Exception as e:  # pragma: no cover
        log.warning("TunableOp load failed: %s", e)
        return False


def freeze(dump_path: str | None = None) -> None:
    """Stop tuning (call after warmup). This torch has no write_file;
    results are flushed to PYTORCH_TUNABLEOP_FILENAME at process exit."""
    if "PYTORCH_TUNABLEOP_ENABLED" in os.environ:
        return  # env-configured: leave the env behaviour alone
    try:
        import torch.cuda.tunable as tunable
        tunable.tuning_enable(False)
    except Exception as e:  # pragma: no cover
        log.warning("TunableOp freeze failed: %s", e)
