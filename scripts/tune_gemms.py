#!/usr/bin/env python3
"""Offline hipBLASLt GEMM tuning for the training step's shapes.

NOTE (round 2): TunableOp measured as a net loss on this stack and is
now opt-in (CHINESENER_TUNABLE=1, see bench.py header and
profiles/gemm_nt_r02.md). This script remains for re-evaluating it on
future ROCm stacks.

Runs the bench training loop with TunableOp tuning enabled, then writes
the winners to gpurun_out/tunableop_gfx950.csv (merge back + commit to
profiles/). Also times 10 post-tuning steps so the expected gain is
recorded in the same log."""
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
OUT = os.path.join(REPO, "gpurun_out")
os.makedirs(OUT, exist_ok=True)

env = dict(os.environ)
env.update({
    "PYTORCH_TUNABLEOP_ENABLED": "1",
    "PYTORCH_TUNABLEOP_TUNING": "1",
    "PYTORCH_TUNABLEOP_FILENAME": os.path.join(OUT, "tunableop_gfx950_.csv"),
    "PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS": "300",
    "PYTORCH_TUNABLEOP_MAX_TUNING_ITERATIONS": "1000",
})
# the tuning pass: every fwd/bwd/optimizer GEMM shape appears in warmup
rc = subprocess.call([sys.executable, os.path.join(REPO, "bench.py"),
                      "--steps", "10", "--warmup", "5"], env=env)
print("tuning pass rc:", rc)
csv = os.path.join(OUT, "tunableop_gfx950_0.csv")
print("csv written:", os.path.exists(csv), csv)
