"""DP equivalence on gloo, world_size 2 (SURVEY.md §4 implication (d)):
bucketed all-reduce grads == single-process grads on the concatenated
batch, and a 2-rank training step keeps replicas identical."""
import os
import pickle
import subprocess
import sys
import tempfile

import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

WORKER = r"""
import os, pickle, sys
sys.path.insert(0, os.environ["REPO"])
sys.path.insert(0, os.path.join(os.environ["REPO"], "tests"))
import torch
import torch.distributed as dist
from chinesener_amd.dist import BucketedDataParallel, init_process_group
from chinesener_amd.models import build_model
from conftest import make_tiny_batch, make_tiny_params

def main():
    rank, world = init_process_group("gloo")
    torch.manual_seed(123)   # same init on every rank (then broadcast anyway)
    name = os.environ["MODEL"]
    params = make_tiny_params(name)
    model = build_model(name, params)
    dp = BucketedDataParallel(model, bucket_cap_mb=0.05)
    batch = make_tiny_batch(name, batch_size=2, seed=rank)
    dp.zero_grad()
    out = model(batch)
    out.loss.backward()
    dp.finalize_backward()
    grads = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}
    if rank == 0:
        with open(os.environ["OUT"], "wb") as f:
            pickle.dump({n: g.numpy() for n, g in grads.items()}, f)
    # all replicas must hold identical averaged grads
    for n, g in sorted(grads.items()):
        gl = [torch.empty_like(g) for _ in range(world)]
        dist.all_gather(gl, g)
        assert torch.allclose(gl[0], gl[1], atol=1e-6), f"grad mismatch {n}"
    dist.destroy_process_group()

main()
"""


def _run_workers(model_name: str, out_path: str, nproc: int = 2):
    with tempfile.NamedTemporaryFile("w", suffix=".py", delete=False) as f:
        f.write(WORKER)
        script = f.name
    procs = []
    for rank in range(nproc):
        env = dict(os.environ, REPO=REPO, MODEL=model_name, OUT=out_path,
                   RANK=str(rank), WORLD_SIZE=str(nproc),
                   MASTER_ADDR="127.0.0.1", MASTER_PORT="29511",
                   LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, script],
                                      env=env, stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    for p in procs:
        out, err = p.communicate(timeout=600)
        assert p.returncode == 0, err[-3000:]
    os.unlink(script)


def test_dp_grads_match_single_process(tmp_path):
    out_path = str(tmp_path / "grads.pkl")
    _run_workers("bilstm_crf", out_path)
    with open(out_path, "rb") as f:
        dp_grads = pickle.load(f)

    # single-process: average grads of the two per-rank batches
    sys.path.insert(0, os.path.join(REPO, "tests"))
    from conftest import make_tiny_batch, make_tiny_params
    from chinesener_amd.models import build_model
    torch.manual_seed(123)
    params = make_tiny_params("bilstm_crf")
    model = build_model("bilstm_crf", params)
    total = None
    for seed in (0, 1):
        model.zero_grad()
        out = model(make_tiny_batch("bilstm_crf", batch_size=2, seed=seed))
        out.loss.backward()
        g = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}
        total = g if total is None else {n: total[n] + g[n] for n in g}
    for n, g in total.items():
        ref = g / 2
        got = torch.tensor(dp_grads[n])
        torch.testing.assert_close(got, ref, atol=1e-5, rtol=1e-4), n


def test_dp_mtl_shared_bert(tmp_path):
    """Shared-BERT multi-task grads flow through the DP engine too."""
    out_path = str(tmp_path / "grads_mtl.pkl")
    _run_workers("bert_bilstm_crf_mtl", out_path)
    with open(out_path, "rb") as f:
        dp_grads = pickle.load(f)
    assert any(n.startswith("bert.") for n in dp_grads)
