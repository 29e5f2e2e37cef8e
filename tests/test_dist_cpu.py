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


def test_dp_unused_param_and_bucket_boundaries():
    """Single-rank gloo: a parameter with no grad this step must not
    wedge finalize_backward (late reduce path), and multi-bucket layout
    must write grads back exactly."""
    import torch.distributed as dist
    from chinesener_amd.dist import BucketedDataParallel, init_process_group
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29533")
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    created = not dist.is_initialized()
    if created:
        init_process_group("gloo")
    try:
        class M(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.a = torch.nn.Linear(64, 64)   # 4160 params
                self.b = torch.nn.Linear(64, 64)
                self.unused = torch.nn.Linear(8, 8)

            def forward(self, x):
                return self.b(self.a(x)).sum()

        torch.manual_seed(0)
        m = M()
        # tiny cap forces several buckets
        dp = BucketedDataParallel(m, bucket_cap_mb=0.01)
        assert len(dp.buckets) >= 2
        x = torch.randn(4, 64)
        dp.zero_grad()
        m(x).backward()
        dp.finalize_backward()          # must not hang on unused bucket
        assert m.unused.weight.grad is None
        # grads equal a plain backward (world=1 => average is identity)
        m2 = M()
        m2.load_state_dict(m.state_dict())
        m2(x).backward()
        torch.testing.assert_close(m.a.weight.grad, m2.a.weight.grad,
                                   atol=1e-6, rtol=1e-5)
        torch.testing.assert_close(m.b.bias.grad, m2.b.bias.grad,
                                   atol=1e-6, rtol=1e-5)
        # second step reuses buckets cleanly
        dp.zero_grad()
        m(x * 2).backward()
        dp.finalize_backward()
    finally:
        if created:
            dist.destroy_process_group()


def test_nerdataset_rank_sharding(tmp_path):
    """DP ranks see disjoint, collectively-exhaustive shards per epoch."""
    from chinesener_amd.data.loader import NerDataset
    pipes = [NerDataset(str(tmp_path), "people_daily", 4, 1, "bilstm_crf",
                        rank=r, world_size=2) for r in range(2)]
    seen = [set(), set()]
    for r, pipe in enumerate(pipes):
        for batch in pipe.iter_batches("train", shuffle=True, drop_last=False):
            for row in batch["token_ids"]:
                seen[r].add(tuple(row.tolist()))
    # shards must not overlap (synthetic rows are distinct w.h.p.)
    assert not (seen[0] & seen[1])
    assert len(seen[0]) + len(seen[1]) >= 100


def test_reduce_in_graph_matches_finalize_backward():
    """reduce_in_graph (the captured-step DP path, no hooks) must
    produce the same averaged grads as the hook + finalize path."""
    import torch.distributed as dist
    from chinesener_amd.dist import BucketedDataParallel, init_process_group
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29534")
    os.environ["RANK"] = "0"
    os.environ["WORLD_SIZE"] = "1"
    created = not dist.is_initialized()
    if created:
        init_process_group("gloo")
    try:
        class M(torch.nn.Module):
            def __init__(self):
                super().__init__()
                self.a = torch.nn.Linear(32, 32)
                self.b = torch.nn.Linear(32, 32)
                self.unused = torch.nn.Linear(4, 4)

            def forward(self, x):
                return self.b(self.a(x)).sum()

        torch.manual_seed(1)
        m = M()
        dp = BucketedDataParallel(m, bucket_cap_mb=0.005)
        x = torch.randn(4, 32)

        # path 1: hooks + finalize
        dp.zero_grad()
        m(x).backward()
        dp.finalize_backward()
        ref = {n: p.grad.clone() for n, p in m.named_parameters()
               if p.grad is not None}

        # path 2: autograd.grad into p.grad + reduce_in_graph
        params = [p for p in m.parameters() if p.requires_grad]
        loss = m(x)
        grads = torch.autograd.grad(loss, params, allow_unused=True)
        with torch.no_grad():
            for p, g in zip(params, grads):
                if p.grad is None and g is None:
                    continue
                if p.grad is None:
                    p.grad = torch.zeros_like(p)
                if g is None:
                    p.grad.zero_()
                else:
                    p.grad.copy_(g)
        dp.reduce_in_graph()
        for n, p in m.named_parameters():
            if n in ref:
                torch.testing.assert_close(p.grad, ref[n],
                                           atol=1e-6, rtol=1e-5)
    finally:
        if created:
            dist.destroy_process_group()


WORKER_GRAPHPATH = """
import os, sys, pickle
sys.path.insert(0, os.environ["REPO"])
sys.path.insert(0, os.path.join(os.environ["REPO"], "tests"))
import torch
import torch.distributed as dist
from chinesener_amd.dist import BucketedDataParallel, init_process_group
from chinesener_amd.models import build_model
from conftest import make_tiny_batch, make_tiny_params

def main():
    rank, world = init_process_group("gloo")
    torch.manual_seed(123)
    name = os.environ["MODEL"]
    params = make_tiny_params(name)
    model = build_model(name, params)
    dp = BucketedDataParallel(model, bucket_cap_mb=0.05)
    batch = make_tiny_batch(name, batch_size=2, seed=rank)
    params_list = [p for p in model.parameters() if p.requires_grad]

    # hook path (eager DP)
    dp.zero_grad()
    model(batch).loss.backward()
    dp.finalize_backward()
    ref = {n: p.grad.clone() for n, p in model.named_parameters()
           if p.grad is not None}

    # graphed-body path: autograd.grad into p.grad + reduce_in_graph
    # (what GraphedTrainStep records; world-2 equivalence)
    loss = model(batch).loss
    grads = torch.autograd.grad(loss, params_list, allow_unused=True)
    with torch.no_grad():
        for p, g in zip(params_list, grads):
            if p.grad is None and g is None:
                continue
            if p.grad is None:
                p.grad = torch.zeros_like(p)
            if g is None:
                p.grad.zero_()
            else:
                p.grad.copy_(g)
    dp.reduce_in_graph()
    for n, p in model.named_parameters():
        if n in ref:
            assert torch.allclose(p.grad, ref[n], atol=1e-6), n
    dist.destroy_process_group()

main()
"""


def test_reduce_in_graph_world2_matches_hooks(tmp_path):
    """World-size-2 gloo: the captured-step DP reduction
    (autograd.grad + reduce_in_graph) produces the same averaged grads
    as the hook + finalize path on every rank."""
    with tempfile.NamedTemporaryFile("w", suffix=".py", delete=False) as f:
        f.write(WORKER_GRAPHPATH)
        script = f.name
    procs = []
    for rank in range(2):
        env = dict(os.environ, REPO=REPO, MODEL="bilstm_crf",
                   RANK=str(rank), WORLD_SIZE="2",
                   MASTER_ADDR="127.0.0.1", MASTER_PORT="29512",
                   LOCAL_RANK=str(rank))
        procs.append(subprocess.Popen([sys.executable, script],
                                      env=env, stdout=subprocess.PIPE,
                                      stderr=subprocess.PIPE, text=True))
    for p in procs:
        out, err = p.communicate(timeout=600)
        assert p.returncode == 0, err[-3000:]
    os.unlink(script)
