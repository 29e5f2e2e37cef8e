"""Isolate the e2e divergence: fused clip vs torch clip, fused bilstm vs
single-direction kernels, 25-step loss curves per variant."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from chinesener_amd import ops
from chinesener_amd.data.loader import make_synthetic_batch
from chinesener_amd.models import build_model
from chinesener_amd.models.bert import BertConfig
from chinesener_amd.train.optimizers import (AdamWeightDecay, LrSchedule,
                                             build_param_groups,
                                             clip_gradients)

assert torch.cuda.is_available() and ops.ext_available()


def params_small():
    cfg = BertConfig(vocab_size=2000, hidden_size=768, num_hidden_layers=2,
                     num_attention_heads=12, intermediate_size=3072)
    return {"vocab_size": 2000, "label_size": 10, "bert_config": cfg,
            "rnn_params": {"hidden_units_list": [128],
                           "cell_activation": "relu", "keep_prob_list": [0.8]},
            "tag2idx": {}, "dropout_rate": 0.1}


def one_backward(model, batch):
    for p in model.parameters():
        p.grad = None
    with torch.autocast("cuda", dtype=torch.bfloat16):
        out = model(batch)
    out.loss.backward()
    return float(out.loss)


def test_clip_equiv():
    torch.manual_seed(0)
    model = build_model("bert_bilstm_crf", params_small()).to("cuda")
    batch = make_synthetic_batch(8, 128, 10, vocab_size=2000, device="cuda")
    one_backward(model, batch)
    saved = {n: p.grad.clone() for n, p in model.named_parameters()
             if p.grad is not None}
    # torch reference clip on copies
    gn = torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
    ref = {n: p.grad.clone() for n, p in model.named_parameters()
           if p.grad is not None}
    # restore + fused clip
    for n, p in model.named_parameters():
        if p.grad is not None:
            p.grad.copy_(saved[n])
    ss = clip_gradients(model, "bert")
    print("torch norm", float(gn), "fused sqrt(sumsq)", float(ss.sqrt()))
    worst = 0.0
    worst_n = ""
    for n, p in model.named_parameters():
        if p.grad is None:
            continue
        d = (p.grad - ref[n]).abs().max().item()
        scale = ref[n].abs().max().item() + 1e-9
        if d / scale > worst:
            worst, worst_n = d / scale, n
    print("max rel grad diff after clip:", worst, worst_n)


def test_bilstm_paths():
    torch.manual_seed(1)
    from chinesener_amd.ops import functional as fn
    B, L, E, h = 8, 128, 768, 128
    x = torch.randn(B, L, E, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    lens = torch.full((B,), L, dtype=torch.int64, device="cuda")
    ws = [torch.randn(E, 4 * h, device="cuda") * 0.02 for _ in range(2)]
    whs = [torch.randn(h, 4 * h, device="cuda") * 0.02 for _ in range(2)]
    bs = [torch.zeros(4 * h, device="cuda") for _ in range(2)]
    for t in ws + whs + bs:
        t.requires_grad_(True)

    out_fused = fn.bilstm(x, ws[0], whs[0], bs[0], ws[1], whs[1], bs[1], lens,
                          "relu")
    g = torch.randn_like(out_fused)
    out_fused.backward(g)
    fused_grads = [t.grad.clone() for t in [x] + ws + whs + bs]
    for t in [x] + ws + whs + bs:
        t.grad = None
    fw = fn._lstm_dir(x, ws[0], whs[0], bs[0], lens, False, "relu")
    bw = fn._lstm_dir(x, ws[1], whs[1], bs[1], lens, True, "relu")
    out_sep = torch.cat([fw, bw], -1)
    print("fwd max diff:", (out_fused - out_sep).abs().max().item())
    out_sep.backward(g)
    sep_grads = [t.grad.clone() for t in [x] + ws + whs + bs]
    for a, b, name in zip(fused_grads, sep_grads,
                          ["dx", "dwi_f", "dwi_b", "dwh_f", "dwh_b", "db_f",
                           "db_b"]):
        print(name, "max diff:", (a - b).abs().max().item(),
              "ref absmax:", b.abs().max().item())


def run_steps(clip_mode, seed=0, n=25):
    torch.manual_seed(seed)
    model = build_model("bert_bilstm_crf", params_small()).to("cuda")
    opt = AdamWeightDecay(build_param_groups(model, 1e-4, 0.01), lr=1e-4)
    sched = LrSchedule("bert", 1e-4, num_train_steps=200, warmup_ratio=0.2)
    batch = make_synthetic_batch(8, 128, 10, vocab_size=2000, device="cuda")
    losses = []
    for step in range(1, n + 1):
        opt.zero_grad(set_to_none=True)
        loss = one_backward(model, batch)
        if clip_mode == "torch":
            torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
        else:
            clip_gradients(model, "bert")
        sched.apply(opt, step)
        opt.step()
        losses.append(loss)
    return losses


def run_instrumented(n=14):
    torch.manual_seed(0)
    model = build_model("bert_bilstm_crf", params_small()).to("cuda")
    opt = AdamWeightDecay(build_param_groups(model, 1e-4, 0.01), lr=1e-4)
    sched = LrSchedule("bert", 1e-4, num_train_steps=200, warmup_ratio=0.2)
    batch = make_synthetic_batch(8, 128, 10, vocab_size=2000, device="cuda")
    for step in range(1, n + 1):
        opt.zero_grad(set_to_none=True)
        loss = one_backward(model, batch)
        grads = [p.grad for p in model.parameters() if p.grad is not None]
        pre = torch.sqrt(sum(g.float().pow(2).sum() for g in grads))
        nfin = sum((~torch.isfinite(g)).sum().item() for g in grads)
        norms = sorted(((float(p.grad.float().norm()), n)
                        for n, p in model.named_parameters()
                        if p.grad is not None), reverse=True)
        ss = clip_gradients(model, "bert")
        post = torch.sqrt(sum(g.float().pow(2).sum() for g in grads))
        print(f"step {step} loss {loss:.1f} pre-norm {float(pre):.3f} "
              f"fused-norm {float(ss.sqrt()):.3f} post-norm {float(post):.3f} "
              f"nonfinite {nfin}")
        print("   top grads:", [(n, round(v, 1)) for v, n in norms[:5]])
        sched.apply(opt, step)
        opt.step()
        # weight sanity
        wmax = max(p.abs().max().item() for p in model.parameters())
        if step % 1 == 0:
            print("   max |w|:", wmax)


def run_repeat(n=30):
    """Same fwd+bwd repeated with identical RNG: any variation = race."""
    torch.manual_seed(0)
    model = build_model("bert_bilstm_crf", params_small()).to("cuda")
    batch = make_synthetic_batch(8, 128, 10, vocab_size=2000, device="cuda")
    results = []
    for i in range(n):
        torch.manual_seed(42)          # identical dropout masks every pass
        for p in model.parameters():
            p.grad = None
        with torch.autocast("cuda", dtype=torch.bfloat16):
            out = model(batch)
        out.loss.backward()
        gn = torch.sqrt(sum(p.grad.float().pow(2).sum()
                            for p in model.parameters()
                            if p.grad is not None))
        results.append((float(out.loss), float(gn)))
    losses = {r[0] for r in results}
    norms = {round(r[1], 2) for r in results}
    print("distinct losses:", sorted(losses))
    print("distinct grad norms:", sorted(norms))


def run_repeat_lstm(n=50):
    """BiLSTM kernel alone, fixed inputs, repeated: fwd hash + grad hash."""
    torch.manual_seed(2)
    from chinesener_amd.ops import functional as fn
    B, L, E, h = 8, 128, 768, 128
    x0 = torch.randn(B, L, E, device="cuda", dtype=torch.bfloat16)
    lens = torch.full((B,), L, dtype=torch.int64, device="cuda")
    ws = [torch.randn(E, 4 * h, device="cuda") * 0.02 for _ in range(2)]
    whs = [torch.randn(h, 4 * h, device="cuda") * 0.02 for _ in range(2)]
    bs = [torch.zeros(4 * h, device="cuda") for _ in range(2)]
    for t in ws + whs + bs:
        t.requires_grad_(True)
    g = torch.randn(B, L, 2 * h, device="cuda", dtype=torch.bfloat16)
    outs, grads = set(), set()
    for i in range(n):
        x = x0.clone().requires_grad_(True)
        out = fn.bilstm(x, ws[0], whs[0], bs[0], ws[1], whs[1], bs[1], lens,
                        "relu")
        out.backward(g)
        outs.add(float(out.float().sum()))
        grads.add(float(x.grad.float().abs().sum()))
        for t in ws + whs + bs:
            t.grad = None
    print("distinct lstm fwd sums:", len(outs), sorted(outs)[:4])
    print("distinct lstm dx sums:", len(grads), sorted(grads)[:4])


def test_adam_fused_vs_foreach():
    """One fused multi-tensor step vs the eager foreach math on identical
    state: any element diff beyond fp32 noise = kernel bug."""
    torch.manual_seed(5)
    from chinesener_amd.train import optimizers as O
    model = build_model("bert_bilstm_crf", params_small()).to("cuda")
    batch = make_synthetic_batch(8, 128, 10, vocab_size=2000, device="cuda")
    one_backward(model, batch)
    import copy
    groups = O.build_param_groups(model, 1e-3, 0.01)
    opt = O.AdamWeightDecay(groups, lr=1e-3)
    snap = {n: p.detach().clone() for n, p in model.named_parameters()}
    gsnap = {n: p.grad.detach().clone() for n, p in model.named_parameters()}
    opt.step()
    fused = {n: p.detach().clone() for n, p in model.named_parameters()}
    # restore and run eager
    with torch.no_grad():
        for n, p in model.named_parameters():
            p.copy_(snap[n])
            p.grad.copy_(gsnap[n])
    opt2 = O.AdamWeightDecay(O.build_param_groups(model, 1e-3, 0.01), lr=1e-3)
    import chinesener_amd.ops as ops_mod
    real = ops_mod.ext_available
    ops_mod.ext_available = lambda: False
    try:
        opt2.step()
    finally:
        ops_mod.ext_available = real
    worst, wn = 0.0, ""
    for n, p in model.named_parameters():
        d = (p.detach() - fused[n]).abs().max().item()
        if d > worst:
            worst, wn = d, n
    print("adam fused-vs-foreach max abs diff:", worst, wn)


def run_act_trace(n=10):
    """Per-step max-activation trace: which module's forward explodes."""
    torch.manual_seed(0)
    model = build_model("bert_bilstm_crf", params_small()).to("cuda")
    from chinesener_amd.train.optimizers import (AdamWeightDecay, LrSchedule,
                                                 build_param_groups)
    opt = AdamWeightDecay(build_param_groups(model, 1e-4, 0.01), lr=1e-4)
    sched = LrSchedule("bert", 1e-4, num_train_steps=200, warmup_ratio=0.2)
    batch = make_synthetic_batch(8, 128, 10, vocab_size=2000, device="cuda")
    acts = {}
    def mk(name):
        def hook(mod, i, o):
            if isinstance(o, torch.Tensor):
                acts[name] = float(o.detach().float().abs().max())
        return hook
    model.bert.embeddings.register_forward_hook(mk("emb"))
    model.bert.layers[0].register_forward_hook(mk("bert0"))
    model.bert.layers[1].register_forward_hook(mk("bert1"))
    model.bilstm.register_forward_hook(mk("bilstm"))
    model.logits.register_forward_hook(mk("logits"))
    for step in range(1, n + 1):
        opt.zero_grad(set_to_none=True)
        loss = one_backward(model, batch)
        gn = torch.sqrt(sum(p.grad.float().pow(2).sum()
                            for p in model.parameters() if p.grad is not None))
        print(f"step {step} loss {loss:.1f} gn {float(gn):.2e} acts {acts}")
        torch.nn.utils.clip_grad_norm_(model.parameters(), 1.0)
        sched.apply(opt, step)
        opt.step()


if __name__ == "__main__":
    test_adam_fused_vs_foreach()
    run_act_trace()
