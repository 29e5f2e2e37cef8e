"""GPU-side attention kernel diagnosis: per-piece comparison vs reference."""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch

from chinesener_amd import ops
from chinesener_amd.ops import reference as ref


def stat(name, a, b):
    d = (a.float() - b.float()).abs()
    flat = d.flatten()
    idx = int(flat.argmax())
    print(f"{name}: max={flat.max():.4f} mean={d.mean():.5f} "
          f"argmax={torch.unravel_index(torch.tensor(idx), a.shape)} "
          f"a={a.flatten()[idx]:.4f} b={b.flatten()[idx]:.4f} "
          f"nan_a={int(a.float().isnan().sum())} nan_b={int(b.float().isnan().sum())}")


def main():
    torch.manual_seed(3)
    B, H, L, D = 2, 4, 128, 64
    q32 = torch.randn(B, H, L, D, device="cuda")
    k32 = torch.randn(B, H, L, D, device="cuda")
    v32 = torch.randn(B, H, L, D, device="cuda")
    lens = torch.tensor([L, L - 41], device="cuda")
    mask = (torch.arange(L, device="cuda")[None, :] < lens[:, None]).long()
    scale = 1.0 / 8.0

    q16, k16, v16 = (t.to(torch.bfloat16) for t in (q32, k32, v32))
    ext = ops.get_ext()
    out, lse = ext.attn_fwd(q16.contiguous(), k16.contiguous(), v16.contiguous(),
                            lens.to(torch.int32), scale)
    out_ref = ref.attention(q32, k32, v32, mask, scale)
    # bf16-input reference for tolerance calibration
    out_ref16 = ref.attention(q16.float(), k16.float(), v16.float(), mask, scale)
    mrow = mask[:, None, :, None].bool()
    stat("fwd out (vs fp32 ref, real rows)", out * mrow, out_ref * mrow)
    stat("fwd out (vs bf16-input ref)", out * mrow, out_ref16 * mrow)
    # lse reference
    s = torch.matmul(q16.float(), k16.float().transpose(-1, -2)) * scale
    s = s + (1 - mask[:, None, None, :].float()) * -1e30
    lse_ref = torch.logsumexp(s.float(), dim=-1)
    stat("lse (real rows)", lse * mask[:, None, :], lse_ref * mask[:, None, :])

    # backward pieces
    g = torch.randn(B, H, L, D, device="cuda") * mask[:, None, :, None]
    g16 = g.to(torch.bfloat16)
    dq, dk, dv = ext.attn_bwd(g16.contiguous(), q16.contiguous(),
                              k16.contiguous(), v16.contiguous(),
                              out.contiguous(), lse, lens.to(torch.int32), scale)
    qr = q16.float().requires_grad_()
    kr = k16.float().requires_grad_()
    vr = v16.float().requires_grad_()
    o2 = ref.attention(qr, kr, vr, mask, scale)
    o2.backward(g)
    stat("dv", dv, vr.grad)
    stat("dk", dk, kr.grad)
    stat("dq", dq, qr.grad)

    # per-row fwd error profile for batch 1 (masked)
    err = (out.float() - out_ref16).abs().amax(dim=-1)  # [B,H,L]
    print("row err profile b=1 h=0 rows 80..92:", err[1, 0, 80:93].tolist())
    print("len[1] =", int(lens[1]))

    # smoke NaN repro: tiny BERT forward trace
    from chinesener_amd.models import build_model
    from chinesener_amd.models.bert import BertConfig
    from chinesener_amd.data.loader import make_synthetic_batch
    torch.manual_seed(0)
    cfg = BertConfig(vocab_size=2000, hidden_size=768, num_hidden_layers=2,
                     num_attention_heads=12, intermediate_size=3072)
    params = {"vocab_size": 2000, "label_size": 10, "bert_config": cfg,
              "rnn_params": {"hidden_units_list": [128],
                             "cell_activation": "relu",
                             "keep_prob_list": [0.8]},
              "tag2idx": {}, "dropout_rate": 0.1}
    model = build_model("bert_bilstm_crf", params).to("cuda")
    batch = make_synthetic_batch(4, 64, 10, vocab_size=2000, device="cuda")

    feats = {}
    def hook(name):
        def f(mod, i, o):
            t = o if torch.is_tensor(o) else o[0]
            feats[name] = (float(t.float().abs().max()),
                           int(t.float().isnan().sum()))
        return f
    model.bert.embeddings.register_forward_hook(hook("emb"))
    for i, layer in enumerate(model.bert.layers):
        layer.register_forward_hook(hook(f"layer{i}"))
    model.bilstm.register_forward_hook(hook("bilstm"))
    model.logits.register_forward_hook(hook("logits"))
    with torch.autocast("cuda", dtype=torch.bfloat16):
        outm = model(batch)
    print("stage stats (absmax, nan count):", feats)
    print("loss:", float(outm.loss))
    torch.cuda.synchronize()


if __name__ == "__main__":
    main()
