"""Minimal 2-rank RCCL probe: init, broadcast, bf16 all_reduce, graph
capture of an all_reduce. Validates the collectives bench.py depends on
— runnable on a 1-GPU box with both ranks oversubscribing cuda:0."""
import os
import sys

import torch
import torch.distributed as dist


def main():
    # standalone world=1 mode: RCCL refuses >1 rank per GPU (verified:
    # "Duplicate GPU detected" on a 1-GPU box), so single-rank RCCL is
    # the deepest RCCL validation a 1-GPU lease allows — it still runs
    # real ncclAllReduce kernels and their hipGraph capture.
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29617")
    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", "1"))
    local = int(os.environ.get("LOCAL_RANK", rank))
    dev_idx = local % max(1, torch.cuda.device_count())
    torch.cuda.set_device(dev_idx)
    device = f"cuda:{dev_idx}"
    print(f"[probe] rank={rank}/{world} device={device}", flush=True)
    from datetime import timedelta
    dist.init_process_group("nccl", timeout=timedelta(minutes=2))
    print(f"[probe] rank={rank} init ok", flush=True)

    t = torch.full((1024,), float(rank + 1), device=device)
    dist.broadcast(t, src=0)
    assert t.mean().item() == 1.0, t.mean().item()
    print(f"[probe] rank={rank} broadcast ok", flush=True)

    b = torch.full((1 << 20,), 1.0, device=device, dtype=torch.bfloat16)
    dist.all_reduce(b)
    assert b.float().mean().item() == world
    print(f"[probe] rank={rank} bf16 all_reduce ok", flush=True)

    w = dist.all_reduce(b, async_op=True)
    w.wait()
    print(f"[probe] rank={rank} async all_reduce ok", flush=True)

    # hipGraph capture containing a collective (bench.py graph step path)
    try:
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                dist.all_reduce(b)
        torch.cuda.current_stream().wait_stream(s)
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            dist.all_reduce(b)
        for _ in range(3):
            g.replay()
        torch.cuda.synchronize()
        print(f"[probe] rank={rank} graph-captured all_reduce ok", flush=True)
    except Exception as e:
        print(f"[probe] rank={rank} graph capture FAILED: {e}", flush=True)

    dist.barrier()
    dist.destroy_process_group()
    print(f"[probe] rank={rank} DONE", flush=True)


if __name__ == "__main__":
    main()
