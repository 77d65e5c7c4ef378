"""Probe: does RCCL accept 2 ranks on one GPU? (simulated-collective mode)"""
import os, sys
import torch
import torch.multiprocessing as mp

def worker(rank, q):
    try:
        os.environ["MASTER_ADDR"] = "127.0.0.1"
        torch.cuda.set_device(0)
        torch.distributed.init_process_group(
            "nccl", init_method="tcp://127.0.0.1:29961", rank=rank,
            world_size=2)
        t = torch.ones(8, device="cuda") * (rank + 1)
        torch.distributed.all_reduce(t)
        torch.cuda.synchronize()
        q.put((rank, t[0].item()))
        torch.distributed.destroy_process_group()
    except Exception as e:
        q.put((rank, f"ERR {e!r}"))

if __name__ == "__main__":
    ctx = mp.get_context("spawn")
    q = ctx.Queue()
    ps = [ctx.Process(target=worker, args=(r, q)) for r in range(2)]
    [p.start() for p in ps]
    res = {}
    for _ in range(2):
        r, v = q.get(timeout=120)
        res[r] = v
    [p.join(timeout=30) for p in ps]
    print("RCCL 2-ranks-1-GPU:", res)
