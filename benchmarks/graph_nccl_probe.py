"""Probe: does RCCL tolerate hipGraph capture of collectives? (world 1)

Informational for enabling graph capture at world>1 in a later round. Run:
    torchrun --standalone --local-addr 127.0.0.1 --nproc-per-node 1 \
        benchmarks/graph_nccl_probe.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import torch.distributed as dist


def main():
    dist.init_process_group("nccl")
    torch.cuda.set_device(0)
    x = torch.ones(1 << 20, device="cuda")
    # warmup collective outside capture
    dist.all_reduce(x)
    torch.cuda.synchronize()
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        dist.all_reduce(x)
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    x.fill_(1.0)
    try:
        with torch.cuda.graph(g):
            dist.all_reduce(x)
            x.mul_(2.0)
        for _ in range(3):
            g.replay()
        torch.cuda.synchronize()
        print("graph+RCCL capture OK; x[0] =", x[0].item())  # expect 1*2 repeated? value check below
    except Exception as e:
        print("graph+RCCL capture FAILED:", repr(e))
    dist.destroy_process_group()


if __name__ == "__main__":
    main()
