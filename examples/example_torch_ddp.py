"""torch.distributed over the glooamd backend: DDP on CPU or MI355X.

Run:  torchrun --standalone --nproc-per-node 2 examples/example_torch_ddp.py
"""
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch
import torch.distributed as dist
import torch.nn as nn

import gloo_amd.pg  # noqa: F401  (registers the "glooamd" backend)


def main():
    dist.init_process_group("glooamd")
    rank = dist.get_rank()
    device = "cuda" if torch.cuda.is_available() else "cpu"
    if device == "cuda":
        torch.cuda.set_device(rank % torch.cuda.device_count())

    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(32, 64), nn.ReLU(), nn.Linear(64, 8))
    model = model.to(device)
    ddp = nn.parallel.DistributedDataParallel(model)
    opt = torch.optim.SGD(ddp.parameters(), lr=0.05)

    torch.manual_seed(1234 + rank)
    for step in range(5):
        x = torch.randn(16, 32, device=device)
        y = torch.randn(16, 8, device=device)
        loss = nn.functional.mse_loss(ddp(x), y)
        opt.zero_grad()
        loss.backward()  # gradients allreduced through gloo_amd
        opt.step()
        if rank == 0:
            print(f"step {step} loss {loss.item():.4f}")

    dist.destroy_process_group()


if __name__ == "__main__":
    main()
