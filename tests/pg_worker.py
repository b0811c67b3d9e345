"""Subprocess worker: init_process_group("glooamd") + a DDP train step."""
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch  # noqa: E402
import torch.distributed as dist  # noqa: E402
import torch.nn as nn  # noqa: E402

import gloo_amd.pg  # noqa: F401,E402  (registers the backend)


def main():
    rank, size, port = int(sys.argv[1]), int(sys.argv[2]), sys.argv[3]
    dist.init_process_group(
        "glooamd",
        init_method=f"tcp://127.0.0.1:{port}",
        rank=rank,
        world_size=size,
    )

    # functional collectives through the dispatcher
    t = torch.full((64,), float(rank + 1))
    dist.all_reduce(t)
    assert torch.allclose(t, torch.full((64,), float(sum(range(1, size + 1)))))

    t = torch.full((8,), float(rank))
    dist.broadcast(t, src=1)
    assert torch.allclose(t, torch.full((8,), 1.0))
    dist.barrier()

    # DDP: gradient bucket allreduce over the backend
    torch.manual_seed(0)
    model = nn.Sequential(nn.Linear(16, 32), nn.ReLU(), nn.Linear(32, 4))
    ddp = nn.parallel.DistributedDataParallel(model)
    opt = torch.optim.SGD(ddp.parameters(), lr=0.05)
    torch.manual_seed(100 + rank)  # different data per rank
    for _ in range(3):
        x = torch.randn(8, 16)
        y = torch.randn(8, 4)
        loss = nn.functional.mse_loss(ddp(x), y)
        opt.zero_grad()
        loss.backward()
        opt.step()
    # parameters must be identical across ranks after DDP steps
    flat = torch.cat([p.detach().reshape(-1) for p in ddp.parameters()])
    mine = flat.clone()
    dist.broadcast(flat, src=0)
    assert torch.allclose(mine, flat, atol=1e-6), "DDP params diverged"

    dist.destroy_process_group()
    print("DDP-OK rank", rank)


if __name__ == "__main__":
    main()
