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

    # object collectives (pickle over the tensor collectives)
    gathered = [None] * size
    dist.all_gather_object(gathered, {"rank": rank, "v": [rank] * 3})
    assert [g["rank"] for g in gathered] == list(range(size))
    olist = [{"x": 1}, "hello"] if rank == 1 else [None, None]
    dist.broadcast_object_list(olist, src=1)
    assert olist[1] == "hello" and olist[0] == {"x": 1}

    # point-to-point, blocking and async, with tags
    if rank == 0:
        dist.send(torch.arange(10.0), dst=1, tag=5)
        got = torch.zeros(10)
        dist.recv(got, src=1, tag=6)
        assert torch.allclose(got, torch.arange(10.0) + 1)
        req = dist.isend(torch.full((4,), 7.0), dst=1, tag=9)
        req.wait()
    elif rank == 1:
        got = torch.zeros(10)
        dist.recv(got, src=0, tag=5)
        dist.send(got + 1, dst=0, tag=6)
        g2 = torch.zeros(4)
        req = dist.irecv(g2, src=0, tag=9)
        req.wait()
        assert torch.allclose(g2, torch.full((4,), 7.0))

    # object gather/scatter through gather()/scatter()
    got = [None] * size if rank == 0 else None
    dist.gather_object({"r": rank}, got, dst=0)
    if rank == 0:
        assert [g["r"] for g in got] == list(range(size))
    out_obj = [None]
    scatter_src = [("obj", i) for i in range(size)] if rank == 0 else None
    dist.scatter_object_list(out_obj, scatter_src, src=0)
    assert out_obj[0] == ("obj", rank)

    # monitored barrier (success path). The module-level wrapper
    # hard-codes the "gloo" backend name, so call the ProcessGroup
    # method our backend implements.
    dist.distributed_c10d._get_default_group().monitored_barrier()

    # batch_isend_irecv (pipeline-parallel shape): both ranks post their
    # sends before any recv; large payloads force the rendezvous, so this
    # deadlocks unless isend/irecv return genuinely pending Works.
    big = 200_000
    send_t = torch.full((big,), float(rank))
    recv_t = torch.zeros(big)
    ops = [dist.P2POp(dist.isend, send_t, (rank + 1) % size),
           dist.P2POp(dist.irecv, recv_t, (rank - 1) % size)]
    for req in dist.batch_isend_irecv(ops):
        req.wait()
    assert torch.allclose(recv_t, torch.full((big,), float((rank - 1) % size)))

    # fused-tensor collectives the dispatcher lowers to *_base
    out = torch.zeros(size * 8)
    dist.all_gather_into_tensor(out, torch.full((8,), float(rank)))
    for r in range(size):
        assert torch.allclose(out[r * 8:(r + 1) * 8], torch.full((8,), float(r)))
    rs_out = torch.zeros(8)
    dist.reduce_scatter_tensor(rs_out, torch.arange(float(size * 8)))
    assert torch.allclose(
        rs_out, size * (torch.arange(8.0) + 8 * rank))
    a2a_out = torch.zeros(size * 4)
    dist.all_to_all_single(a2a_out, torch.arange(float(size * 4)) + rank * 100)
    for s in range(size):
        assert torch.allclose(
            a2a_out[s * 4:(s + 1) * 4],
            torch.arange(4.0) + rank * 4 + s * 100)
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
