"""Subprocess worker for the cross-process HIP IPC test."""
import sys

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch  # noqa: E402

import gloo_amd as ga  # noqa: E402


def main():
    rank, size, storedir = int(sys.argv[1]), int(sys.argv[2]), sys.argv[3]
    store = ga.FileStore(storedir)
    dev = ga.create_tcp_device()
    ctx = ga.Context(rank, size)
    ctx.connect_full_mesh(store, dev)
    ctx.set_timeout(120000)
    torch.cuda.set_device(0)

    n = 1_500_000
    g = torch.Generator("cpu").manual_seed(rank)
    x = torch.rand(n, generator=g, dtype=torch.float32).cuda()
    expect = sum(
        torch.rand(n, generator=torch.Generator("cpu").manual_seed(r),
                   dtype=torch.float32)
        for r in range(size)
    )
    algo = ga._C.HipAllreduceRing(ctx, 0, True, 0)
    algo.run(x.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
    torch.cuda.synchronize()
    got = x.cpu()
    assert torch.allclose(got, expect, rtol=1e-5, atol=1e-4), (
        got - expect).abs().max()
    # second run exercises cross-run flag sequencing over IPC
    algo.run(x.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
    torch.cuda.synchronize()

    # direct engine with >8MiB blocks (chunk pipeline), 3 runs
    nd = 5_000_001 * 2 // size * size  # keep per-rank blocks > 8 MiB
    gd = torch.Generator("cpu").manual_seed(100 + rank)
    y = torch.rand(nd, generator=gd, dtype=torch.float32).cuda()
    refd = sum(
        torch.rand(nd, generator=torch.Generator("cpu").manual_seed(100 + r),
                   dtype=torch.float32)
        for r in range(size))
    direct = ga._C.HipAllreduceDirect(ctx, 0)
    for _ in range(3):
        z = y.clone()
        direct.run(z.data_ptr(), nd, ga.DType.f32, ga.ReduceOp.sum)
        torch.cuda.synchronize()
        assert torch.allclose(z.cpu(), refd, atol=1e-4), (
            (z.cpu() - refd).abs().max())
    print("IPC-OK rank", rank)


if __name__ == "__main__":
    main()
