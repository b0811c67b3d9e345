import faulthandler, sys, threading
sys.path.insert(0, __file__.rsplit('/', 2)[0])
faulthandler.dump_traceback_later(90, exit=True)
import torch
import gloo_amd as ga
from gloo_amd.pg import ProcessGroupGlooAmd

def two_rank(fn, name):
    store = ga.HashStore()
    errors = []
    def worker(rank):
        try:
            dev = ga.create_tcp_device()
            ctx = ga.Context(rank, 2)
            ctx.connect_full_mesh(store, dev)
            ctx.set_timeout(45000)
            torch.cuda.set_device(0)
            fn(ctx, rank)
        except Exception:
            import traceback
            errors.append(traceback.format_exc())
    ths = [threading.Thread(target=worker, args=(r,)) for r in range(2)]
    [t.start() for t in ths]
    [t.join(120) for t in ths]
    assert not errors, errors[0]
    assert not any(t.is_alive() for t in ths), f"{name}: threads leaked"
    print(name, "OK", flush=True)

def ag(ctx, rank):
    n = 1_000_000
    inp = torch.full((n,), float(rank + 1), device="cuda")
    out = torch.zeros(2 * n, device="cuda")
    algo = ga._C.HipAllgatherRing(ctx, 0)
    for _ in range(2):
        algo.run(inp.data_ptr(), out.data_ptr(), n, 4)
        torch.cuda.synchronize()

def rs(ctx, rank):
    n = 500_000
    inp = torch.rand(2 * n).cuda()
    out = torch.zeros(n, device="cuda")
    algo = ga._C.HipReduceScatterRing(ctx, 0)
    for _ in range(2):
        algo.run(inp.data_ptr(), out.data_ptr(), n, ga.DType.f32, ga.ReduceOp.sum)
        torch.cuda.synchronize()

def a2a(ctx, rank):
    n = 300_000
    inp = torch.zeros(2 * n, device="cuda")
    out = torch.zeros(2 * n, device="cuda")
    algo = ga._C.HipAlltoall(ctx, 0)
    for _ in range(2):
        algo.run(inp.data_ptr(), out.data_ptr(), n, 4)
        torch.cuda.synchronize()

two_rank(ag, "AG")
two_rank(rs, "RS")
two_rank(a2a, "A2A")

# PG test
store = ga.HashStore()
errors = []
def pgworker(rank):
    try:
        torch.cuda.set_device(0)
        pg = ProcessGroupGlooAmd(store, rank, 2)
        print("pg ctor done", rank, flush=True)
        t = torch.full((700_000,), float(rank + 1), device="cuda")
        pg.allreduce([t]).wait(); torch.cuda.synchronize()
        print("pg allreduce done", rank, flush=True)
        inp = torch.full((100_000,), float(rank), device="cuda")
        out = torch.zeros(200_000, device="cuda")
        pg._allgather_base(out, inp).wait(); torch.cuda.synchronize()
        print("pg ag done", rank, flush=True)
        rs_in = torch.arange(200_000, dtype=torch.float32, device="cuda")
        rs_out = torch.zeros(100_000, device="cuda")
        pg._reduce_scatter_base(rs_out, rs_in).wait(); torch.cuda.synchronize()
        print("pg rs done", rank, flush=True)
        a2a_in = torch.zeros(100_000, device="cuda")
        a2a_out = torch.zeros(100_000, device="cuda")
        pg.alltoall_base(a2a_out, a2a_in, [], []).wait(); torch.cuda.synchronize()
        print("pg a2a done", rank, flush=True)
        b = torch.zeros(50_000, device="cuda")
        pg.broadcast([b]).wait(); torch.cuda.synchronize()
        print("pg bcast done", rank, flush=True)
    except Exception:
        import traceback
        errors.append(traceback.format_exc())
ths = [threading.Thread(target=pgworker, args=(r,)) for r in range(2)]
[t.start() for t in ths]
[t.join(120) for t in ths]
assert not errors, errors[0]
print("PG OK", flush=True)
