"""torch.distributed ProcessGroup backend backed by gloo_amd.

Register/usage:
    import gloo_amd.pg  # registers backend "glooamd" (cpu + cuda)
    torch.distributed.init_process_group("glooamd", ...)

CPU tensors run the v2 TCP collectives; CUDA tensors route allreduce /
broadcast through the xGMI/IPC device algorithms (csrc/hip/) and the
remaining collectives through pinned-host staging (device-native
versions are per-op upgrades, not API changes).

This is the MI355X counterpart of the reference's role as the `gloo`
backend of torch.distributed.

Note: `torch.distributed.monitored_barrier()` hard-codes the "gloo"
backend name; call the ProcessGroup method instead:
`dist.distributed_c10d._get_default_group().monitored_barrier()`.
"""
import threading
import time
from concurrent.futures import ThreadPoolExecutor
from datetime import timedelta

import torch
import torch.distributed as dist
from torch._C._distributed_c10d import _create_work_from_future
from torch.futures import Future

import gloo_amd as ga

def _map_op(op):
    """c10d passes ReduceOp instances whose hash differs from the enum
    members; compare by equality."""
    if op == dist.ReduceOp.SUM or op == dist.ReduceOp.AVG:
        return ga.ReduceOp.sum  # AVG divides after
    if op == dist.ReduceOp.PRODUCT:
        return ga.ReduceOp.product
    if op == dist.ReduceOp.MIN:
        return ga.ReduceOp.min
    if op == dist.ReduceOp.MAX:
        return ga.ReduceOp.max
    raise ValueError(f"unsupported reduce op {op}")


class _OPSView:
    def __getitem__(self, op):
        return _map_op(op)

    def get(self, op):
        try:
            return _map_op(op)
        except ValueError:
            return None


_OPS = _OPSView()


def _ret_work(result=None):
    fut = Future()
    fut.set_result(result)
    return _create_work_from_future(fut)


class _TorchStoreAdapter(ga.Store):
    """Bridges a c10d Store into the gloo_amd rendezvous interface."""

    def __init__(self, tstore):
        super().__init__()
        self._s = tstore

    def set(self, key, data):
        self._s.set(key, data)

    def get(self, key):
        return bytes(self._s.get(key))

    def wait(self, keys, timeout_ms):
        if timeout_ms and timeout_ms > 0:
            self._s.wait(list(keys), timedelta(milliseconds=timeout_ms))
        else:
            self._s.wait(list(keys))


def _gdtype(t):
    if t.dtype == torch.bool:
        return ga.DType.u8
    return ga.dtype_from_torch(t.dtype)


def _cur_stream(t):
    """The caller stream that produced tensor t (our device algorithms
    order themselves after it)."""
    return torch.cuda.current_stream(t.get_device()).cuda_stream


class ProcessGroupGlooAmd(dist.ProcessGroup):
    def __init__(self, store, rank, size, timeout=timedelta(seconds=300)):
        super().__init__(rank, size)
        if isinstance(store, ga.Store):
            self._store = store
        else:
            self._store = _TorchStoreAdapter(store)
        self._ctx = ga.Context(rank, size)
        self._ctx.set_timeout(int(timeout.total_seconds() * 1000))
        import os

        # Single-node deployments can route the CPU control plane over
        # unix-domain sockets (lower latency/higher throughput than TCP
        # loopback); multi-node needs TCP.
        uds = os.environ.get("GLOO_AMD_UDS", "0") == "1"
        self._ctx.connect_full_mesh(
            self._store, ga.create_tcp_device(use_uds=uds))
        self._lock = threading.Lock()
        # One helper thread: p2p waits complete in post order off the
        # caller's thread (isend/irecv return pending Works).
        self._p2p_pool = ThreadPoolExecutor(
            max_workers=1, thread_name_prefix="glooamd-p2p")
        self._mb_seq = 0
        self._hip_ring = {}  # device -> HipAllreduceRing
        self._hip_bcast = {}  # (device, root) -> HipBroadcastOneToAll
        self._hip_ag = {}  # device -> HipAllgatherRing
        self._hip_rs = {}  # device -> HipReduceScatterRing
        self._hip_a2a = {}  # device -> HipAlltoall
        # Device-native p2p engine (send/recv/gather/scatter without a
        # pinned round trip). Constructed EAGERLY: its mesh exchange is
        # collective, and p2p calls are not a safe collective point.
        self._hip_p2p = None
        self._hip_p2p_dev = -1
        if size > 1 and torch.cuda.is_available():
            self._hip_p2p_dev = torch.cuda.current_device()
            self._hip_p2p = ga._C.HipP2P(self._ctx, self._hip_p2p_dev)

    # -- helpers -------------------------------------------------------------

    def _tag(self):
        return self._ctx.next_slot()

    def _ring(self, device):
        if device not in self._hip_ring:
            import os

            algo = os.environ.get("GLOO_AMD_ALLREDUCE", "auto")
            size = self.size()
            if algo == "ring" or size > 8 or (algo == "auto" and size <= 2):
                self._hip_ring[device] = ga._C.HipAllreduceRing(
                    self._ctx, device)
            elif algo == "hd" and size & (size - 1) == 0:
                self._hip_ring[device] = ga._C.HipAllreduceHalvingDoubling(
                    self._ctx, device)
            else:
                # fully-connected xGMI: one-shot direct allreduce uses all
                # links concurrently
                self._hip_ring[device] = ga._C.HipAllreduceDirect(
                    self._ctx, device)
        return self._hip_ring[device]

    def _bcaster(self, device, root):
        key = (device, root)
        if key not in self._hip_bcast:
            self._hip_bcast[key] = ga._C.HipBroadcastOneToAll(
                self._ctx, device, root)
        return self._hip_bcast[key]

    def _ag(self, device):
        if device not in self._hip_ag:
            self._hip_ag[device] = ga._C.HipAllgatherRing(self._ctx, device)
        return self._hip_ag[device]

    def _rs(self, device):
        if device not in self._hip_rs:
            self._hip_rs[device] = ga._C.HipReduceScatterRing(
                self._ctx, device)
        return self._hip_rs[device]

    def _a2a(self, device):
        if device not in self._hip_a2a:
            self._hip_a2a[device] = ga._C.HipAlltoall(self._ctx, device)
        return self._hip_a2a[device]

    def _staged(self, t, fn):
        """Run a CPU collective on a contiguous host copy of t (CUDA or
        non-contiguous CPU), writing the result back. CUDA tensors stage
        through pinned memory (sync DMA instead of pageable copies)."""
        t_ = t.detach()
        if t_.is_cuda:
            host = torch.empty(t_.numel(), dtype=t_.dtype, pin_memory=True)
            host.copy_(t_.reshape(-1))
        else:
            host = t_.contiguous()
        fn(host)
        t_.copy_(host.view_as(t_))

    # -- collectives ---------------------------------------------------------

    def allreduce(self, tensors, opts=None):
        op = opts.reduceOp if opts is not None else dist.ReduceOp.SUM
        gop = _OPS.get(op)
        if gop is None:
            raise ValueError(f"unsupported reduce op {op}")
        with self._lock:
            for t in tensors:
                t_ = t.detach()
                staged = None
                if not t_.is_contiguous():
                    staged = t_.contiguous()
                    t_ = staged
                if t_.is_cuda:
                    self._ring(t_.get_device()).run(
                        t_.data_ptr(), t_.numel(), _gdtype(t_), gop,
                        stream=_cur_stream(t_))
                else:
                    ga.allreduce(
                        self._ctx, [t_.data_ptr()], t_.numel(), _gdtype(t_),
                        gop, tag=self._tag())
                if op == dist.ReduceOp.AVG:
                    t_.div_(self.size())
                if staged is not None:
                    t.detach().copy_(staged)
        return _ret_work(tensors)

    def broadcast(self, tensors, opts=None):
        root = opts.rootRank if opts is not None else 0
        with self._lock:
            for t in tensors:
                t_ = t.detach()
                staged = None
                if not t_.is_contiguous():
                    staged = t_.contiguous()
                    t_ = staged
                if t_.is_cuda:
                    self._bcaster(t_.get_device(), root).run(
                        t_.data_ptr(), t_.numel() * t_.element_size(),
                        stream=_cur_stream(t_))
                else:
                    ga.broadcast(
                        self._ctx, t_.data_ptr(), 0, t_.numel(), _gdtype(t_),
                        root=root, tag=self._tag())
                if staged is not None:
                    t.detach().copy_(staged)
        return _ret_work(tensors)

    def _allgather_base(self, output, input, opts=None):
        out = output.detach()
        inp = input.detach()
        assert out.is_contiguous() and inp.is_contiguous()
        with self._lock:
            tag = self._tag()
            if out.is_cuda:
                self._ag(out.get_device()).run(
                    inp.data_ptr(), out.data_ptr(), inp.numel(),
                    inp.element_size(), stream=_cur_stream(out))
            else:
                ga.allgather(self._ctx, out.data_ptr(), inp.data_ptr(),
                             inp.numel(), _gdtype(inp), tag=tag)
        return _ret_work(output)

    def allgather(self, output_tensors, input_tensors, opts=None):
        for out_list, inp in zip(output_tensors, input_tensors):
            flat = torch.empty(
                inp.numel() * self.size(), dtype=inp.dtype,
                device="cpu")
            inp_c = inp.detach().contiguous()
            with self._lock:
                h_in = inp_c.cpu() if inp_c.is_cuda else inp_c
                ga.allgather(self._ctx, flat.data_ptr(), h_in.data_ptr(),
                             h_in.numel(), _gdtype(inp), tag=self._tag())
            for r, out in enumerate(out_list):
                out.detach().copy_(
                    flat[r * inp.numel():(r + 1) * inp.numel()].view_as(out))
        return _ret_work(output_tensors)

    def reduce(self, tensors, opts=None):
        root = opts.rootRank if opts is not None else 0
        op = _OPS[opts.reduceOp] if opts is not None else ga.ReduceOp.sum
        for t in tensors:
            t_ = t.detach()
            if t_.is_cuda and t_.is_contiguous():
                # Device-native: run the device allreduce; the root holds
                # the reduced result, non-root contents are unspecified
                # by the c10d contract (here: also the full result).
                with self._lock:
                    self._ring(t_.get_device()).run(
                        t_.data_ptr(), t_.numel(), _gdtype(t_), op,
                        stream=_cur_stream(t_))
                if opts is not None \
                        and opts.reduceOp == dist.ReduceOp.AVG:
                    t_.div_(self.size())
                continue

            def run(host):
                ga.reduce(self._ctx, host.data_ptr(), host.data_ptr(),
                          host.numel(), _gdtype(t_), op, root=root,
                          tag=self._tag())

            with self._lock:
                if t_.is_cuda or not t_.is_contiguous():
                    self._staged(t_, run)
                else:
                    run(t_)
            if opts is not None and opts.reduceOp == dist.ReduceOp.AVG \
                    and self.rank() == root:
                t_.div_(self.size())
        return _ret_work(tensors)

    def _reduce_scatter_base(self, output, input, opts=None):
        op = _OPS[opts.reduceOp] if opts is not None else ga.ReduceOp.sum
        out = output.detach()
        inp = input.detach().contiguous()
        with self._lock:
            tag = self._tag()
            if out.is_cuda:
                self._rs(out.get_device()).run(
                    inp.data_ptr(), out.data_ptr(), out.numel(),
                    _gdtype(out), op, stream=_cur_stream(out))
            else:
                ga.reduce_scatter(self._ctx, out.data_ptr(), inp.data_ptr(),
                                  out.numel(), _gdtype(out), op, tag=tag)
        if opts is not None and opts.reduceOp == dist.ReduceOp.AVG:
            out.div_(self.size())
        return _ret_work(output)

    def reduce_scatter(self, output_tensors, input_tensor_lists, opts=None):
        for out, in_list in zip(output_tensors, input_tensor_lists):
            flat = torch.cat([t.detach().reshape(-1).cpu() for t in in_list])
            h_out = torch.empty(out.numel(), dtype=out.dtype)
            op = _OPS[opts.reduceOp] if opts is not None else ga.ReduceOp.sum
            with self._lock:
                ga.reduce_scatter(self._ctx, h_out.data_ptr(), flat.data_ptr(),
                                  h_out.numel(), _gdtype(out), op,
                                  tag=self._tag())
            out.detach().copy_(h_out.view_as(out))
            if opts is not None and opts.reduceOp == dist.ReduceOp.AVG:
                out.detach().div_(self.size())
        return _ret_work(output_tensors)

    def alltoall_base(self, output, input, output_split_sizes,
                      input_split_sizes, opts=None):
        out = output.detach()
        inp = input.detach().contiguous()
        if (out.is_cuda and not output_split_sizes
                and not input_split_sizes):
            with self._lock:
                self._a2a(out.get_device()).run(
                    inp.data_ptr(), out.data_ptr(),
                    inp.numel() // self.size(), inp.element_size(),
                    stream=_cur_stream(out))
            return _ret_work(output)
        out_staged = out.is_cuda or not out.is_contiguous()
        h_out = (torch.empty(out.numel(), dtype=out.dtype)
                 if out_staged else out)
        h_in = inp.cpu() if inp.is_cuda else inp
        with self._lock:
            tag = self._tag()
            if not output_split_sizes and not input_split_sizes:
                per = inp.numel() // self.size()
                ga.alltoall(self._ctx, h_out.data_ptr(), h_in.data_ptr(),
                            per, _gdtype(inp), tag=tag)
            else:
                row = inp.numel() // inp.shape[0] if inp.dim() > 0 else 1
                in_counts = [int(s) * row for s in input_split_sizes]
                out_counts = [int(s) * row for s in output_split_sizes]
                ga.alltoallv(self._ctx, h_out.data_ptr(), h_in.data_ptr(),
                             in_counts, out_counts, _gdtype(inp), tag=tag)
        if out_staged:
            out.copy_(h_out.view_as(out))
        return _ret_work(output)

    def alltoall(self, output_tensors, input_tensors, opts=None):
        flat_in = torch.cat(
            [t.detach().reshape(-1).cpu() for t in input_tensors])
        counts_in = [t.numel() for t in input_tensors]
        counts_out = [t.numel() for t in output_tensors]
        flat_out = torch.empty(sum(counts_out), dtype=output_tensors[0].dtype)
        with self._lock:
            ga.alltoallv(self._ctx, flat_out.data_ptr(), flat_in.data_ptr(),
                         counts_in, counts_out,
                         _gdtype(input_tensors[0]), tag=self._tag())
        off = 0
        for t in output_tensors:
            t.detach().copy_(flat_out[off:off + t.numel()].view_as(t))
            off += t.numel()
        return _ret_work(output_tensors)

    def gather(self, output_tensors, input_tensors, opts=None):
        root = opts.rootRank if opts is not None else 0
        inp = input_tensors[0].detach().contiguous()
        if self._p2p_device_ok([input_tensors[0]]) and (
                self.rank() != root
                or all(o.is_cuda and o.is_contiguous()
                       for o in output_tensors[0])):
            # Device-native: non-roots stream their block to the root's
            # inbox lanes; the root collects each lane straight into the
            # output tensor (no pinned staging).
            nbytes = inp.numel() * inp.element_size()
            if self.rank() == root:
                outs = output_tensors[0]
                for src in range(self.size()):
                    if src == root:
                        outs[src].detach().copy_(inp.view_as(outs[src]))
                    else:
                        self._hip_p2p.post_recv(
                            src, outs[src].data_ptr(), nbytes,
                            _cur_stream(outs[src]))
                self._hip_p2p.flush_recvs()
            else:
                self._hip_p2p.post_send(
                    root, inp.data_ptr(), nbytes, _cur_stream(inp))
                self._hip_p2p.flush_sends()
            return _ret_work(output_tensors)
        h_in = inp.cpu() if inp.is_cuda else inp
        if self.rank() == root:
            outs = output_tensors[0]
            flat = torch.empty(inp.numel() * self.size(), dtype=inp.dtype)
            with self._lock:
                ga.gather(self._ctx, flat.data_ptr(), h_in.data_ptr(),
                          h_in.numel(), _gdtype(inp), root=root,
                          tag=self._tag())
            for r, out in enumerate(outs):
                out.detach().copy_(
                    flat[r * inp.numel():(r + 1) * inp.numel()].view_as(out))
        else:
            with self._lock:
                ga.gather(self._ctx, 0, h_in.data_ptr(), h_in.numel(),
                          _gdtype(inp), root=root, tag=self._tag())
        return _ret_work(output_tensors)

    def scatter(self, output_tensors, input_tensors, opts=None):
        root = opts.rootRank if opts is not None else 0
        out = output_tensors[0].detach()
        if self._p2p_device_ok([output_tensors[0]]) \
                and out.is_contiguous() and (
                self.rank() != root
                or all(t.is_cuda for t in input_tensors[0])):
            nbytes = out.numel() * out.element_size()
            if self.rank() == root:
                ins = [t.detach().contiguous() for t in input_tensors[0]]
                for dst in range(self.size()):
                    if dst == root:
                        out.copy_(ins[dst].view_as(out))
                    else:
                        self._hip_p2p.post_send(
                            dst, ins[dst].data_ptr(), nbytes,
                            _cur_stream(ins[dst]))
                self._hip_p2p.flush_sends()
            else:
                self._hip_p2p.post_recv(
                    root, out.data_ptr(), nbytes, _cur_stream(out))
                self._hip_p2p.flush_recvs()
            return _ret_work(output_tensors)
        h_out = torch.empty(out.numel(), dtype=out.dtype)
        if self.rank() == root:
            flat = torch.cat(
                [t.detach().reshape(-1).cpu() for t in input_tensors[0]])
            with self._lock:
                ga.scatter(self._ctx, h_out.data_ptr(), flat.data_ptr(),
                           out.numel(), _gdtype(out), root=root,
                           tag=self._tag())
        else:
            with self._lock:
                ga.scatter(self._ctx, h_out.data_ptr(), 0, out.numel(),
                           _gdtype(out), root=root, tag=self._tag())
        out.copy_(h_out.view_as(out))
        return _ret_work(output_tensors)

    def allreduce_coalesced(self, tensors, opts=None):
        return self.allreduce(tensors, opts)

    def allgather_into_tensor_coalesced(self, outputs, inputs, opts=None):
        for out, inp in zip(outputs, inputs):
            self._allgather_base(out, inp, opts).wait()
        return _ret_work(outputs)

    def reduce_scatter_tensor_coalesced(self, outputs, inputs, opts=None):
        for out, inp in zip(outputs, inputs):
            self._reduce_scatter_base(out, inp, opts).wait()
        return _ret_work(outputs)

    def barrier(self, opts=None):
        with self._lock:
            ga.barrier(self._ctx, tag=self._tag())
        return _ret_work(True)

    def monitored_barrier(self, timeout=None, wait_all_ranks=False):
        """Rank 0 collects an ack from every rank and reports WHICH ranks
        failed to arrive (the reference backend's role for this API in
        torch.distributed)."""
        tmo_ms = int(timeout.total_seconds() * 1000) if timeout else 30000
        rank, size = self.rank(), self.size()
        if size == 1:
            return
        with self._lock:
            slot = self._p2p_slot(0xB0000 + (self._mb_seq % 4096))
            self._mb_seq += 1
            byte = torch.zeros(1, dtype=torch.uint8)
            if rank == 0:
                # Post every probe first, then wait with try_wait_recv:
                # a timeout returns False without poisoning the context,
                # so later ranks are still probed truthfully and the
                # context stays usable after the report.
                deadline = time.monotonic() + tmo_ms / 1000.0
                missing, probes = [], []
                for src in range(1, size):
                    try:
                        ub = self._ctx.create_unbound_buffer(
                            byte.data_ptr(), 1)
                        ub.recv(src, slot, 0, 0)
                        probes.append((src, ub))
                    except ga.GlooAmdError:
                        missing.append(src)  # pair already dead
                for src, ub in probes:
                    rem = max(0, int((deadline - time.monotonic()) * 1000))
                    try:
                        ok, _ = ub.try_wait_recv(rem)
                    except ga.GlooAmdError:
                        ok = False  # peer died while probing
                    if not ok:
                        missing.append(src)
                del probes  # detaches any unmatched pending recvs
                if missing:
                    raise RuntimeError(
                        "monitored_barrier: rank(s) "
                        f"{sorted(missing)} failed to arrive within "
                        f"{tmo_ms}ms")
                for dst in range(1, size):
                    ub = self._ctx.create_unbound_buffer(byte.data_ptr(), 1)
                    ub.send(dst, slot + 1, 0, 0)
                    ub.wait_send(tmo_ms)
            else:
                ub = self._ctx.create_unbound_buffer(byte.data_ptr(), 1)
                ub.send(0, slot, 0, 0)
                ub.wait_send(tmo_ms)
                ub2 = self._ctx.create_unbound_buffer(byte.data_ptr(), 1)
                ub2.recv(0, slot + 1, 0, 0)
                ub2.wait_recv(tmo_ms)

    def _p2p_async(self, finish):
        """Run finish() (a blocking wait + optional copy-back) on the p2p
        helper thread and return a torch Work tied to it. Keeps send and
        recv truly asynchronous so batch_isend_irecv-style patterns
        (post everything, then wait) cannot deadlock on the rendezvous."""
        fut = Future()

        def runner():
            try:
                finish()
                fut.set_result(None)
            except Exception as e:  # noqa: BLE001
                try:
                    fut.set_exception(e)
                except Exception:  # pragma: no cover
                    pass

        self._p2p_pool.submit(runner)
        return _create_work_from_future(fut)

    def _p2p_device_ok(self, tensors):
        return (self._hip_p2p is not None
                and all(t.is_cuda and t.get_device() == self._hip_p2p_dev
                        for t in tensors))

    def send(self, tensors, dst_rank, tag=0):
        if self._p2p_device_ok(tensors):
            # Device-native: chunks move straight over xGMI inbox lanes;
            # the post enqueues the whole schedule, the helper thread
            # only syncs (no pinned round trip).
            keep = []
            for t in tensors:
                t_ = t.detach().contiguous()
                keep.append(t_)  # storage alive until the flush
                self._hip_p2p.post_send(
                    dst_rank, t_.data_ptr(),
                    t_.numel() * t_.element_size(), _cur_stream(t_))

            def finish(keep=keep):
                self._hip_p2p.flush_sends()

            return self._p2p_async(finish)
        ubs = []
        for t in tensors:
            t_ = t.detach().contiguous()
            h = t_.cpu() if t_.is_cuda else t_
            ub = self._ctx.create_unbound_buffer(
                h.data_ptr(), h.numel() * h.element_size())
            ub.send(dst_rank, self._p2p_slot(tag))
            ubs.append((ub, h))  # keep the host staging alive until done

        def finish():
            for ub, _h in ubs:
                ub.wait_send()

        return self._p2p_async(finish)

    def recv(self, tensors, src_rank, tag=0):
        if self._p2p_device_ok(tensors) and all(
                t.is_contiguous() for t in tensors):
            for t in tensors:
                t_ = t.detach()
                self._hip_p2p.post_recv(
                    src_rank, t_.data_ptr(),
                    t_.numel() * t_.element_size(), _cur_stream(t_))
            tens = list(tensors)

            def finish(tens=tens):
                self._hip_p2p.flush_recvs()

            return self._p2p_async(finish)
        posted = []
        for t in tensors:
            t_ = t.detach()
            staged = t_.is_cuda or not t_.is_contiguous()
            h = (torch.empty(t_.numel(), dtype=t_.dtype,
                             pin_memory=t_.is_cuda) if staged else t_)
            ub = self._ctx.create_unbound_buffer(
                h.data_ptr(), h.numel() * h.element_size())
            ub.recv(src_rank, self._p2p_slot(tag))
            posted.append((ub, h, t_, staged))

        def finish():
            for ub, h, t_, staged in posted:
                ub.wait_recv()
                if staged:
                    t_.copy_(h.view_as(t_))

        return self._p2p_async(finish)

    @staticmethod
    def _p2p_slot(tag):
        # user tag band, below the context's next_slot() counter space
        return 0x0900_0000_0000_0000 | (tag & 0xFFFFF)

    def getBackendName(self):
        return "glooamd"

    @property
    def options(self):  # some callers poke at this
        return None


def _create_glooamd_pg(prefix_store, rank, world_size, timeout):
    return ProcessGroupGlooAmd(prefix_store, rank, world_size, timeout)


if "GLOOAMD" not in getattr(dist.Backend, "_plugins", {}):
    dist.Backend.register_backend(
        "glooamd", _create_glooamd_pg, devices=["cpu", "cuda"])
