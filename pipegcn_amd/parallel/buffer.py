"""Pipelined boundary feature/gradient exchange (the heart of the framework).

Reimplements the reference's Buffer (/root/reference/helper/feature_buffer.py)
MI355X-first:

 - GPU-direct RCCL send/recv over xGMI on a dedicated side HIP stream — the
   reference's mandatory GPU->pinned->gloo->pinned->GPU staging is gone (its
   CUDA path was never implemented: feature_buffer.py:204-205).
 - ONE serialized comm thread with a FIFO queue instead of a 2L-thread pool:
   RCCL communicators must see the same op order on every rank; the SPMD
   enqueue order (F0..F_{L-1}, B_{L-1}..B1 per epoch) guarantees matching
   without message tags (NCCL has none).
 - Event discipline replacing the reference's global
   `current_stream().synchronize()` at every update/hook entry:
     * ready-event   (compute stream) — producer data (feat/grad) is valid;
       comm stream waits it before gathering/sending.
     * consumed-event (compute stream) — recv buffers of the previous epoch
       were read (concat / scatter-add); comm stream waits it before the new
       transfer overwrites them.
     * done-event    (comm/corr stream) — recvs (+EMA correction) landed;
       compute stream waits it before consuming.
     * host Event    — guarantees the done-event has been *recorded* before
       the consumer waits on it (not completion — no host sync on the GPU
       path).

Semantics preserved exactly (checked against the reference):
 - pipeline off: transfer is synchronous inside update()/the grad hook.
 - pipeline on: epoch E consumes what was sent during epoch E-1; epoch 0
   consumes zeros; staleness is exactly one epoch in both directions.
 - smoothing correction: EMA avg = m*avg + (1-m)*recv on a third stream,
   consumed instead of the raw recv (feature_buffer.py:186-191,137-140).
 - scatter-add of received boundary grads happens every epoch (zeros at
   epoch 0), before the new grad transfer is enqueued.
"""
from __future__ import annotations

import os
import queue
import threading
import time
from typing import List

import torch
import torch.distributed as dist

# PIPEGCN_DEBUG=1: log a checksum of every transfer's send/recv payload
# (per epoch/layer/direction) — the race-hunting facility the reference
# lacks (SURVEY §5 race detection).
_DEBUG = os.environ.get("PIPEGCN_DEBUG", "0") == "1"

from pipegcn_amd import ops
from pipegcn_amd.parallel.transport import RingTransport
from pipegcn_amd.utils.timer import comm_timer


class _CatInto(torch.autograd.Function):
    """cat([feat] + halo) into a persistent buffer (no per-epoch alloc).

    buf rides in a list so autograd does not treat it as a differentiable
    input; the halo blocks are buffers (no grad), so backward only routes
    the leading [num_in] slice back to feat — exactly torch.cat's grad."""

    @staticmethod
    def forward(ctx, feat, buf_list, halo):
        buf = buf_list[0]
        n = feat.shape[0]
        ctx.num_in = n
        with torch.no_grad():
            buf[:n].copy_(feat)
            for h in halo:
                m = h.shape[0]
                buf[n: n + m].copy_(h)
                n += m
        return buf

    @staticmethod
    def backward(ctx, g):
        return g[: ctx.num_in], None, None


class Buffer:
    def __init__(self):
        self._initialized = False

    def init_buffer(self, num_in: int, num_all: int, boundary, recv_shape,
                    layer_size: List[int], use_pp: bool = False,
                    backend: str = "gloo", pipeline: bool = False,
                    corr_feat: bool = False, corr_grad: bool = False,
                    corr_momentum: float = 0.95, device: str = "cpu",
                    group=None, collect_stats: bool = False,
                    dtype: torch.dtype = torch.float32,
                    solo_world: int = 0):
        self._rank = dist.get_rank() if dist.is_initialized() else 0
        self._size = dist.get_world_size() if dist.is_initialized() else 1
        if solo_world > 1:
            # sizing mode (bench.py --solo-of N): full N-rank buffer layout,
            # streams and event choreography, but no peers — transfers skip
            # the wire and recv buffers stay zero (= epoch-0 semantics).
            assert self._size == 1
            self._size = solo_world
        self._num_in = num_in
        self._num_all = num_all
        self._boundary = boundary
        self._recv_shape = recv_shape
        self._layer_size = layer_size
        self._n_layers = len(layer_size)
        self._use_pp = use_pp
        self._pipeline = pipeline
        self._corr_feat, self._corr_grad = corr_feat, corr_grad
        self._corr_momentum = corr_momentum
        self._epoch = 0
        self._cat_buf = {}
        self._device = torch.device(device)
        self._use_cuda = self._device.type == "cuda"
        self._transport = (RingTransport(group)
                           if self._size > 1 and solo_world <= 1 else None)
        self._collect_stats = collect_stats
        self._stats_pairs = []  # (start_evt, end_evt) on comm stream
        self._comm_busy_host = 0.0

        # U-space slice of each peer's halo rows
        self._pl = [None] * self._size
        self._pr = [None] * self._size
        tot = num_in
        for j in range(self._size):
            if recv_shape[j] is not None:
                self._pl[j] = tot
                tot += recv_shape[j]
                self._pr[j] = tot

        L = self._n_layers
        S = self._size
        dev = self._device
        mk = lambda n, f: torch.zeros(n, f, device=dev, dtype=dtype)
        self._f_recv = [[None] * S for _ in range(L)]
        self._b_recv = [[None] * S for _ in range(L)]
        self._f_avg = [[None] * S for _ in range(L)]
        self._b_avg = [[None] * S for _ in range(L)]
        self._f_send = [[None] * S for _ in range(L)]
        for i in range(L):
            if i == 0 and use_pp:
                continue
            for j in range(S):
                if j == self._rank or self._size == 1:
                    continue
                nb = boundary[j].numel()
                nr = recv_shape[j]
                self._f_recv[i][j] = mk(nr, layer_size[i])
                self._f_send[i][j] = mk(nb, layer_size[i])
                if corr_feat:
                    self._f_avg[i][j] = mk(nr, layer_size[i])
                if i > 0:
                    self._b_recv[i][j] = mk(nb, layer_size[i])
                    if corr_grad:
                        self._b_avg[i][j] = mk(nb, layer_size[i])

        # events
        self._f_host_evt = [threading.Event() for _ in range(L)]
        self._b_host_evt = [threading.Event() for _ in range(L)]
        if self._use_cuda:
            cue = torch.cuda.Event
            self._comm_stream = torch.cuda.Stream()
            self._corr_stream = torch.cuda.Stream()
            self._f_done_evt = [cue() for _ in range(L)]
            self._b_done_evt = [cue() for _ in range(L)]
            self._f_consumed_evt = [cue() for _ in range(L)]
            self._b_consumed_evt = [cue() for _ in range(L)]
        else:
            self._comm_stream = self._corr_stream = None
            self._f_done_evt = [None] * L
            self._b_done_evt = [None] * L
            self._f_consumed_evt = [None] * L
            self._b_consumed_evt = [None] * L

        # Force collective creation of the RCCL communicator NOW, on the
        # main thread, at a deterministic point on every rank — before the
        # comm thread issues its first grouped send/recv concurrently with
        # main-thread collectives on the default group.
        if (self._transport is not None and self._use_cuda
                and self._transport.is_nccl):
            dist.all_reduce(torch.zeros(1, device=self._device), group=group)

        # serialized comm thread
        self._queue: queue.Queue = queue.Queue()
        self._exc = None
        self._thread = threading.Thread(target=self._comm_loop, daemon=True)
        self._thread.start()
        self._initialized = True

    # ------------------------------------------------------------------ API

    def next_epoch(self):
        self._epoch += 1

    def update(self, layer: int, feat: torch.Tensor) -> torch.Tensor:
        """Return the halo-extended feature tensor for this conv layer."""
        self._check_err()
        if self._size == 1:
            # single partition: no peers, no transfer
            return feat
        if not self._pipeline:
            with comm_timer.timer(f"forward_{layer}"):
                self._submit_feat(layer, feat)
                self._wait_host(self._f_host_evt[layer])
                self._wait_dev(self._f_done_evt[layer])
        else:
            if self._epoch > 0:
                with comm_timer.timer(f"forward_{layer}"):
                    self._wait_host(self._f_host_evt[layer])
                    self._wait_dev(self._f_done_evt[layer])
        buf = self._feat_concat(layer, feat)
        if self._use_cuda:
            self._f_consumed_evt[layer].record()
        if self._pipeline:
            self._submit_feat(layer, feat)
        if buf.requires_grad:
            buf.register_hook(self._grad_hook(layer))
        return buf

    def _grad_hook(self, layer: int):
        def fn(grad):
            self._check_err()
            if layer == 0:
                return grad  # input features carry no remote grads
            if not grad.is_contiguous():  # e.g. expanded grad from .sum()
                grad = grad.contiguous()
            if not self._pipeline:
                with comm_timer.timer(f"backward_{layer}"):
                    self._submit_grad(layer, grad)
                    self._wait_host(self._b_host_evt[layer])
                    self._wait_dev(self._b_done_evt[layer])
                self._apply_grad(layer, grad)
                if self._use_cuda:
                    self._b_consumed_evt[layer].record()
            else:
                if self._epoch > 0:
                    with comm_timer.timer(f"backward_{layer}"):
                        self._wait_host(self._b_host_evt[layer])
                        self._wait_dev(self._b_done_evt[layer])
                self._apply_grad(layer, grad)
                if self._use_cuda:
                    self._b_consumed_evt[layer].record()
                self._submit_grad(layer, grad)
            return grad

        return fn

    def synchronize(self):
        """Drain all in-flight transfers (end of training / teardown)."""
        self._queue.join()
        self._check_err()

    def pop_comm_stats(self):
        """Return (comm_busy_seconds) measured on the comm stream since the
        last call. Needs collect_stats=True."""
        if self._use_cuda:
            torch.cuda.synchronize()
            busy = sum(a.elapsed_time(b) for a, b in self._stats_pairs) / 1e3
            self._stats_pairs.clear()
        else:
            busy = self._comm_busy_host
            self._comm_busy_host = 0.0
        return busy

    # --------------------------------------------------------- internals

    def _check_err(self):
        if self._exc is not None:
            exc, self._exc = self._exc, None
            raise RuntimeError("comm thread failed") from exc

    def _wait_host(self, evt: threading.Event):
        while not evt.wait(timeout=60.0):
            self._check_err()
        # a failing comm thread sets the event to unblock waiters — check
        # unconditionally so the error surfaces HERE, not an epoch later
        self._check_err()
        evt.clear()

    def _wait_dev(self, evt):
        if self._use_cuda:
            torch.cuda.current_stream().wait_event(evt)

    def _feat_concat(self, layer: int, feat: torch.Tensor) -> torch.Tensor:
        src = self._f_avg if self._corr_feat else self._f_recv
        halo = [src[layer][j] for j in range(self._size)
                if j != self._rank]
        # PERSISTENT concat buffer, written in place: torch.cat here
        # allocates a fresh [num_all, F] every layer every epoch — at the
        # papers100M sizing that is a 26 GiB alloc/free cycle whose
        # fragmentation (35+ GB reserved-but-unallocated) OOMs a run that
        # otherwise fits. Same copy traffic as cat, zero per-epoch
        # allocations. Valid because epoch E's backward (which saves
        # VIEWS of this buffer) completes before epoch E+1's forward
        # overwrites it — the training loop is sequential by construction.
        buf = self._cat_buf.get(layer)
        need = (self._num_all, feat.shape[1])
        if buf is None or buf.shape != need or buf.dtype != feat.dtype                 or buf.device != feat.device:
            buf = torch.empty(need, dtype=feat.dtype, device=feat.device)
            self._cat_buf[layer] = buf
        return _CatInto.apply(feat, [buf], halo)

    def _apply_grad(self, layer: int, grad: torch.Tensor):
        src = self._b_avg if self._corr_grad else self._b_recv
        for j in range(self._size):
            if j == self._rank:
                continue
            ops.scatter_add_rows(grad, self._boundary[j], src[layer][j])

    def _submit_feat(self, layer: int, feat: torch.Tensor):
        ready = self._record_ready()
        self._queue.put(("feat", self._epoch, layer, feat, ready))

    def _submit_grad(self, layer: int, grad: torch.Tensor):
        ready = self._record_ready()
        self._queue.put(("grad", self._epoch, layer, grad, ready))

    def _record_ready(self):
        if self._use_cuda:
            e = torch.cuda.Event()
            e.record()
            return e
        return None

    def _comm_loop(self):
        while True:
            task = self._queue.get()
            try:
                if task is None:
                    return
                kind, epoch, layer, tensor, ready = task
                if kind == "feat":
                    self._do_feat_transfer(epoch, layer, tensor, ready)
                else:
                    self._do_grad_transfer(epoch, layer, tensor, ready)
            except Exception as e:  # surfaced at the next wait
                self._exc = e
                # unblock any waiter
                for evt in self._f_host_evt + self._b_host_evt:
                    evt.set()
            finally:
                self._queue.task_done()
                del task

    def _stats_begin(self):
        if not self._collect_stats:
            return None
        if self._use_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record(self._comm_stream)
            return e
        return time.time()

    def _stats_end(self, t0):
        if not self._collect_stats or t0 is None:
            return
        if self._use_cuda:
            e = torch.cuda.Event(enable_timing=True)
            e.record(self._comm_stream)
            self._stats_pairs.append((t0, e))
        else:
            self._comm_busy_host += time.time() - t0

    def _do_feat_transfer(self, epoch, layer, feat, ready):
        tag = epoch * 2 * self._n_layers + layer
        if self._use_cuda:
            self._comm_stream.wait_event(ready)
            self._comm_stream.wait_event(self._f_consumed_evt[layer])
            t0 = self._stats_begin()
            with torch.cuda.stream(self._comm_stream):
                self._feat_exchange(layer, feat, tag)
                # the gather reads `feat` on the comm stream, but its
                # storage is released by autograd on the compute stream —
                # tell the caching allocator not to reuse it early
                feat.record_stream(self._comm_stream)
            done_stream = self._comm_stream
            if self._corr_feat:
                self._corr_stream.wait_stream(self._comm_stream)
                with torch.cuda.stream(self._corr_stream):
                    for j in range(self._size):
                        if j != self._rank:
                            ops.ema_update(self._f_avg[layer][j],
                                           self._f_recv[layer][j],
                                           self._corr_momentum)
                done_stream = self._corr_stream
            self._f_done_evt[layer].record(done_stream)
            self._stats_end(t0)
        else:
            t0 = self._stats_begin()
            self._feat_exchange(layer, feat, tag)
            if self._corr_feat:
                for j in range(self._size):
                    if j != self._rank:
                        ops.ema_update(self._f_avg[layer][j],
                                       self._f_recv[layer][j],
                                       self._corr_momentum)
            self._stats_end(t0)
        self._f_host_evt[layer].set()

    def _feat_exchange(self, layer, feat, tag):
        for j in range(self._size):
            if j == self._rank:
                continue
            ops.gather_rows_into(feat.detach(), self._boundary[j],
                                 self._f_send[layer][j])
        if self._transport is not None:
            self._transport.all_to_all(self._f_send[layer],
                                       self._f_recv[layer],
                                       key=("f", layer), tag=tag)
        if _DEBUG:
            self._debug_log("feat", tag, self._f_send[layer],
                            self._f_recv[layer])

    def _debug_log(self, kind, tag, send, recv):
        def csum(ts):
            return [round(float(t.double().sum().item()), 4)
                    for t in ts if t is not None]

        print(f"[pipegcn-debug r{self._rank}] {kind} tag={tag} "
              f"send={csum(send)} recv={csum(recv)}", flush=True)

    def _do_grad_transfer(self, epoch, layer, grad, ready):
        tag = epoch * 2 * self._n_layers + layer + self._n_layers
        send = [None] * self._size
        for j in range(self._size):
            if j != self._rank:
                send[j] = grad[self._pl[j]:self._pr[j]]
        if self._use_cuda:
            self._comm_stream.wait_event(ready)
            self._comm_stream.wait_event(self._b_consumed_evt[layer])
            t0 = self._stats_begin()
            with torch.cuda.stream(self._comm_stream):
                if self._transport is not None:
                    self._transport.all_to_all(send, self._b_recv[layer],
                                               key=("b", layer), tag=tag)
                # the sends read row slices of `grad` on the comm stream;
                # autograd frees grad's storage on the compute stream right
                # after the hook returns — same allocator-reuse hazard as
                # the feat path (and ProcessGroupNCCL's internal
                # recordStream is off under TORCH_NCCL_AVOID_RECORD_STREAMS)
                grad.record_stream(self._comm_stream)
                if _DEBUG:
                    self._debug_log("grad", tag, send, self._b_recv[layer])
            done_stream = self._comm_stream
            if self._corr_grad:
                self._corr_stream.wait_stream(self._comm_stream)
                with torch.cuda.stream(self._corr_stream):
                    for j in range(self._size):
                        if j != self._rank:
                            ops.ema_update(self._b_avg[layer][j],
                                           self._b_recv[layer][j],
                                           self._corr_momentum)
                done_stream = self._corr_stream
            self._b_done_evt[layer].record(done_stream)
            self._stats_end(t0)
        else:
            t0 = self._stats_begin()
            if self._transport is not None:
                self._transport.all_to_all(send, self._b_recv[layer],
                                           key=("b", layer), tag=tag)
            if _DEBUG:
                self._debug_log("grad", tag, send, self._b_recv[layer])
            if self._corr_grad:
                for j in range(self._size):
                    if j != self._rank:
                        ops.ema_update(self._b_avg[layer][j],
                                       self._b_recv[layer][j],
                                       self._corr_momentum)
            self._stats_end(t0)
        self._b_host_evt[layer].set()

    def shutdown(self):
        if getattr(self, "_thread", None) is not None:
            self._queue.put(None)
            self._thread.join(timeout=10.0)
            self._thread = None
