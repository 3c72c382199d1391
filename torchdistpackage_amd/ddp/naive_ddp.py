"""NaiveDdp: pure-Python data-parallel with bucketed, overlapped grad all-reduce.

Capability parity with the reference NaiveDDP
(/root/reference/torchdistpackage/ddp/naive_ddp.py:13-230, GradBucket :444-478):
broadcast params at wrap, per-param grad-ready hooks, bucketed all-reduce on a
side stream overlapped with backward, ``num_grad_acc_iter`` support so that
with pipeline parallelism the reduce fires only on the last micro-batch while
staying overlapped, and ``reduce_gradients()`` finalization.

MI355X-first design decisions (not a translation):

- Grad-ready hooks use the public ``Tensor.register_post_accumulate_grad_hook``
  (torch >= 2.1) instead of the fragile
  ``p.expand_as(p).grad_fn.next_functions[0][0]`` AccumulateGrad walk the
  reference uses (naive_ddp.py:84-93) — same firing point, no hidden-ref
  hazard, works on CPU/gloo for tests.
- Buckets are persistent flat buffers (allocated once at warmup, 512-B-aligned
  segments) reduced with ReduceOp.AVG on a dedicated HIP side stream.  Default
  bucket cap is 50 MiB: a ring all-reduce over 8 GPUs moves 2·(n-1)/n ≈ 1.75×
  the payload per link at ≈153 GB/s xGMI per-link bandwidth, so a 50 MiB bucket
  costs ≈0.6 ms — long enough to amortize launch overhead, short enough to
  pipeline several buckets under backward.
- Event-based producer/consumer ordering: the side stream waits on the compute
  stream before packing+reducing; at the end of the iteration the compute
  stream waits on the side stream's event (no global device sync).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import torch
import torch.distributed as dist
import torch.nn as nn

_ALIGN_ELEMS = 128  # 512 B at fp32 — keeps RCCL segment starts aligned


def _align(n: int) -> int:
    return (n + _ALIGN_ELEMS - 1) // _ALIGN_ELEMS * _ALIGN_ELEMS


class GradBucket:
    """Persistent flat buffer holding several params' grads for one all-reduce.

    Reference parity: naive_ddp.py:444-478 (aligned offsets, ready-count reuse
    protocol, push-copies-into-view).
    """

    def __init__(self, params: List[torch.Tensor], dtype, device):
        self.params = params
        self.offsets: List[int] = []
        off = 0
        for p in params:
            self.offsets.append(off)
            off += _align(p.numel())
        self.data = torch.zeros(off, dtype=dtype, device=device)
        self.views = [
            self.data.narrow(0, o, p.numel()).view_as(p)
            for o, p in zip(self.offsets, params)
        ]
        self.ready = 0
        # which members arrived THIS comm iteration (cleared by the wrapper
        # after writeback, not by reset()): a partial-bucket flush must not
        # reduce stale data left in un-pushed views from a prior iteration
        self.pushed = [False] * len(params)
        self.reduced_event: Optional[torch.cuda.Event] = None

    def push(self, idx: int, grad: torch.Tensor) -> bool:
        """Copy grad into its view; True when every member has arrived."""
        self.views[idx].copy_(grad)
        self.pushed[idx] = True
        self.ready += 1
        return self.ready == len(self.params)

    def reset(self):
        self.ready = 0

    def clear_pushed(self):
        for i in range(len(self.pushed)):
            self.pushed[i] = False

    def zero_unpushed(self):
        """Zero views whose param produced no grad this iteration so a
        partial flush contributes zeros instead of stale values."""
        for i, v in enumerate(self.views):
            if not self.pushed[i]:
                v.zero_()


class NaiveDdp(nn.Module):
    """Data-parallel wrapper with overlapped bucketed grad all-reduce.

    Args:
        module: the local model (already on its device).
        group: process group to reduce over (default: tpc 'data' group if
            initialized, else WORLD).
        sync: if True, all-reduce each grad immediately at grad-ready (no
            bucketing, no side stream) — the reference's ``sync=True`` path.
        bucket_cap_mb: flat-bucket capacity in MiB.
        num_grad_acc_iter: reduces fire only on every num_grad_acc_iter-th
            backward (grad accumulation / PP micro-batching); intermediate
            backwards skip communication entirely.
    """

    def __init__(self, module: nn.Module, group: Optional[dist.ProcessGroup] = None,
                 sync: bool = False, bucket_cap_mb: float = 50.0,
                 num_grad_acc_iter: int = 1, broadcast_params: bool = True,
                 reduce_op: str = "avg"):
        super().__init__()
        self.module = module
        # reference parity: reduce_op "avg" (default) or "sum"; the
        # reference's selector is broken (reduce_op.lower missing parens,
        # naive_ddp.py:53 — "sum" is never chosen there)
        assert reduce_op in ("avg", "sum"), reduce_op
        self.reduce_op = reduce_op
        if group is None:
            try:
                from ..dist.topo import tpc
                group = tpc.get_group("data") if tpc.is_mode_inited("data") \
                    else None
            except Exception:
                group = None
        self.group = group
        self.sync = sync
        self.bucket_cap = int(bucket_cap_mb * 1024 * 1024)
        self.num_grad_acc_iter = max(1, num_grad_acc_iter)
        # per-param backward-fire counter: with grad accumulation / PP
        # micro-batching each param's hook fires once per micro-batch; only
        # every num_grad_acc_iter-th fire communicates.
        self._fires: Dict[int, int] = {}

        # expert-parallel params are excluded: they are different per EP rank
        # and sync over 'moe_dp' via MoEDP instead (reference moe_dp.md)
        def _is_expert(p):
            return getattr(p, "expert_parallel", False)
        self._params = [p for p in module.parameters()
                        if p.requires_grad and not _is_expert(p)]
        self._param_bucket: Dict[int, tuple] = {}  # id(p) -> (bucket, idx)
        self._buckets: List[GradBucket] = []
        self._hooks = []
        self._use_gpu = torch.cuda.is_available() and \
            any(p.is_cuda for p in self._params)
        self._reduce_stream = torch.cuda.Stream() if self._use_gpu else None
        self._pending_events: List[torch.cuda.Event] = []
        self._works = []

        if broadcast_params and dist.is_initialized() and \
                dist.get_world_size(self.group) > 1:
            with torch.no_grad():
                for p in module.parameters():
                    if _is_expert(p):
                        continue
                    dist.broadcast(p.data, src=self._group_src(), group=self.group)
                for b in module.buffers():
                    if b.dtype.is_floating_point or b.dtype in (
                            torch.int64, torch.int32):
                        dist.broadcast(b.data, src=self._group_src(),
                                       group=self.group)

        self._build_buckets()
        self._register_hooks()

    # ------------------------------------------------------------------

    def _group_src(self) -> int:
        if self.group is None:
            return 0
        return dist.get_process_group_ranks(self.group)[0]

    def _world(self) -> int:
        return dist.get_world_size(self.group) if dist.is_initialized() else 1

    def _avg(self) -> bool:
        return self.reduce_op == "avg"

    def _build_buckets(self):
        """Pack params into buckets in reverse registration order (grads become
        ready roughly back-to-front during backward)."""
        if self.sync:
            return
        cur: List[torch.Tensor] = []
        cur_bytes = 0
        for p in reversed(self._params):
            nbytes = _align(p.numel()) * p.element_size()
            if cur and cur_bytes + nbytes > self.bucket_cap:
                self._close_bucket(cur)
                cur, cur_bytes = [], 0
            cur.append(p)
            cur_bytes += nbytes
        if cur:
            self._close_bucket(cur)

    def _close_bucket(self, params: List[torch.Tensor]):
        b = GradBucket(params, params[0].dtype, params[0].device)
        for i, p in enumerate(params):
            self._param_bucket[id(p)] = (b, i)
        self._buckets.append(b)

    def _register_hooks(self):
        for p in self._params:
            h = p.register_post_accumulate_grad_hook(self._on_grad_ready)
            self._hooks.append(h)

    # ------------------------------------------------------------------

    def _on_grad_ready(self, p: torch.Tensor):
        if self._world() == 1:
            return
        fires = self._fires.get(id(p), 0) + 1
        self._fires[id(p)] = fires
        if fires % self.num_grad_acc_iter != 0:
            return  # intermediate micro-batch: accumulate only
        if self.sync:
            if self._use_gpu and self._avg():
                dist.all_reduce(p.grad, op=dist.ReduceOp.AVG, group=self.group)
            else:  # gloo has no AVG; or reduce_op == "sum"
                dist.all_reduce(p.grad, op=dist.ReduceOp.SUM, group=self.group)
                if self._avg():
                    p.grad.div_(self._world())
            return
        bucket, idx = self._param_bucket[id(p)]
        if self._use_gpu:
            # the copy into the bucket runs on the compute stream (ordered
            # after grad production); the reduce runs on the side stream.
            full = bucket.push(idx, p.grad)
            if full:
                self._reduce_bucket_async(bucket)
        else:
            full = bucket.push(idx, p.grad)
            if full:
                self._reduce_bucket_cpu(bucket)

    def _reduce_bucket_async(self, bucket: GradBucket):
        cur = torch.cuda.current_stream()
        ev = torch.cuda.Event()
        ev.record(cur)
        with torch.cuda.stream(self._reduce_stream):
            self._reduce_stream.wait_event(ev)
            dist.all_reduce(bucket.data,
                            op=dist.ReduceOp.AVG if self._avg()
                            else dist.ReduceOp.SUM, group=self.group)
            done = torch.cuda.Event()
            done.record(self._reduce_stream)
            bucket.reduced_event = done
            self._pending_events.append(done)
        bucket.reset()

    def _reduce_bucket_cpu(self, bucket: GradBucket):
        work = dist.all_reduce(bucket.data, op=dist.ReduceOp.SUM,
                               group=self.group, async_op=True)
        self._works.append((work, bucket))
        bucket.reset()

    # ------------------------------------------------------------------

    def forward(self, *args, **kwargs):
        return self.module(*args, **kwargs)

    @torch.no_grad()
    def reduce_gradients(self):
        """Finalize an iteration: flush, join the side stream, and write the
        reduced grads back into ``p.grad``.  Call once per optimizer step
        (after the last accumulated micro-batch's backward)."""
        if self._world() == 1:
            return
        if self.sync:
            return

        if self._use_gpu:
            cur = torch.cuda.current_stream()
            for ev in self._pending_events:
                cur.wait_event(ev)
            self._pending_events.clear()
            # flush any bucket that never filled (params with no grad this
            # iter leave it partial): un-pushed views are zeroed first so the
            # collective contributes zeros, never stale values from an
            # earlier iteration
            for b in self._buckets:
                if b.ready > 0:
                    b.zero_unpushed()
                    dist.all_reduce(b.data,
                                    op=dist.ReduceOp.AVG if self._avg()
                                    else dist.ReduceOp.SUM, group=self.group)
                    b.reset()
            for b in self._buckets:
                for p, v, pu in zip(b.params, b.views, b.pushed):
                    if pu and p.grad is not None:
                        p.grad.copy_(v)
                b.clear_pushed()
        else:
            for work, b in self._works:
                work.wait()
                if self._avg():
                    b.data.div_(self._world())
            self._works.clear()
            for b in self._buckets:
                if b.ready > 0:
                    b.zero_unpushed()
                    dist.all_reduce(b.data, op=dist.ReduceOp.SUM,
                                    group=self.group)
                    if self._avg():
                        b.data.div_(self._world())
                    b.reset()
            for b in self._buckets:
                for p, v, pu in zip(b.params, b.views, b.pushed):
                    if pu and p.grad is not None:
                        p.grad.copy_(v)
                b.clear_pushed()

    def remove_hooks(self):
        """Detach all grad hooks (e.g. to re-wrap the module elsewhere)."""
        for h in self._hooks:
            h.remove()
        self._hooks.clear()

    def zero_grad(self, set_to_none: bool = True):
        self.module.zero_grad(set_to_none=set_to_none)

    def __getattr__(self, name):
        try:
            return super().__getattr__(name)
        except AttributeError:
            return getattr(self.module, name)


# reference-compatible alias (the reference exports ``NaiveDDP``)
NaiveDDP = NaiveDdp
