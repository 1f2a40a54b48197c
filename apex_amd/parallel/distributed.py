"""apex_amd.parallel.DistributedDataParallel — bucketed overlapped
gradient all-reduce over RCCL/xGMI.

API parity with the removed apex.parallel.DistributedDataParallel; the full
kwarg surface is pinned by tests/distributed/DDP/ddp_race_condition_test.py:
38-41 (``message_size``, ``gradient_predivide_factor``, ``delay_allreduce``,
``allreduce_trigger_params``, ``num_allreduce_streams``).

MI355X sizing: each MI355X has 7 point-to-point xGMI links (~153 GB/s each);
ring all-reduce is per-link bound, so buckets must be large enough to
amortize per-collective latency — the default ``message_size`` is 16M
*elements* (64 MB fp32 / 32 MB bf16 per bucket), larger than the reference's
CUDA default, because xGMI latency is link-serialized. Collectives run on
``num_allreduce_streams`` side HIP streams, overlapped with the remainder of
backward; gloo (CPU CI) uses async work handles instead of streams.
"""

from contextlib import contextmanager

import torch
import torch.distributed as dist

from .._ext import get_ext
from ..tracing import trace_mark


def flatten(tensors):
    ext = get_ext("apex_C")
    if ext is not None:
        return ext.flatten(tensors)
    return torch._utils._flatten_dense_tensors(tensors)


def unflatten(flat, tensors):
    ext = get_ext("apex_C")
    if ext is not None:
        return ext.unflatten(flat, tensors)
    return torch._utils._unflatten_dense_tensors(flat, tensors)


def flat_dist_call(tensors, call, extra_args=None):
    """Fused broadcast/allreduce over a list of tensors (dtype-bucketed)."""
    buckets = {}
    for t in tensors:
        buckets.setdefault(t.dtype, []).append(t)
    for bucket in buckets.values():
        coalesced = flatten(bucket)
        if extra_args is not None:
            call(coalesced, *extra_args)
        else:
            call(coalesced)
        if call is dist.all_reduce:
            coalesced /= dist.get_world_size()
        for buf, synced in zip(bucket, unflatten(coalesced, bucket)):
            buf.copy_(synced)


class DistributedDataParallel(torch.nn.Module):
    def __init__(
        self,
        module,
        message_size=16_000_000,
        delay_allreduce=False,
        shared_param=None,
        allreduce_trigger_params=None,
        retain_allreduce_buffers=False,
        allreduce_always_fp32=False,
        num_allreduce_streams=1,
        allreduce_communicators=None,
        gradient_average=True,
        gradient_predivide_factor=1.0,
        gradient_average_split_factor=None,
        prof=False,
    ):
        super().__init__()
        if shared_param is not None:
            raise ValueError("shared_param is deprecated (as in the reference)")

        self.module = module
        self.message_size = message_size
        self.delay_allreduce = delay_allreduce
        self.retain_allreduce_buffers = retain_allreduce_buffers
        self.allreduce_always_fp32 = allreduce_always_fp32
        self.gradient_average = gradient_average
        self.gradient_predivide_factor = gradient_predivide_factor
        self.num_allreduce_streams = num_allreduce_streams

        self.world_size = dist.get_world_size()
        self.allreduce_trigger_params = allreduce_trigger_params
        if allreduce_trigger_params is not None:
            self.trigger_ids = {id(p) for p in allreduce_trigger_params}
        else:
            self.trigger_ids = None

        self._disabled = False
        self._use_streams = torch.cuda.is_available() and next(module.parameters()).is_cuda
        if self._use_streams:
            self._streams = [torch.cuda.Stream() for _ in range(num_allreduce_streams)]
        else:
            self._streams = []
        self._next_stream = 0

        # one communicator per all-reduce stream: collectives on a single
        # process group serialize on its internal stream, so genuine overlap
        # of concurrent bucket reductions needs distinct communicators
        # (reference: allreduce_communicators / num_allreduce_streams)
        if allreduce_communicators is not None:
            self._process_groups = list(allreduce_communicators)
        elif num_allreduce_streams > 1 and dist.is_initialized():
            ranks = list(range(self.world_size))
            self._process_groups = [dist.new_group(ranks) for _ in range(num_allreduce_streams)]
        else:
            self._process_groups = [None]
        self._next_pg = 0

        # in-flight (work, bucket_grads, flat, stream) records
        self._pending = []
        self._active_params = []
        self._callback_queued = False
        self.allreduce_buffers = []

        # sync initial weights/buffers from rank 0
        self._sync_params_and_buffers()
        self._register_hooks()

    # --- setup ---
    def _sync_params_and_buffers(self):
        tensors = [p.detach() for p in self.module.parameters()]
        tensors += [b for b in self.module.buffers() if torch.is_tensor(b) and b.dtype.is_floating_point]
        tensors += [b for b in self.module.buffers() if torch.is_tensor(b) and not b.dtype.is_floating_point]
        if tensors:
            flat_dist_call([t for t in tensors if t.numel() > 0], dist.broadcast, (0,))

    def _register_hooks(self):
        self._grad_accs = []
        for p in self.module.parameters():
            if p.requires_grad:
                self._make_hook(p)

    def _make_hook(self, p):
        def hook(*unused):
            if self._disabled:
                return
            self._on_grad_ready(p)

        # post-accumulate-grad hook fires after .grad is final for this param
        handle = p.register_post_accumulate_grad_hook(lambda param: hook())
        self._grad_accs.append(handle)

    # --- per-iteration machinery ---
    def _on_grad_ready(self, p):
        if not self._callback_queued:
            torch.autograd.Variable._execution_engine.queue_callback(self._finish_backward)
            self._callback_queued = True
        if self.delay_allreduce:
            self._active_params.append(p)
            return
        self._active_params.append(p)
        if self.trigger_ids is not None:
            if id(p) in self.trigger_ids:
                self._flush_bucket()
        else:
            pending_elems = sum(q.grad.numel() for q in self._active_params if q.grad is not None)
            if pending_elems >= self.message_size:
                self._flush_bucket()

    def _flush_bucket(self):
        params = [p for p in self._active_params if p.grad is not None]
        self._active_params = []
        if not params:
            return
        # dtype-split buckets
        by_dtype = {}
        for p in params:
            by_dtype.setdefault(p.grad.dtype, []).append(p.grad)
        for grads in by_dtype.values():
            self._allreduce_bucket(grads)

    def _allreduce_bucket(self, grads):
        trace_mark(f"ddp.allreduce_bucket({len(grads)}t)")
        stream = None
        if self._use_streams:
            stream = self._streams[self._next_stream]
            self._next_stream = (self._next_stream + 1) % len(self._streams)
            stream.wait_stream(torch.cuda.current_stream())

        pg = self._process_groups[self._next_pg]
        self._next_pg = (self._next_pg + 1) % len(self._process_groups)
        ctx = torch.cuda.stream(stream) if stream is not None else _nullcontext()
        with ctx:
            flat = flatten(grads)
            orig_dtype = flat.dtype
            if self.allreduce_always_fp32 and flat.dtype != torch.float32:
                flat = flat.float()
            if self.gradient_predivide_factor != 1.0:
                flat.div_(self.gradient_predivide_factor)
            work = dist.all_reduce(flat, async_op=True, group=pg)
        self._pending.append((work, grads, flat, orig_dtype, stream))

    def _finish_backward(self):
        self._callback_queued = False
        # flush the tail bucket (or everything, for delay_allreduce)
        self._flush_bucket()
        self.allreduce_buffers = []
        for work, grads, flat, orig_dtype, stream in self._pending:
            work.wait()
            ctx = torch.cuda.stream(stream) if stream is not None else _nullcontext()
            with ctx:
                if self.gradient_average:
                    post = self.world_size / self.gradient_predivide_factor
                    if post != 1.0:
                        flat.div_(post)
                if flat.dtype != orig_dtype:
                    flat = flat.to(orig_dtype)
                if self.retain_allreduce_buffers:
                    self.allreduce_buffers.append(flat)
                for g, synced in zip(grads, unflatten(flat, grads)):
                    g.copy_(synced)
            if stream is not None:
                torch.cuda.current_stream().wait_stream(stream)
        self._pending = []

    @contextmanager
    def no_sync(self):
        old = self._disabled
        self._disabled = True
        try:
            yield
        finally:
            self._disabled = old

    def forward(self, *args, **kwargs):
        self._active_params = []
        # reset round-robin state so stream/communicator assignment is
        # identical across ranks every iteration (collective matching)
        self._next_stream = 0
        self._next_pg = 0
        return self.module(*args, **kwargs)

    def state_dict(self, *args, **kwargs):
        return self.module.state_dict(*args, **kwargs)

    def load_state_dict(self, *args, **kwargs):
        return self.module.load_state_dict(*args, **kwargs)


class _nullcontext:
    def __enter__(self):
        return None

    def __exit__(self, *a):
        return False


class Reducer:
    """Manual gradient/state reducer (reconstructed removed-apex
    ``apex.parallel.Reducer``): no hooks, no overlap — the user calls
    ``reduce()`` once per accumulation window. Useful when backward timing
    must stay untouched (e.g. gradient accumulation loops)."""

    def __init__(self, module_or_grads_list):
        if isinstance(module_or_grads_list, torch.nn.Module):
            self.module = module_or_grads_list
            flat_dist_call([p.data for p in self.module.parameters()],
                           dist.broadcast, (0,))
        else:
            self.module = None
            self.grads = module_or_grads_list

    def reduce(self):
        if self.module is not None:
            grads = [p.grad.data for p in self.module.parameters()
                     if p.grad is not None]
        else:
            grads = self.grads
        if grads:
            # flat_dist_call divides by world size after the all_reduce
            flat_dist_call(grads, dist.all_reduce)
