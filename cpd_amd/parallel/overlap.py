"""Backward-overlapped gradient pipeline.

The flat bucket is split into contiguous sub-buckets (by parameter order);
each parameter's post-accumulate-grad hook counts down its sub-bucket, and
when a sub-bucket's last gradient lands its APS + low-precision all-reduce
pipeline is issued on a dedicated comm stream — overlapping communication
with the rest of backward (the reference is fully synchronous and
per-parameter; SURVEY.md §7 hard-part 3: buckets carry per-param shift
vectors, so overlap does not change the APS algebra).

Numerics: identical per-parameter APS algebra; with mode="sequential" the
result is bit-identical to the unoverlapped fused path (tested).  With
mode="ring" the ring chunking is per-sub-bucket, so chunk rotation differs
from the single-bucket ring (same semantic class, all ranks bit-agree).
"""
import torch
import torch.distributed as dist

from .. import ops
from .ring import lp_all_reduce_


class OverlapPipeline:
    def __init__(self, bucket, grad_exp, grad_man, use_APS=True,
                 use_kahan=False, mode="ring", wire=None, num_buckets=4):
        # one pipeline per bucket: replace (and unhook) any previous one so
        # re-creating an LPTrainStep never double-registers hooks
        prev = getattr(bucket, "_overlap_pipeline", None)
        if prev is not None:
            prev.remove()
        bucket._overlap_pipeline = self
        self.bucket = bucket
        self.grad_exp = grad_exp
        self.grad_man = grad_man
        self.use_APS = use_APS
        self.use_kahan = use_kahan
        self.mode = mode
        self.wire = wire
        self.enabled = False
        self.events = []

        params = bucket.params
        offsets = bucket.offsets.tolist()
        total = offsets[-1]
        num_buckets = max(1, min(num_buckets, len(params)))
        target = total / num_buckets
        # contiguous param-index boundaries with ~equal element counts
        bounds = [0]
        for i in range(1, len(params)):
            if offsets[i] >= target * len(bounds) and len(bounds) < num_buckets:
                bounds.append(i)
        bounds.append(len(params))
        self.ranges = []  # (param_lo, param_hi, elem_lo, elem_hi, offs_dev)
        dev = bucket.flat.device
        for b in range(len(bounds) - 1):
            plo, phi = bounds[b], bounds[b + 1]
            elo, ehi = offsets[plo], offsets[phi]
            offs = torch.tensor([o - elo for o in offsets[plo:phi + 1]],
                                dtype=torch.int64, device=dev)
            self.ranges.append((plo, phi, elo, ehi, offs))
        self.counts = [phi - plo for (plo, phi, *_rest) in self.ranges]
        self.remaining = list(self.counts)

        self.stream = (torch.cuda.Stream()
                       if bucket.flat.is_cuda else None)

        param_to_bucket = {}
        for b, (plo, phi, *_r) in enumerate(self.ranges):
            for i in range(plo, phi):
                param_to_bucket[i] = b
        self._hooks = []
        for i, p in enumerate(params):
            self._hooks.append(p.register_post_accumulate_grad_hook(
                self._make_hook(param_to_bucket[i])))

    def _make_hook(self, b):
        def hook(_param):
            if not self.enabled:
                return
            self.remaining[b] -= 1
            if self.remaining[b] == 0:
                self._reduce_bucket(b)
        return hook

    def begin_step(self):
        self.remaining = list(self.counts)
        self.events = []
        self.enabled = True

    def _world(self):
        if dist.is_available() and dist.is_initialized():
            return dist.get_world_size()
        return 1

    def _reduce_bucket(self, b):
        _plo, _phi, elo, ehi, offs = self.ranges[b]
        flat = self.bucket.flat[elo:ehi]
        gpu = self.stream is not None
        if gpu:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            self.stream.wait_event(ev)
        ctx = torch.cuda.stream(self.stream) if gpu else _null_ctx()
        with ctx:
            self._pipeline(flat, offs)
            if gpu:
                done = torch.cuda.Event()
                done.record(self.stream)
                self.events.append(done)

    def _pipeline(self, flat, offs):
        W = self._world()
        distributed = dist.is_available() and dist.is_initialized()
        exp, man = self.grad_exp, self.grad_man
        fp32_path = exp == 8 and man == 23 and not self.use_kahan
        if fp32_path and not self.use_APS:
            if distributed:
                dist.all_reduce(flat)
            return
        shifts = None
        wire = self.wire
        if self.use_APS:
            shifts = ops.seg_max_exp(flat, offs, W, aligned=True)
            if distributed:
                dist.all_reduce(shifts, op=dist.ReduceOp.MAX)
            upper = float(2 ** (exp - 1) - 1)
            shifts = (upper - shifts).float()
            ops.scale_quantize_(flat, offs, shifts, man, exp, aligned=True)
            if wire is None and flat.is_cuda and man <= 7:
                wire = "bf16"
        if fp32_path:
            if distributed:
                dist.all_reduce(flat)
        elif distributed and W > 1:
            lp_all_reduce_(flat, exp, man, use_kahan=self.use_kahan,
                           mode=self.mode, wire=wire)
        else:
            from .dist import lp_all_reduce_single_
            lp_all_reduce_single_(flat, exp, man, self.use_kahan)
        if shifts is not None:
            ops.seg_scale_(flat, offs, shifts, -1, aligned=True)

    def finish(self):
        """Call after loss.backward(): drains stragglers (params that never
        got a grad this step) and joins the comm stream."""
        self.enabled = False
        for b, rem in enumerate(self.remaining):
            if 0 < rem < self.counts[b]:
                # partial bucket (some params unused this step): reduce now
                self._reduce_bucket(b)
            elif rem == self.counts[b]:
                # bucket never touched (all params unused): still must join
                # the collective on other ranks — reduce the (zero) grads
                self._reduce_bucket(b)
        if self.stream is not None:
            cur = torch.cuda.current_stream()
            for ev in self.events:
                cur.wait_event(ev)
        self.events = []

    def remove(self):
        for h in self._hooks:
            h.remove()
        self._hooks = []


class _null_ctx:
    def __enter__(self):
        return self

    def __exit__(self, *a):
        return False
