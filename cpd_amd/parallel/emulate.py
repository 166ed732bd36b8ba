"""Node emulation: one GPU pretends to be `emulate_node` ranks.

Reference semantics (mix.py:224-282, duplicated in main.py:156-202 — here a
single shared component): every micro-batch's full gradient set is buffered;
at the boundary the APS max-exponent is taken over ALL buffered copies with
the emulate_node world-factor, each copy is scaled+quantized, and the copies
are summed with the same quantized accumulation the ring applies per hop —
i.e. the ring is replayed locally with the identical HIP kernels, so 1 GPU
with emulate_node=N reproduces the numerics of an N-rank low-precision
reduction.  Buffers are flat bucket clones (288 GB HBM: N copies of a
CNN-sized bucket are negligible).
"""
import torch

from .. import ops


class NodeEmulator:
    def __init__(self, bucket, emulate_node):
        self.bucket = bucket
        self.n = emulate_node
        self.buffers = []

    def store_microbatch(self):
        """Capture the current bucket contents as one emulated rank's
        gradient and clear the bucket for the next micro-batch."""
        self.buffers.append(self.bucket.flat.clone())
        self.bucket.zero_()

    def full(self):
        return len(self.buffers) >= self.n

    def reduce_(self, use_APS=False, grad_exp=5, grad_man=2, use_kahan=False):
        """Local quantized reduction of the buffered micro-batch gradients
        into the bucket (mix.py:254-282 algebra, fused over the flat bucket).
        Leaves the result pre-divided-back (unscaled), ready for the
        cross-rank sum_gradients."""
        assert len(self.buffers) == self.n, \
            f"have {len(self.buffers)} micro-batches, expected {self.n}"
        flat, offsets = self.bucket.flat, self.bucket.offsets
        if self.n == 1:
            flat.copy_(self.buffers[0])
            self.buffers.clear()
            return

        # per-segment max exponent over ALL buffered copies, world factor = N
        max_exp = torch.stack([
            ops.seg_max_exp(b, offsets, self.n, aligned=True)
            for b in self.buffers
        ]).amax(0).contiguous()
        upper = float(2 ** (grad_exp - 1) - 1)
        shifts = upper - max_exp
        # mix.py:268-269: shift 0 when APS off or all-zero sentinel
        zero_mask = max_exp == -100.0
        if not use_APS:
            shifts.zero_()
        else:
            shifts[zero_mask] = 0.0

        res = torch.zeros_like(flat)
        comp = torch.zeros_like(flat) if use_kahan else None
        for b in self.buffers:
            ops.scale_quantize_(b, offsets, shifts, grad_man, grad_exp,
                                aligned=True)
            if use_kahan:
                ops.kahan_qadd_(res, comp, b, grad_man, grad_exp)
            else:
                ops.qadd_(res, b, grad_man, grad_exp)
        flat.copy_(res)
        ops.seg_scale_(flat, offsets, shifts, -1, aligned=True)
        self.buffers.clear()
