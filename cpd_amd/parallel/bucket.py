"""Flat gradient bucket: the data layout behind the fused gradient pipeline.

The reference issues one collective and one quantize launch *per parameter*
(dist_util.py:54-89) plus a host sync per parameter for the APS exponent scan
(dist_util.py:33, mix.py:264).  Here all gradients live in ONE flat fp32
buffer (param.grad tensors are views into it, so backward accumulates into
the bucket for free), segment boundaries are aligned so the fused segmented
kernels (seg_max_exp / scale_quantize_ / seg_scale_) make a single pass, and
the whole bucket is reduced with one collective sequence.  Sized for MI355X:
288 GB HBM makes one big bucket per model the right default.
"""
import torch

ALIGN = 1024  # elements; keeps segments 4 KiB-aligned for clean block mapping


class GradBucket:
    def __init__(self, params, device=None):
        self.params = [p for p in params if p.requires_grad]
        assert self.params, "GradBucket needs at least one parameter"
        device = device or self.params[0].device
        starts, total = [], 0
        for p in self.params:
            starts.append(total)
            total += (p.numel() + ALIGN - 1) // ALIGN * ALIGN
        # pad total so any world size <= 8 divides it (ring chunking)
        total = (total + 8 * ALIGN - 1) // (8 * ALIGN) * (8 * ALIGN)
        self.flat = torch.zeros(total, dtype=torch.float32, device=device)
        self.starts = starts
        # segment s spans [starts[s], starts[s+1]) including its zero pad
        self.offsets = torch.tensor(starts + [total], dtype=torch.int64,
                                    device=device)
        self.attach()

    @staticmethod
    def _grad_view(flat_slice, p):
        # Match the parameter's memory format so autograd accumulates into
        # the bucket without a strided elementwise path (channels_last conv
        # weights produce channels_last grads).
        if (p.dim() == 4 and not p.is_contiguous()
                and p.is_contiguous(memory_format=torch.channels_last)):
            n, c, h, w = p.shape
            return flat_slice.as_strided((n, c, h, w),
                                         (c * h * w, 1, w * c, c))
        return flat_slice.view_as(p)

    def attach(self):
        """(Re)point every param.grad at its view of the flat buffer."""
        for p, st in zip(self.params, self.starts):
            view = self._grad_view(self.flat[st:st + p.numel()], p)
            if p.grad is None or p.grad.data_ptr() != view.data_ptr():
                if p.grad is not None:
                    view.copy_(p.grad.detach())
                p.grad = view

    def check_attached(self):
        """Detect an optimizer having replaced grads (e.g. zero_grad(set_to_none=True));
        re-copy any stray grad back into the bucket."""
        for p, st in zip(self.params, self.starts):
            if p.grad is None or p.grad.data_ptr() != self.flat.data_ptr() + st * 4:
                self.attach()
                return

    def zero_(self):
        self.flat.zero_()

    def numel(self):
        return self.flat.numel()
