#!/usr/bin/env python3
"""Tiny driver that exercises each hot elementwise/segmented kernel a few
times for a rocprofv3 --pmc counter run."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from cpd_amd import ops  # noqa: E402

n = 8 * 1024 * 1024
x = torch.randn(n, device="cuda")
y = torch.randn(n, device="cuda")
c = torch.zeros(n, device="cuda")
offsets = torch.arange(0, n + 1, n // 64, dtype=torch.int64, device="cuda")
shifts = torch.zeros(64, device="cuda")
for _ in range(2):
    ops.quantize_(x, 3, 4)
    ops.qadd_(x, y, 3, 4)
    ops.kahan_qadd_(x, c, y, 3, 4)
    ops.seg_max_exp(x, offsets, 8, aligned=True)
    ops.scale_quantize_(x, offsets, shifts, 3, 4, aligned=True)
    ops.hip_ext().quant_gemm(x[:512 * 512].view(512, 512),
                             y[:512 * 512].view(512, 512), 3, 4)
torch.cuda.synchronize()
print("done")
