#!/usr/bin/env python3
"""Segmented-kernel microbench: aligned fast path vs the generic
window-based path on aligned AND deliberately-odd segment layouts."""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from cpd_amd import ops  # noqa: E402


def t(fn, reps=20):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(reps):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / reps


def main():
    nb = 12 * 1024 * 1024
    flat = torch.randn(nb, device="cuda")
    bounds = sorted({(i * nb // 64) // 1024 * 1024 for i in range(64)} | {nb})
    offsets = torch.tensor([0] + list(bounds), dtype=torch.int64,
                           device="cuda").unique()
    b2 = sorted({(i * nb // 64) + 13 for i in range(1, 64)} | {0, nb})
    off2 = torch.tensor(b2, dtype=torch.int64, device="cuda")
    for tag, off, al in (("aligned-layout fast path", offsets, True),
                         ("aligned-layout generic", offsets, False),
                         ("odd-layout generic", off2, False)):
        s = t(lambda: ops.seg_max_exp(flat, off, 8, aligned=al))
        print(f"seg_max_exp     {tag}: {s * 1e6:7.1f} us  "
              f"{4 * nb / s / 1e12:.3f} TB/s")
        z = torch.zeros(off.numel() - 1, device="cuda")
        sq = t(lambda: ops.scale_quantize_(flat, off, z, 3, 4, aligned=al))
        print(f"scale_quantize_ {tag}: {sq * 1e6:7.1f} us  "
              f"{8 * nb / sq / 1e12:.3f} TB/s")
    a = ops.seg_max_exp(flat, offsets, 8, aligned=True)
    g = ops.seg_max_exp(flat, offsets, 8, aligned=False)
    print("aligned==generic:", bool(torch.equal(a, g)))


if __name__ == "__main__":
    main()
