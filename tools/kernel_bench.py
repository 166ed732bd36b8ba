#!/usr/bin/env python3
"""Microbenchmarks for the cpd_amd HIP kernels on MI355X.

Prints one line per kernel with achieved bandwidth / FLOP rate.  Run on a GPU
box: gpurun -- python tools/kernel_bench.py | tee gpurun_out/kernel_bench.txt
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from cpd_amd import ops  # noqa: E402


def timeit(fn, reps=20, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    times = []
    for _ in range(reps):
        t0 = time.perf_counter()
        fn()
        torch.cuda.synchronize()
        times.append(time.perf_counter() - t0)
    times.sort()
    return times[len(times) // 2]


def report(name, seconds, bytes_moved=None, flops=None):
    msg = f"{name:42s} {seconds * 1e6:10.1f} us"
    if bytes_moved:
        msg += f"  {bytes_moved / seconds / 1e12:8.3f} TB/s"
    if flops:
        msg += f"  {flops / seconds / 1e12:8.2f} TF/s"
    print(msg, flush=True)


def main():
    assert torch.cuda.is_available()
    dev = torch.device("cuda:0")
    n = 64 * 1024 * 1024
    x = torch.randn(n, device=dev)
    y = torch.randn(n, device=dev)
    c = torch.zeros(n, device=dev)

    t = timeit(lambda: ops.quantize_(x, 3, 4))
    report(f"quantize_ e4m3 n={n}", t, bytes_moved=8 * n)
    t = timeit(lambda: ops.quantize(x, 3, 4))
    report("quantize (oop)", t, bytes_moved=8 * n)
    t = timeit(lambda: ops.qadd_(x, y, 3, 4))
    report("qadd_ f32", t, bytes_moved=12 * n)
    t = timeit(lambda: ops.kahan_qadd_(x, c, y, 3, 4))
    report("kahan_qadd_ f32", t, bytes_moved=20 * n)

    xb = x.to(torch.bfloat16)
    yb = y.to(torch.bfloat16)
    t = timeit(lambda: ops.hip_ext().qadd_bf16_(xb, yb, 3, 4))
    report("qadd_bf16_", t, bytes_moved=6 * n)

    # bucket-shaped segmented ops: 64 segments over 12M elements
    nb = 12 * 1024 * 1024
    flat = torch.randn(nb, device=dev)
    bounds = sorted({(i * nb // 64) // 1024 * 1024 for i in range(64)} | {nb})
    offsets = torch.tensor([0] + list(bounds), dtype=torch.int64,
                           device=dev).unique()
    S = offsets.numel() - 1
    shifts = torch.zeros(S, device=dev)
    t = timeit(lambda: ops.seg_max_exp(flat, offsets, 8, aligned=True))
    report(f"seg_max_exp aligned (S={S}, n={nb})", t, bytes_moved=4 * nb)
    t = timeit(lambda: ops.seg_max_exp(flat, offsets, 8, aligned=False))
    report("seg_max_exp generic", t, bytes_moved=4 * nb)
    t = timeit(lambda: ops.scale_quantize_(flat, offsets, shifts, 3, 4,
                                           aligned=True))
    report("scale_quantize_ aligned", t, bytes_moved=8 * nb)
    t = timeit(lambda: ops.scale_quantize_(flat, offsets, shifts, 3, 4,
                                           aligned=False))
    report("scale_quantize_ generic", t, bytes_moved=8 * nb)
    t = timeit(lambda: ops.seg_scale_(flat, offsets, shifts, -1, aligned=True))
    report("seg_scale_ aligned", t, bytes_moved=8 * nb)

    # fused BN fwd+bwd vs eager composition, both layouts, ResNet18/CIFAR
    # layer shapes (b512)
    from cpd_amd.models.fused_bn import FusedBNReLU

    def bn_step(m, x):
        y = m(x)
        y.backward(gy[x.shape])
        x.grad = None

    for (N, C, H, W) in [(512, 64, 32, 32), (512, 128, 16, 16),
                         (512, 256, 8, 8), (512, 512, 4, 4)]:
        gy = {}
        for fmt, tag in ((torch.contiguous_format, "nchw"),
                         (torch.channels_last, "nhwc")):
            x = torch.randn(N, C, H, W, device=dev).clone(memory_format=fmt)
            x.requires_grad_(True)
            gy[x.shape] = torch.randn(N, C, H, W, device=dev).clone(
                memory_format=fmt)
            m = FusedBNReLU(C).cuda().train()
            t = timeit(lambda: bn_step(m, x), reps=10)
            report(f"fusedBN {tag} {N}x{C}x{H}x{W} fwd+bwd", t,
                   bytes_moved=(5 + 2) * N * C * H * W * 4)
            e = torch.nn.Sequential(torch.nn.BatchNorm2d(C),
                                    torch.nn.ReLU()).cuda().train()
            t = timeit(lambda: bn_step(e, x), reps=10)
            report(f"eagerBN {tag} {N}x{C}x{H}x{W} fwd+bwd", t,
                   bytes_moved=(5 + 2) * N * C * H * W * 4)

    # GEMMs
    for sz in (2048, 4096):
        a = torch.randn(sz, sz, device=dev)
        b = torch.randn(sz, sz, device=dev)
        t = timeit(lambda: ops.hip_ext().gemm_f32(a, b), reps=10)
        report(f"gemm_f32 MFMA {sz}^3", t, flops=2 * sz ** 3)
        t = timeit(lambda: a @ b, reps=10)
        report(f"torch.mm (rocBLAS) {sz}^3", t, flops=2 * sz ** 3)

    for sz in (512, 1024, 2048):
        a = torch.randn(sz, sz, device=dev)
        b = torch.randn(sz, sz, device=dev)
        t = timeit(lambda: ops.hip_ext().quant_gemm(a, b, 3, 4), reps=5)
        report(f"quant_gemm e4m3 {sz}^3", t, flops=2 * sz ** 3)
        t = timeit(lambda: ops.hip_ext().quant_gemm(a, b, 2, 5), reps=5)
        report(f"quant_gemm e5m2 {sz}^3", t, flops=2 * sz ** 3)


if __name__ == "__main__":
    main()
