#!/usr/bin/env python3
"""Collect gpurun_out/bm_*.log JSON lines into profiles/bench_matrix_r02.md."""
import glob
import json
import sys

ORDER = ["flagship", "e5m2", "seqmode", "kahan", "noaps", "emu8", "emu32",
         "rn50", "rn18q_cfg4", "rn50q_cfg4"]
LABEL = {
    "flagship": "ResNet18 b512 e4m3+APS, ring (flagship, defaults)",
    "e5m2": "ResNet18 b512 e5m2+APS, ring",
    "seqmode": "ResNet18 b512 e4m3+APS, sequential-emulation mode",
    "kahan": "ResNet18 b512 e4m3+APS+Kahan, ring",
    "noaps": "ResNet18 b512 e4m3 no-APS, ring",
    "emu8": "ResNet18 b512 e4m3+APS, emulate_node=8 (1-GPU ring replay)",
    "emu32": "ResNet18 b512 e5m2+APS+Kahan, emulate_node=32 (config 5)",
    "rn50": "ResNet50 b128 (224²) e5m2+APS+Kahan, ring",
    "rn18q_cfg4": "ResNet18-CIFAR **Quant_Conv** b64 (e5m2 GEMM accumulator"
                  " in-model, config 4)",
    "rn50q_cfg4": "ResNet50 **Quant_Conv** b8 (e5m2 GEMM accumulator"
                  " in-model, config 4)",
}

rows = []
for name in ORDER:
    files = glob.glob(f"gpurun_out/bm_{name}.log")
    if not files:
        continue
    line = None
    for ln in open(files[0]):
        if ln.startswith('{"metric"'):
            line = json.loads(ln)
    if line is None:
        rows.append((name, None))
        continue
    rows.append((name, line))

print("# Bench config matrix — round 2, 1x MI355X (driver-style invocations)")
print()
print("| config | img/s (whole job) | ms/step | hipGraph |")
print("|---|---|---|---|")
for name, r in rows:
    if r is None:
        print(f"| {LABEL[name]} | run failed | — | — |")
        continue
    g = r.get("config", {}).get("hip_graph", False)
    print(f"| {LABEL[name]} | {r['value']:.0f} | {r['ms_per_step']:.2f}"
          f" | {'on' if g else 'off'} |")
print()
print("Context: the reference publishes no throughput; its 8x V100 ResNet50")
print("emulation implies <= ~1070 img/s whole-node (BASELINE.md).")
