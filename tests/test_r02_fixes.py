"""Regression tests for the round-1 review findings (VERDICT weak #7,
ADVICE items 1-4)."""
import math

import pytest
import torch

from cpd_amd import ops
from cpd_amd.quant import float_quantize, float_quantize_
from cpd_amd.parallel.ring import _wire_dtype
from cpd_amd.utils.train_util import DistributedSampler


def test_float_quantize_inplace_noncontiguous():
    # round 1: .contiguous() copied, the copy was mutated, caller's tensor
    # unchanged (VERDICT weak #7).  Must now copy back.
    x = torch.randn(64, 64)
    col = x[:, 3]  # stride-64 view, non-contiguous
    expect = float_quantize(col.clone(), 4, 3)
    out = float_quantize_(col, 4, 3)
    assert out is col
    assert torch.equal(col, expect)
    assert torch.equal(x[:, 3], expect)  # base tensor sees the mutation


def test_float_quantize_inplace_contiguous_identity():
    x = torch.randn(1000)
    expect = float_quantize(x.clone(), 5, 2)
    out = float_quantize_(x, 5, 2)
    assert out is x and torch.equal(x, expect)


def test_aps_max_exp_overflow_guard():
    # max|g| finite but max|g|*W overflows fp32: must NOT return the Inf
    # sentinel 129-with-crushed-shift; falls back to
    # ceil_log2(max) + ceil_log2(W)
    big = 3.0e38  # < FLT_MAX, but * 8 overflows
    flat = torch.tensor([big, 1.0, 0.0, 0.0], dtype=torch.float32)
    offsets = torch.tensor([0, 4], dtype=torch.long)
    e = ops.seg_max_exp(flat, offsets, 8)
    expected = math.ceil(math.log2(big)) + 3  # 129 + 3
    assert float(e[0]) == float(expected)
    # non-overflow parity unchanged
    flat2 = torch.tensor([3.0, -5.0], dtype=torch.float32)
    offsets2 = torch.tensor([0, 2], dtype=torch.long)
    e2 = ops.seg_max_exp(flat2, offsets2, 8)
    assert float(e2[0]) == math.ceil(math.log2(5.0 * 8))
    # genuine Inf still reports 129 (propagates, not crushed)
    flat3 = torch.tensor([float("inf"), 1.0], dtype=torch.float32)
    e3 = ops.seg_max_exp(flat3, offsets2, 2)
    assert float(e3[0]) == 129.0


def test_wire_bf16_rejects_wide_mantissa():
    with pytest.raises(ValueError):
        _wire_dtype(torch.zeros(4), 8, "bf16")
    assert _wire_dtype(torch.zeros(4), 7, "bf16") is torch.bfloat16


def test_wire_debug_check(monkeypatch):
    from cpd_amd.parallel.ring import _check_on_grid
    monkeypatch.setenv("CPD_DEBUG_WIRE", "1")
    on_grid = float_quantize(torch.randn(256), 4, 3)
    _check_on_grid(on_grid, torch.bfloat16)  # no raise
    off_grid = torch.full((8,), 1.0000001)
    with pytest.raises(RuntimeError):
        _check_on_grid(off_grid, torch.bfloat16)


def test_fused_bn_batches_tracked_eager_path():
    from cpd_amd.models.fused_bn import FusedBNReLU
    bn = FusedBNReLU(4)
    bn.train()
    x = torch.randn(2, 4, 3, 3)  # CPU + H*W%4 != 0 -> eager path
    bn(x)
    assert int(bn.num_batches_tracked) == 1
    bn.eval()
    bn(x)
    assert int(bn.num_batches_tracked) == 1  # eval does not count


@pytest.mark.parametrize("n,world", [(10, 3), (7, 2), (8, 4), (5, 4)])
def test_distributed_sampler_len_no_roundup(n, world):
    ds = list(range(n))
    for rank in range(world):
        s = DistributedSampler(ds, world_size=world, rank=rank,
                               round_up=False)
        assert len(s) == len(list(iter(s)))
    # round_up=True unchanged: equal ceil-length on every rank
    for rank in range(world):
        s = DistributedSampler(ds, world_size=world, rank=rank, round_up=True)
        assert len(s) == math.ceil(n / world)
        assert len(list(iter(s))) == len(s)


def test_procedural_dataset_learnable_and_deterministic():
    from cpd_amd.data import ProceduralImages
    a = ProceduralImages(256, seed=0)
    b = ProceduralImages(256, seed=0)
    assert torch.equal(a.images, b.images) and torch.equal(a.labels, b.labels)
    val = ProceduralImages(128, seed=1)
    assert not torch.equal(a.images[:128], val.images)  # distinct split
    assert torch.equal(a.templates, val.templates)      # shared concepts

    # a linear probe learns it fast and generalizes to the other split
    torch.manual_seed(0)
    lin = torch.nn.Linear(3 * 32 * 32, 10)
    opt = torch.optim.Adam(lin.parameters(), lr=1e-3)
    X, Y = a.images.flatten(1), a.labels
    for _ in range(150):
        opt.zero_grad()
        loss = torch.nn.functional.cross_entropy(lin(X), Y)
        loss.backward()
        opt.step()
    acc_tr = (lin(X).argmax(1) == Y).float().mean().item()
    acc_va = (lin(val.images.flatten(1)).argmax(1) == val.labels
              ).float().mean().item()
    assert acc_tr > 0.9, acc_tr
    assert acc_va > 0.5, acc_va


def test_draw_curve_svg(tmp_path):
    log = tmp_path / "x.log"
    log.write_text("\n".join(
        f"* All Loss {2.0 - i * 0.1:.4f} Prec@1 {10 + i * 8:.3f} "
        f"Prec@5 {50 + i * 4:.3f}" for i in range(10)))
    import subprocess
    import sys
    svg = tmp_path / "out.svg"
    r = subprocess.run(
        [sys.executable, "tools/draw_curve.py", str(log), "--svg", str(svg)],
        capture_output=True, text=True, cwd="/root/repo")
    assert r.returncode == 0, r.stderr
    assert "polyline" in svg.read_text()
    assert len(r.stdout.splitlines()) == 11  # header + 10 rows


def test_quantize_model_gemms_conversion():
    """BASELINE config 4 wiring: conv/linear swapped for Quant_* with weights
    preserved; (8,23) conversion stays close to the float model (Kahan fp32
    accumulator vs torch's pairwise sums)."""
    import torch.nn as nn
    from cpd_amd.quant import Quant_Conv, Quant_Linear, quantize_model_gemms

    torch.manual_seed(5)
    m = nn.Sequential(
        nn.Conv2d(3, 8, 3, padding=1), nn.ReLU(),
        nn.Conv2d(8, 8, 3, padding=1, groups=8),  # grouped: must NOT convert
        nn.Flatten(), nn.Linear(8 * 8 * 8, 4))
    x = torch.randn(2, 3, 8, 8)
    want = m(x)
    qm = quantize_model_gemms(m, exp=8, man=23)
    assert isinstance(qm[0], Quant_Conv)
    assert isinstance(qm[2], nn.Conv2d)  # grouped conv untouched
    assert isinstance(qm[4], Quant_Linear)
    got = qm(x)
    torch.testing.assert_close(got, want, rtol=1e-5, atol=1e-5)

    # model registry entry builds and steps
    from cpd_amd.models import build_model
    rq = build_model("resnet18_cifar_quant", exp=5, man=2)
    y = rq(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)
    y.sum().backward()
    assert rq.conv1.weight.grad is not None
