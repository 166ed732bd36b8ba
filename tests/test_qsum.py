"""Tests for the quantized accumulation primitives (qadd_/kahan_qadd_) and the
quantized-accumulator GEMM, against slow pure-Python/numpy references."""
import numpy as np
import pytest
import torch

from cpd_amd import ops
from cpd_amd.quant import quant_gemm
from cpd_amd.quant._oracle import cast_fp_oracle


def seq_qsum_oracle(grads, man, exp):
    res = np.zeros_like(grads[0])
    for g in grads:
        res = cast_fp_oracle(res + g, man, exp)
    return res


def kahan_qsum_oracle(grads, man, exp):
    res = np.zeros_like(grads[0])
    c = np.zeros_like(grads[0])
    for g in grads:
        y = cast_fp_oracle(g - c, man, exp)
        t = cast_fp_oracle(res + y, man, exp)
        c = cast_fp_oracle(cast_fp_oracle(t - res, man, exp) - y, man, exp)
        res = t
    return res


@pytest.mark.parametrize("exp,man", [(4, 3), (5, 2), (8, 23)])
def test_qadd_matches_sequential_oracle(exp, man):
    rng = np.random.default_rng(1)
    grads = [rng.standard_normal(4096).astype(np.float32) for _ in range(8)]
    acc = torch.zeros(4096)
    for g in grads:
        ops.qadd_(acc, torch.from_numpy(g), man, exp)
    want = seq_qsum_oracle(grads, man, exp)
    assert (acc.numpy() == want).all()


@pytest.mark.parametrize("exp,man", [(4, 3), (5, 2)])
def test_kahan_qadd_matches_oracle(exp, man):
    rng = np.random.default_rng(2)
    grads = [rng.standard_normal(4096).astype(np.float32) for _ in range(8)]
    acc = torch.zeros(4096)
    comp = torch.zeros(4096)
    for g in grads:
        ops.kahan_qadd_(acc, comp, torch.from_numpy(g), man, exp)
    want = kahan_qsum_oracle(grads, man, exp)
    assert (acc.numpy() == want).all()


def test_kahan_beats_plain_at_low_precision():
    # the reference's motivating property: Kahan summation loses less
    rng = np.random.default_rng(3)
    grads = [rng.standard_normal(8192).astype(np.float32) * 0.1 for _ in range(16)]
    exact = np.sum(np.stack(grads), axis=0, dtype=np.float64)
    plain = torch.zeros(8192)
    kacc, kcomp = torch.zeros(8192), torch.zeros(8192)
    for g in grads:
        ops.qadd_(plain, torch.from_numpy(g), 3, 5)
        ops.kahan_qadd_(kacc, kcomp, torch.from_numpy(g), 3, 5)
    err_plain = np.abs(plain.numpy() - exact).mean()
    err_kahan = np.abs(kacc.numpy() - exact).mean()
    assert err_kahan < err_plain


def quant_gemm_oracle(a, b, man, exp):
    M, K = a.shape
    N = b.shape[1]
    c = np.zeros((M, N), dtype=np.float32)
    comp = np.zeros((M, N), dtype=np.float32)
    for k in range(K):
        prod = cast_fp_oracle(np.outer(a[:, k], b[k, :]), man, exp)
        y = cast_fp_oracle(prod - comp, man, exp)
        t = cast_fp_oracle(c + y, man, exp)
        comp = cast_fp_oracle(cast_fp_oracle(t - c, man, exp) - y, man, exp)
        c = t
    return c


@pytest.mark.parametrize("exp,man", [(8, 23), (5, 10), (4, 3)])
@pytest.mark.parametrize("shape", [(16, 16, 16), (33, 7, 19), (5, 64, 3), (1, 1, 1)])
def test_quant_gemm_matches_oracle(exp, man, shape):
    M, K, N = shape
    rng = np.random.default_rng(M * 100 + K * 10 + N)
    a = rng.standard_normal((M, K)).astype(np.float32)
    b = rng.standard_normal((K, N)).astype(np.float32)
    got = quant_gemm(torch.from_numpy(a), torch.from_numpy(b), man=man, exp=exp).numpy()
    want = quant_gemm_oracle(a, b, man, exp)
    assert (got == want).all(), np.abs(got - want).max()


def test_quant_gemm_fp32_kahan_close_to_mm():
    # (8,23) quantization is identity on normals -> plain fp32 Kahan GEMM,
    # which should be at least as accurate as torch.mm vs a float64 reference
    torch.manual_seed(0)
    a = torch.randn(64, 128)
    b = torch.randn(128, 32)
    got = quant_gemm(a, b, man=23, exp=8)
    ref64 = (a.double() @ b.double()).float()
    assert (got - ref64).abs().max() < 1e-4
    err_kahan = (got.double() - a.double() @ b.double()).abs().max()
    err_mm = ((a @ b).double() - a.double() @ b.double()).abs().max()
    assert err_kahan <= err_mm + 1e-12


def test_seg_ops():
    rng = np.random.default_rng(5)
    sizes = [100, 1, 4096, 37]
    offsets = torch.tensor(np.concatenate([[0], np.cumsum(sizes)]), dtype=torch.int64)
    flat = torch.from_numpy(rng.standard_normal(int(offsets[-1])).astype(np.float32) * 100)
    W = 8
    me = ops.seg_max_exp(flat, offsets, W)
    # oracle per segment
    from cpd_amd.quant._oracle import ceil_log2_oracle
    for s in range(4):
        seg = flat[offsets[s]:offsets[s + 1]].numpy()
        want = ceil_log2_oracle(np.array([np.abs(seg).max() * W]))[0]
        assert me[s].item() == want
    # all-zero segment sentinel
    flat2 = flat.clone()
    flat2[offsets[2]:offsets[3]] = 0
    me2 = ops.seg_max_exp(flat2, offsets, W)
    assert me2[2].item() == -100.0

    # fused scale+quantize == manual
    shifts = torch.tensor([2.0, -1.0, 0.0, 5.0])
    manual = flat.clone()
    for s in range(4):
        seg = manual[offsets[s]:offsets[s + 1]]
        seg.copy_(torch.from_numpy(
            cast_fp_oracle(seg.numpy() * 2.0 ** shifts[s].item(), 3, 4)))
    fused = flat.clone()
    ops.scale_quantize_(fused, offsets, shifts, 3, 4)
    assert torch.equal(fused, manual)

    # unscale
    ops.seg_scale_(fused, offsets, shifts, -1)
    for s in range(4):
        seg = fused[offsets[s]:offsets[s + 1]]
        want = manual[offsets[s]:offsets[s + 1]] * 2.0 ** -shifts[s].item()
        assert torch.equal(seg, want)
