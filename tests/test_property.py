"""Property-based tests (hypothesis): the native cast vs the numpy oracle
over random formats and adversarial bit patterns, and algebraic invariants
of the quantized accumulation."""
import numpy as np
import torch
from hypothesis import given, settings, strategies as st

from cpd_amd import ops
from cpd_amd.quant import float_quantize
from cpd_amd.quant._oracle import cast_fp_oracle

bits_arrays = st.lists(st.integers(0, 2 ** 32 - 1), min_size=1, max_size=512)


@settings(max_examples=200, deadline=None)
@given(exp=st.integers(1, 8), man=st.integers(0, 23), bits=bits_arrays)
def test_cast_matches_oracle_any_format(exp, man, bits):
    x = np.array(bits, dtype=np.uint32).view(np.float32)
    got = float_quantize(torch.from_numpy(x.copy()), exp, man).numpy()
    want = cast_fp_oracle(x, man, exp)
    nan = np.isnan(got) & np.isnan(want)
    assert (got.view(np.uint32) == want.view(np.uint32))[~nan].all()


@settings(max_examples=100, deadline=None)
@given(exp=st.integers(2, 8), man=st.integers(0, 10),
       vals=st.lists(st.floats(-1e4, 1e4, allow_nan=False, width=32),
                     min_size=2, max_size=64))
def test_qadd_sum_invariants(exp, man, vals):
    """Quantized sequential sum: result is always on the target grid
    (idempotent under re-quantize away from the overflow quirk region) and
    symmetric inputs cancel exactly."""
    g = torch.tensor(vals, dtype=torch.float32)
    acc = torch.zeros_like(g)
    ops.qadd_(acc, g, man, exp)
    # Q(0 + g) == Q(g)
    assert torch.equal(acc, float_quantize(g, exp, man))
    # exact cancellation: Q(x + (-x)) == +-0
    acc2 = float_quantize(g, exp, man).clone()
    neg = -acc2
    finite = torch.isfinite(acc2)
    ops.qadd_(acc2, neg, man, exp)
    assert (acc2[finite] == 0).all()


@settings(max_examples=50, deadline=None)
@given(exp=st.integers(2, 8), man=st.integers(0, 7))
def test_bf16_container_exactness(exp, man):
    """Every (exp<=8, man<=7) grid value round-trips bf16 exactly —
    the wire-format invariant."""
    rng = np.random.default_rng(exp * 31 + man)
    x = (rng.standard_normal(4096) *
         10.0 ** float(rng.integers(-9, 9))).astype(np.float32)
    q = float_quantize(torch.from_numpy(x), exp, man)
    rt = q.to(torch.bfloat16).float()
    nan = torch.isnan(q)
    assert torch.equal(q[~nan], rt[~nan])
