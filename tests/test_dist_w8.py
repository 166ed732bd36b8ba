"""8-GPU readiness hardening — CPU/gloo evidence for the
paths the round-end 8-GPU driver run will exercise:

  * ring all-reduce at W=8 (full node width) vs the rotated sequential oracle
  * overlapped multi-sub-bucket reducer at W=4 (collective-ordering stress)
  * emulate_node=32 (BASELINE config 5 shape: e5m2 + APS, 256-rank-class
    local ring replay)
"""
import os
import sys

import numpy as np
import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cpd_amd.quant._oracle import cast_fp_oracle  # noqa: E402
from tests.test_dist import _grads, _init, _spawn, seq_oracle  # noqa: E402

PORT = 29912


def _ring_worker(rank, world, port, kahan, q):
    from cpd_amd.parallel.ring import ring_lp_all_reduce_
    _init(rank, world, port)
    flat = torch.from_numpy(_grads(world, n=8192 + 40)[rank].copy())
    ring_lp_all_reduce_(flat, 5, 2, use_kahan=kahan)
    q.put((rank, flat.numpy()))
    dist.destroy_process_group()


@pytest.mark.parametrize("kahan", [False, True])
def test_ring_w8_matches_rotated_oracle(kahan):
    """Full node width: 8 ranks, non-divisible length (padded chunks)."""
    world = 8
    port = PORT + int(kahan)
    res = _spawn(_ring_worker, world, port, kahan)
    grads = _grads(world, n=8192 + 40)
    n = grads[0].size
    chunk = (n + world - 1) // world
    padded = chunk * world
    gp = [np.concatenate([g, np.zeros(padded - n, np.float32)])
          for g in grads]
    want = np.empty(padded, np.float32)
    for i in range(world):
        order = [(i + s) % world for s in range(world)]
        sl = slice(i * chunk, (i + 1) * chunk)
        want[sl] = seq_oracle([g[sl] for g in gp], 2, 5, kahan=kahan,
                              order=order)
    for r in range(world):
        assert (res[r] == want[:n]).all(), f"rank {r}"
    assert all((res[r] == res[0]).all() for r in range(world))


def _overlap_worker(rank, world, port, q):
    from cpd_amd.parallel import DistModule
    from cpd_amd.trainers.core import LPTrainStep

    _init(rank, world, port)
    torch.manual_seed(0)
    crit = torch.nn.CrossEntropyLoss()
    results = {}
    for tag, overlap, mode in (("sync_seq", 0, "sequential"),
                               ("ov_seq", 4, "sequential"),
                               ("ov_ring", 4, "ring")):
        torch.manual_seed(0)
        model = torch.nn.Sequential(
            torch.nn.Linear(23, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 64), torch.nn.ReLU(),
            torch.nn.Linear(64, 5))
        dm = DistModule(model)
        opt = torch.optim.SGD([{"params": model.parameters()}], lr=0.05,
                              momentum=0.9)
        step = LPTrainStep(dm, opt, grad_exp=4, grad_man=3, use_APS=True,
                           mode=mode, overlap=overlap)
        gen = torch.Generator().manual_seed(700 + rank)
        for _ in range(3):
            x = torch.randn(16, 23, generator=gen)
            y = torch.randint(0, 5, (16,), generator=gen)
            step.substep(crit(dm(x), y) / step.loss_scale_denom())
        results[tag] = {n: p.detach().clone().numpy()
                        for n, p in model.named_parameters()}
    q.put((rank, results))
    dist.destroy_process_group()


def test_overlap_w4_multibucket():
    """W=4 with 4 sub-buckets over 10 params: the hook-launched collectives
    must issue in a consistent cross-rank order (no deadlock within the join
    timeout), the sequential overlapped result must bit-match the sync
    single-bucket result, and the ring overlapped result must bit-agree
    across all ranks."""
    world = 4
    res = _spawn(_overlap_worker, world, PORT + 10)
    for r in range(world):
        for name in res[r]["sync_seq"]:
            assert np.array_equal(res[r]["sync_seq"][name],
                                  res[r]["ov_seq"][name]), name
    for name in res[0]["ov_ring"]:
        for r in range(1, world):
            assert np.array_equal(res[0]["ov_ring"][name],
                                  res[r]["ov_ring"][name]), name
        assert np.isfinite(res[0]["ov_ring"][name]).all(), name


def test_emulate_node_32_config5_shape():
    """BASELINE config 5: e5m2 grads + APS with emulate_node=32 — the local
    32-way ring replay must equal the 32-term sequential quantized oracle."""
    from cpd_amd.parallel.bucket import GradBucket
    from cpd_amd.parallel.emulate import NodeEmulator

    W = 32
    torch.manual_seed(13)
    model = torch.nn.Sequential(torch.nn.Linear(41, 23),
                                torch.nn.Linear(23, 9))
    bucket = GradBucket(model.parameters())
    em = NodeEmulator(bucket, W)
    total = sum(p.numel() for p in bucket.params)
    grads = _grads(W, n=total, seed=21)
    flats = []
    for w in range(W):
        bucket.zero_()
        off = 0
        for p in bucket.params:
            p.grad.view(-1).copy_(
                torch.from_numpy(grads[w][off:off + p.numel()]))
            off += p.numel()
        flats.append(bucket.flat.clone().numpy())
        em.store_microbatch()
    em.reduce_(use_APS=True, grad_exp=5, grad_man=2, use_kahan=True)

    offsets = bucket.offsets.numpy()
    got = bucket.flat.numpy()
    for s in range(len(offsets) - 1):
        sl = slice(offsets[s], offsets[s + 1])
        segs = [f[sl] for f in flats]
        mx = max(np.abs(g).max() * W for g in segs)
        if mx == 0:
            shift = 0.0
        else:
            m, e = np.frexp(np.float64(mx))
            E = e - 1 if m == 0.5 else e
            shift = (2 ** 4 - 1) - E
        scaled = [cast_fp_oracle(g * np.float32(2.0 ** shift), 2, 5)
                  for g in segs]
        want = seq_oracle(scaled, 2, 5, kahan=True) * np.float32(2.0 ** -shift)
        assert np.array_equal(got[sl], want), f"segment {s}"
