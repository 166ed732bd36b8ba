"""Backward-overlapped reducer: bit-parity with the synchronous fused path
(sequential mode) across 2 gloo ranks, and single-process equivalence."""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

PORT = 29812


def _init(rank, world, port):
    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo")


def _build(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(31, 64), torch.nn.ReLU(),
        torch.nn.Linear(64, 64), torch.nn.ReLU(),
        torch.nn.Linear(64, 7))


def _overlap_worker(rank, world, port, q):
    from cpd_amd.parallel import DistModule
    from cpd_amd.trainers.core import LPTrainStep

    _init(rank, world, port)
    crit = torch.nn.CrossEntropyLoss()
    results = {}
    for tag, overlap in (("sync", 0), ("overlap", 3)):
        model = _build()
        dm = DistModule(model)
        opt = torch.optim.SGD([{"params": model.parameters()}], lr=0.05,
                              momentum=0.9)
        step = LPTrainStep(dm, opt, grad_exp=4, grad_man=3, use_APS=True,
                           mode="sequential", overlap=overlap)
        gen = torch.Generator().manual_seed(500 + rank)
        for it in range(3):
            x = torch.randn(16, 31, generator=gen)
            y = torch.randint(0, 7, (16,), generator=gen)
            step.substep(crit(dm(x), y) / step.loss_scale_denom())
        results[tag] = {n: p.detach().clone().numpy()
                        for n, p in model.named_parameters()}
    q.put((rank, results))
    dist.destroy_process_group()


def test_overlap_bitmatches_sync_sequential_2ranks():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_overlap_worker, args=(r, 2, PORT, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = dict(q.get() for _ in range(2))
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    for r in (0, 1):
        for name in res[r]["sync"]:
            assert np.array_equal(res[r]["sync"][name],
                                  res[r]["overlap"][name]), name
    for name in res[0]["sync"]:
        assert np.array_equal(res[0]["overlap"][name],
                              res[1]["overlap"][name]), name


def test_overlap_single_process_matches_sync():
    from cpd_amd.parallel import DistModule
    from cpd_amd.trainers.core import LPTrainStep

    crit = torch.nn.CrossEntropyLoss()
    outs = {}
    for tag, overlap in (("sync", 0), ("overlap", 2)):
        model = _build(7)
        dm = DistModule(model)
        opt = torch.optim.SGD([{"params": model.parameters()}], lr=0.1)
        step = LPTrainStep(dm, opt, grad_exp=4, grad_man=3, use_APS=True,
                           overlap=overlap)
        gen = torch.Generator().manual_seed(1)
        for it in range(3):
            x = torch.randn(8, 31, generator=gen)
            y = torch.randint(0, 7, (8,), generator=gen)
            step.substep(crit(dm(x), y))
        outs[tag] = {n: p.detach().clone() for n, p in
                     model.named_parameters()}
    for name in outs["sync"]:
        assert torch.equal(outs["sync"][name], outs["overlap"][name]), name
