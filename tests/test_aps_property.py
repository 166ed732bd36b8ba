"""The framework's reason to exist: APS pre-scaling recovers training that
naive low-precision gradient summation destroys (README.md:153-154 claim,
demonstrated deterministically on CPU), plus bf16-wire ring parity."""
import os
import sys

import numpy as np
import torch
import torch.distributed as dist
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

from cpd_amd.parallel import DistModule  # noqa: E402
from cpd_amd.trainers.core import LPTrainStep  # noqa: E402

PORT = 29912


def _train(use_APS, lr=0.5, steps=60, scale=1e-4):
    """Linear regression with tiny gradients: e4m3 underflows raw grads to
    zero; APS's power-of-two pre-scale keeps them alive."""
    torch.manual_seed(0)
    model = torch.nn.Linear(8, 1, bias=False)
    torch.nn.init.zeros_(model.weight)
    dm = DistModule(model)
    opt = torch.optim.SGD([{"params": model.parameters()}], lr=lr)
    step = LPTrainStep(dm, opt, grad_exp=4, grad_man=3, use_APS=use_APS,
                       use_master=False)
    g = torch.Generator().manual_seed(1)
    x = torch.randn(256, 8, generator=g)
    w_true = torch.randn(8, 1, generator=g) * scale
    y = x @ w_true
    for _ in range(steps):
        loss = torch.nn.functional.mse_loss(dm(x), y)
        step.substep(loss)
    final = torch.nn.functional.mse_loss(dm(x), y).item()
    init = torch.nn.functional.mse_loss(torch.zeros_like(y), y).item()
    return final, init


def test_aps_recovers_underflowed_gradients():
    final_aps, init = _train(use_APS=True)
    final_no, _ = _train(use_APS=False)
    # without APS the e4m3 grid flushes the ~1e-8 gradients to zero: no
    # progress; with APS the loss drops by orders of magnitude
    assert final_aps < init * 1e-2, (final_aps, init)
    assert final_no > init * 0.5, (final_no, init)


def _bf16_worker(rank, world, port, q):
    from cpd_amd.parallel.ring import ring_lp_all_reduce_
    from cpd_amd.quant import float_quantize

    os.environ.update(MASTER_ADDR="127.0.0.1", MASTER_PORT=str(port),
                      RANK=str(rank), WORLD_SIZE=str(world))
    dist.init_process_group("gloo")
    rng = np.random.default_rng(21)
    grads = [rng.standard_normal(2048).astype(np.float32) for _ in range(world)]
    # on-grid inputs (as after the APS scale+quantize pass)
    base = float_quantize(torch.from_numpy(grads[rank].copy()), 4, 3)
    f32 = base.clone()
    ring_lp_all_reduce_(f32, 4, 3, wire="f32")
    b16 = base.clone()
    ring_lp_all_reduce_(b16, 4, 3, wire="bf16")
    q.put((rank, f32.numpy(), b16.numpy()))
    dist.destroy_process_group()


def test_bf16_wire_ring_parity_gloo():
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_bf16_worker, args=(r, 2, PORT, q))
             for r in range(2)]
    for p in procs:
        p.start()
    res = {}
    for _ in range(2):
        r, f32, b16 = q.get()
        res[r] = (f32, b16)
    for p in procs:
        p.join(60)
        assert p.exitcode == 0
    for r in res:
        assert np.array_equal(res[r][0], res[r][1]), \
            "bf16 wire must be exact for on-grid e4m3 values"
