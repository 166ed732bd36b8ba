"""Unit tests for the shared training utilities (meters, schedulers,
samplers, checkpoint round-trip with module-prefix fix-up, LARS)."""
import os
import tempfile

import torch

from cpd_amd.utils import (AverageMeter, DistributedGivenIterationSampler,
                           DistributedSampler, GivenIterationSampler,
                           IterLRScheduler, LARS, accuracy, load_state,
                           save_checkpoint)


def test_average_meter_window():
    m = AverageMeter(3)
    for v in [1.0, 2.0, 3.0, 4.0]:
        m.update(v)
    assert m.val == 4.0
    assert m.avg == (2.0 + 3.0 + 4.0) / 3
    c = AverageMeter(0)
    for v in [1.0, 2.0, 3.0]:
        c.update(v)
    assert c.avg == 2.0


def test_accuracy_topk():
    out = torch.tensor([[0.1, 0.9, 0.0], [0.8, 0.1, 0.1], [0.2, 0.3, 0.5]])
    target = torch.tensor([1, 1, 2])
    top1, top2 = accuracy(out, target, topk=(1, 2))
    assert abs(float(top1) - 200.0 / 3) < 1e-4
    assert abs(float(top2) - 100.0) < 1e-4


def test_iter_lr_scheduler():
    opt = torch.optim.SGD([torch.nn.Parameter(torch.zeros(1))], lr=1.0)
    sched = IterLRScheduler(opt, milestones=[3, 5], lr_mults=[0.1, 0.5])
    lrs = []
    for _ in range(6):
        sched.step()
        lrs.append(opt.param_groups[0]["lr"])
    assert lrs == [1.0, 1.0, 1.0, 0.1, 0.1, 0.05]


def test_given_iteration_sampler_coverage():
    data = list(range(100))
    s = GivenIterationSampler(data, total_iter=5, batch_size=10, seed=0)
    idx = list(iter(s))
    assert len(idx) == 50
    assert all(0 <= i < 100 for i in idx)
    # resume offset skips consumed batches
    s2 = GivenIterationSampler(data, total_iter=5, batch_size=10, last_iter=2,
                               seed=0)
    assert list(iter(s2)) == idx[30:]


def test_distributed_given_iteration_sampler_disjoint_shards():
    data = list(range(64))
    shards = [list(iter(DistributedGivenIterationSampler(
        data, total_iter=4, batch_size=4, world_size=2, rank=r, seed=1)))
        for r in range(2)]
    assert len(shards[0]) == len(shards[1]) == 16
    # both ranks drew from one global sequence: their concatenation equals
    # the first 32 entries of the seeded global order
    s_all = DistributedGivenIterationSampler(data, total_iter=8, batch_size=4,
                                             world_size=1, rank=0, seed=1)
    assert shards[0] + shards[1] == list(iter(s_all))


def test_distributed_sampler_partitions():
    data = list(range(10))
    a = list(iter(DistributedSampler(data, world_size=2, rank=0)))
    b = list(iter(DistributedSampler(data, world_size=2, rank=1)))
    assert len(a) == len(b) == 5
    assert sorted(a + b) == sorted(range(10))


def test_checkpoint_roundtrip_with_prefix_fixup():
    model = torch.nn.Sequential(torch.nn.Linear(4, 4))
    wrapped = torch.nn.Sequential()
    wrapped.add_module("module", model[0])  # keys become "module.*"
    with tempfile.TemporaryDirectory() as d:
        path = os.path.join(d, "ck")
        save_checkpoint({"state_dict":
                         {"module." + k if not k.startswith("module.") else k: v
                          for k, v in model.state_dict().items()},
                         "step": 7, "best_prec1": 12.5,
                         "optimizer": {}}, True, path)
        assert os.path.exists(path + ".pth.tar")
        assert os.path.exists(path + "_best.pth.tar")
        fresh = torch.nn.Sequential(torch.nn.Linear(4, 4))
        load_state(path + ".pth.tar", fresh)  # strips "module." prefix
        for (k1, v1), (k2, v2) in zip(fresh.state_dict().items(),
                                      model.state_dict().items()):
            assert torch.equal(v1, v2), k1


def test_lars_update_rule():
    torch.manual_seed(0)
    p = torch.nn.Parameter(torch.randn(10))
    g = torch.randn(10)
    p.grad = g.clone()
    w0 = p.detach().clone()
    lr, mom, wd, tc = 0.5, 0.9, 1e-4, 0.001
    opt = LARS([p], lr=lr, momentum=mom, weight_decay=wd,
               trust_coefficient=tc)
    opt.step()
    local_lr = tc * w0.norm(2) / (g.norm(2) + wd * w0.norm(2))
    buf = lr * local_lr * (g + wd * w0)
    torch.testing.assert_close(p.detach(), w0 - buf)
    # second step applies momentum
    p.grad = g.clone()
    w1 = p.detach().clone()
    opt.step()
    local_lr2 = tc * w1.norm(2) / (g.norm(2) + wd * w1.norm(2))
    buf2 = mom * buf + lr * local_lr2 * (g + wd * w1)
    torch.testing.assert_close(p.detach(), w1 - buf2)
