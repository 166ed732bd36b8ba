"""Edge-case coverage: GradBucket layout invariants, SLURM nodelist parsing,
wire-dtype selection, emulate buffering."""
import torch

from cpd_amd.parallel.bucket import ALIGN, GradBucket
from cpd_amd.parallel.dist import _slurm_master
from cpd_amd.parallel.ring import _wire_dtype


def test_bucket_layout_invariants():
    model = torch.nn.Sequential(
        torch.nn.Linear(100, 50), torch.nn.Linear(50, 3))
    model[1].bias.requires_grad_(False)  # frozen param excluded
    b = GradBucket(model.parameters())
    assert len(b.params) == 3
    offsets = b.offsets.tolist()
    assert all(o % ALIGN == 0 for o in offsets)
    assert b.flat.numel() % (8 * ALIGN) == 0  # any W <= 8 divides it
    # grads are views into flat and padding stays zero
    for p, st in zip(b.params, b.starts):
        assert p.grad.data_ptr() == b.flat.data_ptr() + st * 4
    x = torch.randn(4, 100)
    model(x).sum().backward()
    for p, st in zip(b.params, b.starts):
        lo, hi = st + p.numel(), None
    # padding regions (between param end and next segment start) remain zero
    for (p, st), nxt in zip(zip(b.params, b.starts),
                            offsets[1:]):
        pad = b.flat[st + p.numel():nxt]
        assert (pad == 0).all()
    # zero_() keeps views attached
    b.zero_()
    assert all(p.grad.abs().sum() == 0 for p in b.params)
    b.check_attached()
    for p, st in zip(b.params, b.starts):
        assert p.grad.data_ptr() == b.flat.data_ptr() + st * 4


def test_bucket_reattach_after_set_to_none():
    model = torch.nn.Linear(10, 10)
    b = GradBucket(model.parameters())
    model(torch.randn(2, 10)).sum().backward()
    model.zero_grad(set_to_none=True)  # hostile optimizer behavior
    model(torch.randn(2, 10)).sum().backward()
    b.check_attached()  # re-copies stray grads back into the bucket
    for p, st in zip(b.params, b.starts):
        assert p.grad.data_ptr() == b.flat.data_ptr() + st * 4
        view = b.flat[st:st + p.numel()].view_as(p)
        assert torch.equal(view, p.grad)


def test_slurm_master_parse():
    assert _slurm_master("node[3-7,9]") == "node3"
    assert _slurm_master("node[12,14]") == "node12"
    assert _slurm_master("gpu-a,gpu-b") == "gpu-a"
    assert _slurm_master("single") == "single"
    assert _slurm_master("host[5]") == "host5"


def test_wire_dtype_defaults():
    x = torch.zeros(8)
    assert _wire_dtype(x, 3, None) == torch.float32   # exactness default
    assert _wire_dtype(x, 3, "bf16") == torch.bfloat16
    assert _wire_dtype(x, 23, "f32") == torch.float32


def test_node_emulator_buffer_lifecycle():
    from cpd_amd.parallel.emulate import NodeEmulator

    model = torch.nn.Linear(8, 2)
    b = GradBucket(model.parameters())
    em = NodeEmulator(b, 2)
    for i in range(2):
        model(torch.randn(2, 8)).sum().backward()
        em.store_microbatch()
        assert b.flat.abs().sum() == 0  # cleared after capture
    assert em.full()
    em.reduce_(use_APS=True, grad_exp=4, grad_man=3)
    assert not em.buffers  # buffers released
    assert b.flat.abs().sum() > 0   # combined gradient landed
