"""Model-zoo sanity: shapes, parameter counts, registry, graph executor."""
import torch

from cpd_amd.models import REGISTRY, build_model, davidnet, res_cifar, resnet50


def test_registry_names():
    assert {"res_cifar", "resnet18_cifar", "resnet50", "davidnet"} <= \
        set(REGISTRY)


def test_res_cifar_shapes_and_params():
    m = res_cifar()
    y = m(torch.randn(2, 3, 32, 32))
    assert y.shape == (2, 10)
    n = sum(p.numel() for p in m.parameters())
    assert 11_100_000 < n < 11_300_000  # CIFAR ResNet-18 class
    # dummy-rank calling convention (reference resnet18_cifar.py:73)
    y2 = m(torch.randn(2, 3, 32, 32), 0)
    assert y2.shape == (2, 10)


def test_resnet50_param_count():
    m = resnet50()
    n = sum(p.numel() for p in m.parameters())
    assert n == 25_557_032  # torchvision resnet50 parameter count


def test_build_model_kwargs():
    m = build_model("resnet18_cifar", num_classes=100)
    assert m(torch.randn(1, 3, 32, 32)).shape == (1, 100)
    m2 = build_model("resnet50", fused_bn=False, num_classes=10)
    assert m2(torch.randn(1, 3, 64, 64)).shape == (1, 10)


def test_davidnet_graph_caches_all_nodes():
    m = davidnet()
    out = m({"input": torch.randn(2, 3, 32, 32),
             "target": torch.randint(0, 10, (2,))})
    assert "loss" in out and "correct" in out
    assert out["classifier/logits"].shape == (2, 10)
    # residual adds present in both res layers
    assert "layer1/residual/add" in out and "layer3/residual/add" in out
    # fp16 mode keeps BN fp32
    mh = davidnet().half()
    bn_types = [p.dtype for n, p in mh.named_parameters() if "bn" in n]
    assert all(t == torch.float32 for t in bn_types)
    conv_types = [p.dtype for n, p in mh.named_parameters() if "conv" in n]
    assert all(t == torch.float16 for t in conv_types)


def test_draw_curve_parser(tmp_path):
    from tools.draw_curve import parse_log

    log = tmp_path / "aps.log"
    log.write_text("noise\n* All Loss 0.5432 Prec@1 91.230 Prec@5 99.500\n"
                   "* All Loss 0.4000 Prec@1 92.000 Prec@5 99.600\n")
    rows = parse_log(str(log))
    assert rows == [(0.5432, 91.23, 99.5), (0.4, 92.0, 99.6)]
