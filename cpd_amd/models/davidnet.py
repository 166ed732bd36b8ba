"""DavidNet (DAWNBench CIFAR10 speed-run net) + a dict-DAG graph executor.

Capability parity with the reference's graph-as-nested-dicts network
definition and TorchGraph executor (example/DavidNet/davidnet.py:19-63,
example/DavidNet/utils.py:258-292) — fresh implementation: a network is a
nested dict of modules (or (module, [input paths]) tuples); ``build_graph``
flattens it into topological (name, module, input-names) nodes; ``GraphNet``
executes the DAG caching every node output (loss and correct-count are graph
nodes too), with an optional fp16 mode that keeps BatchNorm in fp32
(utils.py:288-292).
"""
import torch
import torch.nn as nn

__all__ = ["davidnet", "DavidNet", "build_graph", "GraphNet", "net_spec"]

SEP = "/"


class Identity(nn.Module):
    def forward(self, x):
        return x


class Add(nn.Module):
    def forward(self, a, b):
        return a + b


class Flatten(nn.Module):
    def forward(self, x):
        return x.flatten(1)


class Mul(nn.Module):
    def __init__(self, w):
        super().__init__()
        self.w = w

    def forward(self, x):
        return x * self.w


class Correct(nn.Module):
    def forward(self, logits, target):
        return logits.argmax(dim=1) == target


class SumCrossEntropy(nn.Module):
    def __init__(self):
        super().__init__()
        self.ce = nn.CrossEntropyLoss(reduction="sum")

    def forward(self, logits, target):
        return self.ce(logits.float(), target)


def conv_bn(c_in, c_out, bn_weight_init=1.0):
    bn = nn.BatchNorm2d(c_out)
    nn.init.constant_(bn.weight, bn_weight_init)
    return {
        "conv": nn.Conv2d(c_in, c_out, 3, stride=1, padding=1, bias=False),
        "bn": bn,
        "relu": nn.ReLU(True),
    }


def residual(c):
    return {
        "in": Identity(),
        "res1": conv_bn(c, c),
        "res2": conv_bn(c, c),
        "add": (Add(), ["in", f"res2{SEP}relu"]),  # paths relative to here
    }


def net_spec(channels=None, weight=0.125):
    channels = channels or {"prep": 64, "layer1": 128, "layer2": 256,
                            "layer3": 512}
    spec = {
        "prep": conv_bn(3, channels["prep"]),
        "layer1": dict(conv_bn(channels["prep"], channels["layer1"]),
                       pool=nn.MaxPool2d(2)),
        "layer2": dict(conv_bn(channels["layer1"], channels["layer2"]),
                       pool=nn.MaxPool2d(2)),
        "layer3": dict(conv_bn(channels["layer2"], channels["layer3"]),
                       pool=nn.MaxPool2d(2)),
        "classifier": {
            "pool": nn.MaxPool2d(4),
            "flatten": Flatten(),
            "linear": nn.Linear(channels["layer3"], 10, bias=False),
            "logits": Mul(weight),
        },
    }
    spec["layer1"]["residual"] = residual(channels["layer1"])
    spec["layer3"]["residual"] = residual(channels["layer3"])
    return spec


LOSS_NODES = {
    "loss": (SumCrossEntropy(), [f"classifier{SEP}logits", "target"]),
    "correct": (Correct(), [f"classifier{SEP}logits", "target"]),
}


def build_graph(spec, loss_nodes=None):
    """Flatten a nested-dict network spec to [(name, module, [inputs])] in
    definition order; each node's default input is the previous node
    ('input' for the first)."""
    nodes = []

    def walk(d, prefix):
        for key, val in d.items():
            name = SEP.join(prefix + [key])
            if isinstance(val, dict):
                walk(val, prefix + [key])
            elif isinstance(val, tuple):
                mod, inputs = val
                resolved = [
                    i if i in ("input", "target") or SEP.join(prefix + [i]) not
                    in _names else SEP.join(prefix + [i])
                    for i in inputs
                ]
                nodes.append((name, mod, resolved))
                _names.add(name)
            else:
                prev = nodes[-1][0] if nodes else "input"
                nodes.append((name, val, [prev]))
                _names.add(name)

    _names = set()
    walk(spec, [])
    if loss_nodes:
        for key, (mod, inputs) in loss_nodes.items():
            nodes.append((key, mod, list(inputs)))
    return nodes


class GraphNet(nn.Module):
    """DAG executor: forward walks the node list caching every output."""

    def __init__(self, spec, loss_nodes=None):
        super().__init__()
        self.nodes = build_graph(spec, loss_nodes)
        for name, mod, _ in self.nodes:
            self.add_module(name.replace(SEP, "_"), mod)
        self.cache = {}

    def forward(self, inputs):
        # inputs: dict with 'input' (and 'target' when loss nodes exist)
        cache = dict(inputs)
        for name, mod, srcs in self.nodes:
            mod = getattr(self, name.replace(SEP, "_"))
            cache[name] = mod(*[cache[s] for s in srcs])
        self.cache = cache
        return cache

    def half(self):
        """fp16 everywhere except BatchNorm (utils.py:288-292 parity)."""
        for module in self.modules():
            if not isinstance(module, (nn.BatchNorm2d, GraphNet)):
                module.half()
        return self


class DavidNet(GraphNet):
    def __init__(self, with_loss=True):
        super().__init__(net_spec(), LOSS_NODES if with_loss else None)

    def logits(self, x):
        return self.forward({"input": x})[f"classifier{SEP}logits"]


def davidnet(with_loss=True):
    return DavidNet(with_loss=with_loss)
