"""Datasets: synthetic (benchmarks; no network in the target environment) and
a CIFAR-10 python/binary loader with the standard pad-crop/flip/cutout
augmentations (capability parity with the reference's numpy pipeline,
example/DavidNet/utils.py:69-145, fresh implementation)."""
import os
import pickle

import numpy as np
import torch
from torch.utils.data import Dataset

CIFAR_MEAN = np.array([125.31, 122.95, 113.87], dtype=np.float32)
CIFAR_STD = np.array([62.99, 62.09, 66.70], dtype=np.float32)


class SyntheticImages(Dataset):
    """Deterministic random images + labels of a given shape (shared seed so
    every rank regenerates the same data)."""

    def __init__(self, n=50000, shape=(3, 32, 32), num_classes=10, seed=0):
        g = torch.Generator().manual_seed(seed)
        self.images = torch.randn((min(n, 2048),) + tuple(shape), generator=g)
        self.labels = torch.randint(0, num_classes, (min(n, 2048),),
                                    generator=g)
        self.n = n

    def __len__(self):
        return self.n

    def __getitem__(self, i):
        j = i % self.images.shape[0]
        return self.images[j], self.labels[j]


class ProceduralImages(Dataset):
    """Deterministic LEARNABLE classification set (no network in the target
    environment, so the APS-recovers-accuracy experiment — the reference's
    core claim, README.md:153-154 — runs on procedurally generated data).

    Each class owns `templates_per_class` fixed random templates; a sample is
    a randomly scaled template plus unit Gaussian noise.  Fully determined by
    (seed, index); train/val splits use different seeds but share templates,
    so a model that learns the templates generalizes to the val split."""

    def __init__(self, n=16384, shape=(3, 32, 32), num_classes=10, seed=0,
                 templates_per_class=4, snr=0.5):
        g = torch.Generator().manual_seed(777)  # templates shared by splits
        # LOW-FREQUENCY templates (random 4x4 upsampled bilinearly): a conv
        # net with pooling can learn these; full-bandwidth white-noise
        # templates are only learnable by a global matched filter (a linear
        # probe solves them, conv+avgpool architectures cannot)
        lowres = torch.randn(
            (num_classes * templates_per_class, shape[0], 4, 4), generator=g)
        up = torch.nn.functional.interpolate(
            lowres, size=shape[1:], mode="bilinear", align_corners=False)
        up = up / up.std(dim=(1, 2, 3), keepdim=True).clamp_min(1e-6)
        self.templates = up.view((num_classes, templates_per_class)
                                 + tuple(shape))
        gs = torch.Generator().manual_seed(seed)
        self.labels = torch.randint(0, num_classes, (n,), generator=gs)
        v = torch.randint(0, templates_per_class, (n,), generator=gs)
        s = 0.5 + torch.rand((n, 1, 1, 1), generator=gs)
        noise = torch.randn((n,) + tuple(shape), generator=gs)
        self.images = snr * s * self.templates[self.labels, v] + noise

    def __len__(self):
        return self.labels.numel()

    def __getitem__(self, i):
        return self.images[i], int(self.labels[i])


def _load_cifar_batches(root):
    files = [f"data_batch_{i}" for i in range(1, 6)]
    xs, ys = [], []
    for f in files:
        with open(os.path.join(root, f), "rb") as fh:
            d = pickle.load(fh, encoding="bytes")
        xs.append(d[b"data"])
        ys.extend(d[b"labels"])
    x = np.concatenate(xs).reshape(-1, 3, 32, 32).astype(np.float32)
    return x, np.array(ys, dtype=np.int64)


class CIFAR10(Dataset):
    """CIFAR-10 from the standard python pickle batches on local disk."""

    def __init__(self, root, train=True, augment=True, pad=4, cutout=0,
                 seed=0):
        if train:
            x, y = _load_cifar_batches(root)
        else:
            with open(os.path.join(root, "test_batch"), "rb") as fh:
                d = pickle.load(fh, encoding="bytes")
            x = d[b"data"].reshape(-1, 3, 32, 32).astype(np.float32)
            y = np.array(d[b"labels"], dtype=np.int64)
        x = (x - CIFAR_MEAN[None, :, None, None]) / CIFAR_STD[None, :, None, None]
        if train and augment and pad:
            x = np.pad(x, ((0, 0), (0, 0), (pad, pad), (pad, pad)),
                       mode="reflect")
        self.x = x
        self.y = y
        self.train = train
        self.augment = augment and train
        self.pad = pad
        self.cutout = cutout
        self.rng = np.random.default_rng(seed)

    def __len__(self):
        return len(self.y)

    def __getitem__(self, i):
        img = self.x[i]
        if self.augment:
            p = self.pad
            dy, dx = self.rng.integers(0, 2 * p + 1, 2)
            img = img[:, dy:dy + 32, dx:dx + 32]
            if self.rng.random() < 0.5:
                img = img[:, :, ::-1]
            if self.cutout:
                c = self.cutout
                cy = int(self.rng.integers(0, 32 - c + 1))
                cx = int(self.rng.integers(0, 32 - c + 1))
                img = img.copy()
                img[:, cy:cy + c, cx:cx + c] = 0.0
        return torch.from_numpy(np.ascontiguousarray(img)), int(self.y[i])
