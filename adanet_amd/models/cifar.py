"""CIFAR data providers + image preprocessing.

Reference: research/improve_nas/trainer/{cifar10.py:38-152, cifar100.py,
image_processing.py:37-66 (basic augment + cutout), fake_data.py:26
(FakeImageProvider)}. No network exists on these boxes, so the real-data
path accepts pre-downloaded numpy arrays and the default is the synthetic
provider (the reference's own fallback: keras cifar10.load_data with an
offline dummy fallback, cifar10.py:80-104).
"""

from __future__ import annotations

from typing import Iterator, Optional, Tuple

import torch


def random_crop(x: torch.Tensor, pad: int = 4,
                generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Pad+random-crop augmentation (reference image_processing.py:37-50)."""
    B, C, H, W = x.shape
    padded = torch.nn.functional.pad(x, (pad, pad, pad, pad))
    out = torch.empty_like(x)
    ys = torch.randint(0, 2 * pad + 1, (B,), generator=generator)
    xs = torch.randint(0, 2 * pad + 1, (B,), generator=generator)
    for b in range(B):
        out[b] = padded[b, :, ys[b]:ys[b] + H, xs[b]:xs[b] + W]
    return out


def random_flip(x: torch.Tensor,
                generator: Optional[torch.Generator] = None) -> torch.Tensor:
    flip = torch.rand(x.shape[0], generator=generator) < 0.5
    out = x.clone()
    out[flip] = torch.flip(x[flip], dims=[3])
    return out


def cutout(x: torch.Tensor, pad: int = 8,
           generator: Optional[torch.Generator] = None) -> torch.Tensor:
    """Cutout regularization (reference image_processing.py:52-66,
    cutout pad 8 per cifar10.py:72)."""
    B, C, H, W = x.shape
    out = x.clone()
    cy = torch.randint(0, H, (B,), generator=generator)
    cx = torch.randint(0, W, (B,), generator=generator)
    for b in range(B):
        y0, y1 = max(0, cy[b] - pad), min(H, cy[b] + pad)
        x0, x1 = max(0, cx[b] - pad), min(W, cx[b] + pad)
        out[b, :, y0:y1, x0:x1] = 0
    return out


class Provider(object):
    """Base data provider (reference cifar10.py:38)."""

    def __init__(self, n_classes: int, image_shape=(3, 32, 32),
                 batch_size: int = 32, augment: bool = True,
                 cutout_pad: int = 8, seed: Optional[int] = None):
        self.n_classes = n_classes
        self.image_shape = image_shape
        self.batch_size = batch_size
        self.augment = augment
        self.cutout_pad = cutout_pad
        self.seed = seed

    def _data(self, training: bool):  # pragma: no cover - interface
        raise NotImplementedError

    def get_input_fn(self, partition: str = "train", training: bool = True):
        """Returns an input_fn for adanet_amd.Estimator.train/evaluate."""
        X, Y = self._data(training=partition == "train")

        def input_fn():
            def gen():
                g = torch.Generator()
                g.manual_seed(self.seed or 0)
                n = X.shape[0]
                while True:
                    idx = torch.randint(0, n, (self.batch_size,),
                                        generator=g)
                    xb = X[idx]
                    if training and self.augment:
                        xb = random_crop(xb, generator=g)
                        xb = random_flip(xb, generator=g)
                        xb = cutout(xb, self.cutout_pad, generator=g)
                    yield xb, Y[idx]

            return gen()

        return input_fn


class FakeImageProvider(Provider):
    """Deterministic synthetic images (reference fake_data.py:26)."""

    def __init__(self, n_classes: int = 10, n_examples: int = 256, **kwargs):
        super().__init__(n_classes=n_classes, **kwargs)
        self.n_examples = n_examples
        g = torch.Generator().manual_seed(self.seed or 42)
        self._X = torch.randn(n_examples, *self.image_shape, generator=g)
        # teacher-labeled so the task is learnable
        w = torch.randn(
            int(torch.tensor(self.image_shape).prod()), n_classes,
            generator=g)
        self._Y = (self._X.reshape(n_examples, -1) @ w).argmax(dim=1)

    def _data(self, training: bool):
        return self._X, self._Y


def _load_cifar_binaries(data_dir: str, n_classes: int, training: bool):
    """Reads the standard CIFAR binary distribution (the format in
    cifar-10-binary.tar.gz / cifar-100-binary.tar.gz):

      * CIFAR-10:  data_batch_{1..5}.bin / test_batch.bin, records of
        1 label byte + 3072 RGB bytes;
      * CIFAR-100: train.bin / test.bin, records of 2 label bytes
        (coarse, fine) + 3072 RGB bytes.

    Returns (x [N,3,32,32] float in [0,1], y [N] long) or None if the
    files are absent. No torchvision / no network required.
    """
    import os

    import numpy as np
    if n_classes == 10:
        names = (["data_batch_%d.bin" % i for i in range(1, 6)]
                 if training else ["test_batch.bin"])
        rec, label_off = 3073, 0
    else:
        names = ["train.bin"] if training else ["test.bin"]
        rec, label_off = 3074, 1  # fine label is the second byte
    paths = []
    for n in names:
        for sub in ("", "cifar-10-batches-bin", "cifar-100-binary"):
            p = os.path.join(data_dir, sub, n)
            if os.path.exists(p):
                paths.append(p)
                break
    if len(paths) != len(names):
        return None
    xs, ys = [], []
    for p in paths:
        raw = np.fromfile(p, dtype=np.uint8)
        if raw.size % rec:
            return None
        raw = raw.reshape(-1, rec)
        ys.append(raw[:, label_off].astype(np.int64))
        xs.append(raw[:, rec - 3072:].reshape(-1, 3, 32, 32))
    x = torch.from_numpy(np.concatenate(xs)).float() / 255.0
    y = torch.from_numpy(np.concatenate(ys)).long()
    return x, y


class Cifar10Provider(Provider):
    """CIFAR-10 from pre-downloaded data, else synthetic fallback
    (reference cifar10.py:80-104 offline dummy fallback). Accepts either
    the standard binary distribution or a cifar10.npz of
    {x_train,y_train,x_test,y_test} under ``data_dir`` (also taken from
    $ADANET_CIFAR_DIR)."""

    _NPZ = "cifar10.npz"

    def __init__(self, data_dir: Optional[str] = None, **kwargs):
        super().__init__(n_classes=10, **kwargs)
        import os
        self._data_dir = data_dir or os.environ.get("ADANET_CIFAR_DIR")
        self._cache = {}

    @property
    def has_real_data(self) -> bool:
        return self._load(True) is not None

    def _load(self, training: bool):
        key = bool(training)
        if key in self._cache:
            return self._cache[key]
        out = None
        if self._data_dir is not None:
            import os

            import numpy as np
            npz = os.path.join(self._data_dir, self._NPZ)
            if os.path.exists(npz):
                z = np.load(npz)
                xk = "x_train" if training else "x_test"
                yk = "y_train" if training else "y_test"
                if xk in z:
                    x = torch.from_numpy(z[xk]).float()
                    if x.dim() == 4 and x.shape[-1] == 3:  # NHWC -> NCHW
                        x = x.permute(0, 3, 1, 2).contiguous()
                    if float(x.max()) > 1.5:
                        x = x / 255.0
                    out = (x, torch.from_numpy(z[yk]).long().reshape(-1))
            if out is None:
                out = _load_cifar_binaries(self._data_dir, self.n_classes,
                                           training)
        self._cache[key] = out
        return out

    def _data(self, training: bool):
        real = self._load(training)
        if real is not None:
            return real
        fake = FakeImageProvider(n_classes=self.n_classes, n_examples=512,
                                 seed=self.seed)
        return fake._data(training)


class Cifar100Provider(Cifar10Provider):

    _NPZ = "cifar100.npz"

    def __init__(self, data_dir: Optional[str] = None, **kwargs):
        Provider.__init__(self, n_classes=100, **kwargs)
        import os
        self._data_dir = data_dir or os.environ.get("ADANET_CIFAR_DIR")
        self._cache = {}
