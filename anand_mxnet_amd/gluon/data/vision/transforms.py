"""Vision transforms (parity with gluon/data/vision/transforms.py; tensor
implementations — no OpenCV/PIL in this environment)."""
from __future__ import annotations

import numpy as np
import torch

from ...block import Block, HybridBlock
from ...nn.basic_layers import Sequential as Compose_  # base for Compose
from ....ndarray.ndarray import NDArray

__all__ = ["Compose", "Cast", "ToTensor", "Normalize", "Resize",
           "CenterCrop", "RandomResizedCrop", "RandomFlipLeftRight",
           "RandomFlipTopBottom",
           "RandomBrightness", "RandomContrast", "RandomSaturation",
           "RandomHue", "RandomColorJitter", "RandomLighting"]


class Compose(Compose_):
    def __init__(self, transforms):
        super().__init__()
        for t in transforms:
            self.add(t)


class Cast(Block):
    def __init__(self, dtype="float32"):
        super().__init__()
        self._dtype = dtype

    def forward(self, x):
        return x.astype(self._dtype)


class ToTensor(Block):
    """HWC uint8/float [0,255] -> CHW float32 [0,1]."""

    def forward(self, x):
        t = x._t.float() / 255.0
        if t.dim() == 3:
            t = t.permute(2, 0, 1)
        return NDArray(t.contiguous())


class Normalize(Block):
    def __init__(self, mean=0.0, std=1.0):
        super().__init__()
        self._mean = torch.as_tensor(mean, dtype=torch.float32)
        self._std = torch.as_tensor(std, dtype=torch.float32)

    def forward(self, x):
        t = x._t
        mean = self._mean.reshape(-1, 1, 1) if self._mean.dim() else self._mean
        std = self._std.reshape(-1, 1, 1) if self._std.dim() else self._std
        return NDArray((t - mean.to(t.device)) / std.to(t.device))


class Resize(Block):
    def __init__(self, size, keep_ratio=False, interpolation=1):
        super().__init__()
        self._size = (size, size) if isinstance(size, int) else tuple(size)

    def forward(self, x):
        t = x._t
        chw = t.dim() == 3 and t.shape[0] <= 4
        if not chw:
            t = t.permute(2, 0, 1)
        out = torch.nn.functional.interpolate(
            t.unsqueeze(0).float(), size=self._size, mode="bilinear",
            align_corners=False).squeeze(0)
        if not chw:
            out = out.permute(1, 2, 0)
        return NDArray(out)


class CenterCrop(Block):
    def __init__(self, size):
        super().__init__()
        self._size = (size, size) if isinstance(size, int) else tuple(size)

    def forward(self, x):
        t = x._t
        hwc = not (t.dim() == 3 and t.shape[0] <= 4)
        H, W = (t.shape[0], t.shape[1]) if hwc else (t.shape[1], t.shape[2])
        th, tw = self._size
        i, j = max(0, (H - th) // 2), max(0, (W - tw) // 2)
        out = t[i:i + th, j:j + tw] if hwc else t[:, i:i + th, j:j + tw]
        return NDArray(out.contiguous())


class RandomResizedCrop(Block):
    def __init__(self, size, scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3),
                 interpolation=1):
        super().__init__()
        self._size = (size, size) if isinstance(size, int) else tuple(size)
        self._scale = scale
        self._ratio = ratio

    def forward(self, x):
        t = x._t
        hwc = not (t.dim() == 3 and t.shape[0] <= 4)
        H, W = (t.shape[0], t.shape[1]) if hwc else (t.shape[1], t.shape[2])
        area = H * W
        for _ in range(10):
            target_area = np.random.uniform(*self._scale) * area
            ar = np.exp(np.random.uniform(np.log(self._ratio[0]),
                                          np.log(self._ratio[1])))
            w = int(round(np.sqrt(target_area * ar)))
            h = int(round(np.sqrt(target_area / ar)))
            if w <= W and h <= H:
                i = np.random.randint(0, H - h + 1)
                j = np.random.randint(0, W - w + 1)
                crop = t[i:i + h, j:j + w] if hwc else t[:, i:i + h, j:j + w]
                chw = crop.permute(2, 0, 1) if hwc else crop
                out = torch.nn.functional.interpolate(
                    chw.unsqueeze(0).float(), size=self._size,
                    mode="bilinear", align_corners=False).squeeze(0)
                return NDArray(out.permute(1, 2, 0) if hwc else out)
        return CenterCrop(self._size).forward(x)


class RandomFlipLeftRight(Block):
    def forward(self, x):
        if np.random.rand() < 0.5:
            axis = 1 if x._t.dim() == 3 and x._t.shape[0] > 4 else -1
            return NDArray(torch.flip(x._t, (axis,)))
        return x


class RandomFlipTopBottom(Block):
    def forward(self, x):
        if np.random.rand() < 0.5:
            axis = 0 if x._t.dim() == 3 and x._t.shape[0] > 4 else -2
            return NDArray(torch.flip(x._t, (axis,)))
        return x


def _jitter(img_t, factor):
    """Uniform multiplicative jitter in [1-f, 1+f] (reference
    image.py brightness/saturation/contrast aug convention)."""
    alpha = 1.0 + float(np.random.uniform(-factor, factor))
    return img_t.float() * alpha


class RandomBrightness(Block):
    """(reference: gluon/data/vision/transforms.py — RandomBrightness)"""

    def __init__(self, brightness):
        super().__init__()
        self._b = brightness

    def forward(self, x):
        return NDArray(_jitter(x._t, self._b).to(x._t.dtype))


class RandomContrast(Block):
    def __init__(self, contrast):
        super().__init__()
        self._c = contrast

    def forward(self, x):
        t = x._t.float()
        alpha = 1.0 + float(np.random.uniform(-self._c, self._c))
        mean = t.mean()
        return NDArray((t * alpha + mean * (1 - alpha)).to(x._t.dtype))


class RandomSaturation(Block):
    """HWC images; grayscale mix (ITU-R BT.601 weights, reference
    image.py SaturationJitterAug)."""

    def __init__(self, saturation):
        super().__init__()
        self._s = saturation

    def forward(self, x):
        t = x._t.float()
        alpha = 1.0 + float(np.random.uniform(-self._s, self._s))
        coef = torch.tensor([0.299, 0.587, 0.114], dtype=t.dtype)
        gray = (t * coef).sum(-1, keepdim=True)
        return NDArray((t * alpha + gray * (1 - alpha)).to(x._t.dtype))


class RandomHue(Block):
    """Approximate hue rotation in YIQ space (reference image.py
    HueJitterAug linear approximation)."""

    def __init__(self, hue):
        super().__init__()
        self._h = hue

    def forward(self, x):
        t = x._t.float()
        alpha = float(np.random.uniform(-self._h, self._h))
        u, w = np.cos(alpha * np.pi), np.sin(alpha * np.pi)
        bt = np.array([[1.0, 0.0, 0.0], [0.0, u, -w], [0.0, w, u]])
        tyiq = np.array([[0.299, 0.587, 0.114], [0.596, -0.274, -0.321],
                         [0.211, -0.523, 0.311]])
        m = np.linalg.inv(tyiq) @ bt @ tyiq
        mt = torch.tensor(m.T, dtype=t.dtype)
        return NDArray((t @ mt).to(x._t.dtype))


class RandomColorJitter(Block):
    def __init__(self, brightness=0, contrast=0, saturation=0, hue=0):
        super().__init__()
        self._ts = []
        if brightness:
            self._ts.append(RandomBrightness(brightness))
        if contrast:
            self._ts.append(RandomContrast(contrast))
        if saturation:
            self._ts.append(RandomSaturation(saturation))
        if hue:
            self._ts.append(RandomHue(hue))

    def forward(self, x):
        for t in np.random.permutation(self._ts):
            x = t.forward(x)
        return x


class RandomLighting(Block):
    """AlexNet-style PCA lighting noise (reference transforms)."""

    _eigval = np.array([55.46, 4.794, 1.148])
    _eigvec = np.array([[-0.5675, 0.7192, 0.4009],
                        [-0.5808, -0.0045, -0.8140],
                        [-0.5836, -0.6948, 0.4203]])

    def __init__(self, alpha):
        super().__init__()
        self._a = alpha

    def forward(self, x):
        a = np.random.normal(0, self._a, size=(3,))
        rgb = (self._eigvec * a * self._eigval).sum(axis=1)
        return NDArray((x._t.float() +
                        torch.tensor(rgb, dtype=torch.float32))
                       .to(x._t.dtype))
