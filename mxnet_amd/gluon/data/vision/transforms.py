"""Vision transforms (reference gluon/data/vision/transforms.py)."""
import torch

from ...block import Block, HybridBlock
from ....ndarray.ndarray import NDArray


def _t(x):
    if isinstance(x, NDArray):
        if x.is_native:
            # transforms are host-side preprocessing: bridge native
            # samples through numpy (the batch returns to the active
            # runtime via the loader's batchify)
            import numpy as _np
            return torch.from_numpy(_np.ascontiguousarray(x.asnumpy()))
        return x._t
    return x


class Compose(Block):
    def __init__(self, transforms):
        super().__init__()
        self._transforms = transforms

    def forward(self, x):
        for fn in self._transforms:
            x = fn(x)
        return x


class ToTensor(Block):
    """HWC uint8 [0,255] -> CHW float32 [0,1]."""

    def forward(self, x):
        t = _t(x)
        return NDArray(t.permute(2, 0, 1).float() / 255.0)


class Normalize(Block):
    def __init__(self, mean=0.0, std=1.0):
        super().__init__()
        self._mean = torch.as_tensor(mean, dtype=torch.float32)
        self._std = torch.as_tensor(std, dtype=torch.float32)

    def forward(self, x):
        t = _t(x)
        m = self._mean.reshape(-1, 1, 1) if self._mean.dim() else self._mean
        s = self._std.reshape(-1, 1, 1) if self._std.dim() else self._std
        return NDArray((t - m) / s)


class Cast(Block):
    def __init__(self, dtype='float32'):
        super().__init__()
        self._dtype = dtype

    def forward(self, x):
        return x.astype(self._dtype)


class Resize(Block):
    def __init__(self, size, keep_ratio=False, interpolation=1):
        super().__init__()
        self._size = (size, size) if isinstance(size, int) else tuple(size)

    def forward(self, x):
        t = _t(x)
        hwc = t.dim() == 3 and t.shape[-1] in (1, 3)
        if hwc:
            t = t.permute(2, 0, 1)
        y = torch.nn.functional.interpolate(
            t.unsqueeze(0).float(), size=self._size[::-1],
            mode='bilinear', align_corners=False).squeeze(0)
        if hwc:
            y = y.permute(1, 2, 0)
        return NDArray(y.to(_t(x).dtype))


class CenterCrop(Block):
    def __init__(self, size):
        super().__init__()
        self._size = (size, size) if isinstance(size, int) else tuple(size)

    def forward(self, x):
        t = _t(x)
        w, h = self._size
        H, W = t.shape[0], t.shape[1]
        y0 = max((H - h) // 2, 0)
        x0 = max((W - w) // 2, 0)
        return NDArray(t[y0:y0 + h, x0:x0 + w])


class RandomResizedCrop(Block):
    def __init__(self, size, scale=(0.08, 1.0), ratio=(3 / 4, 4 / 3),
                 interpolation=1):
        super().__init__()
        self._size = (size, size) if isinstance(size, int) else tuple(size)
        self._scale = scale
        self._ratio = ratio

    def forward(self, x):
        import random
        import math
        t = _t(x)
        H, W = t.shape[0], t.shape[1]
        area = H * W
        for _ in range(10):
            target_area = random.uniform(*self._scale) * area
            ar = math.exp(random.uniform(math.log(self._ratio[0]),
                                         math.log(self._ratio[1])))
            w = int(round(math.sqrt(target_area * ar)))
            h = int(round(math.sqrt(target_area / ar)))
            if w <= W and h <= H:
                x0 = random.randint(0, W - w)
                y0 = random.randint(0, H - h)
                crop = t[y0:y0 + h, x0:x0 + w]
                return Resize(self._size)(NDArray(crop))
        return Resize(self._size)(NDArray(t))


class RandomFlipLeftRight(Block):
    def forward(self, x):
        import random
        t = _t(x)
        if random.random() < 0.5:
            t = torch.flip(t, dims=[1])
        return NDArray(t)


class RandomFlipTopBottom(Block):
    def forward(self, x):
        import random
        t = _t(x)
        if random.random() < 0.5:
            t = torch.flip(t, dims=[0])
        return NDArray(t)
