"""Image transformation ops (reference python/mxnet/image/image.py +
src/operator/image/).

Tensor-domain transforms (resize/crop/flip/normalize/color jitter) run
on NDArray (HWC uint8/float).  File decode (``imread``/``imdecode``)
needs an image codec, which this offline image does not ship — those
raise with a clear message (the reference used OpenCV there).
"""
import torch

from .ndarray.ndarray import NDArray

__all__ = ['imresize', 'resize_short', 'fixed_crop', 'center_crop',
           'random_crop', 'HorizontalFlipAug', 'color_normalize',
           'imread', 'imdecode', 'CenterCropAug', 'ResizeAug']


def _t(x):
    if isinstance(x, NDArray):
        if x.is_native:
            # image preprocessing is host-side: bridge native arrays
            # through numpy (results wrap back via NDArray(tensor))
            import numpy as _np
            return torch.from_numpy(_np.ascontiguousarray(x.asnumpy()))
        return x.handle
    return x


def imread(*a, **k):
    raise NotImplementedError(
        'imread requires an image codec (OpenCV in the reference); '
        'this offline build operates on decoded arrays — use '
        'mx.nd.array(<decoded HWC array>)')


imdecode = imread


def imresize(src, w, h, interp=1):
    """Resize HWC image to (h, w); interp 1 = bilinear, 0 = nearest
    (reference image.imresize)."""
    t = _t(src)
    orig_dtype = t.dtype
    f = t.float().permute(2, 0, 1).unsqueeze(0)
    mode = 'nearest' if interp == 0 else 'bilinear'
    kwargs = {} if interp == 0 else {'align_corners': False}
    out = torch.nn.functional.interpolate(f, size=(h, w), mode=mode,
                                          **kwargs)
    out = out.squeeze(0).permute(1, 2, 0)
    if orig_dtype == torch.uint8:
        out = out.round().clamp(0, 255).to(torch.uint8)
    else:
        out = out.to(orig_dtype)
    return NDArray(out.contiguous())


def resize_short(src, size, interp=1):
    t = _t(src)
    h, w = t.shape[0], t.shape[1]
    if h > w:
        nh, nw = int(h * size / w), size
    else:
        nh, nw = size, int(w * size / h)
    return imresize(src, nw, nh, interp)


def fixed_crop(src, x0, y0, w, h, size=None, interp=1):
    t = _t(src)
    out = NDArray(t[y0:y0 + h, x0:x0 + w].contiguous())
    if size is not None and (w, h) != size:
        out = imresize(out, size[0], size[1], interp)
    return out


def center_crop(src, size, interp=1):
    t = _t(src)
    h, w = t.shape[0], t.shape[1]
    cw, ch = size
    x0 = max((w - cw) // 2, 0)
    y0 = max((h - ch) // 2, 0)
    return fixed_crop(src, x0, y0, min(cw, w), min(ch, h), size, interp), \
        (x0, y0, cw, ch)


def random_crop(src, size, interp=1):
    t = _t(src)
    h, w = t.shape[0], t.shape[1]
    cw, ch = size
    x0 = int(torch.randint(0, max(w - cw, 0) + 1, (1,)).item())
    y0 = int(torch.randint(0, max(h - ch, 0) + 1, (1,)).item())
    return fixed_crop(src, x0, y0, cw, ch, None, interp), (x0, y0, cw, ch)


def color_normalize(src, mean, std=None):
    t = _t(src).float()
    m = _t(mean) if isinstance(mean, NDArray) else torch.as_tensor(mean)
    t = t - m.to(t.device, t.dtype)
    if std is not None:
        s = _t(std) if isinstance(std, NDArray) else torch.as_tensor(std)
        t = t / s.to(t.device, t.dtype)
    return NDArray(t)


class Augmenter:
    def __call__(self, src):
        raise NotImplementedError


class ResizeAug(Augmenter):
    def __init__(self, size, interp=1):
        self.size, self.interp = size, interp

    def __call__(self, src):
        return resize_short(src, self.size, self.interp)


class CenterCropAug(Augmenter):
    def __init__(self, size, interp=1):
        self.size, self.interp = size, interp

    def __call__(self, src):
        return center_crop(src, self.size, self.interp)[0]


class HorizontalFlipAug(Augmenter):
    def __init__(self, p=0.5):
        self.p = p

    def __call__(self, src):
        if float(torch.rand(1)) < self.p:
            return NDArray(torch.flip(_t(src), dims=[1]).contiguous())
        return src
