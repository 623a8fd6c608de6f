"""Transformer contrib ops (reference src/operator/contrib/transformer.cc:
650-1038 — interleaved self/enc-dec attention matmuls with fp32-accum
strided-batch GEMM, div_sqrt_dim, sliding-window attention masks).

The batched GEMMs run the native MFMA bgemm; layouts follow the
reference: qkv is [seq, batch, heads*3*head_dim] with per-head
interleaved q,k,v projections.
"""
import math

import torch

from . import nn as _nn

__all__ = ['interleaved_matmul_selfatt_qk', 'interleaved_matmul_selfatt_valatt',
           'interleaved_matmul_encdec_qk', 'interleaved_matmul_encdec_valatt',
           'div_sqrt_dim', 'sldwin_atten_mask_like', 'sldwin_atten_score',
           'sldwin_atten_context']


def _split_qkv(qkv, heads):
    # [S, B, H*3*D] -> q,k,v each [B*H, S, D]
    S, B, P = qkv.shape
    D = P // (3 * heads)
    x = qkv.reshape(S, B, heads, 3, D)
    q = x[:, :, :, 0].permute(1, 2, 0, 3).reshape(B * heads, S, D)
    k = x[:, :, :, 1].permute(1, 2, 0, 3).reshape(B * heads, S, D)
    v = x[:, :, :, 2].permute(1, 2, 0, 3).reshape(B * heads, S, D)
    return q.contiguous(), k.contiguous(), v.contiguous(), D


def interleaved_matmul_selfatt_qk(queries_keys_values, heads):
    """[S,B,H*3*D] -> scaled QK^T scores [B*H, S, S]
    (reference transformer.cc:650 _contrib_interleaved_matmul_selfatt_qk)."""
    q, k, _, D = _split_qkv(queries_keys_values, heads)
    scores = _nn.batch_dot(q, k, transpose_b=True)
    return scores / math.sqrt(D)


def interleaved_matmul_selfatt_valatt(queries_keys_values, attention, heads):
    """att [B*H,S,S] x V -> [S, B, H*D] (reference transformer.cc:742)."""
    _, _, v, D = _split_qkv(queries_keys_values, heads)
    S = queries_keys_values.shape[0]
    B = queries_keys_values.shape[1]
    out = _nn.batch_dot(attention.contiguous(), v)          # [B*H, S, D]
    return out.reshape(B, heads, S, D).permute(2, 0, 1, 3) \
              .reshape(S, B, heads * D).contiguous()


def _split_kv(keys_values, heads):
    S, B, P = keys_values.shape
    D = P // (2 * heads)
    x = keys_values.reshape(S, B, heads, 2, D)
    k = x[:, :, :, 0].permute(1, 2, 0, 3).reshape(B * heads, S, D)
    v = x[:, :, :, 1].permute(1, 2, 0, 3).reshape(B * heads, S, D)
    return k.contiguous(), v.contiguous(), D


def interleaved_matmul_encdec_qk(queries, keys_values, heads):
    """queries [Sq,B,H*D], keys_values [Sk,B,H*2*D] -> [B*H,Sq,Sk]."""
    Sq, B, P = queries.shape
    D = P // heads
    q = queries.reshape(Sq, B, heads, D).permute(1, 2, 0, 3) \
               .reshape(B * heads, Sq, D).contiguous()
    k, _, _ = _split_kv(keys_values, heads)
    return _nn.batch_dot(q, k, transpose_b=True) / math.sqrt(D)


def interleaved_matmul_encdec_valatt(keys_values, attention, heads):
    _, v, D = _split_kv(keys_values, heads)
    BH, Sq, _ = attention.shape
    B = keys_values.shape[1]
    out = _nn.batch_dot(attention.contiguous(), v)
    return out.reshape(B, heads, Sq, D).permute(2, 0, 1, 3) \
              .reshape(Sq, B, heads * D).contiguous()


def div_sqrt_dim(data):
    """data / sqrt(last_dim) (reference _contrib_div_sqrt_dim)."""
    return data / math.sqrt(data.shape[-1])


# -- sliding-window (Longformer-style) attention ---------------------------

def sldwin_atten_mask_like(score, dilation, num_heads, w, symmetric=True):
    """Band mask with the same shape as score
    (reference _contrib_sldwin_atten_mask_like transformer.cc:847)."""
    S = score.shape[-2]
    idx = torch.arange(S, device=score.device)
    rel = idx[None, :] - idx[:, None]
    left = w * dilation
    right = w * dilation if symmetric else 0
    band = (rel >= -left) & (rel <= right)
    if dilation > 1:
        band &= (rel % dilation) == 0
    return band.expand_as(score)


def sldwin_atten_score(query, key, dilation, num_heads, w, symmetric=True):
    """Banded QK^T computed densely then masked (the reference stores the
    band compactly; dense+mask is equivalent and MFMA-friendly here)."""
    scores = _nn.batch_dot(query, key, transpose_b=True)
    mask = sldwin_atten_mask_like(scores, dilation, num_heads, w, symmetric)
    return scores.masked_fill(~mask, float('-inf'))


def sldwin_atten_context(score, value, dilation, num_heads, w,
                         symmetric=True):
    return _nn.batch_dot(score.contiguous(), value.contiguous())
