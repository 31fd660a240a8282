"""Pure-PyTorch reference implementations of every framework op.

These define the exact math (reference model/model.py:44-105 and
main.py:251-264) and serve three roles:
1. the CPU training/eval path,
2. the numerical oracle that every HIP kernel is tested against
   (tests/test_kernels_gpu.py),
3. documentation of each kernel's contract.

Shapes: B batch, C contexts (max_path_length), dt/dp terminal/path embed
sizes, E encode size, L label count, T/P terminal/path vocab sizes.
"""

from __future__ import annotations

import math

import torch
import torch.nn.functional as F

NINF = -3.4 * math.pow(10, 38)  # matches reference model/model.py:12


def gather_concat(starts, paths, ends, terminal_weight, path_weight):
    """K1/K2: triple embedding gather + concat (reference model/model.py:48-51).

    starts/paths/ends: int [B, C]; returns [B, C, 2*dt+dp].
    """
    e_s = F.embedding(starts.long(), terminal_weight)
    e_p = F.embedding(paths.long(), path_weight)
    e_e = F.embedding(ends.long(), terminal_weight)
    return torch.cat((e_s, e_p, e_e), dim=2)


def combiner(ccv, weight, gamma, beta, eps: float = 1e-5):
    """K3-K5: context combiner = Linear(no bias) -> LayerNorm -> tanh
    (reference model/model.py:54-57).

    ccv: [B, C, 2dt+dp]; weight: [E, 2dt+dp]; returns [B, C, E].
    """
    x = F.linear(ccv, weight)
    shape = x.shape
    x = F.layer_norm(x.view(-1, shape[-1]), (shape[-1],), gamma, beta, eps)
    return torch.tanh(x).view(shape)


def attention(ccv, a, mask):
    """K7/K8: masked single-query attention scores + softmax
    (reference model/model.py:90-105).

    ccv: [B, C, E]; a: [E]; mask: [B, C] float (1 valid, 0 pad).
    Returns attention [B, C].
    """
    scores = torch.sum(ccv * a.unsqueeze(0).unsqueeze(0), dim=2)
    scores = scores * mask + (1.0 - mask) * NINF
    return F.softmax(scores, dim=1)


def code_vector(ccv, attn):
    """K9: attention-weighted sum (reference model/model.py:68-69)."""
    return torch.sum(ccv * attn.unsqueeze(-1), dim=1)


def attention_code_vector(ccv, a, mask):
    """Fused K7+K8+K9 — contract of the HIP attention kernel."""
    attn = attention(ccv, a, mask)
    return code_vector(ccv, attn), attn


def output_head(cv, weight, bias):
    """K10: label projection (reference model/model.py:83)."""
    return F.linear(cv, weight, bias)


def angular_margin_head(cv, weight, label, cos_m, sin_m, inverse_temp):
    """K11: ArcFace-style head (reference model/model.py:71-80).

    Note the reference computes but never uses th/mm; the actual math is:
    phi = cos*cos_m - sin*sin_m, phi where cos>0 else cos,
    blend by one-hot, scale by inverse_temp.
    """
    cosine = F.linear(F.normalize(cv), F.normalize(weight))
    sine = torch.sqrt(torch.clamp(1.0 - cosine * cosine, min=0.0))
    phi = cosine * cos_m - sine * sin_m
    phi = torch.where(cosine > 0, phi, cosine)
    one_hot = torch.zeros_like(cosine)
    one_hot.scatter_(1, label.view(-1, 1).long(), 1)
    outputs = one_hot * phi + (1.0 - one_hot) * cosine
    return outputs * inverse_temp


def logsoftmax_nll(logits, label, weight):
    """K12: log_softmax + weighted NLL (reference main.py:251-264 with the
    criterion built at main.py:129-130).

    weight: [L] per-class weights (1/freq — effectively all ones, see
    data/vocab.py docstring).  Reduction follows nn.NLLLoss default:
    sum(w_yi * -logp_yi) / sum(w_yi).
    """
    logp = F.log_softmax(logits.float(), dim=1)
    return F.nll_loss(logp, label.long(), weight=weight)
