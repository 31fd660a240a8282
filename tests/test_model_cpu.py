"""Torch-backend model math vs the exact reference formulas
(reference model/model.py:44-105, main.py:251-264)."""

import math

import numpy as np
import torch
import torch.nn.functional as F

from code2vec_amd.models.code2vec import Code2VecTorch, init_logical_params
from code2vec_amd.ops import reference as R
from code2vec_amd.utils.options import Option


def make_option(**kw):
    defaults = dict(
        terminal_count=50, path_count=40, label_count=12,
        max_path_length=10, terminal_embed_size=8, path_embed_size=6,
        encode_size=16, dropout_prob=0.0,
    )
    defaults.update(kw)
    return Option(**defaults)


def random_batch(opt, B=4, rng_seed=0):
    g = torch.Generator().manual_seed(rng_seed)
    C = opt.max_path_length
    starts = torch.randint(0, opt.terminal_count, (B, C), generator=g)
    paths = torch.randint(0, opt.path_count, (B, C), generator=g)
    ends = torch.randint(0, opt.terminal_count, (B, C), generator=g)
    # force some pad entries
    starts[:, -2:] = 0
    label = torch.randint(0, opt.label_count, (B,), generator=g)
    return starts, paths, ends, label


def test_forward_matches_manual_math():
    opt = make_option()
    g = torch.Generator().manual_seed(42)
    model = Code2VecTorch(opt, init_logical_params(opt, g)).eval()
    starts, paths, ends, label = random_batch(opt)

    outputs, cv, attn = model(starts, paths, ends, label)

    # manual recompute of reference model/model.py:44-88
    e_s = model.terminal_embedding[starts]
    e_p = model.path_embedding[paths]
    e_e = model.terminal_embedding[ends]
    ccv = torch.cat((e_s, e_p, e_e), dim=2)
    ccv = ccv @ model.input_weight.t()
    mu = ccv.mean(dim=2, keepdim=True)
    var = ccv.var(dim=2, unbiased=False, keepdim=True)
    ccv = (ccv - mu) / torch.sqrt(var + 1e-5)
    ccv = ccv * model.ln_gamma + model.ln_beta
    ccv = torch.tanh(ccv)
    mask = (starts > 0).float()
    scores = (ccv * model.attention_a).sum(dim=2) * mask + (1 - mask) * R.NINF
    attn_ref = torch.softmax(scores, dim=1)
    cv_ref = (ccv * attn_ref.unsqueeze(-1)).sum(dim=1)
    out_ref = cv_ref @ model.output_weight.t() + model.output_bias

    assert torch.allclose(attn, attn_ref, atol=1e-6)
    assert torch.allclose(cv, cv_ref, atol=1e-6)
    assert torch.allclose(outputs, out_ref, atol=1e-5)


def test_attention_all_pad_row_is_uniform():
    """A fully padded row: all scores NINF -> softmax uniform (matches the
    reference's exact NINF arithmetic, model/model.py:12,93)."""
    opt = make_option()
    model = Code2VecTorch(opt).eval()
    B, C = 2, opt.max_path_length
    starts = torch.zeros(B, C, dtype=torch.long)
    paths = torch.zeros(B, C, dtype=torch.long)
    ends = torch.zeros(B, C, dtype=torch.long)
    label = torch.zeros(B, dtype=torch.long)
    _, _, attn = model(starts, paths, ends, label)
    assert torch.allclose(attn, torch.full((B, C), 1.0 / C), atol=1e-6)


def test_loss_matches_nllloss():
    opt = make_option()
    model = Code2VecTorch(opt).eval()
    starts, paths, ends, label = random_batch(opt)
    outputs, _, _ = model(starts, paths, ends, label)
    weight = torch.ones(opt.label_count)
    loss = model.loss(outputs, label, weight)
    crit = torch.nn.NLLLoss(weight=weight)
    ref = crit(F.log_softmax(outputs, dim=1), label)
    assert torch.allclose(loss, ref, atol=1e-6)


def test_angular_margin_head():
    opt = make_option(angular_margin_loss=True)
    g = torch.Generator().manual_seed(7)
    model = Code2VecTorch(opt, init_logical_params(opt, g)).eval()
    starts, paths, ends, label = random_batch(opt)
    outputs, cv, _ = model(starts, paths, ends, label)

    cos_m = math.cos(opt.angular_margin)
    sin_m = math.sin(opt.angular_margin)
    cosine = F.linear(F.normalize(cv), F.normalize(model.output_weight))
    sine = torch.sqrt((1.0 - cosine.pow(2)).clamp(min=0))
    phi = torch.where(cosine > 0, cosine * cos_m - sine * sin_m, cosine)
    one_hot = torch.zeros_like(cosine).scatter_(1, label.view(-1, 1), 1)
    expected = (one_hot * phi + (1 - one_hot) * cosine) * opt.inverse_temp
    assert torch.allclose(outputs, expected, atol=1e-5)


def test_dropout_train_vs_eval():
    opt = make_option(dropout_prob=0.5)
    model = Code2VecTorch(opt)
    starts, paths, ends, label = random_batch(opt)
    model.eval()
    o1, _, _ = model(starts, paths, ends, label)
    o2, _, _ = model(starts, paths, ends, label)
    assert torch.equal(o1, o2)  # eval: dropout off
    model.train()
    torch.manual_seed(0)
    o3, _, _ = model(starts, paths, ends, label)
    assert not torch.equal(o1, o3)


def test_gradients_flow_everywhere():
    opt = make_option()
    model = Code2VecTorch(opt)
    model.train()
    starts, paths, ends, label = random_batch(opt)
    outputs, _, _ = model(starts, paths, ends, label)
    loss = model.loss(outputs, label, torch.ones(opt.label_count))
    loss.backward()
    for name, p in model.named_parameters():
        assert p.grad is not None, name
        assert torch.isfinite(p.grad).all(), name
