"""The code2vec model — torch (oracle/CPU) and HIP (MI355X) backends.

Math parity target: reference model/model.py:15-106.  Both backends are
initialized from the same logical fp32 tensors (same seeds => identical
starting points), and both return ``(outputs, code_vector, attention)``.

The HIP backend stores parameters in MI355X-native padded layouts:
- embedding tables bf16 [T, TS] / [P, PS] (TS/PS = seg_round(dt/dp):
  16-B granules — dt=100 stores as 104, keeping rows 16-B aligned for
  bf16x8 gathers without the 32-element MFMA padding waste),
- combiner weight bf16 [EP, KP] stored TRANSPOSED (so the MFMA B
  fragment is memory-contiguous; KP = round_up(2*TS+PS, 32),
  EP = round_up(E, 32)),
- output weight bf16 [L, EP]; LN/attention/bias params fp32.
Pad regions are zero and stay zero through training (padding checks in
tests/test_model_gpu.py).
"""

from __future__ import annotations

import math
from typing import Dict, Optional

import torch
import torch.nn as nn
import torch.nn.functional as F

from ..ops import reference as R
from ..ops import round_up, seg_round
from ..ops import functional as Fn


def init_logical_params(option, generator: Optional[torch.Generator] = None) -> Dict[str, torch.Tensor]:
    """Reference-equivalent initial values, in logical fp32 shapes.

    Init schemes follow reference model/model.py:18-42: embeddings N(0,1)
    (nn.Embedding default), input_linear default Linear init (no bias), LN
    ones/zeros, attention xavier_normal on [E,1] then flattened, output
    Linear default with zero bias (or xavier_uniform, no bias, for the
    angular-margin head).
    """
    g = generator
    dt, dp, E, L = (
        option.terminal_embed_size,
        option.path_embed_size,
        option.encode_size,
        option.label_count,
    )
    K = 2 * dt + dp
    p: Dict[str, torch.Tensor] = {}
    p["terminal_embedding"] = torch.randn(option.terminal_count, dt, generator=g)
    p["path_embedding"] = torch.randn(option.path_count, dp, generator=g)
    w_in = torch.empty(E, K)
    nn.init.kaiming_uniform_(w_in, a=math.sqrt(5), generator=g)
    p["input_weight"] = w_in
    p["ln_gamma"] = torch.ones(E)
    p["ln_beta"] = torch.zeros(E)
    attn = torch.zeros(E, 1)
    nn.init.xavier_normal_(attn, generator=g)
    p["attention_a"] = attn.view(-1)
    w_out = torch.empty(L, E)
    if option.angular_margin_loss:
        nn.init.xavier_uniform_(w_out, generator=g)
        p["output_weight"] = w_out
    else:
        nn.init.kaiming_uniform_(w_out, a=math.sqrt(5), generator=g)
        p["output_weight"] = w_out
        p["output_bias"] = torch.zeros(L)
    return p


class Code2VecTorch(nn.Module):
    """Reference-math backend (fp32, stock torch ops) — CPU path + oracle."""

    backend = "torch"

    def __init__(self, option, logical: Optional[Dict[str, torch.Tensor]] = None):
        super().__init__()
        self.option = option
        if logical is None:
            logical = init_logical_params(option)
        self.terminal_embedding = nn.Parameter(logical["terminal_embedding"].clone())
        self.path_embedding = nn.Parameter(logical["path_embedding"].clone())
        self.input_weight = nn.Parameter(logical["input_weight"].clone())
        self.ln_gamma = nn.Parameter(logical["ln_gamma"].clone())
        self.ln_beta = nn.Parameter(logical["ln_beta"].clone())
        self.attention_a = nn.Parameter(logical["attention_a"].clone())
        self.output_weight = nn.Parameter(logical["output_weight"].clone())
        if not option.angular_margin_loss:
            self.output_bias = nn.Parameter(logical["output_bias"].clone())
        else:
            self.output_bias = None
        if option.angular_margin_loss:
            self.cos_m = math.cos(option.angular_margin)
            self.sin_m = math.sin(option.angular_margin)
        self.dropout_p = (
            option.dropout_prob if 0.0 < option.dropout_prob < 1.0 else 0.0
        )

    def forward(self, starts, paths, ends, label):
        opt = self.option
        ccv = R.gather_concat(
            starts, paths, ends, self.terminal_embedding, self.path_embedding
        )
        ccv = R.combiner(ccv, self.input_weight, self.ln_gamma, self.ln_beta)
        if self.dropout_p > 0.0:
            ccv = F.dropout(ccv, p=self.dropout_p, training=self.training)
        mask = (starts > 0).float()
        attn = R.attention(ccv, self.attention_a, mask)
        cv = R.code_vector(ccv, attn)
        if opt.angular_margin_loss:
            outputs = R.angular_margin_head(
                cv, self.output_weight, label, self.cos_m, self.sin_m,
                opt.inverse_temp,
            )
        else:
            outputs = R.output_head(cv, self.output_weight, self.output_bias)
        return outputs, cv, attn

    def loss(self, outputs, label, class_weight):
        return R.logsoftmax_nll(outputs, label, class_weight)


class Code2VecHIP(nn.Module):
    """MI355X backend — padded bf16 parameters, hand-written HIP kernels."""

    backend = "hip"

    def __init__(self, option, logical: Optional[Dict[str, torch.Tensor]] = None,
                 device: Optional[torch.device] = None):
        super().__init__()
        self.option = option
        if logical is None:
            logical = init_logical_params(option)
        device = device or torch.device("cuda")
        dt, dp, E = (
            option.terminal_embed_size,
            option.path_embed_size,
            option.encode_size,
        )
        self.TS = seg_round(dt)
        self.PS = seg_round(dp)
        self.EP = round_up(E)
        # combiner K: segments at offsets [0, TS, TS+PS], summed width
        # padded to the MFMA K granularity (gather zero-fills the tail)
        self.KP = round_up(2 * self.TS + self.PS, 32)
        self.E = E

        def pad2(t, rows, cols):
            out = torch.zeros(rows, cols, dtype=torch.float32)
            out[: t.shape[0], : t.shape[1]] = t
            return out

        def pad1(t, n):
            out = torch.zeros(n, dtype=torch.float32)
            out[: t.shape[0]] = t
            return out

        term = pad2(logical["terminal_embedding"], option.terminal_count, self.TS)
        path = pad2(logical["path_embedding"], option.path_count, self.PS)
        # combiner weight: logical [E, K=2dt+dp] -> padded TRANSPOSED layout
        # [EP, KP] (so MFMA B fragments are contiguous; see combiner.hip).
        w_in = logical["input_weight"]  # [E, K]
        w_pad = torch.zeros(self.EP, self.KP, dtype=torch.float32)
        w_pad[:E, :dt] = w_in[:, :dt]
        w_pad[:E, self.TS : self.TS + dp] = w_in[:, dt : dt + dp]
        w_pad[:E, self.TS + self.PS : self.TS + self.PS + dt] = w_in[:, dt + dp :]

        w_out = torch.zeros(option.label_count, self.EP, dtype=torch.float32)
        w_out[:, :E] = logical["output_weight"]

        self.terminal_embedding = nn.Parameter(term.to(device=device, dtype=torch.bfloat16))
        self.path_embedding = nn.Parameter(path.to(device=device, dtype=torch.bfloat16))
        self.input_weight = nn.Parameter(w_pad.to(device=device, dtype=torch.bfloat16))
        self.ln_gamma = nn.Parameter(pad1(logical["ln_gamma"], self.EP).to(device))
        self.ln_beta = nn.Parameter(pad1(logical["ln_beta"], self.EP).to(device))
        self.attention_a = nn.Parameter(pad1(logical["attention_a"], self.EP).to(device))
        self.output_weight = nn.Parameter(w_out.to(device=device, dtype=torch.bfloat16))
        if not option.angular_margin_loss:
            self.output_bias = nn.Parameter(logical["output_bias"].to(device))
        else:
            self.output_bias = None
        if option.angular_margin_loss:
            self.cos_m = math.cos(option.angular_margin)
            self.sin_m = math.sin(option.angular_margin)
        self.dropout_p = (
            option.dropout_prob if 0.0 < option.dropout_prob < 1.0 else 0.0
        )

    def forward(self, starts, paths, ends, label):
        opt = self.option
        B, C = starts.shape
        starts = starts.to(torch.int32)
        paths = paths.to(torch.int32)
        ends = ends.to(torch.int32)
        if Fn.FUSE_GATHER_COMBINER:
            # fused K1-K6: the concat tensor is never materialized
            y = Fn.FusedGatherCombiner.apply(
                starts, paths, ends, self.terminal_embedding,
                self.path_embedding, self.input_weight, self.ln_gamma,
                self.ln_beta, self.E, self.dropout_p, self.training,
            )
        else:
            x = Fn.GatherConcat.apply(
                starts, paths, ends, self.terminal_embedding,
                self.path_embedding
            )
            y = Fn.CombinerLNTanh.apply(
                x, self.input_weight, self.ln_gamma, self.ln_beta,
                self.E, self.dropout_p, self.training,
            )
        ccv = y.view(B, C, self.EP)
        cv, attn = Fn.AttentionPool.apply(ccv, self.attention_a, starts, self.E)
        if opt.angular_margin_loss:
            outputs = Fn.angular_margin_head_hip(
                cv.to(torch.bfloat16), self.output_weight, label,
                self.cos_m, self.sin_m, opt.inverse_temp,
            )
        else:
            cvb = cv.to(torch.bfloat16)
            if Fn.fused_head_loss_supported(cvb, self.output_weight,
                                            self.training):
                # fused head+loss path: logits are computed WITHOUT autograd
                # tracking; loss() builds the single FusedHeadLoss node whose
                # backward recomputes G (dlogits never materialized).  The
                # returned outputs tensor itself carries no grad — training
                # flows through model.loss(), which is the only consumer
                # (reference main.py:172-174).
                outputs = Fn.head_logits_with_stats(
                    cvb.detach(), self.output_weight.detach(),
                    self.output_bias.detach())
                outputs._c2v_fused_head = (cvb, self.output_weight,
                                           self.output_bias)
            else:
                outputs = Fn.OutputHead.apply(
                    cvb, self.output_weight, self.output_bias
                )
        return outputs, cv[:, : self.E], attn

    def loss(self, outputs, label, class_weight):
        fused = getattr(outputs, "_c2v_fused_head", None)
        if fused is not None:
            cvb, w, bias = fused
            del outputs._c2v_fused_head
            return Fn.FusedHeadLoss.apply(
                outputs.contiguous(), cvb, w, bias, label, class_weight)
        return Fn.FusedLogSoftmaxNLL.apply(
            outputs.contiguous(), label, class_weight)

    # ------------------------------------------------------------------
    def reference_state_dict(self) -> Dict[str, torch.Tensor]:
        """Checkpoint in the reference's state_dict format
        (keys/shapes of reference model/model.py, fp32, unpadded) so
        ``code2vec.model`` files are interchangeable."""
        opt = self.option
        dt, dp, E = opt.terminal_embed_size, opt.path_embed_size, opt.encode_size
        w_pad = self.input_weight.float()  # [EP, KP] transposed layout
        w_in = torch.cat(
            [
                w_pad[:E, :dt],
                w_pad[:E, self.TS : self.TS + dp],
                w_pad[:E, self.TS + self.PS : self.TS + self.PS + dt],
            ],
            dim=1,
        )
        sd = {
            "terminal_embedding.weight": self.terminal_embedding[:, :dt].float().cpu(),
            "path_embedding.weight": self.path_embedding[:, :dp].float().cpu(),
            "input_linear.weight": w_in.cpu(),
            "input_layer_norm.weight": self.ln_gamma[:E].detach().float().cpu(),
            "input_layer_norm.bias": self.ln_beta[:E].detach().float().cpu(),
            "attention_parameter": self.attention_a[:E].detach().float().cpu(),
        }
        if opt.angular_margin_loss:
            sd["output_linear"] = self.output_weight[:, :E].float().cpu()
        else:
            sd["output_linear.weight"] = self.output_weight[:, :E].float().cpu()
            sd["output_linear.bias"] = self.output_bias.detach().float().cpu()
        return sd


def reference_state_dict_torch(model: Code2VecTorch) -> Dict[str, torch.Tensor]:
    """Reference-format state_dict for the torch backend."""
    opt = model.option
    sd = {
        "terminal_embedding.weight": model.terminal_embedding.detach().cpu(),
        "path_embedding.weight": model.path_embedding.detach().cpu(),
        "input_linear.weight": model.input_weight.detach().cpu(),
        "input_layer_norm.weight": model.ln_gamma.detach().cpu(),
        "input_layer_norm.bias": model.ln_beta.detach().cpu(),
        "attention_parameter": model.attention_a.detach().cpu(),
    }
    if opt.angular_margin_loss:
        sd["output_linear"] = model.output_weight.detach().cpu()
    else:
        sd["output_linear.weight"] = model.output_weight.detach().cpu()
        sd["output_linear.bias"] = model.output_bias.detach().cpu()
    return sd


def logical_from_reference_state_dict(sd, option) -> Dict[str, torch.Tensor]:
    """Convert a reference-format state_dict (keys of reference
    model/model.py; our checkpoints use the same format) into logical
    init tensors, so checkpoints load into either backend."""
    p = {
        "terminal_embedding": sd["terminal_embedding.weight"].float(),
        "path_embedding": sd["path_embedding.weight"].float(),
        "input_weight": sd["input_linear.weight"].float(),
        "ln_gamma": sd["input_layer_norm.weight"].float(),
        "ln_beta": sd["input_layer_norm.bias"].float(),
        "attention_a": sd["attention_parameter"].float(),
    }
    if "output_linear" in sd:  # angular-margin head (Parameter, no bias)
        p["output_weight"] = sd["output_linear"].float()
    else:
        p["output_weight"] = sd["output_linear.weight"].float()
        p["output_bias"] = sd["output_linear.bias"].float()
    return p


def build_model(option, backend: str = "auto", logical=None, device=None):
    """Factory: 'hip' on CUDA devices, 'torch' on CPU (or forced)."""
    if backend == "auto":
        dev = device or option.device
        backend = "hip" if (dev is not None and torch.device(dev).type == "cuda") else "torch"
    if backend == "hip":
        return Code2VecHIP(option, logical=logical, device=device or option.device)
    model = Code2VecTorch(option, logical=logical)
    if device is not None:
        model = model.to(device)
    return model
