"""End-to-end GPU tests: HIP backend vs the fp32 torch oracle, padding
invariants, and a short training-loss sanity run."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

from code2vec_amd.data.synthetic import synthetic_batch
from code2vec_amd.engine.optim import FusedAdam
from code2vec_amd.models.code2vec import (
    Code2VecHIP,
    Code2VecTorch,
    init_logical_params,
)
from code2vec_amd.utils.options import Option


def relerr(a, b):
    a = a.float(); b = b.float()
    n = b.norm()
    return float((a - b).norm() / n) if float(n) > 0 else float((a - b).norm())


def make_option(**kw):
    d = dict(terminal_count=3000, path_count=2500, label_count=700,
             max_path_length=40, terminal_embed_size=100, path_embed_size=100,
             encode_size=100, dropout_prob=0.0, batch_size=32)
    d.update(kw)
    return Option(**d)


def make_inputs(opt, B, dev, seed=0, full=False):
    rng = np.random.default_rng(seed)
    s, p, e, y = synthetic_batch(rng, B, opt.max_path_length,
                                 opt.terminal_count, opt.path_count,
                                 opt.label_count, full=full)
    return (torch.from_numpy(s).to(dev), torch.from_numpy(p).to(dev),
            torch.from_numpy(e).to(dev), torch.from_numpy(y).to(dev))


@pytest.fixture(scope="module")
def dev():
    return torch.device("cuda:0")


def test_forward_matches_oracle(dev):
    opt = make_option()
    g = torch.Generator().manual_seed(3)
    logical = init_logical_params(opt, g)
    hip = Code2VecHIP(opt, logical, device=dev).eval()
    ref = Code2VecTorch(opt, logical).to(dev).eval()
    s, p, e, y = make_inputs(opt, 16, dev)
    with torch.no_grad():
        out_h, cv_h, attn_h = hip(s, p, e, y)
        out_r, cv_r, attn_r = ref(s.long(), p.long(), e.long(), y)
    assert relerr(attn_h, attn_r) < 3e-2
    assert relerr(cv_h, cv_r) < 3e-2
    assert relerr(out_h, out_r) < 5e-2
    # prediction agreement (argmax can differ on near-ties; most must agree)
    agree = (out_h.float().argmax(1) == out_r.argmax(1)).float().mean()
    assert agree > 0.8


def test_loss_and_grads_match_oracle(dev):
    opt = make_option()
    g = torch.Generator().manual_seed(4)
    logical = init_logical_params(opt, g)
    hip = Code2VecHIP(opt, logical, device=dev).train()
    ref = Code2VecTorch(opt, logical).to(dev).train()
    s, p, e, y = make_inputs(opt, 24, dev, seed=2)
    w = torch.ones(opt.label_count, device=dev)

    out_h, _, _ = hip(s, p, e, y)
    loss_h = hip.loss(out_h, y, w)
    loss_h.backward()
    out_r, _, _ = ref(s.long(), p.long(), e.long(), y)
    loss_r = ref.loss(out_r, y, w)
    loss_r.backward()

    assert abs(float(loss_h) - float(loss_r)) / float(loss_r) < 2e-2
    E = opt.encode_size
    assert relerr(hip.attention_a.grad[:E], ref.attention_a.grad) < 6e-2
    assert relerr(hip.ln_gamma.grad[:E], ref.ln_gamma.grad) < 6e-2
    assert relerr(hip.output_bias.grad, ref.output_bias.grad) < 6e-2
    assert relerr(hip.output_weight.grad[:, :E], ref.output_weight.grad) < 6e-2


def test_fused_gather_combiner_matches_unfused(dev, monkeypatch):
    """C2V_FUSE=1 path (gather fused into the combiner GEMM + wgrad) must be
    bit-for-bit equivalent in eval forward and match the unfused backward
    closely (both are bf16 MFMA; only summation order differs)."""
    import code2vec_amd.ops.functional as Fn

    opt = make_option()
    g = torch.Generator().manual_seed(7)
    logical = init_logical_params(opt, g)
    s, p, e, y = make_inputs(opt, 24, dev, seed=5)
    w = torch.ones(opt.label_count, device=dev)

    results = []
    for fuse in (False, True):
        monkeypatch.setattr(Fn, "FUSE_GATHER_COMBINER", fuse)
        m = Code2VecHIP(opt, logical, device=dev).train()
        out, _, _ = m(s, p, e, y)
        loss = m.loss(out, y, w)
        loss.backward()
        results.append((
            out.detach(), float(loss), m.terminal_embedding.grad.clone(),
            m.path_embedding.grad.clone(), m.input_weight.grad.clone(),
            m.ln_gamma.grad.clone(),
        ))
    (o0, l0, tg0, pg0, wg0, gg0), (o1, l1, tg1, pg1, wg1, gg1) = results
    assert torch.equal(o0, o1)  # same kernel math for the activations
    assert abs(l0 - l1) / abs(l0) < 1e-3
    assert relerr(tg1, tg0) < 2e-2
    assert relerr(pg1, pg0) < 2e-2
    assert relerr(wg1, wg0) < 2e-2
    assert relerr(gg1, gg0) < 2e-2


def test_pad_regions_stay_zero_after_steps(dev):
    opt = make_option(dropout_prob=0.25)
    hip = Code2VecHIP(opt, device=dev).train()
    optim = FusedAdam(hip.parameters(), lr=0.01)
    w = torch.ones(opt.label_count, device=dev)
    E = opt.encode_size
    dt = opt.terminal_embed_size
    for i in range(3):
        s, p, e, y = make_inputs(opt, 16, dev, seed=10 + i)
        optim.zero_grad()
        out, _, _ = hip(s, p, e, y)
        loss = hip.loss(out, y, w)
        loss.backward()
        optim.step()
    assert torch.all(hip.terminal_embedding.detach()[:, dt:].float() == 0)
    assert torch.all(hip.path_embedding.detach()[:, dt:].float() == 0)
    # input_weight is transposed [EP, KP]: pad rows E..EP and pad K columns
    assert torch.all(hip.input_weight.detach()[E:, :].float() == 0)
    assert torch.all(hip.output_weight.detach()[:, E:].float() == 0)
    assert torch.all(hip.ln_gamma.detach()[E:] == 0)
    assert torch.all(hip.attention_a.detach()[E:] == 0)
    TS, PS = hip.TS, hip.PS
    assert torch.all(hip.input_weight.detach()[:, dt:TS].float() == 0)
    assert torch.all(hip.input_weight.detach()[:, TS + dt:TS + PS].float() == 0)


def test_training_reduces_loss_gpu(dev):
    opt = make_option(dropout_prob=0.25)
    hip = Code2VecHIP(opt, device=dev).train()
    optim = FusedAdam(hip.parameters(), lr=0.01)
    w = torch.ones(opt.label_count, device=dev)
    s, p, e, y = make_inputs(opt, 64, dev, seed=5, full=True)
    losses = []
    for i in range(30):
        optim.zero_grad()
        out, _, _ = hip(s, p, e, y)
        loss = hip.loss(out, y, w)
        loss.backward()
        optim.step()
        losses.append(float(loss))
    assert losses[-1] < losses[0] * 0.5, losses[:3] + losses[-3:]


def test_native_extension_is_loaded(dev):
    """Guard against silent eager fallback: the HIP extension module must be
    the in-tree .so and the model path must call into it."""
    import code2vec_amd.ops as O

    assert O.extension_available()
    mod = O.ext()
    assert "_c2v_hip" in mod.__file__
    assert "code2vec_amd" in mod.__file__


def test_graphed_inference_matches_eager(dev):
    """hipGraph-captured forward == eager forward (BASELINE config 5)."""
    from code2vec_amd.engine.infer import GraphedInference

    opt = make_option()
    hip = Code2VecHIP(opt, device=dev).eval()
    g = GraphedInference(hip, batch_size=16, device=dev)
    s, p, e, y = make_inputs(opt, 16, dev, seed=3)
    out_g, cv_g, attn_g = g.run(s.int(), p.int(), e.int(), y)
    with torch.no_grad():
        out_e, cv_e, attn_e = hip(s, p, e, y)
    assert torch.equal(cv_g, cv_e)
    assert torch.equal(attn_g, attn_e)
    assert torch.equal(out_g.float(), out_e.float())
    # tail batch smaller than capture size
    out_t, cv_t, attn_t = g.run(s.int()[:5], p.int()[:5], e.int()[:5], y[:5])
    assert torch.allclose(cv_t, cv_e[:5])


def test_cli_end_to_end_gpu(tmp_path_factory):
    """Full main.py train on GPU (HIP backend): 2 epochs on a synthetic
    corpus, artifact formats, finite metrics."""
    import main as cli
    from code2vec_amd.data.synthetic import SyntheticSpec, write_synthetic_corpus

    tmp_path = tmp_path_factory.mktemp("gpu_cli")
    files = write_synthetic_corpus(
        str(tmp_path / "data"),
        SyntheticSpec(n_methods=64, n_terminals=120, n_paths=90,
                      max_contexts=30, seed=9),
    )
    out_dir = tmp_path / "out"
    cli.main([
        "--corpus_path", files["corpus_path"],
        "--path_idx_path", files["path_idx_path"],
        "--terminal_idx_path", files["terminal_idx_path"],
        "--model_path", str(out_dir),
        "--vectors_path", str(out_dir / "code.vec"),
        "--max_epoch", "2", "--batch_size", "16",
        "--terminal_embed_size", "100", "--path_embed_size", "100",
        "--encode_size", "100", "--max_path_length", "16",
        "--print_sample_cycle", "0",
    ])
    lines = (out_dir / "code.vec").read_text().splitlines()
    assert len(lines) == 1 + 64
    count, esz = lines[0].split("\t")
    assert (int(count), int(esz)) == (64, 100)
    for row in lines[1:3]:
        _, vec = row.split("\t")
        vals = [float(v) for v in vec.split(" ")]
        assert len(vals) == 100
        assert all(v == v for v in vals)  # finite
    sd = torch.load(out_dir / "code2vec.model", weights_only=True)
    assert sd["terminal_embedding.weight"].shape[1] == 100
    assert sd["input_linear.weight"].shape == (100, 300)


def test_angular_margin_gpu(dev):
    """Optional ArcFace head path on the HIP backend (reference
    model/model.py:71-80)."""
    opt = make_option(angular_margin_loss=True)
    g = torch.Generator().manual_seed(6)
    logical = init_logical_params(opt, g)
    hip = Code2VecHIP(opt, logical, device=dev).train()
    ref = Code2VecTorch(opt, logical).to(dev).train()
    s, p, e, y = make_inputs(opt, 16, dev, seed=4)
    w = torch.ones(opt.label_count, device=dev)
    out_h, _, _ = hip(s, p, e, y)
    out_r, _, _ = ref(s.long(), p.long(), e.long(), y)
    assert relerr(out_h, out_r) < 6e-2
    loss_h = hip.loss(out_h, y, w)
    loss_h.backward()
    assert torch.isfinite(hip.output_weight.grad.float()).all()


@pytest.mark.parametrize("dt,dp,E", [(100, 100, 300), (72, 40, 160), (32, 32, 32)])
def test_nondefault_dims_match_oracle(dev, dt, dp, E):
    """Generality across padded shapes: default CLI encode_size is 300
    (EP=320, NT=20); small/odd embed sizes exercise segment padding."""
    opt = make_option(terminal_embed_size=dt, path_embed_size=dp,
                      encode_size=E, max_path_length=24,
                      terminal_count=800, path_count=700, label_count=150)
    g = torch.Generator().manual_seed(12)
    logical = init_logical_params(opt, g)
    hip = Code2VecHIP(opt, logical, device=dev).train()
    ref = Code2VecTorch(opt, logical).to(dev).train()
    s, p, e, y = make_inputs(opt, 12, dev, seed=8)
    w = torch.ones(opt.label_count, device=dev)
    out_h, cv_h, attn_h = hip(s, p, e, y)
    out_r, cv_r, attn_r = ref(s.long(), p.long(), e.long(), y)
    assert relerr(attn_h, attn_r) < 4e-2
    assert relerr(out_h, out_r) < 6e-2
    loss_h = hip.loss(out_h, y, w)
    loss_r = ref.loss(out_r, y, w)
    assert abs(float(loss_h) - float(loss_r)) / float(loss_r) < 3e-2
    loss_h.backward()
    loss_r.backward()
    E_ = opt.encode_size
    assert relerr(hip.ln_gamma.grad[:E_], ref.ln_gamma.grad) < 8e-2
    assert relerr(hip.terminal_embedding.grad[:, :dt].float(),
                  ref.terminal_embedding.grad) < 8e-2


def test_attention_grad_through_attn_output(dev):
    """Backward when the returned attention itself receives a gradient
    (the has_dattn path of attention_bwd)."""
    from code2vec_amd.ops.functional import AttentionPool
    from code2vec_amd.ops import reference as R
    from code2vec_amd.ops import round_up

    B, C, E = 6, 30, 100
    EP = round_up(E)
    g = torch.Generator().manual_seed(2)
    ccv = torch.zeros(B, C, EP)
    ccv[:, :, :E] = torch.tanh(torch.randn(B, C, E, generator=g))
    a = torch.zeros(EP)
    a[:E] = torch.randn(E, generator=g) * 0.2
    starts = torch.randint(1, 50, (B, C), generator=g, dtype=torch.int32)
    starts[:, -3:] = 0
    ccv_h = ccv.to(dev, torch.bfloat16).requires_grad_(True)
    a_h = a.to(dev).requires_grad_(True)
    starts_d = starts.to(dev)
    cv, attn = AttentionPool.apply(ccv_h, a_h, starts_d, E)
    dcv = torch.randn_like(cv) * 0.1
    dcv[:, E:] = 0
    dattn = torch.randn_like(attn) * 0.1
    torch.autograd.backward([cv, attn], [dcv, dattn])

    ccv_r = ccv.float().to(dev).requires_grad_(True)
    a_r = a.to(dev).requires_grad_(True)
    mask = (starts_d > 0).float()
    cv_r, attn_r = R.attention_code_vector(ccv_r, a_r, mask)
    torch.autograd.backward([cv_r, attn_r], [dcv, dattn])
    assert relerr(ccv_h.grad.float(), ccv_r.grad) < 6e-2
    assert relerr(a_h.grad[:E], a_r.grad[:E]) < 6e-2


def test_torch_backend_on_gpu(dev):
    """--backend torch runs the fp32 reference math on the GPU (oracle mode
    on device); one step must work and produce finite grads."""
    opt = make_option()
    model = Code2VecTorch(opt).to(dev).train()
    s, p, e, y = make_inputs(opt, 8, dev, seed=1)
    out, _, _ = model(s.long(), p.long(), e.long(), y)
    loss = model.loss(out, y, torch.ones(opt.label_count, device=dev))
    loss.backward()
    assert torch.isfinite(model.terminal_embedding.grad).all()


def test_finish_and_step_matches_plain_step(dev):
    """World=1: the overlapped finish_and_step (owner-buffer tables +
    row-chunk Adam) must update parameters exactly like finish()+step()."""
    import numpy as np

    from code2vec_amd.data.synthetic import synthetic_batch
    from code2vec_amd.engine.optim import FusedAdam
    from code2vec_amd.models.code2vec import Code2VecHIP, init_logical_params
    from code2vec_amd.parallel.ddp import BucketedAllReduce
    from code2vec_amd.utils.options import Option

    opt = make_option(dropout_prob=0.0)
    g = torch.Generator().manual_seed(21)
    logical = init_logical_params(opt, g)
    rng = np.random.default_rng(55)
    s, p, e, y = synthetic_batch(rng, 16, opt.max_path_length,
                                 opt.terminal_count, opt.path_count,
                                 opt.label_count)
    s = torch.from_numpy(s).to(dev); p = torch.from_numpy(p).to(dev)
    e = torch.from_numpy(e).to(dev); y = torch.from_numpy(y).to(dev)
    w = torch.ones(opt.label_count, device=dev)

    finals = []
    for mode in ("plain", "owned"):
        m = Code2VecHIP(opt, logical, device=dev).train()
        owned = ([m.terminal_embedding, m.path_embedding]
                 if mode == "owned" else None)
        ddp = BucketedAllReduce(list(m.parameters()), 1, owned_params=owned,
                                owned_chunks=3)
        optim = FusedAdam(m.parameters(), lr=0.01)
        for _ in range(3):
            ddp.zero_grad()
            out, _, _ = m(s, p, e, y)
            loss = m.loss(out, y, w)
            loss.backward()
            if mode == "owned":
                ddp.finish_and_step(optim)
            else:
                ddp.finish()
                optim.step()
        finals.append({n: q.detach().float().cpu()
                       for n, q in m.named_parameters()})
        ddp.close()
    for name in finals[0]:
        assert torch.equal(finals[0][name], finals[1][name]), name


def test_training_bitwise_deterministic(dev):
    """Two identical 3-step training runs produce bitwise-identical
    parameters: every reduction in the framework is fixed-order (no
    fp32 atomic accumulation ordering anywhere in the training path)."""
    import numpy as np

    from code2vec_amd.data.synthetic import synthetic_batch
    from code2vec_amd.engine.optim import FusedAdam
    from code2vec_amd.models.code2vec import Code2VecHIP, init_logical_params
    from code2vec_amd.parallel.ddp import BucketedAllReduce
    from code2vec_amd.ops.functional import reseed_dropout_rng

    opt = make_option(dropout_prob=0.25)  # dropout: counter RNG, reseeded
    g = torch.Generator().manual_seed(31)
    logical = init_logical_params(opt, g)
    rng = np.random.default_rng(66)
    s, p, e, y = synthetic_batch(rng, 16, opt.max_path_length,
                                 opt.terminal_count, opt.path_count,
                                 opt.label_count)
    s = torch.from_numpy(s).to(dev); p = torch.from_numpy(p).to(dev)
    e = torch.from_numpy(e).to(dev); y = torch.from_numpy(y).to(dev)
    w = torch.ones(opt.label_count, device=dev)

    def run():
        reseed_dropout_rng(77)
        m = Code2VecHIP(opt, logical, device=dev).train()
        ddp = BucketedAllReduce(list(m.parameters()), 1)
        optim = FusedAdam(m.parameters(), lr=0.01)
        for _ in range(3):
            ddp.zero_grad()
            out, _, _ = m(s, p, e, y)
            m.loss(out, y, w).backward()
            ddp.finish()
            optim.step()
        return {n: q.detach().float().cpu().clone()
                for n, q in m.named_parameters()}

    a, b = run(), run()
    for name in a:
        assert torch.equal(a[name], b[name]), name


def test_trainer_e2e_variable_task_gpu(dev, tmp_path):
    """Full Trainer run on the HIP backend with the VARIABLE-name task
    (infer_variable + shuffle_variable_indexes) — the numpy builder path
    feeding real GPU training."""
    from code2vec_amd.data.builder import DatasetBuilder
    from code2vec_amd.data.reader import CorpusReader
    from code2vec_amd.data.synthetic import SyntheticSpec, write_synthetic_corpus
    from code2vec_amd.engine.trainer import Trainer, TrainerConfig
    from code2vec_amd.models.code2vec import build_model, init_logical_params
    from code2vec_amd.parallel.dist import DistContext
    from code2vec_amd.utils.options import Option

    files = write_synthetic_corpus(
        str(tmp_path / "data"),
        SyntheticSpec(n_methods=60, n_terminals=300, n_paths=200,
                      max_contexts=20, n_vars_per_method=3, seed=11),
    )
    reader = CorpusReader(files["corpus_path"], files["path_idx_path"],
                          files["terminal_idx_path"], infer_method=True,
                          infer_variable=True, shuffle_variable_indexes=True)
    opt = Option(terminal_count=len(reader.terminal_vocab),
                 path_count=len(reader.path_vocab),
                 label_count=len(reader.label_vocab),
                 max_path_length=16, terminal_embed_size=100,
                 path_embed_size=100, encode_size=100, dropout_prob=0.25,
                 batch_size=16, device=dev)
    builder = DatasetBuilder(reader, opt, seed=3)
    model = build_model(opt, backend="hip",
                        logical=init_logical_params(
                            opt, torch.Generator().manual_seed(2)),
                        device=dev)
    ctx = DistContext(0, 1, 0, dev)

    class A:
        max_epoch = 2; lr = 0.01; beta_min = 0.9; beta_max = 0.999
        weight_decay = 0.0; model_path = str(tmp_path / "out")
        vectors_path = str(tmp_path / "out" / "code.vec")
        test_result_path = str(tmp_path / "out" / "results.tsv")
        env = None; print_sample_cycle = 0
        eval_method = "subtoken"; random_seed = 3; batch_size = 16

    trainer = Trainer(TrainerConfig(A), opt, reader, builder, model, ctx)
    obj = trainer.train()
    assert 0.0 <= obj <= 1.0
    # best-F1 artifacts written in the reference formats
    assert (tmp_path / "out" / "code.vec").exists()
    assert (tmp_path / "out" / "code2vec.model").exists()
    assert (tmp_path / "out" / "results.tsv").exists()


@pytest.mark.parametrize("method", ["exact", "ave_subtoken"])
def test_trainer_eval_methods_gpu(dev, tmp_path, method):
    """The two non-default eval methods through the HIP Trainer."""
    from code2vec_amd.data.builder import DatasetBuilder
    from code2vec_amd.data.reader import CorpusReader
    from code2vec_amd.data.synthetic import SyntheticSpec, write_synthetic_corpus
    from code2vec_amd.engine.trainer import Trainer, TrainerConfig
    from code2vec_amd.models.code2vec import build_model, init_logical_params
    from code2vec_amd.parallel.dist import DistContext
    from code2vec_amd.utils.options import Option

    files = write_synthetic_corpus(
        str(tmp_path / "d"),
        SyntheticSpec(n_methods=50, n_terminals=200, n_paths=150,
                      max_contexts=12, seed=13),
    )
    reader = CorpusReader(files["corpus_path"], files["path_idx_path"],
                          files["terminal_idx_path"])
    opt = Option(terminal_count=len(reader.terminal_vocab),
                 path_count=len(reader.path_vocab),
                 label_count=len(reader.label_vocab),
                 max_path_length=12, terminal_embed_size=100,
                 path_embed_size=100, encode_size=100, dropout_prob=0.0,
                 batch_size=16, eval_method=method, device=dev)
    builder = DatasetBuilder(reader, opt, seed=3)
    model = build_model(opt, backend="hip",
                        logical=init_logical_params(
                            opt, torch.Generator().manual_seed(4)),
                        device=dev)
    ctx = DistContext(0, 1, 0, dev)

    class A:
        max_epoch = 1; lr = 0.01; beta_min = 0.9; beta_max = 0.999
        weight_decay = 0.0; model_path = str(tmp_path / "o")
        vectors_path = str(tmp_path / "o" / "code.vec")
        test_result_path = None; env = None; print_sample_cycle = 0
        eval_method = method; random_seed = 3; batch_size = 16
        no_artifact_export = True

    trainer = Trainer(TrainerConfig(A), opt, reader, builder, model, ctx)
    obj = trainer.train()
    assert 0.0 <= obj <= 1.0


def test_graph_captured_step_matches_eager(dev):
    """A hipGraph-captured training step must produce bitwise-identical
    parameters to the eager step over several replays (device-resident
    dropout offset advances per replay)."""
    import numpy as np

    from code2vec_amd.data.synthetic import synthetic_batch
    from code2vec_amd.engine.optim import FusedAdam
    from code2vec_amd.models.code2vec import Code2VecHIP, init_logical_params
    from code2vec_amd.ops.functional import reseed_dropout_rng
    from code2vec_amd.parallel.ddp import BucketedAllReduce
    from code2vec_amd.utils.options import Option

    opt = Option(terminal_count=600, path_count=500, label_count=1024,
                 max_path_length=16, terminal_embed_size=100,
                 path_embed_size=100, encode_size=100, dropout_prob=0.25)
    g = torch.Generator().manual_seed(41)
    logical = init_logical_params(opt, g)
    rng = np.random.default_rng(77)
    batches = []
    for _ in range(3):
        s, p, e, y = synthetic_batch(rng, 16, opt.max_path_length,
                                     opt.terminal_count, opt.path_count,
                                     opt.label_count)
        batches.append(tuple(torch.from_numpy(a).to(dev)
                             for a in (s, p, e, y)))
    w = torch.ones(opt.label_count, device=dev)

    def run(graphed):
        reseed_dropout_rng(123)
        m = Code2VecHIP(opt, logical, device=dev).train()
        owned = [m.terminal_embedding, m.path_embedding]
        ddp = BucketedAllReduce(list(m.parameters()), 1, owned_params=owned)
        optim = FusedAdam(m.parameters(), lr=0.01)

        static = tuple(t.clone() for t in batches[0])

        def body():
            s, p, e, y = static
            ddp.zero_grad()
            out, _, _ = m(s, p, e, y)
            m.loss(out, y, w).backward()
            ddp.finish_and_step(optim)

        if graphed:
            side = torch.cuda.Stream()
            side.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(side):
                for _ in range(3):
                    body()
            torch.cuda.current_stream().wait_stream(side)
            # fresh model state for the measured run (warmup mutated it)
            reseed_dropout_rng(123)
            m2 = Code2VecHIP(opt, logical, device=dev).train()
            with torch.no_grad():
                for a, b in zip(m.parameters(), m2.parameters()):
                    a.copy_(b)
            for p_, st in optim.state.items():
                st["m"].zero_()
                st["v"].zero_()
                if st["master"] is not None:
                    st["master"].copy_(p_.detach().float().view(-1))
            optim.step_count = 0
            optim.bc_pow.fill_(1.0)  # warmup advanced the device beta^t
            graph = torch.cuda.CUDAGraph()
            with torch.cuda.graph(graph):
                body()
            for b_ in batches:
                for dst, src in zip(static, b_):
                    dst.copy_(src, non_blocking=True)
                graph.replay()
        else:
            for b_ in batches:
                for dst, src in zip(static, b_):
                    dst.copy_(src, non_blocking=True)
                body()
        torch.cuda.synchronize()
        out = {n: q.detach().float().cpu().clone()
               for n, q in m.named_parameters()}
        ddp.close()
        return out

    eager = run(False)
    graphed = run(True)
    for name in eager:
        assert torch.equal(eager[name], graphed[name]), name
