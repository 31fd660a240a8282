"""End-to-end CPU training on a tiny synthetic corpus (BASELINE.json
config 1 shape) + artifact format round-trips."""

import os
import sys

import numpy as np
import pytest
import torch

import main as cli
from code2vec_amd.data.builder import DatasetBuilder
from code2vec_amd.data.reader import CorpusReader
from code2vec_amd.engine.trainer import Trainer, TrainerConfig
from code2vec_amd.models.code2vec import build_model, init_logical_params
from code2vec_amd.parallel.dist import DistContext
from code2vec_amd.utils.options import Option


def run_cli(tmp_path, tiny_corpus, extra=None):
    out_dir = tmp_path / "out"
    argv = [
        "--corpus_path", tiny_corpus["corpus_path"],
        "--path_idx_path", tiny_corpus["path_idx_path"],
        "--terminal_idx_path", tiny_corpus["terminal_idx_path"],
        "--model_path", str(out_dir),
        "--vectors_path", str(out_dir / "code.vec"),
        "--test_result_path", str(out_dir / "test_results.tsv"),
        "--max_epoch", "2",
        "--batch_size", "16",
        "--terminal_embed_size", "12",
        "--path_embed_size", "12",
        "--encode_size", "16",
        "--max_path_length", "12",
        "--no_cuda",
        "--print_sample_cycle", "0",
    ]
    if extra:
        argv.extend(extra)
    cli.main(argv)
    return out_dir


def test_cli_end_to_end(tmp_path, tiny_corpus):
    out_dir = run_cli(tmp_path, tiny_corpus)
    vec_file = out_dir / "code.vec"
    assert vec_file.exists()
    lines = vec_file.read_text().splitlines()
    # header: {len(reader.items)}\t{encode_size} (the reference's quirk:
    # count is items, rows are train+test dataset sizes)
    count, esz = lines[0].split("\t")
    assert int(count) == 48
    assert int(esz) == 16
    assert len(lines) == 1 + 48  # method task: rows == items
    for row in lines[1:]:
        label, vec = row.split("\t")
        values = vec.split(" ")
        assert len(values) == 16
        float(values[0])

    # checkpoint with reference-format keys
    sd = torch.load(out_dir / "code2vec.model", weights_only=True)
    assert "terminal_embedding.weight" in sd
    assert "input_linear.weight" in sd
    assert "attention_parameter" in sd
    assert sd["input_linear.weight"].shape == (16, 36)

    # test-result TSV: id \t correct \t expected \t predicted \t prob
    tsv = (out_dir / "test_results.tsv").read_text().splitlines()
    assert len(tsv) == int(48 * 0.2)
    parts = tsv[0].split("\t")
    assert len(parts) == 5
    int(parts[0]); float(parts[4])
    assert parts[1] in ("True", "False")


@pytest.mark.parametrize("shuffle", [False, True])
def test_cli_variable_task(tmp_path, tiny_corpus, shuffle):
    extra = ["--infer_method_name", "false", "--infer_variable_name", "true"]
    if shuffle:
        extra += ["--shuffle_variable_indexes", "true"]
    out_dir = run_cli(tmp_path, tiny_corpus, extra=extra)
    assert (out_dir / "code.vec").exists()


def test_training_reduces_loss(tiny_corpus):
    """Loss-curve sanity: a few epochs on the tiny corpus reduce train loss."""
    reader = CorpusReader(
        tiny_corpus["corpus_path"], tiny_corpus["path_idx_path"],
        tiny_corpus["terminal_idx_path"],
    )
    opt = Option(
        terminal_count=len(reader.terminal_vocab),
        path_count=len(reader.path_vocab),
        label_count=len(reader.label_vocab),
        max_path_length=12, terminal_embed_size=12, path_embed_size=12,
        encode_size=16, dropout_prob=0.0, batch_size=16,
        device=torch.device("cpu"),
    )
    builder = DatasetBuilder(reader, opt, seed=1)
    g = torch.Generator().manual_seed(0)
    model = build_model(opt, backend="torch", logical=init_logical_params(opt, g))
    ctx = DistContext(0, 1, 0, torch.device("cpu"))

    class A:  # minimal args carrier
        max_epoch = 1; lr = 0.01; beta_min = 0.9; beta_max = 0.999
        weight_decay = 0.0; model_path = "/tmp/c2v_test_out"
        vectors_path = "/tmp/c2v_test_out/code.vec"; test_result_path = None
        env = None; print_sample_cycle = 0; eval_method = "subtoken"
        random_seed = 1; batch_size = 16

    trainer = Trainer(TrainerConfig(A), opt, reader, builder, model, ctx)
    losses = []
    for epoch in range(4):
        loss, _ = trainer._train_epoch(epoch)
        losses.append(loss)
    assert losses[-1] < losses[0]


def test_code_vec_visualizer_parse_roundtrip(tmp_path, tiny_corpus):
    """The exported code.vec parses with the downstream consumer's logic
    (reference visualize_code_vec.py:8-23: skip header, tab-split label,
    space-split floats)."""
    out_dir = run_cli(tmp_path, tiny_corpus)
    from tools.visualize_code_vec import parse_code_vec

    labels, mat = parse_code_vec(str(out_dir / "code.vec"))
    assert len(labels) == mat.shape[0] == 48
    assert mat.shape[1] == 16
    assert np.isfinite(mat).all()


def test_checkpoint_warm_start(tmp_path, tiny_corpus):
    """--init_model loads a previously exported reference-format
    checkpoint (resume capability beyond the reference's save-only)."""
    out1 = run_cli(tmp_path, tiny_corpus)
    ckpt = out1 / "code2vec.model"
    out2 = tmp_path / "out2"
    argv = [
        "--corpus_path", tiny_corpus["corpus_path"],
        "--path_idx_path", tiny_corpus["path_idx_path"],
        "--terminal_idx_path", tiny_corpus["terminal_idx_path"],
        "--model_path", str(out2),
        "--vectors_path", str(out2 / "code.vec"),
        "--max_epoch", "1", "--batch_size", "16",
        "--terminal_embed_size", "12", "--path_embed_size", "12",
        "--encode_size", "16", "--max_path_length", "12",
        "--no_cuda", "--print_sample_cycle", "0",
        "--init_model", str(ckpt),
    ]
    cli.main(argv)
    assert (out2 / "code.vec").exists()
    # warm start actually used the checkpoint weights
    import torch

    sd1 = torch.load(ckpt, weights_only=True)
    from code2vec_amd.models.code2vec import (
        Code2VecTorch, logical_from_reference_state_dict,
    )
    from code2vec_amd.utils.options import Option

    opt = Option(terminal_count=sd1["terminal_embedding.weight"].shape[0],
                 path_count=sd1["path_embedding.weight"].shape[0],
                 label_count=sd1["output_linear.weight"].shape[0],
                 terminal_embed_size=12, path_embed_size=12, encode_size=16)
    m = Code2VecTorch(opt, logical_from_reference_state_dict(sd1, opt))
    assert torch.equal(m.terminal_embedding.detach(),
                       sd1["terminal_embedding.weight"])


@pytest.mark.parametrize("method", ["exact", "ave_subtoken"])
def test_eval_methods_end_to_end(tmp_path, tiny_corpus, method):
    out_dir = run_cli(tmp_path, tiny_corpus, extra=["--eval_method", method])
    assert (out_dir / "code.vec").exists()


def test_print_sample_runs(tmp_path, tiny_corpus, caplog):
    """print_sample walks a loader and logs one correctly-predicted
    example's contexts with attention (reference main.py:362-390)."""
    import logging
    from code2vec_amd.engine.export import print_sample
    from code2vec_amd.engine.loader import BatchIterator

    reader = CorpusReader(
        tiny_corpus["corpus_path"], tiny_corpus["path_idx_path"],
        tiny_corpus["terminal_idx_path"],
    )
    opt = Option(
        terminal_count=len(reader.terminal_vocab),
        path_count=len(reader.path_vocab),
        label_count=len(reader.label_vocab),
        max_path_length=12, terminal_embed_size=12, path_embed_size=12,
        encode_size=16, dropout_prob=0.0, batch_size=16,
        device=torch.device("cpu"),
    )
    builder = DatasetBuilder(reader, opt, seed=1)
    model = build_model(opt, backend="torch")
    data = builder.refresh_train_dataset(0)
    loader = BatchIterator(data, 16, shuffle=False, device=torch.device("cpu"))
    with caplog.at_level(logging.INFO):
        print_sample(reader, model, loader, opt, torch.device("cpu"))
    # either no correct prediction (silent) or a full sample dump
    if caplog.records:
        assert any("label" in r.getMessage() for r in caplog.records)


def test_seed_determinism_end_to_end(tiny_corpus):
    """Same seed => identical epoch-0 loss (the reference's unseeded
    run-to-run variance is a design delta we intentionally remove)."""
    def one_loss():
        reader = CorpusReader(
            tiny_corpus["corpus_path"], tiny_corpus["path_idx_path"],
            tiny_corpus["terminal_idx_path"],
        )
        opt = Option(
            terminal_count=len(reader.terminal_vocab),
            path_count=len(reader.path_vocab),
            label_count=len(reader.label_vocab),
            max_path_length=12, terminal_embed_size=12, path_embed_size=12,
            encode_size=16, dropout_prob=0.0, batch_size=16,
            device=torch.device("cpu"),
        )
        builder = DatasetBuilder(reader, opt, seed=21)
        g = torch.Generator().manual_seed(21)
        model = build_model(opt, backend="torch",
                            logical=init_logical_params(opt, g))
        ctx = DistContext(0, 1, 0, torch.device("cpu"))

        class A:
            max_epoch = 1; lr = 0.01; beta_min = 0.9; beta_max = 0.999
            weight_decay = 0.0; model_path = "/tmp/c2v_det"
            vectors_path = "/tmp/c2v_det/code.vec"; test_result_path = None
            env = None; print_sample_cycle = 0; eval_method = "subtoken"
            random_seed = 21; batch_size = 16

        t = Trainer(TrainerConfig(A), opt, reader, builder, model, ctx)
        loss, _ = t._train_epoch(0)
        return loss

    assert one_loss() == one_loss()


def test_find_hyperparams_with_stub_optuna(tmp_path, tiny_corpus, monkeypatch):
    """HPO wiring (reference main.py:429-488) exercised end to end with a
    minimal optuna stand-in (optuna itself is not installed here): the
    objective must mutate Option in place, train, report per-epoch values
    and honor pruning."""
    import sys
    import types

    import main as main_mod

    calls = {"suggest": [], "reports": 0, "pruned": 0}

    class Trial:
        def __init__(self, n):
            self.n = n

        def suggest_float(self, name, lo, hi, log=False):
            calls["suggest"].append(name)
            return float(lo)

        def report(self, value, step):
            calls["reports"] += 1

        def should_prune(self):
            # prune the second trial after its first epoch
            return self.n == 1 and calls["reports"] > 0

    class TrialPruned(Exception):
        pass

    class Study:
        best_params = {"adam_lr": 1e-5}
        best_value = 0.5

        def optimize(self, objective, n_trials):
            for n in range(n_trials):
                try:
                    objective(Trial(n))
                except TrialPruned:
                    calls["pruned"] += 1

    stub = types.ModuleType("optuna")
    stub.TrialPruned = TrialPruned
    stub.create_study = lambda pruner=None: Study()
    stub.pruners = types.SimpleNamespace(MedianPruner=lambda: None)
    monkeypatch.setitem(sys.modules, "optuna", stub)

    main_mod.main([
        "--corpus_path", tiny_corpus["corpus_path"],
        "--path_idx_path", tiny_corpus["path_idx_path"],
        "--terminal_idx_path", tiny_corpus["terminal_idx_path"],
        "--find_hyperparams", "--num_trials", "2", "--max_epoch", "2",
        "--no_cuda", "--encode_size", "16", "--terminal_embed_size", "8",
        "--path_embed_size", "8", "--batch_size", "8",
        "--model_path", str(tmp_path / "out"),
        "--vectors_path", str(tmp_path / "out" / "code.vec"),
    ])
    # same search space as the reference (main.py:447-456)
    assert set(calls["suggest"]) == {"encode_size", "dropout_prob",
                                     "batch_size", "adam_lr", "weight_decay"}
    assert calls["reports"] >= 2   # per-epoch trial.report
    assert calls["pruned"] == 1    # TrialPruned raised and handled


def test_floyd_metric_line_format(tmp_path, tiny_corpus, capsys):
    """--env floyd prints the machine-readable metric lines via print()
    (reference main.py:183-190 contract)."""
    import main as main_mod

    main_mod.main([
        "--corpus_path", tiny_corpus["corpus_path"],
        "--path_idx_path", tiny_corpus["path_idx_path"],
        "--terminal_idx_path", tiny_corpus["terminal_idx_path"],
        "--max_epoch", "1", "--no_cuda", "--env", "floyd",
        "--encode_size", "16", "--terminal_embed_size", "8",
        "--path_embed_size", "8", "--batch_size", "8",
        "--model_path", str(tmp_path / "out"),
        "--vectors_path", str(tmp_path / "out" / "code.vec"),
    ])
    out = capsys.readouterr().out
    for name in ("train_loss", "test_loss", "accuracy", "precision",
                 "recall", "f1", "best_f1"):
        assert f'{{"metric": "{name}", "value": ' in out, name
    assert "epoch 0" in out


def test_bench_driver_contract(tmp_path):
    """bench.py is the driver's measurement contract: with no GPU it must
    still run (torch reference model), finish quickly at the tiny config,
    and print ONE final JSON line carrying every field the driver and the
    judge consume."""
    import json
    import subprocess

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--config", "tiny",
         "--steps", "2", "--warmup", "1", "--pool", "2"],
        capture_output=True, text=True, timeout=420, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    line = out.stdout.strip().splitlines()[-1]
    d = json.loads(line)
    for key in ("metric", "value", "unit", "n_gpus", "steps", "warmup",
                "ms_per_step", "higher_is_better", "scaling", "vs_baseline",
                "dtype", "data", "config"):
        assert key in d, key
    assert d["n_gpus"] == 1 and d["steps"] == 2 and d["warmup"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["higher_is_better"] is True and d["scaling"] == "weak"
    assert isinstance(d["config"], dict) and "global_batch" in d["config"]
    assert d["dtype"] == "fp32"  # CPU fallback states its real dtype

    # --mode infer carries the same contract shape
    out = subprocess.run(
        [sys.executable, os.path.join(repo, "bench.py"), "--config", "tiny",
         "--mode", "infer", "--steps", "2", "--warmup", "1", "--pool", "2"],
        capture_output=True, text=True, timeout=420, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    d = json.loads(out.stdout.strip().splitlines()[-1])
    assert d["metric"] == "infer_path_contexts_per_sec"
    assert d["dtype"] == "fp32" and d["value"] > 0


@pytest.mark.timeout(600)
def test_bench_torchrun_world2_cpu():
    """Rehearse the driver's multi-GPU invocation shape on CPU/gloo:
    torchrun --nproc-per-node 2 bench.py --gpus 2 must produce exactly one
    JSON metric line (rank 0), with n_gpus/global_batch scaled."""
    import json
    import socket
    import subprocess

    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]; s.close()
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(repo, "bench.py"), "--gpus", "2", "--steps", "2",
         "--warmup", "1", "--config", "tiny", "--pool", "2"],
        capture_output=True, text=True, timeout=540, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [ln for ln in out.stdout.splitlines()
                  if ln.startswith("{") and "path_contexts_per_sec" in ln]
    assert len(json_lines) == 1, out.stdout[-1500:]
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 2
    assert d["config"]["parallelism"] == "dp2"
    assert d["config"]["global_batch"] == 64  # 2 ranks x tiny batch 32


@pytest.mark.timeout(600)
def test_bench_torchrun_world4_cpu():
    """4 gloo ranks through the same launcher shape — more ranks than
    owned-chunk splits and a deeper bucket partition than world 2."""
    import json
    import socket
    import subprocess

    s = socket.socket(); s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]; s.close()
    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", str(port),
         os.path.join(repo, "bench.py"), "--gpus", "4", "--steps", "2",
         "--warmup", "1", "--config", "tiny", "--pool", "2"],
        capture_output=True, text=True, timeout=540, cwd=repo)
    assert out.returncode == 0, out.stderr[-2000:]
    json_lines = [ln for ln in out.stdout.splitlines()
                  if ln.startswith("{") and "path_contexts_per_sec" in ln]
    assert len(json_lines) == 1, out.stdout[-1500:]
    d = json.loads(json_lines[0])
    assert d["n_gpus"] == 4 and d["config"]["parallelism"] == "dp4"


def test_cli_flag_parity_with_reference():
    """Every reference flag (reference main.py:37-79) must
    exist with the reference's exact default; our additions are
    new-flag-only (never a changed default)."""
    p = cli.build_parser()
    defaults = {a.dest: a.default for a in p._actions if a.dest != "help"}
    reference = {
        "random_seed": 123,
        "corpus_path": "./dataset/corpus.txt",
        "path_idx_path": "./dataset/path_idxs.txt",
        "terminal_idx_path": "./dataset/terminal_idxs.txt",
        "batch_size": 32,
        "terminal_embed_size": 100,
        "path_embed_size": 100,
        "encode_size": 300,
        "max_path_length": 200,
        "model_path": "./output",
        "vectors_path": "./output/code.vec",
        "test_result_path": None,
        "max_epoch": 40,
        "lr": 0.01,
        "beta_min": 0.9,
        "beta_max": 0.999,
        "weight_decay": 0.0,
        "dropout_prob": 0.25,
        "no_cuda": False,
        "gpu": "cuda:0",
        "num_workers": 4,
        "env": None,
        "print_sample_cycle": 10,
        "eval_method": "subtoken",
        "find_hyperparams": False,
        "num_trials": 100,
        "angular_margin_loss": False,
        "angular_margin": 0.5,
        "inverse_temp": 30.0,
        "infer_method_name": True,
        "infer_variable_name": False,
        "shuffle_variable_indexes": False,
    }
    for flag, want in reference.items():
        assert flag in defaults, f"missing reference flag --{flag}"
        assert defaults[flag] == want, (flag, defaults[flag], want)
    # eval_method choices parity (main.py:68)
    ev = next(a for a in p._actions if a.dest == "eval_method")
    assert set(ev.choices) == {"subtoken", "exact", "ave_subtoken"}


def test_checkpoint_key_set_matches_reference_format():
    """EXACT state-dict key set of the reference model (model/model.py:
    21-42) — a checkpoint written here must load into the reference's
    nn.Module and vice versa."""
    from code2vec_amd.models.code2vec import (
        Code2VecTorch, init_logical_params, reference_state_dict_torch)
    from code2vec_amd.utils.options import Option

    opt = Option(terminal_count=50, path_count=40, label_count=10,
                 max_path_length=8, terminal_embed_size=6,
                 path_embed_size=6, encode_size=12, dropout_prob=0.0,
                 batch_size=4, device=torch.device("cpu"))
    g = torch.Generator().manual_seed(3)
    m = Code2VecTorch(opt, init_logical_params(opt, g))
    sd = reference_state_dict_torch(m)
    assert set(sd.keys()) == {
        "terminal_embedding.weight",      # model.py:21
        "path_embedding.weight",          # model.py:22
        "input_linear.weight",            # model.py:23 (bias=False)
        "input_layer_norm.weight",        # model.py:24
        "input_layer_norm.bias",
        "attention_parameter",            # model.py:31
        "output_linear.weight",           # model.py:41
        "output_linear.bias",             # model.py:42
    }
    assert sd["terminal_embedding.weight"].shape == (50, 6)
    assert sd["output_linear.weight"].shape == (10, 12)
    assert sd["input_linear.weight"].shape == (12, 18)
    assert sd["attention_parameter"].shape == (12,)


def test_all_tool_clis_have_working_help():
    """Every tools/ CLI must answer --help (or print a graceful GPU-needed
    message) with exit code 0 — no raw tracebacks for a user probing the
    toolbox on a CPU box."""
    import subprocess

    repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    tools = ["gen_corpus.py", "extract_paths.py", "derive_label_vocab.py",
             "visualize_code_vec.py", "kbench.py", "prof_summary.py",
             "pmc_summary.py"]
    for t in tools:
        out = subprocess.run(
            [sys.executable, os.path.join(repo, "tools", t), "--help"],
            capture_output=True, text=True, timeout=90, cwd=repo)
        assert out.returncode == 0, (t, out.stderr[-400:])


def test_option_field_parity_with_reference():
    """Option carries every reference Option field (main.py:93-115)."""
    from code2vec_amd.utils.options import Option

    opt = Option(terminal_count=5, path_count=4, label_count=3,
                 device=torch.device("cpu"))
    for field in ("max_path_length", "terminal_count", "path_count",
                  "label_count", "terminal_embed_size", "path_embed_size",
                  "encode_size", "dropout_prob", "batch_size",
                  "eval_method", "angular_margin_loss", "angular_margin",
                  "inverse_temp", "device"):
        assert hasattr(opt, field), field
