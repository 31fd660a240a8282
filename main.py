#!/usr/bin/env python3
"""code2vec_amd CLI — flag-compatible with the reference main.py:37-81.

Every reference flag is accepted with the same name, type and default;
MI355X-specific additions (--precision, --backend, --bucket_mb) have no
reference counterpart.  Multi-GPU data parallelism: launch with
``torchrun --nproc-per-node N main.py ...`` (one process per GPU, RCCL).
"""

from __future__ import annotations

import argparse
import logging

import torch

from code2vec_amd.data.builder import DatasetBuilder
from code2vec_amd.data.reader import CorpusReader
from code2vec_amd.engine.trainer import Trainer, TrainerConfig
from code2vec_amd.models.code2vec import build_model, init_logical_params
from code2vec_amd.parallel.dist import init_distributed
from code2vec_amd.utils.options import Option

logger = logging.getLogger()


def strtobool(v: str) -> bool:
    v = v.lower()
    if v in ("y", "yes", "t", "true", "on", "1"):
        return True
    if v in ("n", "no", "f", "false", "off", "0"):
        return False
    raise ValueError(f"invalid truth value {v!r}")


def build_parser() -> argparse.ArgumentParser:
    p = argparse.ArgumentParser()
    p.add_argument("--random_seed", type=int, default=123)
    p.add_argument("--corpus_path", type=str, default="./dataset/corpus.txt")
    p.add_argument("--path_idx_path", type=str, default="./dataset/path_idxs.txt")
    p.add_argument("--terminal_idx_path", type=str, default="./dataset/terminal_idxs.txt")
    p.add_argument("--batch_size", type=int, default=32)
    p.add_argument("--terminal_embed_size", type=int, default=100)
    p.add_argument("--path_embed_size", type=int, default=100)
    p.add_argument("--encode_size", type=int, default=300)
    p.add_argument("--max_path_length", type=int, default=200)
    p.add_argument("--model_path", type=str, default="./output")
    p.add_argument("--vectors_path", type=str, default="./output/code.vec")
    p.add_argument("--test_result_path", type=str, default=None)
    p.add_argument("--max_epoch", type=int, default=40)
    p.add_argument("--lr", type=float, default=0.01)
    p.add_argument("--beta_min", type=float, default=0.9)
    p.add_argument("--beta_max", type=float, default=0.999)
    p.add_argument("--weight_decay", type=float, default=0.0)
    p.add_argument("--dropout_prob", type=float, default=0.25)
    p.add_argument("--no_cuda", action="store_true", default=False)
    p.add_argument("--gpu", type=str, default="cuda:0")
    p.add_argument("--num_workers", type=int, default=4)  # accepted; pipeline is array-sliced
    p.add_argument("--env", type=str, default=None)
    p.add_argument("--print_sample_cycle", type=int, default=10)
    p.add_argument("--eval_method", type=str, default="subtoken",
                   choices=["subtoken", "exact", "ave_subtoken"])
    p.add_argument("--find_hyperparams", action="store_true", default=False)
    p.add_argument("--num_trials", type=int, default=100)
    p.add_argument("--angular_margin_loss", action="store_true", default=False)
    p.add_argument("--angular_margin", type=float, default=0.5)
    p.add_argument("--inverse_temp", type=float, default=30.0)
    p.add_argument("--infer_method_name", type=strtobool, default=True)
    p.add_argument("--infer_variable_name", type=strtobool, default=False)
    p.add_argument("--shuffle_variable_indexes", type=strtobool, default=False)
    # MI355X-native additions
    p.add_argument("--precision", type=str, default="bf16", choices=["bf16", "fp32"],
                   help="GPU compute precision (HIP backend is bf16)")
    p.add_argument("--backend", type=str, default="auto",
                   choices=["auto", "hip", "torch"])
    p.add_argument("--bucket_mb", type=int, default=32,
                   help="gradient all-reduce bucket size (MiB)")
    p.add_argument("--no_artifact_export", action="store_true", default=False,
                   help="MI355X extra: skip the best-F1 code.vec/checkpoint "
                        "export (useful for pure throughput/F1 benchmark "
                        "runs on top11-scale corpora where the text export "
                        "dominates wall time)")
    p.add_argument("--init_model", type=str, default=None,
                   help="warm-start from a code2vec.model checkpoint "
                        "(reference state_dict format)")
    return p


def setup_logging() -> None:
    logger.setLevel(logging.INFO)
    fmt = logging.Formatter("%(asctime)s: %(message)s", "%m/%d/%Y %I:%M:%S %p")
    console = logging.StreamHandler()
    console.setFormatter(fmt)
    logger.addHandler(console)


def make_context(args):
    use_cuda = not args.no_cuda and torch.cuda.is_available()
    ctx = init_distributed("cuda" if use_cuda else "cpu")
    if use_cuda and ctx.world_size == 1:
        # single process honors --gpu (reference main.py:83)
        ctx.device = torch.device(args.gpu)
        torch.cuda.set_device(ctx.device)
    logger.info("device: {0}".format(ctx.device))
    return ctx


def prepare(args, ctx):
    torch.manual_seed(args.random_seed)
    reader = CorpusReader(
        args.corpus_path, args.path_idx_path, args.terminal_idx_path,
        infer_method=args.infer_method_name,
        infer_variable=args.infer_variable_name,
        shuffle_variable_indexes=args.shuffle_variable_indexes,
    )
    option = Option.from_args(args, reader, ctx.device)
    builder = DatasetBuilder(
        reader, option, seed=args.random_seed,
        rank=ctx.rank, world_size=ctx.world_size,
    )
    return reader, option, builder


def train(args) -> None:
    ctx = make_context(args)
    reader, option, builder = prepare(args, ctx)
    if args.init_model:
        from code2vec_amd.models.code2vec import logical_from_reference_state_dict

        sd = torch.load(args.init_model, map_location="cpu", weights_only=True)
        logical = logical_from_reference_state_dict(sd, option)
    else:
        gen = torch.Generator().manual_seed(args.random_seed)
        logical = init_logical_params(option, gen)
    backend = args.backend
    if backend == "auto":
        backend = "hip" if ctx.device.type == "cuda" else "torch"
    model = build_model(option, backend=backend, logical=logical, device=ctx.device)
    cfg = TrainerConfig(args)
    trainer = Trainer(cfg, option, reader, builder, model, ctx)
    trainer.train()


def find_optimal_hyperparams(args) -> None:
    """Optuna HPO (reference main.py:429-488)."""
    try:
        import optuna
    except ImportError as e:
        raise SystemExit(
            "--find_hyperparams needs the optuna package, which is not "
            "installed in this environment"
        ) from e

    ctx = make_context(args)
    reader, option, builder = prepare(args, ctx)

    def objective(trial):
        option.encode_size = int(trial.suggest_float("encode_size", 100, 300, log=True))
        option.dropout_prob = trial.suggest_float("dropout_prob", 0.5, 0.9, log=True)
        option.batch_size = int(trial.suggest_float("batch_size", 256, 2048, log=True))
        args.lr = trial.suggest_float("adam_lr", 1e-5, 1e-1, log=True)
        args.weight_decay = trial.suggest_float("weight_decay", 1e-10, 1e-3, log=True)
        args.batch_size = option.batch_size
        gen = torch.Generator().manual_seed(args.random_seed)
        logical = init_logical_params(option, gen)
        backend = args.backend
        if backend == "auto":
            backend = "hip" if ctx.device.type == "cuda" else "torch"
        model = build_model(option, backend=backend, logical=logical, device=ctx.device)
        cfg = TrainerConfig(args)
        trainer = Trainer(cfg, option, reader, builder, model, ctx, trial=trial)
        return trainer.train()

    study = optuna.create_study(pruner=optuna.pruners.MedianPruner())
    study.optimize(objective, n_trials=args.num_trials)
    if args.env == "floyd":
        print("best hyperparams: {0}".format(study.best_params))
        print("best value: {0}".format(study.best_value))
    else:
        logger.info("optimal hyperparams: {0}".format(study.best_params))
        logger.info("best value: {0}".format(study.best_value))


def main(argv=None) -> None:
    setup_logging()
    args = build_parser().parse_args(argv)
    if args.find_hyperparams:
        find_optimal_hyperparams(args)
    else:
        train(args)


if __name__ == "__main__":
    main()
