"""Training / evaluation driver — the reference main.py:118-297 rebuilt
around DP, fused kernels and async input, with identical observable
behavior (metric-line format, early stopping, best-F1 artifact export).
"""

from __future__ import annotations

import logging
import os
import time
from os import path as osp
from typing import Optional

import torch

from concurrent.futures import ThreadPoolExecutor

from ..data.builder import DatasetBuilder
from ..models.code2vec import (
    Code2VecHIP,
    reference_state_dict_torch,
)
from ..parallel.dist import DistContext, all_reduce_sum_list, all_reduce_sum_scalar
from ..parallel.ddp import BucketedAllReduce
from . import metrics as M
from .export import print_sample, write_code_vectors, write_vector_header
from .loader import BatchIterator
from .optim import FusedAdam
from ..ops import functional as Fn

logger = logging.getLogger(__name__)


class _Roctx:
    """roctx ranges (torch nvtx == roctx on ROCm) — visible in rocprofv3
    --marker-trace; no-ops on CPU."""

    def __init__(self, enabled: bool) -> None:
        self.enabled = enabled

    def push(self, name: str) -> None:
        if self.enabled:
            torch.cuda.nvtx.range_push(name)

    def pop(self) -> None:
        if self.enabled:
            torch.cuda.nvtx.range_pop()


class TrainerConfig:
    def __init__(self, args) -> None:
        self.max_epoch = args.max_epoch
        self.lr = args.lr
        self.beta_min = args.beta_min
        self.beta_max = args.beta_max
        self.weight_decay = args.weight_decay
        self.model_path = args.model_path
        self.vectors_path = args.vectors_path
        self.test_result_path = args.test_result_path
        self.env = args.env
        self.print_sample_cycle = args.print_sample_cycle
        self.eval_method = args.eval_method
        self.random_seed = args.random_seed
        self.batch_size = args.batch_size
        self.no_artifact_export = getattr(args, "no_artifact_export", False)


class Trainer:
    def __init__(
        self,
        config: TrainerConfig,
        option,
        reader,
        builder: DatasetBuilder,
        model,
        ctx: DistContext,
        trial=None,
    ) -> None:
        self.cfg = config
        self.option = option
        self.reader = reader
        self.builder = builder
        self.model = model
        self.ctx = ctx
        self.trial = trial
        self.device = ctx.device

        # loss class weights = 1/label_freq (reference main.py:129-130)
        freq = torch.tensor(
            reader.label_vocab.get_freq_list(), dtype=torch.float32
        )
        self.class_weight = (1.0 / freq).to(self.device)

        if isinstance(model, Code2VecHIP):
            self.optimizer = FusedAdam(
                model.parameters(),
                lr=config.lr,
                betas=(config.beta_min, config.beta_max),
                weight_decay=config.weight_decay,
            )
        else:
            self.optimizer = torch.optim.Adam(
                model.parameters(),
                lr=config.lr,
                betas=(config.beta_min, config.beta_max),
                weight_decay=config.weight_decay,
            )

        # the embedding tables take the owner-buffer chunked-overlap path
        # (their grads are produced last in backward; see parallel/ddp.py)
        owned = (
            [model.terminal_embedding, model.path_embedding]
            if isinstance(model, Code2VecHIP) else None
        )
        self.ddp = BucketedAllReduce(
            list(model.parameters()), ctx.world_size, owned_params=owned
        )
        self.ddp.broadcast_parameters()
        self._overlapped_step = isinstance(self.optimizer, FusedAdam)

        self.roctx = _Roctx(self.device.type == "cuda")
        # epoch N+1's host-side rebuild runs here while N trains (the
        # native C++ builder releases the GIL)
        self._build_pool = ThreadPoolExecutor(max_workers=1)
        self._next_train = None
        self.summary_writer = None
        if config.env == "tensorboard" and ctx.is_rank0:
            from tensorboardX import SummaryWriter  # optional dep

            self.summary_writer = SummaryWriter()

    # ------------------------------------------------------------------
    def _emit_metric(self, name: str, value, epoch: Optional[int] = None) -> None:
        """Reference metric-line contract (main.py:183-205)."""
        if not self.ctx.is_rank0:
            return
        line = '{{"metric": "{0}", "value": {1}}}'.format(name, value)
        if self.cfg.env == "floyd":
            print(line)
        else:
            logger.info(line)
        if self.summary_writer is not None and epoch is not None:
            self.summary_writer.add_scalar("metric/" + name, value, epoch)

    # ------------------------------------------------------------------
    def train(self) -> float:
        """Epoch loop (reference _train, main.py:143-248).
        Returns 1 - best-epoch F1 (the optuna objective)."""
        cfg = self.cfg
        f1 = 0.0
        best_f1 = None
        last_loss = None
        last_accuracy = None
        bad_count = 0

        try:
            for epoch in range(cfg.max_epoch):
                t0 = time.perf_counter()
                train_loss, n_contexts = self._train_epoch(epoch)
                train_loss = all_reduce_sum_scalar(train_loss, self.ctx)
                n_contexts = all_reduce_sum_scalar(n_contexts, self.ctx)
                dt = time.perf_counter() - t0

                test_loader = self._make_loader(train=False, epoch=epoch)
                test_loss, accuracy, precision, recall, f1 = self.evaluate(test_loader)

                if self.ctx.is_rank0:
                    if self.cfg.env == "floyd":
                        print("epoch {0}".format(epoch))
                    else:
                        logger.info("epoch %d", epoch)
                self._emit_metric("train_loss", train_loss, epoch)
                self._emit_metric("test_loss", test_loss, epoch)
                self._emit_metric("accuracy", accuracy, epoch)
                self._emit_metric("precision", precision, epoch)
                self._emit_metric("recall", recall, epoch)
                self._emit_metric("f1", f1, epoch)
                self._emit_metric(
                    "path_contexts_per_sec", n_contexts / dt if dt > 0 else 0.0, epoch
                )

                if self.trial is not None:
                    import optuna

                    self.trial.report(1.0 - f1, epoch)
                    if self.trial.should_prune():
                        raise optuna.TrialPruned()

                if (
                    epoch > 1
                    and cfg.print_sample_cycle > 0
                    and epoch % cfg.print_sample_cycle == 0
                    and self.trial is None
                    and self.ctx.is_rank0
                ):
                    print_sample(self.reader, self.model, test_loader,
                                 self.option, self.device)

                if best_f1 is None or best_f1 < f1:
                    self._emit_metric("best_f1", f1, epoch)
                    best_f1 = f1
                    if (self.trial is None and self.ctx.is_rank0
                            and not cfg.no_artifact_export):
                        self._export_best(epoch)

                # early stop: OR-logic reset (reference main.py:233-242)
                if (
                    last_loss is None
                    or train_loss < last_loss
                    or last_accuracy is None
                    or last_accuracy < accuracy
                ):
                    last_loss = train_loss
                    last_accuracy = accuracy
                    bad_count = 0
                else:
                    bad_count += 1
                if bad_count > 10:
                    if self.ctx.is_rank0:
                        print(
                            "early stop loss:{0}, bad:{1}".format(train_loss, bad_count)
                        )
                        print_sample(self.reader, self.model, test_loader,
                                     self.option, self.device)
                    break
        finally:
            if self.summary_writer is not None:
                self.summary_writer.close()
            # drain any orphaned background epoch build (early stop can
            # leave one in flight) and retire the worker thread
            pending = self._next_train
            if pending is not None:
                try:
                    pending[1].result()
                except Exception:  # noqa: BLE001 - build died with train
                    pass
                self._next_train = None
            self._build_pool.shutdown(wait=True)
            # deregister the owned-grad callbacks: a later model in the
            # same process could reuse this model's table addresses and
            # silently hand its grads to a stale callback otherwise
            self.ddp.close()

        return 1.0 - f1

    # ------------------------------------------------------------------
    def _make_loader(self, train: bool, epoch: int,
                     keep_pending: bool = False) -> BatchIterator:
        if train:
            pending = self._next_train
            if pending is not None and pending[0] == epoch:
                data = pending[1].result()
                self._next_train = None
            else:
                if pending is not None and not keep_pending:
                    # await before replacing: the orphaned background build
                    # writes the same (tag, parity) pinned pool a synchronous
                    # rebuild of pending[0] would also write
                    pending[1].result()
                    self._next_train = None
                # keep_pending: the export path rebuilds THIS epoch (parity
                # epoch%2) while the pending build writes parity (epoch+1)%2
                # — disjoint pools, safe to run concurrently
                data = self.builder.refresh_train_dataset(epoch)
        else:
            data = self.builder.refresh_test_dataset(epoch)
        return BatchIterator(
            data,
            batch_size=self.cfg.batch_size,
            shuffle=True,
            seed=self.cfg.random_seed + (0 if train else 104729) + epoch,
            device=self.device,
        )

    def _train_epoch(self, epoch: int):
        loader = self._make_loader(train=True, epoch=epoch)
        # kick off next epoch's rebuild in the background.  NOTE: the
        # builder's pinned pool is double-buffered per (tag, parity) so the
        # background build never overwrites the epoch currently training.
        self._next_train = (
            epoch + 1,
            self._build_pool.submit(self.builder.refresh_train_dataset,
                                    epoch + 1),
        )
        model = self.model
        model.train()
        total_loss = torch.zeros((), dtype=torch.float64, device=self.device)
        n_contexts = 0
        self.roctx.push(f"epoch_{epoch}")
        for batch in loader:
            self.roctx.push("step")
            starts = batch["starts"].to(self.device)
            paths = batch["paths"].to(self.device)
            ends = batch["ends"].to(self.device)
            label = batch["label"].to(self.device)

            self.ddp.zero_grad()
            self.roctx.push("forward")
            outputs, _, _ = model(starts, paths, ends, label)
            loss = model.loss(outputs, label, self.class_weight)
            self.roctx.pop()
            self.roctx.push("backward")
            loss.backward()
            self.roctx.pop()
            self.roctx.push("optimizer")
            if self._overlapped_step:
                # comm-overlapped: non-table Adam + chunkwise table Adam
                # pipelined against the in-flight table all-reduces
                self.ddp.finish_and_step(self.optimizer)
            else:
                self.ddp.finish()
                self.optimizer.step()
            self.roctx.pop()
            self.roctx.pop()

            # deferred loss reduction: no per-step .item() sync
            total_loss += loss.detach().double()
            n_contexts += starts.numel()
        self.roctx.pop()
        return float(total_loss.item()), float(n_contexts)

    # ------------------------------------------------------------------
    def evaluate(self, loader: BatchIterator):
        """reference test() (main.py:267-297), DP-aware: subtoken counts are
        all-reduced; exact-match labels are gathered to rank 0."""
        model = self.model
        model.eval()
        loss_acc = torch.zeros((), dtype=torch.float64, device=self.device)
        expected = []
        actual = []
        with torch.no_grad():
            for batch in loader:
                starts = batch["starts"].to(self.device)
                paths = batch["paths"].to(self.device)
                ends = batch["ends"].to(self.device)
                label = batch["label"].to(self.device)
                outputs, _, _ = model(starts, paths, ends, label)
                loss = model.loss(outputs, label, self.class_weight)
                # deferred accumulation — no per-batch .item()/.cpu() sync
                loss_acc += loss.detach().double()
                expected.append(label)
                actual.append(Fn.row_max_argmax(outputs.detach())[1])

        if expected:
            expected = torch.cat(expected).cpu().tolist()
            actual = torch.cat(actual).cpu().tolist()
        test_loss = all_reduce_sum_scalar(float(loss_acc.item()), self.ctx)
        method = self.cfg.eval_method
        if method == "subtoken":
            m, ec, ac = M.subtoken_match_counts(expected, actual, self.reader.label_vocab)
            m, ec, ac = all_reduce_sum_list([m, ec, ac], self.ctx)
            stats = M.subtoken_stats_from_counts(m, ec, ac)
        elif method == "exact":
            # gather shards (small host lists)
            if self.ctx.initialized:
                import torch.distributed as dist

                gathered_e = [None] * self.ctx.world_size
                gathered_a = [None] * self.ctx.world_size
                dist.all_gather_object(gathered_e, expected)
                dist.all_gather_object(gathered_a, actual)
                expected = [x for lst in gathered_e for x in lst]
                actual = [x for lst in gathered_a for x in lst]
            stats = M.exact_match(expected, actual)
        elif method == "ave_subtoken":
            if self.ctx.initialized:
                import torch.distributed as dist

                gathered_e = [None] * self.ctx.world_size
                gathered_a = [None] * self.ctx.world_size
                dist.all_gather_object(gathered_e, expected)
                dist.all_gather_object(gathered_a, actual)
                expected = [x for lst in gathered_e for x in lst]
                actual = [x for lst in gathered_a for x in lst]
            stats = M.averaged_subtoken_match(expected, actual, self.reader.label_vocab)
        else:
            raise ValueError(f"unknown eval_method {method}")
        accuracy, precision, recall, f1 = stats
        return test_loss, accuracy, precision, recall, f1

    # ------------------------------------------------------------------
    def _export_best(self, epoch: int) -> None:
        """Best-F1 artifacts (reference main.py:216-231): code.vec header +
        train + test vectors + checkpoint."""
        cfg = self.cfg
        os.makedirs(osp.dirname(osp.abspath(cfg.vectors_path)), exist_ok=True)
        os.makedirs(cfg.model_path, exist_ok=True)
        write_vector_header(
            cfg.vectors_path, len(self.reader.items), self.option.encode_size
        )
        train_loader = self._make_loader(train=True, epoch=epoch,
                                         keep_pending=True)
        test_loader = self._make_loader(train=False, epoch=epoch)
        # hipGraph-captured forward for the batched export (config 5)
        from .infer import graphed_export_forward

        graphed = graphed_export_forward(self.model, self.cfg.batch_size,
                                         self.device)
        write_code_vectors(
            self.reader, self.model, train_loader, self.option,
            cfg.vectors_path, "a", None, self.device, graphed=graphed,
        )
        write_code_vectors(
            self.reader, self.model, test_loader, self.option,
            cfg.vectors_path, "a", cfg.test_result_path, self.device,
            graphed=graphed,
        )
        if isinstance(self.model, Code2VecHIP):
            sd = self.model.reference_state_dict()
        else:
            sd = reference_state_dict_torch(self.model)
        torch.save(sd, osp.join(cfg.model_path, "code2vec.model"))
