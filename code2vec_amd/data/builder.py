"""Per-epoch dataset construction — seeded, rank-aware.

Replicates the *distribution* of the reference's per-epoch resampling
(reference model/dataset_builder.py:112-210): every epoch, each method's
path-context bag is re-shuffled and truncated to ``max_path_length``
contexts (implicit data augmentation that is load-bearing for F1), the
``@method_0`` terminal is replaced by ``@question``, and the three index
sequences are zero-padded.

Differences from the reference, by design:
- the reference's ``random`` module is never seeded (its 80/20 split and
  resampling differ run-to-run); here all randomness derives from
  ``(seed, epoch, rank)`` so DP replicas agree and runs reproduce,
- items are sharded across DP ranks per epoch,
- outputs are int32 numpy arrays (half the H2D bytes of the reference's
  int64 tensors); the model casts as needed.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass
from typing import List, Optional, Sequence

import numpy as np

from .reader import CodeItem, CorpusReader

logger = logging.getLogger(__name__)

try:  # native (C++/OpenMP) epoch builder — falls back to numpy if absent
    import torch as _torch
    from . import _c2v_host as _native
except Exception:  # noqa: BLE001
    _native = None


@dataclass
class EpochData:
    """One epoch's tensors for one rank (cf. reference CodeDataset rows)."""

    ids: List[int]
    starts: np.ndarray  # [N, C] int32
    paths: np.ndarray   # [N, C] int32
    ends: np.ndarray    # [N, C] int32
    labels: np.ndarray  # [N] int64

    def __len__(self) -> int:
        return self.starts.shape[0]


def _filter_variable_aliases(aliases: dict) -> List[str]:
    return [a for a in aliases if a.startswith("@var_")]


class DatasetBuilder:
    """Train/test split + per-epoch tensor construction.

    Mirrors reference model/dataset_builder.py:19-62 with seeded RNG.
    """

    def __init__(
        self,
        reader: CorpusReader,
        option,
        split_ratio: float = 0.2,
        seed: int = 123,
        rank: int = 0,
        world_size: int = 1,
    ) -> None:
        self.reader = reader
        self.option = option
        self.seed = seed
        self.rank = rank
        self.world_size = world_size

        rng = np.random.default_rng([seed, 0xC0DE])
        order = rng.permutation(len(reader.items))
        test_count = int(len(reader.items) * split_ratio)
        self.test_items = [reader.items[i] for i in order[:test_count]]
        self.train_items = [reader.items[i] for i in order[test_count:]]
        logger.info("train item size: %d", len(self.train_items))
        logger.info("test item size: %d", len(self.test_items))

        train_sz, test_sz = self._dataset_sizes()
        logger.info("train dataset size: %d", train_sz)
        logger.info("test dataset size: %d", test_sz)

        self.train_dataset: Optional[EpochData] = None
        self.test_dataset: Optional[EpochData] = None
        self._flat_cache = {}

        logger.info("OOV rate: %s", self.out_of_vocabulary_rate())

    # ------------------------------------------------------------------
    def _dataset_sizes(self):
        train_sz = test_sz = 0
        if self.reader.infer_method:
            train_sz += len(self.train_items)
            test_sz += len(self.test_items)
        if self.reader.infer_variable:
            train_sz += sum(
                len(_filter_variable_aliases(it.aliases)) for it in self.train_items
            )
            test_sz += sum(
                len(_filter_variable_aliases(it.aliases)) for it in self.test_items
            )
        return train_sz, test_sz

    def _get_labels(self, normalized_label: str) -> Sequence[str]:
        """Label subtokens for OOV accounting (reference :64-71)."""
        if self.option.eval_method == "exact":
            return [normalized_label]
        idx = self.reader.label_vocab.stoi[normalized_label]
        return self.reader.label_vocab.itosubtokens[idx]

    def out_of_vocabulary_rate(self) -> float:
        """Fraction of test label subtokens unseen in train labels
        (reference model/dataset_builder.py:72-110)."""
        train_vocab = set()
        match = 0
        count = 0
        if self.reader.infer_method:
            for it in self.train_items:
                train_vocab.update(self._get_labels(it.normalized_label))
        if self.reader.infer_variable:
            for it in self.train_items:
                for alias in _filter_variable_aliases(it.aliases):
                    train_vocab.update(self._get_labels(it.aliases[alias]))
        if self.reader.infer_method:
            for it in self.test_items:
                tokens = self._get_labels(it.normalized_label)
                match += sum(1 for t in tokens if t in train_vocab)
                count += len(tokens)
        if self.reader.infer_variable:
            for it in self.test_items:
                for alias in _filter_variable_aliases(it.aliases):
                    tokens = self._get_labels(it.aliases[alias])
                    match += sum(1 for t in tokens if t in train_vocab)
                    count += len(tokens)
        if count == 0:
            return 0.0
        return 1.0 - match / count

    # ------------------------------------------------------------------
    def refresh_train_dataset(self, epoch: int = 0) -> EpochData:
        """Rebuild the train tensors for ``epoch`` on this rank's shard."""
        self.train_dataset = self._refresh(self.train_items, "train", epoch, 1)
        return self.train_dataset

    def refresh_test_dataset(self, epoch: int = 0) -> EpochData:
        self.test_dataset = self._refresh(self.test_items, "test", epoch, 2)
        return self.test_dataset

    def _refresh(self, base: List[CodeItem], tag: str, epoch: int,
                 stream: int) -> EpochData:
        shard_idx = self._shard_indices(len(base), epoch,
                                        equal_shards=(tag == "train"))
        if (
            _native is not None
            and self.reader.infer_method
            and not self.reader.infer_variable
        ):
            # pool double-buffered by epoch parity: epoch N+1 can build in a
            # background thread while N's buffers feed H2D copies
            return self._build_native(base, tag, f"{tag}{epoch % 2}",
                                      shard_idx, epoch, stream)
        items = [base[i] for i in shard_idx]
        data = self.build_data(items, self.option.max_path_length,
                               epoch=epoch, stream=stream)
        if tag == "train" and self.world_size > 1 and self.reader.infer_variable:
            # variable task: rows per item vary, so equal ITEM counts do not
            # give equal ROW (=> batch) counts across ranks.  Every rank can
            # compute every rank's row count from the shared reader state, so
            # truncate to the global min deterministically — no collective.
            data = self._truncate_rows(data, base, epoch)
        return data

    def _rows_per_item(self, base: List[CodeItem]) -> np.ndarray:
        key = ("rows", id(base))
        cached = self._flat_cache.get(key)
        if cached is None:
            m = 1 if self.reader.infer_method else 0
            cached = np.fromiter(
                (m + len(_filter_variable_aliases(it.aliases)) for it in base),
                count=len(base), dtype=np.int64,
            )
            self._flat_cache[key] = cached
        return cached

    def _truncate_rows(self, data: EpochData, base: List[CodeItem],
                       epoch: int) -> EpochData:
        rows = self._rows_per_item(base)
        rng = np.random.default_rng([self.seed, epoch, 0x5A5A])
        order = rng.permutation(len(base))
        per = len(base) // self.world_size
        min_rows = min(
            int(rows[order[r :: self.world_size][:per]].sum())
            for r in range(self.world_size)
        )
        if len(data) <= min_rows:
            return data
        return EpochData(
            ids=data.ids[:min_rows],
            starts=data.starts[:min_rows],
            paths=data.paths[:min_rows],
            ends=data.ends[:min_rows],
            labels=data.labels[:min_rows],
        )

    def _shard_indices(self, n: int, epoch: int,
                       equal_shards: bool = False) -> np.ndarray:
        """Per-rank item indexes for one epoch.

        Train shards are truncated to a common size floor(n/world): every
        rank then runs the SAME number of batches per epoch, so the
        per-step gradient all-reduces stay in lockstep (a rank with one
        extra batch would fire collectives no peer matches — deadlock).
        The dropped tail (< world items) rotates with the epoch-seeded
        permutation, so all items are still seen across epochs.  Test
        shards keep every item (eval has no in-loop collectives)."""
        if self.world_size <= 1:
            return np.arange(n)
        rng = np.random.default_rng([self.seed, epoch, 0x5A5A])
        order = rng.permutation(n)
        if equal_shards:
            per = n // self.world_size
            return order[self.rank :: self.world_size][:per]
        return order[self.rank :: self.world_size]

    # ------------------------------------------------------------------
    def _get_flat(self, base: List[CodeItem], tag: str):
        """Flatten a split's contexts once: offsets, [total,3] contexts,
        labels, ids (feeds the C++ OpenMP epoch builder)."""
        cached = self._flat_cache.get(tag)
        if cached is not None:
            return cached
        counts = np.fromiter((it.path_contexts.shape[0] for it in base),
                             count=len(base), dtype=np.int64)
        offsets = np.zeros(len(base) + 1, dtype=np.int64)
        np.cumsum(counts, out=offsets[1:])
        contexts = (
            np.concatenate([it.path_contexts for it in base], axis=0)
            if len(base) else np.empty((0, 3), dtype=np.int32)
        )
        label_stoi = self.reader.label_vocab.stoi
        labels = np.fromiter(
            (label_stoi[it.normalized_label] for it in base),
            count=len(base), dtype=np.int64,
        )
        ids = [it.id for it in base]
        cached = (
            _torch.from_numpy(offsets),
            _torch.from_numpy(np.ascontiguousarray(contexts)),
            labels,
            ids,
        )
        self._flat_cache[tag] = cached
        return cached

    def _pinned_pool(self, tag: str, N: int, C: int):
        """Persistent (optionally pinned) per-epoch output buffers: the
        native builder writes straight into them, so the loader's H2D path
        needs no per-epoch pin_memory copy of ~GB index arrays."""
        key = ("pool", tag, N, C)
        bufs = self._flat_cache.get(key)
        if bufs is None:
            pin = _torch.cuda.is_available()
            bufs = tuple(
                _torch.empty(N, C, dtype=_torch.int32, pin_memory=pin)
                for _ in range(3)
            )
            self._flat_cache[key] = bufs
        elif _torch.cuda.is_available():
            # the previous epoch's async H2D copies may still read these
            # buffers; fence before overwriting (NOTE: an EpochData aliases
            # its pool — it is invalidated by the next refresh of its split)
            _torch.cuda.synchronize()
        return bufs

    def _build_native(self, base, flat_tag, pool_tag, shard_idx, epoch,
                      stream) -> EpochData:
        offsets_t, contexts_t, labels, ids = self._get_flat(base, flat_tag)
        C = self.option.max_path_length
        N = len(shard_idx)
        item_idx = _torch.from_numpy(np.ascontiguousarray(shard_idx, dtype=np.int64))
        starts, paths, ends = self._pinned_pool(pool_tag, N, C)
        seed = hash((self.seed, epoch, stream, self.rank)) & 0x7FFFFFFFFFFFFFFF
        # -1 when the corpus has no @method_0 occurrences (matches nothing)
        method_token_index = self.reader.terminal_vocab.stoi.get(
            "@method_0", -1)
        _native.build_method_epoch(
            offsets_t, contexts_t, item_idx, starts, paths, ends,
            method_token_index, self.reader.QUESTION_TOKEN_INDEX, seed,
        )
        return EpochData(
            ids=[ids[i] for i in shard_idx],
            starts=starts.numpy(),
            paths=paths.numpy(),
            ends=ends.numpy(),
            labels=labels[shard_idx],
        )

    # ------------------------------------------------------------------
    def build_data(
        self,
        items: List[CodeItem],
        max_path_length: int,
        epoch: int = 0,
        stream: int = 1,
    ) -> EpochData:
        """Resample + pad one epoch's tensors.

        Method-name task (reference :122-150): shuffle the item's bag, keep
        the first ``max_path_length`` contexts, replace @method_0 with
        @question in start/end, pad with 0.
        Variable-name task (reference :152-204): one sample per @var alias,
        contexts filtered to those touching the variable.
        """
        C = max_path_length
        reader = self.reader
        rng = np.random.default_rng([self.seed, epoch, stream, self.rank])
        q = reader.QUESTION_TOKEN_INDEX

        ids: List[int] = []
        labels: List[int] = []
        starts_rows: List[np.ndarray] = []
        paths_rows: List[np.ndarray] = []
        ends_rows: List[np.ndarray] = []

        if reader.infer_method:
            method_token_index = reader.terminal_vocab.stoi.get(
                "@method_0", -1)
            label_stoi = reader.label_vocab.stoi
            for item in items:
                pcs = item.path_contexts
                n = pcs.shape[0]
                if n > C:
                    sel = rng.permutation(n)[:C]
                    chosen = pcs[sel]
                else:
                    chosen = pcs[rng.permutation(n)] if n > 1 else pcs
                s = chosen[:, 0].copy()
                p = chosen[:, 1]
                e = chosen[:, 2].copy()
                s[s == method_token_index] = q
                e[e == method_token_index] = q
                ids.append(item.id)
                labels.append(label_stoi[item.normalized_label])
                starts_rows.append(s)
                paths_rows.append(np.asarray(p))
                ends_rows.append(e)

        if reader.infer_variable:
            self._build_variable_rows(
                items, C, rng, ids, labels, starts_rows, paths_rows, ends_rows
            )

        N = len(starts_rows)
        starts = np.zeros((N, C), dtype=np.int32)
        paths = np.zeros((N, C), dtype=np.int32)
        ends = np.zeros((N, C), dtype=np.int32)
        for i, (s, p, e) in enumerate(zip(starts_rows, paths_rows, ends_rows)):
            k = min(len(s), C)
            starts[i, :k] = s[:k]
            paths[i, :k] = p[:k]
            ends[i, :k] = e[:k]
        return EpochData(
            ids=ids,
            starts=starts,
            paths=paths,
            ends=ends,
            labels=np.asarray(labels, dtype=np.int64),
        )

    def _build_variable_rows(
        self, items, C, rng, ids, labels, starts_rows, paths_rows, ends_rows
    ) -> None:
        reader = self.reader
        q = reader.QUESTION_TOKEN_INDEX
        term_stoi = reader.terminal_vocab.stoi
        label_stoi = reader.label_vocab.stoi
        variable_indexes = np.asarray(reader.variable_indexes, dtype=np.int32)
        # identity remap unless shuffling is on (reference :157-164)
        remap = {int(v): int(v) for v in variable_indexes}
        for item in items:
            alias_names = _filter_variable_aliases(item.aliases)
            if not alias_names:
                continue
            alias_indexes = [term_stoi[a] for a in alias_names]
            if reader.shuffle_variable_indexes:
                perm = rng.permutation(len(variable_indexes))
                remap = {
                    int(k): int(variable_indexes[perm[i]])
                    for i, k in enumerate(variable_indexes)
                }
            pcs = item.path_contexts
            alias_set = np.isin(pcs[:, 0], alias_indexes) | np.isin(
                pcs[:, 2], alias_indexes
            )
            var_pcs = pcs[alias_set]
            var_pcs = var_pcs[rng.permutation(var_pcs.shape[0])]
            for alias_name, var_idx in zip(alias_names, alias_indexes):
                mask = (var_pcs[:, 0] == var_idx) | (var_pcs[:, 2] == var_idx)
                sel = var_pcs[mask]
                s = sel[:, 0].copy()
                p = sel[:, 1]
                e = sel[:, 2].copy()
                for arr in (s, e):
                    other = arr != var_idx
                    arr[~other] = q
                    if reader.shuffle_variable_indexes:
                        for j in np.nonzero(other)[0]:
                            arr[j] = remap.get(int(arr[j]), int(arr[j]))
                ids.append(item.id)
                labels.append(label_stoi[item.aliases[alias_name]])
                starts_rows.append(s[:C])
                paths_rows.append(np.asarray(p[:C]))
                ends_rows.append(e[:C])
