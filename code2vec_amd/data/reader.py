"""Corpus reader — parses the path-context corpus format.

Input contract (reference model/dataset_reader.py:72-128, format written by
create_path_contexts.ipynb cell 11): blank-line-separated records with line
prefixes ``#<id>``, ``label:``, ``class:``, ``paths:`` (mode 1 ->
``start\\tpath\\tend`` int triples), ``vars:`` (mode 2 ->
``originalName\\talias`` pairs), ``doc:`` (discarded).

The reader stores path-contexts in flat numpy arrays (one big int32 triple
array + per-item offsets) instead of per-item Python lists, so the per-epoch
resampling pass (builder.py) is vectorizable and shardable across DP ranks.
"""

from __future__ import annotations

import logging
from dataclasses import dataclass, field
from typing import Dict, List, Optional

import numpy as np

from .vocab import (
    QUESTION_TOKEN_INDEX,
    QUESTION_TOKEN_NAME,
    Vocab,
    get_method_subtokens,
    normalize_method_name,
    read_vocab_file,
)

logger = logging.getLogger(__name__)

try:  # native C++ corpus parser (GIL-released single pass)
    import torch as _torch  # noqa: F401  (loads libc10 for the extension)
    from . import _c2v_host as _native
except Exception:  # noqa: BLE001
    _native = None


@dataclass
class CodeItem:
    """One method's record (reference model/dataset.py:40-49)."""

    id: Optional[int] = None
    label: Optional[str] = None
    normalized_label: Optional[str] = None
    source: Optional[str] = None
    aliases: Dict[str, str] = field(default_factory=dict)
    # [n_contexts, 3] int32 (start, path, end); question-token offset applied.
    path_contexts: np.ndarray = field(
        default_factory=lambda: np.empty((0, 3), dtype=np.int32)
    )


class CorpusReader:
    """Reads vocab files + corpus, building the label vocab on the fly.

    Mirrors reference model/dataset_reader.py:44-128:
    - path vocab read with no extra tokens (:48),
    - terminal vocab read with extra_tokens=["@question"] (:51),
    - ``variable_indexes`` = terminal indexes of names starting "@var_" (:54-56),
    - corpus start/end terminal indexes shifted by +QUESTION_TOKEN_INDEX
      (:112-115); path indexes are NOT shifted,
    - label normalization/lowercasing and label-vocab build (:97-102).
    """

    def __init__(
        self,
        corpus_path: str,
        path_index_path: str,
        terminal_index_path: str,
        infer_method: bool = True,
        infer_variable: bool = False,
        shuffle_variable_indexes: bool = False,
    ) -> None:
        self.path_vocab = read_vocab_file(path_index_path)
        logger.info("path vocab size: %d", len(self.path_vocab))

        self.terminal_vocab = read_vocab_file(
            terminal_index_path, extra_tokens=[QUESTION_TOKEN_NAME]
        )
        logger.info("terminal vocab size: %d", len(self.terminal_vocab))

        self.variable_indexes = [
            idx for term, idx in self.terminal_vocab.stoi.items()
            if term.startswith("@var_")
        ]
        logger.info("variable index size: %d", len(self.variable_indexes))

        self.shuffle_variable_indexes = shuffle_variable_indexes
        self.QUESTION_TOKEN_NAME = QUESTION_TOKEN_NAME
        self.QUESTION_TOKEN_INDEX = QUESTION_TOKEN_INDEX
        self.infer_method = infer_method
        self.infer_variable = infer_variable

        self.label_vocab = Vocab()
        self.items: List[CodeItem] = []
        if _native is not None:
            self._load_native(corpus_path)
        else:
            self._load(corpus_path)
        self._validate_indices()

        logger.info("label vocab size: %d", len(self.label_vocab))
        logger.info("corpus: %d", len(self.items))

    def _validate_indices(self) -> None:
        """Fail loudly on out-of-range corpus indices at load time.

        The replicated vocab quirk (duplicate names in an index file are
        skipped, so a file's max index can exceed len(vocab)-1) makes
        out-of-range indices reachable with real data.  The reference
        fails in nn.Embedding (table rows == len(vocab)); our HIP gather
        must never see such an index, so we validate once on the host
        here instead of bounds-checking in kernels."""
        t_hi = len(self.terminal_vocab)
        p_hi = len(self.path_vocab)
        flat = getattr(self, "_flat_contexts", None)
        if flat is not None and flat.shape[0] > 0:
            m = flat.max(axis=0)  # native path: one pass over the flat array
            s_max, p_max, e_max = int(m[0]), int(m[1]), int(m[2])
            lo = int(flat.min())
        else:
            s_max = p_max = e_max = -1
            lo = 0
            for it in self.items:
                pcs = it.path_contexts
                if pcs.shape[0] == 0:
                    continue
                m = pcs.max(axis=0)
                s_max = max(s_max, int(m[0]))
                p_max = max(p_max, int(m[1]))
                e_max = max(e_max, int(m[2]))
                lo = min(lo, int(pcs.min()))
        if lo < 0 or max(s_max, e_max) >= t_hi or p_max >= p_hi:
            raise ValueError(
                "corpus contains out-of-range indices: "
                f"start/end max {max(s_max, e_max)} (terminal vocab {t_hi} "
                f"rows incl. @question shift), path max {p_max} (path vocab "
                f"{p_hi} rows), min {lo} (must be >= 0). The corpus does "
                "not match the given index files."
            )

    def _load_native(self, corpus_path: str) -> None:
        """Native parse (data/csrc/epoch_builder.cpp::parse_corpus) +
        Python-side label normalization/vocab build (regex parity with the
        reference); path_contexts become zero-copy views of one flat
        array."""
        res = _native.parse_corpus(corpus_path)
        self._flat_contexts = res["contexts"].numpy()
        ids = res["ids"].numpy()
        offsets = res["offsets"].numpy()
        voffsets = res["var_offsets"].numpy()
        contexts = res["contexts"].numpy()
        labels = res["labels"]
        sources = res["sources"]
        vorig = res["var_originals"]
        valias = res["var_aliases"]
        label_vocab = self.label_vocab
        infer_method = self.infer_method
        infer_variable = self.infer_variable
        norm_cache = {}

        def normalized(raw: str):
            hit = norm_cache.get(raw)
            if hit is None:
                n = normalize_method_name(raw)
                hit = (n.lower(), get_method_subtokens(n))
                norm_cache[raw] = hit
            return hit

        for i in range(len(ids)):
            item = CodeItem()
            item.id = int(ids[i]) if ids[i] >= 0 else None
            raw = labels[i]
            if raw:
                item.label = raw
                lower, subtokens = normalized(raw)
                item.normalized_label = lower
                if infer_method:
                    label_vocab.append(lower, subtokens=subtokens)
            item.source = sources[i] or None
            item.path_contexts = contexts[offsets[i]:offsets[i + 1]]
            for v in range(voffsets[i], voffsets[i + 1]):
                alias_name = valias[v]
                lower, subtokens = normalized(vorig[v])
                item.aliases[alias_name] = lower
                if infer_variable and alias_name.startswith("@var_"):
                    label_vocab.append(lower, subtokens=subtokens)
            self.items.append(item)

    def _load(self, corpus_path: str) -> None:
        items = self.items
        label_vocab = self.label_vocab
        infer_method = self.infer_method
        infer_variable = self.infer_variable
        q = QUESTION_TOKEN_INDEX

        cur: Optional[CodeItem] = None
        triples: List[int] = []  # flat start,path,end ints for the current item
        parse_mode = 0

        def flush() -> None:
            nonlocal cur, triples
            if cur is not None:
                arr = np.asarray(triples, dtype=np.int32).reshape(-1, 3)
                cur.path_contexts = arr
                items.append(cur)
            cur = None
            triples = []

        with open(corpus_path, mode="r", encoding="utf-8") as f:
            for lineno, raw in enumerate(f, start=1):
                line = raw.strip(" \r\n\t")
                if line == "":
                    flush()
                    continue
                if cur is None:
                    cur = CodeItem()
                    parse_mode = 0
                if line.startswith("#"):
                    cur.id = int(line[1:])
                elif line.startswith("label:"):
                    label = line[6:]
                    cur.label = label
                    normalized = normalize_method_name(label)
                    subtokens = get_method_subtokens(normalized)
                    lower = normalized.lower()
                    cur.normalized_label = lower
                    if infer_method:
                        label_vocab.append(lower, subtokens=subtokens)
                elif line.startswith("class:"):
                    cur.source = line[6:]
                elif line.startswith("paths:"):
                    parse_mode = 1
                elif line.startswith("vars:"):
                    parse_mode = 2
                elif line.startswith("doc:"):
                    pass  # parsed and discarded (reference :109-110)
                elif parse_mode == 1:
                    try:
                        s, p, e = line.split("\t")
                        triples.extend((int(s) + q, int(p), int(e) + q))
                    except ValueError as exc:
                        raise ValueError(
                            f"malformed path-context line {lineno} in "
                            f"{corpus_path}"
                        ) from exc
                elif parse_mode == 2:
                    fields = line.split("\t")
                    if len(fields) < 2:
                        raise ValueError(
                            f"malformed vars line {lineno} in {corpus_path}"
                        )
                    original_name, alias_name = fields[:2]
                    normalized = normalize_method_name(original_name)
                    subtokens = get_method_subtokens(normalized)
                    lower = normalized.lower()
                    cur.aliases[alias_name] = lower
                    if infer_variable and alias_name.startswith("@var_"):
                        label_vocab.append(lower, subtokens=subtokens)
        flush()
