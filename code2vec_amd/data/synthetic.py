"""Synthetic corpus / vocab generation in the reference's file formats.

Two uses:
- tests: write a small corpus + vocab files and round-trip them through
  CorpusReader (format spec: reference model/dataset_reader.py:72-128 and
  create_path_contexts.ipynb cell 11),
- benchmarking: generate batched index tensors of a named shape
  (BASELINE.json configs) directly, skipping the file round trip.
"""

from __future__ import annotations

import os
from dataclasses import dataclass

import numpy as np

_SYLLABLES = [
    "get", "set", "run", "read", "write", "parse", "load", "store", "make",
    "build", "find", "count", "list", "map", "node", "tree", "path", "token",
    "value", "index", "name", "file", "data", "item", "meta", "graph",
]


def _method_name(rng: np.random.Generator) -> str:
    k = int(rng.integers(1, 4))
    parts = [str(rng.choice(_SYLLABLES))]
    for _ in range(k):
        s = str(rng.choice(_SYLLABLES))
        parts.append(s.capitalize())
    return "".join(parts)


# Heaps-law parameters fit to the prefix-unique-label curve of the
# reference's REAL method list (dataset/methods.txt: 9,916 methods,
# 2,280 unique normalized names): V(n) = K * n^beta.  Evaluated at
# top11's 605,945 methods (top11_dataset/params.txt:9) this predicts
# ~72.4k unique normalized labels — the label-vocab scale the synthetic
# corpus reproduces (tools/derive_label_vocab.py shows the derivation).
HEAPS_K = 1.01
HEAPS_BETA = 0.840


def name_stream(rng: np.random.Generator, n: int):
    """Yield n method names whose unique-normalized-count growth follows
    the Heaps law fit above: method i mints a NEW name with probability
    dV/dn = K*beta*i^(beta-1), else reuses an earlier OCCURRENCE drawn
    uniformly (preferential attachment -> Zipfian name reuse, like real
    Java corpora full of get/set/run variants)."""
    seen = set()
    draws: list = []

    def mint() -> str:
        for _ in range(64):
            nm = _method_name(rng)
            if nm.lower() not in seen:
                return nm
        # syllable space crowded: extend until unique
        nm = _method_name(rng)
        while nm.lower() in seen:
            nm += str(rng.choice(_SYLLABLES)).capitalize()
        return nm

    for i in range(1, n + 1):
        p_new = min(1.0, HEAPS_K * HEAPS_BETA * float(i) ** (HEAPS_BETA - 1.0))
        if not draws or rng.random() < p_new:
            nm = mint()
            seen.add(nm.lower())
        else:
            nm = draws[int(rng.integers(0, len(draws)))]
        draws.append(nm)
        yield nm


@dataclass
class SyntheticSpec:
    n_methods: int = 64
    n_terminals: int = 200        # file-side terminal vocab size (excl. PAD)
    n_paths: int = 300            # file-side path vocab size (excl. PAD)
    max_contexts: int = 40        # contexts per method (uniform 1..max)
    n_vars_per_method: int = 2
    seed: int = 1234
    # fraction of a method's contexts drawn from its label's signature
    # pool instead of uniform noise: >0 makes the name PREDICTABLE from
    # the bag of contexts, so training produces a meaningful F1 curve
    label_signal: float = 0.0
    sig_pool: int = 24            # signature triples per label


def write_synthetic_corpus(out_dir: str, spec: SyntheticSpec) -> dict:
    """Write corpus.txt / path_idxs.txt / terminal_idxs.txt in out_dir.

    Terminal index 0 is <PAD/>; @method_0 and @var_N terminals are included
    so the @question replacement path is exercised.  Returns file paths.
    """
    rng = np.random.default_rng(spec.seed)
    os.makedirs(out_dir, exist_ok=True)

    terminal_names = ["<PAD/>", "@method_0"]
    for i in range(spec.n_vars_per_method * 4):
        terminal_names.append(f"@var_{i}")
    i = 0
    while len(terminal_names) < spec.n_terminals:
        terminal_names.append(f"term{i}")
        i += 1
    terminal_path = os.path.join(out_dir, "terminal_idxs.txt")
    with open(terminal_path, "w", encoding="utf-8") as f:
        for idx, name in enumerate(terminal_names):
            f.write(f"{idx}\t{name}\n")

    path_path = os.path.join(out_dir, "path_idxs.txt")
    with open(path_path, "w", encoding="utf-8") as f:
        f.write("0\t<PAD/>\n")
        for idx in range(1, spec.n_paths):
            f.write(f"{idx}\tClassOrInterfaceDeclaration{idx}↑MethodDeclaration↓NameExpr\n")

    import zlib

    sig_cache: dict = {}

    def signature(label: str) -> np.ndarray:
        """Deterministic per-label pool of (s, p, e) triples."""
        pool = sig_cache.get(label)
        if pool is None:
            srng = np.random.default_rng(zlib.crc32(label.encode()))
            pool = np.stack([
                srng.integers(1, spec.n_terminals - 1, spec.sig_pool),
                srng.integers(1, spec.n_paths, spec.sig_pool),
                srng.integers(1, spec.n_terminals - 1, spec.sig_pool),
            ], axis=1)
            sig_cache[label] = pool
        return pool

    var_block = "vars:\n" + "".join(
        f"someVar{v}\t@var_{v}\n" for v in range(spec.n_vars_per_method)
    )
    corpus_path = os.path.join(out_dir, "corpus.txt")
    names = name_stream(rng, spec.n_methods)
    with open(corpus_path, "w", encoding="utf-8", buffering=1 << 22) as f:
        chunk: list = []
        for mid, name in zip(range(spec.n_methods), names):
            n_ctx = int(rng.integers(1, spec.max_contexts + 1))
            # raw file-side terminal indexes (reader adds +1 for @question)
            s = rng.integers(1, spec.n_terminals - 1, n_ctx)
            p = rng.integers(1, spec.n_paths, n_ctx)
            e = rng.integers(1, spec.n_terminals - 1, n_ctx)
            if spec.label_signal > 0.0:
                k = int(round(n_ctx * spec.label_signal))
                if k:
                    pool = signature(name.lower())
                    pick = rng.integers(0, pool.shape[0], k)
                    s[:k], p[:k], e[:k] = (pool[pick, 0], pool[pick, 1],
                                           pool[pick, 2])
            rows = "\n".join(f"{int(a)}\t{int(b)}\t{int(c)}"
                             for a, b, c in zip(s, p, e))
            chunk.append(
                f"#{mid}\nlabel:{name}\nclass:Synth{mid % 7}.java\n"
                f"doc: synthetic method {mid}\npaths:\n{rows}\n{var_block}\n"
            )
            if len(chunk) >= 1024:
                f.write("".join(chunk))
                chunk.clear()
        f.write("".join(chunk))

    return {
        "corpus_path": corpus_path,
        "path_idx_path": path_path,
        "terminal_idx_path": terminal_path,
    }


class _IndexVocab:
    """Minimal Vocab-compatible stand-in for generated name spaces."""

    def __init__(self, names, with_subtokens=False):
        self.stoi = {n: i for i, n in enumerate(names)}
        self.itos = dict(enumerate(names))
        self.itosubtokens = (
            {i: [n] for i, n in enumerate(names)} if with_subtokens else {}
        )
        self.freq = {i: 1 for i in range(len(names))}

    def get_freq_list(self):
        return [self.freq[i] for i in range(len(self.stoi))]

    def __len__(self):
        return len(self.stoi)

    def len(self):  # noqa: A003 - reference API parity
        return len(self.stoi)


class SyntheticReader:
    """In-memory CorpusReader stand-in at benchmark scale — the same
    .items/.vocab surface DatasetBuilder consumes, built directly from
    numpy (no multi-GB corpus file round-trip).  Used by bench.py
    --real-pipeline to route the flagship benchmark through the REAL
    input pipeline (native epoch builder, pinned pools, BatchIterator)."""

    QUESTION_TOKEN_NAME = "@question"
    QUESTION_TOKEN_INDEX = 1

    def __init__(self, n_methods, terminal_count, path_count, label_count,
                 max_contexts=400, seed=0):
        from .reader import CodeItem

        rng = np.random.default_rng(seed)
        term_names = ["<PAD/>", "@question", "@method_0"] + [
            f"t{i}" for i in range(terminal_count - 3)
        ]
        path_names = ["<PAD/>"] + [f"p{i}" for i in range(path_count - 1)]
        label_names = [f"lbl{i}" for i in range(label_count)]
        self.terminal_vocab = _IndexVocab(term_names)
        self.path_vocab = _IndexVocab(path_names)
        self.label_vocab = _IndexVocab(label_names, with_subtokens=True)
        self.variable_indexes = []
        self.shuffle_variable_indexes = False
        self.infer_method = True
        self.infer_variable = False

        counts = rng.integers(1, max_contexts + 1, n_methods)
        offsets = np.zeros(n_methods + 1, dtype=np.int64)
        np.cumsum(counts, out=offsets[1:])
        total = int(offsets[-1])
        flat = np.empty((total, 3), dtype=np.int32)
        flat[:, 0] = rng.integers(2, terminal_count, total)
        flat[:, 1] = rng.integers(1, path_count, total)
        flat[:, 2] = rng.integers(2, terminal_count, total)
        labels = rng.integers(0, label_count, n_methods)
        self.items = [
            CodeItem(id=i, label=label_names[labels[i]],
                     normalized_label=label_names[labels[i]],
                     path_contexts=flat[offsets[i]:offsets[i + 1]])
            for i in range(n_methods)
        ]


def synthetic_batch(
    rng: np.random.Generator,
    batch_size: int,
    contexts: int,
    terminal_count: int,
    path_count: int,
    label_count: int,
    full: bool = True,
):
    """Random index batch of the model's input shape (for benchmarks).

    ``full=True`` gives every method exactly ``contexts`` real contexts (the
    dense case: real corpora at C=200 are usually truncations of larger
    bags).  Returns int32 starts/paths/ends [B,C] and int64 labels [B].
    """
    B, C = batch_size, contexts
    starts = rng.integers(1, terminal_count, size=(B, C), dtype=np.int32)
    paths = rng.integers(1, path_count, size=(B, C), dtype=np.int32)
    ends = rng.integers(1, terminal_count, size=(B, C), dtype=np.int32)
    if not full:
        # ragged: random valid prefix per row, rest padded with 0
        lens = rng.integers(1, C + 1, size=B)
        mask = np.arange(C)[None, :] >= lens[:, None]
        starts[mask] = 0
        paths[mask] = 0
        ends[mask] = 0
    labels = rng.integers(0, label_count, size=B).astype(np.int64)
    return starts, paths, ends, labels
