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


@dataclass
class SyntheticSpec:
    n_methods: int = 64
    n_terminals: int = 200        # file-side terminal vocab size (excl. PAD)
    n_paths: int = 300            # file-side path vocab size (excl. PAD)
    max_contexts: int = 40        # contexts per method (uniform 1..max)
    n_vars_per_method: int = 2
    seed: int = 1234


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

    corpus_path = os.path.join(out_dir, "corpus.txt")
    with open(corpus_path, "w", encoding="utf-8") as f:
        for mid in range(spec.n_methods):
            name = _method_name(rng)
            f.write(f"#{mid}\n")
            f.write(f"label:{name}\n")
            f.write(f"class:Synth{mid % 7}.java\n")
            f.write(f"doc: synthetic method {mid}\n")
            f.write("paths:\n")
            n_ctx = int(rng.integers(1, spec.max_contexts + 1))
            for _ in range(n_ctx):
                # raw file-side terminal indexes (reader adds +1 for @question)
                s = int(rng.integers(1, spec.n_terminals - 1))
                p = int(rng.integers(1, spec.n_paths))
                e = int(rng.integers(1, spec.n_terminals - 1))
                f.write(f"{s}\t{p}\t{e}\n")
            f.write("vars:\n")
            for v in range(spec.n_vars_per_method):
                f.write(f"someVar{v}\t@var_{v}\n")
            f.write("\n")

    return {
        "corpus_path": corpus_path,
        "path_idx_path": path_path,
        "terminal_idx_path": terminal_path,
    }


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
