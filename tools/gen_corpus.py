"""Generate a synthetic dataset in the reference corpus file format
(corpus.txt / terminal_idxs.txt / path_idxs.txt — the files the Scala
extractor of reference create_path_contexts.ipynb cell 11 writes; format
spec in docs/PREPROCESSING.md).  Lets a user exercise the full training
CLI without Java sources:

    python tools/gen_corpus.py --out ./dataset --methods 10000
    python main.py --corpus_path ./dataset/corpus.txt \
        --path_idx_path ./dataset/path_idxs.txt \
        --terminal_idx_path ./dataset/terminal_idxs.txt
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from code2vec_amd.data.synthetic import SyntheticSpec, write_synthetic_corpus


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--out", default="./dataset")
    ap.add_argument("--methods", type=int, default=1000)
    ap.add_argument("--terminals", type=int, default=2000,
                    help="terminal vocab size (excl. PAD)")
    ap.add_argument("--paths", type=int, default=3000,
                    help="path vocab size (excl. PAD)")
    ap.add_argument("--max_contexts", type=int, default=200)
    ap.add_argument("--vars_per_method", type=int, default=2)
    ap.add_argument("--seed", type=int, default=1234)
    ap.add_argument("--label_signal", type=float, default=0.0,
                    help="fraction of each method's contexts drawn from its "
                         "label's signature pool (makes names predictable "
                         "from the context bag -> meaningful F1 curves)")
    a = ap.parse_args()
    spec = SyntheticSpec(n_methods=a.methods, n_terminals=a.terminals,
                         n_paths=a.paths, max_contexts=a.max_contexts,
                         n_vars_per_method=a.vars_per_method, seed=a.seed,
                         label_signal=a.label_signal)
    files = write_synthetic_corpus(a.out, spec)
    for k, v in files.items():
        print(f"{k}: {v}")


if __name__ == "__main__":
    main()
