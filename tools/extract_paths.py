#!/usr/bin/env python3
"""Offline Java -> path-context extractor (native C++; the reference's
create_path_contexts.ipynb equivalent).  Reads a methods list
("javaFile\\tmethodName" per line, like the reference dataset/methods.txt),
parses each Java file, and writes corpus.txt / terminal_idxs.txt /
path_idxs.txt / actual_methods.txt / params.txt in the reference format:

    python tools/extract_paths.py --methods ./methods.txt \\
        --src-root ./java_src --out ./dataset \\
        [--max-length 8] [--max-width 3]

Then train exactly as with reference-extracted data:

    python main.py --corpus_path ./dataset/corpus.txt \\
        --path_idx_path ./dataset/path_idxs.txt \\
        --terminal_idx_path ./dataset/terminal_idxs.txt
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch  # noqa: F401  (loads libc10 for the extension)

from code2vec_amd.data import _c2v_extract as X


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--methods", required=True,
                    help="methods list: javaFile<TAB>methodName per line")
    ap.add_argument("--src-root", default="",
                    help="prefix joined to the javaFile paths")
    ap.add_argument("--out", required=True)
    ap.add_argument("--max-length", type=int, default=8)
    ap.add_argument("--max-width", type=int, default=3)
    a = ap.parse_args()
    os.makedirs(a.out, exist_ok=True)
    stats = X.extract_to_dataset(a.methods, a.src_root, a.out,
                                 a.max_length, a.max_width)
    for k, v in sorted(stats.items()):
        print(f"{k}: {v}")


if __name__ == "__main__":
    main()
