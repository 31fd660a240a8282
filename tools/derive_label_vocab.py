#!/usr/bin/env python3
"""Derive the top11-scale label-vocab size from REAL reference data.

The reference never publishes its top11 label vocab.  Its repo does ship
the real method list of the tiny dataset (dataset/methods.txt: 9,916
methods).  This tool:
1. fits Heaps' law V(n) = K * n^beta to the prefix-unique curve of the
   normalized method names in a real methods.txt,
2. evaluates the fit at top11's 605,945 methods (top11_dataset/params.txt:9),
3. simulates the synthetic generator's Heaps-calibrated name process
   (code2vec_amd/data/synthetic.py::name_stream) at the same scale, so
   the bench config's label_count matches what a generated top11-scale
   corpus actually produces.

    python tools/derive_label_vocab.py [--methods-file PATH] [--n 605945]
"""
import argparse
import os
import re
import sys

import numpy as np

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from code2vec_amd.data.synthetic import HEAPS_BETA, HEAPS_K, name_stream


def fit_heaps(names):
    norm = [re.sub(r"[_0-9]+", "", n).lower() for n in names]
    seen, curve = set(), []
    for x in norm:
        seen.add(x)
        curve.append(len(seen))
    curve = np.asarray(curve)
    ns = np.unique(np.logspace(1.5, np.log10(len(norm)), 30).astype(int))
    A = np.vstack([np.log(ns), np.ones(len(ns))]).T
    beta, logK = np.linalg.lstsq(A, np.log(curve[ns - 1]), rcond=None)[0]
    return float(np.exp(logK)), float(beta), len(seen)


def main():
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument("--methods-file",
                    default="/root/reference/dataset/methods.txt")
    ap.add_argument("--n", type=int, default=605945)
    ap.add_argument("--seed", type=int, default=123)
    a = ap.parse_args()

    if os.path.exists(a.methods_file):
        names = []
        with open(a.methods_file) as f:
            for line in f:
                parts = line.rstrip("\n").split("\t")
                if len(parts) >= 2:
                    names.append(parts[1])
        K, beta, uniq = fit_heaps(names)
        print(f"methods.txt: {len(names)} methods, {uniq} unique normalized")
        print(f"Heaps fit: V(n) = {K:.3f} * n^{beta:.3f} "
              f"(shipped constants: {HEAPS_K} * n^{HEAPS_BETA})")
        print(f"V({a.n}) = {int(K * a.n ** beta)} (fit)")

    rng = np.random.default_rng(a.seed)
    uniq = len({nm.lower() for nm in name_stream(rng, a.n)})
    print(f"simulated generator unique labels at n={a.n}: {uniq}")


if __name__ == "__main__":
    main()
