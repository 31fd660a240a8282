#!/usr/bin/env python3
"""code.vec -> TensorBoard embedding projector (reference
visualize_code_vec.py:1-23 equivalent; parser split out for testing)."""

from __future__ import annotations

import argparse

import numpy as np


def parse_code_vec(path: str):
    """Skip the header line; rows are ``label\\tv0 v1 ...`` (reference
    visualize_code_vec.py:13-15)."""
    labels = []
    vectors = []
    with open(path, mode="r", encoding="utf-8") as f:
        first = True
        for line in f:
            if first:
                first = False
                continue
            line = line.rstrip("\n")
            if not line:
                continue
            label, vec = line.split("\t")
            labels.append(label)
            vectors.append([float(x) for x in vec.split(" ")])
    return labels, np.asarray(vectors, dtype=np.float32)


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--vectors_path", type=str, default="output/code.vec")
    args = ap.parse_args()
    labels, mat = parse_code_vec(args.vectors_path)
    try:
        from tensorboardX import SummaryWriter
    except ImportError as e:
        raise SystemExit(
            "tensorboardX is not installed in this environment; "
            "parse succeeded ({} vectors of dim {}).".format(*mat.shape)
        ) from e
    import torch

    writer = SummaryWriter()
    writer.add_embedding(torch.from_numpy(mat), metadata=labels)
    writer.close()


if __name__ == "__main__":
    main()
