"""Artifact export: code.vec writer, test-result TSV, sample printer.

Byte-format parity with reference main.py:226-231,362-423:
- code.vec header line is ``{len(reader.items)}\\t{encode_size}`` even though
  the row count written is train+test dataset sizes (the reference's quirk,
  preserved on purpose),
- each row: ``label_name\\tv0 v1 ...`` with str(float) rendering,
- test-result TSV rows: ``id\\tcorrect\\texpected\\tpredicted\\tprob``.
"""

from __future__ import annotations

import logging
from typing import Optional

import torch

from ..ops import functional as Fn

logger = logging.getLogger(__name__)


def _forward(model, batch, device, graphed=None):
    starts = batch["starts"].to(device)
    paths = batch["paths"].to(device)
    ends = batch["ends"].to(device)
    label = batch["label"].to(device)
    if graphed is not None and starts.shape[0] <= graphed.B:
        outputs, code_vector, attn = graphed.run(starts, paths, ends, label)
    else:
        outputs, code_vector, attn = model(starts, paths, ends, label)
    return starts, label, outputs, code_vector, attn


def write_code_vectors(
    reader,
    model,
    loader,
    option,
    vector_file: str,
    mode: str,
    test_result_file: Optional[str],
    device,
    graphed=None,
) -> None:
    """Append one loader's code vectors (reference main.py:393-423).
    ``graphed``: optional GraphedInference (hipGraph-captured forward)."""
    model.eval()
    itos = reader.label_vocab.itos
    with torch.no_grad():
        fr = open(test_result_file, "w") if test_result_file is not None else None
        try:
            with open(vector_file, mode) as fv:
                for batch in loader:
                    ids = batch["id"]
                    _, label, outputs, code_vector, _ = _forward(
                        model, batch, device, graphed)
                    preds_prob, preds_label = Fn.row_max_argmax(outputs.detach())
                    cv = code_vector.float().cpu()
                    label_cpu = label.cpu()
                    preds_cpu = preds_label.cpu()
                    prob_cpu = preds_prob.cpu()
                    for i in range(cv.shape[0]):
                        label_name = itos[int(label_cpu[i])]
                        vec = cv[i]
                        fv.write(
                            label_name + "\t"
                            + " ".join(str(e.item()) for e in vec) + "\n"
                        )
                        if fr is not None:
                            pred_name = itos[int(preds_cpu[i])]
                            fr.write(
                                "{0}\t{1}\t{2}\t{3}\t{4}\n".format(
                                    int(ids[i]),
                                    label_name == pred_name,
                                    label_name,
                                    pred_name,
                                    float(prob_cpu[i]),
                                )
                            )
        finally:
            if fr is not None:
                fr.close()


def write_vector_header(vector_file: str, item_count: int, encode_size: int) -> None:
    with open(vector_file, "w") as f:
        f.write("{0}\t{1}\n".format(item_count, encode_size))


def print_sample(reader, model, loader, option, device) -> None:
    """Print one correctly-predicted example's contexts with attention
    (reference main.py:362-390)."""
    model.eval()
    term_itos = reader.terminal_vocab.itos
    path_itos = reader.path_vocab.itos
    label_itos = reader.label_vocab.itos
    with torch.no_grad():
        for batch in loader:
            starts_d, label, outputs, _, attn = _forward(model, batch, device)
            _, preds_label = Fn.row_max_argmax(outputs.detach())
            starts = batch["starts"]
            paths = batch["paths"]
            ends = batch["ends"]
            label_cpu = label.cpu()
            preds_cpu = preds_label.cpu()
            attn_cpu = attn.float().cpu()
            for i in range(starts.shape[0]):
                if int(preds_cpu[i]) == int(label_cpu[i]):
                    for c in range(starts.shape[1]):
                        s_name = term_itos.get(int(starts[i, c]), "")
                        if s_name == "<PAD/>":
                            continue
                        p_name = path_itos.get(int(paths[i, c]), "")
                        e_name = term_itos.get(int(ends[i, c]), "")
                        logger.info(
                            "%s %s %s [%s]", s_name, p_name, e_name, attn_cpu[i, c]
                        )
                    logger.info("expected label: %s", label_itos[int(label_cpu[i])])
                    logger.info("actual label:   %s", label_itos[int(preds_cpu[i])])
                    return
