"""Vocabulary with the reference's exact index / normalization semantics.

Parity notes (cites are into /root/reference):
- ``normalize_method_name`` strips ``[_0-9]+`` (model/dataset.py:55,86-88).
- ``get_method_subtokens`` splits camelCase with the regex
  ``([a-z]+)([A-Z][a-z]+)|([A-Z][a-z]+)`` and lowercases
  (model/dataset.py:56,90-92).
- ``append`` increments ``freq`` only on the FIRST occurrence of a name
  (model/dataset.py:64-74) — so every label ends up with freq == 1 and the
  1/freq loss weighting is effectively uniform.  This quirk is replicated
  on purpose: "fixing" it would change the loss.
- Vocab files are ``index\\tname`` lines; extra tokens are inserted starting
  at index 1 and every file index > 0 is shifted by ``len(extra_tokens)``
  (model/dataset_reader.py:22-41).  Index 0 is ``<PAD/>``.
"""

from __future__ import annotations

import re
from typing import Dict, List, Optional

PAD_INDEX = 0
QUESTION_TOKEN_INDEX = 1
QUESTION_TOKEN_NAME = "@question"

_REDUNDANT_SYMBOL_CHARS = re.compile(r"[_0-9]+")
_METHOD_SUBTOKEN_SEPARATOR = re.compile(r"([a-z]+)([A-Z][a-z]+)|([A-Z][a-z]+)")


def normalize_method_name(method_name: str) -> str:
    """Strip underscores and digits (reference model/dataset.py:86-88)."""
    return _REDUNDANT_SYMBOL_CHARS.sub("", method_name)


def get_method_subtokens(method_name: str) -> List[str]:
    """Split a camelCase name into lowercased subtokens
    (reference model/dataset.py:90-92)."""
    return [
        x.lower()
        for x in _METHOD_SUBTOKEN_SEPARATOR.split(method_name)
        if x is not None and x != ""
    ]


class Vocab:
    """String <-> index vocabulary (reference model/dataset.py:52-93)."""

    __slots__ = ("stoi", "itos", "itosubtokens", "freq")

    def __init__(self) -> None:
        self.stoi: Dict[str, int] = {}
        self.itos: Dict[int, str] = {}
        self.itosubtokens: Dict[int, List[str]] = {}
        self.freq: Dict[int, int] = {}

    def append(
        self,
        name: str,
        index: Optional[int] = None,
        subtokens: Optional[List[str]] = None,
    ) -> None:
        # NOTE: freq is bumped only inside the first-occurrence branch —
        # replicating reference model/dataset.py:64-74 exactly.
        if name not in self.stoi:
            if index is None:
                index = len(self.stoi)
            if self.freq.get(index) is None:
                self.freq[index] = 0
            self.stoi[name] = index
            self.itos[index] = name
            if subtokens is not None:
                self.itosubtokens[index] = subtokens
            self.freq[index] += 1

    def get_freq_list(self) -> List[int]:
        """Dense freq list indexed 0..len-1 (reference model/dataset.py:76-81)."""
        return [self.freq[i] for i in range(len(self))]

    def __len__(self) -> int:
        return len(self.stoi)

    # Reference spells this ``vocab.len()``; keep both for API parity.
    def len(self) -> int:  # noqa: A003
        return len(self.stoi)


def read_vocab_file(filename: str, extra_tokens: Optional[List[str]] = None) -> Vocab:
    """Read an ``index\\tname`` vocab file with extra-token index shifting.

    Semantics of reference model/dataset_reader.py:22-41:
    - extra tokens get indexes 1..len(extra_tokens),
    - file indexes > 0 shift up by len(extra_tokens); index 0 is kept,
    - a line with no second column maps the empty string.
    """
    extra_tokens = extra_tokens or []
    vocab = Vocab()
    extra_size = len(extra_tokens)
    idx = 1
    for name in extra_tokens:
        vocab.append(name, idx)
        idx += 1
    with open(filename, mode="r", encoding="utf-8") as f:
        for line in f:
            data = line.strip(" \r\n\t").split("\t")
            if not data or data[0] == "":
                continue
            index = int(data[0])
            if index > 0:
                index += extra_size
            name = data[1] if len(data) > 1 else ""
            vocab.append(name, index)
    return vocab
