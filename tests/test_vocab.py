"""Vocab semantics parity tests (reference model/dataset.py:52-93,
model/dataset_reader.py:15-41)."""

from code2vec_amd.data.vocab import (
    QUESTION_TOKEN_INDEX,
    Vocab,
    get_method_subtokens,
    normalize_method_name,
    read_vocab_file,
)


def test_normalize_method_name():
    assert normalize_method_name("get_value_2") == "getvalue"
    assert normalize_method_name("toString") == "toString"
    assert normalize_method_name("__init__9x") == "initx"


def test_subtoken_split():
    assert get_method_subtokens("getValue") == ["get", "value"]
    assert get_method_subtokens("toString") == ["to", "string"]
    # regex keeps leading lowercase run + [A-Z][a-z]+ groups, lowercased
    assert get_method_subtokens("readHttpHeader") == ["read", "http", "header"]
    assert get_method_subtokens("x") == ["x"]


def test_freq_first_occurrence_only():
    """The reference bumps freq only on first insert — every name keeps
    freq == 1 no matter how many times it is appended."""
    v = Vocab()
    v.append("foo")
    v.append("foo")
    v.append("foo")
    assert v.freq[v.stoi["foo"]] == 1
    assert v.get_freq_list() == [1]


def test_vocab_len_and_both_apis():
    v = Vocab()
    v.append("a")
    v.append("b")
    assert len(v) == 2
    assert v.len() == 2


def test_read_vocab_file_index_shift(tmp_path):
    p = tmp_path / "terms.txt"
    p.write_text("0\t<PAD/>\n1\tfoo\n2\tbar\n3\n", encoding="utf-8")
    v = read_vocab_file(str(p), extra_tokens=["@question"])
    # extra token at index 1; file indexes > 0 shifted by 1
    assert v.stoi["@question"] == QUESTION_TOKEN_INDEX == 1
    assert v.stoi["<PAD/>"] == 0
    assert v.stoi["foo"] == 2
    assert v.stoi["bar"] == 3
    # line with no name column -> empty-string name
    assert v.stoi[""] == 4
    assert len(v) == 5


def test_read_vocab_file_no_extras(tmp_path):
    p = tmp_path / "paths.txt"
    p.write_text("0\t<PAD/>\n1\tp1\n", encoding="utf-8")
    v = read_vocab_file(str(p))
    assert v.stoi["p1"] == 1
