"""Corpus parse + per-epoch builder semantics (reference
model/dataset_reader.py:72-128, model/dataset_builder.py:112-210)."""

import numpy as np
import pytest

from code2vec_amd.data.builder import DatasetBuilder
from code2vec_amd.data.reader import CorpusReader
from code2vec_amd.utils.options import Option


def make_reader(tiny_corpus, **kw):
    return CorpusReader(
        tiny_corpus["corpus_path"],
        tiny_corpus["path_idx_path"],
        tiny_corpus["terminal_idx_path"],
        **kw,
    )


def make_option(reader, **kw):
    defaults = dict(max_path_length=16, terminal_embed_size=8,
                    path_embed_size=8, encode_size=12)
    defaults.update(kw)
    return Option(
        terminal_count=len(reader.terminal_vocab),
        path_count=len(reader.path_vocab),
        label_count=len(reader.label_vocab),
        **defaults,
    )


def test_reader_question_offset(tiny_corpus):
    r = make_reader(tiny_corpus)
    # terminal vocab has @question at 1; start/end corpus indexes shifted +1
    assert r.terminal_vocab.stoi["@question"] == 1
    assert r.terminal_vocab.stoi["<PAD/>"] == 0
    # raw file terminal index i appears as i+1 in path_contexts start/end
    for item in r.items[:5]:
        assert item.path_contexts[:, 0].min() >= 2  # raw >= 1, +1 shift
        assert item.path_contexts[:, 1].min() >= 1  # paths unshifted


def test_reader_items_and_labels(tiny_corpus):
    r = make_reader(tiny_corpus)
    assert len(r.items) == 48
    for item in r.items:
        assert item.normalized_label in r.label_vocab.stoi
        idx = r.label_vocab.stoi[item.normalized_label]
        assert idx in r.label_vocab.itosubtokens
        assert item.id is not None
        assert item.aliases  # vars: section parsed


def test_reader_variable_indexes(tiny_corpus):
    r = make_reader(tiny_corpus)
    for idx in r.variable_indexes:
        assert r.terminal_vocab.itos[idx].startswith("@var_")


def test_builder_split_and_shapes(tiny_corpus):
    r = make_reader(tiny_corpus)
    opt = make_option(r)
    b = DatasetBuilder(r, opt, seed=11)
    assert len(b.train_items) + len(b.test_items) == len(r.items)
    assert len(b.test_items) == int(len(r.items) * 0.2)

    data = b.refresh_train_dataset(epoch=0)
    C = opt.max_path_length
    assert data.starts.shape == (len(b.train_items), C)
    assert data.starts.dtype == np.int32
    assert data.labels.dtype == np.int64
    # padding is zero; non-pad values positive
    assert (data.starts >= 0).all()


def test_builder_resamples_per_epoch(tiny_corpus):
    r = make_reader(tiny_corpus)
    opt = make_option(r, max_path_length=4)  # force truncation
    b = DatasetBuilder(r, opt, seed=11)
    # NOTE: an EpochData aliases the builder's persistent buffers and is
    # invalidated by the next refresh of the same split — copy to compare.
    d0_paths = b.refresh_train_dataset(epoch=0).paths.copy()
    d0_labels = list(b.refresh_train_dataset(epoch=0).labels.tolist())
    d1 = b.refresh_train_dataset(epoch=1)
    # same items, different resampled contexts
    assert d0_labels == d1.labels.tolist()
    assert not np.array_equal(d0_paths, d1.paths)
    # deterministic per (seed, epoch)
    d0b = b.refresh_train_dataset(epoch=0)
    assert np.array_equal(d0_paths, d0b.paths)


def test_builder_question_replacement(tiny_corpus):
    r = make_reader(tiny_corpus)
    opt = make_option(r)
    b = DatasetBuilder(r, opt, seed=3)
    method_idx = r.terminal_vocab.stoi["@method_0"]
    data = b.refresh_train_dataset(epoch=0)
    assert not np.any(data.starts == method_idx)
    assert not np.any(data.ends == method_idx)


def test_builder_rank_sharding(tiny_corpus):
    r = make_reader(tiny_corpus)
    opt = make_option(r)
    shards = []
    for rank in range(2):
        b = DatasetBuilder(r, opt, seed=5, rank=rank, world_size=2)
        d = b.refresh_train_dataset(epoch=0)
        shards.append(set(d.ids))
    assert shards[0].isdisjoint(shards[1])
    # equal shard sizes => equal per-rank step counts (collective lockstep);
    # the dropped tail is < world_size items
    assert len(shards[0]) == len(shards[1])
    b_all = DatasetBuilder(r, opt, seed=5)
    d_all = b_all.refresh_train_dataset(epoch=0)
    all_ids = set(d_all.ids)
    union = shards[0] | shards[1]
    assert union <= all_ids
    assert len(all_ids) - len(union) < 2
    # the epoch-seeded permutation rotates which items land in the dropped
    # tail, so across epochs everything is seen
    seen = set(union)
    for epoch in range(1, 6):
        for rank in range(2):
            b = DatasetBuilder(r, opt, seed=5, rank=rank, world_size=2)
            seen |= set(b.refresh_train_dataset(epoch=epoch).ids)
    assert seen == all_ids


def test_builder_variable_task(tiny_corpus):
    r = make_reader(tiny_corpus, infer_method=False, infer_variable=True)
    opt = make_option(r)
    b = DatasetBuilder(r, opt, seed=3)
    data = b.refresh_train_dataset(epoch=0)
    # one row per @var alias per item that has matching contexts
    assert len(data) > 0
    # all labels exist in the variable label vocab
    assert all(0 <= l < len(r.label_vocab) for l in data.labels)


def test_oov_rate_bounds(tiny_corpus):
    r = make_reader(tiny_corpus)
    opt = make_option(r)
    b = DatasetBuilder(r, opt, seed=11)
    rate = b.out_of_vocabulary_rate()
    assert 0.0 <= rate <= 1.0


def test_native_builder_matches_python_semantics(tiny_corpus):
    """The C++ OpenMP epoch builder produces rows that are valid resamples
    of each item's bag (same labels, contexts drawn w/o replacement, 
    @method_0 replaced) — structure-equal to the numpy fallback."""
    from code2vec_amd.data import builder as B

    if B._native is None:
        pytest.skip("native builder not built")
    r = make_reader(tiny_corpus)
    opt = make_option(r, max_path_length=8)
    b = DatasetBuilder(r, opt, seed=17)
    native = b.refresh_train_dataset(epoch=0)

    # numpy fallback on the same shard
    idx = b._shard_indices(len(b.train_items), 0)
    items = [b.train_items[i] for i in idx]
    py = b.build_data(items, opt.max_path_length, epoch=0, stream=1)

    assert native.labels.tolist() == py.labels.tolist()
    assert native.starts.shape == py.starts.shape
    q = r.QUESTION_TOKEN_INDEX
    mtok = r.terminal_vocab.stoi["@method_0"]
    for row, item in zip(range(len(items)), items):
        pcs = item.path_contexts
        n = min(pcs.shape[0], opt.max_path_length)
        # non-pad count matches
        npad = int((native.paths[row] != 0).sum())
        assert npad == n or pcs.shape[0] > opt.max_path_length
        # every sampled triple exists in the item's bag (mod @question swap)
        allowed = set()
        for s, p, e in pcs.tolist():
            s2 = q if s == mtok else s
            e2 = q if e == mtok else e
            allowed.add((s2, p, e2))
        for c in range(n):
            trip = (int(native.starts[row, c]), int(native.paths[row, c]),
                    int(native.ends[row, c]))
            assert trip in allowed
        # no duplicates (sampling w/o replacement) when bag has no dups
        trips = [tuple(t) for t in
                 np.stack([native.starts[row, :n], native.paths[row, :n],
                           native.ends[row, :n]], axis=1).tolist()]
        if len(allowed) == pcs.shape[0]:
            assert len(set(trips)) == len(trips)


def test_reader_edge_cases(tmp_path):
    """Parser robustness: empty paths section, no-vars method, missing id,
    no trailing blank line, doc lines, blank-line runs."""
    corpus = tmp_path / "corpus.txt"
    corpus.write_text(
        "#1\n"
        "label:emptyPaths\n"
        "class:A.java\n"
        "paths:\n"
        "\n"
        "\n"
        "label:noIdNoVars\n"
        "doc: some doc text\n"
        "paths:\n"
        "1\t1\t2\n"
        "2\t2\t1\n",  # no trailing blank line
        encoding="utf-8",
    )
    terms = tmp_path / "terms.txt"
    terms.write_text("0\t<PAD/>\n1\t@method_0\n2\tfoo\n3\tbar\n",
                     encoding="utf-8")
    pathsf = tmp_path / "paths.txt"
    pathsf.write_text("0\t<PAD/>\n1\tP1\n2\tP2\n", encoding="utf-8")

    r = CorpusReader(str(corpus), str(pathsf), str(terms))
    assert len(r.items) == 2
    assert r.items[0].path_contexts.shape == (0, 3)  # empty bag
    assert r.items[1].id is None
    assert r.items[1].path_contexts.shape == (2, 3)
    # +1 question shift on start/end, not on path
    assert r.items[1].path_contexts[0].tolist() == [2, 1, 3]

    # a zero-context method builds an all-pad row; model handles it
    opt = make_option(r, max_path_length=4)
    b = DatasetBuilder(r, opt, seed=1, split_ratio=0.0)
    data = b.refresh_train_dataset(0)
    assert len(data) == 2
    row_empty = data.ids.index(1)
    assert (data.starts[row_empty] == 0).all()


def test_native_parser_matches_python(tiny_corpus):
    """C++ parse_corpus produces identical records/vocabs to the Python
    parser (including +1 question shift, aliases, doc-skip)."""
    from code2vec_amd.data import reader as RD

    if RD._native is None:
        pytest.skip("native parser not built")
    r_nat = make_reader(tiny_corpus)
    # force the python path
    r_py = CorpusReader.__new__(CorpusReader)
    r_py.path_vocab = r_nat.path_vocab
    r_py.terminal_vocab = r_nat.terminal_vocab
    r_py.infer_method = True
    r_py.infer_variable = False
    from code2vec_amd.data.vocab import Vocab
    r_py.label_vocab = Vocab()
    r_py.items = []
    r_py._load(tiny_corpus["corpus_path"])

    assert len(r_nat.items) == len(r_py.items)
    assert r_nat.label_vocab.stoi == r_py.label_vocab.stoi
    assert r_nat.label_vocab.itosubtokens == r_py.label_vocab.itosubtokens
    for a, b in zip(r_nat.items, r_py.items):
        assert a.id == b.id
        assert a.label == b.label
        assert a.normalized_label == b.normalized_label
        assert a.source == b.source
        assert a.aliases == b.aliases
        assert np.array_equal(a.path_contexts, b.path_contexts)


def test_gen_corpus_cli_output_parses(tmp_path):
    """tools/gen_corpus.py writes the reference file format end to end:
    the reader must parse its output with matching counts."""
    import subprocess
    import sys
    import os

    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    out = tmp_path / "gen"
    res = subprocess.run(
        [sys.executable, os.path.join(root, "tools", "gen_corpus.py"),
         "--out", str(out), "--methods", "37", "--terminals", "60",
         "--paths", "50", "--max_contexts", "9", "--seed", "3"],
        capture_output=True, text=True, timeout=120)
    assert res.returncode == 0, res.stderr
    r = CorpusReader(str(out / "corpus.txt"), str(out / "path_idxs.txt"),
                     str(out / "terminal_idxs.txt"))
    assert len(r.items) == 37
    assert all(len(it.path_contexts) >= 1 for it in r.items)
    assert all(it.normalized_label for it in r.items)
    # terminal vocab includes PAD + @question shift semantics
    assert r.terminal_vocab.stoi["@question"] == 1


@pytest.mark.parametrize("variant", ["crlf", "no_trailing_blank",
                                     "no_label", "empty_paths",
                                     "extra_blanks", "spaces"])
def test_native_parser_matches_python_edge_corpora(tmp_path, variant):
    """C++ and Python parsers must agree on adversarial corpus shapes:
    CRLF line endings, missing trailing blank line, records without
    labels or paths, runs of blank lines, stray spaces."""
    from code2vec_amd.data import reader as RD
    from code2vec_amd.data.vocab import Vocab

    if RD._native is None:
        pytest.skip("native parser not built")

    base = (
        "#0\n"
        "label:fooBar\n"
        "class:A.java\n"
        "doc: something to discard\n"
        "paths:\n"
        "1\t2\t3\n"
        "4\t5\t6\n"
        "vars:\n"
        "counter\t@var_0\n"
        "\n"
        "#1\n"
        "label:bazQux9\n"
        "paths:\n"
        "7\t8\t9\n"
        "\n"
    )
    if variant == "crlf":
        text = base.replace("\n", "\r\n")
    elif variant == "no_trailing_blank":
        text = base.rstrip("\n") + "\n"  # last record not blank-terminated
    elif variant == "no_label":
        text = "#0\npaths:\n1\t2\t3\n\n" + base
    elif variant == "empty_paths":
        text = "#5\nlabel:onlyVars\nvars:\nx\t@var_0\n\n" + base
    elif variant == "extra_blanks":
        text = base.replace("\n\n", "\n\n\n\n")
    elif variant == "spaces":
        text = base.replace("label:fooBar", "label:fooBar  ")

    cp = tmp_path / "corpus.txt"
    cp.write_text(text)

    def parse(force_python):
        r = CorpusReader.__new__(CorpusReader)
        r.infer_method = True
        r.infer_variable = False
        r.label_vocab = Vocab()
        r.items = []
        if force_python:
            r._load(str(cp))
        else:
            r._load_native(str(cp))
        return r

    rn, rp = parse(False), parse(True)
    assert len(rn.items) == len(rp.items)
    assert rn.label_vocab.stoi == rp.label_vocab.stoi
    for a, b in zip(rn.items, rp.items):
        assert a.id == b.id
        assert a.label == b.label
        assert a.normalized_label == b.normalized_label
        assert a.source == b.source
        assert a.aliases == b.aliases
        assert np.array_equal(np.asarray(a.path_contexts),
                              np.asarray(b.path_contexts))


def test_reader_rejects_out_of_range_indices(tmp_path, tiny_corpus):
    """Out-of-range corpus indices fail loudly at load time (the HIP
    gather never bounds-checks; the reference fails in nn.Embedding)."""
    bad = tmp_path / "bad_corpus.txt"
    with open(tiny_corpus["corpus_path"]) as f:
        text = f.read()
    # append a record whose path index is far beyond the path vocab
    text += "\n#9999\nlabel:badMethod\npaths:\n2\t999999\t3\n\n"
    bad.write_text(text)
    with pytest.raises(ValueError, match="out-of-range"):
        CorpusReader(str(bad), tiny_corpus["path_idx_path"],
                     tiny_corpus["terminal_idx_path"])
    # and a bad terminal index (after the +1 question shift)
    bad2 = tmp_path / "bad_corpus2.txt"
    bad2.write_text(text.replace("2\t999999\t3", "888888\t1\t3"))
    with pytest.raises(ValueError, match="out-of-range"):
        CorpusReader(str(bad2), tiny_corpus["path_idx_path"],
                     tiny_corpus["terminal_idx_path"])


def test_variable_task_equal_rows_across_ranks(tiny_corpus):
    """With the variable task on, per-rank ROW counts (not just item
    counts) must be equal, or DP ranks run different step counts."""
    r = make_reader(tiny_corpus, infer_method=True, infer_variable=True)
    opt = make_option(r)
    for epoch in range(3):
        sizes = []
        for rank in range(2):
            b = DatasetBuilder(r, opt, seed=11, rank=rank, world_size=2)
            sizes.append(len(b.refresh_train_dataset(epoch=epoch)))
        assert sizes[0] == sizes[1], sizes


def test_synthetic_reader_builder_integration():
    """SyntheticReader (bench --real-pipeline) must drive DatasetBuilder
    exactly like a file-backed CorpusReader."""
    from code2vec_amd.data.synthetic import SyntheticReader

    r = SyntheticReader(n_methods=100, terminal_count=50, path_count=40,
                        label_count=20, max_contexts=30, seed=3)
    opt = Option(terminal_count=50, path_count=40, label_count=20,
                 max_path_length=16, terminal_embed_size=8,
                 path_embed_size=8, encode_size=12)
    b = DatasetBuilder(r, opt, seed=5)
    d = b.refresh_train_dataset(epoch=0)
    assert len(d) == len(b.train_items)
    assert d.starts.shape == (len(d), 16)
    assert (d.labels >= 0).all() and (d.labels < 20).all()
    assert (d.starts < 50).all() and (d.paths < 40).all()
    # resampling across epochs still applies
    d0 = d.paths.copy()
    d1 = b.refresh_train_dataset(epoch=1)
    assert not np.array_equal(d0, d1.paths)


class TestMalformedCorpus:
    """Corrupt corpora must fail LOUDLY with a line number — the native
    parser used to fabricate PAD-ish triples from malformed lines
    silently (the reference's int()/tuple-unpack crashes on them)."""

    def _mk(self, tmp_path, corpus_text):
        c = tmp_path / "c.txt"
        c.write_text(corpus_text)
        t = tmp_path / "t.txt"
        t.write_text("0\t<PAD/>\n1\tfoo\n2\tbar\n")
        p = tmp_path / "p.txt"
        p.write_text("0\t<PAD/>\n1\tpp\n")
        return str(c), str(p), str(t)

    @pytest.mark.parametrize("bad", [
        "1\tx\t2",        # non-integer field
        "1\t1",           # missing field
        "1\t1\t2\t3",     # extra field (reference unpack would crash too)
        "",               # (blank ends the record; covered elsewhere)
    ])
    def test_malformed_context_line_raises(self, tmp_path, bad):
        if bad == "":
            return  # blank line = record separator, legal
        c, p, t = self._mk(tmp_path,
                           f"#1\nlabel:getFoo\npaths:\n{bad}\n\n")
        with pytest.raises((RuntimeError, ValueError),
                           match="malformed|invalid literal|not enough|"
                                 "too many"):
            CorpusReader(c, p, t)

    def test_malformed_vars_line_raises(self, tmp_path):
        c, p, t = self._mk(tmp_path,
                           "#1\nlabel:a\nvars:\nnameonly\npaths:\n1\t1\t2\n\n")
        with pytest.raises((RuntimeError, ValueError), match="malformed|"):
            CorpusReader(c, p, t)

    def test_error_carries_line_number(self, tmp_path):
        c, p, t = self._mk(tmp_path,
                           "#1\nlabel:getFoo\npaths:\n1\t1\t2\n1\tBAD\t2\n\n")
        try:
            CorpusReader(c, p, t)
            assert False, "expected a parse error"
        except (RuntimeError, ValueError) as e:
            assert "5" in str(e) or "literal" in str(e)

    def test_python_fallback_parser_same_contract(self, tmp_path,
                                                  monkeypatch):
        """With the native extension disabled, the pure-Python parser
        must produce the SAME items on good corpora and the same
        line-numbered errors on malformed ones."""
        import code2vec_amd.data.reader as rd

        good = "#1\nlabel:getFoo\npaths:\n1\t1\t2\n\n"
        c, p, t = self._mk(tmp_path, good)
        ref = CorpusReader(c, p, t)
        monkeypatch.setattr(rd, "_native", None)
        alt = rd.CorpusReader(c, p, t)
        assert len(alt.items) == len(ref.items) == 1
        assert (alt.items[0].path_contexts == ref.items[0].path_contexts).all()
        assert alt.items[0].normalized_label == ref.items[0].normalized_label
        c2, p2, t2 = self._mk(tmp_path, "#1\nlabel:a\npaths:\n1\tX\t2\n\n")
        with pytest.raises(ValueError, match="malformed.*line 4"):
            rd.CorpusReader(c2, p2, t2)


def test_batch_iterator_edge_sizes():
    """batch > dataset, empty dataset, ragged tail — all legal."""
    from types import SimpleNamespace

    import torch

    from code2vec_amd.engine.loader import BatchIterator

    def mk(n):
        return SimpleNamespace(
            ids=list(range(n)),
            starts=np.zeros((n, 4), dtype=np.int64),
            paths=np.zeros((n, 4), dtype=np.int64),
            ends=np.zeros((n, 4), dtype=np.int64),
            labels=np.zeros(n, dtype=np.int64))

    cpu = torch.device("cpu")
    it = BatchIterator(mk(3), 8, shuffle=True, seed=1, device=cpu)
    assert [b["label"].shape[0] for b in it] == [3] and len(it) == 1
    it = BatchIterator(mk(0), 4, shuffle=False, seed=1, device=cpu)
    assert list(it) == [] and len(it) == 0
    it = BatchIterator(mk(9), 4, shuffle=True, seed=1, device=cpu)
    assert [b["label"].shape[0] for b in it] == [4, 4, 1]


def test_builder_epoch_invariants_random_corpora():
    """Invariants of every built epoch, over randomized tiny corpora:
    fixed [N, C] shape, zero-padding after each method's true length,
    indices in vocab range, @method_0 never surviving in starts/ends
    (replaced by @question = 1), labels in label-vocab range."""
    from code2vec_amd.data.builder import DatasetBuilder
    from code2vec_amd.data.synthetic import SyntheticSpec, write_synthetic_corpus
    from code2vec_amd.data.reader import CorpusReader
    from code2vec_amd.utils.options import Option
    import tempfile

    import torch

    for seed in (11, 23, 47):
        with tempfile.TemporaryDirectory() as d:
            files = write_synthetic_corpus(
                d, SyntheticSpec(n_methods=30, n_terminals=50, n_paths=40,
                                 max_contexts=12, seed=seed))
            reader = CorpusReader(files["corpus_path"],
                                  files["path_idx_path"],
                                  files["terminal_idx_path"])
            C = 8
            opt = Option(terminal_count=len(reader.terminal_vocab),
                         path_count=len(reader.path_vocab),
                         label_count=len(reader.label_vocab),
                         max_path_length=C, terminal_embed_size=8,
                         path_embed_size=8, encode_size=8,
                         dropout_prob=0.0, batch_size=4,
                         device=torch.device("cpu"))
            builder = DatasetBuilder(reader, opt, seed=seed)
            mzero = reader.terminal_vocab.stoi.get("@method_0", -1)
            for epoch in (0, 1):
                ds = builder.refresh_train_dataset(epoch)
                for arr in (ds.starts, ds.paths, ds.ends):
                    assert arr.shape[1] == C
                    assert arr.min() >= 0
                assert ds.starts.max() < len(reader.terminal_vocab)
                assert ds.ends.max() < len(reader.terminal_vocab)
                assert ds.paths.max() < len(reader.path_vocab)
                assert ds.labels.min() >= 0
                assert ds.labels.max() < len(reader.label_vocab)
                if mzero > 0:
                    # the method's own anonymized name must be masked
                    # (stoi already carries the +@question shift, and the
                    # corpus start/end values get the same shift at read)
                    assert not (ds.starts == mzero).any()
                    assert not (ds.ends == mzero).any()
                # pad runs are suffixes: once a column is 0 for a row in
                # starts, paths and ends are 0 there too
                pad = ds.starts == 0
                assert (ds.paths[pad] == 0).all()
                assert (ds.ends[pad] == 0).all()
