"""Multi-process CPU (gloo, world_size=2) tests of the DP layer:
bucketed all-reduce equivalence vs single-process large-batch training."""

import os
import sys

import numpy as np
import pytest
import torch
import torch.multiprocessing as mp

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def _free_port():
    """OS-assigned free TCP port (avoids TIME_WAIT collisions when the
    suite runs twice in quick succession)."""
    import socket

    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


def _worker(rank, world, port, result_file):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    from code2vec_amd.models.code2vec import Code2VecTorch, init_logical_params
    from code2vec_amd.parallel.ddp import BucketedAllReduce
    from code2vec_amd.utils.options import Option
    from code2vec_amd.data.synthetic import synthetic_batch

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    opt = Option(terminal_count=200, path_count=150, label_count=40,
                 max_path_length=12, terminal_embed_size=16,
                 path_embed_size=16, encode_size=20, dropout_prob=0.0)
    g = torch.Generator().manual_seed(7)
    model = Code2VecTorch(opt, init_logical_params(opt, g))
    # small buckets + low direct threshold: exercises BOTH the bucketed
    # path (LN/attention/bias) and the direct per-tensor path (embeddings)
    ddp = BucketedAllReduce(list(model.parameters()), world,
                            bucket_bytes=4096, direct_threshold=8192)
    ddp.broadcast_parameters()
    optim = torch.optim.Adam(model.parameters(), lr=0.01)
    w = torch.ones(opt.label_count)

    # each rank gets its own half of a fixed global batch
    rng = np.random.default_rng(99)
    s, p, e, y = synthetic_batch(rng, 8 * world, opt.max_path_length,
                                 opt.terminal_count, opt.path_count,
                                 opt.label_count)
    sl = slice(rank * 8, (rank + 1) * 8)
    s = torch.from_numpy(s[sl]); p = torch.from_numpy(p[sl])
    e = torch.from_numpy(e[sl]); y = torch.from_numpy(y[sl])

    for step in range(3):
        ddp.zero_grad()
        out, _, _ = model(s.long(), p.long(), e.long(), y)
        loss = model.loss(out, y, w)
        loss.backward()
        ddp.finish()
        optim.step()

    if rank == 0:
        torch.save({n: p.detach().clone() for n, p in model.named_parameters()},
                   result_file)
    dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("world", [2, 4])
def test_ddp2_matches_single_process_large_batch(tmp_path, world):
    """DP=2/4 with batch shards + grad averaging == single process on
    the full batch (same init, fp32 => near-bitwise)."""
    ctx = mp.get_context("spawn")
    port = _free_port()
    result_file = str(tmp_path / "params.pt")
    procs = [ctx.Process(target=_worker, args=(r, world, port, result_file))
             for r in range(world)]
    for pr in procs:
        pr.start()
    for pr in procs:
        pr.join(timeout=240)
        assert pr.exitcode == 0
    params = torch.load(result_file, weights_only=True)

    # single-process reference on the concatenated batch
    from code2vec_amd.models.code2vec import Code2VecTorch, init_logical_params
    from code2vec_amd.utils.options import Option
    from code2vec_amd.data.synthetic import synthetic_batch

    opt = Option(terminal_count=200, path_count=150, label_count=40,
                 max_path_length=12, terminal_embed_size=16,
                 path_embed_size=16, encode_size=20, dropout_prob=0.0)
    g = torch.Generator().manual_seed(7)
    model = Code2VecTorch(opt, init_logical_params(opt, g))
    optim = torch.optim.Adam(model.parameters(), lr=0.01)
    w = torch.ones(opt.label_count)
    rng = np.random.default_rng(99)
    s, p, e, y = synthetic_batch(rng, 8 * world, opt.max_path_length,
                                 opt.terminal_count, opt.path_count,
                                 opt.label_count)
    s = torch.from_numpy(s); p = torch.from_numpy(p)
    e = torch.from_numpy(e); y = torch.from_numpy(y)
    for step in range(3):
        optim.zero_grad()
        out, _, _ = model(s.long(), p.long(), e.long(), y)
        # average of the two half-batch means == full-batch mean (equal sizes)
        loss = model.loss(out, y, w)
        loss.backward()
        optim.step()

    for name, p_ref in model.named_parameters():
        p_ddp = params[name]
        assert torch.allclose(p_ddp, p_ref, atol=1e-5), name


def _worker_trainer(rank, world, port, tmpdir, result_file, n_methods=40):
    """Full Trainer epoch under gloo world=2 (sharded data, reduced metrics)."""
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    from code2vec_amd.data.builder import DatasetBuilder
    from code2vec_amd.data.reader import CorpusReader
    from code2vec_amd.data.synthetic import SyntheticSpec, write_synthetic_corpus
    from code2vec_amd.engine.trainer import Trainer, TrainerConfig
    from code2vec_amd.models.code2vec import build_model, init_logical_params
    from code2vec_amd.parallel.dist import DistContext
    from code2vec_amd.utils.options import Option

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    files = write_synthetic_corpus(
        os.path.join(tmpdir, "data"),
        SyntheticSpec(n_methods=n_methods, n_terminals=60, n_paths=50,
                      max_contexts=16, seed=5),
    )
    reader = CorpusReader(files["corpus_path"], files["path_idx_path"],
                          files["terminal_idx_path"])
    opt = Option(terminal_count=len(reader.terminal_vocab),
                 path_count=len(reader.path_vocab),
                 label_count=len(reader.label_vocab),
                 max_path_length=10, terminal_embed_size=12,
                 path_embed_size=12, encode_size=16, dropout_prob=0.0,
                 batch_size=8, device=torch.device("cpu"))
    builder = DatasetBuilder(reader, opt, seed=3, rank=rank, world_size=world)
    g = torch.Generator().manual_seed(1)
    model = build_model(opt, backend="torch", logical=init_logical_params(opt, g))
    ctx = DistContext(rank, world, rank, torch.device("cpu"))

    class A:
        max_epoch = 2; lr = 0.01; beta_min = 0.9; beta_max = 0.999
        weight_decay = 0.0; model_path = os.path.join(tmpdir, f"out{rank}")
        vectors_path = os.path.join(tmpdir, f"out{rank}", "code.vec")
        test_result_path = None; env = None; print_sample_cycle = 0
        eval_method = "subtoken"; random_seed = 3; batch_size = 8

    trainer = Trainer(TrainerConfig(A), opt, reader, builder, model, ctx)
    obj = trainer.train()
    if rank == 0:
        with open(result_file, "w") as f:
            f.write(str(obj))
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_trainer_world2_runs(tmp_path):
    ctx = mp.get_context("spawn")
    result_file = str(tmp_path / "obj.txt")
    port = _free_port()
    procs = [
        ctx.Process(target=_worker_trainer,
                    args=(r, 2, port, str(tmp_path), result_file))
        for r in range(2)
    ]
    for pr in procs:
        pr.start()
    for pr in procs:
        pr.join(timeout=240)
        assert pr.exitcode == 0
    obj = float(open(result_file).read())
    assert 0.0 <= obj <= 1.0
    # artifact export is a rank-0-only side effect (trainer.py guards on
    # ctx.is_rank0): rank 0's dir holds checkpoint+vectors, rank 1's none
    assert os.path.exists(str(tmp_path / "out0" / "code2vec.model"))
    assert os.path.exists(str(tmp_path / "out0" / "code.vec"))
    assert not os.path.exists(str(tmp_path / "out1" / "code2vec.model"))
    assert not os.path.exists(str(tmp_path / "out1" / "code.vec"))


@pytest.mark.timeout(300)
def test_trainer_world2_uneven_dataset(tmp_path):
    """n_methods=41 -> 33 train items, not divisible by world*batch: without
    equal-size shards one rank runs an extra step and its collectives
    deadlock (the ADVICE.md stride-sharding finding).  Must complete."""
    ctx = mp.get_context("spawn")
    result_file = str(tmp_path / "obj.txt")
    port = _free_port()
    procs = [
        ctx.Process(target=_worker_trainer,
                    args=(r, 2, port, str(tmp_path), result_file, 41))
        for r in range(2)
    ]
    for pr in procs:
        pr.start()
    for pr in procs:
        pr.join(timeout=240)
        assert pr.exitcode == 0
    obj = float(open(result_file).read())
    assert 0.0 <= obj <= 1.0


def _worker_early_cb(rank, world, port, result_file):
    """Exercise the early-gradient-callback path (the one the embedding
    backward uses on GPU): cb launches the all-reduce before the hook."""
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    from code2vec_amd.parallel.ddp import BucketedAllReduce
    from code2vec_amd.ops import functional as Fn

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    p = torch.nn.Parameter(torch.zeros(4096))
    ddp = BucketedAllReduce([p], world, direct_threshold=1024)
    # early callbacks were removed (see ddp._on_grad_ready note): no
    # registration must happen
    assert p.data_ptr() not in Fn.EARLY_GRAD_CALLBACKS

    # step 1: hook-path direct reduce
    g = torch.full((4096,), float(rank + 1))
    p.grad = g
    ddp._on_grad_ready(p)
    ddp.finish()
    expected = (sum(range(1, world + 1)) / world)
    assert torch.allclose(p.grad, torch.full((4096,), expected)), p.grad[0]

    # step 2: second cycle after reset
    ddp.zero_grad()
    g2 = torch.full((4096,), float(10 * (rank + 1)))
    p.grad = g2
    ddp._on_grad_ready(p)
    ddp.finish()
    expected2 = 10 * sum(range(1, world + 1)) / world
    assert torch.allclose(p.grad, torch.full((4096,), expected2)), p.grad[0]

    if rank == 0:
        with open(result_file, "w") as f:
            f.write("ok")
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_early_grad_callback_world2(tmp_path):
    ctx = mp.get_context("spawn")
    result_file = str(tmp_path / "cb.txt")
    port = _free_port()
    procs = [ctx.Process(target=_worker_early_cb,
                         args=(r, 2, port, result_file)) for r in range(2)]
    for pr in procs:
        pr.start()
    for pr in procs:
        pr.join(timeout=240)
        assert pr.exitcode == 0
    assert open(result_file).read() == "ok"


def _worker_owned(rank, world, port, result_file):
    """Owner-buffer chunked all-reduce + pipelined row-chunk step under
    gloo: must equal the dense all-reduce + full step on the same grads."""
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    from code2vec_amd.ops import functional as Fn
    from code2vec_amd.parallel.ddp import BucketedAllReduce

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    class MockSGD:
        """Implements the FusedAdam step/step_rows contract on CPU."""

        def __init__(self, params, lr=0.1):
            self.params = list(params)
            self.lr = lr

        def step(self, exclude_ids=None):
            for p in self.params:
                if p.grad is None or (exclude_ids and id(p) in exclude_ids):
                    continue
                p.data -= self.lr * p.grad

        def step_rows(self, p, grad2d, lo, hi):
            p.data[lo:hi] -= self.lr * grad2d[lo:hi].float()

    torch.manual_seed(7)
    N, W = 37, 8  # odd row count: uneven chunk tail
    init = torch.randn(N, W)
    p_owned = torch.nn.Parameter(init.clone())
    p_plain = torch.nn.Parameter(init.clone())
    grads = [torch.randn(N, W) * (rank + 1 + s) for s in range(2)]

    ddp = BucketedAllReduce([p_owned], world, owned_params=[p_owned],
                            owned_chunks=4)
    optim = MockSGD([p_owned])
    key = p_owned.data_ptr()
    assert key in Fn.OWNED_GRAD_KEYS
    for g in grads:
        ddp.zero_grad()
        Fn.EARLY_GRAD_CALLBACKS[key](g.clone())  # as the backward would
        ddp.finish_and_step(optim)

    # dense reference on the plain param
    for g in grads:
        gd = g.clone()
        dist.all_reduce(gd)
        p_plain.data -= 0.1 * (gd / world)

    assert torch.allclose(p_owned.detach(), p_plain.detach(), atol=1e-6)

    # the non-pipelined finish() publishes the reduced grad as p.grad
    ddp.zero_grad()
    Fn.EARLY_GRAD_CALLBACKS[key](grads[0].clone())
    ddp.finish()
    gd = grads[0].clone()
    dist.all_reduce(gd)
    assert torch.allclose(p_owned.grad, gd / world, atol=1e-6)
    ddp.close()
    assert key not in Fn.OWNED_GRAD_KEYS

    if rank == 0:
        with open(result_file, "w") as f:
            f.write("ok")
    dist.destroy_process_group()


@pytest.mark.timeout(300)
def test_owned_chunked_allreduce_world2(tmp_path):
    ctx = mp.get_context("spawn")
    result_file = str(tmp_path / "owned.txt")
    port = _free_port()
    procs = [ctx.Process(target=_worker_owned,
                         args=(r, 2, port, result_file)) for r in range(2)]
    for pr in procs:
        pr.start()
    for pr in procs:
        pr.join(timeout=240)
        assert pr.exitcode == 0
    assert open(result_file).read() == "ok"


def _worker_tp_head(rank, world, port, result_file):
    """Vocab-sharded head+loss (parallel/tp_head.py) == dense reference:
    loss, dcv, and the reassembled dW/dbias shards must match."""
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    from code2vec_amd.ops import reference as R
    from code2vec_amd.parallel.tp_head import (shard_rows,
                                               vocab_parallel_head_loss)

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)

    torch.manual_seed(11)
    B, E, L = 16, 32, 51  # odd L: uneven last shard
    cv = torch.randn(B, E)
    w = torch.randn(L, E) * 0.2
    b = torch.randn(L) * 0.1
    y = torch.randint(0, L, (B,))
    wt = torch.rand(L) + 0.5

    per = (L + world - 1) // world
    cvh = cv.clone().requires_grad_(True)
    wsh = shard_rows(w, rank, world).clone().requires_grad_(True)
    bsh = shard_rows(b, rank, world).clone().requires_grad_(True)
    loss = vocab_parallel_head_loss(cvh, wsh, bsh, y,
                                    shard_rows(wt, rank, world),
                                    rank * per, L)
    loss.backward()

    cvr = cv.clone().requires_grad_(True)
    wr = w.clone().requires_grad_(True)
    br = b.clone().requires_grad_(True)
    ref = R.logsoftmax_nll(cvr @ wr.t() + br, y, wt)
    ref.backward()

    assert torch.allclose(loss, ref, atol=1e-5), (float(loss), float(ref))
    assert torch.allclose(cvh.grad, cvr.grad, atol=1e-5)
    assert torch.allclose(wsh.grad, shard_rows(wr.grad, rank, world),
                          atol=1e-5)
    assert torch.allclose(bsh.grad, shard_rows(br.grad, rank, world),
                          atol=1e-5)
    if rank == 0:
        with open(result_file, "w") as f:
            f.write("ok")
    dist.destroy_process_group()


@pytest.mark.timeout(300)
@pytest.mark.parametrize("world", [2, 3])
def test_vocab_parallel_head_matches_dense(tmp_path, world):
    ctx = mp.get_context("spawn")
    port = _free_port()
    result_file = str(tmp_path / "tp.txt")
    procs = [ctx.Process(target=_worker_tp_head,
                         args=(r, world, port, result_file))
             for r in range(world)]
    for pr in procs:
        pr.start()
    for pr in procs:
        pr.join(timeout=240)
        assert pr.exitcode == 0
    assert open(result_file).read() == "ok"
