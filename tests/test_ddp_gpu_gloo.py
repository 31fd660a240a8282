"""DP integration on ONE GPU: two processes share cuda:0 over the gloo
backend (gloo moves CUDA tensors through host memory, so no one-GPU NCCL
restriction).  This exercises the exact production path of the 8-GPU run —
HIP model, autograd hooks, EARLY_GRAD_CALLBACKS launched inside backward,
grad-tensor adoption, finish()/zero_grad cycling — end to end."""

import os
import sys

import pytest
import torch
import torch.multiprocessing as mp

pytestmark = pytest.mark.gpu

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


def _worker(rank, world, port, result_file, mode="plain"):
    import torch.distributed as dist

    sys.path.insert(0, REPO)
    import numpy as np

    from code2vec_amd.data.synthetic import synthetic_batch
    from code2vec_amd.engine.optim import FusedAdam
    from code2vec_amd.models.code2vec import Code2VecHIP, init_logical_params
    from code2vec_amd.parallel.ddp import BucketedAllReduce
    from code2vec_amd.utils.options import Option

    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    dev = torch.device("cuda:0")
    torch.cuda.set_device(dev)

    opt = Option(terminal_count=2000, path_count=1500, label_count=300,
                 max_path_length=24, terminal_embed_size=100,
                 path_embed_size=100, encode_size=100, dropout_prob=0.0,
                 batch_size=16, device=dev)
    g = torch.Generator().manual_seed(5)
    model = Code2VecHIP(opt, init_logical_params(opt, g), device=dev).train()
    owned = ([model.terminal_embedding, model.path_embedding]
             if mode == "owned" else None)
    ddp = BucketedAllReduce(list(model.parameters()), world,
                            direct_threshold=64 * 1024, owned_params=owned,
                            owned_chunks=3)
    ddp.broadcast_parameters()
    optim = FusedAdam(model.parameters(), lr=0.01)
    w = torch.ones(opt.label_count, device=dev)

    rng = np.random.default_rng(100 + rank)  # DIFFERENT data per rank
    for step in range(3):
        s, p, e, y = synthetic_batch(rng, 16, opt.max_path_length,
                                     opt.terminal_count, opt.path_count,
                                     opt.label_count)
        s = torch.from_numpy(s).to(dev); p = torch.from_numpy(p).to(dev)
        e = torch.from_numpy(e).to(dev); y = torch.from_numpy(y).to(dev)
        ddp.zero_grad()
        out, _, _ = model(s, p, e, y)
        loss = model.loss(out, y, w)
        loss.backward()
        if mode == "owned":
            # chunked owner-buffer all-reduce + pipelined per-chunk Adam
            ddp.finish_and_step(optim)
        else:
            ddp.finish()  # raises loudly if grad adoption failed
            optim.step()
    torch.cuda.synchronize()

    # replicas must stay bit-identical after synchronized updates
    digest = float(model.terminal_embedding.detach().float().abs().sum())
    digests = [None] * world
    dist.all_gather_object(digests, digest)
    assert all(d == digests[0] for d in digests), digests

    if rank == 0:
        with open(result_file, "w") as f:
            f.write(f"ok {digests[0]}")
    dist.destroy_process_group()


@pytest.mark.timeout(600)
def test_hip_dp_world2_one_gpu(tmp_path):
    """Both DP paths on the HIP model under gloo world=2; their results
    must agree with each other (chunked owner-buffer all-reduce +
    pipelined row-chunk Adam == plain reduce + full step)."""
    digests = {}
    for i, mode in enumerate(("plain", "owned")):
        ctx = mp.get_context("spawn")
        result_file = str(tmp_path / f"dp_{mode}.txt")
        port = _free_port()
        procs = [
            ctx.Process(target=_worker,
                        args=(r, 2, port, result_file, mode))
            for r in range(2)
        ]
        for pr in procs:
            pr.start()
        for pr in procs:
            pr.join(timeout=500)
            assert pr.exitcode == 0
        txt = open(result_file).read()
        assert txt.startswith("ok")
        digests[mode] = txt.split()[1]
    assert digests["plain"] == digests["owned"], digests
