"""Communication-backend primitives on a 2-process gloo world (the
reference's tests/distributed/test_dist_ops.py exercised its lazy c10d
backend; here the primitives are BucketedAllReduce / all_reduce_coalesced /
the PP tensor p2p protocol over plain torch.distributed)."""
import numpy as np
import os
import torch

from tests.utils.distributed import run_multiprocess


def _bucketed_worker(rank, world, outdir):
    import torch.distributed as dist
    from torchacc_amd.dist.backend import (BucketedAllReduce,
                                           all_reduce_coalesced)
    group = dist.group.WORLD
    sizes = (1000, 17, 4096)
    params = [torch.nn.Parameter(torch.zeros(n)) for n in sizes]

    def rank_grads(r):
        torch.manual_seed(100 + r)
        return [torch.randn(n) for n in sizes]

    grads = rank_grads(rank)
    expected = [sum(rank_grads(r)[i] for r in range(world)) / world
                for i in range(len(sizes))]
    reducer = BucketedAllReduce(group, bucket_mb=1)
    for p, g in zip(params, grads):
        p.grad = g.clone()
        reducer.add(p.grad)
    reducer.finalize(world)
    ok_bucketed = all(
        torch.allclose(p.grad, e, atol=1e-6)
        for p, e in zip(params, expected))

    # coalesced all-reduce
    ts = [torch.full((5,), float(rank + 1)), torch.full((3,), 2.0 * rank)]
    all_reduce_coalesced(ts, group=group)
    exp0 = sum(float(r + 1) for r in range(world))
    exp1 = sum(2.0 * r for r in range(world))
    ok_coalesced = torch.allclose(ts[0], torch.full((5,), exp0)) and \
        torch.allclose(ts[1], torch.full((3,), exp1))

    # p2p tensor protocol (meta handshake + payload)
    from torchacc_amd.dist.pp import p2p
    ok_p2p = True
    if rank == 0:
        payload = [torch.arange(6, dtype=torch.float32).reshape(2, 3),
                   torch.ones(4, dtype=torch.int64)]
        p2p.send_tensors(payload, 1, torch.device("cpu"))
    else:
        got = p2p.recv_tensors(0, torch.device("cpu"))
        ok_p2p = (len(got) == 2 and got[0].shape == (2, 3) and
                  got[0].dtype == torch.float32 and
                  torch.equal(got[0],
                              torch.arange(6, dtype=torch.float32)
                              .reshape(2, 3)) and
                  torch.equal(got[1], torch.ones(4, dtype=torch.int64)))
    np.save(os.path.join(outdir, f"ok_{rank}.npy"),
            np.array([ok_bucketed, ok_coalesced, ok_p2p]))


def test_backend_primitives(tmp_path):
    run_multiprocess(_bucketed_worker, world_size=2, args=(str(tmp_path),))
    for r in range(2):
        ok = np.load(tmp_path / f"ok_{r}.npy")
        assert ok.all(), f"rank {r}: [bucketed, coalesced, p2p] = {ok}"
