"""Multi-process CPU test (gloo, world_size 2) of bench.py's distributed
timing protocol: per-rank independent jobs (weak scaling, no data-path
collective), barrier + MAX-over-ranks reduction of elapsed time."""
import os

import pytest
import torch
import torch.multiprocessing as mp


def _rank_main(rank, world, port, results):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    # per-rank independent "job" (sleep-free deterministic work)
    elapsed = 0.1 * (rank + 1)
    dist.barrier()
    t = torch.tensor([elapsed], dtype=torch.float64)
    dist.all_reduce(t, op=dist.ReduceOp.MAX)
    results[rank] = float(t.item())
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(120)
def test_gloo_max_over_ranks():
    world = 2
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29741
        ps = [ctx.Process(target=_rank_main, args=(r, world, port, results))
              for r in range(world)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(110)
            assert p.exitcode == 0
        # both ranks agree on the MAX elapsed
        assert abs(results[0] - 0.2) < 1e-9
        assert results[0] == results[1]
