"""Job-queue service (toplingdb_amd/dcompact_queue.py) CPU tests:
deterministic LPT assignment, and the full coordinator protocol
(broadcast -> identical assignment on every rank -> execute -> gather)
over gloo with world_size 8 — the CPU stand-in for the 8-GPU node
(BASELINE.json configs[4])."""
import os
import sys

import pytest
import torch.multiprocessing as mp

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from toplingdb_amd import dcompact_queue as dq


def test_lpt_assign_balance_and_determinism():
    weights = [100, 1, 50, 50, 100, 7, 7, 7, 90, 3]
    a1 = dq.lpt_assign(weights, 4)
    a2 = dq.lpt_assign(weights, 4)
    assert a1 == a2
    loads = [0] * 4
    for j, r in enumerate(a1):
        loads[r] += weights[j]
    # greedy LPT bound: max load <= avg + max_weight
    assert max(loads) <= sum(weights) / 4 + max(weights)
    # every job assigned exactly once to a valid rank
    assert all(0 <= r < 4 for r in a1) and len(a1) == len(weights)


def test_job_mix_shape(tmp_path):
    jobs = dq.build_job_mix(str(tmp_path), gen_sst=None, n_jobs=64)
    assert len(jobs) == 64
    kinds = {j["kind"] for j in jobs}
    assert kinds == {"L0L1", "L1L2"}
    assert all(j["runs"] == (4 if j["kind"] == "L0L1" else 8) for j in jobs)
    assert len({j["job_id"] for j in jobs}) == 64


def _rank_main(rank, world, port, results):
    import torch.distributed as dist
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    dist.init_process_group("gloo", rank=rank, world_size=world)
    coord = dq.QueueCoordinator(dist, rank, world)
    # rank 0 owns the job list; others pass None-shaped metadata
    jobs = None
    if rank == 0:
        jobs = [{"job_id": i, "w": (i * 37) % 11 + 1} for i in range(64)]
    jobs = coord.broadcast_jobs(jobs if rank == 0 else [])
    assert len(jobs) == 64  # metadata reached every rank
    weights = [j["w"] for j in jobs]
    mine, assign = coord.my_jobs(jobs, weights)
    # "execute" the local share: record job ids + the full assignment hash
    local = {"rank": rank, "done": sorted(j["job_id"] for j in mine),
             "assign": assign}
    gathered = coord.gather_results(local)
    if rank == 0:
        # every rank derived the SAME assignment with no extra messages
        for g in gathered:
            assert g["assign"] == gathered[0]["assign"]
        # every job executed exactly once across the node
        all_done = sorted(i for g in gathered for i in g["done"])
        assert all_done == list(range(64))
        results["ok"] = True
    dist.barrier()
    dist.destroy_process_group()


@pytest.mark.timeout(180)
def test_queue_world8_gloo():
    world = 8
    ctx = mp.get_context("spawn")
    with ctx.Manager() as mgr:
        results = mgr.dict()
        port = 29743
        ps = [ctx.Process(target=_rank_main, args=(r, world, port, results))
              for r in range(world)]
        for p in ps:
            p.start()
        for p in ps:
            p.join(170)
            assert p.exitcode == 0
        assert results.get("ok")
