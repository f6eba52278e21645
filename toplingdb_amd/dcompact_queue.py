"""dcompact_queue — the node-level job queue of the MI355X dcompact worker
(BASELINE.json configs[4]: concurrent mixed compaction jobs sharded across
the GPUs of one node).

Mirrors the reference deployment shape: a dcompact node receives many
independent compaction jobs (job/attempt dirs, isolation —
db/compaction/compaction_executor.cc:305-330) and runs them concurrently;
nothing in the data path crosses jobs.  MI355X-native layout: one worker
PROCESS per GPU (torch.distributed, backend "nccl" = RCCL over xGMI on the
box, "gloo" in CPU tests), job METADATA broadcast from rank 0 as the only
collective, then a deterministic weighted assignment every
rank computes identically — zero further coordination, zero data-path
traffic between GPUs.

Assignment: LPT (longest-processing-time greedy) on input bytes — jobs
sorted by descending weight, each placed on the currently lightest rank.
Deterministic given the broadcast metadata, so every rank derives the same
schedule without another message.
"""
import heapq
import json
import os


def lpt_assign(weights, world):
    """weights: per-job input-byte weights -> list of rank ids per job.
    Deterministic: ties broken by rank id, then job order."""
    heap = [(0, r) for r in range(world)]
    heapq.heapify(heap)
    order = sorted(range(len(weights)), key=lambda j: (-weights[j], j))
    assign = [0] * len(weights)
    for j in order:
        load, r = heapq.heappop(heap)
        assign[j] = r
        heapq.heappush(heap, (load + weights[j], r))
    return assign


def build_job_mix(workdir, gen_sst, n_jobs=64, seed=0x5A1AD, world=8):
    """BASELINE configs[4] job mix: alternating L0→L1 (4-way, ~16 MiB runs)
    and L1→L2 (8-way, ~64 MiB runs) jobs over fully-overlapping uniform key
    ranges (worst-case overlap; the Zipfian skew of the reference workload
    concentrates overlap the same way).  Returns job descriptor dicts; the
    input SST generation is partitioned round-robin over ranks by the
    caller (rank r generates jobs j with j % world == r)."""
    jobs = []
    for j in range(n_jobs):
        small = j % 2 == 0
        runs = 4 if small else 8
        entries = 230_000 if small else 920_000
        jobs.append({
            "job_id": 1000 + j,
            "kind": "L0L1" if small else "L1L2",
            "runs": runs,
            "entries": entries,
            "seed": seed + j * 131,
            "paths": [os.path.join(workdir, "mix_j%d_r%d.sst" % (j, r))
                      for r in range(runs)],
            "bottommost": 0 if small else 1,
        })
    return jobs


def gen_job_inputs(job, gen_sst, compression=1):
    for r, p in enumerate(job["paths"]):
        if not os.path.exists(p):
            gen_sst(p, seed=job["seed"] + r, num_entries=job["entries"],
                    seq_base=1 + r * job["entries"], compression=compression)


class QueueCoordinator:
    """Broadcast job metadata once, assign deterministically, execute the
    local share, gather results.  `dist` is torch.distributed (initialized)
    or None for world=1."""

    def __init__(self, dist, rank, world):
        self.dist = dist
        self.rank = rank
        self.world = world

    def broadcast_jobs(self, jobs):
        """rank 0's job list -> every rank (the only collective on this
        path; metadata is a few KB)."""
        if self.dist is None or self.world == 1:
            return jobs
        payload = [json.dumps(jobs) if self.rank == 0 else None]
        self.dist.broadcast_object_list(payload, src=0)
        return json.loads(payload[0])

    def my_jobs(self, jobs, weights=None):
        if weights is None:
            weights = [sum(os.path.getsize(p) for p in j["paths"])
                       if all(os.path.exists(p) for p in j["paths"])
                       else j["runs"] * j["entries"] * 124
                       for j in jobs]
        assign = lpt_assign(weights, self.world)
        return [j for j, a in zip(jobs, assign) if a == self.rank], assign

    def gather_results(self, local_results):
        """all ranks -> rank 0: list of per-rank result lists."""
        if self.dist is None or self.world == 1:
            return [local_results]
        out = [None] * self.world
        self.dist.all_gather_object(out, local_results)
        return out
