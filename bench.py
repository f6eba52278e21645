#!/usr/bin/env python3
"""bench.py — dcompact-worker throughput on MI355X.

Metric (BASELINE.json): compacted MB/s (input SST bytes) per node, on the
8-way L1→L2 merge workload (BASELINE.json configs[2]): 8 × ~64 MiB
snappy-compressed SSTs, 16 B keys / 100 B values, bottommost output,
target_file_size 64 MiB.

A step = one END-TO-END compaction job: the timed region covers the input
SST file reads (tmpfs), H2D staging, the full GPU pipeline, plan FSM, D2H
and output file writes — everything the reference CPU worker's hot loop
pays (its input reads are inside the timed path, table/block_fetcher.cc:242).
A secondary `hbm_resident` figure (inputs staged in HBM before the region)
is reported alongside, never as `value`.

`cpu_baseline` runs the CPU oracle worker on ALL host cores (one job per
core, jobs-in-parallel — the production dcompact worker's own concurrency
model) over a bounded sample of the same 8-way workload shape; the
single-core figure is reported next to it.

Weak scaling: each rank owns independent jobs (distinct seeds) — compaction
jobs share nothing; no data-path collective.

  python bench.py --gpus N --steps K --warmup W
(driver launches N>1 via torch.distributed.run; RANK/LOCAL_RANK read from
the env; barriers over the default process group.)
"""
import argparse
import ctypes
import json
import os
import shutil
import sys
import time

REPO = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, REPO)

METRIC = "compacted MB/s (input SST bytes) per node, 8-way L1→L2 merge, 16 B keys"


def log(msg):
    print(msg, file=sys.stderr, flush=True)


def gen_inputs(dcw, work_dir, rank, runs, entries, compression, tag=""):
    paths = []
    for r in range(runs):
        p = os.path.join(work_dir, "in%s_r%d_%d.sst" % (tag, rank, r))
        if not os.path.exists(p):
            dcw.gen_sst(p, seed=0x746F706C696E6721 + rank * 1000 + r +
                        (0 if not tag else 777_000),
                        num_entries=entries, seq_base=1 + r * entries,
                        compression=compression)
        paths.append([p])
    return paths


# ---- all-core CPU baseline (runs in forked children BEFORE GPU init) ----
_CPU_CTX = {}


def _cpu_one_job(i):
    import oracle
    sdir = os.path.join(_CPU_CTX["work"], "cpu_out_%d" % i)
    os.makedirs(sdir, exist_ok=True)
    jo = oracle.make_job(_CPU_CTX["runs"], sdir,
                         compression=_CPU_CTX["compression"],
                         target_file_size=64 << 20, bottommost_level=1)
    r = oracle.execute(jo)
    shutil.rmtree(sdir, ignore_errors=True)  # bound tmpfs usage
    return r["in_bytes"]


def cpu_baseline(args, work, dcw, rank):
    """Oracle worker (the spec-v4 restatement, kind "port"): single-core and
    all-host-cores jobs-in-parallel, on a bounded 8-way sample of the same
    workload shape (SURVEY.md §8d's plan)."""
    import oracle
    entries = max(10_000, args.entries_per_run // 4)  # ~16 MiB/run sample
    sample = gen_inputs(dcw, work, rank, args.runs, entries,
                        args.compression, tag="cpu")
    sdir = os.path.join(work, "cpu_out_1t")
    os.makedirs(sdir, exist_ok=True)
    jo = oracle.make_job(sample, sdir, compression=args.compression,
                         target_file_size=64 << 20, bottommost_level=1)
    t0 = time.time()
    ro = oracle.execute(jo)
    t1 = time.time() - t0
    shutil.rmtree(sdir, ignore_errors=True)
    single = ro["in_bytes"] / t1 / 1e6
    cores = len(os.sched_getaffinity(0))
    _CPU_CTX.update(work=work, runs=sample, compression=args.compression)
    from concurrent.futures import ProcessPoolExecutor
    t0 = time.time()
    with ProcessPoolExecutor(max_workers=cores) as pe:
        totals = list(pe.map(_cpu_one_job, range(cores)))
    wall = time.time() - t0
    allcore = sum(totals) / wall / 1e6
    return {
        "value": round(allcore, 2),
        "unit": "MB/s",
        "cores": cores,
        "kind": "port",
        "single_core": round(single, 2),
        "sample": "%d jobs in parallel (1/core), each an 8-way merge of "
                  "8 x %.0f MiB SSTs (%.0f MiB/job), oracle worker"
                  % (cores, ro["in_bytes"] / len(sample) / 2**20,
                     ro["in_bytes"] / 2**20),
    }


def load_pmc_traffic(kernel):
    """Per-launch HBM bytes for `kernel` from the committed PMC capture
    (profiles/pmc_per_launch.json, produced by tools/parse_pmc.py from
    rocprofv3 --pmc runs; fetch doubled per the gfx950 FETCH_SIZE
    correction).  None when no capture matches."""
    p = os.path.join(REPO, "profiles", "pmc_per_launch.json")
    try:
        with open(p) as f:
            d = json.load(f)
        k = d["kernels"].get(kernel)
        if k is None:
            return None
        return float(k["fetch_bytes_per_launch"]) + \
            float(k.get("write_bytes_per_launch", 0.0))
    except Exception:
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)  # >= 2x jobs-in-flight
    ap.add_argument("--warmup", type=int, default=4)
    ap.add_argument("--runs", type=int, default=8)
    ap.add_argument("--entries-per-run", type=int, default=920_000,
                    help="~64 MiB snappy SST per run at 16B/100B")
    ap.add_argument("--compression", type=int, default=1, help="0=none 1=snappy")
    ap.add_argument("--workdir", default="/dev/shm/dcw_bench")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
    ap.add_argument("--skip-hbm-resident", action="store_true",
                    help="skip the secondary staged-inputs region")
    ap.add_argument("--table-factory", choices=["bbt", "dzt"], default="bbt",
                    help="dzt = DcwZipTable searchable-compressed output "
                         "(BASELINE configs[3]); bbt = BlockBasedTable")
    ap.add_argument("--job-mix", choices=["uniform", "mixed"], default="uniform",
                    help="mixed = BASELINE configs[4] job mix (alternating "
                         "L0→L1 4-way and L1→L2 8-way jobs) through the "
                         "node job queue; uniform = the headline configs[2] "
                         "workload (still queue-dispatched at N>1)")
    ap.add_argument("--mix-jobs", type=int, default=0,
                    help="job count for --job-mix mixed (0 = 8 per rank)")
    ap.add_argument("--jobs-in-flight", type=int, default=0,
                    help="concurrent compaction jobs per GPU (the production "
                         "dcompact worker runs concurrent jobs per node, "
                         "BASELINE.json configs[4]); 1 = strictly sequential; "
                         "0 = auto: 10 at 1 rank, scaled down with ranks so "
                         "per-node host threads and pinned memory stay sane")
    args = ap.parse_args()

    rank = int(os.environ.get("RANK", "0"))
    world = int(os.environ.get("WORLD_SIZE", str(args.gpus)))
    local_rank = int(os.environ.get("LOCAL_RANK", str(rank)))

    import torch
    dist = None
    if world > 1:
        import torch.distributed as tdist
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29571")
        backend = "nccl" if torch.cuda.is_available() else "gloo"
        if os.environ.get("DCW_FORCE_GLOO"):
            backend = "gloo"  # multi-rank smoke on a single-GPU box
        tdist.init_process_group(backend=backend, rank=rank, world_size=world)
        dist = tdist
    if not torch.cuda.is_available():
        log("bench.py: no GPU visible — this bench runs on an MI355X box")
        sys.exit(2)
    local_rank = min(local_rank, torch.cuda.device_count() - 1)
    torch.cuda.set_device(local_rank)

    import toplingdb_amd as dcw

    work = os.path.join(args.workdir, "r%d" % rank)
    shutil.rmtree(work, ignore_errors=True)
    os.makedirs(work, exist_ok=True)

    t_gen = time.time()
    runs = gen_inputs(dcw, work, rank, args.runs, args.entries_per_run,
                      args.compression)
    in_bytes = sum(os.path.getsize(r[0]) for r in runs)
    log("rank %d: generated %d runs, %.1f MiB in %.1fs" %
        (rank, len(runs), in_bytes / 2**20, time.time() - t_gen))

    # CPU baseline before GPU init (forked children must not inherit a live
    # HIP context); rank 0 at N=1 only
    cpu_base = None
    if rank == 0 and world == 1 and not args.skip_cpu_baseline:
        t0 = time.time()
        cpu_base = cpu_baseline(args, work, dcw, rank)
        log("cpu_baseline: %.1f MB/s on %d cores (single core %.1f) in %.1fs"
            % (cpu_base["value"], cpu_base["cores"], cpu_base["single_core"],
               time.time() - t0))

    dcw.init(local_rank)

    jif = args.jobs_in_flight
    if jif <= 0:
        jif = 10 if world == 1 else max(2, 12 // world)
    jif = max(1, jif)
    slot_dirs = []
    for i in range(jif):
        d = os.path.join(work, "out%d" % i)
        os.makedirs(d, exist_ok=True)
        slot_dirs.append(d)

    lib = dcw.lib()
    lib.dcw_kernel_stats_json.restype = ctypes.c_int32
    lib.dcw_kernel_stats_json.argtypes = [ctypes.c_char_p, ctypes.c_uint32]

    otf = 1 if args.table_factory == "dzt" else 0

    def job(slot, staged=0):
        return dcw.make_job(runs, slot_dirs[slot], compression=args.compression,
                            target_file_size=64 << 20, bottommost_level=1,
                            staged_handle=staged, output_table_factory=otf)

    from concurrent.futures import ThreadPoolExecutor
    pool = ThreadPoolExecutor(max_workers=jif)

    # ---- node job queue (toplingdb_amd/dcompact_queue.py): metadata
    # broadcast from rank 0 (the only collective), deterministic LPT
    # assignment, per-rank concurrent execution ----
    from toplingdb_amd import dcompact_queue as dq
    coord = dq.QueueCoordinator(dist, rank, world)

    def rank_input_paths(r):
        w = os.path.join(args.workdir, "r%d" % r)
        return [[os.path.join(w, "in_r%d_%d.sst" % (r, i))]
                for i in range(args.runs)]

    my_queue_jobs = None
    queue_total_bytes = None
    mix_jobs_meta = None
    if args.job_mix == "mixed":
        # BASELINE configs[4] mix through the queue: generation sharded
        # round-robin over ranks, then broadcast + LPT by input bytes
        n_mix = args.mix_jobs if args.mix_jobs > 0 else 8 * world
        mixdir = os.path.join(args.workdir, "mix")
        os.makedirs(mixdir, exist_ok=True)
        all_mix = dq.build_job_mix(mixdir, dcw.gen_sst, n_jobs=n_mix,
                                   world=world)
        for j in all_mix:
            if j["job_id"] % world == rank:
                dq.gen_job_inputs(j, dcw.gen_sst,
                                  compression=args.compression)
        if dist:
            dist.barrier()
        mix_jobs_meta = coord.broadcast_jobs(all_mix if rank == 0 else [])
        weights = [sum(os.path.getsize(p) for p in j["paths"])
                   for j in mix_jobs_meta]
        my_queue_jobs, _ = coord.my_jobs(mix_jobs_meta, weights)
        queue_total_bytes = sum(weights)
    elif world > 1:
        # headline workload, queue-dispatched: steps*world equal jobs, job
        # i over rank (i % world)'s input set; any rank may execute any set
        meta = [{"job_id": 5000 + i, "input_rank": i % world}
                for i in range(args.steps * world)]
        meta = coord.broadcast_jobs(meta if rank == 0 else [])
        my_queue_jobs, _ = coord.my_jobs(meta, [1] * len(meta))
        queue_total_bytes = in_bytes * len(meta)

    def exec_queue_job(jm, slot):
        if "paths" in jm:  # mixed-mode job
            return dcw.execute(dcw.make_job(
                [[p] for p in jm["paths"]], slot_dirs[slot],
                compression=args.compression, target_file_size=64 << 20,
                bottommost_level=jm["bottommost"]))
        return dcw.execute(dcw.make_job(
            rank_input_paths(jm["input_rank"]), slot_dirs[slot],
            compression=args.compression, target_file_size=64 << 20,
            bottommost_level=1))

    def run_queue():
        futs = [pool.submit(exec_queue_job, jm, i % jif)
                for i, jm in enumerate(my_queue_jobs)]
        return [f.result() for f in futs]

    def run_steps(k, staged=0):
        futs = [pool.submit(lambda i: dcw.execute(job(i % jif, staged=staged)), i)
                for i in range(k)]
        return [f.result() for f in futs]

    def barrier():
        if dist:
            dist.barrier()
        torch.cuda.synchronize()

    def timed(k, staged=0):
        barrier()
        t0 = time.time()
        last = run_steps(k, staged=staged)[-1]
        barrier()
        elapsed = time.time() - t0
        if dist:  # MAX over ranks
            t = torch.tensor([elapsed], dtype=torch.float64,
                             device="cuda" if dist.get_backend() == "nccl" else "cpu")
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elapsed = float(t.item())
        return elapsed, last

    # ---- warmup (end-to-end jobs: same shape as the timed region;
    #      own-rank inputs only — other ranks may still be generating) ----
    run_steps(max(args.warmup, jif))
    lib.dcw_kernel_stats_reset()

    # ---- PRIMARY: end-to-end (input read + H2D inside the region).
    # At N>1 (or --job-mix mixed) the jobs go through the node queue:
    # broadcast metadata, deterministic LPT dispatch, each rank executes
    # its share (the barrier in timed() also guarantees every rank's
    # generated inputs exist before any cross-rank job starts) ----
    if my_queue_jobs is not None:
        def run_queue_region():
            rs = run_queue()
            return rs[-1] if rs else None
        barrier()
        t0 = time.time()
        last = run_queue_region()
        barrier()
        elapsed = time.time() - t0
        if dist:
            t = torch.tensor([elapsed], dtype=torch.float64,
                             device="cuda" if dist.get_backend() == "nccl" else "cpu")
            dist.all_reduce(t, op=dist.ReduceOp.MAX)
            elapsed = float(t.item())
    else:
        elapsed, last = timed(args.steps)

    # per-kernel stats -> roofline for the dominant kernel
    buf = ctypes.create_string_buffer(16384)
    lib.dcw_kernel_stats_json(buf, 16384)
    kstats = json.loads(buf.value.decode())
    dom_name, dom = max(kstats.items(), key=lambda kv: kv[1]["ms"]) if kstats else (None, None)
    roofline = None
    if dom and dom["ms"] > 0:
        per_launch_ms = dom["ms"] / dom["launches"]
        per_launch_alg = dom["alg_bytes"] / dom["launches"]
        achieved_gbps = per_launch_alg / (per_launch_ms * 1e-3) / 1e9
        roofline = {
            "bound": "hbm",
            "kernel": dom_name,
            "achieved": round(achieved_gbps, 1),
            "peak": 8000.0,
            "unit": "GB/s",
            "frac": round(achieved_gbps / 8000.0, 4),
            "traffic": load_pmc_traffic(dom_name),
            "alg_bytes_per_launch": round(per_launch_alg),
            "ms_per_launch": round(per_launch_ms, 4),
        }

    # ---- SECONDARY: inputs already resident in HBM (staged) ----
    hbm_resident = None
    if not args.skip_hbm_resident and my_queue_jobs is None:
        handle = dcw.stage_inputs(job(0))
        run_steps(max(2, jif // 2), staged=handle)  # short re-warm
        e2, _ = timed(args.steps, staged=handle)
        dcw.release_staged(handle)
        hbm_resident = {
            "value": round(in_bytes * args.steps * world / e2 / 1e6, 2),
            "ms_per_step": round(e2 / args.steps * 1000, 3),
            "note": "inputs staged in HBM before the region (round-1 primary)",
        }

    if my_queue_jobs is not None:
        total_in = queue_total_bytes
        n_jobs = (len(mix_jobs_meta) if args.job_mix == "mixed"
                  else args.steps * world)
        per_rank_steps = max(1, n_jobs // max(world, 1))
    else:
        total_in = in_bytes * args.steps * world
        per_rank_steps = args.steps
    value = total_in / elapsed / 1e6  # MB/s, whole job aggregate
    out = {
        "metric": METRIC,
        "value": round(value, 2),
        "unit": "MB/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed / per_rank_steps * 1000, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # no published number for this path
        "dtype": "u8",
        "data": "synthetic",
        "config": {
            "workload": ("DcwZipTable build: searchable-compressed output "
                         "(BASELINE.json configs[3])" if otf else
                         "8-way L1→L2 merge, 8×~64MiB snappy SSTs, "
                         "16B key / 100B value (BASELINE.json configs[2])"),
            "output_table_factory": "dzt" if otf else "bbt",
            "runs": args.runs,
            "entries_per_run": args.entries_per_run,
            "input_bytes_per_job": in_bytes,
            "compression": "snappy" if args.compression else "none",
            "target_file_size": 64 << 20,
            "input": "tmpfs file read + parse + H2D inside the timed region "
                     "(per job)",
            "output": "tmpfs (/dev/shm), D2H + file write inside the timed region",
            "jobs_in_flight": jif,
            "dispatch": ("node job queue: metadata broadcast + LPT"
                         if my_queue_jobs is not None else "per-rank loop"),
            "mix_jobs": (len(mix_jobs_meta) if mix_jobs_meta else None),
        },
        "roofline": roofline,
        "cpu_baseline": cpu_base,
        "hbm_resident": hbm_resident,
        "phase_usec_last_step": ({k: last[k] for k in
                                  ("t_read_usec", "t_h2d_usec", "t_gpu_usec",
                                   "t_plan_usec", "t_d2h_usec", "t_write_usec",
                                   "work_time_usec")} if last else None),
        "kernels": kstats,
    }
    if rank == 0:
        print(json.dumps(out))
    dcw.shutdown()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
