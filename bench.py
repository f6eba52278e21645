#!/usr/bin/env python3
"""bench.py — dcompact-worker throughput on MI355X.

Metric (BASELINE.json): compacted MB/s (input SST bytes) per node, on the
8-way L1→L2 merge workload (BASELINE.json configs[2]): 8 × ~64 MiB
snappy-compressed SSTs, 16 B keys / 100 B values, bottommost output,
target_file_size 64 MiB.

A step = one compaction job over the staged inputs (inputs resident in HBM
when the timed region starts; outputs are D2H'd, assembled and written to
tmpfs inside the timed region).  Weak scaling: each rank owns independent
jobs (distinct seeds) — compaction jobs share nothing
(SURVEY.md §8e); no data-path collective.

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


def gen_inputs(dcw, work_dir, rank, runs, entries, compression):
    paths = []
    for r in range(runs):
        p = os.path.join(work_dir, "in_r%d_%d.sst" % (rank, r))
        if not os.path.exists(p):
            dcw.gen_sst(p, seed=0x746F706C696E6721 + rank * 1000 + r,
                        num_entries=entries, seq_base=1 + r * entries,
                        compression=compression)
        paths.append([p])
    return paths


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
    ap.add_argument("--cpu-baseline-runs", type=int, default=4,
                    help="bounded oracle sample (number of input runs)")
    ap.add_argument("--skip-cpu-baseline", action="store_true")
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
        tdist.init_process_group(
            backend="nccl" if torch.cuda.is_available() else "gloo",
            rank=rank, world_size=world)
        dist = tdist
    if not torch.cuda.is_available():
        log("bench.py: no GPU visible — this bench runs on an MI355X box")
        sys.exit(2)
    torch.cuda.set_device(local_rank)

    import toplingdb_amd as dcw
    dcw.init(local_rank)

    work = os.path.join(args.workdir, "r%d" % rank)
    outd = os.path.join(work, "out")
    shutil.rmtree(work, ignore_errors=True)
    os.makedirs(outd, exist_ok=True)

    t_gen = time.time()
    runs = gen_inputs(dcw, work, rank, args.runs, args.entries_per_run,
                      args.compression)
    in_bytes = sum(os.path.getsize(r[0]) for r in runs)
    log("rank %d: generated %d runs, %.1f MiB in %.1fs" %
        (rank, len(runs), in_bytes / 2**20, time.time() - t_gen))

    jif = args.jobs_in_flight
    if jif <= 0:
        jif = 10 if world == 1 else max(2, 12 // world)
    jif = max(1, jif)
    slot_dirs = []
    for i in range(jif):
        d = os.path.join(work, "out%d" % i)
        os.makedirs(d, exist_ok=True)
        slot_dirs.append(d)

    def job(slot, staged=0):
        return dcw.make_job(runs, slot_dirs[slot], compression=args.compression,
                            target_file_size=64 << 20, bottommost_level=1,
                            staged_handle=staged)

    handle = dcw.stage_inputs(job(0))
    lib = dcw.lib()
    lib.dcw_kernel_stats_json.restype = ctypes.c_int32
    lib.dcw_kernel_stats_json.argtypes = [ctypes.c_char_p, ctypes.c_uint32]

    def step(i):
        # slot picks the output dir; concurrent jobs share the staged
        # (read-only) inputs but nothing else
        res = dcw.execute(job(i % jif, staged=handle))
        return res

    from concurrent.futures import ThreadPoolExecutor
    pool = ThreadPoolExecutor(max_workers=jif)

    def run_steps(k):
        futs = [pool.submit(step, i) for i in range(k)]
        return [f.result() for f in futs]

    # warmup
    last = None
    rs = run_steps(max(args.warmup, jif))
    last = rs[-1]
    lib.dcw_kernel_stats_reset()

    def barrier():
        if dist:
            dist.barrier()
        torch.cuda.synchronize()

    barrier()
    t0 = time.time()
    last = run_steps(args.steps)[-1]
    barrier()
    elapsed = time.time() - t0
    # MAX over ranks
    if dist:
        t = torch.tensor([elapsed], dtype=torch.float64,
                         device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    # per-kernel stats -> roofline for the dominant kernel
    buf = ctypes.create_string_buffer(16384)
    lib.dcw_kernel_stats_json(buf, 16384)
    kstats = json.loads(buf.value.decode())
    dom_name, dom = max(kstats.items(), key=lambda kv: kv[1]["ms"]) if kstats else (None, None)
    roofline = None
    if dom and dom["ms"] > 0:
        achieved_gbps = dom["alg_bytes"] / (dom["ms"] * 1e-3) / 1e9
        roofline = {
            "bound": "hbm",
            "kernel": dom_name,
            "achieved": round(achieved_gbps, 1),
            "peak": 8000.0,
            "unit": "GB/s",
            "frac": round(achieved_gbps / 8000.0, 4),
            "traffic": None,
        }

    # CPU baseline (oracle restatement, "port"), rank 0 at N=1 only
    cpu_baseline = None
    if rank == 0 and world == 1 and not args.skip_cpu_baseline:
        import oracle
        nb = min(args.cpu_baseline_runs, len(runs))
        sample_runs = runs[:nb]
        sdir = os.path.join(work, "cpu_out")
        os.makedirs(sdir, exist_ok=True)
        jo = oracle.make_job(sample_runs, sdir, compression=args.compression,
                             target_file_size=64 << 20, bottommost_level=1)
        tb0 = time.time()
        ro = oracle.execute(jo)
        tb = time.time() - tb0
        mbps = ro["in_bytes"] / tb / 1e6
        cpu_baseline = {
            "value": round(mbps, 2),
            "unit": "MB/s",
            "cores": 1,
            "kind": "port",
            "sample": "%d of %d input runs (%.0f MiB) through the oracle worker, 1 thread"
                      % (nb, len(runs), ro["in_bytes"] / 2**20),
        }

    total_in = in_bytes * args.steps * world
    value = total_in / elapsed / 1e6  # MB/s, whole job aggregate
    out = {
        "metric": METRIC,
        "value": round(value, 2),
        "unit": "MB/s",
        "n_gpus": world,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": round(elapsed / args.steps * 1000, 3),
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,  # no published number for this path
        "dtype": "u8",
        "data": "synthetic",
        "config": {
            "workload": "8-way L1→L2 merge, 8×~64MiB snappy SSTs, "
                        "16B key / 100B value (BASELINE.json configs[2])",
            "runs": args.runs,
            "entries_per_run": args.entries_per_run,
            "input_bytes_per_job": in_bytes,
            "compression": "snappy" if args.compression else "none",
            "target_file_size": 64 << 20,
            "output": "tmpfs (/dev/shm), D2H + file write inside the timed region",
            "jobs_in_flight": jif,
        },
        "roofline": roofline,
        "cpu_baseline": cpu_baseline,
        "phase_usec_last_step": {k: last[k] for k in
                                 ("t_h2d_usec", "t_gpu_usec", "t_plan_usec",
                                  "t_d2h_usec", "t_write_usec", "work_time_usec")},
        "kernels": kstats,
    }
    if rank == 0:
        print(json.dumps(out))
    dcw.release_staged(handle)
    dcw.shutdown()
    if dist:
        dist.destroy_process_group()


if __name__ == "__main__":
    main()
