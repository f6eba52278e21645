"""Decompress scaling probe: per-launch time vs block count.

One wave decodes one block, so at B blocks the launch occupies
min(B, 16*256) resident decoders.  If the per-block serial time is
constant from B=1 (no sharing) to B=100k (full chip), the decoder is
genuinely latency-chain bound; if small-B runs are much faster per
block, the resident decoders contend for a shared resource (DS issue,
L1, scheduler) and fewer-decoders-per-CU layouts deserve another look.

Writes one line per size: blocks, decompress ms/launch, us/block.
"""
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import toplingdb_amd as dcw  # noqa: E402

dcw.init(0)
lib = dcw._lib

# ~33 entries of ~124 B raw per 4 KiB block
for tag, entries in (("1blk", 30), ("4blk", 130), ("64blk", 2100),
                     ("1kblk", 33000), ("16kblk", 530000),
                     ("100kblk", 2000000)):
    work = "/tmp/dsp_%s" % tag
    os.makedirs(work, exist_ok=True)
    p = os.path.join(work, "in.sst")
    if not os.path.exists(p):
        dcw.gen_sst(p, seed=1, num_entries=entries, compression=1)
    outd = os.path.join(work, "out")
    os.makedirs(outd, exist_ok=True)
    lib.dcw_kernel_stats_reset()
    # two executions; stats average over both (first may include warmup)
    for _ in range(2):
        for f in os.listdir(outd):
            os.unlink(os.path.join(outd, f))
        dcw.execute(dcw.make_job([[p]], outd, compression=1))
    import ctypes
    buf = ctypes.create_string_buffer(16384)
    lib.dcw_kernel_stats_json(buf, 16384)
    ks = json.loads(buf.value.decode())
    d = ks.get("decompress")
    if not d:
        print(tag, "no decompress launches")
        continue
    per_launch = d["ms"] / d["launches"]
    blocks = max(1, entries // 33)
    print("%-8s blocks~%-7d launches=%d ms/launch=%.4f us/block=%.2f"
          % (tag, blocks, d["launches"], per_launch,
             per_launch * 1000 / blocks))
dcw.shutdown()
