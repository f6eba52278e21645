"""tmpfs read-bandwidth microbench (host-side diagnostic for the worker's
input-load phase).  Measures aggregate GB/s of segmented preads from
/dev/shm at several thread counts, into pageable and (if torch+GPU
present) pinned destinations — the same access pattern load_inputs uses.

Usage: python tools/read_bw.py [probe_dir]
"""
import concurrent.futures as cf
import mmap
import os
import sys
import time

DIR = sys.argv[1] if len(sys.argv) > 1 else "/dev/shm/dcw_read_probe"
NFILES = 8
FSIZE = 256 << 20
SEG = 16 << 20


def setup():
    os.makedirs(DIR, exist_ok=True)
    blk = os.urandom(1 << 20) * 16  # 16 MB pattern
    for i in range(NFILES):
        p = os.path.join(DIR, "f%d" % i)
        if os.path.exists(p) and os.path.getsize(p) == FSIZE:
            continue
        with open(p, "wb") as f:
            for _ in range(FSIZE // len(blk)):
                f.write(blk)
    return [os.path.join(DIR, "f%d" % i) for i in range(NFILES)]


def read_file(path, dst):
    fd = os.open(path, os.O_RDONLY)
    try:
        off = 0
        while off < FSIZE:
            n = os.preadv(fd, [dst[off:off + SEG]], off)
            if n <= 0:
                break
            off += n
    finally:
        os.close(fd)
    return off


def bench(paths, nthreads, bufs, label):
    t0 = time.time()
    with cf.ThreadPoolExecutor(nthreads) as ex:
        futs = []
        for i in range(nthreads):
            futs.append(ex.submit(read_file, paths[i % NFILES],
                                  bufs[i % len(bufs)]))
        total = sum(f.result() for f in futs)
    dt = time.time() - t0
    print("%-28s threads=%2d  %.1f GB in %.3fs = %.1f GB/s" %
          (label, nthreads, total / 1e9, dt, total / dt / 1e9))


def main():
    paths = setup()
    pageable = [memoryview(bytearray(FSIZE)) for _ in range(4)]
    for t in (1, 4, 10, 24, 40):
        bench(paths, t, pageable, "pageable dst")
    try:
        import torch
        if torch.cuda.is_available():
            pinned = [memoryview(
                torch.empty(FSIZE, dtype=torch.uint8, pin_memory=True)
                .numpy()) for _ in range(4)]
            for t in (1, 10, 40):
                bench(paths, t, pinned, "pinned dst")
    except Exception as e:
        print("pinned leg skipped:", e)


if __name__ == "__main__":
    main()
