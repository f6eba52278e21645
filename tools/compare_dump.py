#!/usr/bin/env python3
"""Dev tool: compare a dcw_debug survivor dump (gpurun_out/dbg.{meta,kv})
against the oracle's output KV stream for the same job (the oracle's output
files ARE the survivor stream)."""
import os
import struct
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import oracle


def load_dump(prefix):
    kvs = []
    with open(prefix + ".kv", "rb") as f:
        data = f.read()
    i = 0
    while i < len(data):
        kl = data[i]
        k = data[i + 1:i + 1 + kl]
        vl = struct.unpack("<I", data[i + 1 + kl:i + 5 + kl])[0]
        v = data[i + 5 + kl:i + 5 + kl + vl]
        kvs.append((k, v))
        i += 5 + kl + vl
    n = len(kvs)
    with open(prefix + ".meta", "rb") as f:
        m = f.read()
    shared = m[:n]
    klen = m[n:2 * n]
    vlen = struct.unpack("<%dI" % n, m[2 * n:2 * n + 4 * n])
    return kvs, shared, klen, vlen


def main():
    prefix = sys.argv[1] if len(sys.argv) > 1 else "gpurun_out/dbg"
    oracle_dir = sys.argv[2] if len(sys.argv) > 2 else "/tmp/plan_case/out"
    kvs, shared, klen, vlen = load_dump(prefix)
    print("gpu survivors:", len(kvs))
    okvs = []
    for name in sorted(os.listdir(oracle_dir)):
        if name.endswith(".sst"):
            with open(os.path.join(oracle_dir, name), "rb") as f:
                okvs += oracle.read_sst(f.read())
    print("oracle survivors:", len(okvs))
    n = min(len(kvs), len(okvs))
    bad = 0
    for i in range(n):
        gk, gv = kvs[i]
        ok, ov = okvs[i]
        if gk != ok or gv != ov:
            print("KV DIFF at %d:\n  gpu k=%s v[:20]=%s\n  orc k=%s v[:20]=%s" %
                  (i, gk.hex(), gv[:20].hex(), ok.hex(), ov[:20].hex()))
            bad += 1
            if bad > 5:
                return
    print("kv identical" if bad == 0 and len(kvs) == len(okvs) else "kv DIFFER")
    # shared check
    bad = 0
    for i in range(n):
        exp = 0
        if i:
            a, b = okvs[i - 1][0], okvs[i][0]
            m = min(len(a), len(b))
            while exp < m and a[exp] == b[exp]:
                exp += 1
        if shared[i] != exp:
            print("SHARED DIFF at %d: gpu=%d expected=%d  key[i-1]=%s key[i]=%s" %
                  (i, shared[i], exp, okvs[i - 1][0].hex(), okvs[i][0].hex()))
            bad += 1
            if bad > 5:
                return
        if klen[i] != len(okvs[i][0]) or vlen[i] != len(okvs[i][1]):
            print("LEN DIFF at %d: klen %d/%d vlen %d/%d" %
                  (i, klen[i], len(okvs[i][0]), vlen[i], len(okvs[i][1])))
            bad += 1
    print("shared/len identical" if bad == 0 else "meta DIFFER")


if __name__ == "__main__":
    main()
