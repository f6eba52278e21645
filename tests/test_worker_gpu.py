"""GPU parity tests: the dcompact worker's output SSTs must be BIT-IDENTICAL
to the CPU oracle's on the same job descriptors (the §8 coverage contract).
All tests here need a real MI355X (marked gpu; run via gpurun)."""
import os
import random

import pytest

import oracle
import toplingdb_amd as dcw

pytestmark = pytest.mark.gpu


@pytest.fixture(scope="module", autouse=True)
def gpu():
    dcw.init(0)
    yield
    dcw.shutdown()


def run_both(tmp_path, runs, **kw):
    out_gpu = tmp_path / "out_gpu"
    out_orc = tmp_path / "out_orc"
    out_gpu.mkdir()
    out_orc.mkdir()
    jd = dcw.make_job(runs, str(out_gpu), **kw)
    rg = dcw.execute(jd)
    jo = oracle.make_job(runs, str(out_orc), **kw)
    ro = oracle.execute(jo)
    return rg, ro


def assert_identical(rg, ro):
    assert rg["out_entries"] == ro["out_entries"]
    assert len(rg["files"]) == len(ro["files"])
    for fg, fo in zip(rg["files"], ro["files"]):
        assert fg["file_number"] == fo["file_number"]
        with open(fg["path"], "rb") as a, open(fo["path"], "rb") as b:
            da, db = a.read(), b.read()
        assert da == db, "GPU SST %s differs from oracle (sizes %d vs %d)" % (
            fg["path"], len(da), len(db))
        assert fg["smallest"] == fo["smallest"]
        assert fg["largest"] == fo["largest"]
        assert fg["smallest_seqno"] == fo["smallest_seqno"]
        assert fg["largest_seqno"] == fo["largest_seqno"]


def gen_runs(tmp_path, n_runs, entries_per_run, seed0=1, compression=0):
    runs = []
    for r in range(n_runs):
        p = str(tmp_path / ("in_%d.sst" % r))
        dcw.gen_sst(p, seed=seed0 + r, num_entries=entries_per_run,
                    seq_base=1 + r * entries_per_run, compression=compression)
        runs.append([p])
    return runs


def test_tiny_2way(tmp_path):
    runs = gen_runs(tmp_path, 2, 1000)
    rg, ro = run_both(tmp_path, runs, target_file_size=64 << 20)
    assert_identical(rg, ro)


def test_2way_64k_entries_nocomp(tmp_path):
    # config-2 shape scaled down: 2 runs, no compression, bottommost
    runs = gen_runs(tmp_path, 2, 65536)
    rg, ro = run_both(tmp_path, runs)
    assert_identical(rg, ro)


def test_8way_snappy_filecuts(tmp_path):
    # config-3 shape scaled down: 8 overlapping runs, snappy in+out,
    # multiple output files
    runs = gen_runs(tmp_path, 8, 30000, compression=1)
    rg, ro = run_both(tmp_path, runs, compression=1, target_file_size=2 << 20)
    assert_identical(rg, ro)
    assert len(rg["files"]) > 1


def test_nocomp_filecuts(tmp_path):
    runs = gen_runs(tmp_path, 4, 40000)
    rg, ro = run_both(tmp_path, runs, target_file_size=2 << 20)
    assert_identical(rg, ro)
    assert len(rg["files"]) > 1


def _write_kv_run(tmp_path, name, kvs, compression=0):
    """kvs: list of (ukey bytes, seq, type, value bytes), presorted."""
    es = [(oracle.make_ikey(k, s, t), v) for k, s, t, v in kvs]
    p = str(tmp_path / name)
    opts = oracle.default_table_opts(compression=compression)
    with open(p, "wb") as f:
        f.write(oracle.build_sst(es, opts))
    return p


def ikey_sort(kvs):
    return sorted(kvs, key=lambda e: (e[0], -((e[1] << 8) | e[2])))


@pytest.mark.parametrize("seed", [1234, 99, 424242])
def test_tombstones_and_snapshots_fuzz(tmp_path, seed):
    # randomized Put/Delete/SingleDelete streams with snapshots; GPU FSM vs
    # oracle FSM must agree bit-for-bit.  SingleDelete contract: never mix
    # SD and non-SD writes for the same key (reference requirement); model
    # each key as either SD-managed or Put/Delete-managed.
    rnd = random.Random(seed)
    nkeys = 400
    sd_keys = {k for k in range(nkeys) if rnd.random() < 0.4}
    seq = 1
    all_entries = []
    for k in range(nkeys):
        uk = b"k%014d" % k
        nver = rnd.randrange(1, 6)
        vers = []
        last_was_put = False
        for _ in range(nver):
            if k in sd_keys:
                # alternate put/SD, SD only directly deleting a put
                if last_was_put and rnd.random() < 0.6:
                    vers.append((uk, seq, 7, b""))
                    last_was_put = False
                else:
                    vers.append((uk, seq, 1, b"v%d" % seq))
                    last_was_put = True
            else:
                t = 0 if rnd.random() < 0.25 else 1
                vers.append((uk, seq, t, b"" if t == 0 else b"v%d" % seq))
            seq += 1
        all_entries.extend(vers)
    # distribute over 3 runs; each run must be internally sorted and a key's
    # versions must appear newest-first overall, so split by version rank
    runs_kv = [[], [], []]
    for e in all_entries:
        runs_kv[rnd.randrange(3)].append(e)
    runs = []
    for i, kv in enumerate(runs_kv):
        kv = ikey_sort(kv)
        runs.append([_write_kv_run(tmp_path, "f%d.sst" % i, kv)])
    snaps = sorted(rnd.sample(range(1, seq), 2))
    for bottom in (1, 0):
        out1 = tmp_path / ("g%d" % bottom)
        out2 = tmp_path / ("o%d" % bottom)
        out1.mkdir()
        out2.mkdir()
        jd = dcw.make_job(runs, str(out1), snapshots=snaps,
                          earliest_write_conflict_snapshot=snaps[0],
                          bottommost_level=bottom)
        jo = oracle.make_job(runs, str(out2), snapshots=snaps,
                             earliest_write_conflict_snapshot=snaps[0],
                             bottommost_level=bottom)
        rg = dcw.execute(jd)
        ro = oracle.execute(jo)
        assert_identical(rg, ro)


def test_staged_execute_matches_unstaged(tmp_path):
    runs = gen_runs(tmp_path, 2, 20000)
    out1 = tmp_path / "a"
    out2 = tmp_path / "b"
    out1.mkdir()
    out2.mkdir()
    jd1 = dcw.make_job(runs, str(out1))
    r1 = dcw.execute(jd1)
    jd2 = dcw.make_job(runs, str(out2))
    h = dcw.stage_inputs(jd2)
    jd2.staged_handle = h
    r2 = dcw.execute(jd2)
    dcw.release_staged(h)
    for f1, f2 in zip(r1["files"], r2["files"]):
        with open(f1["path"], "rb") as a, open(f2["path"], "rb") as b:
            assert a.read() == b.read()


def test_refuses_merge_operand(tmp_path):
    p = _write_kv_run(tmp_path, "m.sst", [(b"k%014d" % 1, 5, 2, b"op")])
    out = tmp_path / "out"
    out.mkdir()
    jd = dcw.make_job([[p]], str(out))
    with pytest.raises(RuntimeError):
        dcw.execute(jd)


def test_config2_exact_shape(tmp_path):
    # BASELINE.json configs[1]: 2-way merge of two ~64 MiB SSTs, 16B keys,
    # no compression, 1 GPU — full size, bit-exact vs oracle.
    runs = gen_runs(tmp_path, 2, 530_000)
    rg, ro = run_both(tmp_path, runs, compression=0, target_file_size=64 << 20)
    assert_identical(rg, ro)


def test_multi_file_level_run_gpu(tmp_path):
    # one run made of several non-overlapping files (a level-N run)
    import oracle as o
    level_files = []
    for part in range(3):
        # 16-byte user keys (must match the generator's uniform key length)
        kvs = [(b"k%015d" % i, 100 + i, 1, b"L1-%d" % i)
               for i in range(part * 5000, (part + 1) * 5000)]
        es = [(o.make_ikey(k, s, t), v) for k, s, t, v in kvs]
        p = str(tmp_path / ("l1_%d.sst" % part))
        with open(p, "wb") as f:
            f.write(o.build_sst(es))
        level_files.append(p)
    l0 = gen_runs(tmp_path, 1, 8000, seed0=99)[0]
    rg, ro = run_both(tmp_path, [l0, level_files])
    assert_identical(rg, ro)


def test_all_tombstones_bottommost_zero_files(tmp_path):
    # bottommost + no snapshots: every Delete is dropped (CompactionIterator
    # bottommost-delete rule) -> zero survivors -> zero output files
    kvs = [(b"k%014d" % i, 1 + i, 0, b"") for i in range(5000)]
    p = _write_kv_run(tmp_path, "d.sst", kvs)
    rg, ro = run_both(tmp_path, [[p]], bottommost_level=1)
    assert rg["out_entries"] == 0 and ro["out_entries"] == 0
    assert len(rg["files"]) == 0 and len(ro["files"]) == 0


def test_large_values_multi_chunk(tmp_path):
    # 1 KiB values force multiple plan->emit chunks per output file and
    # exercise the image-buffer pre-reserve/realloc guard under async D2H
    runs = []
    for r in range(2):
        p = str(tmp_path / ("lv%d.sst" % r))
        dcw.gen_sst(p, seed=7 + r, num_entries=40000, value_len=1024,
                    seq_base=1 + r * 40000, compression=1)
        runs.append([p])
    rg, ro = run_both(tmp_path, runs, compression=1, target_file_size=16 << 20)
    assert_identical(rg, ro)
    assert len(rg["files"]) > 1


def test_concurrent_jobs_bit_exact(tmp_path):
    # two different jobs executing concurrently from two threads must each
    # produce byte-identical output to their sequential oracle runs (the
    # worker pools GpuJob instances; production runs concurrent jobs)
    import concurrent.futures
    runs_a = gen_runs(tmp_path, 4, 60000, seed0=11, compression=1)
    runs_b = gen_runs(tmp_path, 4, 60000, seed0=77, compression=1)
    dirs = {}
    for tag in ("ga", "gb", "oa", "ob"):
        d = tmp_path / tag
        d.mkdir()
        dirs[tag] = str(d)
    kw = dict(compression=1, target_file_size=8 << 20)
    with concurrent.futures.ThreadPoolExecutor(2) as ex:
        fa = ex.submit(dcw.execute, dcw.make_job(runs_a, dirs["ga"], **kw))
        fb = ex.submit(dcw.execute, dcw.make_job(runs_b, dirs["gb"], **kw))
        ra, rb = fa.result(), fb.result()
    oa = oracle.execute(oracle.make_job(runs_a, dirs["oa"], **kw))
    ob = oracle.execute(oracle.make_job(runs_b, dirs["ob"], **kw))
    assert_identical(ra, oa)
    assert_identical(rb, ob)


def test_nondefault_table_options(tmp_path):
    # block_size / restart-interval / deviation variants flow through the
    # plan FSM, emit kernel and index builder identically to the oracle
    runs = gen_runs(tmp_path, 2, 40000, compression=1)
    for bs, ri, dev, comp in ((8192, 8, 10, 0), (2048, 4, 25, 1),
                              (4096, 1, 0, 1), (8192, 16, 10, 1),
                              (16384, 16, 10, 1)):
        og = tmp_path / ("g_%d_%d_%d" % (bs, ri, dev))
        oo = tmp_path / ("o_%d_%d_%d" % (bs, ri, dev))
        og.mkdir()
        oo.mkdir()
        kw = dict(compression=comp, target_file_size=8 << 20, block_size=bs,
                  block_restart_interval=ri, block_size_deviation=dev)
        rg = dcw.execute(dcw.make_job(runs, str(og), **kw))
        ro = oracle.execute(oracle.make_job(runs, str(oo), **kw))
        assert_identical(rg, ro)


def test_checksum_type_variants(tmp_path):
    # kNoChecksum(0) and kCRC32c(1) flow through k_checksum / the verify
    # pass and the trailers bit-compare; inputs are built with the same
    # checksum type so the decode side verifies them too
    for cs in (0, 1):
        runs = []
        for r in range(2):
            p = str(tmp_path / ("cs%d_%d.sst" % (cs, r)))
            dcw.gen_sst(p, seed=300 + cs * 10 + r, num_entries=20000,
                        seq_base=1 + r * 20000, compression=1,
                        checksum_type=cs)
            runs.append([p])
        og = tmp_path / ("gcs%d" % cs)
        oo = tmp_path / ("ocs%d" % cs)
        og.mkdir()
        oo.mkdir()
        kw = dict(compression=1, checksum_type=cs)
        rg = dcw.execute(dcw.make_job(runs, str(og), **kw))
        ro = oracle.execute(oracle.make_job(runs, str(oo), **kw))
        assert_identical(rg, ro)


def test_cancel_pending_job(tmp_path):
    # dcw_cancel before execute: the job aborts with DCW_CANCELLED (30)
    # and produces no output files (include/dcw.h cancel contract)
    runs = gen_runs(tmp_path, 1, 5000)
    out = tmp_path / "out"
    out.mkdir()
    jd = dcw.make_job(runs, str(out))
    jd.job_id = 7777
    dcw.cancel(7777)
    with pytest.raises(RuntimeError, match="cancel"):
        dcw.execute(jd)
    assert list(out.iterdir()) == []
    # the request was consumed: the same job now runs normally
    jd2 = dcw.make_job(runs, str(out))
    jd2.job_id = 7777
    r = dcw.execute(jd2)
    assert r["out_entries"] == 5000


def test_corrupted_input_fails_loudly(tmp_path):
    # flip one byte inside a data block: the worker must verify input
    # block checksums and fail the job (block_fetcher.cc semantics), not
    # emit output from corrupt data
    runs = gen_runs(tmp_path, 1, 20000, compression=1)
    p = runs[0][0]
    blob = bytearray(open(p, "rb").read())
    blob[len(blob) // 3] ^= 0xFF
    open(p, "wb").write(bytes(blob))
    out = tmp_path / "out"
    out.mkdir()
    jd = dcw.make_job(runs, str(out))
    with pytest.raises(RuntimeError):
        dcw.execute(jd)


def test_oversize_incompressible_entry(tmp_path):
    # regression (round-1 review): a single large entry is always admitted
    # into a block past block_size; a poorly-compressible one used to overrun
    # its fixed per-block compressed slot (silent device OOB write).  The
    # slot is now sized from the chunk's largest planned block.
    rnd = random.Random(31337)
    kvs = []
    for i in range(4000):
        if i % 500 == 250:
            v = bytes(rnd.getrandbits(8) for _ in range(12000))  # incompressible
        else:
            v = b"v" * 100
        kvs.append((b"k%014d" % i, 1 + i, 1, v))
    p = _write_kv_run(tmp_path, "big.sst", kvs, compression=0)
    rg, ro = run_both(tmp_path, [[p]], compression=1, bottommost_level=1)
    assert_identical(rg, ro)


def test_dense_restarts_over_255(tmp_path):
    # regression (round-1 review): >255 restarts per block used to wrap the
    # packed 8-bit restart count in the GPU block plan (silent corruption).
    # block_restart_interval=1 + empty values + 16 KiB blocks -> ~700
    # restarts per block; bit-exact vs oracle.
    runs = []
    for r in range(2):
        p = str(tmp_path / ("dr%d.sst" % r))
        dcw.gen_sst(p, seed=101 + r, num_entries=30000, key_len=8, value_len=0,
                    seq_base=1 + r * 30000)
        runs.append([p])
    rg, ro = run_both(tmp_path, runs, block_size=16384,
                      block_restart_interval=1)
    assert_identical(rg, ro)


def test_rejects_unsupported_checksum_type(tmp_path):
    # kXXHash(2)/kXXHash64(3) are declared in dcw.h but not implemented by
    # the worker: the job must be REFUSED (DB falls back local), not written
    # with zero checksums
    runs = gen_runs(tmp_path, 1, 1000)
    out = tmp_path / "out"
    out.mkdir()
    jd = dcw.make_job(runs, str(out), checksum_type=2)
    with pytest.raises(RuntimeError, match="checksum"):
        dcw.execute(jd)


def test_index_restart_interval_honored(tmp_path):
    # index_block_restart_interval flows into the index block for both
    # worker and oracle (bit-compare covers the index bytes)
    runs = gen_runs(tmp_path, 2, 20000)
    rg, ro = run_both(tmp_path, runs, index_block_restart_interval=4)
    assert_identical(rg, ro)
    # and differs from the default-1 output (the field is not ignored)
    og = tmp_path / "g1"
    og.mkdir()
    r1 = dcw.execute(dcw.make_job(runs, str(og)))
    b_custom = open(rg["files"][0]["path"], "rb").read()
    b_default = open(r1["files"][0]["path"], "rb").read()
    assert b_custom != b_default


def test_grandparent_cuts_match_oracle(tmp_path):
    # grandparent-aware file cutting (ShouldStopBefore boundary rules,
    # compaction_outputs.cc:231-352): GPU worker vs oracle, bit-exact,
    # including mid-block cuts.  Dense grandparents with huge sizes force
    # max_compaction_bytes cuts; smaller ones exercise the dynamic-size
    # rules.
    runs = gen_runs(tmp_path, 2, 60000)
    import random
    rnd = random.Random(5)
    # grandparent ranges over the 16B uniform key space
    gps = []
    lo = b"\x00" * 16
    for g in range(64):
        hi = bytes([4 * g + rnd.randrange(1, 4)]) + bytes(
            rnd.randrange(256) for _ in range(15))
        if hi <= lo:
            continue
        gps.append((lo, hi, rnd.choice([1 << 20, 1 << 26, 1 << 30])))
        lo = hi + b"\x01"
        lo = lo[:16]
    for mcb in [1 << 30, 8 << 30]:
        og = tmp_path / ("g%d" % (mcb >> 30))
        oo = tmp_path / ("o%d" % (mcb >> 30))
        og.mkdir()
        oo.mkdir()
        jd = dcw.make_job(runs, str(og), target_file_size=2 << 20,
                          max_compaction_bytes=mcb, grandparents=gps)
        jo = oracle.make_job(runs, str(oo), target_file_size=2 << 20,
                             max_compaction_bytes=mcb, grandparents=gps)
        rg = dcw.execute(jd)
        ro = oracle.execute(jo)
        assert_identical(rg, ro)
