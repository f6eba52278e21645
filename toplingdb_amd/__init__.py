# toplingdb_amd — Python bindings for the MI355X dcompact worker (libdcw.so).
#
# The PRODUCT package: a GPU compaction-offload worker for ToplingDB's
# dcompact seam (CompactionExecutor, db/compaction/compaction_executor.h).
# The compute path is hand-written HIP for gfx950 (csrc/dcw_kernels.hip);
# this module is ctypes plumbing for tests and bench.py.  It never imports
# the oracle.
import ctypes as C
import os

_HERE = os.path.dirname(os.path.abspath(__file__))


class Run(C.Structure):
    _fields_ = [("files", C.POINTER(C.c_char_p)), ("num_files", C.c_uint32)]


class Grandparent(C.Structure):
    _fields_ = [
        ("smallest_ukey", C.POINTER(C.c_uint8)),
        ("smallest_len", C.c_uint32),
        ("largest_ukey", C.POINTER(C.c_uint8)),
        ("largest_len", C.c_uint32),
        ("file_size", C.c_uint64),
    ]


class LevelFiles(C.Structure):
    _fields_ = [("files", C.POINTER(Grandparent)), ("num_files", C.c_uint32)]


class JobDesc(C.Structure):
    _fields_ = [
        ("struct_size", C.c_uint32),
        ("job_id", C.c_int32),
        ("runs", C.POINTER(Run)),
        ("num_runs", C.c_uint32),
        ("output_dir", C.c_char_p),
        ("cf_id", C.c_uint32),
        ("cf_name", C.c_char_p),
        ("output_level", C.c_int32),
        ("bottommost_level", C.c_int32),
        ("compression", C.c_uint32),
        ("target_file_size", C.c_uint64),
        ("max_compaction_bytes", C.c_uint64),
        ("snapshots", C.POINTER(C.c_uint64)),
        ("num_snapshots", C.c_uint32),
        ("earliest_write_conflict_snapshot", C.c_uint64),
        ("next_file_number", C.c_uint64),
        ("db_id", C.c_char_p),
        ("db_session_id", C.c_char_p),
        ("db_host_id", C.c_char_p),
        ("current_time", C.c_uint64),
        ("oldest_ancester_time", C.c_uint64),
        ("grandparents", C.POINTER(Grandparent)),
        ("num_grandparents", C.c_uint32),
        ("levels_below_valid", C.c_int32),
        ("levels_below", C.POINTER(LevelFiles)),
        ("num_levels_below", C.c_uint32),
        ("block_size", C.c_uint32),
        ("block_restart_interval", C.c_uint32),
        ("format_version", C.c_uint32),
        ("checksum_type", C.c_uint32),
        ("index_block_restart_interval", C.c_uint32),
        ("level_compaction_dynamic_file_size", C.c_uint32),
        ("block_size_deviation", C.c_uint64),
        ("comparator_name", C.c_char_p),
        ("output_table_factory", C.c_uint32),
        ("bloom_millibits_per_key", C.c_uint32),
        ("flush_kv", C.POINTER(C.c_uint8)),
        ("flush_kv_bytes", C.c_uint64),
        ("flush_offsets", C.POINTER(C.c_uint64)),
        ("flush_num_entries", C.c_uint64),
        ("staged_handle", C.c_uint64),
    ]


class OutputFile(C.Structure):
    _fields_ = [
        ("path", C.c_char * 512),
        ("file_number", C.c_uint64),
        ("file_size", C.c_uint64),
        ("smallest_ikey", C.c_uint8 * 64),
        ("smallest_len", C.c_uint32),
        ("largest_ikey", C.c_uint8 * 64),
        ("largest_len", C.c_uint32),
        ("smallest_seqno", C.c_uint64),
        ("largest_seqno", C.c_uint64),
        ("num_entries", C.c_uint64),
    ]


class JobResult(C.Structure):
    _fields_ = [
        ("status", C.c_int32),
        ("error", C.c_char * 256),
        ("files", C.POINTER(OutputFile)),
        ("num_files", C.c_uint32),
        ("in_bytes", C.c_uint64),
        ("out_bytes", C.c_uint64),
        ("in_entries", C.c_uint64),
        ("out_entries", C.c_uint64),
        ("work_time_usec", C.c_uint64),
        ("t_read_usec", C.c_uint64),
        ("t_h2d_usec", C.c_uint64),
        ("t_gpu_usec", C.c_uint64),
        ("t_plan_usec", C.c_uint64),
        ("t_d2h_usec", C.c_uint64),
        ("t_write_usec", C.c_uint64),
    ]


_lib = None


def lib():
    """Load libdcw.so (in-tree).  Fails loudly if the HIP extension is not
    built — there is no CPU fallback in the product."""
    global _lib
    if _lib is None:
        path = os.path.join(_HERE, "libdcw.so")
        if not os.path.exists(path):
            raise RuntimeError(
                "toplingdb_amd/libdcw.so not built — run `make -C toplingdb_amd/csrc` "
                "(hipcc --offload-arch=gfx950)")
        _lib = C.CDLL(path)
        _lib.dcw_init.restype = C.c_int32
        _lib.dcw_init.argtypes = [C.c_int32]
        _lib.dcw_execute.restype = C.c_int32
        _lib.dcw_execute.argtypes = [C.POINTER(JobDesc), C.POINTER(JobResult)]
        _lib.dcw_free_result.argtypes = [C.POINTER(JobResult)]
        _lib.dcw_stage_inputs.restype = C.c_uint64
        _lib.dcw_stage_inputs.argtypes = [C.POINTER(JobDesc)]
        _lib.dcw_release_staged.argtypes = [C.c_uint64]
        _lib.dcw_gen_sst.restype = C.c_int32
        _lib.dcw_gen_sst.argtypes = [
            C.c_char_p, C.c_uint64, C.c_uint64, C.c_uint32, C.c_uint32,
            C.c_uint64, C.c_uint32, C.c_uint32, C.c_uint64, C.c_char_p,
            C.c_char_p, C.c_uint64
        ]
        _lib.dcw_version.restype = C.c_char_p
        _lib.dcw_cancel.argtypes = [C.c_int32]
    return _lib


def init(device=0):
    rc = lib().dcw_init(device)
    if rc != 0:
        raise RuntimeError("dcw_init failed (no usable gfx950 device?)")


def shutdown():
    lib().dcw_shutdown()


def cancel(job_id: int):
    lib().dcw_cancel(job_id)


def version() -> str:
    return lib().dcw_version().decode()


def gen_sst(path, seed, num_entries, key_len=16, value_len=100, seq_base=1,
            compression=0, checksum_type=4, file_number=1,
            db_id="DCW-TEST-DB-ID", db_session_id="DCWTESTSESSION",
            current_time=1757900000):
    """Synthetic input SST (harness; stands in for the DB host's
    fillrandom/flush write path — never used by dcw_execute)."""
    rc = lib().dcw_gen_sst(path.encode(), seed, num_entries, key_len, value_len,
                           seq_base, compression, checksum_type, file_number,
                           db_id.encode(), db_session_id.encode(), current_time)
    if rc != 0:
        raise RuntimeError("dcw_gen_sst failed")


def make_job(runs, output_dir, **kw) -> JobDesc:
    """Same layout/defaults as oracle.make_job (the shared boundary spec)."""
    d = JobDesc()
    d._keep = []
    d.struct_size = C.sizeof(JobDesc)
    run_arr = (Run * len(runs))()
    for i, files in enumerate(runs):
        arr = (C.c_char_p * len(files))(*[f.encode() for f in files])
        d._keep.append(arr)
        run_arr[i].files = arr
        run_arr[i].num_files = len(files)
    d._keep.append(run_arr)
    d.runs = run_arr
    d.num_runs = len(runs)
    d.output_dir = output_dir.encode()
    d.cf_id = 0
    d.cf_name = b"default"
    d.output_level = kw.pop("output_level", 2)
    d.bottommost_level = kw.pop("bottommost_level", 1)
    d.compression = kw.pop("compression", 0)
    d.target_file_size = kw.pop("target_file_size", 64 << 20)
    d.max_compaction_bytes = kw.pop("max_compaction_bytes", 25 * (64 << 20))
    snaps = kw.pop("snapshots", [])
    if snaps:
        sarr = (C.c_uint64 * len(snaps))(*snaps)
        d._keep.append(sarr)
        d.snapshots = sarr
        d.num_snapshots = len(snaps)
    d.earliest_write_conflict_snapshot = kw.pop(
        "earliest_write_conflict_snapshot", (1 << 56) - 1)
    d.next_file_number = kw.pop("next_file_number", 100)
    d.db_id = kw.pop("db_id", "DCW-TEST-DB-ID").encode()
    d.db_session_id = kw.pop("db_session_id", "DCWTESTSESSION").encode()
    d.db_host_id = kw.pop("db_host_id", "dcw-host").encode()
    d.current_time = kw.pop("current_time", 1757900000)
    d.oldest_ancester_time = kw.pop("oldest_ancester_time", 1757800000)
    gps = kw.pop("grandparents", [])
    if gps:
        garr = (Grandparent * len(gps))()
        for i, (sm, lg, fsz) in enumerate(gps):
            smb = (C.c_uint8 * len(sm)).from_buffer_copy(sm)
            lgb = (C.c_uint8 * len(lg)).from_buffer_copy(lg)
            d._keep += [smb, lgb]
            garr[i].smallest_ukey = smb
            garr[i].smallest_len = len(sm)
            garr[i].largest_ukey = lgb
            garr[i].largest_len = len(lg)
            garr[i].file_size = fsz
        d._keep.append(garr)
        d.grandparents = garr
        d.num_grandparents = len(gps)
    lvls = kw.pop("levels_below", [])
    d.levels_below_valid = kw.pop("levels_below_valid", 1)
    larr = (LevelFiles * max(len(lvls), 1))()
    for i, files in enumerate(lvls):
        rarr = (Grandparent * max(len(files), 1))()
        for j, (sm, lg, fsz) in enumerate(files):
            smb = (C.c_uint8 * len(sm)).from_buffer_copy(sm)
            lgb = (C.c_uint8 * len(lg)).from_buffer_copy(lg)
            d._keep += [smb, lgb]
            rarr[j].smallest_ukey = smb
            rarr[j].smallest_len = len(sm)
            rarr[j].largest_ukey = lgb
            rarr[j].largest_len = len(lg)
            rarr[j].file_size = fsz
        d._keep.append(rarr)
        larr[i].files = rarr
        larr[i].num_files = len(files)
    d._keep.append(larr)
    d.levels_below = larr
    d.num_levels_below = len(lvls)
    d.block_size = kw.pop("block_size", 4096)
    d.block_restart_interval = kw.pop("block_restart_interval", 16)
    d.format_version = 5
    d.checksum_type = kw.pop("checksum_type", 4)
    d.index_block_restart_interval = 1
    d.level_compaction_dynamic_file_size = 1
    d.block_size_deviation = kw.pop("block_size_deviation", 10)
    d.comparator_name = b"leveldb.BytewiseComparator"
    d.output_table_factory = kw.pop("output_table_factory", 0)
    d.bloom_millibits_per_key = kw.pop("bloom_millibits_per_key", 0)
    flush_entries = kw.pop("flush_entries", None)
    if flush_entries is not None:
        # sorted [(internal_key, value)] -> the raw flush record blob
        blob = bytearray()
        offs = []
        for k, v in flush_entries:
            offs.append(len(blob))
            blob += len(k).to_bytes(4, "little") + k
            blob += len(v).to_bytes(4, "little") + v
        offs.append(len(blob))
        bbuf = (C.c_uint8 * len(blob)).from_buffer_copy(bytes(blob))
        obuf = (C.c_uint64 * len(offs))(*offs)
        d._keep += [bbuf, obuf]
        d.flush_kv = bbuf
        d.flush_kv_bytes = len(blob)
        d.flush_offsets = obuf
        d.flush_num_entries = len(flush_entries)
    d.staged_handle = kw.pop("staged_handle", 0)
    for k, v in kw.items():
        setattr(d, k, v)
    return d


def execute(desc: JobDesc):
    res = JobResult()
    rc = lib().dcw_execute(C.byref(desc), C.byref(res))
    if rc != 0:
        err = res.error.decode(errors="replace")
        lib().dcw_free_result(C.byref(res))
        raise RuntimeError("dcw_execute failed (%d): %s" % (rc, err))
    files = []
    for i in range(res.num_files):
        f = res.files[i]
        files.append(dict(
            path=f.path.decode(),
            file_number=f.file_number,
            file_size=f.file_size,
            smallest=bytes(f.smallest_ikey[:f.smallest_len]),
            largest=bytes(f.largest_ikey[:f.largest_len]),
            smallest_seqno=f.smallest_seqno,
            largest_seqno=f.largest_seqno,
            num_entries=f.num_entries,
        ))
    out = dict(files=files, in_bytes=res.in_bytes, out_bytes=res.out_bytes,
               in_entries=res.in_entries, out_entries=res.out_entries,
               work_time_usec=res.work_time_usec,
               t_read_usec=res.t_read_usec, t_h2d_usec=res.t_h2d_usec,
               t_gpu_usec=res.t_gpu_usec, t_plan_usec=res.t_plan_usec,
               t_d2h_usec=res.t_d2h_usec, t_write_usec=res.t_write_usec)
    lib().dcw_free_result(C.byref(res))
    return out


def stage_inputs(desc: JobDesc) -> int:
    h = lib().dcw_stage_inputs(C.byref(desc))
    if h == 0:
        raise RuntimeError("dcw_stage_inputs failed")
    return h


def release_staged(handle: int):
    lib().dcw_release_staged(handle)
