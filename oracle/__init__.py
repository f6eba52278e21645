# oracle/__init__.py — ctypes wrapper around liborc.so.
# TEST INFRASTRUCTURE ONLY: imported by tests/, __graft_entry__.smoke()'s
# checker and bench.py's cpu_baseline leg — never by the product package.
import ctypes as C
import os

_HERE = os.path.dirname(os.path.abspath(__file__))


def _load():
    path = os.path.join(_HERE, "liborc.so")
    if not os.path.exists(path):
        raise RuntimeError(
            "oracle/liborc.so not built — run `make -C oracle` (gcc only)")
    return C.CDLL(path)


_lib = _load()


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


_lib.orc_crc32c.restype = C.c_uint32
_lib.orc_crc32c.argtypes = [C.c_void_p, C.c_size_t]
_lib.orc_crc32c_masked.restype = C.c_uint32
_lib.orc_crc32c_masked.argtypes = [C.c_void_p, C.c_size_t]
_lib.orc_xxh3_64.restype = C.c_uint64
_lib.orc_xxh3_64.argtypes = [C.c_void_p, C.c_size_t]
_lib.orc_xxph3_64.restype = C.c_uint64
_lib.orc_xxph3_64.argtypes = [C.c_void_p, C.c_size_t]
_lib.orc_block_checksum.restype = C.c_uint32
_lib.orc_block_checksum.argtypes = [C.c_uint32, C.c_void_p, C.c_size_t, C.c_uint8]
_lib.orc_snappy_max_compressed.restype = C.c_size_t
_lib.orc_snappy_max_compressed.argtypes = [C.c_size_t]
_lib.orc_snappy_compress.restype = C.c_size_t
_lib.orc_snappy_compress.argtypes = [C.c_char_p, C.c_size_t, C.c_void_p]
_lib.orc_snappy_uncompressed_len.restype = C.c_size_t
_lib.orc_snappy_uncompressed_len.argtypes = [C.c_char_p, C.c_size_t]
_lib.orc_snappy_uncompress.restype = C.c_size_t
_lib.orc_snappy_uncompress.argtypes = [C.c_char_p, C.c_size_t, C.c_void_p, C.c_size_t]
_lib.orc_ikey_compare.restype = C.c_int
_lib.orc_ikey_compare.argtypes = [C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
_lib.orc_execute.restype = C.c_int32
_lib.orc_execute.argtypes = [C.POINTER(JobDesc), C.POINTER(JobResult)]
_lib.orc_free_result.argtypes = [C.POINTER(JobResult)]


def crc32c(data: bytes) -> int:
    return _lib.orc_crc32c(data, len(data))


def crc32c_masked(data: bytes) -> int:
    return _lib.orc_crc32c_masked(data, len(data))


def xxh3_64(data: bytes) -> int:
    return _lib.orc_xxh3_64(data, len(data))


def xxph3_64(data: bytes) -> int:
    return _lib.orc_xxph3_64(data, len(data))


def block_checksum(cstype: int, data: bytes, last_byte: int) -> int:
    return _lib.orc_block_checksum(cstype, data, len(data), last_byte)


def snappy_compress(data: bytes) -> bytes:
    out = C.create_string_buffer(_lib.orc_snappy_max_compressed(len(data)))
    n = _lib.orc_snappy_compress(data, len(data), out)
    return out.raw[:n]


def snappy_uncompress(data: bytes) -> bytes:
    ulen = _lib.orc_snappy_uncompressed_len(data, len(data))
    if ulen == C.c_size_t(-1).value:
        raise ValueError("bad snappy preamble")
    out = C.create_string_buffer(max(ulen, 1))
    n = _lib.orc_snappy_uncompress(data, len(data), out, ulen)
    if n != ulen:
        raise ValueError("snappy corruption")
    return out.raw[:ulen]


def ikey_compare(a: bytes, b: bytes) -> int:
    return _lib.orc_ikey_compare(a, len(a), b, len(b))


# ---- dictionary snappy ("DZT dict codec v1") ----
def snap_dict_table(dictionary: bytes):
    tab = (C.c_uint32 * 2048)()
    _lib.orc_snap_dict_table(dictionary, len(dictionary), tab)
    return tab


def snappy_compress_dict(dictionary: bytes, data: bytes) -> bytes:
    tab = snap_dict_table(dictionary)
    out = C.create_string_buffer(_lib.orc_snappy_max_compressed(len(data)) + 16)
    n = _lib.orc_snappy_compress_dict(dictionary, len(dictionary), tab, data,
                                      len(data), out)
    return out.raw[:n]


def snappy_uncompress_dict(dictionary: bytes, data: bytes) -> bytes:
    ulen = _lib.orc_snappy_uncompressed_len(data, len(data))
    if ulen == C.c_size_t(-1).value:
        raise ValueError("bad snappy preamble")
    out = C.create_string_buffer(max(ulen, 1))
    n = _lib.orc_snappy_uncompress_dict(dictionary, len(dictionary), data,
                                        len(data), out, ulen)
    if n != ulen:
        raise ValueError("dict snappy corruption")
    return out.raw[:ulen]


# ---- DcwZipTable reader (searchability verification) ----
def dzt_read(data: bytes):
    """Full scan of a DZT1 file -> list of (internal_key, value)."""
    r = _lib.orc_dzt_open(data, len(data))
    if not r:
        raise ValueError("DZT open failed")
    out = []

    @_DZTCB
    def cb(_arg, k, klen, v, vlen):
        out.append((C.string_at(k, klen), C.string_at(v, vlen)))
        return 0

    rc = _lib.orc_dzt_iterate(r, cb, None)
    _lib.orc_dzt_close(r)
    if rc != 0:
        raise ValueError("DZT iterate failed")
    return out


def dzt_get(data: bytes, user_key: bytes):
    """Point lookup -> (value, tag) or None (the searchable property)."""
    r = _lib.orc_dzt_open(data, len(data))
    if not r:
        raise ValueError("DZT open failed")
    buf = Buf()
    tag = C.c_uint64()
    rc = _lib.orc_dzt_get(r, user_key, len(user_key), C.byref(buf),
                          C.byref(tag))
    out = None
    if rc == 0:
        out = (C.string_at(buf.data, buf.size), tag.value)
        _lib.orc_buf_free(C.byref(buf))
    _lib.orc_dzt_close(r)
    if rc < 0:
        raise ValueError("DZT corruption")
    return out


# ---- table builder (for golden tests) ----
class TableOpts(C.Structure):
    _fields_ = [
        ("block_size", C.c_uint32),
        ("block_restart_interval", C.c_uint32),
        ("index_block_restart_interval", C.c_uint32),
        ("format_version", C.c_uint32),
        ("checksum_type", C.c_uint32),
        ("compression", C.c_uint32),
        ("block_size_deviation", C.c_uint64),
        ("db_id", C.c_char_p),
        ("db_session_id", C.c_char_p),
        ("db_host_id", C.c_char_p),
        ("cf_name", C.c_char_p),
        ("cf_id", C.c_uint32),
        ("orig_file_number", C.c_uint64),
        ("creation_time", C.c_uint64),
        ("file_creation_time", C.c_uint64),
        ("oldest_key_time", C.c_uint64),
        ("level_at_creation", C.c_int32),
    ]


class Buf(C.Structure):
    _fields_ = [("data", C.POINTER(C.c_uint8)), ("size", C.c_size_t), ("cap", C.c_size_t)]


_lib.orc_table_opts_default.argtypes = [C.POINTER(TableOpts)]
_lib.orc_table_builder_new.restype = C.c_void_p
_lib.orc_table_builder_new.argtypes = [C.POINTER(TableOpts)]
_lib.orc_table_builder_add.argtypes = [C.c_void_p, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
_lib.orc_table_builder_file_size.restype = C.c_uint64
_lib.orc_table_builder_file_size.argtypes = [C.c_void_p]
_lib.orc_table_builder_finish.restype = C.c_int
_lib.orc_table_builder_finish.argtypes = [C.c_void_p, C.POINTER(Buf)]
_lib.orc_table_builder_delete.argtypes = [C.c_void_p]
_lib.orc_table_builder_add_tombstone.argtypes = [
    C.c_void_p, C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t, C.c_uint64]
_lib.orc_snap_dict_table.argtypes = [C.c_char_p, C.c_uint32,
                                     C.POINTER(C.c_uint32)]
_lib.orc_snappy_compress_dict.restype = C.c_size_t
_lib.orc_snappy_compress_dict.argtypes = [C.c_char_p, C.c_uint32,
                                          C.POINTER(C.c_uint32), C.c_char_p,
                                          C.c_size_t, C.c_void_p]
_lib.orc_snappy_uncompress_dict.restype = C.c_size_t
_lib.orc_snappy_uncompress_dict.argtypes = [C.c_char_p, C.c_size_t, C.c_char_p,
                                            C.c_size_t, C.c_void_p, C.c_size_t]
_DZTCB = C.CFUNCTYPE(C.c_int, C.c_void_p, C.POINTER(C.c_uint8), C.c_uint32,
                     C.POINTER(C.c_uint8), C.c_uint32)
_lib.orc_dzt_open.restype = C.c_void_p
_lib.orc_dzt_open.argtypes = [C.c_char_p, C.c_size_t]
_lib.orc_dzt_close.argtypes = [C.c_void_p]
_lib.orc_dzt_num_entries.restype = C.c_uint64
_lib.orc_dzt_num_entries.argtypes = [C.c_void_p]
_lib.orc_dzt_iterate.restype = C.c_int
_lib.orc_dzt_iterate.argtypes = [C.c_void_p, _DZTCB, C.c_void_p]
_lib.orc_dzt_get.restype = C.c_int
_lib.orc_dzt_get.argtypes = [C.c_void_p, C.c_char_p, C.c_uint32,
                             C.POINTER(Buf), C.POINTER(C.c_uint64)]
_lib.orc_buf_free.argtypes = [C.POINTER(Buf)]


def default_table_opts(**kw) -> TableOpts:
    t = TableOpts()
    _lib.orc_table_opts_default(C.byref(t))
    # keep the strings alive on the struct
    t._keep = []
    for k, v in kw.items():
        if isinstance(v, str):
            v = v.encode()
        if isinstance(v, bytes):
            t._keep.append(v)
        setattr(t, k, v)
    return t


def build_sst(entries, opts: TableOpts = None, tombstones=()) -> bytes:
    """entries: iterable of (internal_key: bytes, value: bytes), sorted.
    tombstones: iterable of (start_ukey, end_ukey, seq) -> the
    "rocksdb.range_del" meta block."""
    if opts is None:
        opts = default_table_opts()
    b = _lib.orc_table_builder_new(C.byref(opts))
    for k, v in entries:
        _lib.orc_table_builder_add(b, k, len(k), v, len(v))
    for st, en, seq in tombstones:
        _lib.orc_table_builder_add_tombstone(b, st, len(st), en, len(en), seq)
    out = Buf()
    _lib.orc_table_builder_finish(b, C.byref(out))
    data = C.string_at(out.data, out.size)
    _lib.orc_buf_free(C.byref(out))
    _lib.orc_table_builder_delete(b)
    return data


# ---- table reader ----
_lib.orc_table_open.restype = C.c_void_p
_lib.orc_table_open.argtypes = [C.c_char_p, C.c_size_t, C.c_char_p, C.c_size_t]
_lib.orc_table_close.argtypes = [C.c_void_p]
_KVCB = C.CFUNCTYPE(C.c_int, C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t,
                    C.POINTER(C.c_uint8), C.c_size_t)
_lib.orc_table_iterate.restype = C.c_int
_lib.orc_table_iterate.argtypes = [C.c_void_p, _KVCB, C.c_void_p]


def read_sst(data: bytes):
    """Returns list of (internal_key, value)."""
    err = C.create_string_buffer(160)
    r = _lib.orc_table_open(data, len(data), err, 160)
    if not r:
        raise ValueError("open failed: %s" % err.value.decode())
    out = []

    @_KVCB
    def cb(_arg, k, klen, v, vlen):
        out.append((C.string_at(k, klen), C.string_at(v, vlen)))
        return 0

    rc = _lib.orc_table_iterate(r, cb, None)
    _lib.orc_table_close(r)
    if rc != 0:
        raise ValueError("iterate failed")
    return out


_TOMBCB = C.CFUNCTYPE(C.c_int, C.c_void_p, C.POINTER(C.c_uint8), C.c_size_t,
                      C.POINTER(C.c_uint8), C.c_size_t, C.c_uint64)
_lib.orc_table_tombstones.restype = C.c_int64
_lib.orc_table_tombstones.argtypes = [C.c_void_p, _TOMBCB, C.c_void_p]


def read_tombstones(data: bytes):
    """[(start_ukey, end_ukey, seq)] from an SST's range-del meta block."""
    err = C.create_string_buffer(160)
    r = _lib.orc_table_open(data, len(data), err, 160)
    if not r:
        raise ValueError("open failed: %s" % err.value.decode())
    out = []

    @_TOMBCB
    def cb(_arg, s_, sl, e_, el, seq):
        out.append((C.string_at(s_, sl), C.string_at(e_, el), seq))
        return 0

    n = _lib.orc_table_tombstones(r, cb, None)
    _lib.orc_table_close(r)
    if n < 0:
        raise ValueError("tombstone block corrupt")
    return out


def make_ikey(user_key: bytes, seq: int, vtype: int) -> bytes:
    return user_key + ((seq << 8) | vtype).to_bytes(8, "little")


def make_job(runs, output_dir, **kw) -> JobDesc:
    """runs: list of list-of-paths. kw overrides JobDesc fields."""
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
    def _ranges(triples):
        arr = (Grandparent * len(triples))()
        for i, (sm, lg, fsz) in enumerate(triples):
            smb = (C.c_uint8 * len(sm)).from_buffer_copy(sm)
            lgb = (C.c_uint8 * len(lg)).from_buffer_copy(lg)
            d._keep += [smb, lgb]
            arr[i].smallest_ukey = smb
            arr[i].smallest_len = len(sm)
            arr[i].largest_ukey = lgb
            arr[i].largest_len = len(lg)
            arr[i].file_size = fsz
        d._keep.append(arr)
        return arr

    # levels_below: list (one per level below output) of [(sm, lg, size)]
    lvls = kw.pop("levels_below", None)
    d.levels_below_valid = kw.pop("levels_below_valid",
                                  1 if lvls is not None else 1)
    if lvls is None:
        lvls = []
    if lvls or d.levels_below_valid:
        larr = (LevelFiles * max(len(lvls), 1))()
        for i, files in enumerate(lvls):
            rarr = _ranges(files)
            larr[i].files = rarr
            larr[i].num_files = len(files)
        d._keep.append(larr)
        d.levels_below = larr
        d.num_levels_below = len(lvls)
    d.block_size = kw.pop("block_size", 4096)
    d.block_restart_interval = kw.pop("block_restart_interval", 16)
    d.format_version = 5
    d.checksum_type = kw.pop("checksum_type", 4)  # kXXH3
    d.index_block_restart_interval = 1
    d.level_compaction_dynamic_file_size = kw.pop(
        "level_compaction_dynamic_file_size", 1)
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
    for k, v in kw.items():
        setattr(d, k, v)
    return d


def execute(desc: JobDesc):
    """Run the oracle worker; returns dict result. Raises on failure."""
    res = JobResult()
    rc = _lib.orc_execute(C.byref(desc), C.byref(res))
    if rc != 0:
        err = res.error.decode(errors="replace")
        _lib.orc_free_result(C.byref(res))
        raise RuntimeError("oracle execute failed (%d): %s" % (rc, err))
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
               work_time_usec=res.work_time_usec)
    _lib.orc_free_result(C.byref(res))
    return out
