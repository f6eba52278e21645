// dcw_host.cpp — implementation of dcw_host.h (see header for citations).
#include "dcw_host.h"

#include <algorithm>
#include <cstdio>
#include <cstring>

namespace dcw {

Crc32cTables g_crc;
namespace {
struct CrcInit {
  CrcInit() { crc32c_build_tables(&g_crc); }
} g_crc_init;

void put_fixed32(std::string& s, uint32_t v) { s.append((const char*)&v, 4); }
[[maybe_unused]] void put_fixed64(std::string& s, uint64_t v) { s.append((const char*)&v, 8); }
[[maybe_unused]] void put_varint32(std::string& s, uint32_t v) {
  uint8_t t[5];
  s.append((const char*)t, varint32_put(t, v));
}
void put_varint64(std::string& s, uint64_t v) {
  uint8_t t[10];
  s.append((const char*)t, varint64_put(t, v));
}
void put_varsigned64(std::string& s, int64_t v) { // util/coding.h zigzag
  put_varint64(s, ((uint64_t)v << 1) ^ (uint64_t)(v >> 63));
}
} // namespace

// ---------------- BlockBuilder ----------------
void BlockBuilder::AddWithLastKey(const uint8_t* key, size_t klen,
                                  const uint8_t* val, size_t vlen,
                                  const uint8_t* last_key, size_t last_len) {
  size_t buffer_size = buf_.size();
  if (buffer_size < last_len) last_len = buffer_size; // empty at block start
  size_t shared = 0;
  if (counter_ >= interval_) {
    restarts_.push_back((uint32_t)buffer_size);
    counter_ = 0;
  } else {
    size_t n = std::min(klen, last_len);
    while (shared < n && key[shared] == last_key[shared]) shared++;
  }
  size_t non_shared = klen - shared;
  uint8_t hdr[15];
  int hn = varint32_put(hdr, (uint32_t)shared);
  hn += varint32_put(hdr + hn, (uint32_t)non_shared);
  if (!uvde_) hn += varint32_put(hdr + hn, (uint32_t)vlen);
  buf_.append((const char*)hdr, hn);
  buf_.append((const char*)key + shared, non_shared);
  buf_.append((const char*)val, vlen);
  counter_++;
}

void BlockBuilder::Add(const std::string& key, const std::string& value,
                       const std::string* delta_value) {
  size_t buffer_size = buf_.size();
  size_t last_len = std::min(buffer_size, last_key_.size());
  size_t shared = 0;
  if (counter_ >= interval_) {
    restarts_.push_back((uint32_t)buffer_size);
    counter_ = 0;
  } else {
    size_t n = std::min(key.size(), last_len);
    while (shared < n && key[shared] == last_key_[shared]) shared++;
  }
  size_t non_shared = key.size() - shared;
  uint8_t hdr[15];
  int hn = varint32_put(hdr, (uint32_t)shared);
  hn += varint32_put(hdr + hn, (uint32_t)non_shared);
  const std::string* v = &value;
  if (uvde_) {
    if (shared != 0) v = delta_value;
  } else {
    hn += varint32_put(hdr + hn, (uint32_t)value.size());
  }
  buf_.append((const char*)hdr, hn);
  buf_.append(key.data() + shared, non_shared);
  buf_.append(*v);
  counter_++;
  last_key_ = key;
}

size_t BlockBuilder::EstimateSizeAfterKV(size_t klen, size_t vlen) const {
  size_t est = CurrentSizeEstimate() + klen + vlen;
  if (counter_ >= interval_) est += 4;
  est += 4; // sizeof(int32) for shared varint (block_builder.cc:118)
  est += (size_t)varint_len(klen);
  est += (size_t)varint_len(vlen);
  return est;
}

std::string BlockBuilder::Finish() {
  std::string out = buf_;
  for (uint32_t r : restarts_) put_fixed32(out, r);
  put_fixed32(out, (uint32_t)restarts_.size()); // kDataBlockBinarySearch
  return out;
}

// ---------------- input SST parsing ----------------
extern "C" size_t ZSTD_decompress(void*, size_t, const void*, size_t);
extern "C" unsigned ZSTD_isError(size_t);

ParsedSst parse_sst(const uint8_t* data, size_t size) {
  ParsedSst r;
  if (size < 53) {
    r.error = "file too small";
    return r;
  }
  const uint8_t* f = data + size - 53;
  if (load64(f + 45) != kTableMagic) {
    r.error = "bad magic";
    return r;
  }
  r.checksum_type = f[0];
  const uint8_t* p = f + 1;
  const uint8_t* lim = f + 41;
  uint64_t mi_off, mi_sz, idx_off, idx_sz;
  int a;
  if ((a = varint64_get(p, lim, &mi_off)) < 0) { r.error = "footer"; return r; }
  p += a;
  if ((a = varint64_get(p, lim, &mi_sz)) < 0) { r.error = "footer"; return r; }
  p += a;
  if ((a = varint64_get(p, lim, &idx_off)) < 0) { r.error = "footer"; return r; }
  p += a;
  if ((a = varint64_get(p, lim, &idx_sz)) < 0) { r.error = "footer"; return r; }
  if (idx_off + idx_sz + kTrailerSize > size) {
    r.error = "index handle out of range";
    return r;
  }
  // read + verify + maybe decompress index block
  const uint8_t* ib = data + idx_off;
  uint8_t type = ib[idx_sz];
  uint32_t stored = load32(ib + idx_sz + 1);
  if (r.checksum_type != 0 &&
      stored != block_checksum(r.checksum_type, &g_crc, ib, idx_sz, type)) {
    r.error = "index checksum mismatch";
    return r;
  }
  std::string dec;
  if (type == 1 /*snappy*/) {
    size_t ul = snappy_uncompressed_len(ib, idx_sz);
    if (ul == (size_t)-1) { r.error = "index snappy"; return r; }
    dec.resize(ul);
    if (snappy_uncompress(ib, idx_sz, (uint8_t*)dec.data(), ul) != ul) {
      r.error = "index snappy corrupt";
      return r;
    }
    ib = (const uint8_t*)dec.data();
    idx_sz = ul;
  } else if (type == 7 /*zstd: host decode, same framing as data blocks*/) {
    uint32_t ul;
    int hn = varint32_get(ib, ib + (idx_sz < 5 ? idx_sz : 5), &ul);
    if (hn < 0) { r.error = "index zstd preamble"; return r; }
    dec.resize(ul);
    size_t got = ZSTD_decompress((void*)dec.data(), ul, ib + hn, idx_sz - hn);
    if (ZSTD_isError(got) || got != ul) {
      r.error = "index zstd corrupt";
      return r;
    }
    ib = (const uint8_t*)dec.data();
    idx_sz = ul;
  } else if (type != 0) {
    r.error = "index compression unsupported";
    return r;
  }
  if (idx_sz < 8) { r.error = "index too small"; return r; }
  uint32_t nrestarts = load32(ib + idx_sz - 4) & 0x7fffffff;
  size_t data_end = idx_sz - 4 - 4 * (size_t)nrestarts;
  p = ib;
  const uint8_t* dl = ib + data_end;
  uint64_t prev_off = 0, prev_sz = 0;
  bool have_prev = false;
  while (p < dl) {
    uint32_t shared, non_shared;
    if ((a = varint32_get(p, dl, &shared)) < 0) break;
    p += a;
    if ((a = varint32_get(p, dl, &non_shared)) < 0) break;
    p += a;
    p += non_shared;
    uint64_t off, sz;
    if (shared == 0) {
      if ((a = varint64_get(p, dl, &off)) < 0) break;
      p += a;
      if ((a = varint64_get(p, dl, &sz)) < 0) break;
      p += a;
    } else { // delta-encoded IndexValue (format.cc IndexValue::EncodeTo)
      uint64_t zz;
      if ((a = varint64_get(p, dl, &zz)) < 0) break;
      p += a;
      if (!have_prev) break;
      int64_t d = (int64_t)(zz >> 1) ^ -(int64_t)(zz & 1);
      off = prev_off + prev_sz + kTrailerSize;
      sz = (uint64_t)((int64_t)prev_sz + d);
    }
    r.data_blocks.push_back({off, sz});
    prev_off = off;
    prev_sz = sz;
    have_prev = true;
  }
  // metaindex -> "rocksdb.range_del" meta block -> tombstones
  // (block_based_table_builder.cc:1735-1743; the block is uncompressed)
  if (mi_off + mi_sz + kTrailerSize <= size && mi_sz >= 8) {
    const uint8_t* mb = data + mi_off;
    if (r.checksum_type == 0 ||
        load32(mb + mi_sz + 1) ==
            block_checksum(r.checksum_type, &g_crc, mb, mi_sz, mb[mi_sz])) {
      uint32_t mnr = load32(mb + mi_sz - 4) & 0x7fffffff;
      size_t mend = mi_sz - 4 - 4 * (size_t)mnr;
      const uint8_t* mp = mb;
      const uint8_t* ml = mb + mend;
      std::string mkey;
      uint64_t rd_off = 0, rd_sz = 0;
      bool rd_found = false;
      while (mp < ml) {
        uint32_t sh, ns, vl;
        if ((a = varint32_get(mp, ml, &sh)) < 0) break;
        mp += a;
        if ((a = varint32_get(mp, ml, &ns)) < 0) break;
        mp += a;
        if ((a = varint32_get(mp, ml, &vl)) < 0) break;
        mp += a;
        if (sh > mkey.size()) break;
        mkey.resize(sh);
        mkey.append((const char*)mp, ns);
        mp += ns;
        if (mkey == "rocksdb.range_del") {
          const uint8_t* vp = mp;
          int b1 = varint64_get(vp, mp + vl, &rd_off);
          if (b1 > 0 && varint64_get(vp + b1, mp + vl, &rd_sz) > 0)
            rd_found = true;
        }
        mp += vl;
      }
      if (rd_found && rd_off + rd_sz + kTrailerSize <= size && rd_sz >= 8) {
        const uint8_t* rb = data + rd_off;
        if (r.checksum_type != 0 &&
            load32(rb + rd_sz + 1) !=
                block_checksum(r.checksum_type, &g_crc, rb, rd_sz, rb[rd_sz])) {
          r.error = "range_del block checksum mismatch";
          return r;
        }
        uint32_t rnr = load32(rb + rd_sz - 4) & 0x7fffffff;
        size_t rend = rd_sz - 4 - 4 * (size_t)rnr;
        const uint8_t* rp = rb;
        const uint8_t* rl = rb + rend;
        std::string rkey;
        while (rp < rl) {
          uint32_t sh, ns, vl;
          if ((a = varint32_get(rp, rl, &sh)) < 0) break;
          rp += a;
          if ((a = varint32_get(rp, rl, &ns)) < 0) break;
          rp += a;
          if ((a = varint32_get(rp, rl, &vl)) < 0) break;
          rp += a;
          if (sh > rkey.size() || rp + ns + vl > rl) break;
          rkey.resize(sh);
          rkey.append((const char*)rp, ns);
          rp += ns;
          if (rkey.size() < 9) break;
          uint64_t tag = load64((const uint8_t*)rkey.data() + rkey.size() - 8);
          SstTombstone t;
          t.start.assign(rkey, 0, rkey.size() - 8);
          t.end.assign((const char*)rp, vl);
          t.seq = tag >> 8;
          r.tombstones.push_back(std::move(t));
          rp += vl;
        }
      }
    }
  }
  r.ok = true;
  return r;
}

// ---------------- output tail ----------------
SstIndexEntry append_block(std::string& out, const TableOpts& o,
                           const uint8_t* data, size_t n, bool try_compress) {
  uint8_t type = 0;
  std::string comp;
  const uint8_t* body = data;
  size_t bn = n;
  if (try_compress && o.compression == 1) {
    comp.resize(snappy_max_compressed(n));
    std::vector<uint32_t> tab(1u << kSnapHashBits, 0xffffffffu);
    size_t cn = snappy_compress_block(data, n, (uint8_t*)comp.data(), tab.data());
    if (cn <= ((uint64_t)896 * n) >> 10) { // GoodCompressionRatio default
      body = (const uint8_t*)comp.data();
      bn = cn;
      type = 1;
    }
  }
  SstIndexEntry h{out.size(), bn};
  out.append((const char*)body, bn);
  uint8_t tr[5];
  tr[0] = type;
  store32(tr + 1, block_checksum(o.checksum_type, &g_crc, body, bn, type));
  out.append((const char*)tr, 5);
  return h;
}

std::string build_tail(const TableOpts& o, const TailStats& st,
                       const std::vector<SstIndexEntry>& handles,
                       const std::vector<std::string>& separators,
                       bool sep_is_user_key, uint64_t tail_start,
                       const std::string& filter_content,
                       uint64_t num_filter_entries) {
  std::string out;
  // full filter block FIRST in the tail (uncompressed;
  // block_based_table_builder.cc:1488-1526, layout comment :1950-1955)
  SstIndexEntry filter_handle{0, 0};
  if (!filter_content.empty()) {
    filter_handle = append_block(out, o, (const uint8_t*)filter_content.data(),
                                 filter_content.size(), false);
    filter_handle.off += tail_start;
  }
  // index block (restart_interval=1, value delta encoding for fmt>=4)
  BlockBuilder ib(o.index_block_restart_interval, true);
  for (size_t i = 0; i < handles.size(); i++) {
    std::string full;
    put_varint64(full, handles[i].off);
    put_varint64(full, handles[i].size);
    std::string delta;
    if (i > 0) put_varsigned64(delta, (int64_t)handles[i].size - (int64_t)handles[i - 1].size);
    std::string key = separators[i];
    if (sep_is_user_key) key.resize(key.size() - 8);
    ib.Add(key, full, &delta);
  }
  std::string index_contents = ib.Finish();
  uint64_t index_size_prop = index_contents.size() + kTrailerSize;
  SstIndexEntry index_handle = append_block(out, o, (const uint8_t*)index_contents.data(),
                                            index_contents.size(),
                                            true /*enable_index_compression*/);
  index_handle.off += tail_start;

  // properties block (sorted names; restart_interval=INT32_MAX)
  std::map<std::string, std::string> props;
  auto addi = [&](const char* n, uint64_t v) {
    std::string s;
    put_varint64(s, v);
    props[n] = s;
  };
  auto adds = [&](const char* n, const std::string& v) { props[n] = v; };
  addi("rocksdb.original.file.number", o.orig_file_number);
  addi("rocksdb.raw.key.size", st.raw_key_size);
  addi("rocksdb.raw.value.size", st.raw_value_size);
  addi("rocksdb.data.size", st.data_size);
  addi("rocksdb.index.size", index_size_prop);
  addi("rocksdb.index.key.is.user.key", sep_is_user_key ? 1 : 0);
  addi("rocksdb.index.value.is.delta.encoded", 1);
  addi("rocksdb.num.entries", st.num_entries);
  addi("rocksdb.num.filter_entries", num_filter_entries);
  addi("rocksdb.deleted.keys", st.num_deletions);
  addi("rocksdb.merge.operands", st.num_merge_operands);
  addi("rocksdb.num.range-deletions", st.num_range_deletions);
  addi("rocksdb.num.data.blocks", st.num_data_blocks);
  addi("rocksdb.filter.size", filter_content.size());
  addi("rocksdb.format.version", o.format_version);
  addi("rocksdb.fixed.key.length", 0);
  addi("rocksdb.column.family.id", o.cf_id);
  addi("rocksdb.creation.time", o.creation_time);
  addi("rocksdb.oldest.key.time", o.oldest_key_time);
  if (o.file_creation_time > 0)
    addi("rocksdb.file.creation.time", o.file_creation_time);
  addi("rocksdb.tail.start.offset", tail_start);
  if (!o.db_id.empty()) adds("rocksdb.creating.db.identity", o.db_id);
  if (!o.db_session_id.empty())
    adds("rocksdb.creating.session.identity", o.db_session_id);
  if (!o.db_host_id.empty()) adds("rocksdb.creating.host.identity", o.db_host_id);
  adds("rocksdb.comparator", "leveldb.BytewiseComparator");
  if (o.bloom_millibits_per_key)
    adds("rocksdb.filter.policy", "bloomfilter");
  adds("rocksdb.merge.operator", "nullptr");
  adds("rocksdb.prefix.extractor.name", "nullptr");
  adds("rocksdb.property.collectors", "[]");
  if (!o.cf_name.empty()) adds("rocksdb.column.family.name", o.cf_name);
  adds("rocksdb.compression", o.compression == 1 ? "Snappy" : "NoCompression");
  adds("rocksdb.compression_options",
       "window_bits=-14; level=32767; strategy=0; max_dict_bytes=0; "
       "zstd_max_train_bytes=0; enabled=0; max_dict_buffer_bytes=0; "
       "use_zstd_dict_trainer=1; ");
  {
    std::string v;
    put_fixed32(v, 0); // kBinarySearch
    adds("rocksdb.block.based.table.index.type", v);
    adds("rocksdb.block.based.table.prefix.filtering", "0");
    adds("rocksdb.block.based.table.whole.key.filtering", "1");
  }
  BlockBuilder pb(0x7fffffff, false);
  for (auto& kv : props) pb.Add(kv.first, kv.second);
  std::string pcontents = pb.Finish();
  SstIndexEntry props_handle = append_block(out, o, (const uint8_t*)pcontents.data(),
                                            pcontents.size(), false);
  props_handle.off += tail_start;

  // metaindex (restart_interval=1; keys sorted: "fullfilter." < "rocksdb.")
  BlockBuilder mi(1, false);
  if (!filter_content.empty()) {
    std::string fv;
    put_varint64(fv, filter_handle.off);
    put_varint64(fv, filter_handle.size);
    mi.Add("fullfilter.rocksdb.BuiltinBloomFilter", fv);
  }
  std::string hv;
  put_varint64(hv, props_handle.off);
  put_varint64(hv, props_handle.size);
  mi.Add("rocksdb.properties", hv);
  std::string mcontents = mi.Finish();
  SstIndexEntry mi_handle = append_block(out, o, (const uint8_t*)mcontents.data(),
                                         mcontents.size(), false);
  mi_handle.off += tail_start;

  // footer
  uint8_t footer[53];
  memset(footer, 0, sizeof(footer));
  footer[0] = (uint8_t)o.checksum_type;
  int fn = 1;
  fn += varint64_put(footer + fn, mi_handle.off);
  fn += varint64_put(footer + fn, mi_handle.size);
  fn += varint64_put(footer + fn, index_handle.off);
  fn += varint64_put(footer + fn, index_handle.size);
  store32(footer + 41, o.format_version);
  store64(footer + 45, kTableMagic);
  out.append((const char*)footer, 53);
  return out;
}

// ---------------- separator shortening ----------------
void shorten_separator(std::string& start, const uint8_t* limit, size_t limit_len) {
  size_t ustart_len = start.size() - 8;
  size_t ulimit_len = limit_len - 8;
  const uint8_t* us = (const uint8_t*)start.data();
  size_t min_len = std::min(ustart_len, ulimit_len);
  size_t di = 0;
  while (di < min_len && us[di] == limit[di]) di++;
  std::string tmp;
  if (di >= min_len) return; // prefix: do not shorten
  uint8_t sb = us[di], lb = limit[di];
  if (sb >= lb) return;
  if (di < ulimit_len - 1 || sb + 1 < lb) {
    tmp.assign((const char*)us, di + 1);
    tmp[di]++;
  } else {
    di++;
    while (di < ustart_len) {
      if (us[di] < 0xff) {
        tmp.assign((const char*)us, di + 1);
        tmp[di]++;
        break;
      }
      di++;
    }
    if (tmp.empty()) return;
  }
  // accept iff shorter-or-equal and logically larger (index_builder.cc:85-90)
  size_t n = std::min(ustart_len, tmp.size());
  int c = memcmp(us, tmp.data(), n);
  bool lt = c < 0 || (c == 0 && ustart_len < tmp.size());
  if (!(tmp.size() <= ustart_len && lt)) return;
  uint64_t tag = (kMaxSeq << 8) | kTypeWideColumnEntity;
  tmp.append((const char*)&tag, 8);
  start.swap(tmp);
}

// ---------------- plan FSM: pure block planning ----------------
std::vector<PlannedBlock> plan_blocks(const PlanIn& in, size_t from,
                                      const TableOpts& o, uint64_t min_bytes,
                                      std::vector<uint32_t>* eoff_out) {
  std::vector<PlannedBlock> out;
  if (eoff_out) eoff_out->clear();
  uint64_t dev_limit = ((o.block_size * (100 - o.block_size_deviation)) + 99) / 100;
  uint64_t produced = 0;
  size_t i = from;
  while (i < in.n && produced < min_bytes) {
    // one block
    uint32_t first = (uint32_t)i;
    uint64_t bytes = 0; // entry bytes in buffer
    uint32_t nrestarts = 1, counter = 0;
    while (i < in.n) {
      size_t klen = in.klen[i], vlen = in.vlen[i];
      uint64_t curr = 8 + bytes + 4 * (nrestarts - 1);
      if (bytes > 0) { // flush policy (flush_block_policy.cc:37-52)
        if (curr >= o.block_size) break;
        uint64_t after = curr + klen + vlen + 4 + varint_len(klen) +
                         varint_len(vlen) + (counter >= o.block_restart_interval ? 4 : 0);
        if (after > o.block_size && curr > dev_limit && dev_limit != 0) break;
      }
      size_t shared = in.shared[i];
      if (counter >= o.block_restart_interval) {
        nrestarts++;
        counter = 0;
        shared = 0;
      } else if (bytes == 0) {
        shared = 0; // first entry of the block: last_key truncated to empty
      }
      size_t non_shared = klen - shared;
      if (eoff_out) eoff_out->push_back((uint32_t)bytes);
      bytes += varint_len(shared) + varint_len(non_shared) + varint_len(vlen) +
               non_shared + vlen;
      counter++;
      i++;
    }
    PlannedBlock b;
    b.first = first;
    b.count = (uint32_t)(i - first);
    b.num_restarts = nrestarts;
    b.unc_size = (uint32_t)(bytes + 4 * nrestarts + 4);
    out.push_back(b);
    produced += b.unc_size + kTrailerSize;
  }
  return out;
}

// ---------------- TableWriter (sstgen + partial blocks) ----------------
void TableWriter::FlushData() {
  if (data_block_.empty()) return;
  std::string contents = data_block_.Finish();
  data_block_.Reset();
  pending_ = append_block(file_, o_, (const uint8_t*)contents.data(),
                          contents.size(), true);
  st_.data_size = file_.size();
  st_.num_data_blocks++;
  has_pending_ = true;
}
void TableWriter::AddIndexEntry(const uint8_t* next_key, size_t next_len) {
  std::string sep = last_key_;
  if (next_key != nullptr) {
    shorten_separator(sep, next_key, next_len);
    if (!sep_key_plus_seq_) {
      size_t su = sep.size() - 8, nu = next_len - 8;
      if (su == nu && memcmp(sep.data(), next_key, su) == 0) sep_key_plus_seq_ = true;
    }
  }
  handles_.push_back(pending_);
  separators_.push_back(sep);
  has_pending_ = false;
}
void TableWriter::Add(const uint8_t* ikey, size_t klen, const uint8_t* val,
                      size_t vlen) {
  bool should_flush = false;
  if (!data_block_.empty()) {
    size_t curr = data_block_.CurrentSizeEstimate();
    uint64_t dev_limit = ((o_.block_size * (100 - o_.block_size_deviation)) + 99) / 100;
    if (curr >= o_.block_size)
      should_flush = true;
    else if (dev_limit != 0) {
      size_t after = data_block_.EstimateSizeAfterKV(klen, vlen);
      should_flush = after > o_.block_size && curr > dev_limit;
    }
  }
  if (should_flush) {
    FlushData();
    AddIndexEntry(ikey, klen);
  }
  data_block_.AddWithLastKey(ikey, klen, val, vlen,
                             (const uint8_t*)last_key_.data(), last_key_.size());
  last_key_.assign((const char*)ikey, klen);
  st_.num_entries++;
  st_.raw_key_size += klen;
  st_.raw_value_size += vlen;
  uint8_t type = ikey[klen - 8];
  if (type == kTypeDeletion || type == kTypeSingleDeletion) st_.num_deletions++;
  else if (type == kTypeMerge) st_.num_merge_operands++;
}
std::string TableWriter::Finish() {
  FlushData();
  if (has_pending_) AddIndexEntry(nullptr, 0);
  uint64_t tail_start = file_.size();
  file_ += build_tail(o_, st_, handles_, separators_, !sep_key_plus_seq_, tail_start);
  return std::move(file_);
}

// ---------------- synthetic input generator ----------------
static uint64_t splitmix64(uint64_t& s) {
  uint64_t z = (s += 0x9E3779B97f4A7C15ULL);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

int gen_sst_file(const char* path, uint64_t seed, uint64_t num_entries,
                 uint32_t key_len, uint32_t value_len, uint64_t seq_base,
                 const TableOpts& opts) {
  // keys: key_len uniform random bytes, deduplicated, sorted (db_bench
  // fillrandom style); values: value_len bytes, first half pseudorandom from
  // the key, second half repeats the first (≈50% snappy-compressible).
  std::vector<std::string> keys;
  keys.reserve(num_entries);
  uint64_t s = seed;
  for (uint64_t i = 0; i < num_entries; i++) {
    std::string k(key_len, 0);
    for (uint32_t j = 0; j < key_len; j += 8) {
      uint64_t r = splitmix64(s);
      for (uint32_t b = 0; b < 8 && j + b < key_len; b++)
        k[j + b] = (char)(r >> (8 * b));
    }
    keys.push_back(std::move(k));
  }
  std::sort(keys.begin(), keys.end());
  keys.erase(std::unique(keys.begin(), keys.end()), keys.end());

  TableWriter w(opts);
  std::string val(value_len, 0);
  uint64_t seq = seq_base;
  for (auto& k : keys) {
    uint64_t vs = load64((const uint8_t*)k.data()) ^ 0x6a09e667f3bcc909ULL;
    uint32_t half = value_len / 2;
    for (uint32_t j = 0; j < half; j += 8) {
      uint64_t r = splitmix64(vs);
      for (uint32_t b = 0; b < 8 && j + b < half; b++) val[j + b] = (char)(r >> (8 * b));
    }
    for (uint32_t j = half; j < value_len; j++) val[j] = val[j - half];
    uint8_t ikey[64];
    memcpy(ikey, k.data(), key_len);
    uint64_t tag = (seq << 8) | kTypeValue;
    store64(ikey + key_len, tag);
    seq++;
    w.Add(ikey, key_len + 8, (const uint8_t*)val.data(), value_len);
  }
  std::string file = w.Finish();
  FILE* f = fopen(path, "wb");
  if (!f) return -1;
  fwrite(file.data(), 1, file.size(), f);
  fclose(f);
  return 0;
}

} // namespace dcw
