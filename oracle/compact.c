/* compact.c — oracle compaction worker: k-way merge + CompactionIterator
 * semantics + output file cutting.  TEST INFRASTRUCTURE (see oracle.h).
 *
 * Restated from:
 *  - merge order:      table/compaction_merging_iterator.cc:239-348 (min-heap;
 *                      ties resolved to the lower child index = run order)
 *  - visibility FSM:   db/compaction/compaction_iterator.cc:475-1082
 *                      (NextFromInput), :156-231 (SeekToFirst/Next,
 *                      has_outputted_key_), :1274-1341 (PrepareOutput
 *                      seq-zeroing), :1343-1396 (findEarliestVisibleSnapshot)
 *  - output cutting:   db/compaction/compaction_outputs.cc:121-420
 *                      (UpdateGrandparentBoundaryInfo / ShouldStopBefore /
 *                      AddToOutput), sstableKeyCompare (compaction.cc:26-43)
 *  - file meta:        db/compaction/compaction_job.cc:2230-2340
 *                      (OpenCompactionOutputFile: file numbers, times)
 *
 * Envelope (round 1): value types {Put, Delete, SingleDelete}; no merge
 * operator, no range deletions, no compaction filter, no user timestamps,
 * no snapshot checker.  Jobs outside the envelope return an error, like the
 * worker refusing a job (DB side then runs local,
 * compaction_job.cc:648-655).
 */
#include "oracle.h"

#include <inttypes.h>
#include <stdio.h>
#include <stdlib.h>
#include <string.h>
#include <sys/time.h>

#define KMAXSEQ DCW_MAX_SEQUENCE
#define MAX_IKEY 512

/* internal hooks from table.c */
extern uint8_t* orc__read_block_for_iter(orc_table_reader* r, size_t bi,
                                         size_t* out_n, char* err, size_t cap);
extern size_t orc__table_nblocks(const orc_table_reader* r);

static uint64_t now_usec(void) {
  struct timeval tv;
  gettimeofday(&tv, NULL);
  return (uint64_t)tv.tv_sec * 1000000 + (uint64_t)tv.tv_usec;
}

/* ---------- streaming run iterator over a list of SST files ---------- */
typedef struct file_image {
  uint8_t* data;
  size_t size;
} file_image;

typedef struct run_iter {
  const dcw_run* run;
  uint32_t file_idx;
  file_image img;
  orc_table_reader* rd;
  /* block cursor */
  size_t blk_idx;
  uint8_t* blk;
  size_t blk_n, data_end, pos;
  /* current entry */
  uint8_t key[MAX_IKEY];
  size_t klen;
  const uint8_t* val;
  size_t vlen;
  int valid;
  char err[160];
} run_iter;

static int ri_load_file(run_iter* it);

static int ri_next_block(run_iter* it) {
  free(it->blk);
  it->blk = NULL;
  for (;;) {
    if (it->rd == NULL) return 0;
    if (it->blk_idx >= orc__table_nblocks(it->rd)) {
      /* next file in run */
      orc_table_close(it->rd);
      it->rd = NULL;
      free(it->img.data);
      it->img.data = NULL;
      it->file_idx++;
      if (it->file_idx >= it->run->num_files) return 0;
      if (ri_load_file(it) != 0) return -1;
      continue;
    }
    break;
  }
  char err[128];
  it->blk = orc__read_block_for_iter(it->rd, it->blk_idx, &it->blk_n, err, sizeof(err));
  if (!it->blk) {
    snprintf(it->err, sizeof(it->err), "run read: %s", err);
    return -1;
  }
  it->blk_idx++;
  uint32_t footer_u32;
  memcpy(&footer_u32, it->blk + it->blk_n - 4, 4);
  uint32_t nrestarts = footer_u32 & 0x7fffffff;
  it->data_end = it->blk_n - 4 - 4 * (size_t)nrestarts;
  it->pos = 0;
  it->klen = 0;
  return 1;
}

static int ri_load_file(run_iter* it) {
  const char* path = it->run->files[it->file_idx];
  FILE* f = fopen(path, "rb");
  if (!f) {
    snprintf(it->err, sizeof(it->err), "open %s failed", path);
    return -1;
  }
  fseek(f, 0, SEEK_END);
  long sz = ftell(f);
  fseek(f, 0, SEEK_SET);
  it->img.data = (uint8_t*)malloc((size_t)sz);
  it->img.size = (size_t)sz;
  if (fread(it->img.data, 1, (size_t)sz, f) != (size_t)sz) {
    fclose(f);
    snprintf(it->err, sizeof(it->err), "read %s failed", path);
    return -1;
  }
  fclose(f);
  char err[128];
  it->rd = orc_table_open(it->img.data, it->img.size, err, sizeof(err));
  if (!it->rd) {
    snprintf(it->err, sizeof(it->err), "%s: %s", path, err);
    return -1;
  }
  it->blk_idx = 0;
  return 0;
}

static int ri_advance(run_iter* it) { /* returns 1 ok, 0 exhausted, -1 err */
  for (;;) {
    if (it->blk && it->pos < it->data_end) {
      const uint8_t* p = it->blk + it->pos;
      const uint8_t* lim = it->blk + it->data_end;
      uint32_t shared, non_shared, vlen;
      int a = orc_varint32_get(p, lim, &shared);
      if (a < 0) goto corrupt;
      p += a;
      a = orc_varint32_get(p, lim, &non_shared);
      if (a < 0) goto corrupt;
      p += a;
      a = orc_varint32_get(p, lim, &vlen);
      if (a < 0) goto corrupt;
      p += a;
      if (shared + non_shared > MAX_IKEY || p + non_shared + vlen > lim) goto corrupt;
      if (shared > it->klen) goto corrupt;
      memcpy(it->key + shared, p, non_shared);
      it->klen = shared + non_shared;
      p += non_shared;
      it->val = p;
      it->vlen = vlen;
      it->pos = (size_t)(p + vlen - it->blk);
      it->valid = 1;
      return 1;
    }
    int r = ri_next_block(it);
    if (r <= 0) {
      it->valid = 0;
      return r;
    }
  }
corrupt:
  snprintf(it->err, sizeof(it->err), "corrupt data block in run");
  it->valid = 0;
  return -1;
}

static int ri_init(run_iter* it, const dcw_run* run) {
  memset(it, 0, sizeof(*it));
  it->run = run;
  if (run->num_files == 0) {
    it->valid = 0;
    return 0;
  }
  if (ri_load_file(it) != 0) return -1;
  return ri_advance(it) < 0 ? -1 : 0;
}
static void ri_free(run_iter* it) {
  free(it->blk);
  orc_table_close(it->rd);
  free(it->img.data);
}

/* ---------- k-way merge (ties -> lowest run index) ---------- */
typedef struct kmerge {
  run_iter* runs;
  uint32_t k;
  int cur; /* -1 when exhausted */
  char err[160];
} kmerge;

static int km_find_min(kmerge* m) {
  int best = -1;
  for (uint32_t i = 0; i < m->k; i++) {
    if (!m->runs[i].valid) continue;
    if (best < 0 ||
        orc_ikey_compare(m->runs[i].key, m->runs[i].klen, m->runs[best].key,
                         m->runs[best].klen) < 0)
      best = (int)i;
  }
  m->cur = best;
  return best;
}
static int km_init(kmerge* m, const dcw_run* runs, uint32_t k) {
  m->runs = (run_iter*)calloc(k, sizeof(run_iter));
  m->k = k;
  for (uint32_t i = 0; i < k; i++)
    if (ri_init(&m->runs[i], &runs[i]) != 0) {
      snprintf(m->err, sizeof(m->err), "%s", m->runs[i].err);
      return -1;
    }
  km_find_min(m);
  return 0;
}
static void km_free(kmerge* m) {
  for (uint32_t i = 0; i < m->k; i++) ri_free(&m->runs[i]);
  free(m->runs);
}
static int km_valid(const kmerge* m) { return m->cur >= 0; }
static const uint8_t* km_key(const kmerge* m, size_t* klen) {
  *klen = m->runs[m->cur].klen;
  return m->runs[m->cur].key;
}
static const uint8_t* km_val(const kmerge* m, size_t* vlen) {
  *vlen = m->runs[m->cur].vlen;
  return m->runs[m->cur].val;
}
static int km_next(kmerge* m) {
  if (ri_advance(&m->runs[m->cur]) < 0) {
    snprintf(m->err, sizeof(m->err), "%s", m->runs[m->cur].err);
    return -1;
  }
  km_find_min(m);
  return 0;
}

/* ---------- range-deletion aggregator (envelope subset) ----------
 * Restates CompactionRangeDelAggregator::ShouldDelete
 * (db/range_del_aggregator.cc:407-413) + the CompactionIterator drop site
 * (compaction_iterator.cc:1056-1063) for the SUPPORTED ENVELOPE:
 * bottommost output, no snapshots, no grandparents.  In that envelope all
 * tombstones are obsolete at emission (<= earliest_snapshot = +inf at
 * bottommost), so outputs carry no range-del block and the only effect is
 * dropping covered point keys; with a single snapshot stripe the rule is
 * simply: drop key iff some tombstone [start,end) contains its user key
 * with tombstone seq > key seq.  Fragmentation
 * (db/range_tombstone_fragmenter.cc) reduces to interval max-seq here. */
typedef struct rd_frag {
  uint8_t* start;
  size_t start_len;
  uint64_t max_seq; /* over tombstones covering [start, next_start) */
} rd_frag;
typedef struct rd_aggr {
  /* raw tombstones */
  uint8_t** s;
  size_t* sl;
  uint8_t** e;
  size_t* el;
  uint64_t* seq;
  size_t n, cap;
  /* fragments: boundaries sorted asc; frag i covers [b[i], b[i+1]) */
  rd_frag* frags;
  size_t nfrags;
} rd_aggr;

static int uk_cmp(const uint8_t* a, size_t al, const uint8_t* b, size_t bl) {
  size_t m = al < bl ? al : bl;
  int c = memcmp(a, b, m);
  if (c) return c;
  return al < bl ? -1 : (al > bl ? 1 : 0);
}

static int rd_tomb_cb(void* arg, const uint8_t* start_uk, size_t slen,
                      const uint8_t* end_uk, size_t elen, uint64_t seq) {
  rd_aggr* a = (rd_aggr*)arg;
  if (a->n == a->cap) {
    a->cap = a->cap ? a->cap * 2 : 16;
    a->s = (uint8_t**)realloc(a->s, a->cap * sizeof(void*));
    a->sl = (size_t*)realloc(a->sl, a->cap * sizeof(size_t));
    a->e = (uint8_t**)realloc(a->e, a->cap * sizeof(void*));
    a->el = (size_t*)realloc(a->el, a->cap * sizeof(size_t));
    a->seq = (uint64_t*)realloc(a->seq, a->cap * sizeof(uint64_t));
  }
  a->s[a->n] = (uint8_t*)malloc(slen ? slen : 1);
  memcpy(a->s[a->n], start_uk, slen);
  a->sl[a->n] = slen;
  a->e[a->n] = (uint8_t*)malloc(elen ? elen : 1);
  memcpy(a->e[a->n], end_uk, elen);
  a->el[a->n] = elen;
  a->seq[a->n] = seq;
  a->n++;
  return 0;
}

static void rd_build_frags(rd_aggr* a) {
  size_t nb = a->n * 2;
  if (!nb) return;
  /* boundary list = all starts + ends, sorted unique */
  const uint8_t** bp = (const uint8_t**)malloc(nb * sizeof(void*));
  size_t* bl = (size_t*)malloc(nb * sizeof(size_t));
  for (size_t i = 0; i < a->n; i++) {
    bp[2 * i] = a->s[i];
    bl[2 * i] = a->sl[i];
    bp[2 * i + 1] = a->e[i];
    bl[2 * i + 1] = a->el[i];
  }
  /* insertion sort (tombstone counts are small) */
  for (size_t i = 1; i < nb; i++)
    for (size_t j = i; j > 0 && uk_cmp(bp[j], bl[j], bp[j - 1], bl[j - 1]) < 0;
         j--) {
      const uint8_t* tp = bp[j];
      size_t tl = bl[j];
      bp[j] = bp[j - 1];
      bl[j] = bl[j - 1];
      bp[j - 1] = tp;
      bl[j - 1] = tl;
    }
  a->frags = (rd_frag*)calloc(nb, sizeof(rd_frag));
  a->nfrags = 0;
  /* every boundary starts a fragment [b_i, b_{i+1}); max_seq 0 marks a
   * gap (incl. the terminator starting at the last boundary) */
  for (size_t i = 0; i + 1 <= nb - 1 + 1; i++) {
    if (i + 1 < nb && uk_cmp(bp[i], bl[i], bp[i + 1], bl[i + 1]) == 0)
      continue;
    uint64_t mx = 0;
    if (i + 1 < nb)
      for (size_t t = 0; t < a->n; t++)
        if (uk_cmp(a->s[t], a->sl[t], bp[i], bl[i]) <= 0 &&
            uk_cmp(bp[i], bl[i], a->e[t], a->el[t]) < 0 && a->seq[t] > mx)
          mx = a->seq[t];
    rd_frag* f = &a->frags[a->nfrags++];
    f->start = (uint8_t*)malloc(bl[i] ? bl[i] : 1);
    memcpy(f->start, bp[i], bl[i]);
    f->start_len = bl[i];
    f->max_seq = mx;
    if (i + 1 == nb) break;
  }
  free(bp);
  free(bl);
}

/* covered iff the fragment containing uk has max_seq > seq; keys past the
 * last boundary are uncovered (the last interval ends at the max end
 * boundary which is itself a fragment start with max_seq 0) */
static int rd_covers(const rd_aggr* a, const uint8_t* uk, size_t ul,
                     uint64_t seq) {
  if (!a->nfrags) return 0;
  size_t lo = 0, hi = a->nfrags;
  while (lo < hi) { /* first frag with start > uk */
    size_t mid = (lo + hi) / 2;
    if (uk_cmp(a->frags[mid].start, a->frags[mid].start_len, uk, ul) <= 0)
      lo = mid + 1;
    else
      hi = mid;
  }
  if (lo == 0) return 0; /* before the first tombstone */
  return a->frags[lo - 1].max_seq > seq;
}

static void rd_free(rd_aggr* a) {
  for (size_t i = 0; i < a->n; i++) {
    free(a->s[i]);
    free(a->e[i]);
  }
  free(a->s);
  free(a->sl);
  free(a->e);
  free(a->el);
  free(a->seq);
  for (size_t f = 0; f < a->nfrags; f++) free(a->frags[f].start);
  free(a->frags);
  memset(a, 0, sizeof(*a));
}

/* ---------- compaction iterator (compaction_iterator.cc FSM) ---------- */
typedef struct citer {
  kmerge* in;
  const dcw_job_desc* d;
  const rd_aggr* rd; /* range-deletion aggregator (NULL when no tombstones) */
  /* config */
  int visible_at_tip;
  uint64_t earliest_snapshot;
  /* output slot */
  int valid;
  uint8_t key[MAX_IKEY];
  size_t klen;
  const uint8_t* val;
  size_t vlen;
  uint64_t seq;
  uint8_t type;
  /* state */
  int has_current_user_key;
  uint8_t cur_ukey[MAX_IKEY];
  size_t cur_ukey_len;
  uint64_t current_user_key_sequence;
  uint64_t current_user_key_snapshot;
  int has_outputted_key;
  int last_key_seq_zeroed;
  int clear_and_output_next_key;
  int at_next;
  /* stats */
  uint64_t num_input_records, num_output_records;
  /* KeyNotExistsBeyondOutputLevel monotonic level pointers (compaction.cc:564) */
  size_t level_ptrs[16];
  char err[256];
  int failed;
} citer;

static int ukey_cmp2(const uint8_t* a, size_t alen, const uint8_t* b, size_t blen) {
  size_t n = alen < blen ? alen : blen;
  int c = memcmp(a, b, n);
  if (c) return c;
  return alen < blen ? -1 : (alen > blen ? 1 : 0);
}

/* Compaction::KeyNotExistsBeyondOutputLevel (compaction.cc:548-586) */
static int key_not_exists_beyond(citer* c, const uint8_t* ukey, size_t ulen) {
  const dcw_job_desc* d = c->d;
  if (d->bottommost_level) return 1;
  if (!d->levels_below_valid) return 0; /* reference worker branch (:555-556) */
  for (uint32_t lvl = 0; lvl < d->num_levels_below && lvl < 16; lvl++) {
    const dcw_level_files* lf = &d->levels_below[lvl];
    for (; c->level_ptrs[lvl] < lf->num_files; c->level_ptrs[lvl]++) {
      const dcw_grandparent* f = &lf->files[c->level_ptrs[lvl]];
      if (ukey_cmp2(ukey, ulen, f->largest_ukey, f->largest_len) <= 0) {
        if (ukey_cmp2(ukey, ulen, f->smallest_ukey, f->smallest_len) >= 0)
          return 0;
        break;
      }
    }
  }
  return 1;
}

/* findEarliestVisibleSnapshot (compaction_iterator.cc:1343-1369, no checker):
 * first snapshot >= seq; prev = the one below (0 if none). */
static uint64_t find_earliest_visible(const citer* c, uint64_t seq, uint64_t* prev) {
  const uint64_t* s = c->d->snapshots;
  uint32_t n = c->d->num_snapshots;
  uint32_t lo = 0, hi = n;
  while (lo < hi) {
    uint32_t mid = (lo + hi) / 2;
    if (s[mid] < seq)
      lo = mid + 1;
    else
      hi = mid;
  }
  *prev = lo > 0 ? s[lo - 1] : 0;
  return lo < n ? s[lo] : KMAXSEQ;
}

static void ci_parse(const uint8_t* k, size_t klen, uint64_t* seq, uint8_t* type,
                     size_t* ukey_len) {
  uint64_t tag;
  memcpy(&tag, k + klen - 8, 8);
  *seq = tag >> 8;
  *type = (uint8_t)tag;
  *ukey_len = klen - 8;
}

static int ukey_eq(const uint8_t* a, size_t alen, const uint8_t* b, size_t blen) {
  return alen == blen && memcmp(a, b, alen) == 0;
}

static void ci_next_from_input(citer* c) {
  c->at_next = 0;
  c->valid = 0;
  kmerge* in = c->in;
  while (!c->valid && km_valid(in)) {
    size_t klen, vlen;
    const uint8_t* k = km_key(in, &klen);
    const uint8_t* v = km_val(in, &vlen);
    c->num_input_records++;
    uint64_t seq;
    uint8_t type;
    size_t ulen;
    ci_parse(k, klen, &seq, &type, &ulen);
    if (!(type == DCW_TYPE_VALUE || type == DCW_TYPE_DELETION ||
          type == DCW_TYPE_SINGLE_DELETION)) {
      snprintf(c->err, sizeof(c->err), "value type 0x%x outside worker envelope", type);
      c->failed = 1;
      return;
    }
    /* copy current key/value into the output slot */
    memcpy(c->key, k, klen);
    c->klen = klen;
    c->val = v;
    c->vlen = vlen;
    c->seq = seq;
    c->type = type;

    int user_key_changed =
        !c->has_current_user_key || !ukey_eq(k, ulen, c->cur_ukey, c->cur_ukey_len);
    if (user_key_changed) {
      c->current_user_key_sequence = KMAXSEQ;
      c->current_user_key_snapshot = 0;
      c->has_current_user_key = 1;
      memcpy(c->cur_ukey, k, ulen);
      c->cur_ukey_len = ulen;
      c->has_outputted_key = 0;
      c->last_key_seq_zeroed = 0;
    }
    uint64_t last_sequence = c->current_user_key_sequence;
    (void)last_sequence;
    c->current_user_key_sequence = seq;
    uint64_t last_snapshot = c->current_user_key_snapshot;
    uint64_t prev_snapshot = 0;
    c->current_user_key_snapshot =
        c->visible_at_tip ? c->earliest_snapshot
                          : find_earliest_visible(c, seq, &prev_snapshot);

    if (c->clear_and_output_next_key) {
      /* Optimization 3 (compaction_iterator.cc:640-667): keep this Put,
       * drop its data */
      if (type != DCW_TYPE_VALUE) {
        snprintf(c->err, sizeof(c->err), "unexpected key type after kept SD");
        c->failed = 1;
        return;
      }
      c->val = (const uint8_t*)"";
      c->vlen = 0;
      c->valid = 1; /* kKeepSDAndClearPut */
      c->clear_and_output_next_key = 0;
    } else if (type == DCW_TYPE_SINGLE_DELETION) {
      /* peek ahead (compaction_iterator.cc:722-...) */
      if (km_next(in) != 0) goto in_err;
      size_t nklen;
      uint64_t nseq;
      uint8_t ntype;
      size_t nulen;
      const uint8_t* nk = km_valid(in) ? km_key(in, &nklen) : NULL;
      if (nk) ci_parse(nk, nklen, &nseq, &ntype, &nulen);
      if (nk && ukey_eq(nk, nulen, c->cur_ukey, c->cur_ukey_len)) {
        if (c->last_key_seq_zeroed) {
          if (km_next(in) != 0) goto in_err; /* drop SD and next */
        } else if (prev_snapshot == 0 || nseq > prev_snapshot) {
          if (ntype == DCW_TYPE_SINGLE_DELETION) {
            /* two SDs in a row: skip the first (input already advanced) */
          } else if (ntype == DCW_TYPE_DELETION) {
            /* SD+DEL mix: enforce_single_del_contracts (default true) ->
             * Corruption (compaction_iterator.cc:779-800) */
            snprintf(c->err, sizeof(c->err),
                     "Found SD and DEL on the same key (contract violation)");
            c->failed = 1;
            return;
          } else if (c->has_outputted_key ||
                     seq <= c->d->earliest_write_conflict_snapshot ||
                     (c->earliest_snapshot <
                          c->d->earliest_write_conflict_snapshot &&
                      seq <= c->earliest_snapshot)) {
            /* drop both SD and the value */
            if (km_next(in) != 0) goto in_err;
          } else {
            c->valid = 1; /* kKeepSDForConflictCheck */
            c->clear_and_output_next_key = 1;
          }
        } else {
          c->valid = 1; /* kKeepSDForSnapshot */
        }
      } else {
        /* end of input or different key (compaction_iterator.cc:856-890) */
        c->has_current_user_key = 0;
        if (seq <= c->earliest_snapshot &&
            key_not_exists_beyond(c, c->cur_ukey, c->cur_ukey_len)) {
          /* drop SD */
        } else if (c->last_key_seq_zeroed) {
          /* drop */
        } else {
          c->valid = 1; /* kKeepSD */
        }
      }
      if (c->valid) c->at_next = 1;
    } else if (last_snapshot == c->current_user_key_snapshot ||
               (last_snapshot > 0 && last_snapshot < c->current_user_key_snapshot)) {
      /* rule (A): hidden by newer entry for the same user key (:894-916) */
      if (km_next(in) != 0) goto in_err;
    } else if (type == DCW_TYPE_DELETION && seq <= c->earliest_snapshot &&
               key_not_exists_beyond(c, c->cur_ukey, c->cur_ukey_len)) {
      /* obsolete deletion marker (:917-952) */
      if (km_next(in) != 0) goto in_err;
    } else if (type == DCW_TYPE_DELETION && c->d->bottommost_level) {
      /* bottommost delete: skip versions in the same snapshot range (:953-995) */
      if (km_next(in) != 0) goto in_err;
      for (;;) {
        if (!km_valid(in)) break;
        size_t nklen;
        const uint8_t* nk = km_key(in, &nklen);
        uint64_t nseq;
        uint8_t ntype;
        size_t nulen;
        ci_parse(nk, nklen, &nseq, &ntype, &nulen);
        (void)ntype;
        if (!ukey_eq(nk, nulen, c->cur_ukey, c->cur_ukey_len)) break;
        if (!(prev_snapshot == 0 || nseq > prev_snapshot)) break;
        if (km_next(in) != 0) goto in_err;
      }
      if (km_valid(in)) {
        size_t nklen;
        const uint8_t* nk = km_key(in, &nklen);
        uint64_t nseq;
        uint8_t ntype;
        size_t nulen;
        ci_parse(nk, nklen, &nseq, &ntype, &nulen);
        (void)nseq;
        (void)ntype;
        if (ukey_eq(nk, nulen, c->cur_ukey, c->cur_ukey_len)) {
          c->valid = 1; /* kKeepDel */
          c->at_next = 1;
        }
      }
    } else if (c->rd && rd_covers(c->rd, c->cur_ukey, c->cur_ukey_len, seq)) {
      /* dropped by a range tombstone (compaction_iterator.cc:1056-1063:
       * the ShouldDelete check sits in the final keep branch only) */
      if (km_next(in) != 0) goto in_err;
    } else {
      c->valid = 1; /* kNewUserKey */
    }
  }
  return;
in_err:
  snprintf(c->err, sizeof(c->err), "%s", c->in->err);
  c->failed = 1;
}

/* PrepareOutput seq-zeroing (compaction_iterator.cc:1286-1328) */
static void ci_prepare_output(citer* c) {
  if (!c->valid) return;
  if (c->d->bottommost_level && c->seq <= c->earliest_snapshot &&
      c->type != DCW_TYPE_MERGE) {
    if (c->type == DCW_TYPE_DELETION || c->type == DCW_TYPE_SINGLE_DELETION) {
      snprintf(c->err, sizeof(c->err), "unexpected tombstone in seq-zero path");
      c->failed = 1;
      return;
    }
    c->seq = 0;
    uint64_t tag = (uint64_t)c->type; /* (0<<8)|type */
    memcpy(c->key + c->klen - 8, &tag, 8);
    c->last_key_seq_zeroed = 1;
  }
}

static void ci_seek_to_first(citer* c) {
  ci_next_from_input(c);
  ci_prepare_output(c);
}
static void ci_next(citer* c) {
  if (!c->at_next) {
    if (km_next(c->in) != 0) {
      snprintf(c->err, sizeof(c->err), "%s", c->in->err);
      c->failed = 1;
      return;
    }
  }
  ci_next_from_input(c);
  if (c->valid) c->has_outputted_key = 1; /* compaction_iterator.cc:223-226 */
  ci_prepare_output(c);
}

/* ---------- output file cutting (compaction_outputs.cc) ---------- */
typedef struct outputs {
  const dcw_job_desc* d;
  orc_table_builder* builder;
  /* DcwZipTable output mode (d->output_table_factory == 1, dzt.c spec):
   * the cut rule runs on UNCOMPRESSED bytes at value-block boundaries */
  orc_dzt_builder* zb;
  uint64_t dzt_unc;       /* key-record + value bytes so far (this file) */
  uint64_t dzt_vb_ulen;   /* open value block's uncompressed bytes */
  uint32_t dzt_vb_count;  /* open value block's entry count */
  uint64_t current_output_file_size;
  /* grandparent state (compaction_outputs.cc:121-230) */
  size_t grandparent_index;
  uint64_t grandparent_overlapped_bytes;
  size_t grandparent_boundary_switched_num;
  int being_grandparent_gap;
  int seen_key;
  /* per-file meta */
  uint64_t file_number;
  uint8_t smallest[MAX_IKEY], largest[MAX_IKEY];
  size_t smallest_len, largest_len;
  uint64_t smallest_seqno, largest_seqno, file_entries;
  /* results */
  dcw_output_file* files;
  uint32_t num_files, files_cap;
  uint64_t next_file_number;
  uint64_t total_out_bytes;
  char err[256];
} outputs;

static int ukey_cmp(const uint8_t* a, size_t alen, const uint8_t* b, size_t blen) {
  size_t n = alen < blen ? alen : blen;
  int c = memcmp(a, b, n);
  if (c) return c;
  return alen < blen ? -1 : (alen > blen ? 1 : 0);
}

/* UpdateGrandparentBoundaryInfo; keys vs grandparent USER-key bounds
 * (sstableKeyCompare == user-key compare; no range-tombstone sentinels in
 * the job desc, so ties compare equal) */
static uint64_t gp_cur_overlap(const outputs* o, const uint8_t* ikey, size_t iklen) {
  /* GetCurrentKeyGrandparentOverlappedBytes (compaction_outputs.cc:189-229) */
  if (o->being_grandparent_gap || o->d->num_grandparents == 0) return 0;
  uint64_t b = o->d->grandparents[o->grandparent_index].file_size;
  size_t ulen = iklen - 8;
  for (int64_t i = (int64_t)o->grandparent_index - 1;
       i >= 0 && ukey_cmp(ikey, ulen, o->d->grandparents[i].largest_ukey,
                          o->d->grandparents[i].largest_len) == 0;
       i--)
    b += o->d->grandparents[i].file_size;
  return b;
}
static size_t gp_update(outputs* o, const uint8_t* ikey, size_t iklen) {
  const dcw_job_desc* d = o->d;
  if (d->num_grandparents == 0) return 0;
  size_t ulen = iklen - 8;
  size_t switched = 0;
  while (o->grandparent_index < d->num_grandparents) {
    const dcw_grandparent* g = &d->grandparents[o->grandparent_index];
    if (o->being_grandparent_gap) {
      if (ukey_cmp(ikey, ulen, g->smallest_ukey, g->smallest_len) < 0) break;
      if (o->seen_key) {
        switched++;
        o->grandparent_boundary_switched_num++;
        o->grandparent_overlapped_bytes += g->file_size;
      }
      o->being_grandparent_gap = 0;
    } else {
      int cmp = ukey_cmp(ikey, ulen, g->largest_ukey, g->largest_len);
      if (cmp < 0 ||
          (cmp == 0 && (o->grandparent_index == d->num_grandparents - 1 ||
                        ukey_cmp(ikey, ulen,
                                 d->grandparents[o->grandparent_index + 1].smallest_ukey,
                                 d->grandparents[o->grandparent_index + 1].smallest_len) < 0)))
        break;
      if (o->seen_key) {
        switched++;
        o->grandparent_boundary_switched_num++;
      }
      o->being_grandparent_gap = 1;
      o->grandparent_index++;
    }
  }
  if (!o->seen_key && !o->being_grandparent_gap) {
    o->grandparent_overlapped_bytes = gp_cur_overlap(o, ikey, iklen);
  }
  o->seen_key = 1;
  return switched;
}

/* ShouldStopBefore (compaction_outputs.cc:231-352) */
static int out_should_stop_before(outputs* o, const uint8_t* ikey, size_t iklen) {
  const dcw_job_desc* d = o->d;
  uint64_t previous_overlapped = o->grandparent_overlapped_bytes;
  size_t crossed = 0;
  if (d->output_level > 0) crossed = gp_update(o, ikey, iklen);
  if (o->builder == NULL) return 0;
  if (d->output_level == 0) return 0;
  if (o->current_output_file_size >= d->target_file_size) return 1;
  if (crossed > 0) {
    if (o->grandparent_overlapped_bytes + o->current_output_file_size >
        d->max_compaction_bytes)
      return 1;
    size_t skippable = o->being_grandparent_gap ? 2 : 3;
    if (d->level_compaction_dynamic_file_size && crossed >= skippable &&
        o->grandparent_overlapped_bytes - previous_overlapped >
            d->target_file_size / 8)
      return 1;
    if (d->level_compaction_dynamic_file_size &&
        o->current_output_file_size >=
            ((d->target_file_size + 99) / 100) *
                (50 + (o->grandparent_boundary_switched_num * 5 < 40
                           ? o->grandparent_boundary_switched_num * 5
                           : 40)))
      return 1;
  }
  return 0;
}

static void fill_topts(outputs* o, orc_table_opts* t) {
  orc_table_opts_default(t);
  const dcw_job_desc* d = o->d;
  if (d->block_size) t->block_size = d->block_size;
  if (d->block_restart_interval) t->block_restart_interval = d->block_restart_interval;
  if (d->index_block_restart_interval)
    t->index_block_restart_interval = d->index_block_restart_interval;
  if (d->format_version) t->format_version = d->format_version;
  t->checksum_type = d->checksum_type;
  t->compression = d->compression;
  if (d->block_size_deviation) t->block_size_deviation = d->block_size_deviation;
  t->db_id = d->db_id;
  t->db_session_id = d->db_session_id;
  t->db_host_id = d->db_host_id;
  t->cf_name = d->cf_name;
  t->cf_id = d->cf_id;
  t->orig_file_number = o->file_number;
  t->creation_time = d->oldest_ancester_time ? d->oldest_ancester_time : d->current_time;
  t->file_creation_time = d->current_time;
  t->oldest_key_time = 0;
  t->level_at_creation = d->output_level;
  t->bloom_millibits_per_key = d->bloom_millibits_per_key;
}

static void out_open(outputs* o) {
  o->file_number = o->next_file_number++;
  orc_table_opts t;
  fill_topts(o, &t);
  if (o->d->output_table_factory != 1) o->builder = orc_table_builder_new(&t);
  o->zb = NULL; /* DZT builder created in out_add once ukey_len is known */
  o->dzt_unc = 0;
  o->dzt_vb_ulen = 0;
  o->dzt_vb_count = 0;
  o->smallest_len = o->largest_len = 0;
  o->smallest_seqno = KMAXSEQ;
  o->largest_seqno = 0;
  o->file_entries = 0;
}

static int out_close(outputs* o) {
  orc_buf file = {0};
  if (o->zb) {
    orc_dzt_builder_finish(o->zb, &file);
    orc_dzt_builder_delete(o->zb);
    o->zb = NULL;
  } else {
    orc_table_builder_finish(o->builder, &file);
    orc_table_builder_delete(o->builder);
    o->builder = NULL;
  }
  char path[600];
  snprintf(path, sizeof(path), "%s/%06" PRIu64 ".sst", o->d->output_dir,
           o->file_number);
  FILE* f = fopen(path, "wb");
  if (!f) {
    snprintf(o->err, sizeof(o->err), "cannot write %s", path);
    orc_buf_free(&file);
    return -1;
  }
  fwrite(file.data, 1, file.size, f);
  fclose(f);
  if (o->num_files == o->files_cap) {
    o->files_cap = o->files_cap ? o->files_cap * 2 : 8;
    o->files = (dcw_output_file*)realloc(o->files, o->files_cap * sizeof(dcw_output_file));
  }
  dcw_output_file* of = &o->files[o->num_files++];
  memset(of, 0, sizeof(*of));
  snprintf(of->path, sizeof(of->path), "%s", path);
  of->file_number = o->file_number;
  of->file_size = file.size;
  memcpy(of->smallest_ikey, o->smallest, o->smallest_len > 64 ? 64 : o->smallest_len);
  of->smallest_len = (uint32_t)o->smallest_len;
  memcpy(of->largest_ikey, o->largest, o->largest_len > 64 ? 64 : o->largest_len);
  of->largest_len = (uint32_t)o->largest_len;
  of->smallest_seqno = o->smallest_seqno == KMAXSEQ ? 0 : o->smallest_seqno;
  of->largest_seqno = o->largest_seqno;
  of->num_entries = o->file_entries;
  o->total_out_bytes += file.size;
  orc_buf_free(&file);
  o->current_output_file_size = 0;
  /* reset grandparent accounting (compaction_outputs.cc:374-380) */
  o->grandparent_boundary_switched_num = 0;
  return 0;
}

/* AddToOutput (compaction_outputs.cc:356-420) */
static int out_add(outputs* o, const uint8_t* key, size_t klen, const uint8_t* val,
                   size_t vlen) {
  if (o->d->output_table_factory == 1) {
    /* DcwZipTable output (dzt.c spec): cut when this entry would start a
     * new value block and the file's uncompressed bytes reached target */
    int vb_closes = o->zb && (o->dzt_vb_count >= 256 ||
                              (o->dzt_vb_count > 0 &&
                               o->dzt_vb_ulen + vlen > 16384));
    if (o->zb && vb_closes && o->dzt_unc >= o->d->target_file_size) {
      if (out_close(o) != 0) return -1;
      vb_closes = 0;
    }
    if (o->zb == NULL && o->builder == NULL) out_open(o);
    if (o->zb == NULL) o->zb = ({
          orc_table_opts t;
          fill_topts(o, &t);
          orc_dzt_builder_new(&t, (uint32_t)(klen - 8));
        });
    if (vb_closes) {
      o->dzt_vb_count = 0;
      o->dzt_vb_ulen = 0;
    }
    o->dzt_vb_count++;
    o->dzt_vb_ulen += vlen;
    int krec = orc_dzt_builder_add(o->zb, key, klen, val, vlen);
    if (krec < 0) {
      snprintf(o->err, sizeof(o->err), "DZT: non-uniform user key length");
      return -1;
    }
    o->dzt_unc += (uint64_t)krec + vlen;
    o->current_output_file_size = o->dzt_unc;
    uint64_t tag;
    memcpy(&tag, key + klen - 8, 8);
    uint64_t seq = tag >> 8;
    if (o->file_entries == 0) {
      memcpy(o->smallest, key, klen);
      o->smallest_len = klen;
    }
    memcpy(o->largest, key, klen);
    o->largest_len = klen;
    if (seq < o->smallest_seqno) o->smallest_seqno = seq;
    if (seq > o->largest_seqno) o->largest_seqno = seq;
    o->file_entries++;
    return 0;
  }
  if (out_should_stop_before(o, key, klen) && o->builder != NULL) {
    if (out_close(o) != 0) return -1;
    o->grandparent_overlapped_bytes = gp_cur_overlap(o, key, klen);
  }
  if (o->builder == NULL) out_open(o);
  orc_table_builder_add(o->builder, key, klen, val, vlen);
  o->current_output_file_size = orc_table_builder_file_size(o->builder);
  uint64_t tag;
  memcpy(&tag, key + klen - 8, 8);
  uint64_t seq = tag >> 8;
  if (o->file_entries == 0) {
    memcpy(o->smallest, key, klen);
    o->smallest_len = klen;
  }
  memcpy(o->largest, key, klen);
  o->largest_len = klen;
  if (seq < o->smallest_seqno) o->smallest_seqno = seq;
  if (seq > o->largest_seqno) o->largest_seqno = seq;
  o->file_entries++;
  return 0;
}

/* ---------- the job ---------- */
int32_t orc_execute(const dcw_job_desc* d, dcw_job_result* res) {
  memset(res, 0, sizeof(*res));
  uint64_t t0 = now_usec();
  if (d->comparator_name && strcmp(d->comparator_name, "leveldb.BytewiseComparator") != 0) {
    snprintf(res->error, sizeof(res->error), "unsupported comparator");
    res->status = 2;
    return 2;
  }
  /* flush offload: reduce the raw sorted KV stream to a
   * single-run compaction by materializing it as an uncompressed temp SST
   * (BlockBasedTable encode/decode round-trips the stream exactly, so the
   * reduction is semantics-preserving; BuildTable = same iterator pass,
   * db/builder.cc:56) */
  char flush_tmp[600];
  flush_tmp[0] = 0;
  const char* flush_files[1];
  dcw_run flush_run;
  dcw_job_desc d2;
  if (d->flush_kv != NULL) {
    if (d->num_runs) {
      snprintf(res->error, sizeof(res->error), "flush job must carry no runs");
      res->status = 2;
      return 2;
    }
    orc_table_opts t;
    orc_table_opts_default(&t);
    orc_table_builder* tb = orc_table_builder_new(&t);
    for (uint64_t i = 0; i < d->flush_num_entries; i++) {
      const uint8_t* p = d->flush_kv + d->flush_offsets[i];
      uint32_t klen, vl;
      memcpy(&klen, p, 4);
      memcpy(&vl, p + 4 + klen, 4);
      orc_table_builder_add(tb, p + 4, klen, p + 8 + klen, vl);
    }
    orc_buf img = {0};
    orc_table_builder_finish(tb, &img);
    orc_table_builder_delete(tb);
    snprintf(flush_tmp, sizeof(flush_tmp), "%s/.flush_input.tmp",
             d->output_dir);
    FILE* fp = fopen(flush_tmp, "wb");
    if (!fp || fwrite(img.data, 1, img.size, fp) != img.size) {
      if (fp) fclose(fp);
      orc_buf_free(&img);
      snprintf(res->error, sizeof(res->error), "flush temp write failed");
      res->status = 3;
      return 3;
    }
    fclose(fp);
    orc_buf_free(&img);
    d2 = *d;
    flush_files[0] = flush_tmp;
    flush_run.files = flush_files;
    flush_run.num_files = 1;
    d2.runs = &flush_run;
    d2.num_runs = 1;
    d = &d2;
  }

  kmerge km;
  memset(&km, 0, sizeof(km));
  if (km_init(&km, d->runs, d->num_runs) != 0) {
    snprintf(res->error, sizeof(res->error), "%s", km.err);
    km_free(&km);
    res->status = 3;
    return 3;
  }
  uint64_t in_bytes = 0;
  /* input bytes = sum of input file sizes (metric numerator) */
  for (uint32_t r = 0; r < d->num_runs; r++)
    for (uint32_t f = 0; f < d->runs[r].num_files; f++) {
      FILE* fp = fopen(d->runs[r].files[f], "rb");
      if (fp) {
        fseek(fp, 0, SEEK_END);
        in_bytes += (uint64_t)ftell(fp);
        fclose(fp);
      }
    }

  /* range-deletion tombstones from every input's meta block
   * (MakeInputIterator adds all inputs' tombstones up front,
   * db/version_set.cc:7269-7352) */
  rd_aggr rd;
  memset(&rd, 0, sizeof(rd));
  for (uint32_t r = 0; r < d->num_runs; r++)
    for (uint32_t f = 0; f < d->runs[r].num_files; f++) {
      FILE* fp = fopen(d->runs[r].files[f], "rb");
      if (!fp) continue;
      fseek(fp, 0, SEEK_END);
      long fsz = ftell(fp);
      fseek(fp, 0, SEEK_SET);
      uint8_t* img = (uint8_t*)malloc((size_t)fsz);
      size_t got = fread(img, 1, (size_t)fsz, fp);
      fclose(fp);
      if (got == (size_t)fsz) {
        char terr[128];
        orc_table_reader* tr = orc_table_open(img, got, terr, sizeof(terr));
        if (tr) {
          (void)orc_table_tombstones(tr, rd_tomb_cb, &rd);
          orc_table_close(tr);
        }
      }
      free(img);
    }
  if (rd.n) {
    /* supported envelope (see rd_aggr header comment); outside it the
     * worker refuses and the DB runs the job locally */
    if (!d->bottommost_level || d->num_snapshots || d->num_grandparents) {
      snprintf(res->error, sizeof(res->error),
               "range deletions outside supported envelope "
               "(bottommost, no snapshots, no grandparents)");
      rd_free(&rd);
      km_free(&km);
      res->status = 6;
      return 6;
    }
    rd_build_frags(&rd);
  }

  citer ci;
  memset(&ci, 0, sizeof(ci));
  ci.in = &km;
  ci.d = d;
  ci.rd = rd.n ? &rd : NULL;
  ci.visible_at_tip = d->num_snapshots == 0;
  ci.earliest_snapshot = d->num_snapshots ? d->snapshots[0] : KMAXSEQ;

  outputs out;
  memset(&out, 0, sizeof(out));
  out.d = d;
  out.being_grandparent_gap = 1;
  out.next_file_number = d->next_file_number;

  ci_seek_to_first(&ci);
  while (ci.valid && !ci.failed) {
    if (out_add(&out, ci.key, ci.klen, ci.val, ci.vlen) != 0) {
      snprintf(res->error, sizeof(res->error), "%s", out.err);
      res->status = 4;
      km_free(&km);
      rd_free(&rd);
      free(out.files);
      return 4;
    }
    ci.num_output_records++;
    ci_next(&ci);
  }
  if (ci.failed) {
    snprintf(res->error, sizeof(res->error), "%s", ci.err);
    res->status = 5;
    if (out.builder) orc_table_builder_delete(out.builder);
    km_free(&km);
    rd_free(&rd);
    free(out.files);
    return 5;
  }
  if (out.builder || out.zb) {
    if (out_close(&out) != 0) {
      snprintf(res->error, sizeof(res->error), "%s", out.err);
      res->status = 4;
      km_free(&km);
      rd_free(&rd);
      free(out.files);
      return 4;
    }
  }
  km_free(&km);
  rd_free(&rd);
  if (flush_tmp[0]) {
    remove(flush_tmp);
    in_bytes = d->flush_kv_bytes; /* the metric numerator is the raw stream */
  }
  res->status = 0;
  res->files = out.files;
  res->num_files = out.num_files;
  res->in_bytes = in_bytes;
  res->out_bytes = out.total_out_bytes;
  res->in_entries = ci.num_input_records;
  res->out_entries = ci.num_output_records;
  res->work_time_usec = now_usec() - t0;
  return 0;
}

void orc_free_result(dcw_job_result* res) {
  free(res->files);
  res->files = NULL;
  res->num_files = 0;
}
