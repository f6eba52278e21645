// host simulation of snap_dec_wave's batch/round structure vs oracle
#include <cstdint>
#include <cstring>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <random>
extern "C" {
  size_t orc_snappy_compress(const uint8_t*, size_t, uint8_t*);
  size_t orc_snappy_uncompress(const uint8_t*, size_t, uint8_t*, size_t);
}
static inline uint64_t ld64(const uint8_t* p){uint64_t v;memcpy(&v,p,8);return v;}
static inline void st64(uint8_t* p,uint64_t v){memcpy(p,&v,8);}
struct Op { uint32_t dst, src, len, kind; };
// mirrors snap_dec_wave: lockstep parse == single parse; rounds as on device
static uint32_t dec_wave_sim(const uint8_t* in, uint32_t n, uint8_t* out, uint32_t cap) {
  uint32_t ulen = 0, ip = 0;
  { uint64_t h = ld64(in); uint32_t s = 0;
    for (;;) { if (ip >= n || ip >= 5) return 0;
      uint8_t b = (uint8_t)(h >> (8*ip)); ulen |= (uint32_t)(b & 0x7f) << s; ip++;
      if (!(b & 0x80)) break; s += 7; } }
  if (ulen > cap) return 0;
  uint32_t opos = 0;
  while (ip < n) {
    Op ops[64]; uint32_t nops = 0;
    while (ip < n && nops < 64) {
      uint64_t h = ld64(in + ip); uint8_t tag = (uint8_t)h;
      uint32_t len, srcp, kind;
      if ((tag & 3) == 0) {
        len = (uint32_t)(tag >> 2) + 1; uint32_t hb = 1;
        if (len > 60) { uint32_t nb = len - 60;
          if (ip + 1 + nb > n) return 0;
          len = (uint32_t)((h >> 8) & (0xffffffffull >> (8*(4-nb)))) + 1; hb = 1 + nb; }
        ip += hb;
        if (ip + len > n || opos + len > ulen) return 0;
        srcp = ip; kind = 1; ip += len;
      } else {
        uint32_t offset, hb;
        if ((tag & 3) == 1) { len = ((uint32_t)(tag>>2)&7)+4; offset = ((uint32_t)(tag>>5)<<8)|(uint8_t)(h>>8); hb = 2; }
        else if ((tag & 3) == 2) { len = (uint32_t)(tag>>2)+1; offset = (uint32_t)(h>>8)&0xffffu; hb = 3; }
        else { len = (uint32_t)(tag>>2)+1; offset = (uint32_t)(h>>8); hb = 5; }
        if (ip + hb > n) return 0;
        ip += hb;
        if (offset == 0 || offset > opos || opos + len > ulen) return 0;
        srcp = opos - offset; kind = 2;
      }
      ops[nops] = {opos, srcp, len, kind};
      opos += len; nops++;
    }
    uint64_t done = 0; uint32_t frontier = 0;
    while (frontier < nops) {
      uint32_t fdst = ops[frontier].dst;
      bool any = false;
      uint64_t newly = 0;
      for (uint32_t i = 0; i < nops; i++) {
        if ((done >> i) & 1) continue;
        Op& o = ops[i];
        bool ready = (o.kind == 1) || (o.src + o.len <= fdst) || (i == frontier);
        if (!ready) continue;
        any = true; newly |= 1ull << i;
        {
          const uint8_t* sbase = o.kind == 1 ? in : out;
          uint32_t offset = o.kind == 1 ? 0xffffffffu : o.dst - o.src;
          uint32_t t = 0;
          if (offset >= 16) {
            for (; t + 16 <= o.len; t += 16) {
              uint64_t a = ld64(sbase + o.src + t), b = ld64(sbase + o.src + t + 8);
              st64(out + o.dst + t, a); st64(out + o.dst + t + 8, b);
            }
            for (; t < o.len; t++) out[o.dst + t] = sbase[o.src + t];
          } else if (offset >= 8) {
            for (; t + 8 <= o.len; t += 8) st64(out + o.dst + t, ld64(out + o.src + t));
            for (; t < o.len; t++) out[o.dst + t] = out[o.src + t];
          } else {
            uint32_t p = o.dst, src = o.src, end = o.dst + o.len;
            while (p < end) {
              uint32_t d = p - src;
              if (end - p < 8) { for (; p < end; p++) out[p] = out[p - offset]; break; }
              if (d < 8) {
                st64(out + p, ld64(out + src));
                p += d < end - p ? d : end - p;
              } else {
                uint32_t dist = d;
                for (; p + 8 <= end; p += 8) st64(out + p, ld64(out + p - dist));
                for (; p < end; p++) out[p] = out[p - offset];
                break;
              }
            }
          }
        }
      }
      if (!any) { printf("STALL\n"); return 0; }
      done |= newly;
      while (frontier < nops && ((done >> frontier) & 1)) frontier++;
    }
  }
  return opos == ulen ? ulen : 0;
}
int main(){
  std::mt19937 rng(19);
  for (int it = 0; it < 30000; it++) {
    int n = rng() % 4992 + 1;
    std::vector<uint8_t> d(n);
    int mode = it % 6;
    for (int i = 0; i < n; i++) {
      if (mode == 0) d[i] = rng();
      else if (mode == 1) d[i] = (i % (1 + it % 7));
      else if (mode == 2) d[i] = i >= 64 ? d[i-64] ^ (rng()%16==0) : rng();
      else if (mode == 3) d[i] = i >= 8 && rng()%8 ? d[i-8] : rng();
      else if (mode == 4) d[i] = i >= 2048 && rng()%4 ? d[i-2048] : (rng()%3==0 ? rng() : 'a');
      else d[i] = i >= 3 && rng()%5 ? d[i-3] : rng();   // offset-3 chains (round stress)
    }
    std::vector<uint8_t> enc(16 + n + n/6 + 1024);
    size_t en = orc_snappy_compress(d.data(), n, enc.data());
    std::vector<uint8_t> fast(n + 16), ref(n + 1);
    size_t rn = orc_snappy_uncompress(enc.data(), en, ref.data(), n);
    if (rn != (size_t)n) { printf("REF FAIL %d\n", it); return 1; }
    uint32_t fn = dec_wave_sim(enc.data(), (uint32_t)en, fast.data(), n);
    if (fn != (uint32_t)n || memcmp(fast.data(), d.data(), n)) {
      printf("WAVE MISMATCH it=%d mode=%d n=%d fn=%u\n", it, mode, n, fn); return 1; }
    std::vector<uint8_t> bad(enc.begin(), enc.begin()+en);
    bad[rng() % en] ^= 1 << (rng() % 8);
    std::vector<uint8_t> f2(n + 16), r2(n + 1);
    size_t a = orc_snappy_uncompress(bad.data(), en, r2.data(), n);
    uint32_t b = dec_wave_sim(bad.data(), (uint32_t)en, f2.data(), n);
    if ((a != 0) != (b != 0)) { printf("ACCEPT DISAGREE %d a=%zu b=%u\n", it, a, b); return 1; }
    if (a && memcmp(r2.data(), f2.data(), a)) { printf("CORRUPT DIFF %d\n", it); return 1; }
  }
  printf("wave fuzz OK\n");
  return 0;
}
