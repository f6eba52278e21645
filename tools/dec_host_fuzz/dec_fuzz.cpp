// host-compiled copy of snap_dec_lds for fuzzing against the oracle codec
#include <cstdint>
#include <cstring>
#include <cstdio>
#include <cstdlib>
#include <vector>
#include <random>
#define __device__
#define __forceinline__ inline
#include "/tmp/snap_dec_body.inc"
extern "C" {
  size_t orc_snappy_compress(const uint8_t*, size_t, uint8_t*);
  size_t orc_snappy_uncompress(const uint8_t*, size_t, uint8_t*, size_t);
}
int main(){
  std::mt19937 rng(7);
  for (int it = 0; it < 20000; it++) {
    // generate data with tunable redundancy to hit all op forms
    int n = rng() % 4992 + 1;
    std::vector<uint8_t> d(n);
    int mode = it % 5;
    for (int i = 0; i < n; i++) {
      if (mode == 0) d[i] = rng();                        // incompressible
      else if (mode == 1) d[i] = (i % (1 + it % 7));      // tiny periods (offset 1..7)
      else if (mode == 2) d[i] = i >= 64 ? d[i-64] ^ (rng()%16==0) : rng();  // offset 64
      else if (mode == 3) d[i] = i >= 8 && rng()%8 ? d[i-8] : rng();  // offset 8 mixed
      else d[i] = i >= 2048 && rng()%4 ? d[i-2048] : (rng()%3==0 ? rng() : 'a'); // far offsets
    }
    std::vector<uint8_t> enc(16 + n + n/6 + 1024);
    size_t en = orc_snappy_compress(d.data(), n, enc.data());
    std::vector<uint8_t> ref(n+1), fast(n + 16);
    size_t rn = orc_snappy_uncompress(enc.data(), en, ref.data(), n);
    if (rn != (size_t)n) { printf("REF FAIL it=%d\n", it); return 1; }
    uint32_t fn = snap_dec_lds_host(enc.data(), (uint32_t)en, fast.data(), n);
    uint32_t fn0 = snap_dec_lds_host0(enc.data(), (uint32_t)en, fast.data(), n);
    if (fn0 != fn) { printf("PIPE/NOPIPE DIFF it=%d\n", it); return 1; }
    if (fn != (uint32_t)n || memcmp(fast.data(), d.data(), n)) {
      printf("FAST MISMATCH it=%d mode=%d n=%d fn=%u\n", it, mode, n, fn); return 1;
    }
    // corrupt-input agreement: flip a byte, both must agree on accept/reject+output
    std::vector<uint8_t> bad(enc.begin(), enc.begin()+en);
    bad[rng() % en] ^= 1 << (rng() % 8);
    std::vector<uint8_t> r2(n+1), f2(n + 16);
    size_t a = orc_snappy_uncompress(bad.data(), en, r2.data(), n);
    uint32_t b = snap_dec_lds_host(bad.data(), (uint32_t)en, f2.data(), n);
    if ((a != 0) != (b != 0)) { printf("ACCEPT DISAGREE it=%d a=%zu b=%u\n", it, a, b); return 1; }
    if (a && memcmp(r2.data(), f2.data(), a)) { printf("CORRUPT OUT DIFF it=%d\n", it); return 1; }
  }
  printf("dec fuzz OK\n");
  return 0;
}
