/* ref_xxh_main.c — oracle/_ref harness: compiles the REFERENCE's vendored
 * xxHash header (util/xxhash.h, upstream xxHash) from where it lies under
 * /root/reference and prints XXH3_64bits digests, to pin the oracle's XXH3
 * restatement and to generate tests/golden/xxh3_vectors.json.
 * Build recipe: oracle/Makefile (target _ref/xxh_ref; only when
 * /root/reference is present — never shipped, output under oracle/_ref/).
 *
 * Usage: xxh_ref <len...>   — for each len, hashes the deterministic
 * byte pattern b[i] = (i*2654435761 >> 24) & 0xff and prints
 * "<len> <hex64>" per line.
 */
#include <stdint.h>
#include <stdio.h>
#include <stdlib.h>

#define XXH_INLINE_ALL
#include "util/xxhash.h" /* resolved via -I$(REF) */

int main(int argc, char** argv) {
  static uint8_t buf[1 << 22];
  for (size_t i = 0; i < sizeof(buf); i++)
    buf[i] = (uint8_t)((i * 2654435761ULL) >> 24);
  for (int a = 1; a < argc; a++) {
    size_t len = (size_t)strtoull(argv[a], NULL, 10);
    if (len > sizeof(buf)) len = sizeof(buf);
    unsigned long long h = (unsigned long long)XXH3_64bits(buf, len);
    printf("%zu %016llx\n", len, h);
  }
  return 0;
}
