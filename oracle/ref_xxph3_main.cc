/* ref_xxph3_main.cc — golden-vector generator for XXPH3 (the filter hash),
 * compiled against the REFERENCE's own util/xxph3.h (oracle/_ref pattern;
 * dev container only — the committed vectors travel instead).
 * Prints: len hex_input hash for deterministic inputs of len 0..128. */
#include <stdint.h>
#include <stdio.h>
#include <string.h>

#include "util/xxph3.h"

int main(void) {
  uint8_t buf[256];
  uint32_t x = 0x12345678u;
  for (int i = 0; i < 256; i++) { /* xorshift32 bytes */
    x ^= x << 13; x ^= x >> 17; x ^= x << 5;
    buf[i] = (uint8_t)x;
  }
  for (int len = 0; len <= 128; len++) {
    unsigned long long h = (unsigned long long)XXPH3_64bits(buf, (size_t)len);
    printf("%d ", len);
    for (int i = 0; i < len; i++) printf("%02x", buf[i]);
    printf(" %016llx\n", h);
  }
  return 0;
}
