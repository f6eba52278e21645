"""Extract snap_dec_lds (and its LDS helpers) from dcw_kernels.hip into
/tmp/snap_dec_body.inc for host compilation by dec_fuzz.cpp."""
import os

root = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
src = open(os.path.join(root, "toplingdb_amd/csrc/dcw_kernels.hip")).read()
# slice 1: the LDS load/store helpers (stop before the device-only sync)
i = src.index("__device__ __forceinline__ uint64_t lds_ld64")
j = src.index("__device__ __forceinline__ void wave_lds_sync2", i)
# slice 2: the templated serial decoder only (snap_dec_wave is device-only)
k = src.index("template <int PIPE>\n__device__ uint32_t snap_dec_lds")
depth = 0
e = src.index("{", k)
while True:
    if src[e] == "{":
        depth += 1
    elif src[e] == "}":
        depth -= 1
        if depth == 0:
            break
    e += 1
body = (src[i:j] + src[k:e + 1]).replace("__device__ __forceinline__",
                                         "static inline")
body = body.replace("__device__ ", "static ")
body += ("\nstatic uint32_t snap_dec_lds_host(const uint8_t* in, uint32_t n,"
         " uint8_t* out, uint32_t cap){ return snap_dec_lds<1>(in,n,out,cap); }\n"
         "static uint32_t snap_dec_lds_host0(const uint8_t* in, uint32_t n,"
         " uint8_t* out, uint32_t cap){ return snap_dec_lds<0>(in,n,out,cap); }\n")
open("/tmp/snap_dec_body.inc", "w").write(body)
print("extracted", e - i, "bytes")
