#ifndef KV_HOST_SHIM
#define KV_HOST_SHIM
#define KV_HOST_TEST 1
#define __device__
#define __host__
#define __global__
#define __constant__
#define __forceinline__ inline
#define __noinline__
#define __launch_bounds__(...)
#include <stdint.h>
static inline uint64_t __umul64hi(uint64_t a, uint64_t b) {
  return (uint64_t)(((unsigned __int128)a * b) >> 64);
}
#endif
/* Host-compilation shim: compiles the DEVICE headers (kv_secp_device.h,
 * kv_secp_kernels.hip) with g++ so the EC arithmetic — including the fe26
 * magnitude asserts enabled by KV_HOST_TEST — can be fuzzed against the
 * oracle and exact bigint arithmetic inside the CPU test suite. Test
 * infrastructure only; the product path compiles the same headers with hipcc
 * for gfx950. */
