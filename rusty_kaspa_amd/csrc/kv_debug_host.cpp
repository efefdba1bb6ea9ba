#include <hip/hip_runtime.h>
#include <stdio.h>
#include <unistd.h>
extern "C" __global__ void kv_debug_verify(volatile int *, const uint8_t *, int *);
extern "C" __global__ void kv_debug_pure(volatile int *, int, int *);
extern "C" int kv_debug_run_pure(int timeout_s) {
  int *progress = nullptr;
  hipHostMalloc((void **)&progress, 64, hipHostMallocMapped);
  progress[0] = 0;
  int *d_out;
  hipMalloc(&d_out, 64);
  int *d_progress = nullptr;
  hipHostGetDevicePointer((void **)&d_progress, progress, 0);
  hipStream_t s;
  hipStreamCreate(&s);
  hipLaunchKernelGGL(kv_debug_pure, dim3(1), dim3(64), 0, s, d_progress, 256, d_out);
  printf("pure launched, err=%s\n", hipGetErrorString(hipGetLastError()));
  fflush(stdout);
  int last = -1;
  for (int t = 0; t < timeout_s * 10; t++) {
    usleep(100000);
    int cur = progress[0];
    if (cur != last) { printf("t=%.1fs pure progress=%d\n", t/10.0, cur); fflush(stdout); last = cur; }
    if (cur == 110) break;
    if (t % 20 == 19) { printf("  streamQuery=%d\n", (int)hipStreamQuery(s)); fflush(stdout); }
  }
  printf(progress[0]==110 ? "PURE DONE\n" : "PURE HUNG at %d\n", progress[0]);
  fflush(stdout);
  return progress[0];
}
extern "C" int kv_debug_run(const uint8_t *tuple, int timeout_s) {
  int *progress = nullptr;
  if (hipHostMalloc((void **)&progress, 64, hipHostMallocMapped) != hipSuccess) {
    printf("hostmalloc failed\n");
    return -1;
  }
  progress[0] = 0;
  uint8_t *d_tuple;
  int *d_out;
  hipMalloc(&d_tuple, 128);
  hipMalloc(&d_out, 64);
  hipMemcpy(d_tuple, tuple, 128, hipMemcpyHostToDevice);
  int *d_progress = nullptr;
  hipHostGetDevicePointer((void **)&d_progress, progress, 0);
  hipStream_t s;
  hipStreamCreate(&s);
  hipLaunchKernelGGL(kv_debug_verify, dim3(1), dim3(64), 0, s, d_progress, d_tuple, d_out);
  printf("launched, err=%s\n", hipGetErrorString(hipGetLastError()));
  fflush(stdout);
  int last = -1;
  for (int t = 0; t < timeout_s * 10; t++) {
    usleep(100000);
    int cur = progress[0];
    if (cur != last) {
      printf("t=%.1fs progress=%d\n", t / 10.0, cur);
      fflush(stdout);
      last = cur;
    }
    if (cur == 10) break;
    if (hipStreamQuery(s) == hipSuccess && cur == last && cur != 10) {
      /* kernel finished without reaching 10?? */
    }
  }
  if (progress[0] == 10) {
    int out[16];
    hipMemcpy(out, d_out, 64, hipMemcpyDeviceToHost);
    printf("DONE status=%d\n", out[7]);
  } else {
    printf("HUNG at progress=%d\n", progress[0]);
  }
  fflush(stdout);
  return progress[0];
}
