// Phase-cost probe for mlp_step_fused_kernel: build with -DPROBE_SKIP_*
// to excise a phase and difference the step time (timing only — skipped
// phases leave garbage math, never use for numerics).
//
//   hipcc --offload-arch=gfx950 -O3 -o benchmarks/probe_base.bin benchmarks/probe_floor.hip
//   hipcc ... -DPROBE_SKIP_FWDBWD -o benchmarks/probe_nofwd.bin ...
//
#include "../unionml_amd/ops/hip/tabular_kernels.hip"

#include <cstdio>
#include <cstdlib>

#define CHECK(x) do { hipError_t e = (x); if (e != hipSuccess) { \
  fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(e), __FILE__, __LINE__); \
  exit(1); } } while (0)

int main(int argc, char** argv) {
  const int B = argc > 1 ? atoi(argv[1]) : 2048;
  const int iters = argc > 2 ? atoi(argv[2]) : 2000;
  const int n_wg = (B + 127) / 128;

  unsigned short *Xbf, *W1bf, *W2bf, *bfmirror;
  int *y, *t_dev;
  unsigned* counter;
  float *master, *m, *v, *slabs, *loss;
  unsigned short* wimg;
  CHECK(hipMalloc(&Xbf, (size_t)B * 64 * 2));
  CHECK(hipMalloc(&W1bf, 64 * 32 * 2));
  CHECK(hipMalloc(&W2bf, 32 * 16 * 2));
  CHECK(hipMalloc(&bfmirror, (NPARAM + 8) * 2));
  CHECK(hipMalloc(&y, (size_t)B * 4));
  CHECK(hipMalloc(&t_dev, 4));
  CHECK(hipMalloc(&counter, 4));
  CHECK(hipMalloc(&master, NPARAM * 4));
  CHECK(hipMalloc(&m, NPARAM * 4));
  CHECK(hipMalloc(&v, NPARAM * 4));
  CHECK(hipMalloc(&slabs, (size_t)n_wg * SLAB * 4));
  CHECK(hipMalloc(&loss, 4));
  CHECK(hipMalloc(&wimg, 4224 * 2));
  CHECK(hipMemset(wimg, 0, 4224 * 2));
  CHECK(hipMemset(Xbf, 0, (size_t)B * 64 * 2));
  CHECK(hipMemset(W1bf, 0, 64 * 32 * 2));
  CHECK(hipMemset(W2bf, 0, 32 * 16 * 2));
  CHECK(hipMemset(y, 0, (size_t)B * 4));
  CHECK(hipMemset(t_dev, 0, 4));
  CHECK(hipMemset(counter, 0, 4));
  CHECK(hipMemset(master, 0, NPARAM * 4));
  CHECK(hipMemset(m, 0, NPARAM * 4));
  CHECK(hipMemset(v, 0, NPARAM * 4));
  CHECK(hipMemset(slabs, 0, (size_t)n_wg * SLAB * 4));

  for (int i = 0; i < 100; ++i) {
    if (launch_mlp_step_fused(Xbf, y, B, W1bf, W2bf, master, bfmirror, m, v,
                              t_dev, slabs, counter, loss, 1.0f / B, 1e-3f,
                              0.9f, 0.999f, 1e-8f, n_wg, nullptr, wimg, 0) != 0) {
      fprintf(stderr, "launch failed\n");
      return 1;
    }
  }
  CHECK(hipDeviceSynchronize());

  hipEvent_t t0, t1;
  CHECK(hipEventCreate(&t0));
  CHECK(hipEventCreate(&t1));
  CHECK(hipEventRecord(t0, 0));
  for (int i = 0; i < iters; ++i) {
    launch_mlp_step_fused(Xbf, y, B, W1bf, W2bf, master, bfmirror, m, v, t_dev,
                          slabs, counter, loss, 1.0f / B, 1e-3f, 0.9f, 0.999f,
                          1e-8f, n_wg, nullptr, wimg, 0);
  }
  CHECK(hipEventRecord(t1, 0));
  CHECK(hipEventSynchronize(t1));
  float ms = 0.f;
  CHECK(hipEventElapsedTime(&ms, t0, t1));
  printf("B=%d n_wg=%d us_per_launch=%.3f\n", B, n_wg, ms * 1000.0f / iters);
  return 0;
}
