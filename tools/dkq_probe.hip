// Standalone perf bisect for the fused dK+dQ kernel (no torch import —
// seconds per run).  Times each DBG variant of bwd_dkq_kernel plus the
// split-plan kernels on the s=65536 headline shape, so one gpurun call
// pinpoints which component (K^T reads, ds_add reduce, global atomics,
// LDS size, barriers) carries a slowdown.
//
// Build: hipcc --offload-arch=gfx950 -O3 -std=c++17 tools/dkq_probe.hip -o tools/dkq_probe
#include "../burst_attn_amd/csrc/attn_bwd.hip"

#include <stdio.h>
#include <vector>

#define CK(x)                                                        \
  do {                                                               \
    hipError_t e_ = (x);                                             \
    if (e_ != hipSuccess) {                                          \
      printf("HIP error %s @%d\n", hipGetErrorString(e_), __LINE__); \
      return 1;                                                      \
    }                                                                \
  } while (0)

__global__ void fill_kernel(_Float16* p, size_t n, unsigned seed) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) {
    unsigned h = (unsigned)i * 2654435761u + seed;
    h ^= h >> 13;
    p[i] = (_Float16)(((float)(h & 1023) / 512.f) - 1.f);
  }
}
__global__ void fillf_kernel(float* p, size_t n, float v) {
  size_t i = (size_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = v;
}

int main(int argc, char** argv) {
  const int B = 1, S = (argc > 1) ? atoi(argv[1]) : 65536, N = 32, D = 128;
  using T = _Float16;
  const size_t nel = (size_t)B * S * N * D;
  T *q, *k, *v, *dout;
  float *delta, *lse, *dq, *dk;
  CK(hipMalloc(&q, nel * 2));
  CK(hipMalloc(&k, nel * 2));
  CK(hipMalloc(&v, nel * 2));
  CK(hipMalloc(&dout, nel * 2));
  CK(hipMalloc(&delta, (size_t)B * N * S * 4));
  CK(hipMalloc(&lse, (size_t)B * N * S * 4));
  CK(hipMalloc(&dq, nel * 4));
  CK(hipMalloc(&dk, nel * 4));
  int thr = 256;
  fill_kernel<<<(nel + thr - 1) / thr, thr>>>(q, nel, 1);
  fill_kernel<<<(nel + thr - 1) / thr, thr>>>(k, nel, 2);
  fill_kernel<<<(nel + thr - 1) / thr, thr>>>(v, nel, 3);
  fill_kernel<<<(nel + thr - 1) / thr, thr>>>(dout, nel, 4);
  size_t nml = (size_t)B * N * S;
  fillf_kernel<<<(nml + thr - 1) / thr, thr>>>(delta, nml, 0.5f);
  fillf_kernel<<<(nml + thr - 1) / thr, thr>>>(lse, nml, 11.f);
  CK(hipMemset(dq, 0, nel * 4));
  CK(hipMemset(dk, 0, nel * 4));
  CK(hipDeviceSynchronize());

  const float scale = 0.0883883f;
  int64_t s4[3] = {(int64_t)S * N * D, (int64_t)N * D, (int64_t)D};
  int64_t s2[2] = {(int64_t)N * S, (int64_t)S};
  dim3 gdkq((unsigned)N, (unsigned)((S + 255) / 256), 1);
  dim3 gkv((unsigned)((S + 255) / 256), (unsigned)N, 1);
  dim3 gdq((unsigned)((S + 255) / 256), (unsigned)N, 1);

  float* dv2;
  CK(hipMalloc(&dv2, nel * 4));
  CK(hipMemset(dv2, 0, nel * 4));
#define DKQ_A                                                              \
  dout, q, k, v, delta, lse, dq, dk, dv2, S, S, N, s4[0], s4[1], s4[2],    \
      s4[0], s4[1], s4[2], s4[0], s4[1], s4[2], s4[0], s4[1], s4[2],       \
      s2[0], s2[1], s2[0], s2[1], s4[0], s4[1], s4[2], s4[0], s4[1],       \
      s4[2], s4[0], s4[1], s4[2], scale, 0

  hipEvent_t e0, e1;
  CK(hipEventCreate(&e0));
  CK(hipEventCreate(&e1));
  auto bench = [&](const char* name, auto launch) {
    launch();  // warm
    (void)hipDeviceSynchronize();
    hipError_t le = hipGetLastError();
    if (le != hipSuccess) {
      printf("%-28s LAUNCH ERROR: %s\n", name, hipGetErrorString(le));
      return;
    }
    (void)hipEventRecord(e0);
    launch();
    launch();
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms = 0;
    (void)hipEventElapsedTime(&ms, e0, e1);
    printf("%-28s %8.2f ms\n", name, ms / 2);
  };

  bench("dkq FUSE0 (flip dK)", [&] {
    bwd_dkq_kernel<T, 128, 0, 0, 64><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkq FUSE1 DBG0 (full)", [&] {
    bwd_dkq_kernel<T, 128, 1, 0, 64><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkq FUSE1 DBG1 (noGlbAtom)", [&] {
    bwd_dkq_kernel<T, 128, 1, 1, 64><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkq FUSE1 DBG2 (dsWrite)", [&] {
    bwd_dkq_kernel<T, 128, 1, 2, 64><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkq FUSE1 DBG3 (no dQ sec)", [&] {
    bwd_dkq_kernel<T, 128, 1, 3, 64><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkq FUSE1 DBG4 (no KT rd)", [&] {
    bwd_dkq_kernel<T, 128, 1, 4, 64><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkq FUSE1 DBG5 (no reduce)", [&] {
    bwd_dkq_kernel<T, 128, 1, 5, 64><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkq FUSE1 DBG0 SCRW32", [&] {
    bwd_dkq_kernel<T, 128, 1, 0, 32><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkvf (fused dK+dV)", [&] {
    bwd_dkq_kernel<T, 128, 0, 0, 32, 1><<<gdkq, 512>>>(DKQ_A);
  });
  bench("dkv MODE1 (split dK)", [&] {
    bwd_dkv_kernel<T, 128, 1, 64><<<gkv, 512>>>(
        dout, q, k, v, delta, lse, dk, nullptr, S, S, N, s4[0], s4[1], s4[2],
        s4[0], s4[1], s4[2], s4[0], s4[1], s4[2], s4[0], s4[1], s4[2], s2[0],
        s2[1], s2[0], s2[1], s4[0], s4[1], s4[2], scale, 0);
  });
  bench("dkv MODE0 (dV)", [&] {
    bwd_dkv_kernel<T, 128, 0, 64><<<gkv, 512>>>(
        dout, q, k, v, delta, lse, dk, nullptr, S, S, N, s4[0], s4[1], s4[2],
        s4[0], s4[1], s4[2], s4[0], s4[1], s4[2], s4[0], s4[1], s4[2], s2[0],
        s2[1], s2[0], s2[1], s4[0], s4[1], s4[2], scale, 0);
  });
  bench("dq kernel DQP=1", [&] {
    bwd_dq_kernel<T, 128, 1><<<gdq, 512>>>(
        dout, q, k, v, delta, lse, dq, S, S, N, s4[0], s4[1], s4[2], s4[0],
        s4[1], s4[2], s4[0], s4[1], s4[2], s4[0], s4[1], s4[2], s2[0], s2[1],
        s2[0], s2[1], s4[0], s4[1], s4[2], scale, 0);
  });
  bench("dq kernel", [&] {
    bwd_dq_kernel<T, 128><<<gdq, 512>>>(
        dout, q, k, v, delta, lse, dq, S, S, N, s4[0], s4[1], s4[2], s4[0],
        s4[1], s4[2], s4[0], s4[1], s4[2], s4[0], s4[1], s4[2], s2[0], s2[1],
        s2[0], s2[1], s4[0], s4[1], s4[2], scale, 0);
  });
  return 0;
}
