// BLS batch-verify: Miller-loop kernels (per-lane, wave-split,
// cooperative small-batch). Split TU — see m3x_bls_common.hh.
#include "m3x_bls_common.hh"
#include <cstdio>
#include <cstdlib>

using namespace m3xb;

namespace {
// small-batch variant: ONE WAVE PER SET (cooperative miller_w). At tiny n
// (block import: ~131 sets) the per-lane kernel is latency-bound — a
// single set's serial Miller loop is tens of ms on one lane — while n
// cooperative waves spread across 256 CUs cut that ~10x. Crossover is
// empirical (M3X_SMALL_MILLER sets the threshold).
__global__ __launch_bounds__(64) void k_bls_miller_small(uint64_t n,
                                                         BlsWork w) {
  __shared__ fp12m f;
  __shared__ f12w_ws ws;
  __shared__ miller_ws mws;
  uint64_t i = blockIdx.x;
  int lane = threadIdx.x;
  if (i >= n) return;
  if (*w.fail) {
    if (lane == 0) f12_one(w.fparts[i]);
    return;
  }
  miller_w(f, w.p_scaled[i], w.h2c[i], ws, mws, lane);
  __syncthreads();
  if (lane == 0) w.fparts[i] = f;
}

__global__ __launch_bounds__(64, 1) void k_bls_miller(uint64_t n, BlsWork w) {
  // fp12 state stays in thread-local scratch: an LDS-resident variant
  // measured 2x SLOWER (123ms vs 63ms on C2) — the L1/L2-cached spill
  // traffic beats per-limb ds_read latency for this access pattern.
  // (An in-wave LDS tree fold of the 64 per-set values also regressed:
  // +5ms in the kernel and the 64x-smaller reduce became LATENCY-bound
  // at 4 blocks — reverted; measured round 2.)
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  fp12m f, tmp;
  if (*w.fail == 0)
    miller_raw(f, tmp, w.p_scaled[i], w.h2c[i]);
  else
    f12_one(f);
  w.fparts[i] = f;
}

// WAVE-SPLIT variant (round 2): 2n lanes — lane i = set i's high bit-half,
// lane n+i = set i's low half. Doubles the wave count (2/SIMD at the C2
// 64k-set shape) and halves each lane's dependent chain; the 2n partial
// products feed the same GT reduction (their product = the n full
// Millers' product). Each wave is role-uniform: no intra-wave divergence.
__global__ __launch_bounds__(64, 2)
__attribute__((amdgpu_waves_per_eu(2))) void k_bls_miller_split(
    uint64_t n, BlsWork w) {
  // launch_bounds min-waves 2: cap the allocation at 256 VGPRs so the two
  // half-Miller waves of a SIMD actually co-reside (the whole point of
  // the split; at the default the allocator takes 511 -> 1 wave/SIMD)
  uint64_t lane = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (lane >= 2 * n) return;
  int role = lane < n ? 0 : 1;
  uint64_t i = role == 0 ? lane : lane - n;
  fp12m f;
  if (*w.fail == 0)
    miller_half(f, w.p_scaled[i], w.h2c[i], role);
  else
    f12_one(f);
  w.fparts[lane] = f;
}

} // namespace

namespace m3xk {

// returns the number of fp12 partials written (n, or 2n for the split)
uint64_t launch_miller(hipStream_t s, uint64_t n, BlsWork w) {
  uint64_t small_thresh = 2048; // measured crossover (tmp_bench/c4probe)
  if (const char *e = getenv("M3X_SMALL_MILLER"))
    small_thresh = strtoull(e, nullptr, 10);
  static int use_split = -1;
  if (use_split < 0) {
    const char *e = getenv("M3X_MILLER_SPLIT");
    use_split = (e && e[0] == '1') ? 1 : 0; // measured SLOWER at C2 (57
    // vs 44 ms: the 256-VGPR cap spills more than co-residency saves)
  }
  uint32_t blocks = (uint32_t)((n + 63) / 64);
  if (n <= small_thresh) {
    hipLaunchKernelGGL(k_bls_miller_small, dim3((uint32_t)n), dim3(64), 0, s,
                       n, w);
    return n;
  }
  if (use_split) {
    uint32_t blocks2 = (uint32_t)((2 * n + 63) / 64);
    hipLaunchKernelGGL(k_bls_miller_split, dim3(blocks2), dim3(64), 0, s, n,
                       w);
    return 2 * n;
  }
  hipLaunchKernelGGL(k_bls_miller, dim3(blocks), dim3(64), 0, s, n, w);
  return n;
}

} // namespace m3xk
