// BLS batch-verify: GT/sig reductions + the cooperative finish
// (final exponentiation) kernel. Split TU — see m3x_bls_common.hh.
#include "m3x_bls_common.hh"
#include <cstdio>
#include <cstdlib>

using namespace m3xb;

namespace {
// two-stage GT-product reduction: each block folds its contiguous span of
// per-set miller values (thread-strided local products, then an LDS tree)
// into one output element; a second 1-block launch folds the partials.
__global__ __launch_bounds__(256) void k_bls_reduce_gt(
    const fp12m *__restrict__ in, uint64_t n, fp12m *__restrict__ out) {
  __shared__ fp12m lds[256];
  uint64_t per = (n + gridDim.x - 1) / gridDim.x;
  uint64_t lo = (uint64_t)blockIdx.x * per;
  uint64_t hi = lo + per < n ? lo + per : n;
  fp12m local, t;
  f12_one(local);
  for (uint64_t i = lo + threadIdx.x; i < hi; i += 256) {
    f12_mul_nn(t, local, in[i]);
    f12_copy(local, t);
  }
  lds[threadIdx.x] = local;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if ((int)threadIdx.x < s) {
      f12_mul_nn(t, lds[threadIdx.x], lds[threadIdx.x + s]);
      f12_copy(lds[threadIdx.x], t);
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) out[blockIdx.x] = lds[0];
}

// two-stage sum of r_i*sigma_i, same shape
__global__ __launch_bounds__(256) void k_bls_reduce_sig(
    const g2j *__restrict__ in, uint64_t n, g2j *__restrict__ out) {
  __shared__ g2j lds[256];
  uint64_t per = (n + gridDim.x - 1) / gridDim.x;
  uint64_t lo = (uint64_t)blockIdx.x * per;
  uint64_t hi = lo + per < n ? lo + per : n;
  g2j local;
  fp2_zero(local.x);
  fp2_zero(local.y);
  fp2_zero(local.z);
  for (uint64_t i = lo + threadIdx.x; i < hi; i += 256)
    g2j_add(local, local, in[i]);
  lds[threadIdx.x] = local;
  __syncthreads();
  for (int s = 128; s > 0; s >>= 1) {
    if ((int)threadIdx.x < s) {
      g2j t;
      g2j_add(t, lds[threadIdx.x], lds[threadIdx.x + s]);
      lds[threadIdx.x] = t;
    }
    __syncthreads();
  }
  if (threadIdx.x == 0) out[blockIdx.x] = lds[0];
}

// final: f_total *= miller(-g1, sig_sum); final_exp; compare to one.
// One wave, cooperative fp12 ops in LDS (the per-batch serial tail: the
// 36 coefficient products of each Fp12 multiply fan across lanes).
__global__ __launch_bounds__(64) void k_bls_finish(BlsWork w) {
  __shared__ fp12m sh[7];
  __shared__ f12w_ws ws;
  __shared__ miller_ws mws;
  int lane = threadIdx.x;
  if (*w.fail) {
    if (lane == 0) *w.verdict = 0;
    return;
  }
  g1j ng1;
  {
    g1a g;
    g1_gen(g);
    ng1.x = g.x;
    fp_neg(ng1.y, g.y);
    fp_one(ng1.z);
  }
  if (lane == 0) f12_copy(sh[0], w.gt_parts[0]);
  f12w_sync();
  // sig_sum stays Jacobian: the Q-Jacobian Miller loop needs no inversion
  miller_w(sh[1], ng1, w.sig_sum[0], ws, mws, lane);
  f12_mul_w(sh[1], sh[0], sh[1], ws, lane); // f_total
  final_exp_w(sh[2], sh[1], &sh[3], ws, lane);
  if (lane == 0) *w.verdict = f12_is_one(sh[2]) ? 1 : 0;
}

} // namespace

namespace m3xk {

void launch_reduce(hipStream_t s, uint64_t n_parts, uint64_t n, BlsWork w) {
  uint32_t rblocks = (uint32_t)((n_parts + 255) / 256);
  if (rblocks > 256) rblocks = 256;
  hipLaunchKernelGGL(k_bls_reduce_gt, dim3(rblocks), dim3(256), 0, s,
                     w.fparts, n_parts, w.gt_stage);
  hipLaunchKernelGGL(k_bls_reduce_gt, dim3(1), dim3(256), 0, s, w.gt_stage,
                     (uint64_t)rblocks, w.gt_parts);
  uint32_t sblocks = (uint32_t)((n + 255) / 256);
  if (sblocks > 256) sblocks = 256;
  hipLaunchKernelGGL(k_bls_reduce_sig, dim3(sblocks), dim3(256), 0, s,
                     w.rsig, n, w.sig_stage);
  hipLaunchKernelGGL(k_bls_reduce_sig, dim3(1), dim3(256), 0, s, w.sig_stage,
                     (uint64_t)sblocks, w.sig_sum);
}

void launch_finish(hipStream_t s, BlsWork w) {
  hipLaunchKernelGGL(k_bls_finish, dim3(1), dim3(64), 0, s, w);
}

} // namespace m3xk
