// BLS batch-verify: prepare-side kernels (pk decompress, pubkey
// aggregation, sigma decompress, scalar-mult classes). Split TU
// (round 2): the one big TU compiled in ~19 min; three compile in
// parallel. Work layout + launcher seams: m3x_bls_common.hh.
#include "m3x_bls_common.hh"
#include <cstdio>
#include <cstdlib>

using namespace m3xb;

namespace {
__device__ void order_be_bytes(uint8_t be[32]) {
#pragma unroll
  for (int i = 0; i < 4; i++) {
    uint64_t limb = BLS_ORDER[3 - i];
#pragma unroll
    for (int j = 0; j < 8; j++) be[8 * i + j] = (uint8_t)(limb >> (56 - 8 * j));
  }
}

__global__ __launch_bounds__(64, 1) void k_bls_pk_decompress(const uint8_t *__restrict__ comp,
                                    uint64_t n, uint8_t *__restrict__ uncomp,
                                    int32_t *__restrict__ status) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  g1a p;
  if (g1_decompress(p, comp + 48 * i) != 0) {
    status[i] = -1;
    return;
  }
  if (p.inf) { // infinity pubkey rejected (generic_public_key.rs:86-94)
    status[i] = -2;
    return;
  }
  uint8_t be[32];
  order_be_bytes(be);
  g1j t;
  g1j_mul_be(t, p, be, 32);
  if (!g1j_is_inf(t)) {
    status[i] = -4; // subgroup check failed
    return;
  }
  g1_to_uncomp(p, uncomp + 96 * i);
  status[i] = 0;
}

// collect the k>1 sets (their aggregation runs wave-parallel)
__global__ void k_bls_scan_agg(const uint32_t *__restrict__ offs, uint64_t n,
                               BlsWork w) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  if (offs[i + 1] - offs[i] > 1) {
    uint32_t pos = atomicAdd(w.agg_count, 1u);
    w.agg_idx[pos] = i;
  }
}

// one WAVE per k>1 set: each lane partial-sums a strided slice of the
// set's pubkeys in Jacobian form, then an LDS tree reduce (6 levels)
__global__ __launch_bounds__(64, 1) void k_bls_aggregate_w(
    const uint8_t *__restrict__ pks, const uint32_t *__restrict__ offs,
    BlsWork w) {
  __shared__ g1j lds[64];
  // grid-stride over the aggregate list: the grid is FIXED (8192 blocks)
  // so a k=1-only workload costs ~nothing — launching n blocks of
  // early-exits measured 183 ms at n=1M (rocprof r02)
  for (uint32_t b = blockIdx.x; b < *w.agg_count; b += gridDim.x) {
  uint64_t set = w.agg_idx[b];
  uint32_t k0 = offs[set], k1 = offs[set + 1];
  int lane = threadIdx.x;
  g1j acc;
  fp_zero(acc.x);
  fp_zero(acc.y);
  fp_zero(acc.z);
  bool bad = false;
  for (uint32_t k = k0 + lane; k < k1; k += 64) {
    g1a pk;
    if (g1_from_uncomp_trusted(pk, pks + 96 * (uint64_t)k) != 0) {
      bad = true;
      break;
    }
    g1j_add_aff(acc, acc, pk);
  }
  if (bad) atomicOr(w.fail, 1);
  lds[lane] = acc;
  __syncthreads();
  for (int sft = 32; sft > 0; sft >>= 1) {
    if (lane < sft) {
      g1j t;
      g1j_add(t, lds[lane], lds[lane + sft]);
      lds[lane] = t;
    }
    __syncthreads();
  }
  if (lane == 0) w.apk[set] = lds[0];
  __syncthreads(); // next grid-stride iteration reuses lds
  }
}

// FUSED per-set prepare (kept for LARGE batches): at high occupancy the
// kernel is ISSUE-bound and the shared-doubling dual-scalar chain does
// less total work than the split form (rocprof r02g: 450 vs 657+116 ms
// at 1M sets); the split form wins in the small-batch LATENCY regime
// (32 vs 37 ms at 64k). Dispatch picks by n.
__global__ __launch_bounds__(64, 1) void k_bls_prepare(const uint8_t *__restrict__ sigs,
                              const uint8_t *__restrict__ pks,
                              const uint32_t *__restrict__ offs,
                              const uint64_t *__restrict__ rands, uint64_t n,
                              BlsWork w) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  g2a sig;
  if (g2_decompress(sig, sigs + 96 * i) != 0) {
    atomicOr(w.fail, 1);
    return;
  }
  uint32_t k0 = offs[i], k1 = offs[i + 1];
  if (k1 <= k0) {
    atomicOr(w.fail, 1);
    return;
  }
  g1j apk;
  if (k1 - k0 == 1) {
    g1a pk;
    if (g1_from_uncomp_trusted(pk, pks + 96 * (uint64_t)k0) != 0) {
      atomicOr(w.fail, 1);
      return;
    }
    g1j_from_aff(apk, pk);
  } else {
    apk = w.apk[i]; // precomputed by k_bls_aggregate_w
  }
  if (g1j_is_inf(apk)) { // aggregate at infinity -> invalid
    atomicOr(w.fail, 1);
    return;
  }
  uint8_t rbe[8];
#pragma unroll
  for (int b = 0; b < 8; b++) rbe[b] = (uint8_t)(rands[i] >> (56 - 8 * b));
  g1j rp;
  g1j_mul_be_j(rp, apk, rbe, 8); // P stays Jacobian end-to-end
  w.p_scaled[i] = rp;
  if (sig.inf) {
    // infinity is a valid subgroup element; contributes nothing
    fp2_zero(w.rsig[i].x);
    fp2_zero(w.rsig[i].y);
    fp2_zero(w.rsig[i].z);
  } else {
    // [r]sigma and the psi subgroup check's [|x|]sigma share sigma's
    // doubling chain (blst.rs:73-77 deferred subgroup check)
    g2j rsig_j, xsig_j;
    g2j_mul2_u64(rsig_j, xsig_j, sig, rands[i], BLS_X_ABS);
    w.rsig[i] = rsig_j;
    // psi(sigma) must equal -[|x|]sigma (x < 0): cross-multiplied compare
    g2a ps;
    psi_g2(ps, sig);
    bool ok;
    if (g2j_is_inf(xsig_j)) {
      ok = false;
    } else {
      fp2 z2, z3, lx, ly, ny;
      fp2_sqr(z2, xsig_j.z);
      fp2_mul(z3, z2, xsig_j.z);
      fp2_mul(lx, ps.x, z2);
      fp2_neg(ny, xsig_j.y);
      fp2_mul(ly, ps.y, z3);
      ok = fp2_eq(lx, xsig_j.x) && fp2_eq(ly, ny);
    }
    if (!ok) {
      atomicOr(w.fail, 1);
      return;
    }
  }
}

// [k]B for an affine G2 base with MIXED adds (24 vs 43 fp-muls per add);
// MSB-first double-and-add. Separate per-scalar chains with mixed adds
// cost LESS total than the round-1 shared-doubling chain of FULL adds.
__device__ inline void g2_mul_u64_aff(g2j &r, const g2a &base, uint64_t k) {
  g2j acc;
  fp2_zero(acc.x);
  fp2_zero(acc.y);
  fp2_zero(acc.z);
  for (int b = 63; b >= 0; b--) {
    g2j_dbl(acc, acc);
    if ((k >> b) & 1) g2j_add_aff(acc, acc, base);
  }
  r = acc;
}

// [k]B for an affine G1 base with mixed adds
__device__ inline void g1_mul_u64_aff(g1j &r, const g1a &base, uint64_t k) {
  g1j acc;
  fp_zero(acc.x);
  fp_zero(acc.y);
  fp_zero(acc.z);
  for (int b = 63; b >= 0; b--) {
    g1j_dbl(acc, acc);
    if ((k >> b) & 1) g1j_add_aff(acc, acc, base);
  }
  r = acc;
}

// LATENCY-REGIME register-budget twins: at n <= 2^18 these kernels run
// 1-2 waves/SIMD, so granting each wave the idle register file (512 or
// 256 VGPRs vs the ~130 the default allocation picks) trades nothing
// and removes scratch spill round-trips from the serial chains. The
// default-budget forms stay for the high-occupancy regime.
__global__ __launch_bounds__(64) __attribute__((amdgpu_waves_per_eu(1, 1)))
void k_bls_sigdec_lat(const uint8_t *__restrict__ sigs, uint64_t n,
                      BlsWork w) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  g2a sig;
  if (g2_decompress(sig, sigs + 96 * i) != 0) {
    atomicOr(w.fail, 1);
    sig.inf = 1;
  }
  g2j out;
  if (sig.inf) {
    fp2_zero(out.x);
    fp2_zero(out.y);
    fp2_zero(out.z);
  } else {
    out.x = sig.x;
    out.y = sig.y;
    fp2_one(out.z);
  }
  w.sig_aff[i] = out;
}

// THREE-class latency-regime mult pass (3n lanes): A [r]sigma,
// B [|x|]sigma + psi subgroup check, C [r]apk on G1. Same total work as
// the 2-class form but the critical lane shrinks from (G1 mult + x-mult)
// to max(one chain) and wave count rises 3n/64 — pure latency win for
// small/medium batches.
__global__ __launch_bounds__(64) __attribute__((amdgpu_waves_per_eu(2, 2)))
void k_bls_prep_mults3(const uint8_t *__restrict__ pks,
                       const uint32_t *__restrict__ offs,
                       const uint64_t *__restrict__ rands, uint64_t n,
                       BlsWork w) {
  uint64_t lane = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (lane >= 3 * n) return;
  int cls = lane < n ? 0 : (lane < 2 * n ? 1 : 2);
  uint64_t i = lane - (uint64_t)cls * n;
  if (cls == 0) { // A: rsig[i] = [r_i] sigma
    g2j sj = w.sig_aff[i];
    if (g2j_is_inf(sj)) {
      fp2_zero(w.rsig[i].x);
      fp2_zero(w.rsig[i].y);
      fp2_zero(w.rsig[i].z);
      return;
    }
    g2a sig;
    sig.x = sj.x;
    sig.y = sj.y;
    sig.inf = 0;
    g2j rs;
    g2_mul_u64_aff(rs, sig, rands[i]);
    w.rsig[i] = rs;
    return;
  }
  if (cls == 1) { // B: deferred psi subgroup check
    g2j sj = w.sig_aff[i];
    if (g2j_is_inf(sj)) return; // infinity is a valid element
    g2a sig;
    sig.x = sj.x;
    sig.y = sj.y;
    sig.inf = 0;
    g2j xsig_j;
    g2_mul_u64_aff(xsig_j, sig, BLS_X_ABS);
    g2a ps;
    psi_g2(ps, sig);
    bool ok;
    if (g2j_is_inf(xsig_j)) {
      ok = false;
    } else {
      fp2 z2, z3, lx, ly, ny;
      fp2_sqr(z2, xsig_j.z);
      fp2_mul(z3, z2, xsig_j.z);
      fp2_mul(lx, ps.x, z2);
      fp2_neg(ny, xsig_j.y);
      fp2_mul(ly, ps.y, z3);
      ok = fp2_eq(lx, xsig_j.x) && fp2_eq(ly, ny);
    }
    if (!ok) atomicOr(w.fail, 1);
    return;
  }
  // C: p_scaled[i] = [r_i] apk
  uint32_t k0 = offs[i], k1 = offs[i + 1];
  if (k1 <= k0) {
    atomicOr(w.fail, 1);
    return;
  }
  g1j rp;
  if (k1 - k0 == 1) {
    g1a pk;
    if (g1_from_uncomp_trusted(pk, pks + 96 * (uint64_t)k0) != 0) {
      atomicOr(w.fail, 1);
      return;
    }
    if (pk.inf) {
      atomicOr(w.fail, 1);
      return;
    }
    g1_mul_u64_aff(rp, pk, rands[i]);
  } else {
    g1j apk = w.apk[i];
    if (g1j_is_inf(apk)) {
      atomicOr(w.fail, 1);
      return;
    }
    uint8_t rbe[8];
#pragma unroll
    for (int b = 0; b < 8; b++)
      rbe[b] = (uint8_t)(rands[i] >> (56 - 8 * b));
    g1j_mul_be_j(rp, apk, rbe, 8);
  }
  w.p_scaled[i] = rp;
}

} // namespace

namespace m3xk {

void launch_pk_decompress(hipStream_t s, const uint8_t *comp_d, uint64_t n,
                          uint8_t *unc_d, int32_t *st_d) {
  uint32_t blocks = (uint32_t)((n + 63) / 64);
  hipLaunchKernelGGL(k_bls_pk_decompress, dim3(blocks), dim3(64), 0, s,
                     comp_d, n, unc_d, st_d);
}

void launch_aggregate(hipStream_t s, const uint8_t *pks_dev,
                      const uint32_t *offs_dev, uint64_t n, BlsWork w) {
  uint32_t blocks = (uint32_t)((n + 63) / 64);
  hipLaunchKernelGGL(k_bls_scan_agg, dim3(blocks), dim3(64), 0, s, offs_dev,
                     n, w);
  uint32_t agg_blocks = n < 32768 ? (uint32_t)n : 32768;
  hipLaunchKernelGGL(k_bls_aggregate_w, dim3(agg_blocks), dim3(64), 0, s,
                     pks_dev, offs_dev, w);
}

void launch_prepare(hipStream_t s, const uint8_t *sigs_dev,
                    const uint8_t *pks_dev, const uint32_t *offs_dev,
                    const uint64_t *rands_dev, uint64_t n, BlsWork w) {
  uint32_t blocks = (uint32_t)((n + 63) / 64);
  if (n > (1ull << 18)) {
    // issue-bound regime: the fused shared-chain kernel does less work
    hipLaunchKernelGGL(k_bls_prepare, dim3(blocks), dim3(64), 0, s,
                       sigs_dev, pks_dev, offs_dev, rands_dev, n, w);
  } else {
    // latency regime: decompress pass + three wave-uniform mult classes
    hipLaunchKernelGGL(k_bls_sigdec_lat, dim3(blocks), dim3(64), 0, s,
                       sigs_dev, n, w);
    uint32_t blocks3 = (uint32_t)((3 * n + 63) / 64);
    hipLaunchKernelGGL(k_bls_prep_mults3, dim3(blocks3), dim3(64), 0, s,
                       pks_dev, offs_dev, rands_dev, n, w);
  }
}

} // namespace m3xk
