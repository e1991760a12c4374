// MI355X-native batched BLS12-381 signature-set verification (hot path #1).
//
// Implements the blst.rs:37-119 contract (see include/m3x_consensus.h and
// DESIGN.md): per set — decompress + subgroup-check sigma (deferred checks
// exactly as generic_aggregate_signature.rs:161-176 + blst.rs:73-77),
// aggregate the set's pubkeys, scale by the host-drawn 64-bit r_i, hash the
// message to G2 (RFC 9380), accumulate one Miller loop per set; then one
// extra pair e(-g1, sum r_i sigma_i), one final exponentiation, compare to
// one. CDNA4 shape: one set per lane, kernels split by divergence class,
// grids ≫256 workgroups; integer VALU workload (no MFMA).
#include "bls_device.hh"
#include "m3x_ctx.hh"
#include "../../include/m3x_consensus.h"
#include <cstdio>
#include <cstdlib>

// M3X_DEBUG_SYNC=1: synchronize + trace after each BLS kernel (debug aid)
static bool dbg_sync() {
  static int v = -1;
  if (v < 0) {
    const char *e = getenv("M3X_DEBUG_SYNC");
    v = (e && e[0] == '1') ? 1 : 0;
  }
  return v == 1;
}
#define DBG_STEP(ctx, name)                                                      do {                                                                             if (dbg_sync()) {                                                                hipError_t _e = hipStreamSynchronize((ctx)->stream);                           fprintf(stderr, "[m3x dbg] %s: %s\n", name, hipGetErrorString(_e));            fflush(stderr);                                                              }                                                                            } while (0)

using namespace m3xb;

namespace {

struct BlsWork {
  g1j *apk;        // [n] precomputed aggregate pubkeys (k>1 sets)
  uint64_t *agg_idx; // [n] indices of k>1 sets (count in agg_count[0])
  uint32_t *agg_count;
  g1j *p_scaled;   // [n] r_i * aggregate pubkey (Jacobian)
  g2j *h2c;        // [n] hash_to_curve(msg), Jacobian (no inversion)
  uint8_t *uni;    // [n*256] expand_message_xmd output (h2c pass 1)
  g2j *h2c_pts;    // [2n] per-point sswu+iso outputs (h2c pass 2)
  g2j *rsig;       // [n] r_i * sigma (jacobian)
  g2j *sig_aff;    // [n] decompressed sigma (z=1 affine; z=0 infinity)
  fp12m *fparts;   // [n] per-set miller values
  int *fail;       // [1]
  g2j *sig_stage;  // [256] stage-1 partial sums
  g2j *sig_sum;    // [1]
  fp12m *gt_stage; // [256] stage-1 partial products
  fp12m *gt_parts; // [1]
  int *verdict;    // [1]
};

__device__ void order_be_bytes(uint8_t be[32]) {
#pragma unroll
  for (int i = 0; i < 4; i++) {
    uint64_t limb = BLS_ORDER[3 - i];
#pragma unroll
    for (int j = 0; j < 8; j++) be[8 * i + j] = (uint8_t)(limb >> (56 - 8 * j));
  }
}

// ---------------------------------------------------------------- kernels

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


// ---- WAVE-SPLIT prepare (round 2; was one ~30ms/lane serial kernel) ----
// pass 1: decompress sigma (n lanes; the sqrt pow chain dominates)
__global__ __launch_bounds__(64, 1) void k_bls_sigdec(
    const uint8_t *__restrict__ sigs, uint64_t n, BlsWork w) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  g2a sig;
  if (g2_decompress(sig, sigs + 96 * i) != 0) {
    atomicOr(w.fail, 1);
    sig.inf = 1; // harmless placeholder; verdict is already forced false
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

// pass 2: the scalar-mult work, TWO wave-uniform classes over 2n lanes
// (2 waves/SIMD at the C2 shape):
//   class A (lane i):   rsig[i] = [r_i] sigma_i
//   class B (lane n+i): p_scaled[i] = [r_i] apk_i  AND the deferred
//                       psi-subgroup check psi(sigma) == -[|x|]sigma
//                       (blst.rs:73-77)
__global__ __launch_bounds__(64, 1) void k_bls_prep_mults(
    const uint8_t *__restrict__ pks, const uint32_t *__restrict__ offs,
    const uint64_t *__restrict__ rands, uint64_t n, BlsWork w) {
  uint64_t lane = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (lane >= 2 * n) return;
  if (lane < n) {
    uint64_t i = lane;
    g2j sj = w.sig_aff[i];
    if (g2j_is_inf(sj)) {
      // infinity contributes nothing to the signature sum
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
  uint64_t i = lane - n;
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
    if (pk.inf) { // aggregate at infinity -> invalid
      atomicOr(w.fail, 1);
      return;
    }
    g1_mul_u64_aff(rp, pk, rands[i]); // mixed adds on the affine base
  } else {
    g1j apk = w.apk[i]; // precomputed by k_bls_aggregate_w
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
  // deferred subgroup check (skipped for infinity: valid element)
  g2j sj = w.sig_aff[i];
  if (!g2j_is_inf(sj)) {
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
  }
}

__global__ __launch_bounds__(64, 1) void k_bls_h2c(const uint8_t *__restrict__ msgs, uint64_t n,
                          BlsWork w) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  m3xb::h2c_g2(w.h2c[i], msgs + 32 * i);
}

// ---- WAVE-SPLIT h2c (round 2): three passes so the heavy SSWU work runs
// at 2n lanes (2 waves/SIMD at the C2 shape) and each lane's dependent
// chain is one point, not two ----
__global__ __launch_bounds__(64, 1) void k_bls_h2c_expand(
    const uint8_t *__restrict__ msgs, uint64_t n, BlsWork w) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  expand_message_xmd32(msgs + 32 * i, w.uni + 256 * i);
}

__global__ __launch_bounds__(64, 2) void k_bls_h2c_map(uint64_t n,
                                                       BlsWork w) {
  uint64_t lane = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (lane >= 2 * n) return;
  int role = lane < n ? 0 : 1; // which of the two RO points
  uint64_t i = role == 0 ? lane : lane - n;
  const uint8_t *uni = w.uni + 256 * i + 128 * role;
  fp2 u;
  h2f_from_be64(u.c0, uni);
  h2f_from_be64(u.c1, uni + 64);
  g2a q;
  sswu_g2(q, u);
  g2j pt, cleared;
  iso_map_g2_j(pt, q);
  // cofactor clearing is a homomorphism (h_eff scalar mult + psi), so
  // clearing each RO point separately at 2n lanes (2 waves/SIMD hiding
  // the dependent dbl-chain latency) and summing afterwards equals
  // clearing the sum — the fin pass is then one mixed add per set
  clear_cofactor_g2j(cleared, pt);
  w.h2c_pts[lane] = cleared;
}

__global__ __launch_bounds__(64, 1) void k_bls_h2c_fin(uint64_t n,
                                                       BlsWork w) {
  uint64_t i = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= n) return;
  g2j s = w.h2c_pts[i];
  g2j_add(s, s, w.h2c_pts[n + i]);
  w.h2c[i] = s;
}

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

// one wave aggregates n compressed signatures: lane-strided decompress +
// Jacobian partial sums, LDS tree reduce, lane 0 compresses the total
__global__ __launch_bounds__(64, 1) void k_bls_sig_aggregate_w(
    const uint8_t *__restrict__ sigs, uint64_t n, uint8_t *__restrict__ out,
    int *__restrict__ fail) {
  __shared__ g2j lds[64];
  int lane = threadIdx.x;
  g2j acc;
  fp2_zero(acc.x);
  fp2_zero(acc.y);
  fp2_zero(acc.z);
  bool bad = false;
  for (uint64_t i = lane; i < n; i += 64) {
    g2a sg;
    if (g2_decompress(sg, sigs + 96 * i) != 0) {
      bad = true;
      break;
    }
    if (!sg.inf) {
      g2j sj;
      g2j_from_aff(sj, sg);
      g2j_add(acc, acc, sj);
    }
  }
  if (bad) atomicOr(fail, 1);
  lds[lane] = acc;
  __syncthreads();
  for (int sft = 32; sft > 0; sft >>= 1) {
    if (lane < sft) {
      g2j t;
      g2j_add(t, lds[lane], lds[lane + sft]);
      lds[lane] = t;
    }
    __syncthreads();
  }
  if (lane == 0 && !*fail) {
    g2a a;
    g2j_to_aff(a, lds[0]);
    // compress (ZCash): c1 || c0 big-endian, flags in byte 0
    if (a.inf) {
      for (int i = 0; i < 96; i++) out[i] = 0;
      out[0] = 0xC0;
    } else {
      fp_to_be48(a.x.c1, out);
      fp_to_be48(a.x.c0, out + 48);
      out[0] |= 0x80;
      if (fp2_gt_half_lex(a.y)) out[0] |= 0x20;
    }
  }
}

// one-thread test kernel: RFC 9380 h2c with an arbitrary DST, affine
// uncompressed output — pins the device sswu/iso/cofactor code against
// the literal RFC vectors (tests/golden/rfc9380_vectors.json)
__global__ void k_bls_h2c_dst(const uint8_t *__restrict__ msg,
                              uint32_t msg_len,
                              const uint8_t *__restrict__ dst,
                              uint32_t dst_len, uint8_t *__restrict__ out,
                              uint8_t *__restrict__ uni_out) {
  if (blockIdx.x != 0 || threadIdx.x != 0) return;
  uint8_t uni[256];
  expand_message_xmd_gen(msg, msg_len, dst, dst_len, 256, uni);
  for (int i = 0; i < 256; i++) uni_out[i] = uni[i];
  g2j h;
  h2c_g2_from_uniform(h, uni);
  g2a a;
  g2j_to_aff(a, h);
  g2_to_uncomp_dev(a, out);
}

// one-thread test kernel: general expand_message_xmd only
__global__ void k_bls_expand_dst(const uint8_t *__restrict__ msg,
                                 uint32_t msg_len,
                                 const uint8_t *__restrict__ dst,
                                 uint32_t dst_len, uint32_t len_in_bytes,
                                 uint8_t *__restrict__ out) {
  if (blockIdx.x != 0 || threadIdx.x != 0) return;
  expand_message_xmd_gen(msg, msg_len, dst, dst_len, len_in_bytes, out);
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

__global__ __launch_bounds__(64) __attribute__((amdgpu_waves_per_eu(2, 2)))
void k_bls_prep_mults_lat(const uint8_t *__restrict__ pks,
                          const uint32_t *__restrict__ offs,
                          const uint64_t *__restrict__ rands, uint64_t n,
                          BlsWork w) {
  uint64_t lane = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (lane >= 2 * n) return;
  if (lane < n) {
    uint64_t i = lane;
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
  uint64_t i = lane - n;
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
  g2j sj = w.sig_aff[i];
  if (!g2j_is_inf(sj)) {
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
  }
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

__global__ __launch_bounds__(64) __attribute__((amdgpu_waves_per_eu(2, 2)))
void k_bls_h2c_map_lat(uint64_t n, BlsWork w) {
  uint64_t lane = (uint64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (lane >= 2 * n) return;
  int role = lane < n ? 0 : 1;
  uint64_t i = role == 0 ? lane : lane - n;
  const uint8_t *uni = w.uni + 256 * i + 128 * role;
  fp2 u;
  h2f_from_be64(u.c0, uni);
  h2f_from_be64(u.c1, uni + 64);
  g2a q;
  sswu_g2(q, u);
  g2j pt, cleared;
  iso_map_g2_j(pt, q);
  clear_cofactor_g2j(cleared, pt);
  w.h2c_pts[lane] = cleared;
}

int run_verify(m3x_ctx *ctx, const void *msgs_dev, const void *sigs_dev,
               const void *pks_dev, const void *offs_dev,
               const void *rands_dev, uint64_t n, int32_t *out) {
  BlsWork w;
  uint64_t bytes = 0;
  auto align = [](uint64_t x) { return (x + 255) & ~255ull; };
  uint64_t off_apk = bytes; bytes += align(n * sizeof(g1j));
  uint64_t off_aggidx = bytes; bytes += align(n * 8);
  uint64_t off_aggcnt = bytes; bytes += 256;
  uint64_t off_p = bytes; bytes += align(n * sizeof(g1j));
  uint64_t off_h = bytes; bytes += align(n * sizeof(g2j));
  uint64_t off_uni = bytes; bytes += align(n * 256);
  uint64_t off_hpts = bytes; bytes += align(2 * n * sizeof(g2j));
  uint64_t off_r = bytes; bytes += align(n * sizeof(g2j));
  uint64_t off_sa = bytes; bytes += align(n * sizeof(g2j));
  uint64_t off_f = bytes; bytes += align(2 * n * sizeof(fp12m)); // split miller: 2n partials
  uint64_t off_fail = bytes; bytes += 256;
  uint64_t off_sig_stage = bytes; bytes += align(256 * sizeof(g2j));
  uint64_t off_sum = bytes; bytes += align(sizeof(g2j));
  uint64_t off_gt_stage = bytes; bytes += align(256 * sizeof(fp12m));
  uint64_t off_gt = bytes; bytes += align(sizeof(fp12m));
  uint64_t off_v = bytes; bytes += 256;
  int rc = m3x::ensure_scratch(ctx, &ctx->scratch_a, &ctx->scratch_a_bytes,
                               bytes);
  if (rc != M3X_OK) return rc;
  uint8_t *base = ctx->scratch_a;
  w.apk = reinterpret_cast<g1j *>(base + off_apk);
  w.agg_idx = reinterpret_cast<uint64_t *>(base + off_aggidx);
  w.agg_count = reinterpret_cast<uint32_t *>(base + off_aggcnt);
  w.p_scaled = reinterpret_cast<g1j *>(base + off_p);
  w.h2c = reinterpret_cast<g2j *>(base + off_h);
  w.uni = base + off_uni;
  w.h2c_pts = reinterpret_cast<g2j *>(base + off_hpts);
  w.rsig = reinterpret_cast<g2j *>(base + off_r);
  w.sig_aff = reinterpret_cast<g2j *>(base + off_sa);
  w.fparts = reinterpret_cast<fp12m *>(base + off_f);
  w.fail = reinterpret_cast<int *>(base + off_fail);
  w.sig_stage = reinterpret_cast<g2j *>(base + off_sig_stage);
  w.sig_sum = reinterpret_cast<g2j *>(base + off_sum);
  w.gt_stage = reinterpret_cast<fp12m *>(base + off_gt_stage);
  w.gt_parts = reinterpret_cast<fp12m *>(base + off_gt);
  w.verdict = reinterpret_cast<int *>(base + off_v);
  M3X_HIP_CHECK(hipMemsetAsync(w.fail, 0, 4, ctx->stream));
  M3X_HIP_CHECK(hipMemsetAsync(w.agg_count, 0, 4, ctx->stream));
  uint32_t blocks = (uint32_t)((n + 63) / 64);
  m3x::time_begin(ctx, M3X_K_BLS_AGG);
  hipLaunchKernelGGL(k_bls_scan_agg, dim3(blocks), dim3(64), 0, ctx->stream,
                     (const uint32_t *)offs_dev, n, w);
  {
    uint32_t agg_blocks = n < 32768 ? (uint32_t)n : 32768;
    hipLaunchKernelGGL(k_bls_aggregate_w, dim3(agg_blocks), dim3(64), 0,
                       ctx->stream, (const uint8_t *)pks_dev,
                       (const uint32_t *)offs_dev, w);
  }
  m3x::time_end(ctx, M3X_K_BLS_AGG);
  m3x::time_begin(ctx, M3X_K_BLS_PREPARE);
  if (n > (1ull << 18)) {
    // issue-bound regime: the fused shared-chain kernel does less work
    hipLaunchKernelGGL(k_bls_prepare, dim3(blocks), dim3(64), 0,
                       ctx->stream, (const uint8_t *)sigs_dev,
                       (const uint8_t *)pks_dev, (const uint32_t *)offs_dev,
                       (const uint64_t *)rands_dev, n, w);
  } else {
    // latency regime: decompress pass + two wave-uniform mult classes,
    // full-register-budget twins
    hipLaunchKernelGGL(k_bls_sigdec_lat, dim3(blocks), dim3(64), 0,
                       ctx->stream, (const uint8_t *)sigs_dev, n, w);
    uint32_t blocks3 = (uint32_t)((3 * n + 63) / 64);
    hipLaunchKernelGGL(k_bls_prep_mults3, dim3(blocks3), dim3(64), 0,
                       ctx->stream, (const uint8_t *)pks_dev,
                       (const uint32_t *)offs_dev,
                       (const uint64_t *)rands_dev, n, w);
  }
  m3x::time_end(ctx, M3X_K_BLS_PREPARE);
  DBG_STEP(ctx, "prepare");
  // h2c is independent of prepare: run it on the second stream so the two
  // ~1-wave/SIMD kernels co-reside (both fit at 2 waves/SIMD by VGPR count)
  m3x::time_begin_s(ctx, M3X_K_BLS_H2C, ctx->stream2);
  if (n > 64) { // split h2c pays at small n too (per-lane chain is ONE
                // point + clear, vs two points + clear in the fused form)
    uint32_t blocks2 = (uint32_t)((2 * n + 63) / 64);
    hipLaunchKernelGGL(k_bls_h2c_expand, dim3(blocks), dim3(64), 0,
                       ctx->stream2, (const uint8_t *)msgs_dev, n, w);
    if (n <= (1ull << 18))
      hipLaunchKernelGGL(k_bls_h2c_map_lat, dim3(blocks2), dim3(64), 0,
                         ctx->stream2, n, w);
    else
      hipLaunchKernelGGL(k_bls_h2c_map, dim3(blocks2), dim3(64), 0,
                         ctx->stream2, n, w);
    hipLaunchKernelGGL(k_bls_h2c_fin, dim3(blocks), dim3(64), 0,
                       ctx->stream2, n, w);
  } else {
    hipLaunchKernelGGL(k_bls_h2c, dim3(blocks), dim3(64), 0, ctx->stream2,
                       (const uint8_t *)msgs_dev, n, w);
  }
  m3x::time_end_s(ctx, M3X_K_BLS_H2C, ctx->stream2);
  M3X_HIP_CHECK(hipEventRecord(ctx->ev_s2, ctx->stream2));
  M3X_HIP_CHECK(hipStreamWaitEvent(ctx->stream, ctx->ev_s2, 0));
  DBG_STEP(ctx, "h2c");
  m3x::time_begin(ctx, M3X_K_BLS_MILLER);
  // small batches are latency-bound on the per-lane kernel: go wave-per-set
  uint64_t small_thresh = 2048; // measured crossover (tmp_bench/c4probe)
  if (const char *e = getenv("M3X_SMALL_MILLER")) small_thresh = strtoull(e, nullptr, 10);
  uint64_t n_parts = n; // fp12 partials feeding the GT reduce
  static int use_split = -1;
  if (use_split < 0) {
    const char *e = getenv("M3X_MILLER_SPLIT");
    use_split = (e && e[0] == '1') ? 1 : 0; // measured SLOWER at C2 (57
    // vs 44 ms: the 256-VGPR cap spills more than co-residency saves)
  }
  if (n <= small_thresh) {
    hipLaunchKernelGGL(k_bls_miller_small, dim3((uint32_t)n), dim3(64), 0,
                       ctx->stream, n, w);
  } else if (use_split) {
    n_parts = 2 * n; // wave-split: two half-Millers per set
    uint32_t blocks2 = (uint32_t)((2 * n + 63) / 64);
    hipLaunchKernelGGL(k_bls_miller_split, dim3(blocks2), dim3(64), 0,
                       ctx->stream, n, w);
  } else {
    hipLaunchKernelGGL(k_bls_miller, dim3(blocks), dim3(64), 0, ctx->stream,
                       n, w);
  }
  m3x::time_end(ctx, M3X_K_BLS_MILLER);
  DBG_STEP(ctx, "miller");
  uint32_t rblocks = (uint32_t)((n_parts + 255) / 256);
  if (rblocks > 256) rblocks = 256;
  m3x::time_begin(ctx, M3X_K_BLS_REDUCE);
  hipLaunchKernelGGL(k_bls_reduce_gt, dim3(rblocks), dim3(256), 0,
                     ctx->stream, w.fparts, n_parts, w.gt_stage);
  hipLaunchKernelGGL(k_bls_reduce_gt, dim3(1), dim3(256), 0, ctx->stream,
                     w.gt_stage, (uint64_t)rblocks, w.gt_parts);
  hipLaunchKernelGGL(k_bls_reduce_sig, dim3(rblocks), dim3(256), 0,
                     ctx->stream, w.rsig, n, w.sig_stage);
  hipLaunchKernelGGL(k_bls_reduce_sig, dim3(1), dim3(256), 0, ctx->stream,
                     w.sig_stage, (uint64_t)rblocks, w.sig_sum);
  m3x::time_end(ctx, M3X_K_BLS_REDUCE);
  DBG_STEP(ctx, "reduce");
  m3x::time_begin(ctx, M3X_K_BLS_FINISH);
  hipLaunchKernelGGL(k_bls_finish, dim3(1), dim3(64), 0, ctx->stream, w);
  m3x::time_end(ctx, M3X_K_BLS_FINISH);
  DBG_STEP(ctx, "finish");
  int32_t verdict = 0;
  M3X_HIP_CHECK(hipMemcpyAsync(&verdict, w.verdict, 4, hipMemcpyDeviceToHost,
                               ctx->stream));
  M3X_HIP_CHECK(hipStreamSynchronize(ctx->stream));
  *out = verdict;
  return M3X_OK;
}

} // namespace

extern "C" {

int32_t m3x_bls_pk_decompress(m3x_ctx *ctx, const uint8_t *comp, uint64_t n,
                              uint8_t *uncomp, int32_t *status) {
  if (!ctx || n == 0) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  // single cleanup path: partial allocation failure frees prior buffers,
  // and every copy is checked (a failed upload must not run the kernel on
  // stale device memory and report "invalid")
  uint8_t *comp_d = nullptr, *unc_d = nullptr;
  int32_t *st_d = nullptr;
  int32_t rc = M3X_ERR_HIP;
  if (hipMalloc(&comp_d, n * 48) != hipSuccess) goto out;
  if (hipMalloc(&unc_d, n * 96) != hipSuccess) goto out;
  if (hipMalloc(&st_d, n * 4) != hipSuccess) goto out;
  if (hipMemcpyAsync(comp_d, comp, n * 48, hipMemcpyHostToDevice,
                     ctx->stream) != hipSuccess)
    goto out;
  {
    uint32_t blocks = (uint32_t)((n + 63) / 64);
    hipLaunchKernelGGL(k_bls_pk_decompress, dim3(blocks), dim3(64), 0,
                       ctx->stream, comp_d, n, unc_d, st_d);
  }
  if (hipMemcpyAsync(uncomp, unc_d, n * 96, hipMemcpyDeviceToHost,
                     ctx->stream) != hipSuccess)
    goto out;
  if (hipMemcpyAsync(status, st_d, n * 4, hipMemcpyDeviceToHost,
                     ctx->stream) != hipSuccess)
    goto out;
  if (hipStreamSynchronize(ctx->stream) == hipSuccess) rc = M3X_OK;
out:
  if (comp_d) (void)hipFree(comp_d);
  if (unc_d) (void)hipFree(unc_d);
  if (st_d) (void)hipFree(st_d);
  return rc;
}

int32_t m3x_bls_expand_test(m3x_ctx *ctx, const uint8_t *msg,
                            uint32_t msg_len, const uint8_t *dst,
                            uint32_t dst_len, uint32_t len_in_bytes,
                            uint8_t *out) {
  if (!ctx || msg_len > 544 || dst_len == 0 || dst_len > 255 ||
      len_in_bytes > 256)
    return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint8_t *buf_d = nullptr;
  int32_t rc = M3X_ERR_HIP;
  if (hipMalloc(&buf_d, 1024 + 256) != hipSuccess) return M3X_ERR_HIP;
  if (msg_len &&
      hipMemcpyAsync(buf_d, msg, msg_len, hipMemcpyHostToDevice,
                     ctx->stream) != hipSuccess)
    goto out;
  if (hipMemcpyAsync(buf_d + 768, dst, dst_len, hipMemcpyHostToDevice,
                     ctx->stream) != hipSuccess)
    goto out;
  hipLaunchKernelGGL(k_bls_expand_dst, dim3(1), dim3(1), 0, ctx->stream,
                     buf_d, msg_len, buf_d + 768, dst_len, len_in_bytes,
                     buf_d + 1024);
  if (hipMemcpyAsync(out, buf_d + 1024, len_in_bytes, hipMemcpyDeviceToHost,
                     ctx->stream) != hipSuccess)
    goto out;
  if (hipStreamSynchronize(ctx->stream) == hipSuccess) rc = M3X_OK;
out:
  (void)hipFree(buf_d);
  return rc;
}

int32_t m3x_bls_h2c_test(m3x_ctx *ctx, const uint8_t *msg, uint32_t msg_len,
                         const uint8_t *dst, uint32_t dst_len,
                         uint8_t out_uncomp[192], uint8_t out_uniform[256]) {
  if (!ctx || msg_len > 544 || dst_len == 0 || dst_len > 255)
    return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint8_t *buf_d = nullptr;
  int32_t rc = M3X_ERR_HIP;
  if (hipMalloc(&buf_d, 1024 + 192 + 256) != hipSuccess) return M3X_ERR_HIP;
  if (msg_len &&
      hipMemcpyAsync(buf_d, msg, msg_len, hipMemcpyHostToDevice,
                     ctx->stream) != hipSuccess)
    goto out;
  if (hipMemcpyAsync(buf_d + 768, dst, dst_len, hipMemcpyHostToDevice,
                     ctx->stream) != hipSuccess)
    goto out;
  hipLaunchKernelGGL(k_bls_h2c_dst, dim3(1), dim3(1), 0, ctx->stream, buf_d,
                     msg_len, buf_d + 768, dst_len, buf_d + 1024,
                     buf_d + 1024 + 192);
  if (hipMemcpyAsync(out_uncomp, buf_d + 1024, 192, hipMemcpyDeviceToHost,
                     ctx->stream) != hipSuccess)
    goto out;
  if (out_uniform &&
      hipMemcpyAsync(out_uniform, buf_d + 1024 + 192, 256,
                     hipMemcpyDeviceToHost, ctx->stream) != hipSuccess)
    goto out;
  if (hipStreamSynchronize(ctx->stream) == hipSuccess) rc = M3X_OK;
out:
  (void)hipFree(buf_d);
  return rc;
}

int32_t m3x_bls_sig_aggregate(m3x_ctx *ctx, const uint8_t *sigs, uint64_t n,
                              uint8_t out_sig[96]) {
  if (!ctx || n == 0) return M3X_ERR_ARG;
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint8_t *sigs_d, *out_d;
  int *fail_d;
  M3X_HIP_CHECK(hipMalloc(&sigs_d, n * 96));
  M3X_HIP_CHECK(hipMalloc(&out_d, 96));
  M3X_HIP_CHECK(hipMalloc(&fail_d, 4));
  hipError_t e1 = hipMemcpyAsync(sigs_d, sigs, n * 96,
                                 hipMemcpyHostToDevice, ctx->stream);
  (void)hipMemsetAsync(fail_d, 0, 4, ctx->stream);
  hipLaunchKernelGGL(k_bls_sig_aggregate_w, dim3(1), dim3(64), 0, ctx->stream,
                     sigs_d, n, out_d, fail_d);
  int fail = 1;
  (void)hipMemcpyAsync(out_sig, out_d, 96, hipMemcpyDeviceToHost, ctx->stream);
  (void)hipMemcpyAsync(&fail, fail_d, 4, hipMemcpyDeviceToHost, ctx->stream);
  hipError_t e2 = hipStreamSynchronize(ctx->stream);
  (void)hipFree(sigs_d);
  (void)hipFree(out_d);
  (void)hipFree(fail_d);
  if (e1 != hipSuccess || e2 != hipSuccess) return M3X_ERR_HIP;
  return fail ? M3X_ERR_ARG : M3X_OK;
}

int32_t m3x_bls_verify_sets_dev(m3x_ctx *ctx, const void *msgs_dev,
                                const void *sigs_dev, const void *pks_dev,
                                const void *pk_offsets_dev,
                                const void *rands_dev, uint64_t n) {
  if (!ctx) return M3X_ERR_ARG;
  if (n == 0) return 0; // blst.rs:42-44 (host normally short-circuits)
  std::lock_guard<std::recursive_mutex> lk(ctx->mu);
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  int32_t verdict = 0;
  int rc = run_verify(ctx, msgs_dev, sigs_dev, pks_dev, pk_offsets_dev,
                      rands_dev, n, &verdict);
  if (rc != M3X_OK) return rc;
  return verdict;
}

int32_t m3x_bls_verify_sets(m3x_ctx *ctx, const uint8_t *msgs,
                            const uint8_t *sigs, const uint8_t *pks,
                            const uint32_t *pk_offsets, const uint64_t *rands,
                            uint64_t n) {
  if (!ctx) return M3X_ERR_ARG;
  if (n == 0) return 0;
  M3X_HIP_CHECK(hipSetDevice(ctx->device));
  uint64_t nk = pk_offsets[n];
  // checked uploads + single cleanup path: an infrastructure failure must
  // surface as M3X_ERR_HIP, never as a "signature invalid" verdict from a
  // kernel run over stale device memory
  uint8_t *msgs_d = nullptr, *sigs_d = nullptr, *pks_d = nullptr;
  uint32_t *offs_d = nullptr;
  uint64_t *rands_d = nullptr;
  int32_t rc = M3X_ERR_HIP;
  if (hipMalloc(&msgs_d, n * 32) != hipSuccess) goto out;
  if (hipMalloc(&sigs_d, n * 96) != hipSuccess) goto out;
  if (hipMalloc(&pks_d, nk * 96) != hipSuccess) goto out;
  if (hipMalloc(&offs_d, (n + 1) * 4) != hipSuccess) goto out;
  if (hipMalloc(&rands_d, n * 8) != hipSuccess) goto out;
  if (hipMemcpy(msgs_d, msgs, n * 32, hipMemcpyHostToDevice) != hipSuccess)
    goto out;
  if (hipMemcpy(sigs_d, sigs, n * 96, hipMemcpyHostToDevice) != hipSuccess)
    goto out;
  if (hipMemcpy(pks_d, pks, nk * 96, hipMemcpyHostToDevice) != hipSuccess)
    goto out;
  if (hipMemcpy(offs_d, pk_offsets, (n + 1) * 4, hipMemcpyHostToDevice) !=
      hipSuccess)
    goto out;
  if (hipMemcpy(rands_d, rands, n * 8, hipMemcpyHostToDevice) != hipSuccess)
    goto out;
  rc = m3x_bls_verify_sets_dev(ctx, msgs_d, sigs_d, pks_d, offs_d, rands_d,
                               n);
out:
  if (msgs_d) (void)hipFree(msgs_d);
  if (sigs_d) (void)hipFree(sigs_d);
  if (pks_d) (void)hipFree(pks_d);
  if (offs_d) (void)hipFree(offs_d);
  if (rands_d) (void)hipFree(rands_d);
  return rc;
}

} // extern "C"
