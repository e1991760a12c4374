// MI355X-native batched BLS12-381 signature-set verification (hot path #1).
//
// Implements the blst.rs:37-119 contract (see include/m3x_consensus.h and
// DESIGN.md): per set — decompress + subgroup-check sigma (deferred checks
// exactly as generic_aggregate_signature.rs:161-176 + blst.rs:73-77),
// aggregate the set's pubkeys, scale by the host-drawn 64-bit r_i, hash the
// message to G2 (RFC 9380), accumulate one Miller loop per set; then one
// extra pair e(-g1, sum r_i sigma_i), one final exponentiation, compare to
// one. CDNA4 shape: one set per lane (wave-split variants by regime),
// kernels split by divergence class, grids >>256 workgroups; integer VALU
// workload (no MFMA). Prepare/h2c kernels live in m3x_bls_prep.hip /
// m3x_bls_h2c.hip (parallel-compiling TUs; seams in m3x_bls_common.hh).
#include "m3x_bls_common.hh"
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
  m3xk::launch_aggregate(ctx->stream, (const uint8_t *)pks_dev,
                         (const uint32_t *)offs_dev, n, w);
  m3x::time_end(ctx, M3X_K_BLS_AGG);
  m3x::time_begin(ctx, M3X_K_BLS_PREPARE);
  m3xk::launch_prepare(ctx->stream, (const uint8_t *)sigs_dev,
                       (const uint8_t *)pks_dev, (const uint32_t *)offs_dev,
                       (const uint64_t *)rands_dev, n, w);
  m3x::time_end(ctx, M3X_K_BLS_PREPARE);
  DBG_STEP(ctx, "prepare");
  // h2c is independent of prepare: run it on the second stream so the two
  // ~1-wave/SIMD kernels co-reside (both fit at 2 waves/SIMD by VGPR count)
  m3x::time_begin_s(ctx, M3X_K_BLS_H2C, ctx->stream2);
  m3xk::launch_h2c(ctx->stream2, (const uint8_t *)msgs_dev, n, w);
  m3x::time_end_s(ctx, M3X_K_BLS_H2C, ctx->stream2);
  M3X_HIP_CHECK(hipEventRecord(ctx->ev_s2, ctx->stream2));
  M3X_HIP_CHECK(hipStreamWaitEvent(ctx->stream, ctx->ev_s2, 0));
  DBG_STEP(ctx, "h2c");
  m3x::time_begin(ctx, M3X_K_BLS_MILLER);
  uint64_t n_parts = m3xk::launch_miller(ctx->stream, n, w);
  m3x::time_end(ctx, M3X_K_BLS_MILLER);
  DBG_STEP(ctx, "miller");
  m3x::time_begin(ctx, M3X_K_BLS_REDUCE);
  m3xk::launch_reduce(ctx->stream, n_parts, n, w);
  m3x::time_end(ctx, M3X_K_BLS_REDUCE);
  DBG_STEP(ctx, "reduce");
  m3x::time_begin(ctx, M3X_K_BLS_FINISH);
  m3xk::launch_finish(ctx->stream, w);
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
  m3xk::launch_pk_decompress(ctx->stream, comp_d, n, unc_d, st_d);
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
  m3xk::launch_expand_test(ctx->stream, buf_d, msg_len, buf_d + 768,
                           dst_len, len_in_bytes, buf_d + 1024);
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
  m3xk::launch_h2c_dst_test(ctx->stream, buf_d, msg_len, buf_d + 768,
                            dst_len, buf_d + 1024, buf_d + 1024 + 192);
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
