// finishbench — phase-level timing of the one-wave finish path
// (k_bls_finish's building blocks: miller_w, final_exp_w, f12_inv_w,
// f12_pow_xabs_w, f12_mul_w/f12_sqr_w). One block of 64 threads, wall
// clock (s_memrealtime, constant 100 MHz) stamped by lane 0 around each
// phase. Timing is data-independent for this arithmetic, so inputs are
// arbitrary in-range field values.
//
// Build: hipcc --offload-arch=gfx950 -O3 -I lighthouse_amd/csrc \
//          tools/finishbench.hip -o gpurun_out/finishbench
#include <hip/hip_runtime.h>
#include <cstdio>
#include <cstdint>
#include "bls_device.hh"

using namespace m3xb;

#define NSTAMP 16

__global__ __launch_bounds__(64) void k_finishbench(uint64_t *stamp,
                                                    fp12m *out) {
  __shared__ fp12m sh[7];
  __shared__ f12w_ws ws;
  __shared__ miller_ws mws;
  int lane = threadIdx.x;
  // inputs: a smallish fp12 "miller value" and a scaled-generator g2j
  if (lane == 0) {
    f12_one(sh[0]);
    for (int i = 0; i < 12; i++) {
      fp t;
      fp_one(t);
      for (int j = 0; j < i + 2; j++) fp_add(t, t, t);
      fp_add(sh[0].s[i], sh[0].s[i], t);
    }
  }
  g1j ng1;
  {
    g1a g;
    g1_gen(g);
    ng1.x = g.x;
    fp_neg(ng1.y, g.y);
    fp_one(ng1.z);
  }
  g2j Qj;
  {
    g2a g;
    FP_LOAD_C(g.x.c0, BLS_G2X_C0);
    FP_LOAD_C(g.x.c1, BLS_G2X_C1);
    FP_LOAD_C(g.y.c0, BLS_G2Y_C0);
    FP_LOAD_C(g.y.c1, BLS_G2Y_C1);
    g.inf = false;
    g2j t;
    g2j_from_aff(t, g);
    g2j_dbl(t, t); // non-trivial Z
    g2j_dbl(Qj, t);
  }
  __syncthreads();
#define STAMP(k)                                                             \
  do {                                                                       \
    __syncthreads();                                                         \
    if (lane == 0) stamp[k] = wall_clock64();                                \
    __syncthreads();                                                         \
  } while (0)

  STAMP(0);
  miller_w(sh[1], ng1, Qj, ws, mws, lane); // full cooperative miller
  STAMP(1);
  final_exp_w(sh[2], sh[1], &sh[3], ws, lane); // full final exp
  STAMP(2);
  f12_inv_w(sh[3], sh[1], sh[4], sh[5], ws, lane); // one fp12 inverse
  STAMP(3);
  f12_pow_xabs_w(sh[4], sh[1], ws, lane); // one |x|-power chain
  STAMP(4);
  for (int i = 0; i < 64; i++) f12_mul_w(sh[5], sh[5], sh[1], ws, lane);
  STAMP(5); // 64 cooperative generic muls
  for (int i = 0; i < 64; i++) f12_sqr_w(sh[5], ws, lane);
  STAMP(6); // 64 cooperative squarings
  // product phase only (lane<36 fp2_mul into ws) — isolates the math
  for (int i = 0; i < 64; i++) {
    if (lane < 36) {
      int pi = lane / 6, pj = lane % 6;
      fp2 ai, bj, t;
      f12_get(sh[5], pi, ai);
      f12_get(sh[1], pj, bj);
      fp2_mul(t, ai, bj);
      ws.t[lane] = t;
    }
    __syncthreads();
  }
  STAMP(7);
  // accumulate phase only (lane<6 fold of ws.t)
  for (int it = 0; it < 64; it++) {
    if (lane < 6) {
      fp2 acc, h, t;
      fp2_zero(acc);
      fp2_zero(h);
      for (int i = 0; i <= lane; i++) {
        t = ws.t[i * 6 + (lane - i)];
        fp2_add(acc, acc, t);
      }
      for (int i = lane + 1; i < 6; i++) {
        t = ws.t[i * 6 + (lane + 6 - i)];
        fp2_add(h, h, t);
      }
      fp2_mul_xi(h, h);
      fp2_add(acc, acc, h);
      f12_set(sh[5], lane, acc);
    }
    __syncthreads();
  }
  STAMP(8);
  // bare fp2_mul, all 64 lanes, no LDS (pure issue cost reference)
  {
    fp2 x = Qj.x, y = Qj.y;
    for (int i = 0; i < 64; i++) fp2_mul(x, x, y);
    if (lane == 0) sh[6].s[2] = x.c0;
  }
  STAMP(9);
  // phase-1-equivalent: the serial T-chain alone on lane 0
  if (lane == 0) {
    g2j T = Qj;
    int idx = 0;
    for (int i = 62; i >= 0; i--) {
      mws.T[idx][0] = T.x;
      mws.T[idx][1] = T.y;
      mws.T[idx][2] = T.z;
      idx++;
      g2j_dbl(T, T);
      if ((BLS_X_ABS >> i) & 1) {
        mws.T[idx][0] = T.x;
        mws.T[idx][1] = T.y;
        mws.T[idx][2] = T.z;
        idx++;
        g2j_add(T, T, Qj);
      }
    }
    sh[6].s[0] = T.x.c0; // keep alive
  }
  STAMP(10);
  // serial fp_inv on lane 0 (the f12_inv_w norm tail)
  if (lane == 0) {
    fp v = sh[0].s[1], r;
    fp_inv(r, v);
    sh[6].s[1] = r;
  }
  STAMP(11);
  if (lane == 0) {
    for (int i = 0; i < 7; i++) f12_copy(out[i], sh[i]);
  }
}

// per-lane h2c latency breakdown (all 64 lanes compute the same message,
// matching the per-lane production kernel's regime)
__global__ __launch_bounds__(64) void k_h2cbench(uint64_t *stamp,
                                                 g2j *out) {
  int lane = threadIdx.x;
  uint8_t msg[32];
  for (int i = 0; i < 32; i++) msg[i] = (uint8_t)(i * 7 + 3);
#define HSTAMP(k)                                                            \
  do {                                                                       \
    __syncthreads();                                                         \
    if (lane == 0) stamp[k] = wall_clock64();                                \
    __syncthreads();                                                         \
  } while (0)
  HSTAMP(0);
  uint8_t uni[256];
  expand_message_xmd32(msg, uni);
  HSTAMP(1);
  fp2 u0, u1;
  h2f_from_be64(u0.c0, uni);
  h2f_from_be64(u0.c1, uni + 64);
  h2f_from_be64(u1.c0, uni + 128);
  h2f_from_be64(u1.c1, uni + 192);
  HSTAMP(2);
  fp2 uu[2] = {u0, u1};
  g2a qp[2];
  sswu_g2_dual(qp, uu);
  HSTAMP(3);
  g2j s, t;
  iso_map_g2_j(s, qp[0]);
  iso_map_g2_j(t, qp[1]);
  HSTAMP(4);
  g2j_add(s, s, t);
  HSTAMP(5);
  g2j r;
  clear_cofactor_g2j(r, s);
  HSTAMP(6);
  if (lane == 0) out[0] = r;
}

int main() {
  uint64_t *stamp_d;
  fp12m *out_d;
  (void)hipMalloc(&stamp_d, NSTAMP * 8);
  (void)hipMalloc(&out_d, 7 * sizeof(fp12m));
  // warmup + timed
  for (int it = 0; it < 3; it++) {
    hipLaunchKernelGGL(k_finishbench, dim3(1), dim3(64), 0, 0, stamp_d,
                       out_d);
  }
  (void)hipDeviceSynchronize();
  uint64_t st[NSTAMP];
  (void)hipMemcpy(st, stamp_d, NSTAMP * 8, hipMemcpyDeviceToHost);
  const char *names[] = {"miller_w (full)",   "final_exp_w (full)",
                         "f12_inv_w",         "f12_pow_xabs_w",
                         "64x f12_mul_w",     "64x f12_sqr_w",
                         "64x product-only",  "64x accumulate-only",
                         "64x bare fp2_mul",  "serial T-chain",
                         "serial fp_inv"};
  // wall_clock64 is the constant ~100 MHz counter
  double mhz = 100.0;
  for (int k = 0; k < 11; k++) {
    double ms = (double)(st[k + 1] - st[k]) / (mhz * 1000.0);
    printf("%-20s %10.3f ms\n", names[k], ms);
  }
  printf("total                %10.3f ms\n",
         (double)(st[11] - st[0]) / (mhz * 1000.0));
  // h2c per-lane latency breakdown
  g2j *h2c_d;
  (void)hipMalloc(&h2c_d, sizeof(g2j));
  for (int it = 0; it < 3; it++)
    hipLaunchKernelGGL(k_h2cbench, dim3(1), dim3(64), 0, 0, stamp_d, h2c_d);
  (void)hipDeviceSynchronize();
  (void)hipMemcpy(st, stamp_d, NSTAMP * 8, hipMemcpyDeviceToHost);
  const char *hn[] = {"expand_message", "h2f x4", "sswu x2", "iso_map x2",
                      "from_aff+add",   "clear_cofactor"};
  printf("-- h2c per-lane latency --\n");
  for (int k = 0; k < 6; k++)
    printf("%-20s %10.3f ms\n", hn[k],
           (double)(st[k + 1] - st[k]) / (mhz * 1000.0));
  printf("h2c total            %10.3f ms\n",
         (double)(st[6] - st[0]) / (mhz * 1000.0));
  return 0;
}
