// Shared declarations for the split BLS translation units (round 2:
// the single TU compiled in ~19 min; three TUs compile in parallel).
#pragma once
#include <hip/hip_runtime.h>
#include <stdint.h>
#include "bls_device.hh"
#include "m3x_ctx.hh"
#include "../../include/m3x_consensus.h"

using namespace m3xb; // internal header (g1j/g2j/fp12m field types)

// scratch-resident work layout for one batch-verify call (offsets carved
// in run_verify, m3x_bls.hip)
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

// per-TU launcher seams (each TU owns its kernels; run_verify sequences
// them around the shared timing slots)
namespace m3xk {
void launch_pk_decompress(hipStream_t s, const uint8_t *comp_d, uint64_t n,
                          uint8_t *unc_d, int32_t *st_d);
void launch_aggregate(hipStream_t s, const uint8_t *pks_dev,
                      const uint32_t *offs_dev, uint64_t n, BlsWork w);
void launch_prepare(hipStream_t s, const uint8_t *sigs_dev,
                    const uint8_t *pks_dev, const uint32_t *offs_dev,
                    const uint64_t *rands_dev, uint64_t n, BlsWork w);
void launch_h2c(hipStream_t s, const uint8_t *msgs_dev, uint64_t n,
                BlsWork w);
uint64_t launch_miller(hipStream_t s, uint64_t n, BlsWork w);
void launch_reduce(hipStream_t s, uint64_t n_parts, uint64_t n, BlsWork w);
void launch_finish(hipStream_t s, BlsWork w);
void launch_h2c_dst_test(hipStream_t s, const uint8_t *msg_d,
                         uint32_t msg_len, const uint8_t *dst_d,
                         uint32_t dst_len, uint8_t *out_d, uint8_t *uni_d);
void launch_expand_test(hipStream_t s, const uint8_t *msg_d,
                        uint32_t msg_len, const uint8_t *dst_d,
                        uint32_t dst_len, uint32_t len_in_bytes,
                        uint8_t *out_d);
} // namespace m3xk
