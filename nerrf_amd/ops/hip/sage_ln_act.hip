// Fused GraphSAGE layer tail (training path) — CDNA4 (gfx950), D = 128.
//
//   fwd: y = h + LayerNorm(dropout(GELU(zs + zn))) * gamma + beta
//   bwd: dz (shared by both GEMM branches), dh (= dy, folded by caller),
//        dgamma/dbeta (f32 atomic partials)
//
// The eager training layer ran this as ~5 elementwise/reduce launches
// forward and ~7 backward, per layer, x28 layers (z-add, GELU, dropout,
// LN, residual).  One wave handles one row (D = 128 -> 2 elements/lane),
// so the LN statistics are wave shuffles; dropout uses a counter-based
// RNG keyed on (seed, element index) and is RECOMPUTED in backward —
// no mask tensor exists.  Saved for backward: the pre-activation s = zs+zn
// (bf16) and the LN mean/rstd (f32 per row).
//
// Numerics: erf-exact GELU (matches torch), biased LN variance, eps 1e-5,
// fp32 math throughout the row.  Validated against the eager PyTorch layer
// (tests/test_ops_gpu.py; dropout path cross-checked via the debug mask
// output).  SURVEY §2a GNN-layer fusion — training-side counterpart of
// sage_fused.hip.
#include "common.h"

namespace nerrf {

#define SLN_D 128

typedef __bf16 lbf16x2 __attribute__((ext_vector_type(2)));

__device__ __forceinline__ unsigned sln_hash(unsigned a, unsigned b) {
  // 2-round xxhash-style mix: counter-based, replayable in backward
  unsigned h = a * 2654435761u ^ b * 2246822519u;
  h ^= h >> 15;
  h *= 2654435761u;
  h ^= h >> 13;
  return h;
}

__device__ __forceinline__ float sln_gelu(float s) {
  return 0.5f * s * (1.0f + erff(s * 0.70710678118654752f));
}

__device__ __forceinline__ float sln_gelu_grad(float s) {
  const float cdf = 0.5f * (1.0f + erff(s * 0.70710678118654752f));
  const float pdf = 0.3989422804014327f * __expf(-0.5f * s * s);
  return cdf + s * pdf;
}

// one wave per row; 2 elements per lane
__global__ void sage_ln_act_fwd_kernel(
    const __hip_bfloat16* __restrict__ h,     // [N, 128]
    const __hip_bfloat16* __restrict__ zs,    // [N, 128]
    const __hip_bfloat16* __restrict__ zn,    // [N, 128]
    const __hip_bfloat16* __restrict__ gamma, // [128]
    const __hip_bfloat16* __restrict__ beta,  // [128]
    __hip_bfloat16* __restrict__ y,           // [N, 128]
    __hip_bfloat16* __restrict__ s_save,      // [N, 128] pre-activation
    float* __restrict__ stat_save,            // [N, 2] (mean, rstd)
    unsigned char* __restrict__ mask_dbg,     // [N, 128] or nullptr (tests)
    long n_rows, float drop_p, unsigned seed) {
  const long row = blockIdx.x * (blockDim.x / NERRF_WAVE) + threadIdx.x / NERRF_WAVE;
  const int lane = threadIdx.x % NERRF_WAVE;
  if (row >= n_rows) return;
  const long base = row * SLN_D + lane * 2;
  const lbf16x2 zs2 = *reinterpret_cast<const lbf16x2*>(zs + base);
  const lbf16x2 zn2 = *reinterpret_cast<const lbf16x2*>(zn + base);
  float x0, x1, s0, s1;
  s0 = (float)zs2[0] + (float)zn2[0];
  s1 = (float)zs2[1] + (float)zn2[1];
  *reinterpret_cast<lbf16x2*>(s_save + base) = lbf16x2{(__bf16)s0, (__bf16)s1};
  // torch rounds s to bf16 between the GEMM add and GELU; match it so the
  // backward recompute from s_save is self-consistent
  s0 = (float)(__bf16)s0;
  s1 = (float)(__bf16)s1;
  x0 = sln_gelu(s0);
  x1 = sln_gelu(s1);
  if (drop_p > 0.0f) {
    const float keep = 1.0f - drop_p;
    const unsigned r0 = sln_hash(seed, (unsigned)(base));
    const unsigned r1 = sln_hash(seed, (unsigned)(base + 1));
    const bool k0 = (r0 >> 8) * (1.0f / 16777216.0f) >= drop_p;
    const bool k1 = (r1 >> 8) * (1.0f / 16777216.0f) >= drop_p;
    x0 = k0 ? x0 / keep : 0.0f;
    x1 = k1 ? x1 / keep : 0.0f;
    if (mask_dbg != nullptr) {
      mask_dbg[base] = k0;
      mask_dbg[base + 1] = k1;
    }
  }
  const float mean = wave_reduce_sum(x0 + x1) * (1.0f / SLN_D);
  const float d0 = x0 - mean, d1 = x1 - mean;
  const float var = wave_reduce_sum(d0 * d0 + d1 * d1) * (1.0f / SLN_D);
  const float rstd = rsqrtf(var + 1e-5f);
  if (lane == 0) {
    stat_save[row * 2] = mean;
    stat_save[row * 2 + 1] = rstd;
  }
  const lbf16x2 h2 = *reinterpret_cast<const lbf16x2*>(h + base);
  const lbf16x2 g2 = *reinterpret_cast<const lbf16x2*>(gamma + lane * 2);
  const lbf16x2 b2 = *reinterpret_cast<const lbf16x2*>(beta + lane * 2);
  const float y0 = (float)h2[0] + d0 * rstd * (float)g2[0] + (float)b2[0];
  const float y1 = (float)h2[1] + d1 * rstd * (float)g2[1] + (float)b2[1];
  *reinterpret_cast<lbf16x2*>(y + base) = lbf16x2{(__bf16)y0, (__bf16)y1};
}

__global__ void sage_ln_act_bwd_kernel(
    const __hip_bfloat16* __restrict__ dy,     // [N, 128]
    const __hip_bfloat16* __restrict__ s_save, // [N, 128]
    const float* __restrict__ stat_save,       // [N, 2]
    const __hip_bfloat16* __restrict__ gamma,  // [128]
    __hip_bfloat16* __restrict__ dz,           // [N, 128] grad for zs AND zn
    float* __restrict__ dgamma_part,           // [grid, 128] f32 partials
    float* __restrict__ dbeta_part,            // [grid, 128]
    long n_rows, float drop_p, unsigned seed) {
  // per-block LDS accumulation of the dgamma/dbeta column sums — one
  // global write per block per column instead of one atomic per row
  __shared__ float acc_g[SLN_D];
  __shared__ float acc_b[SLN_D];
  const int tid = threadIdx.x;
  if (tid < SLN_D) {
    acc_g[tid] = 0.0f;
    acc_b[tid] = 0.0f;
  }
  __syncthreads();
  const long row = blockIdx.x * (blockDim.x / NERRF_WAVE) + threadIdx.x / NERRF_WAVE;
  const int lane = threadIdx.x % NERRF_WAVE;
  if (row < n_rows) {
  const long base = row * SLN_D + lane * 2;
  const lbf16x2 s2 = *reinterpret_cast<const lbf16x2*>(s_save + base);
  const float s0 = (float)s2[0], s1 = (float)s2[1];
  // recompute x (GELU + replayed dropout) — the mask never hit memory
  float x0 = sln_gelu(s0), x1 = sln_gelu(s1);
  float k0f = 1.0f, k1f = 1.0f;
  if (drop_p > 0.0f) {
    const float keep = 1.0f - drop_p;
    const unsigned r0 = sln_hash(seed, (unsigned)(base));
    const unsigned r1 = sln_hash(seed, (unsigned)(base + 1));
    const bool k0 = (r0 >> 8) * (1.0f / 16777216.0f) >= drop_p;
    const bool k1 = (r1 >> 8) * (1.0f / 16777216.0f) >= drop_p;
    k0f = k0 ? 1.0f / keep : 0.0f;
    k1f = k1 ? 1.0f / keep : 0.0f;
    x0 *= k0f;
    x1 *= k1f;
  }
  const float mean = stat_save[row * 2];
  const float rstd = stat_save[row * 2 + 1];
  const float xh0 = (x0 - mean) * rstd, xh1 = (x1 - mean) * rstd;
  const lbf16x2 dy2 = *reinterpret_cast<const lbf16x2*>(dy + base);
  const float dy0 = (float)dy2[0], dy1 = (float)dy2[1];
  const lbf16x2 g2 = *reinterpret_cast<const lbf16x2*>(gamma + lane * 2);
  const float dxh0 = dy0 * (float)g2[0], dxh1 = dy1 * (float)g2[1];
  const float m1 = wave_reduce_sum(dxh0 + dxh1) * (1.0f / SLN_D);
  const float m2 = wave_reduce_sum(dxh0 * xh0 + dxh1 * xh1) * (1.0f / SLN_D);
  const float dx0 = (dxh0 - m1 - xh0 * m2) * rstd;
  const float dx1 = (dxh1 - m1 - xh1 * m2) * rstd;
  const float dz0 = dx0 * k0f * sln_gelu_grad(s0);
  const float dz1 = dx1 * k1f * sln_gelu_grad(s1);
  *reinterpret_cast<lbf16x2*>(dz + base) = lbf16x2{(__bf16)dz0, (__bf16)dz1};
  atomicAdd(acc_g + lane * 2, dy0 * xh0);
  atomicAdd(acc_g + lane * 2 + 1, dy1 * xh1);
  atomicAdd(acc_b + lane * 2, dy0);
  atomicAdd(acc_b + lane * 2 + 1, dy1);
  }
  __syncthreads();
  if (tid < SLN_D) {
    dgamma_part[(long)blockIdx.x * SLN_D + tid] = acc_g[tid];
    dbeta_part[(long)blockIdx.x * SLN_D + tid] = acc_b[tid];
  }
}

void launch_sage_ln_act_fwd(const void* h, const void* zs, const void* zn,
                            const void* gamma, const void* beta, void* y,
                            void* s_save, float* stat_save,
                            unsigned char* mask_dbg, long n_rows, float drop_p,
                            unsigned seed, hipStream_t s) {
  const int waves_per_block = 8;
  const int block = waves_per_block * NERRF_WAVE;
  const long grid = (n_rows + waves_per_block - 1) / waves_per_block;
  sage_ln_act_fwd_kernel<<<(int)grid, block, 0, s>>>(
      (const __hip_bfloat16*)h, (const __hip_bfloat16*)zs,
      (const __hip_bfloat16*)zn, (const __hip_bfloat16*)gamma,
      (const __hip_bfloat16*)beta, (__hip_bfloat16*)y,
      (__hip_bfloat16*)s_save, stat_save, mask_dbg, n_rows, drop_p, seed);
}

void launch_sage_ln_act_bwd(const void* dy, const void* s_save,
                            const float* stat_save, const void* gamma,
                            void* dz, float* dgamma_part, float* dbeta_part,
                            long n_rows, float drop_p, unsigned seed,
                            hipStream_t s) {
  const int waves_per_block = 8;
  const int block = waves_per_block * NERRF_WAVE;
  const long grid = (n_rows + waves_per_block - 1) / waves_per_block;
  sage_ln_act_bwd_kernel<<<(int)grid, block, 0, s>>>(
      (const __hip_bfloat16*)dy, (const __hip_bfloat16*)s_save, stat_save,
      (const __hip_bfloat16*)gamma, (__hip_bfloat16*)dz, dgamma_part,
      dbeta_part, n_rows, drop_p, seed);
}

}  // namespace nerrf
