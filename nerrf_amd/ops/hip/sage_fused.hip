// Fully-fused GraphSAGE-T layer forward — CDNA4 (gfx950), D = 128 only.
//
// One launch per layer computes, for a 64-node tile per block:
//   agg   = weighted mean of K sampled neighbor rows        (global gather)
//   z     = h @ W_self^T + agg @ W_nbr^T + b                (MFMA, K = 128)
//   out   = h + LayerNorm(GELU(z)) * gamma + beta           (residual)
// — the inference path of models/graphsage.SageLayer in a single kernel
// (eager: gather + 2 hipBLASLt GEMMs + ~5 elementwise/reduce launches).
// Inference-only: training keeps the autograd path.  (Model spec:
// reference docs architecture.mdx:49-53 — GraphSAGE-T, 28 layers; the
// kernel design itself has no upstream counterpart.)
//
// Geometry: 8 waves (512 threads); wave w owns rows [(w>>1)*16, +16) and the
// column half (w&1)*64 of the 64x128 output tile -> per GEMM 4 accumulators
// of v_mfma_f32_16x16x32_bf16 over K=128 (4 k-steps).
//
// LDS (96 KB): x tile + agg tile (bf16, ((row&15)<<4) XOR swizzle for
// conflict-free ds_read_b128 A-fragments), W_self + W_nbr (bf16, same
// swizzle on the B side, cooperatively staged from L2), and a fp32 z-stage
// that ALIASES the W area once the MFMAs are done (time-disjoint) so the
// LayerNorm phase sees whole rows.
//
// Fragment maps: as lstm_step_fused.hip (A row=l&15 / B col=l&15, k-chunk
// (l>>4)*8; C/D row=(l>>4)*4+r, col=l&15).  Numerics: fp32 accumulation,
// erf-exact GELU, torch-LayerNorm semantics (biased variance, eps 1e-5).
// Validated against the eval-mode PyTorch layer (tests/test_ops_gpu.py).
#include "common.h"

namespace nerrf {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define SAGE_D 128
#define SAGE_BM 64
#define ROW_B (SAGE_D * 2)  // 256 B per row

__device__ __forceinline__ unsigned sage_swz(unsigned row, unsigned byte_col) {
  return row * ROW_B + (byte_col ^ ((row & 15u) << 4));
}

__launch_bounds__(512)
__global__ void sage_layer_fwd_kernel(
    const __hip_bfloat16* __restrict__ h,       // [N, 128] layer input
    const long* __restrict__ nbr_idx,           // [N, K]
    const float* __restrict__ nbr_w,            // [N, K]
    const __hip_bfloat16* __restrict__ w_self,  // [128, 128] (out, in)
    const __hip_bfloat16* __restrict__ w_nbr,   // [128, 128]
    const __hip_bfloat16* __restrict__ bias,    // [128] (w_nbr bias)
    const __hip_bfloat16* __restrict__ gamma,   // [128] LN weight
    const __hip_bfloat16* __restrict__ beta,    // [128] LN bias
    __hip_bfloat16* __restrict__ out,           // [N, 128]
    int n_nodes, int k) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* x_lds = smem;                         // 16 KB (swizzled)
  char* agg_lds = smem + SAGE_BM * ROW_B;     // 16 KB (swizzled)
  char* ws_lds = agg_lds + SAGE_BM * ROW_B;   // 32 KB (swizzled)
  char* wn_lds = ws_lds + SAGE_D * ROW_B;     // 32 KB (swizzled)
  float* z_lds = reinterpret_cast<float*>(ws_lds);  // aliased after MFMAs

  const int row0 = blockIdx.x * SAGE_BM;
  const int tid = threadIdx.x;
  const int wave = tid / NERRF_WAVE;
  const int lane = tid % NERRF_WAVE;

  // ---- stage x tile (64 rows x 256 B) and both W (128 rows x 256 B) ------
  {
    const int r = tid / 8;          // 0..63
    const int c16 = (tid % 8) * 2;  // two 16-B chunks per thread
    const long grow = (long)(row0 + r);
#pragma unroll
    for (int cc = 0; cc < 2; ++cc) {
      uint4 v = make_uint4(0, 0, 0, 0);
      if (grow < n_nodes)
        v = *reinterpret_cast<const uint4*>(
            reinterpret_cast<const char*>(h) + grow * ROW_B + (c16 + cc) * 16);
      *reinterpret_cast<uint4*>(x_lds + sage_swz(r, (c16 + cc) * 16)) = v;
    }
    const int wr = tid / 4;         // 0..127
    const int wc16 = (tid % 4) * 4; // four 16-B chunks per thread
#pragma unroll
    for (int cc = 0; cc < 4; ++cc) {
      *reinterpret_cast<uint4*>(ws_lds + sage_swz(wr, (wc16 + cc) * 16)) =
          *reinterpret_cast<const uint4*>(
              reinterpret_cast<const char*>(w_self) + wr * ROW_B + (wc16 + cc) * 16);
      *reinterpret_cast<uint4*>(wn_lds + sage_swz(wr, (wc16 + cc) * 16)) =
          *reinterpret_cast<const uint4*>(
              reinterpret_cast<const char*>(w_nbr) + wr * ROW_B + (wc16 + cc) * 16);
    }
  }

  // ---- gather: agg rows (weighted neighbor mean), wave w -> rows w*8..+8 --
  for (int r = wave * 8; r < wave * 8 + 8; ++r) {
    const long node = (long)(row0 + r);
    float acc0 = 0.0f, acc1 = 0.0f;
    if (node < n_nodes) {
      float w_lane = (lane < k) ? nbr_w[node * k + lane] : 0.0f;
      long i_lane = (lane < k) ? nbr_idx[node * k + lane] : 0;
      const float inv_denom = 1.0f / fmaxf(wave_reduce_sum(w_lane), 1e-6f);
      for (int kk = 0; kk < k; ++kk) {
        const float wk = __shfl(w_lane, kk, NERRF_WAVE) * inv_denom;
        const long nb = __shfl(i_lane, kk, NERRF_WAVE);
        const ushort2 v = reinterpret_cast<const ushort2*>(h + nb * SAGE_D)[lane];
        acc0 = fmaf(wk, __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&v.x)), acc0);
        acc1 = fmaf(wk, __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&v.y)), acc1);
      }
    }
    __hip_bfloat16 b0 = __float2bfloat16(acc0);
    __hip_bfloat16 b1 = __float2bfloat16(acc1);
    uint2 packed;
    packed.x = (unsigned)*reinterpret_cast<const unsigned short*>(&b0) |
               ((unsigned)*reinterpret_cast<const unsigned short*>(&b1) << 16);
    *reinterpret_cast<unsigned*>(agg_lds + sage_swz(r, lane * 4)) = packed.x;
  }
  __syncthreads();

  // ---- dual MFMA: z = x @ Ws^T + agg @ Wn^T ------------------------------
  const int row_grp = (wave >> 1) * 16;  // this wave's 16-row group
  const int col_half = (wave & 1) * 64;  // this wave's 64-col half
  f32x4 accS[4], accN[4];
#pragma unroll
  for (int cf = 0; cf < 4; ++cf) {
    accS[cf] = f32x4{0.f, 0.f, 0.f, 0.f};
    accN[cf] = f32x4{0.f, 0.f, 0.f, 0.f};
  }
  const int arow = row_grp + (lane & 15);
  const int kbyte = (lane >> 4) * 16;
#pragma unroll
  for (int ks = 0; ks < SAGE_D / 32; ++ks) {
    const unsigned koff = ks * 64 + kbyte;
    const bf16x8 a_x = *reinterpret_cast<const bf16x8*>(x_lds + sage_swz(arow, koff));
    const bf16x8 a_g = *reinterpret_cast<const bf16x8*>(agg_lds + sage_swz(arow, koff));
#pragma unroll
    for (int cf = 0; cf < 4; ++cf) {
      const int o = col_half + cf * 16 + (lane & 15);
      const bf16x8 b_s = *reinterpret_cast<const bf16x8*>(ws_lds + sage_swz(o, koff));
      const bf16x8 b_n = *reinterpret_cast<const bf16x8*>(wn_lds + sage_swz(o, koff));
      accS[cf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_x, b_s, accS[cf], 0, 0, 0);
      accN[cf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(a_g, b_n, accN[cf], 0, 0, 0);
    }
  }

  // ---- z = GELU(accS + accN + bias) staged fp32 (aliases the W area) -----
  __syncthreads();  // all waves done reading W before aliasing
#pragma unroll
  for (int cf = 0; cf < 4; ++cf) {
    const int col = col_half + cf * 16 + (lane & 15);
    const float b = __bfloat162float(bias[col]);
#pragma unroll
    for (int r = 0; r < 4; ++r) {
      const int row = row_grp + (lane >> 4) * 4 + r;
      const float zv = accS[cf][r] + accN[cf][r] + b;
      const float gelu = 0.5f * zv * (1.0f + erff(zv * 0.70710678118654752f));
      z_lds[row * SAGE_D + col] = gelu;
    }
  }
  __syncthreads();

  // ---- LayerNorm + residual: wave w -> rows w*8..+8; lane -> col pair ----
  for (int r = wave * 8; r < wave * 8 + 8; ++r) {
    const long node = (long)(row0 + r);
    if (node >= n_nodes) continue;
    const float z0 = z_lds[r * SAGE_D + 2 * lane];
    const float z1 = z_lds[r * SAGE_D + 2 * lane + 1];
    const float sum = wave_reduce_sum(z0 + z1);
    const float mean = sum * (1.0f / SAGE_D);
    const float d0 = z0 - mean, d1 = z1 - mean;
    const float var = wave_reduce_sum(d0 * d0 + d1 * d1) * (1.0f / SAGE_D);
    const float rstd = rsqrtf(var + 1e-5f);
    // original input (swizzled bf16 pair at byte col lane*4)
    const unsigned xv = *reinterpret_cast<const unsigned*>(x_lds + sage_swz(r, lane * 4));
    const float x0 = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&xv));
    const unsigned xhi = xv >> 16;
    const float x1 = __bfloat162float(*reinterpret_cast<const __hip_bfloat16*>(&xhi));
    const float g0 = __bfloat162float(gamma[2 * lane]);
    const float g1 = __bfloat162float(gamma[2 * lane + 1]);
    const float be0 = __bfloat162float(beta[2 * lane]);
    const float be1 = __bfloat162float(beta[2 * lane + 1]);
    __hip_bfloat16 o0 = __float2bfloat16(x0 + d0 * rstd * g0 + be0);
    __hip_bfloat16 o1 = __float2bfloat16(x1 + d1 * rstd * g1 + be1);
    ushort2 packed;
    packed.x = *reinterpret_cast<const unsigned short*>(&o0);
    packed.y = *reinterpret_cast<const unsigned short*>(&o1);
    reinterpret_cast<ushort2*>(out + node * SAGE_D)[lane] = packed;
  }
}

void launch_sage_layer_fwd(const void* h, const long* nbr_idx,
                           const float* nbr_w, const void* w_self,
                           const void* w_nbr, const void* bias,
                           const void* gamma, const void* beta, void* out,
                           int n_nodes, int k, hipStream_t s) {
  const int grid = (n_nodes + SAGE_BM - 1) / SAGE_BM;
  const size_t lds = (SAGE_BM * 2 + SAGE_D * 2) * ROW_B;  // 96 KB
  sage_layer_fwd_kernel<<<grid, 512, lds, s>>>(
      (const __hip_bfloat16*)h, nbr_idx, nbr_w, (const __hip_bfloat16*)w_self,
      (const __hip_bfloat16*)w_nbr, (const __hip_bfloat16*)bias,
      (const __hip_bfloat16*)gamma, (const __hip_bfloat16*)beta,
      (__hip_bfloat16*)out, n_nodes, k);
}

}  // namespace nerrf
