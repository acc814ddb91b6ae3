// Fully-fused LSTM recurrent step on MFMA — CDNA4 (gfx950), H = 256 only.
//
// One launch computes, for a 32-row batch tile per block:
//     gates_pre = h_prev @ W_hh^T + xg + bias      (MFMA, K = 256)
//     i,f,o = sigmoid; g = tanh; c' = f*c + i*g; h' = o*tanh(c'); mask
// eliminating the separate hipBLASLt GEMM and its [B,4H] HBM round trip
// (gates_pre never leaves the CU; the hidden-state tile lives in LDS).
//
// Geometry: 8 waves (512 threads); wave w owns hidden columns [w*32, w*32+32)
// across all four gates -> 4 gates x 2 col-frags x 2 row-frags = 16
// accumulators of v_mfma_f32_16x16x32_bf16 over K=256 (8 k-slices of 32).
//
// Operand staging (both through LDS in full cache lines — fragment-shaped
// global loads are TA-issue-bound on CDNA4 and measured 2.4x slower here):
//   * h tile [32][256] bf16, ((row&15)<<4) XOR byte swizzle ->
//     bank-conflict-free 16-lane ds_read_b128 A-fragments;
//   * W_hh pre-tiled on the host to [8][1024][32] (k-slice-contiguous); each
//     64 KB k-slice is cooperatively loaded as contiguous 128 B per thread
//     into 80 B-padded LDS rows (20-dword stride => the 16 consecutive-g
//     B-fragment reads land on 16 distinct banks, conflict-free).
//   * the gates staging buffer for the epilogue ALIASES the W-slice buffer
//     (time-disjoint), keeping the block at 112 KB LDS.
//
// MFMA fragment maps (v_mfma_f32_16x16x32_bf16):
//   A: lane l -> row l&15,  k in [(l>>4)*8, +8)   (8 contiguous bf16, 16 B)
//   B: lane l -> col l&15,  k in [(l>>4)*8, +8)
//   C/D: lane l, reg r -> row (l>>4)*4 + r, col l&15
// Validated against torch.mm via the `raw_gates` debug mode (tests).
#include "common.h"

namespace nerrf {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define LSTM_H 256
#define LSTM_G (4 * LSTM_H)
#define BM 32                  // batch rows per block
#define H_BYTES (LSTM_H * 2)   // 512 B per h row
#define KSLICE 32              // k columns per W slice
#define WROW_B 80              // padded LDS bytes per W row (64 data + 16 pad)

__device__ __forceinline__ unsigned h_swz(unsigned row, unsigned byte_col) {
  return row * H_BYTES + (byte_col ^ ((row & 15u) << 4));
}

template <bool RAW>
__launch_bounds__(512)
__global__ void lstm_step_fused_kernel(
    const __hip_bfloat16* __restrict__ h_prev,   // [B, 256]
    const __hip_bfloat16* __restrict__ w_tiled,  // [8, 1024, 32] k-sliced
    const __hip_bfloat16* __restrict__ xg,       // [B, 1024]
    const __hip_bfloat16* __restrict__ bias,     // [1024]
    const __hip_bfloat16* __restrict__ c_prev,   // [B, 256]
    const float* __restrict__ mask,              // [B] or nullptr
    __hip_bfloat16* __restrict__ h_out,          // [B, 256]
    __hip_bfloat16* __restrict__ c_out,          // [B, 256]
    __hip_bfloat16* __restrict__ gates_act,      // [B, 1024]
    int batch) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* h_lds = smem;                       // 16 KB, swizzled
  char* w_lds = smem + BM * H_BYTES;        // 80 KB (1024 rows x 80 B)
  // epilogue gates tile aliases the W buffer (64 KB of the 80)
  __hip_bfloat16* gacc = reinterpret_cast<__hip_bfloat16*>(w_lds);

  const int row0 = blockIdx.x * BM;
  const int tid = threadIdx.x;
  const int wave = tid / NERRF_WAVE;
  const int lane = tid % NERRF_WAVE;

  // ---- stage h tile (swizzled) -------------------------------------------
  {
    const int r = tid / 16;
    const int chunk = tid % 16;
    const long grow = (long)(row0 + r);
    for (int cc = chunk; cc < 32; cc += 16) {
      uint4 v = make_uint4(0, 0, 0, 0);
      if (grow < batch)
        v = *reinterpret_cast<const uint4*>(
            reinterpret_cast<const char*>(h_prev) + grow * H_BYTES + cc * 16);
      *reinterpret_cast<uint4*>(h_lds + h_swz(r, cc * 16)) = v;
    }
  }

  f32x4 acc[4][2][2];
#pragma unroll
  for (int g = 0; g < 4; ++g)
#pragma unroll
    for (int cf = 0; cf < 2; ++cf)
#pragma unroll
      for (int rf = 0; rf < 2; ++rf) acc[g][cf][rf] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int arow = lane & 15;
  const int kbyte = (lane >> 4) * 16;  // this lane's 16-B k-chunk offset

#pragma unroll 1
  for (int ks = 0; ks < LSTM_H / KSLICE; ++ks) {
    // cooperative W-slice load: 64 KB contiguous -> padded LDS rows.
    // thread t covers g rows 2t and 2t+1 (64 B each, 4x16 B writes).
    __syncthreads();  // previous slice fully consumed
    {
      const long src = (long)ks * LSTM_G * KSLICE;  // elements
      const int g0 = tid * 2;
#pragma unroll
      for (int gg = 0; gg < 2; ++gg) {
        const int g = g0 + gg;
        const uint4* wsrc = reinterpret_cast<const uint4*>(w_tiled + src + (long)g * KSLICE);
#pragma unroll
        for (int cc = 0; cc < 4; ++cc)
          *reinterpret_cast<uint4*>(w_lds + g * WROW_B + cc * 16) = wsrc[cc];
      }
    }
    __syncthreads();

    bf16x8 a_frag[2];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf)
      a_frag[rf] = *reinterpret_cast<const bf16x8*>(
          h_lds + h_swz(rf * 16 + arow, ks * (KSLICE * 2) + kbyte));
#pragma unroll
    for (int g = 0; g < 4; ++g) {
#pragma unroll
      for (int cf = 0; cf < 2; ++cf) {
        const int gcol = g * LSTM_H + wave * 32 + cf * 16 + (lane & 15);
        const bf16x8 b_frag =
            *reinterpret_cast<const bf16x8*>(w_lds + gcol * WROW_B + kbyte);
#pragma unroll
        for (int rf = 0; rf < 2; ++rf)
          acc[g][cf][rf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[rf], b_frag, acc[g][cf][rf], 0, 0, 0);
      }
    }
  }

  // ---- spill accumulators to the (aliased) gates LDS tile -----------------
  __syncthreads();  // all waves done with w_lds before aliasing it
#pragma unroll
  for (int g = 0; g < 4; ++g)
#pragma unroll
    for (int cf = 0; cf < 2; ++cf)
#pragma unroll
      for (int rf = 0; rf < 2; ++rf)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = rf * 16 + (lane >> 4) * 4 + r;
          const int col = g * LSTM_H + wave * 32 + cf * 16 + (lane & 15);
          gacc[row * LSTM_G + col] = __float2bfloat16(acc[g][cf][rf][r]);
        }
  __syncthreads();

  // ---- pointwise epilogue -------------------------------------------------
  {
    const int r = tid / 16;
    const int d0 = (tid % 16) * 16;
    const long grow = (long)(row0 + r);
    if (grow < batch) {
      const float m = (mask != nullptr) ? mask[grow] : 1.0f;
      float gates[4][16];
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const long gbase = (long)g * LSTM_H + d0;
#pragma unroll
        for (int j = 0; j < 16; ++j)
          gates[g][j] = __bfloat162float(gacc[r * LSTM_G + gbase + j]) +
                        __bfloat162float(xg[grow * LSTM_G + gbase + j]) +
                        __bfloat162float(bias[gbase + j]);
      }
      if (RAW) {
#pragma unroll
        for (int g = 0; g < 4; ++g)
#pragma unroll
          for (int j = 0; j < 16; ++j)
            gates_act[grow * LSTM_G + g * LSTM_H + d0 + j] =
                __float2bfloat16(gates[g][j]);
        return;
      }
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const float i = sigmoidf_(gates[0][j]);
        const float f = sigmoidf_(gates[1][j]);
        const float gg = tanhf(gates[2][j]);
        const float o = sigmoidf_(gates[3][j]);
        const float cvj = __bfloat162float(c_prev[grow * LSTM_H + d0 + j]);
        float cn = f * cvj + i * gg;
        float hn = o * tanhf(cn);
        const float hp = __bfloat162float(h_prev[grow * LSTM_H + d0 + j]);
        cn = m * cn + (1.0f - m) * cvj;
        hn = m * hn + (1.0f - m) * hp;
        c_out[grow * LSTM_H + d0 + j] = __float2bfloat16(cn);
        h_out[grow * LSTM_H + d0 + j] = __float2bfloat16(hn);
        gates_act[grow * LSTM_G + d0 + j] = __float2bfloat16(i);
        gates_act[grow * LSTM_G + LSTM_H + d0 + j] = __float2bfloat16(f);
        gates_act[grow * LSTM_G + 2 * LSTM_H + d0 + j] = __float2bfloat16(gg);
        gates_act[grow * LSTM_G + 3 * LSTM_H + d0 + j] = __float2bfloat16(o);
      }
    }
  }
}

void launch_lstm_step_fused(const void* h_prev, const void* w_tiled,
                            const void* xg, const void* bias,
                            const void* c_prev, const float* mask, void* h_out,
                            void* c_out, void* gates_act, int batch, bool raw,
                            hipStream_t s) {
  const int grid = (batch + BM - 1) / BM;
  const size_t lds = BM * H_BYTES + LSTM_G * WROW_B;  // 16 KB + 80 KB
  if (raw) {
    lstm_step_fused_kernel<true><<<grid, 512, lds, s>>>(
        (const __hip_bfloat16*)h_prev, (const __hip_bfloat16*)w_tiled,
        (const __hip_bfloat16*)xg, (const __hip_bfloat16*)bias,
        (const __hip_bfloat16*)c_prev, mask, (__hip_bfloat16*)h_out,
        (__hip_bfloat16*)c_out, (__hip_bfloat16*)gates_act, batch);
  } else {
    lstm_step_fused_kernel<false><<<grid, 512, lds, s>>>(
        (const __hip_bfloat16*)h_prev, (const __hip_bfloat16*)w_tiled,
        (const __hip_bfloat16*)xg, (const __hip_bfloat16*)bias,
        (const __hip_bfloat16*)c_prev, mask, (__hip_bfloat16*)h_out,
        (__hip_bfloat16*)c_out, (__hip_bfloat16*)gates_act, batch);
  }
}

}  // namespace nerrf
