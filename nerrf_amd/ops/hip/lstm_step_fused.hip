// Fully-fused LSTM recurrent step on MFMA — CDNA4 (gfx950), H = 256 only.
//
// One launch computes, for a 32-row batch tile per block:
//     gates_pre = h_prev @ W_hh^T + xg + bias      (MFMA, K = 256)
//     i,f,o = sigmoid; g = tanh; c' = f*c + i*g; h' = o*tanh(c'); mask
// eliminating the separate hipBLASLt GEMM and its [B,4H] HBM round trip
// (gates_pre never touches HBM; the hidden-state tile is staged in LDS).
//
// Geometry: 8 waves (512 threads) per block; wave w owns hidden columns
// [w*32, w*32+32) across all four gates -> 4 gates x 2 col-frags x 2 row-frags
// = 16 accumulators of v_mfma_f32_16x16x32_bf16 over K=256 (8 k-steps).
// h tile [32][256] bf16 sits in LDS with a ((row&15)<<4) XOR byte swizzle so
// the 16-lane ds_read_b128 A-fragment reads are bank-conflict-free
// (cdna_hip_programming.md §6 Guideline 4); W_hh streams from L2 (512 KB,
// resident) as B-fragments.  The pointwise epilogue re-tiles gates through a
// second LDS buffer so every HBM access (xg, c, h', c', gates_act) is a
// coalesced 32 B-per-thread row segment.
//
// MFMA fragment maps used (v_mfma_f32_16x16x32_bf16):
//   A: lane l -> row l&15,  k in [(l>>4)*8, +8)   (8 contiguous bf16, 16 B)
//   B: lane l -> col l&15,  k in [(l>>4)*8, +8)
//   C/D: lane l, reg r -> row (l>>4)*4 + r, col l&15
// Validated against torch.mm by the `raw_gates` debug mode (tests).
#include "common.h"

namespace nerrf {

typedef __bf16 bf16x8 __attribute__((ext_vector_type(8)));
typedef float f32x4 __attribute__((ext_vector_type(4)));

#define LSTM_H 256
#define LSTM_G (4 * LSTM_H)
#define BM 32          // batch rows per block
#define H_BYTES (LSTM_H * 2)  // 512 B per h row

__device__ __forceinline__ unsigned swz(unsigned row, unsigned byte_col) {
  return row * H_BYTES + (byte_col ^ ((row & 15u) << 4));
}

template <bool RAW>
__launch_bounds__(512, 1)
__global__ void lstm_step_fused_kernel(
    const __hip_bfloat16* __restrict__ h_prev,  // [B, 256]
    const __hip_bfloat16* __restrict__ w_hh,    // [1024, 256] row-major
    const __hip_bfloat16* __restrict__ xg,      // [B, 1024]
    const __hip_bfloat16* __restrict__ bias,    // [1024]
    const __hip_bfloat16* __restrict__ c_prev,  // [B, 256]
    const float* __restrict__ mask,             // [B] or nullptr
    __hip_bfloat16* __restrict__ h_out,         // [B, 256]
    __hip_bfloat16* __restrict__ c_out,         // [B, 256]
    __hip_bfloat16* __restrict__ gates_act,     // [B, 1024]
    int batch) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* h_lds = smem;                        // 32*512 = 16 KB (swizzled)
  __hip_bfloat16* gacc = reinterpret_cast<__hip_bfloat16*>(smem + BM * H_BYTES);  // [32][1024] 64 KB

  const int row0 = blockIdx.x * BM;
  const int tid = threadIdx.x;
  const int wave = tid / NERRF_WAVE;   // 0..7
  const int lane = tid % NERRF_WAVE;

  // ---- stage h tile into LDS (swizzled), 32 B per thread ------------------
  {
    const int r = tid / 16;            // 0..31
    const int chunk = tid % 16;        // 16-B chunks 0..15 (+16)
    const long grow = (long)(row0 + r);
    for (int cc = chunk; cc < 32; cc += 16) {
      uint4 v = make_uint4(0, 0, 0, 0);
      if (grow < batch)
        v = *reinterpret_cast<const uint4*>(
            reinterpret_cast<const char*>(h_prev) + grow * H_BYTES + cc * 16);
      *reinterpret_cast<uint4*>(h_lds + swz(r, cc * 16)) = v;
    }
  }
  __syncthreads();

  // ---- MFMA main loop -----------------------------------------------------
  f32x4 acc[4][2][2];  // [gate][col_frag][row_frag]
#pragma unroll
  for (int g = 0; g < 4; ++g)
#pragma unroll
    for (int cf = 0; cf < 2; ++cf)
#pragma unroll
      for (int rf = 0; rf < 2; ++rf) acc[g][cf][rf] = f32x4{0.f, 0.f, 0.f, 0.f};

  const int arow = lane & 15;          // A-fragment row within 16-row frag
  const int kchunk = (lane >> 4) * 8;  // k offset of this lane's 8 elements
#pragma unroll 1
  for (int kk = 0; kk < LSTM_H; kk += 32) {
    bf16x8 a_frag[2];
#pragma unroll
    for (int rf = 0; rf < 2; ++rf) {
      const unsigned r = rf * 16 + arow;
      a_frag[rf] = *reinterpret_cast<const bf16x8*>(h_lds + swz(r, (kk + kchunk) * 2));
    }
#pragma unroll
    for (int g = 0; g < 4; ++g) {
#pragma unroll
      for (int cf = 0; cf < 2; ++cf) {
        const int gcol = g * LSTM_H + wave * 32 + cf * 16 + (lane & 15);
        const bf16x8 b_frag = *reinterpret_cast<const bf16x8*>(
            w_hh + (long)gcol * LSTM_H + kk + kchunk);
#pragma unroll
        for (int rf = 0; rf < 2; ++rf)
          acc[g][cf][rf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              a_frag[rf], b_frag, acc[g][cf][rf], 0, 0, 0);
      }
    }
  }

  // ---- spill accumulators to the gates LDS tile (bf16) --------------------
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

  // ---- pointwise epilogue: thread -> (row, 16-wide d chunk) ---------------
  {
    const int r = tid / 16;
    const int d0 = (tid % 16) * 16;
    const long grow = (long)(row0 + r);
    if (grow < batch) {
      const float m = (mask != nullptr) ? mask[grow] : 1.0f;
      float hv[16], cv[16];
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        cv[j] = __bfloat162float(c_prev[grow * LSTM_H + d0 + j]);
        hv[j] = 0.0f;
      }
      float gates[4][16];
#pragma unroll
      for (int g = 0; g < 4; ++g) {
        const long gbase = (long)g * LSTM_H + d0;
#pragma unroll
        for (int j = 0; j < 16; ++j) {
          const float pre = __bfloat162float(gacc[r * LSTM_G + gbase + j]) +
                            __bfloat162float(xg[grow * LSTM_G + gbase + j]) +
                            __bfloat162float(bias[gbase + j]);
          gates[g][j] = pre;
        }
      }
      if (RAW) {
        // debug/test mode: emit raw gates_pre, skip the state update
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
        float cn = f * cv[j] + i * gg;
        float hn = o * tanhf(cn);
        const float hp = __bfloat162float(h_prev[grow * LSTM_H + d0 + j]);
        cn = m * cn + (1.0f - m) * cv[j];
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

void launch_lstm_step_fused(const void* h_prev, const void* w_hh,
                            const void* xg, const void* bias,
                            const void* c_prev, const float* mask, void* h_out,
                            void* c_out, void* gates_act, int batch, bool raw,
                            hipStream_t s) {
  const int grid = (batch + BM - 1) / BM;
  const size_t lds = BM * H_BYTES + BM * LSTM_G * 2;  // 16 KB + 64 KB
  if (raw) {
    lstm_step_fused_kernel<true><<<grid, 512, lds, s>>>(
        (const __hip_bfloat16*)h_prev, (const __hip_bfloat16*)w_hh,
        (const __hip_bfloat16*)xg, (const __hip_bfloat16*)bias,
        (const __hip_bfloat16*)c_prev, mask, (__hip_bfloat16*)h_out,
        (__hip_bfloat16*)c_out, (__hip_bfloat16*)gates_act, batch);
  } else {
    lstm_step_fused_kernel<false><<<grid, 512, lds, s>>>(
        (const __hip_bfloat16*)h_prev, (const __hip_bfloat16*)w_hh,
        (const __hip_bfloat16*)xg, (const __hip_bfloat16*)bias,
        (const __hip_bfloat16*)c_prev, mask, (__hip_bfloat16*)h_out,
        (__hip_bfloat16*)c_out, (__hip_bfloat16*)gates_act, batch);
  }
}

}  // namespace nerrf
