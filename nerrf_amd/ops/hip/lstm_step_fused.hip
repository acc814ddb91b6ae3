// Fully-fused LSTM recurrent step on MFMA — CDNA4 (gfx950), H = 256 only.
//
// One launch computes, for a 32-row batch tile per block:
//     gates_pre = h_prev @ W_hh^T + xg + bias      (MFMA, K = 256)
//     i,f,o = sigmoid; g = tanh; c' = f*c + i*g; h' = o*tanh(c'); mask
// eliminating the separate hipBLASLt GEMM and its [B,4H] HBM round trip.
//
// v3 structure (v1: fragment-shaped W loads straight from L2 => TA-bound;
// v2: synchronous LDS W staging => full L2 latency exposed per k-slice;
// both measured slower than GEMM+pointwise — see profiles/PROFILES.md):
//   * 8 waves (512 threads); wave w owns hidden cols [w*32, w*32+32) across
//     all 4 gates -> 16 accumulators of v_mfma_f32_16x16x32_bf16, K=256.
//   * h tile [32][256] bf16 in LDS, ((row&15)<<4) XOR swizzle =>
//     conflict-free ds_read_b128 A-fragments.
//   * W_hh pre-tiled [8][1024][32] (k-slice contiguous).  Per k-slice each
//     thread owns 128 B of W; the NEXT slice's global loads are issued
//     BEFORE this slice's MFMAs (async-STAGE split, guide §6 G15) and
//     ds_written after the barrier — HBM/L2 latency hides under compute.
//     LDS W rows padded to 80 B (20-dword stride: the 16 consecutive-g
//     B-fragment lanes land on 16 distinct banks).
//   * epilogue: thread -> (row, 16-wide d chunk); ALL global and LDS
//     traffic is 16-B vectorised (uint4 / 2x ds_read_b128 per gate) —
//     the scalar-bf16 epilogue was the dominant v2 cost at 1 block/CU.
//   * gates LDS tile aliases the W buffer (time-disjoint).
//
// MFMA fragment maps (v_mfma_f32_16x16x32_bf16):
//   A: lane l -> row l&15,  k in [(l>>4)*8, +8);  B: lane l -> col l&15, same k
//   C/D: lane l, reg r -> row (l>>4)*4 + r, col l&15
// Validated against torch.mm via `raw_gates` mode and the fp32 reference
// (tests/test_ops_gpu.py).
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

__device__ __forceinline__ float bf_hi(unsigned u) {
  union { unsigned u; __hip_bfloat162 b2; } cv{u};
  return __bfloat162float(cv.b2.y);
}
__device__ __forceinline__ float bf_lo(unsigned u) {
  union { unsigned u; __hip_bfloat162 b2; } cv{u};
  return __bfloat162float(cv.b2.x);
}
__device__ __forceinline__ unsigned pack_bf2(float lo, float hi) {
  union { unsigned u; __hip_bfloat162 b2; } cv;
  cv.b2 = __hip_bfloat162(__float2bfloat16(lo), __float2bfloat16(hi));
  return cv.u;
}

// unpack a 16-B vector of 8 bf16 into 8 floats starting at dst
__device__ __forceinline__ void unpack8(const uint4 v, float* dst) {
  dst[0] = bf_lo(v.x); dst[1] = bf_hi(v.x);
  dst[2] = bf_lo(v.y); dst[3] = bf_hi(v.y);
  dst[4] = bf_lo(v.z); dst[5] = bf_hi(v.z);
  dst[6] = bf_lo(v.w); dst[7] = bf_hi(v.w);
}

__device__ __forceinline__ uint4 pack16(const float* src) {
  uint4 v;
  v.x = pack_bf2(src[0], src[1]);
  v.y = pack_bf2(src[2], src[3]);
  v.z = pack_bf2(src[4], src[5]);
  v.w = pack_bf2(src[6], src[7]);
  return v;
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
  __hip_bfloat16* gacc = reinterpret_cast<__hip_bfloat16*>(w_lds);  // aliased later

  const int row0 = blockIdx.x * BM;
  const int tid = threadIdx.x;
  const int wave = tid / NERRF_WAVE;
  const int lane = tid % NERRF_WAVE;

  // ---- stage h tile (swizzled) + issue slice-0 W loads --------------------
  const int wg0 = tid * 2;  // this thread's two W rows (g, g+1)
  uint4 wreg[2][4];         // register-staged W slice: 2 rows x 64 B
  {
    const uint4* wsrc = reinterpret_cast<const uint4*>(w_tiled + (long)wg0 * KSLICE);
#pragma unroll
    for (int gg = 0; gg < 2; ++gg)
#pragma unroll
      for (int cc = 0; cc < 4; ++cc) wreg[gg][cc] = wsrc[gg * 4 + cc];
  }
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
  const int kbyte = (lane >> 4) * 16;

#pragma unroll 1
  for (int ks = 0; ks < LSTM_H / KSLICE; ++ks) {
    // write the register-staged slice into padded LDS rows
    __syncthreads();  // previous slice fully consumed
#pragma unroll
    for (int gg = 0; gg < 2; ++gg)
#pragma unroll
      for (int cc = 0; cc < 4; ++cc)
        *reinterpret_cast<uint4*>(w_lds + (wg0 + gg) * WROW_B + cc * 16) = wreg[gg][cc];
    __syncthreads();
    // issue NEXT slice's loads before this slice's MFMAs (latency hides
    // under the 16-MFMA compute phase)
    if (ks + 1 < LSTM_H / KSLICE) {
      const uint4* wsrc = reinterpret_cast<const uint4*>(
          w_tiled + (long)(ks + 1) * LSTM_G * KSLICE + (long)wg0 * KSLICE);
#pragma unroll
      for (int gg = 0; gg < 2; ++gg)
#pragma unroll
        for (int cc = 0; cc < 4; ++cc) wreg[gg][cc] = wsrc[gg * 4 + cc];
    }

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

  // ---- pointwise epilogue: fully 16-B vectorised --------------------------
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
        const uint4* gl = reinterpret_cast<const uint4*>(gacc + r * LSTM_G + gbase);
        const uint4* xl = reinterpret_cast<const uint4*>(xg + grow * LSTM_G + gbase);
        const uint4* bl = reinterpret_cast<const uint4*>(bias + gbase);
        float a8[8], b8[8], c8[8];
#pragma unroll
        for (int half = 0; half < 2; ++half) {
          unpack8(gl[half], a8);
          unpack8(xl[half], b8);
          unpack8(bl[half], c8);
#pragma unroll
          for (int j = 0; j < 8; ++j)
            gates[g][half * 8 + j] = a8[j] + b8[j] + c8[j];
        }
      }
      if (RAW) {
#pragma unroll
        for (int g = 0; g < 4; ++g) {
          uint4* go = reinterpret_cast<uint4*>(gates_act + grow * LSTM_G + g * LSTM_H + d0);
          go[0] = pack16(&gates[g][0]);
          go[1] = pack16(&gates[g][8]);
        }
        return;
      }
      float cv[16], hv[16], iv[16], fv[16], gv[16], ov[16];
      {
        const uint4* cl = reinterpret_cast<const uint4*>(c_prev + grow * LSTM_H + d0);
        const uint4* hl = reinterpret_cast<const uint4*>(h_prev + grow * LSTM_H + d0);
        unpack8(cl[0], cv); unpack8(cl[1], cv + 8);
        unpack8(hl[0], hv); unpack8(hl[1], hv + 8);
      }
#pragma unroll
      for (int j = 0; j < 16; ++j) {
        const float i = sigmoidf_(gates[0][j]);
        const float f = sigmoidf_(gates[1][j]);
        const float gg = tanhf(gates[2][j]);
        const float o = sigmoidf_(gates[3][j]);
        float cn = f * cv[j] + i * gg;
        float hn = o * tanhf(cn);
        cn = m * cn + (1.0f - m) * cv[j];
        hn = m * hn + (1.0f - m) * hv[j];
        iv[j] = i; fv[j] = f; gv[j] = gg; ov[j] = o;
        cv[j] = cn; hv[j] = hn;  // reuse as outputs
      }
      uint4* co = reinterpret_cast<uint4*>(c_out + grow * LSTM_H + d0);
      uint4* ho = reinterpret_cast<uint4*>(h_out + grow * LSTM_H + d0);
      co[0] = pack16(cv); co[1] = pack16(cv + 8);
      ho[0] = pack16(hv); ho[1] = pack16(hv + 8);
      uint4* ga = reinterpret_cast<uint4*>(gates_act + grow * LSTM_G + d0);
      ga[0] = pack16(iv); ga[1] = pack16(iv + 8);
      ga = reinterpret_cast<uint4*>(gates_act + grow * LSTM_G + LSTM_H + d0);
      ga[0] = pack16(fv); ga[1] = pack16(fv + 8);
      ga = reinterpret_cast<uint4*>(gates_act + grow * LSTM_G + 2 * LSTM_H + d0);
      ga[0] = pack16(gv); ga[1] = pack16(gv + 8);
      ga = reinterpret_cast<uint4*>(gates_act + grow * LSTM_G + 3 * LSTM_H + d0);
      ga[0] = pack16(ov); ga[1] = pack16(ov + 8);
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
