// Fused LSTM recurrent step — CDNA4 (gfx950), bf16, H = 256 only.
//
// Round-1 ran each timestep as {hipBLASLt GEMM writes hg -> fused pointwise
// reads hg}: the [B, 4H] bf16 pre-activation slab (131 MB at the production
// B = 64k) made a full HBM round trip per step per direction, and the
// backward did the mirror trip for grad_gates @ W_hh.  These kernels fuse
// the recurrent GEMM with the gate pointwise in both directions:
//
//   fwd:  one launch computes pre = h_prev @ W_hh^T in MFMA f32
//         accumulators and applies the LSTM epilogue in-register —
//         hg never exists in HBM.  Traffic/step: 394 MB vs 656 MB split.
//   bwd:  one launch computes the gate gradients, stages grad_gates
//         through LDS, and runs grad_h = ghp + gg @ W_hh as a second MFMA
//         phase of the same launch — gg goes to HBM once (the weight-grad
//         GEMMs still need it), not written+reread.
//
// Why this succeeds where round-1's lstm_step_fused (32-row blocks, full-W
// re-read per block — profiles/PROFILES.md ladder) lost: BM = 64 halves the
// W L2 traffic again, W fragments stream from L2 under 8-wave MFMA cover
// instead of synchronous LDS slices, the xg operand is prefetched into LDS
// during the GEMM phase (register-staged writes, no glds-drain trap), and
// the epilogue is wave-local: each wave owns 32 hidden units and computes
// all four of their gate column groups {j, H+j, 2H+j, 3H+j}, so no
// cross-wave gate exchange is needed.
//
// Geometry (both kernels): 512 threads = 8 waves; BM = 64 batch rows per
// block; wave w owns hidden units [32w, 32w+32).  MFMA
// v_mfma_f32_16x16x32_bf16, fragment maps as sage_fused.hip
// (A row = l&15, B out-col = l&15, k-chunk (l>>4)*8; C/D row = (l>>4)*4+r,
// col = l&15).  LDS at the 160 KiB cap (1 block/CU by design; overlap
// comes from the in-block phase structure).
//
// Numerics: f32 gate accumulation (the split path rounds hg to bf16 first,
// so the fused forward is slightly MORE precise); activated gates stored
// bf16 exactly as the split path.  Validated against
// ops/reference.py::lstm_pointwise_{fwd,bwd}_ref + torch.mm
// (tests/test_ops_gpu.py).  SURVEY.md §2a "fused LSTM cell" obligation.
#include "common.h"

namespace nerrf {

typedef __bf16 rbf16x8 __attribute__((ext_vector_type(8)));
typedef float rf32x4 __attribute__((ext_vector_type(4)));

#define REC_H 256
#define REC_G 1024
#define REC_BM 64
#define REC_HROW_B 512   // h tile row bytes (256 * 2)
#define REC_GROW_B 2048  // gate tile row bytes (1024 * 2)

// XOR swizzle for conflict-free ds_read_b128 column-slice reads
__device__ __forceinline__ unsigned rec_swz(unsigned row_byte0, unsigned row,
                                            unsigned byte_col) {
  return row_byte0 + (byte_col ^ ((row & 15u) << 4));
}

// ---------------------------------------------------------------------------
// forward: h_out/c_out/gates = LSTM(h_prev @ W_hh^T + xg + bias)
// ---------------------------------------------------------------------------

__launch_bounds__(512)
__global__ void lstm_rec_fwd_kernel(
    const __hip_bfloat16* __restrict__ h_prev,  // [B(hprev_stride), 256]
    const __hip_bfloat16* __restrict__ w_hh,    // [1024, 256] (out, in)
    const __hip_bfloat16* __restrict__ xg,      // [B(xg_stride), 1024]
    const __hip_bfloat16* __restrict__ bias,    // [1024]
    const __hip_bfloat16* __restrict__ c_prev,  // [B, 256]
    const float* __restrict__ mask,             // [B] or nullptr
    __hip_bfloat16* __restrict__ h_out,         // [B(hout_stride), 256]
    __hip_bfloat16* __restrict__ c_out,         // [B, 256]
    __hip_bfloat16* __restrict__ gates_act,     // [B, 1024] or nullptr
    int batch, long hprev_stride, long xg_stride, long hout_stride) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* h_lds = smem;                           // 32 KB, swizzled
  char* gpre_lds = smem + REC_BM * REC_HROW_B;  // 128 KB, linear bf16

  const int row0 = blockIdx.x * REC_BM;
  const int tid = threadIdx.x;
  const int wave = tid / NERRF_WAVE;
  const int lane = tid % NERRF_WAVE;
  const int j0 = wave * 32;  // this wave's hidden-unit slice

  // ---- stage h tile: 64 rows x 512 B, swizzled ---------------------------
  {
    const int r = tid >> 3;               // 0..63
    const int c0 = tid & 7;               // interleaved: consecutive lanes
    const long grow = (long)(row0 + r);   // read consecutive 16-B chunks
#pragma unroll
    for (int cc = 0; cc < 4; ++cc) {
      const int chunk = c0 + cc * 8;
      uint4 v = make_uint4(0, 0, 0, 0);
      if (grow < batch)
        v = *reinterpret_cast<const uint4*>(
            reinterpret_cast<const char*>(h_prev + grow * hprev_stride) +
            chunk * 16);
      *reinterpret_cast<uint4*>(
          h_lds + rec_swz(r * REC_HROW_B, r, chunk * 16)) = v;
    }
  }
  __syncthreads();

  // ---- MFMA phase, gate-major: one gate's accumulators (8 f32x4) live at
  // a time, dumped to LDS before the next gate — the all-gates-resident
  // variant held 128 acc + fragment regs and spilled 44 VGPRs into the
  // MFMA loop.  A fragments re-read per gate from LDS (cheap; 16 KB/wave)
  // and the LDS dump of gate g overlaps the next gate's loads. ----------
  const int frag_col = lane & 15;
  const int kchunk = (lane >> 4) * 8;  // element offset within k-step
#pragma unroll 1
  for (int g = 0; g < 4; ++g) {
    rf32x4 acc[4][2];
#pragma unroll
    for (int rf = 0; rf < 4; ++rf)
#pragma unroll
      for (int cf = 0; cf < 2; ++cf) acc[rf][cf] = rf32x4{0.f, 0.f, 0.f, 0.f};
#pragma unroll
    for (int ks = 0; ks < REC_H / 32; ++ks) {
      // B fragments from L2 (w_hh is ~0.5 MB, shared by every block)
      rbf16x8 bfr[2];
#pragma unroll
      for (int cf = 0; cf < 2; ++cf) {
        const int orow = g * REC_H + j0 + cf * 16 + frag_col;
        bfr[cf] = *reinterpret_cast<const rbf16x8*>(
            w_hh + (long)orow * REC_H + ks * 32 + kchunk);
      }
#pragma unroll
      for (int rf = 0; rf < 4; ++rf) {
        const int arow = rf * 16 + frag_col;
        const rbf16x8 afr = *reinterpret_cast<const rbf16x8*>(
            h_lds +
            rec_swz(arow * REC_HROW_B, arow, ks * 64 + (lane >> 4) * 16));
#pragma unroll
        for (int cf = 0; cf < 2; ++cf)
          acc[rf][cf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr, bfr[cf], acc[rf][cf], 0, 0, 0);
      }
    }
    // dump this gate's pre-activations to LDS (bf16) so the pointwise
    // phase can do 16-B row-major global I/O — scalar 2-B global access
    // is the 2-2.5x CDNA4 loss the v1 epilogue measured (Guideline 13)
#pragma unroll
    for (int cf = 0; cf < 2; ++cf) {
      const int j = j0 + cf * 16 + frag_col;
#pragma unroll
      for (int rf = 0; rf < 4; ++rf)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int row = rf * 16 + (lane >> 4) * 4 + r;
          *reinterpret_cast<__hip_bfloat16*>(
              gpre_lds + row * REC_GROW_B + (g * REC_H + j) * 2) =
              __float2bfloat16(acc[rf][cf][r]);
        }
    }
  }
  __syncthreads();

  // ---- pointwise phase: thread -> (row, 8-col chunk); all global traffic
  // 16-B and row-contiguous (2 rows x 512 B per wave-instruction).
  // unroll 1: a 4x unroll keeps ~64 VT8 registers live and spills ---------
#pragma unroll 1
  for (int p = 0; p < 4; ++p) {
    const int row = p * 16 + (tid >> 5);  // 0..63
    const int ch = (tid & 31) * 8;        // hidden-unit chunk base (8 els)
    const long grow = (long)(row0 + row);
    if (grow >= batch) continue;
    const char* pre_p = gpre_lds + row * REC_GROW_B;
    const rbf16x8 p_i = *reinterpret_cast<const rbf16x8*>(pre_p + ch * 2);
    const rbf16x8 p_f =
        *reinterpret_cast<const rbf16x8*>(pre_p + (REC_H + ch) * 2);
    const rbf16x8 p_g =
        *reinterpret_cast<const rbf16x8*>(pre_p + (2 * REC_H + ch) * 2);
    const rbf16x8 p_o =
        *reinterpret_cast<const rbf16x8*>(pre_p + (3 * REC_H + ch) * 2);
    const __hip_bfloat16* xrow = xg + grow * xg_stride;
    const rbf16x8 x_i = *reinterpret_cast<const rbf16x8*>(xrow + ch);
    const rbf16x8 x_f = *reinterpret_cast<const rbf16x8*>(xrow + REC_H + ch);
    const rbf16x8 x_g =
        *reinterpret_cast<const rbf16x8*>(xrow + 2 * REC_H + ch);
    const rbf16x8 x_o =
        *reinterpret_cast<const rbf16x8*>(xrow + 3 * REC_H + ch);
    const rbf16x8 b_i = *reinterpret_cast<const rbf16x8*>(bias + ch);
    const rbf16x8 b_f = *reinterpret_cast<const rbf16x8*>(bias + REC_H + ch);
    const rbf16x8 b_g =
        *reinterpret_cast<const rbf16x8*>(bias + 2 * REC_H + ch);
    const rbf16x8 b_o =
        *reinterpret_cast<const rbf16x8*>(bias + 3 * REC_H + ch);
    const rbf16x8 cp_v =
        *reinterpret_cast<const rbf16x8*>(c_prev + grow * REC_H + ch);
    rbf16x8 hp_v{};
    const float m = (mask != nullptr) ? mask[grow] : 1.0f;
    if (mask != nullptr)
      hp_v = *reinterpret_cast<const rbf16x8*>(h_prev + grow * hprev_stride +
                                               ch);
    rbf16x8 ho_v, co_v, ga_i, ga_f, ga_g, ga_o;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float i = sigmoidf_(__bfloat162float(p_i[e]) +
                                __bfloat162float(x_i[e]) +
                                __bfloat162float(b_i[e]));
      const float f = sigmoidf_(__bfloat162float(p_f[e]) +
                                __bfloat162float(x_f[e]) +
                                __bfloat162float(b_f[e]));
      const float g = tanhf(__bfloat162float(p_g[e]) +
                            __bfloat162float(x_g[e]) +
                            __bfloat162float(b_g[e]));
      const float o = sigmoidf_(__bfloat162float(p_o[e]) +
                                __bfloat162float(x_o[e]) +
                                __bfloat162float(b_o[e]));
      const float cp = __bfloat162float(cp_v[e]);
      float cn = f * cp + i * g;
      float hn = o * tanhf(cn);
      if (mask != nullptr) {
        cn = m * cn + (1.0f - m) * cp;
        hn = m * hn + (1.0f - m) * __bfloat162float(hp_v[e]);
      }
      co_v[e] = (__bf16)cn;
      ho_v[e] = (__bf16)hn;
      ga_i[e] = (__bf16)i;
      ga_f[e] = (__bf16)f;
      ga_g[e] = (__bf16)g;
      ga_o[e] = (__bf16)o;
    }
    *reinterpret_cast<rbf16x8*>(c_out + grow * REC_H + ch) = co_v;
    *reinterpret_cast<rbf16x8*>(h_out + grow * hout_stride + ch) = ho_v;
    if (gates_act != nullptr) {
      __hip_bfloat16* gp = gates_act + grow * REC_G + ch;
      *reinterpret_cast<rbf16x8*>(gp) = ga_i;
      *reinterpret_cast<rbf16x8*>(gp + REC_H) = ga_f;
      *reinterpret_cast<rbf16x8*>(gp + 2 * REC_H) = ga_g;
      *reinterpret_cast<rbf16x8*>(gp + 3 * REC_H) = ga_o;
    }
  }
}

// ---------------------------------------------------------------------------
// backward: grad_gates + grad_c_prev + (grad_h = ghp + gg @ W_hh) fused
// ---------------------------------------------------------------------------

__launch_bounds__(512)
__global__ void lstm_rec_bwd_kernel(
    const __hip_bfloat16* __restrict__ grad_h,     // [B, 256] recurrent grad
    const __hip_bfloat16* __restrict__ grad_out_t, // [B(gout_stride), 256] or null
    const __hip_bfloat16* __restrict__ grad_c,     // [B, 256]
    const __hip_bfloat16* __restrict__ gates_act,  // [B, 1024]
    const __hip_bfloat16* __restrict__ c_prev,     // [B, 256]
    const __hip_bfloat16* __restrict__ w_hh_t,     // [256, 1024] = W_hh^T contig
    const float* __restrict__ mask,                // [B] or nullptr
    __hip_bfloat16* __restrict__ grad_gates,  // [B(row-stride gg_stride), 1024]
    __hip_bfloat16* __restrict__ grad_c_prev,      // [B, 256]
    __hip_bfloat16* __restrict__ grad_h_out,       // [B, 256]
    int batch, long gout_stride, long gg_stride) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* gg_lds = smem;                         // 128 KB, swizzled
  char* ghp_lds = smem + REC_BM * REC_GROW_B;  // 32 KB, linear bf16

  const int row0 = blockIdx.x * REC_BM;
  const int tid = threadIdx.x;
  const int wave = tid / NERRF_WAVE;
  const int lane = tid % NERRF_WAVE;
  const int j0 = wave * 32;

  // ---- phase 1: gate gradients; per pass a wave covers 16 rows x 4
  // 8-unit chunks of its j slice (16 rows x 64 B per instruction — the
  // lane->row map of v1 made every load 2048-B-strided / uncoalesced) ----
#pragma unroll
  for (int p = 0; p < 4; ++p) {
    const int row = p * 16 + (lane >> 2);  // 0..63
    const int cc = lane & 3;
    const int jc = j0 + cc * 8;
    const long grow = (long)(row0 + row);
    const bool live = grow < batch;
    const float m = (live && mask != nullptr) ? mask[grow] : 1.0f;
    rbf16x8 ga_i{}, ga_f{}, ga_g{}, ga_o{}, cp_v{}, gh_v{}, gc_v{}, go_v{};
    if (live) {
      const __hip_bfloat16* gp = gates_act + grow * REC_G + jc;
      ga_i = *reinterpret_cast<const rbf16x8*>(gp);
      ga_f = *reinterpret_cast<const rbf16x8*>(gp + REC_H);
      ga_g = *reinterpret_cast<const rbf16x8*>(gp + 2 * REC_H);
      ga_o = *reinterpret_cast<const rbf16x8*>(gp + 3 * REC_H);
      cp_v = *reinterpret_cast<const rbf16x8*>(c_prev + grow * REC_H + jc);
      gh_v = *reinterpret_cast<const rbf16x8*>(grad_h + grow * REC_H + jc);
      gc_v = *reinterpret_cast<const rbf16x8*>(grad_c + grow * REC_H + jc);
      if (grad_out_t != nullptr)
        go_v = *reinterpret_cast<const rbf16x8*>(grad_out_t +
                                                 grow * gout_stride + jc);
    }
    rbf16x8 gg_i, gg_f, gg_g, gg_o, gcp_v, ghp_v;
#pragma unroll
    for (int e = 0; e < 8; ++e) {
      const float i = __bfloat162float(ga_i[e]);
      const float f = __bfloat162float(ga_f[e]);
      const float g = __bfloat162float(ga_g[e]);
      const float o = __bfloat162float(ga_o[e]);
      const float cp = __bfloat162float(cp_v[e]);
      const float tcn = tanhf(f * cp + i * g);
      float gh_in = __bfloat162float(gh_v[e]);
      if (grad_out_t != nullptr) gh_in += __bfloat162float(go_v[e]);
      const float gc_in = __bfloat162float(gc_v[e]);
      const float gh = gh_in * m;
      const float gc = gc_in * m;
      const float d_o = gh * tcn;
      const float d_c = gc + gh * o * (1.0f - tcn * tcn);
      gcp_v[e] = (__bf16)(d_c * f + gc_in * (1.0f - m));
      ghp_v[e] = (__bf16)(gh_in * (1.0f - m));
      gg_i[e] = (__bf16)(d_c * g * i * (1.0f - i));
      gg_f[e] = (__bf16)(d_c * cp * f * (1.0f - f));
      gg_g[e] = (__bf16)(d_c * i * (1.0f - g * g));
      gg_o[e] = (__bf16)(d_o * o * (1.0f - o));
    }
    // grad_gates to HBM (the weight-grad GEMMs read it) AND to LDS (the
    // in-launch grad_h GEMM reads it as swizzled A fragments)
    if (live) {
      __hip_bfloat16* op = grad_gates + grow * gg_stride + jc;
      *reinterpret_cast<rbf16x8*>(op) = gg_i;
      *reinterpret_cast<rbf16x8*>(op + REC_H) = gg_f;
      *reinterpret_cast<rbf16x8*>(op + 2 * REC_H) = gg_g;
      *reinterpret_cast<rbf16x8*>(op + 3 * REC_H) = gg_o;
      *reinterpret_cast<rbf16x8*>(grad_c_prev + grow * REC_H + jc) = gcp_v;
    }
    *reinterpret_cast<rbf16x8*>(
        gg_lds + rec_swz(row * REC_GROW_B, row, jc * 2)) = gg_i;
    *reinterpret_cast<rbf16x8*>(
        gg_lds + rec_swz(row * REC_GROW_B, row, (REC_H + jc) * 2)) = gg_f;
    *reinterpret_cast<rbf16x8*>(
        gg_lds + rec_swz(row * REC_GROW_B, row, (2 * REC_H + jc) * 2)) = gg_g;
    *reinterpret_cast<rbf16x8*>(
        gg_lds + rec_swz(row * REC_GROW_B, row, (3 * REC_H + jc) * 2)) = gg_o;
    *reinterpret_cast<rbf16x8*>(ghp_lds + row * REC_HROW_B +
                                ((jc * 2) ^ ((row & 7u) << 4))) = ghp_v;
  }
  __syncthreads();

  // ---- phase 2: grad_h[64,256] = gg[64,1024] @ W_hh[1024,256] ------------
  rf32x4 acc[4][2];
#pragma unroll
  for (int rf = 0; rf < 4; ++rf)
#pragma unroll
    for (int cf = 0; cf < 2; ++cf) acc[rf][cf] = rf32x4{0.f, 0.f, 0.f, 0.f};

  const int frag_col = lane & 15;
  const int kchunk = (lane >> 4) * 8;
#pragma unroll 8
  for (int ks = 0; ks < REC_G / 32; ++ks) {
    rbf16x8 bfr[2];
#pragma unroll
    for (int cf = 0; cf < 2; ++cf) {
      const int orow = j0 + cf * 16 + frag_col;  // output hidden unit
      bfr[cf] = *reinterpret_cast<const rbf16x8*>(
          w_hh_t + (long)orow * REC_G + ks * 32 + kchunk);
    }
    rbf16x8 afr[4];
#pragma unroll
    for (int rf = 0; rf < 4; ++rf) {
      const int arow = rf * 16 + frag_col;
      afr[rf] = *reinterpret_cast<const rbf16x8*>(
          gg_lds + rec_swz(arow * REC_GROW_B, arow,
                           ks * 64 + (lane >> 4) * 16));
    }
#pragma unroll
    for (int rf = 0; rf < 4; ++rf)
#pragma unroll
      for (int cf = 0; cf < 2; ++cf)
        acc[rf][cf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
            afr[rf], bfr[cf], acc[rf][cf], 0, 0, 0);
  }

  // ---- epilogue: grad_h_out = acc + ghp ----------------------------------
#pragma unroll
  for (int cf = 0; cf < 2; ++cf) {
    const int j = j0 + cf * 16 + frag_col;
#pragma unroll
    for (int rf = 0; rf < 4; ++rf) {
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int row = rf * 16 + (lane >> 4) * 4 + r;
        const long grow = (long)(row0 + row);
        if (grow >= batch) continue;
        const float ghp = __bfloat162float(
            *reinterpret_cast<const __hip_bfloat16*>(
                ghp_lds + row * REC_HROW_B +
                (((unsigned)(j * 2) & ~15u) ^ ((row & 7u) << 4)) +
                ((j * 2) & 15)));
        grad_h_out[grow * REC_H + j] = __float2bfloat16(acc[rf][cf][r] + ghp);
      }
    }
  }
}

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

void launch_lstm_rec_fwd(const void* h_prev, const void* w_hh, const void* xg,
                         const void* bias, const void* c_prev,
                         const float* mask, void* h_out, void* c_out,
                         void* gates_act, int batch, long hprev_stride,
                         long xg_stride, long hout_stride, hipStream_t s) {
  const int grid = (batch + REC_BM - 1) / REC_BM;
  const size_t lds = REC_BM * (REC_HROW_B + REC_GROW_B);  // 160 KB
  lstm_rec_fwd_kernel<<<grid, 512, lds, s>>>(
      (const __hip_bfloat16*)h_prev, (const __hip_bfloat16*)w_hh,
      (const __hip_bfloat16*)xg, (const __hip_bfloat16*)bias,
      (const __hip_bfloat16*)c_prev, mask, (__hip_bfloat16*)h_out,
      (__hip_bfloat16*)c_out, (__hip_bfloat16*)gates_act, batch, hprev_stride,
      xg_stride, hout_stride);
}

void launch_lstm_rec_bwd(const void* grad_h, const void* grad_out_t,
                         const void* grad_c, const void* gates_act,
                         const void* c_prev, const void* w_hh_t,
                         const float* mask, void* grad_gates,
                         void* grad_c_prev, void* grad_h_out, int batch,
                         long gout_stride, long gg_stride, hipStream_t s) {
  const int grid = (batch + REC_BM - 1) / REC_BM;
  const size_t lds = REC_BM * (REC_GROW_B + REC_HROW_B);  // 160 KB
  lstm_rec_bwd_kernel<<<grid, 512, lds, s>>>(
      (const __hip_bfloat16*)grad_h, (const __hip_bfloat16*)grad_out_t,
      (const __hip_bfloat16*)grad_c, (const __hip_bfloat16*)gates_act,
      (const __hip_bfloat16*)c_prev, (const __hip_bfloat16*)w_hh_t, mask,
      (__hip_bfloat16*)grad_gates, (__hip_bfloat16*)grad_c_prev,
      (__hip_bfloat16*)grad_h_out, batch, gout_stride, gg_stride);
}

}  // namespace nerrf
