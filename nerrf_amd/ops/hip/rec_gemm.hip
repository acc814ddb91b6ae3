// Recurrent-step GEMM for the BiLSTM hidden path — CDNA4 (gfx950).
//
// The training recurrence issues 400 fwd + 400 bwd GEMMs per step of shape
// hg[B,1024] = h[B,256] @ W_hh[1024,256]^T (B ~ 64k sequences at the bench
// config).  hipBLASLt's best tunings (MT256x256x64 / MT256x256x32,
// profiles/train_kstats_r02.txt) run them at ~2.2 TB/s effective — the
// K=256 mainloop is too short to hide L2/HBM latency behind, and the C
// write epilogue dominates (C is 128 MB vs A's 32 MB per call).
//
// This kernel is built for exactly that shape:
//   * NT layout end-to-end: A rows and W rows are both k-contiguous, so
//     every MFMA fragment is a single 16-B load (no LDS transpose on the
//     load side at all; the proj kernels above need staging because they
//     reuse A across two directions).
//   * A strip [BM=128, 256] staged once in 64 KB of XOR-swizzled LDS and
//     reused by all four 256-column chunks of N=1024.
//   * W is 0.5 MB total: per-wave column slices keep each fragment in L2
//     after the first block touches it.
//   * The C tile is converted to bf16 and round-tripped through a small
//     per-wave LDS buffer so global stores are 16-B per lane (the 2-B
//     scalar acc-layout stores in proj_fwd_dual are its measured
//     bottleneck per Guideline 13: half-line write segments).
//   * K/32 = 8 MFMA steps fully unrolled with the B fragments for step
//     ks+1 prefetched during step ks (same software pipeline as
//     proj_fwd_dual, stream_gemm.hip:105-131).
//
// The same kernel shape also serves the recurrence backward's
// grad_h[B,256] = gg[B,1024] @ W_hh[1024,256]: pass W^T contiguous
// ([256,1024], already materialised once per (layer,direction) for the
// forward) and the roles of K/N swap to K=1024, N=256 — rec_dgrad below.
//
// Validated against torch.matmul fp32 (tests/test_ops_gpu.py).
#include "common.h"

namespace nerrf {

typedef __bf16 rbf16x8 __attribute__((ext_vector_type(8)));
typedef float rf32x4 __attribute__((ext_vector_type(4)));

#define RG_BM 128

__device__ __forceinline__ unsigned rg_swz(unsigned row, unsigned byte_col,
                                           unsigned row_bytes) {
  return row * row_bytes + (byte_col ^ ((row & 15u) << 4));
}

// ---------------------------------------------------------------------------
// rec_gemm_fwd: C[M, 1024] = A[M, 256] @ W[1024, 256]^T
// 512 threads = 8 waves; wave w owns a 32-col slice of each 256-col chunk.
// LDS: 64 KB A strip + 8 KB/wave C staging (reused per rf tile).
// ---------------------------------------------------------------------------
__launch_bounds__(512)
__global__ void rec_gemm_fwd_kernel(
    const __hip_bfloat16* __restrict__ a,  // [M(row-stride a_stride), 256]
    const __hip_bfloat16* __restrict__ w,  // [1024, 256]
    __hip_bfloat16* __restrict__ c,        // [M(row-stride c_stride), 1024]
    long m_rows, long a_stride, long c_stride) {
  constexpr int K = 256;
  constexpr int ROW_B = K * 2;  // 512 B per A row
  extern __shared__ __attribute__((aligned(16))) char smem[];
  char* a_lds = smem;                       // 64 KB swizzled A strip
  char* st_lds = smem + RG_BM * ROW_B;      // 8 waves x 1 KB C staging

  const long row0 = (long)blockIdx.x * RG_BM;
  const int tid = threadIdx.x;
  const int lane = tid % NERRF_WAVE;
  const int wave = tid / NERRF_WAVE;

  // ---- stage the A strip once: 128 rows x 512 B --------------------------
  {
    const int r = tid >> 2;      // 0..127
    const int c0 = tid & 3;      // interleaved 16-B chunks within the row
    const long grow = row0 + r;
#pragma unroll
    for (int cc = 0; cc < 8; ++cc) {
      const int chunk = c0 + cc * 4;
      uint4 v = make_uint4(0, 0, 0, 0);
      if (grow < m_rows)
        v = *reinterpret_cast<const uint4*>(
            reinterpret_cast<const char*>(a + grow * a_stride) + chunk * 16);
      *reinterpret_cast<uint4*>(a_lds + rg_swz(r, chunk * 16, ROW_B)) = v;
    }
  }
  __syncthreads();

  const int frag_col = lane & 15;
  const int kchunk_b = (lane >> 4) * 16;  // byte offset within 64-B k-step
  char* my_st = st_lds + wave * 1024;     // [16 rows][32 cols] bf16

#pragma unroll 1
  for (int nc = 0; nc < 4; ++nc) {  // 256-col chunks of N = 1024
    const int col0 = nc * 256 + wave * 32;
    rf32x4 acc[8][2];
#pragma unroll
    for (int rf = 0; rf < 8; ++rf)
#pragma unroll
      for (int cf = 0; cf < 2; ++cf) acc[rf][cf] = rf32x4{0.f, 0.f, 0.f, 0.f};
    const char* wrow0 = reinterpret_cast<const char*>(w) +
                        (long)(col0 + frag_col) * ROW_B + kchunk_b;
    const char* wrow1 = wrow0 + 16 * ROW_B;
    rbf16x8 b_cur[2], b_nxt[2];
    b_cur[0] = *reinterpret_cast<const rbf16x8*>(wrow0);
    b_cur[1] = *reinterpret_cast<const rbf16x8*>(wrow1);
#pragma unroll 2
    for (int ks = 0; ks < K / 32; ++ks) {
      if (ks + 1 < K / 32) {
        b_nxt[0] = *reinterpret_cast<const rbf16x8*>(wrow0 + (ks + 1) * 64);
        b_nxt[1] = *reinterpret_cast<const rbf16x8*>(wrow1 + (ks + 1) * 64);
      }
      rbf16x8 afr[8];
#pragma unroll
      for (int rf = 0; rf < 8; ++rf)
        afr[rf] = *reinterpret_cast<const rbf16x8*>(
            a_lds + rg_swz(rf * 16 + frag_col, ks * 64 + kchunk_b, ROW_B));
      __builtin_amdgcn_sched_group_barrier(0x100, 8, 0);  // DS_READ x8
      __builtin_amdgcn_sched_group_barrier(0x8, 16, 0);   // MFMA x16
#pragma unroll
      for (int rf = 0; rf < 8; ++rf)
#pragma unroll
        for (int cf = 0; cf < 2; ++cf)
          acc[rf][cf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[rf], b_cur[cf], acc[rf][cf], 0, 0, 0);
      b_cur[0] = b_nxt[0];
      b_cur[1] = b_nxt[1];
    }
    // ---- store: acc layout -> per-wave LDS -> 16-B row-vector stores -----
    // acc element r of (rf, cf) is C row rf*16 + (lane>>4)*4 + r, col
    // col0 + cf*16 + (lane&15); staging one rf tile ([16, 32] bf16 = 1 KB)
    // at a time lets lane l re-read row l>>2, 16-B chunk l&3 and write a
    // full 64-B segment per 4 lanes (vs 2-B scalar stores straight from
    // the acc layout).  The buffers are per-wave, so the write->read
    // hazard is intra-wave only: a scheduling fence pins program order and
    // hipcc inserts the lgkmcnt wait from the may-alias dependence
    // (block barriers here measured no faster — 48.4 vs 51.2 us).
#pragma unroll
    for (int rf = 0; rf < 8; ++rf) {
#pragma unroll
      for (int cf = 0; cf < 2; ++cf) {
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int srow = (lane >> 4) * 4 + r;
          *reinterpret_cast<__hip_bfloat16*>(
              my_st + srow * 64 + (cf * 16 + frag_col) * 2) =
              __float2bfloat16(acc[rf][cf][r]);
        }
      }
      // per-wave private buffer: the write->read hazard is intra-wave, so
      // a scheduling fence suffices (hipcc inserts the lgkmcnt wait from
      // the may-alias dependence); no block barrier needed
      __builtin_amdgcn_sched_barrier(0);
      const int srow = lane >> 2;
      const long grow = row0 + rf * 16 + srow;
      if (grow < m_rows) {
        uint4 v = *reinterpret_cast<const uint4*>(
            my_st + srow * 64 + (lane & 3) * 16);
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(c + grow * c_stride) +
            (col0 + (lane & 3) * 8) * 2) = v;
      }
      __builtin_amdgcn_sched_barrier(0);  // staging reused by next rf tile
    }
  }
}

void launch_rec_gemm_fwd(const void* a, const void* w, void* c, long m_rows,
                         long a_stride, long c_stride, hipStream_t s) {
  const int grid = (int)((m_rows + RG_BM - 1) / RG_BM);
  const size_t lds = RG_BM * 512 + 8 * 1024;  // 64 KB A + 8 KB staging
  rec_gemm_fwd_kernel<<<grid, 512, lds, s>>>(
      (const __hip_bfloat16*)a, (const __hip_bfloat16*)w, (__hip_bfloat16*)c,
      m_rows, a_stride, c_stride);
}


// ---------------------------------------------------------------------------
// rec_gemm_dgrad: C[M, 256] = A[M, 1024] @ Wt[256, 1024]^T + D[M, 256]
// (the recurrence backward's grad_h = grad_h_pass + gg @ W_hh; Wt = W_hh^T
// contiguous, so this is the same NT form as the forward with K/N swapped.)
//
// K=1024 means the A strip cannot sit in LDS whole (256 KB); it streams
// through a double-buffered [128, 128] chunk (2 x 32 KB).  N=256 is covered
// by the 8 waves in ONE pass (wave w owns cols w*32..w*32+31), so each A
// element is read from HBM exactly once and every fragment is k-contiguous.
// The D addend rides the vectorized epilogue: the C tile round-trips LDS
// as bf16 rows and D is read with the same 16-B pattern.
// ---------------------------------------------------------------------------
__launch_bounds__(512)
__global__ void rec_gemm_dgrad_kernel(
    const __hip_bfloat16* __restrict__ a,   // [M(row-stride a_stride), 1024]
    const __hip_bfloat16* __restrict__ wt,  // [256, 1024] (= W_hh^T)
    const __hip_bfloat16* __restrict__ d,   // [M, 256] addend or nullptr
    __hip_bfloat16* __restrict__ c,         // [M, 256]
    long m_rows, long a_stride) {
  constexpr int K = 1024;
  constexpr int WROW_B = K * 2;   // 2048 B per Wt row
  constexpr int CK = 128;         // k-chunk staged per round
  constexpr int CH_B = CK * 2;    // 256 B per LDS row
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto ch_lds = [&](int buf) -> char* {            // 2 x 32 KB chunk bufs
    return smem + buf * (RG_BM * CH_B);
  };
  char* st_lds = smem + 2 * RG_BM * CH_B;          // 8 waves x 1 KB

  const long row0 = (long)blockIdx.x * RG_BM;
  const int tid = threadIdx.x;
  const int lane = tid % NERRF_WAVE;
  const int wave = tid / NERRF_WAVE;
  const int frag_col = lane & 15;
  const int kchunk_b = (lane >> 4) * 16;
  char* my_st = st_lds + wave * 1024;

  // chunk-staging map: 4 threads per row, each covering 4 interleaved
  // 16-B chunks of the row's 256-B chunk slice
  const int s_r = tid >> 2;
  const int s_c0 = tid & 3;
  const long s_row = (row0 + s_r < m_rows) ? row0 + s_r : m_rows - 1;
  const char* s_base = reinterpret_cast<const char*>(a + s_row * a_stride);

  auto stage = [&](int buf, int kc) {
#pragma unroll
    for (int cc = 0; cc < 4; ++cc) {
      const int chunk = s_c0 + cc * 4;
      uint4 v = *reinterpret_cast<const uint4*>(
          s_base + kc * CH_B + chunk * 16);
      *reinterpret_cast<uint4*>(ch_lds(buf) + rg_swz(s_r, chunk * 16, CH_B)) = v;
    }
  };

  stage(0, 0);

  rf32x4 acc[8][2];
#pragma unroll
  for (int rf = 0; rf < 8; ++rf)
#pragma unroll
    for (int cf = 0; cf < 2; ++cf) acc[rf][cf] = rf32x4{0.f, 0.f, 0.f, 0.f};

  const int col0 = wave * 32;
  const char* wrow0 = reinterpret_cast<const char*>(wt) +
                      (long)(col0 + frag_col) * WROW_B + kchunk_b;
  const char* wrow1 = wrow0 + 16 * WROW_B;

  for (int kc = 0; kc < K / CK; ++kc) {
    __syncthreads();  // chunk kc staged (and prior reads of this buf done)
    if (kc + 1 < K / CK) stage((kc + 1) & 1, kc + 1);
    const char* cur = ch_lds(kc & 1);
#pragma unroll
    for (int ks = 0; ks < CK / 32; ++ks) {
      rbf16x8 bfr[2];
      bfr[0] = *reinterpret_cast<const rbf16x8*>(wrow0 + kc * CH_B + ks * 64);
      bfr[1] = *reinterpret_cast<const rbf16x8*>(wrow1 + kc * CH_B + ks * 64);
      rbf16x8 afr[8];
#pragma unroll
      for (int rf = 0; rf < 8; ++rf)
        afr[rf] = *reinterpret_cast<const rbf16x8*>(
            cur + rg_swz(rf * 16 + frag_col, ks * 64 + kchunk_b, CH_B));
#pragma unroll
      for (int rf = 0; rf < 8; ++rf)
#pragma unroll
        for (int cf = 0; cf < 2; ++cf)
          acc[rf][cf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[rf], bfr[cf], acc[rf][cf], 0, 0, 0);
    }
    __syncthreads();  // this buf's reads done before it restages at kc+2
  }

  // ---- epilogue: stage per-16-row tiles, add D, 16-B stores --------------
#pragma unroll 1
  for (int rf = 0; rf < 8; ++rf) {
#pragma unroll
    for (int cf = 0; cf < 2; ++cf)
#pragma unroll
      for (int r = 0; r < 4; ++r) {
        const int srow = (lane >> 4) * 4 + r;
        *reinterpret_cast<__hip_bfloat16*>(
            my_st + srow * 64 + (cf * 16 + frag_col) * 2) =
            __float2bfloat16(acc[rf][cf][r]);
      }
    __builtin_amdgcn_sched_barrier(0);
    const int srow = lane >> 2;
    const long grow = row0 + rf * 16 + srow;
    if (grow < m_rows) {
      rbf16x8 v = *reinterpret_cast<const rbf16x8*>(
          my_st + srow * 64 + (lane & 3) * 16);
      const long cb = grow * 256 + col0 + (lane & 3) * 8;
      if (d != nullptr) {
        const rbf16x8 dv = *reinterpret_cast<const rbf16x8*>(d + cb);
#pragma unroll
        for (int i = 0; i < 8; ++i)
          v[i] = (__bf16)((float)v[i] + (float)dv[i]);
      }
      *reinterpret_cast<rbf16x8*>(c + cb) = v;
    }
    __builtin_amdgcn_sched_barrier(0);
  }
}

void launch_rec_gemm_dgrad(const void* a, const void* wt, const void* d,
                           void* c, long m_rows, long a_stride,
                           hipStream_t s) {
  const int grid = (int)((m_rows + RG_BM - 1) / RG_BM);
  const size_t lds = 2 * RG_BM * 256 + 8 * 1024;  // 64 KB chunks + 8 KB st
  rec_gemm_dgrad_kernel<<<grid, 512, lds, s>>>(
      (const __hip_bfloat16*)a, (const __hip_bfloat16*)wt,
      (const __hip_bfloat16*)d, (__hip_bfloat16*)c, m_rows, a_stride);
}

}  // namespace nerrf
