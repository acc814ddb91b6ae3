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
//
// v2: A fragments load DIRECTLY from global — the NT layout means every
// fragment is one k-contiguous 16-B load, and the 64-KB A strip stays
// L2-resident across the four n-chunks, so LDS staging (v1) only added a
// block barrier and ds-traffic.  The C tile is the only LDS user: each
// wave dumps its whole [128, 32] bf16 slice (8 KB), one barrier, then
// re-reads row-major for 16-B global stores (v1 staged per 16-row tile =
// 16 barriers per chunk; measured 1.17x blas at M=64k, this removes its
// dominant stall).
// ---------------------------------------------------------------------------
__launch_bounds__(512)
__global__ void rec_gemm_fwd_kernel(
    const __hip_bfloat16* __restrict__ a,  // [M(row-stride a_stride), 256]
    const __hip_bfloat16* __restrict__ w,  // [1024, 256]
    __hip_bfloat16* __restrict__ c,        // [M(row-stride c_stride), 1024]
    long m_rows, long a_stride, long c_stride) {
  constexpr int K = 256;
  constexpr int ROW_B = K * 2;  // 512 B per A/W row
  extern __shared__ __attribute__((aligned(16))) char smem[];

  const long row0 = (long)blockIdx.x * RG_BM;
  const int tid = threadIdx.x;
  const int lane = tid % NERRF_WAVE;
  const int wave = tid / NERRF_WAVE;

  const int frag_col = lane & 15;
  const int kchunk_b = (lane >> 4) * 16;  // byte offset within 64-B k-step
  char* my_st = smem + wave * 8192;       // [128 rows][32 cols] bf16

  // A addressing: one vector base + uniform per-fragment offsets (the
  // fragment rows differ by rf*16 rows, a scalar quantity — keeping the
  // offsets uniform saves a VGPR-resident pointer array; tail blocks
  // clamp each row individually, garbage rows are never stored)
  const char* arow[8];
#pragma unroll
  for (int rf = 0; rf < 8; ++rf) {
    long r = row0 + rf * 16 + frag_col;
    if (r >= m_rows) r = m_rows - 1;  // tail: garbage rows never stored
    arow[rf] = reinterpret_cast<const char*>(a + r * a_stride) + kchunk_b;
  }

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
#pragma unroll 2
    for (int ks = 0; ks < K / 32; ++ks) {
      rbf16x8 bfr[2];
      bfr[0] = *reinterpret_cast<const rbf16x8*>(wrow0 + ks * 64);
      bfr[1] = *reinterpret_cast<const rbf16x8*>(wrow1 + ks * 64);
      rbf16x8 afr[8];
#pragma unroll
      for (int rf = 0; rf < 8; ++rf)
        afr[rf] = *reinterpret_cast<const rbf16x8*>(arow[rf] + ks * 64);
      // let all 10 VMEM issue, then the MFMA block (partial vmcnt waits)
      __builtin_amdgcn_sched_group_barrier(0x20, 10, 0);  // VMEM read x10
      __builtin_amdgcn_sched_group_barrier(0x8, 16, 0);   // MFMA x16
#pragma unroll
      for (int rf = 0; rf < 8; ++rf)
#pragma unroll
        for (int cf = 0; cf < 2; ++cf)
          acc[rf][cf] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(
              afr[rf], bfr[cf], acc[rf][cf], 0, 0, 0);
    }
    // ---- store: acc layout -> per-wave LDS slice -> 16-B row stores ------
    // acc element r of (rf, cf) is C row rf*16 + (lane>>4)*4 + r, col
    // col0 + cf*16 + (lane&15); the full [128, 32] bf16 slice stages in
    // 8 KB of this wave's LDS, then lane l stores rows l>>2 (+16 per rr)
    // as 16-B chunks — 64-B segments per 4 lanes vs 2-B scalar stores.
    if (nc) __syncthreads();  // previous chunk's reads done before overwrite
#pragma unroll
    for (int rf = 0; rf < 8; ++rf)
#pragma unroll
      for (int cf = 0; cf < 2; ++cf)
#pragma unroll
        for (int r = 0; r < 4; ++r) {
          const int srow = rf * 16 + (lane >> 4) * 4 + r;
          const int cb = (cf * 16 + frag_col) * 2;
          // 16-B-granular XOR swizzle: the four srows alive in one write
          // instruction differ by 4, so (srow>>2)&3 spreads their bank
          // groups (without it they all land on banks 0-7)
          const int cbs = cb ^ (((srow >> 2) & 3) << 4);
          *reinterpret_cast<__hip_bfloat16*>(my_st + srow * 64 + cbs) =
              __float2bfloat16(acc[rf][cf][r]);
        }
    __syncthreads();
#pragma unroll
    for (int rr = 0; rr < 8; ++rr) {
      const int srow = rr * 16 + (lane >> 2);
      const long grow = row0 + srow;
      if (grow < m_rows) {
        const int cbs = ((lane & 3) * 16) ^ (((srow >> 2) & 3) << 4);
        uint4 v = *reinterpret_cast<const uint4*>(my_st + srow * 64 + cbs);
        *reinterpret_cast<uint4*>(
            reinterpret_cast<char*>(c + grow * c_stride) +
            (col0 + (lane & 3) * 8) * 2) = v;
      }
    }
  }
}

void launch_rec_gemm_fwd(const void* a, const void* w, void* c, long m_rows,
                         long a_stride, long c_stride, hipStream_t s) {
  const int grid = (int)((m_rows + RG_BM - 1) / RG_BM);
  const size_t lds = 8 * 8192;  // 8 waves x 8 KB C staging
  rec_gemm_fwd_kernel<<<grid, 512, lds, s>>>(
      (const __hip_bfloat16*)a, (const __hip_bfloat16*)w, (__hip_bfloat16*)c,
      m_rows, a_stride, c_stride);
}

}  // namespace nerrf
