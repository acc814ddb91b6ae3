// Weighted neighbor gather-mean for GraphSAGE-T — CDNA4 (gfx950).
//
// out[n, :] = sum_k w[n,k] * h[idx[n,k], :] / max(sum_k w[n,k], eps)
//
// Design (MI355X): one wave64 per node row; the K neighbor ids / weights are
// read by lanes 0..K-1 and broadcast with wave shuffles (no LDS round trip);
// feature rows stream through coalesced per-lane loads — for bf16 D%128==0
// each lane moves 4 B (ushort2) per row so a wave reads a 256 B contiguous
// span per neighbor row.  Memory-bound by design: the MFMA work of the layer
// (the two 128x128 GEMMs) stays in hipBLASLt; this kernel covers the
// irregular gather that a library GEMM cannot.
//
// Counterpart of the "sampled SpMM / segment-reduce" obligation in
// SURVEY.md §2a (GraphSAGE-T neighbor aggregation); validated against
// nerrf_amd/ops/reference.py::gather_mean_ref.
#include "common.h"

namespace nerrf {

template <typename T>
__global__ void gather_mean_fwd_kernel(
    const T* __restrict__ h,        // [N, D]
    const long* __restrict__ idx,   // [N, K]
    const float* __restrict__ w,    // [N, K]
    T* __restrict__ out,            // [N, D]
    int n_nodes, int dim, int k) {
  const int wave_in_block = threadIdx.x / NERRF_WAVE;
  const int lane = threadIdx.x % NERRF_WAVE;
  const int waves_per_block = blockDim.x / NERRF_WAVE;
  for (int node = blockIdx.x * waves_per_block + wave_in_block; node < n_nodes;
       node += gridDim.x * waves_per_block) {
    // lane k' < K holds (idx, w) of neighbor k'
    float w_lane = (lane < k) ? w[(long)node * k + lane] : 0.0f;
    long i_lane = (lane < k) ? idx[(long)node * k + lane] : 0;
    float denom = wave_reduce_sum(w_lane);
    denom = fmaxf(denom, 1e-6f);
    const float inv_denom = 1.0f / denom;

    // accumulate D columns, lane-strided
    constexpr int MAX_COLS = 8;  // supports D up to 512 per wave pass
    float acc[MAX_COLS];
    const int cols = (dim + NERRF_WAVE - 1) / NERRF_WAVE;
#pragma unroll
    for (int cc = 0; cc < MAX_COLS; ++cc) acc[cc] = 0.0f;
    for (int kk = 0; kk < k; ++kk) {
      const float wk = __shfl(w_lane, kk, NERRF_WAVE) * inv_denom;
      const long nb = __shfl(i_lane, kk, NERRF_WAVE);
      const T* row = h + (long)nb * dim;
      for (int cc = 0; cc < cols; ++cc) {
        const int d = lane + cc * NERRF_WAVE;
        if (d < dim) acc[cc] = fmaf(wk, to_f32(row[d]), acc[cc]);
      }
    }
    T* orow = out + (long)node * dim;
    for (int cc = 0; cc < cols; ++cc) {
      const int d = lane + cc * NERRF_WAVE;
      if (d < dim) orow[d] = from_f32<T>(acc[cc]);
    }
  }
}

// Specialised bf16 path for D % 128 == 0: ushort2 per lane per row (4 B/lane).
__global__ void gather_mean_fwd_bf16_v2_kernel(
    const __hip_bfloat16* __restrict__ h,
    const long* __restrict__ idx,
    const float* __restrict__ w,
    __hip_bfloat16* __restrict__ out,
    int n_nodes, int dim, int k) {
  const int wave_in_block = threadIdx.x / NERRF_WAVE;
  const int lane = threadIdx.x % NERRF_WAVE;
  const int waves_per_block = blockDim.x / NERRF_WAVE;
  const int pairs = dim / 2;           // bf16 pairs per row
  const int cols = pairs / NERRF_WAVE; // ushort2 loads per lane
  for (int node = blockIdx.x * waves_per_block + wave_in_block; node < n_nodes;
       node += gridDim.x * waves_per_block) {
    float w_lane = (lane < k) ? w[(long)node * k + lane] : 0.0f;
    long i_lane = (lane < k) ? idx[(long)node * k + lane] : 0;
    float denom = fmaxf(wave_reduce_sum(w_lane), 1e-6f);
    const float inv_denom = 1.0f / denom;
    constexpr int MAX_PAIR_COLS = 4;  // D up to 512
    float acc0[MAX_PAIR_COLS], acc1[MAX_PAIR_COLS];
#pragma unroll
    for (int cc = 0; cc < MAX_PAIR_COLS; ++cc) { acc0[cc] = 0.f; acc1[cc] = 0.f; }
    for (int kk = 0; kk < k; ++kk) {
      const float wk = __shfl(w_lane, kk, NERRF_WAVE) * inv_denom;
      const long nb = __shfl(i_lane, kk, NERRF_WAVE);
      const ushort2* row = reinterpret_cast<const ushort2*>(h + (long)nb * dim);
      for (int cc = 0; cc < cols; ++cc) {
        const ushort2 v = row[lane + cc * NERRF_WAVE];
        __hip_bfloat16 b0 = *reinterpret_cast<const __hip_bfloat16*>(&v.x);
        __hip_bfloat16 b1 = *reinterpret_cast<const __hip_bfloat16*>(&v.y);
        acc0[cc] = fmaf(wk, __bfloat162float(b0), acc0[cc]);
        acc1[cc] = fmaf(wk, __bfloat162float(b1), acc1[cc]);
      }
    }
    ushort2* orow = reinterpret_cast<ushort2*>(out + (long)node * dim);
    for (int cc = 0; cc < cols; ++cc) {
      __hip_bfloat16 b0 = __float2bfloat16(acc0[cc]);
      __hip_bfloat16 b1 = __float2bfloat16(acc1[cc]);
      ushort2 v;
      v.x = *reinterpret_cast<const unsigned short*>(&b0);
      v.y = *reinterpret_cast<const unsigned short*>(&b1);
      orow[lane + cc * NERRF_WAVE] = v;
    }
  }
}

// Backward: grad_h[m, :] += sum_{(n,k): idx[n,k]==m} (w[n,k]/denom_n) * grad_out[n, :]
// fp32 atomics into a workspace (MI355X has native global fp32 atomic add);
// the caller casts the workspace to the grad dtype.
template <typename T>
__global__ void gather_mean_bwd_kernel(
    const T* __restrict__ grad_out,  // [N, D]
    const long* __restrict__ idx,    // [N, K]
    const float* __restrict__ w,     // [N, K]
    float* __restrict__ grad_h,      // [M, D] fp32 workspace (zeroed)
    int n_nodes, int dim, int k) {
  const int wave_in_block = threadIdx.x / NERRF_WAVE;
  const int lane = threadIdx.x % NERRF_WAVE;
  const int waves_per_block = blockDim.x / NERRF_WAVE;
  for (int node = blockIdx.x * waves_per_block + wave_in_block; node < n_nodes;
       node += gridDim.x * waves_per_block) {
    float w_lane = (lane < k) ? w[(long)node * k + lane] : 0.0f;
    long i_lane = (lane < k) ? idx[(long)node * k + lane] : 0;
    float denom = fmaxf(wave_reduce_sum(w_lane), 1e-6f);
    const float inv_denom = 1.0f / denom;
    const T* grow = grad_out + (long)node * dim;
    const int cols = (dim + NERRF_WAVE - 1) / NERRF_WAVE;
    float gval[8];
    for (int cc = 0; cc < cols; ++cc) {
      const int d = lane + cc * NERRF_WAVE;
      gval[cc] = (d < dim) ? to_f32(grow[d]) : 0.0f;
    }
    for (int kk = 0; kk < k; ++kk) {
      const float wk = __shfl(w_lane, kk, NERRF_WAVE) * inv_denom;
      if (wk == 0.0f) continue;
      const long nb = __shfl(i_lane, kk, NERRF_WAVE);
      float* dst = grad_h + (long)nb * dim;
      for (int cc = 0; cc < cols; ++cc) {
        const int d = lane + cc * NERRF_WAVE;
        if (d < dim) atomicAdd(dst + d, wk * gval[cc]);
      }
    }
  }
}

template __global__ void gather_mean_fwd_kernel<float>(
    const float*, const long*, const float*, float*, int, int, int);
template __global__ void gather_mean_fwd_kernel<__hip_bfloat16>(
    const __hip_bfloat16*, const long*, const float*, __hip_bfloat16*, int, int, int);
template __global__ void gather_mean_bwd_kernel<float>(
    const float*, const long*, const float*, float*, int, int, int);
template __global__ void gather_mean_bwd_kernel<__hip_bfloat16>(
    const __hip_bfloat16*, const long*, const float*, float*, int, int, int);

// Backward over the reverse index (entries sorted by destination node):
//   grad_h[m, :] += sum_{e: rev_dst[e]==m} rev_w[e] * grad_out[rev_src[e], :]
// (rev_w already contains w/denom).  Load-balanced segmented reduce: each
// wave owns a fixed 256-entry tile regardless of per-node in-degree (a
// wave-per-node schedule serialises on hub nodes — a popular process node
// can carry 10^4-10^5 reverse entries), accumulating in registers while the
// destination stays constant and flushing with fp32 atomics at segment
// boundaries (rare: sorted entries change destination at most
// nodes-per-tile times).
template <typename T>
__global__ void gather_mean_bwd_csr_kernel(
    const T* __restrict__ grad_out,   // [N, D]
    const long* __restrict__ rev_dst, // [E] sorted
    const long* __restrict__ rev_src, // [E]
    const float* __restrict__ rev_w,  // [E]
    float* __restrict__ grad_h,       // [M, D] fp32 workspace (zeroed)
    long n_entries, int dim) {
  const int ENT = 64;  // entries per wave tile (small => enough waves to fill 256 CUs)
  const int wave_in_block = threadIdx.x / NERRF_WAVE;
  const int lane = threadIdx.x % NERRF_WAVE;
  const int waves_per_block = blockDim.x / NERRF_WAVE;
  const long n_tiles = (n_entries + ENT - 1) / ENT;
  const int cols = (dim + NERRF_WAVE - 1) / NERRF_WAVE;
  for (long tile = blockIdx.x * waves_per_block + wave_in_block; tile < n_tiles;
       tile += (long)gridDim.x * waves_per_block) {
    const long e0 = tile * ENT;
    const long e1 = (e0 + ENT < n_entries) ? e0 + ENT : n_entries;
    float acc[8];
#pragma unroll
    for (int cc = 0; cc < 8; ++cc) acc[cc] = 0.0f;
    long cur_m = -1;
    for (long base = e0; base < e1; base += NERRF_WAVE) {
      const long e = base + lane;
      const float w_lane = (e < e1) ? rev_w[e] : 0.0f;
      const long s_lane = (e < e1) ? rev_src[e] : 0;
      const long m_lane = (e < e1) ? rev_dst[e] : -1;
      const int cnt = (int)((e1 - base < NERRF_WAVE) ? (e1 - base) : NERRF_WAVE);
      for (int j = 0; j < cnt; ++j) {
        const long m = __shfl(m_lane, j, NERRF_WAVE);
        if (m != cur_m) {
          if (cur_m >= 0) {
            float* dst = grad_h + cur_m * dim;
            for (int cc = 0; cc < cols; ++cc) {
              const int d = lane + cc * NERRF_WAVE;
              if (d < dim && acc[cc] != 0.0f) atomicAdd(dst + d, acc[cc]);
              acc[cc] = 0.0f;
            }
          }
          cur_m = m;
        }
        const float wk = __shfl(w_lane, j, NERRF_WAVE);
        const long src = __shfl(s_lane, j, NERRF_WAVE);
        const T* row = grad_out + src * dim;
        for (int cc = 0; cc < cols; ++cc) {
          const int d = lane + cc * NERRF_WAVE;
          if (d < dim) acc[cc] = fmaf(wk, to_f32(row[d]), acc[cc]);
        }
      }
    }
    if (cur_m >= 0) {
      float* dst = grad_h + cur_m * dim;
      for (int cc = 0; cc < cols; ++cc) {
        const int d = lane + cc * NERRF_WAVE;
        if (d < dim && acc[cc] != 0.0f) atomicAdd(dst + d, acc[cc]);
      }
    }
  }
}

template __global__ void gather_mean_bwd_csr_kernel<float>(
    const float*, const long*, const long*, const float*, float*, long, int);
template __global__ void gather_mean_bwd_csr_kernel<__hip_bfloat16>(
    const __hip_bfloat16*, const long*, const long*, const float*, float*, long, int);

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static inline int grid_for(long waves_needed, int waves_per_block) {
  long blocks = (waves_needed + waves_per_block - 1) / waves_per_block;
  // >> 256 workgroups to fill 256 CUs / 8 XCDs; cap and grid-stride the rest
  if (blocks > 8192) blocks = 8192;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

void launch_gather_mean_fwd(const void* h, const long* idx, const float* w,
                            void* out, int n, int dim, int k, bool bf16,
                            hipStream_t s) {
  const int block = 256;
  const int wpb = block / NERRF_WAVE;
  const int grid = grid_for(n, wpb);
  if (bf16 && dim % 128 == 0 && dim <= 512) {
    gather_mean_fwd_bf16_v2_kernel<<<grid, block, 0, s>>>(
        (const __hip_bfloat16*)h, idx, w, (__hip_bfloat16*)out, n, dim, k);
  } else if (bf16) {
    gather_mean_fwd_kernel<__hip_bfloat16><<<grid, block, 0, s>>>(
        (const __hip_bfloat16*)h, idx, w, (__hip_bfloat16*)out, n, dim, k);
  } else {
    gather_mean_fwd_kernel<float><<<grid, block, 0, s>>>(
        (const float*)h, idx, w, (float*)out, n, dim, k);
  }
}

void launch_gather_mean_bwd_csr(const void* gout, const long* rev_dst,
                                const long* rev_src, const float* rev_w,
                                float* gh, long n_entries, int dim, bool bf16,
                                hipStream_t s) {
  const int block = 256;
  const int wpb = block / NERRF_WAVE;
  const long n_tiles = (n_entries + 63) / 64;
  const int grid = grid_for(n_tiles, wpb);
  if (bf16) {
    gather_mean_bwd_csr_kernel<__hip_bfloat16><<<grid, block, 0, s>>>(
        (const __hip_bfloat16*)gout, rev_dst, rev_src, rev_w, gh, n_entries, dim);
  } else {
    gather_mean_bwd_csr_kernel<float><<<grid, block, 0, s>>>(
        (const float*)gout, rev_dst, rev_src, rev_w, gh, n_entries, dim);
  }
}

void launch_gather_mean_bwd(const void* gout, const long* idx, const float* w,
                            float* gh, int n, int dim, int k, bool bf16,
                            hipStream_t s) {
  const int block = 256;
  const int wpb = block / NERRF_WAVE;
  const int grid = grid_for(n, wpb);
  if (bf16) {
    gather_mean_bwd_kernel<__hip_bfloat16><<<grid, block, 0, s>>>(
        (const __hip_bfloat16*)gout, idx, w, gh, n, dim, k);
  } else {
    gather_mean_bwd_kernel<float><<<grid, block, 0, s>>>(
        (const float*)gout, idx, w, gh, n, dim, k);
  }
}

}  // namespace nerrf
