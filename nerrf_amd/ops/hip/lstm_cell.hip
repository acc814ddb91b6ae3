// Fused LSTM gate pointwise + state update — CDNA4 (gfx950).
//
// Forward:  given pre-activation gates [B, 4H] (= x W_ih + h W_hh + b,
// the GEMMs stay in hipBLASLt), previous (h, c) and a per-row validity mask,
// computes in ONE launch what eager PyTorch does in ~12 elementwise kernels:
//   i,f,o = sigmoid; g = tanh; c' = f*c + i*g; h' = o*tanh(c');
//   masked rows pass (h, c) through unchanged.
// Also writes the post-activation gates (saved for backward).
//
// Backward: the matching fused gate-gradient kernel; the two outer GEMMs
// (grad_gates @ W_hh, grad_gates^T @ h) stay in hipBLASLt.
//
// Counterpart of the "fused LSTM cell" obligation in SURVEY.md §2a;
// validated against nerrf_amd/ops/reference.py::lstm_pointwise_{fwd,bwd}_ref.
#include "common.h"

namespace nerrf {

// xg_stride: row stride (elements) of the xg slice — lets the sequence op
// read gate slabs straight out of a [T, B, 2*4H] dual-direction projection.
// hout_stride: row stride of h_out — lets both directions write straight
// into the [T, B, 2H] concatenated output (no torch.cat afterwards).
template <typename T>
__global__ void lstm_pointwise_fwd_kernel(
    const T* __restrict__ hg,         // [B, 4H] = h_prev @ W_hh^T (beta=0 GEMM)
    const T* __restrict__ xg,         // [B(row-stride xg_stride), 4H]
    const T* __restrict__ bias,       // [4H]
    const T* __restrict__ c_prev,     // [B, H]
    const T* __restrict__ h_prev,     // [B(row-stride hprev_stride), H]
    const float* __restrict__ mask,   // [B] or nullptr
    T* __restrict__ h_out,            // [B(row-stride hout_stride), H]
    T* __restrict__ c_out,            // [B, H]
    T* __restrict__ gates_act,        // [B, 4H] or nullptr (inference)
    long batch, int hdim, long xg_stride, long hout_stride,
    long hprev_stride) {
  const long total = batch * hdim;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long b = t / hdim;
    const int d = (int)(t % hdim);
    const long g0 = b * 4 * hdim + d;
    const long x0 = b * xg_stride + d;
    const float ip = to_f32(hg[g0]) + to_f32(xg[x0]) + to_f32(bias[d]);
    const float fp = to_f32(hg[g0 + hdim]) + to_f32(xg[x0 + hdim]) + to_f32(bias[d + hdim]);
    const float gp = to_f32(hg[g0 + 2 * hdim]) + to_f32(xg[x0 + 2 * hdim]) + to_f32(bias[d + 2 * hdim]);
    const float op = to_f32(hg[g0 + 3 * hdim]) + to_f32(xg[x0 + 3 * hdim]) + to_f32(bias[d + 3 * hdim]);
    const float i = sigmoidf_(ip);
    const float f = sigmoidf_(fp);
    const float g = tanhf(gp);
    const float o = sigmoidf_(op);
    const float cp = to_f32(c_prev[t]);
    float cn = f * cp + i * g;
    float hn = o * tanhf(cn);
    if (mask != nullptr) {
      const float m = mask[b];
      cn = m * cn + (1.0f - m) * cp;
      hn = m * hn + (1.0f - m) * to_f32(h_prev[b * hprev_stride + d]);
    }
    c_out[t] = from_f32<T>(cn);
    h_out[b * hout_stride + d] = from_f32<T>(hn);
    if (gates_act != nullptr) {  // inference skips the backward-only store
      gates_act[g0] = from_f32<T>(i);
      gates_act[g0 + hdim] = from_f32<T>(f);
      gates_act[g0 + 2 * hdim] = from_f32<T>(g);
      gates_act[g0 + 3 * hdim] = from_f32<T>(o);
    }
  }
}

template <typename T>
__global__ void lstm_pointwise_bwd_kernel(
    const T* __restrict__ grad_h,     // [B, H] recurrent grad
    const T* __restrict__ grad_out_t, // [B(row-stride gout_stride), H] or nullptr
    const T* __restrict__ grad_c,     // [B, H]
    const T* __restrict__ gates_act,  // [B, 4H]
    const T* __restrict__ c_prev,     // [B, H]
    const float* __restrict__ mask,   // [B] or nullptr
    T* __restrict__ grad_gates,       // [B(row-stride gg_stride), 4H]
    T* __restrict__ grad_c_prev,      // [B, H]
    T* __restrict__ grad_h_pass,      // [B, H]
    long batch, int hdim, long gout_stride, long gg_stride) {
  const long total = batch * hdim;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long b = t / hdim;
    const int d = (int)(t % hdim);
    const long g0 = b * 4 * hdim + d;
    const long o0 = b * gg_stride + d;
    const float i = to_f32(gates_act[g0]);
    const float f = to_f32(gates_act[g0 + hdim]);
    const float g = to_f32(gates_act[g0 + 2 * hdim]);
    const float o = to_f32(gates_act[g0 + 3 * hdim]);
    const float cp = to_f32(c_prev[t]);
    const float m = (mask != nullptr) ? mask[b] : 1.0f;
    const float tcn = tanhf(f * cp + i * g);  // tanh of UNMASKED c_new
    float gh_in = to_f32(grad_h[t]);
    if (grad_out_t != nullptr) gh_in += to_f32(grad_out_t[b * gout_stride + d]);
    const float gc_in = to_f32(grad_c[t]);
    const float gh = gh_in * m;
    const float gc = gc_in * m;
    const float d_o = gh * tcn;
    const float d_c = gc + gh * o * (1.0f - tcn * tcn);
    const float d_i = d_c * g;
    const float d_f = d_c * cp;
    const float d_g = d_c * i;
    grad_c_prev[t] = from_f32<T>(d_c * f + gc_in * (1.0f - m));
    grad_h_pass[t] = from_f32<T>(gh_in * (1.0f - m));
    grad_gates[o0] = from_f32<T>(d_i * i * (1.0f - i));
    grad_gates[o0 + hdim] = from_f32<T>(d_f * f * (1.0f - f));
    grad_gates[o0 + 2 * hdim] = from_f32<T>(d_g * (1.0f - g * g));
    grad_gates[o0 + 3 * hdim] = from_f32<T>(d_o * o * (1.0f - o));
  }
}

// ---------------------------------------------------------------------------
// vectorised variants: V adjacent hidden elements per thread so every global
// access is a 8/16-byte load (scalar bf16 is a measured 2-2.5x loss on CDNA4).
// Used whenever hdim % V == 0 (H=256 in production).
// ---------------------------------------------------------------------------

template <typename T, int V>
struct alignas(sizeof(T) * V) VecT {
  T v[V];
};

template <typename T, int V>
__global__ void lstm_pointwise_fwd_vec_kernel(
    const T* __restrict__ hg, const T* __restrict__ xg,
    const T* __restrict__ bias, const T* __restrict__ c_prev,
    const T* __restrict__ h_prev, const float* __restrict__ mask,
    T* __restrict__ h_out, T* __restrict__ c_out, T* __restrict__ gates_act,
    long batch, int hdim, long xg_stride, long hout_stride,
    long hprev_stride) {
  using VT = VecT<T, V>;
  const int hv = hdim / V;
  const long total = batch * hv;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long b = t / hv;
    const int dv = (int)(t % hv) * V;
    const long g0 = b * 4 * hdim + dv;
    const long x0 = b * xg_stride + dv;
    const long c0 = b * (long)hdim + dv;
    const VT hg_i = *reinterpret_cast<const VT*>(hg + g0);
    const VT hg_f = *reinterpret_cast<const VT*>(hg + g0 + hdim);
    const VT hg_g = *reinterpret_cast<const VT*>(hg + g0 + 2 * hdim);
    const VT hg_o = *reinterpret_cast<const VT*>(hg + g0 + 3 * hdim);
    const VT xg_i = *reinterpret_cast<const VT*>(xg + x0);
    const VT xg_f = *reinterpret_cast<const VT*>(xg + x0 + hdim);
    const VT xg_g = *reinterpret_cast<const VT*>(xg + x0 + 2 * hdim);
    const VT xg_o = *reinterpret_cast<const VT*>(xg + x0 + 3 * hdim);
    const VT b_i = *reinterpret_cast<const VT*>(bias + dv);
    const VT b_f = *reinterpret_cast<const VT*>(bias + dv + hdim);
    const VT b_g = *reinterpret_cast<const VT*>(bias + dv + 2 * hdim);
    const VT b_o = *reinterpret_cast<const VT*>(bias + dv + 3 * hdim);
    const VT cp_v = *reinterpret_cast<const VT*>(c_prev + c0);
    VT hp_v;
    const float m = (mask != nullptr) ? mask[b] : 1.0f;
    if (mask != nullptr)
      hp_v = *reinterpret_cast<const VT*>(h_prev + b * hprev_stride + dv);
    VT ho_v, co_v, ga_i, ga_f, ga_g, ga_o;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const float i = sigmoidf_(to_f32(hg_i.v[j]) + to_f32(xg_i.v[j]) + to_f32(b_i.v[j]));
      const float f = sigmoidf_(to_f32(hg_f.v[j]) + to_f32(xg_f.v[j]) + to_f32(b_f.v[j]));
      const float g = tanhf(to_f32(hg_g.v[j]) + to_f32(xg_g.v[j]) + to_f32(b_g.v[j]));
      const float o = sigmoidf_(to_f32(hg_o.v[j]) + to_f32(xg_o.v[j]) + to_f32(b_o.v[j]));
      const float cp = to_f32(cp_v.v[j]);
      float cn = f * cp + i * g;
      float hn = o * tanhf(cn);
      if (mask != nullptr) {
        cn = m * cn + (1.0f - m) * cp;
        hn = m * hn + (1.0f - m) * to_f32(hp_v.v[j]);
      }
      co_v.v[j] = from_f32<T>(cn);
      ho_v.v[j] = from_f32<T>(hn);
      ga_i.v[j] = from_f32<T>(i);
      ga_f.v[j] = from_f32<T>(f);
      ga_g.v[j] = from_f32<T>(g);
      ga_o.v[j] = from_f32<T>(o);
    }
    *reinterpret_cast<VT*>(c_out + c0) = co_v;
    *reinterpret_cast<VT*>(h_out + b * hout_stride + dv) = ho_v;
    if (gates_act != nullptr) {
      *reinterpret_cast<VT*>(gates_act + g0) = ga_i;
      *reinterpret_cast<VT*>(gates_act + g0 + hdim) = ga_f;
      *reinterpret_cast<VT*>(gates_act + g0 + 2 * hdim) = ga_g;
      *reinterpret_cast<VT*>(gates_act + g0 + 3 * hdim) = ga_o;
    }
  }
}

template <typename T, int V>
__global__ void lstm_pointwise_bwd_vec_kernel(
    const T* __restrict__ grad_h, const T* __restrict__ grad_out_t,
    const T* __restrict__ grad_c, const T* __restrict__ gates_act,
    const T* __restrict__ c_prev, const float* __restrict__ mask,
    T* __restrict__ grad_gates, T* __restrict__ grad_c_prev,
    T* __restrict__ grad_h_pass, float* __restrict__ bias_accum,
    long batch, int hdim, long gout_stride, long gg_stride) {
  using VT = VecT<T, V>;
  // optional fused bias-grad: per-block LDS partial of sum_b(grad_gates),
  // atomically folded into bias_accum[4H] at block end — removes the
  // separate 13-GB-per-direction gg re-read (gg2.sum(0)) from the step
  extern __shared__ float bias_lds[];
  if (bias_accum != nullptr)
    for (int i = threadIdx.x; i < 4 * hdim; i += blockDim.x) bias_lds[i] = 0.0f;
  if (bias_accum != nullptr) __syncthreads();
  const int hv = hdim / V;
  const long total = batch * hv;
  for (long t = blockIdx.x * (long)blockDim.x + threadIdx.x; t < total;
       t += (long)gridDim.x * blockDim.x) {
    const long b = t / hv;
    const int dv = (int)(t % hv) * V;
    const long g0 = b * 4 * hdim + dv;
    const long o0 = b * gg_stride + dv;
    const long c0 = b * (long)hdim + dv;
    const VT ga_i = *reinterpret_cast<const VT*>(gates_act + g0);
    const VT ga_f = *reinterpret_cast<const VT*>(gates_act + g0 + hdim);
    const VT ga_g = *reinterpret_cast<const VT*>(gates_act + g0 + 2 * hdim);
    const VT ga_o = *reinterpret_cast<const VT*>(gates_act + g0 + 3 * hdim);
    const VT cp_v = *reinterpret_cast<const VT*>(c_prev + c0);
    const VT gh_v = *reinterpret_cast<const VT*>(grad_h + c0);
    const VT gc_v = *reinterpret_cast<const VT*>(grad_c + c0);
    VT go_v;
    if (grad_out_t != nullptr)
      go_v = *reinterpret_cast<const VT*>(grad_out_t + b * gout_stride + dv);
    const float m = (mask != nullptr) ? mask[b] : 1.0f;
    VT gg_i, gg_f, gg_g, gg_o, gcp_v, ghp_v;
#pragma unroll
    for (int j = 0; j < V; ++j) {
      const float i = to_f32(ga_i.v[j]);
      const float f = to_f32(ga_f.v[j]);
      const float g = to_f32(ga_g.v[j]);
      const float o = to_f32(ga_o.v[j]);
      const float cp = to_f32(cp_v.v[j]);
      const float tcn = tanhf(f * cp + i * g);
      float gh_in = to_f32(gh_v.v[j]);
      if (grad_out_t != nullptr) gh_in += to_f32(go_v.v[j]);
      const float gc_in = to_f32(gc_v.v[j]);
      const float gh = gh_in * m;
      const float gc = gc_in * m;
      const float d_o = gh * tcn;
      const float d_c = gc + gh * o * (1.0f - tcn * tcn);
      const float d_i = d_c * g;
      const float d_f = d_c * cp;
      const float d_g = d_c * i;
      gcp_v.v[j] = from_f32<T>(d_c * f + gc_in * (1.0f - m));
      ghp_v.v[j] = from_f32<T>(gh_in * (1.0f - m));
      gg_i.v[j] = from_f32<T>(d_i * i * (1.0f - i));
      gg_f.v[j] = from_f32<T>(d_f * f * (1.0f - f));
      gg_g.v[j] = from_f32<T>(d_g * (1.0f - g * g));
      gg_o.v[j] = from_f32<T>(d_o * o * (1.0f - o));
    }
    *reinterpret_cast<VT*>(grad_c_prev + c0) = gcp_v;
    *reinterpret_cast<VT*>(grad_h_pass + c0) = ghp_v;
    *reinterpret_cast<VT*>(grad_gates + o0) = gg_i;
    *reinterpret_cast<VT*>(grad_gates + o0 + hdim) = gg_f;
    *reinterpret_cast<VT*>(grad_gates + o0 + 2 * hdim) = gg_g;
    *reinterpret_cast<VT*>(grad_gates + o0 + 3 * hdim) = gg_o;
    if (bias_accum != nullptr) {
#pragma unroll
      for (int j = 0; j < V; ++j) {
        atomicAdd(bias_lds + dv + j, to_f32(gg_i.v[j]));
        atomicAdd(bias_lds + hdim + dv + j, to_f32(gg_f.v[j]));
        atomicAdd(bias_lds + 2 * hdim + dv + j, to_f32(gg_g.v[j]));
        atomicAdd(bias_lds + 3 * hdim + dv + j, to_f32(gg_o.v[j]));
      }
    }
  }
  if (bias_accum != nullptr) {
    __syncthreads();
    // 64 striped replicas: ~8k blocks all adding to ONE [4H] vector
    // serialize on the per-address atomic queues (~100 us/launch measured);
    // striping by block id cuts each address's chain 64x.  The caller sums
    // the [64, 4H] buffer once per direction.
    float* rep = bias_accum + (long)(blockIdx.x & 63) * (4 * hdim);
    for (int i = threadIdx.x; i < 4 * hdim; i += blockDim.x)
      if (bias_lds[i] != 0.0f) atomicAdd(rep + i, bias_lds[i]);
  }
}

template __global__ void lstm_pointwise_fwd_vec_kernel<__hip_bfloat16, 8>(
    const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
    const __hip_bfloat16*, const __hip_bfloat16*, const float*,
    __hip_bfloat16*, __hip_bfloat16*, __hip_bfloat16*, long, int, long, long,
    long);
template __global__ void lstm_pointwise_fwd_vec_kernel<float, 4>(
    const float*, const float*, const float*, const float*, const float*,
    const float*, float*, float*, float*, long, int, long, long, long);
template __global__ void lstm_pointwise_bwd_vec_kernel<__hip_bfloat16, 8>(
    const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
    const __hip_bfloat16*, const __hip_bfloat16*, const float*,
    __hip_bfloat16*, __hip_bfloat16*, __hip_bfloat16*, float*, long, int,
    long, long);
template __global__ void lstm_pointwise_bwd_vec_kernel<float, 4>(
    const float*, const float*, const float*, const float*, const float*,
    const float*, float*, float*, float*, float*, long, int, long, long);

template __global__ void lstm_pointwise_fwd_kernel<float>(
    const float*, const float*, const float*, const float*, const float*, const float*,
    float*, float*, float*, long, int, long, long, long);
template __global__ void lstm_pointwise_fwd_kernel<__hip_bfloat16>(
    const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
    const __hip_bfloat16*, const float*, __hip_bfloat16*, __hip_bfloat16*, __hip_bfloat16*,
    long, int, long, long, long);
template __global__ void lstm_pointwise_bwd_kernel<float>(
    const float*, const float*, const float*, const float*, const float*, const float*,
    float*, float*, float*, long, int, long, long);
template __global__ void lstm_pointwise_bwd_kernel<__hip_bfloat16>(
    const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*, const __hip_bfloat16*,
    const __hip_bfloat16*, const float*, __hip_bfloat16*, __hip_bfloat16*, __hip_bfloat16*, long, int, long, long);

// ---------------------------------------------------------------------------
// host launchers
// ---------------------------------------------------------------------------

static inline int grid_elems(long total, int block) {
  long blocks = (total + block - 1) / block;
  if (blocks > 8192) blocks = 8192;
  if (blocks < 1) blocks = 1;
  return (int)blocks;
}

void launch_lstm_pointwise_fwd(const void* hg, const void* xg, const void* bias,
                               const void* c_prev, const void* h_prev,
                               const float* mask, void* h_out, void* c_out,
                               void* gates_act, long batch, int hdim,
                               long xg_stride, long hout_stride,
                               long hprev_stride, bool bf16, hipStream_t s) {
  const int block = 256;
  // vectorised path (16-B global accesses) whenever the layout allows
  const int v = bf16 ? 8 : 4;
  const bool vec = hdim % v == 0 && xg_stride % v == 0 &&
                   hout_stride % v == 0 && hprev_stride % v == 0;
  if (bf16) {
    if (vec) {
      const int grid = grid_elems(batch * (hdim / v), block);
      lstm_pointwise_fwd_vec_kernel<__hip_bfloat16, 8><<<grid, block, 0, s>>>(
          (const __hip_bfloat16*)hg, (const __hip_bfloat16*)xg,
          (const __hip_bfloat16*)bias, (const __hip_bfloat16*)c_prev,
          (const __hip_bfloat16*)h_prev, mask, (__hip_bfloat16*)h_out,
          (__hip_bfloat16*)c_out, (__hip_bfloat16*)gates_act, batch, hdim,
          xg_stride, hout_stride, hprev_stride);
      return;
    }
    const int grid = grid_elems(batch * hdim, block);
    lstm_pointwise_fwd_kernel<__hip_bfloat16><<<grid, block, 0, s>>>(
        (const __hip_bfloat16*)hg, (const __hip_bfloat16*)xg,
        (const __hip_bfloat16*)bias, (const __hip_bfloat16*)c_prev,
        (const __hip_bfloat16*)h_prev, mask, (__hip_bfloat16*)h_out,
        (__hip_bfloat16*)c_out, (__hip_bfloat16*)gates_act, batch, hdim,
        xg_stride, hout_stride, hprev_stride);
  } else {
    if (vec) {
      const int grid = grid_elems(batch * (hdim / v), block);
      lstm_pointwise_fwd_vec_kernel<float, 4><<<grid, block, 0, s>>>(
          (const float*)hg, (const float*)xg, (const float*)bias,
          (const float*)c_prev, (const float*)h_prev, mask, (float*)h_out,
          (float*)c_out, (float*)gates_act, batch, hdim, xg_stride,
          hout_stride, hprev_stride);
      return;
    }
    const int grid = grid_elems(batch * hdim, block);
    lstm_pointwise_fwd_kernel<float><<<grid, block, 0, s>>>(
        (const float*)hg, (const float*)xg, (const float*)bias,
        (const float*)c_prev, (const float*)h_prev, mask, (float*)h_out,
        (float*)c_out, (float*)gates_act, batch, hdim, xg_stride, hout_stride,
        hprev_stride);
  }
}

void launch_lstm_pointwise_bwd(const void* grad_h, const void* grad_out_t,
                               const void* grad_c, const void* gates_act,
                               const void* c_prev, const float* mask,
                               void* grad_gates, void* grad_c_prev,
                               void* grad_h_pass, float* bias_accum,
                               long batch, int hdim, long gout_stride,
                               long gg_stride, bool bf16, hipStream_t s) {
  const int block = 256;
  const int v = bf16 ? 8 : 4;
  const bool vec = hdim % v == 0 && gout_stride % v == 0 && gg_stride % v == 0;
  if (bf16) {
    if (vec) {
      const int grid = grid_elems(batch * (hdim / v), block);
      lstm_pointwise_bwd_vec_kernel<__hip_bfloat16, 8>
          <<<grid, block, bias_accum ? 4 * hdim * sizeof(float) : 0, s>>>(
          (const __hip_bfloat16*)grad_h, (const __hip_bfloat16*)grad_out_t,
          (const __hip_bfloat16*)grad_c, (const __hip_bfloat16*)gates_act,
          (const __hip_bfloat16*)c_prev, mask, (__hip_bfloat16*)grad_gates,
          (__hip_bfloat16*)grad_c_prev, (__hip_bfloat16*)grad_h_pass,
          bias_accum, batch, hdim, gout_stride, gg_stride);
      return;
    }
    const int grid = grid_elems(batch * hdim, block);
    lstm_pointwise_bwd_kernel<__hip_bfloat16><<<grid, block, 0, s>>>(
        (const __hip_bfloat16*)grad_h, (const __hip_bfloat16*)grad_out_t,
        (const __hip_bfloat16*)grad_c, (const __hip_bfloat16*)gates_act,
        (const __hip_bfloat16*)c_prev, mask, (__hip_bfloat16*)grad_gates,
        (__hip_bfloat16*)grad_c_prev, (__hip_bfloat16*)grad_h_pass, batch, hdim,
        gout_stride, gg_stride);
  } else {
    if (vec) {
      const int grid = grid_elems(batch * (hdim / v), block);
      lstm_pointwise_bwd_vec_kernel<float, 4>
          <<<grid, block, bias_accum ? 4 * hdim * sizeof(float) : 0, s>>>(
          (const float*)grad_h, (const float*)grad_out_t, (const float*)grad_c,
          (const float*)gates_act, (const float*)c_prev, mask,
          (float*)grad_gates, (float*)grad_c_prev, (float*)grad_h_pass,
          bias_accum, batch, hdim, gout_stride, gg_stride);
      return;
    }
    const int grid = grid_elems(batch * hdim, block);
    lstm_pointwise_bwd_kernel<float><<<grid, block, 0, s>>>(
        (const float*)grad_h, (const float*)grad_out_t, (const float*)grad_c,
        (const float*)gates_act, (const float*)c_prev, mask, (float*)grad_gates,
        (float*)grad_c_prev, (float*)grad_h_pass, batch, hdim, gout_stride,
        gg_stride);
  }
}

}  // namespace nerrf
