// Fused softmax cross-entropy over a bf16 logits chunk (gfx950, HBM-bound).
//
// Semantics anchor: transformers fixed_cross_entropy as used by the
// reference's chunk_loss (ref chunk_loss.py:109-127, eager.py:22-36):
// fp32 log-softmax over each row, NLL at the label, grads scaled by
// grad_scale = 1/num_items. The [rows, V] chunk never exceeds
// chunk_size x V; the caller loops chunks (dlogits feeds the dW/dx GEMMs).
//
// One block (4 waves) per row; pass 1 computes max+sumexp online, pass 2
// writes dlogits = (softmax - onehot) * grad_scale.

#include "vh_common.h"

__global__ void k_ce_fwd(const bf16x8* __restrict__ logits,
                         const int64_t* __restrict__ labels,
                         float* __restrict__ loss_rows,
                         bf16x8* __restrict__ dlogits, int64_t rows,
                         int64_t Vv, int64_t V, float grad_scale,
                         int64_t ignore_index) {
  __shared__ float red_m[4], red_s[4];
  int64_t r = blockIdx.x;
  if (r >= rows) return;
  int64_t label = labels[r];
  const bf16x8* row = logits + r * Vv;
  bf16x8* drow = dlogits + r * Vv;

  bool valid = (label != ignore_index);
  int lane = threadIdx.x & (kWave - 1);
  int wave = threadIdx.x / kWave;

  // pass 1: online max / sumexp (fp32)
  float m = -3.0e38f, s = 0.f;
  for (int64_t c = threadIdx.x; c < Vv; c += blockDim.x) {
    bf16x8 v = row[c];
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int64_t col = c * 8 + j;
      float f = (col < V) ? bf2f(v.v[j]) : -3.0e38f;
      if (f > m) {
        s *= __expf(m - f);
        m = f;
      }
      s += __expf(f - m);
    }
  }
#pragma unroll
  for (int off = kWave / 2; off > 0; off >>= 1) {
    float m2 = __shfl_down(m, off, kWave);
    float s2 = __shfl_down(s, off, kWave);
    float mn = fmaxf(m, m2);
    s = s * __expf(m - mn) + s2 * __expf(m2 - mn);
    m = mn;
  }
  if (lane == 0) { red_m[wave] = m; red_s[wave] = s; }
  __syncthreads();
  if (threadIdx.x == 0) {
    float mm = red_m[0], ss = red_s[0];
    for (int wv = 1; wv < 4; ++wv) {
      float mn = fmaxf(mm, red_m[wv]);
      ss = ss * __expf(mm - mn) + red_s[wv] * __expf(red_m[wv] - mn);
      mm = mn;
    }
    red_m[0] = mm;
    red_s[0] = ss;
  }
  __syncthreads();
  m = red_m[0];
  s = red_s[0];
  float lse = m + __logf(s);
  float inv_s = 1.0f / s;

  // pass 2: dlogits; also fetch the label logit for the loss
  for (int64_t c = threadIdx.x; c < Vv; c += blockDim.x) {
    bf16x8 v = row[c];
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int64_t col = c * 8 + j;
      float f = bf2f(v.v[j]);
      float p = __expf(f - m) * inv_s;
      float gradv = valid ? (p - (col == label ? 1.0f : 0.0f)) * grad_scale : 0.0f;
      o.v[j] = (col < V) ? f2bf(gradv) : (bf16_t)0;
      if (col == label && valid && loss_rows != nullptr)
        loss_rows[r] = lse - f;
    }
    drow[c] = o;
  }
  if (!valid && threadIdx.x == 0 && loss_rows != nullptr) loss_rows[r] = 0.f;
}

extern "C" int vh_ce_fwd_bf16(const uint16_t* logits, const int64_t* labels,
                              float* loss_rows, uint16_t* dlogits,
                              int64_t rows, int64_t V, float grad_scale,
                              int64_t ignore_index, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(V % 8 == 0, "V %% 8 != 0 (V=%lld)", (long long)V);
  int64_t Vv = V / 8;
  if (rows == 0) return 0;
  hipLaunchKernelGGL(k_ce_fwd, dim3((uint32_t)rows), dim3(256), 0, s,
                     reinterpret_cast<const bf16x8*>(logits), labels,
                     loss_rows, reinterpret_cast<bf16x8*>(dlogits), rows, Vv,
                     V, grad_scale, ignore_index);
  VH_HIP(hipGetLastError());
  return 0;
}
