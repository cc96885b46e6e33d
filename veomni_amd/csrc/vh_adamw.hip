// Fused AdamW over a device pointer table — ONE kernel sweep for the whole
// parameter set (torch's multi_tensor_apply issued ~1.8k chunked launches
// per step on the 30B model, ~3.5x the traffic roofline).
//
// Semantics: torch AdamW (decoupled weight decay, bias correction), fp32
// math, bf16 storage for p/g/m/v (our training keeps optimizer state in
// param dtype). Optional grad_scale divides grads in-register (the
// grad-clip fold). All tensor sizes must be multiples of 8 (bf16x8 I/O).
#include "vh_common.h"

namespace {

struct AdamArgs {
  const uint64_t* p_ptrs;   // [T] device addresses
  const uint64_t* g_ptrs;
  const uint64_t* m_ptrs;
  const uint64_t* v_ptrs;
  const int64_t* prefix;    // [T+1] inclusive elem prefix (8-multiples)
  int T;
  float lr, beta1, beta2, eps, weight_decay;
  float bc1, bc2;           // 1 - beta^t
  const float* grad_scale;  // nullable
};

__global__ void k_adamw(AdamArgs a, int64_t total_vec) {
  // per-BLOCK contiguous chunk, threads stride within it: wave reads stay
  // coalesced while the tensor id is MONOTONE per thread — one binary
  // search at entry, then an O(1) forward walk (the grid-stride form paid
  // an 11-step prefix search per 8 elements: measured 86 ms -> the walk
  // version targets the ~53 ms HBM bound for the 427 GB sweep)
  const int64_t chunk =
      (total_vec + gridDim.x - 1) / gridDim.x;
  const int64_t v_begin = (int64_t)blockIdx.x * chunk + threadIdx.x;
  const int64_t v_end = min((int64_t)(blockIdx.x + 1) * chunk, total_vec);
  const int64_t stride = blockDim.x;
  float inv_scale = 1.f;
  if (a.grad_scale != nullptr) inv_scale = 1.f / *a.grad_scale;
  const float wd_mul = 1.f - a.lr * a.weight_decay;
  const float step_lr = a.lr / a.bc1;
  const float inv_bc2 = 1.f / a.bc2;
  int lo = 0;
  {
    int64_t e0 = v_begin * 8;
    int hi = a.T;
    while (lo + 1 < hi) {
      int mid = (lo + hi) >> 1;
      if (e0 >= a.prefix[mid]) lo = mid; else hi = mid;
    }
  }
  for (int64_t vi = v_begin; vi < v_end; vi += stride) {
    int64_t e = vi * 8;
    while (lo + 1 < a.T && e >= a.prefix[lo + 1]) ++lo;
    int64_t off = e - a.prefix[lo];
    bf16x8* pp = reinterpret_cast<bf16x8*>(a.p_ptrs[lo]) + (off >> 3);
    const bf16x8* gp = reinterpret_cast<const bf16x8*>(a.g_ptrs[lo]) + (off >> 3);
    bf16x8* mp = reinterpret_cast<bf16x8*>(a.m_ptrs[lo]) + (off >> 3);
    bf16x8* vp = reinterpret_cast<bf16x8*>(a.v_ptrs[lo]) + (off >> 3);
    bf16x8 pv = *pp, gv = *gp, mv = *mp, vv = *vp;
    bf16x8 po, mo, vo;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float p = bf2f(pv.v[j]);
      float g = bf2f(gv.v[j]) * inv_scale;
      float m = bf2f(mv.v[j]);
      float v = bf2f(vv.v[j]);
      p *= wd_mul;
      m = a.beta1 * m + (1.f - a.beta1) * g;
      v = a.beta2 * v + (1.f - a.beta2) * g * g;
      float denom = sqrtf(v * inv_bc2) + a.eps;
      p -= step_lr * m / denom;
      po.v[j] = f2bf(p);
      mo.v[j] = f2bf(m);
      vo.v[j] = f2bf(v);
    }
    *pp = po;
    *mp = mo;
    *vp = vo;
  }
}

}  // namespace

extern "C" int vh_adamw_bf16(const uint64_t* p_ptrs, const uint64_t* g_ptrs,
                             const uint64_t* m_ptrs, const uint64_t* v_ptrs,
                             const int64_t* prefix, int T, int64_t total,
                             float lr, float beta1, float beta2, float eps,
                             float weight_decay, int step,
                             const float* grad_scale, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(total % 8 == 0, "total %% 8 != 0");
  AdamArgs a;
  a.p_ptrs = p_ptrs; a.g_ptrs = g_ptrs; a.m_ptrs = m_ptrs; a.v_ptrs = v_ptrs;
  a.prefix = prefix; a.T = T;
  a.lr = lr; a.beta1 = beta1; a.beta2 = beta2; a.eps = eps;
  a.weight_decay = weight_decay;
  a.bc1 = 1.f - powf(beta1, (float)step);
  a.bc2 = 1.f - powf(beta2, (float)step);
  a.grad_scale = grad_scale;
  int64_t total_vec = total / 8;
  int blocks = (int)((total_vec + 255) / 256);
  if (blocks > 4096) blocks = 4096;
  hipLaunchKernelGGL(k_adamw, dim3(blocks), dim3(256), 0, s, a, total_vec);
  VH_HIP(hipGetLastError());
  return 0;
}
