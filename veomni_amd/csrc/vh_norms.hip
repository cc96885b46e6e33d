// RMSNorm / RoPE / SwiGLU — HBM-bound kernels for gfx950.
//
// Semantics anchors:
//   RMSNorm  — eager Qwen3MoeRMSNorm (patched_modeling_qwen3_moe_gpu.py
//              :380-400): fp32 variance, downcast BEFORE the weight multiply.
//   RoPE     — rotate-half (same file :94-111); backward = negated sin.
//   SwiGLU   — LigerSiLUMulFunction slot (ops/liger/__init__.py:130-141).
//
// Design: bf16x8 (16 B/lane) loads per guide G13; wave-shuffle fp32 reductions;
// one WAVE per row (rows are 128–8192 elements, 4 waves per block stride rows).

#include "vh_common.h"

// --------------------------------------------------------------- RMSNorm fwd
// Block-per-row (256 threads): one or two bf16x8 chunks per thread stay in
// registers across the reduce, so the second pass re-reads nothing from HBM.
// Wave shuffle + 4-slot LDS cross-wave reduce.
__global__ __launch_bounds__(256) void k_rmsnorm_fwd(
    const bf16x8* __restrict__ x, const bf16x8* __restrict__ w,
    bf16x8* __restrict__ y, float* __restrict__ rstd, int64_t T, int64_t Hv,
    float eps, float invH) {
  __shared__ float red[4];
  constexpr int MAXC = 4;  // up to 4 chunks/thread = H <= 8192
  int wave = threadIdx.x / kWave;
  int lane = threadIdx.x & (kWave - 1);
  for (int64_t r = blockIdx.x; r < T; r += gridDim.x) {
    const bf16x8* xr = x + r * Hv;
    bf16x8 v[MAXC];
    int nc = 0;
    float ss = 0.f;
    for (int64_t c = threadIdx.x; c < Hv; c += blockDim.x) {
      bf16x8 vv = xr[c];
      if (nc < MAXC) v[nc] = vv;
      ++nc;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(vv.v[j]);
        ss += f * f;
      }
    }
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1) ss += __shfl_down(ss, off, kWave);
    if (lane == 0) red[wave] = ss;
    __syncthreads();
    float tot = red[0] + red[1] + red[2] + red[3];
    float rs = rsqrtf(tot * invH + eps);
    if (threadIdx.x == 0 && rstd != nullptr) rstd[r] = rs;
    bf16x8* yr = y + r * Hv;
    int ci = 0;
    for (int64_t c = threadIdx.x; c < Hv; c += blockDim.x) {
      bf16x8 vv = (ci < MAXC) ? v[ci] : xr[c];
      ++ci;
      bf16x8 wv = w[c];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        // downcast the normalized value to bf16 FIRST, then bf16 multiply
        float nh = bf2f(f2bf(bf2f(vv.v[j]) * rs));
        o.v[j] = f2bf(bf2f(wv.v[j]) * nh);
      }
      yr[c] = o;
    }
    __syncthreads();
  }
}

// wave-per-row variant for small rows (per-head q/k norms: H = head_dim)
__global__ void k_rmsnorm_fwd_small(const bf16x8* __restrict__ x,
                                    const bf16x8* __restrict__ w,
                                    bf16x8* __restrict__ y,
                                    float* __restrict__ rstd, int64_t T,
                                    int64_t Hv, float eps, float invH) {
  int wave = (blockIdx.x * blockDim.x + threadIdx.x) / kWave;
  int lane = threadIdx.x & (kWave - 1);
  int num_waves = (gridDim.x * blockDim.x) / kWave;
  for (int64_t r = wave; r < T; r += num_waves) {
    const bf16x8* xr = x + r * Hv;
    bf16x8 v = {};
    float ss = 0.f;
    if (lane < Hv) {
      v = xr[lane];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float f = bf2f(v.v[j]);
        ss += f * f;
      }
    }
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1) ss += __shfl_down(ss, off, kWave);
    ss = __shfl(ss, 0, kWave);
    float rs = rsqrtf(ss * invH + eps);
    if (lane == 0 && rstd != nullptr) rstd[r] = rs;
    if (lane < Hv) {
      bf16x8 wv = w[lane];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float nh = bf2f(f2bf(bf2f(v.v[j]) * rs));
        o.v[j] = f2bf(bf2f(wv.v[j]) * nh);
      }
      y[r * Hv + lane] = o;
    }
  }
}

extern "C" int vh_rmsnorm_fwd_bf16(const uint16_t* x, const uint16_t* w,
                                   uint16_t* y, float* rstd, int64_t T,
                                   int64_t H, float eps, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(H % 8 == 0, "H %% 8 != 0 (H=%lld)", (long long)H);
  VH_CHECK(H <= 65536, "H too large");
  int64_t Hv = H / 8;
  if (Hv <= kWave) {
    int blocks = (int)((T + 3) / 4);
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(k_rmsnorm_fwd_small, dim3(blocks), dim3(256), 0, s,
                       reinterpret_cast<const bf16x8*>(x),
                       reinterpret_cast<const bf16x8*>(w),
                       reinterpret_cast<bf16x8*>(y), rstd, T, Hv, eps,
                       1.0f / (float)H);
  } else {
    int blocks = (int)(T < 2048 ? (T ? T : 1) : 2048);
    hipLaunchKernelGGL(k_rmsnorm_fwd, dim3(blocks), dim3(256), 0, s,
                       reinterpret_cast<const bf16x8*>(x),
                       reinterpret_cast<const bf16x8*>(w),
                       reinterpret_cast<bf16x8*>(y), rstd, T, Hv, eps,
                       1.0f / (float)H);
  }
  VH_HIP(hipGetLastError());
  return 0;
}

// --------------------------------------------------------------- RMSNorm bwd
// dx = rs * (g - x * rs^2/H * dot(g, x)),  g = dy * w  (fp32)
// dw += dy * bf16(x * rs): accumulated in per-lane REGISTERS (CHUNKS = Hv/64
// column chunks per lane) across the wave's rows, stored as per-wave partial
// rows in a scratch buffer, then summed by k_dw_reduce — no fp32 atomics on
// shared addresses anywhere. Earlier versions measured: 8 LDS fp32
// atomicAdds per bf16x8 per row (ds_add_f32 under 4-wave same-address
// contention is ~600 cycles/op on gfx950 — see the attention-backward probe
// notes in DESIGN.md) capped the kernel at ~1 TB/s; a global fp32 atomicAdd
// epilogue serialized ~2k same-address chains and was equally slow.
template <int CHUNKS>
__global__ __launch_bounds__(256, 2) void k_rmsnorm_bwd_reg(
    const bf16x8* __restrict__ dy, const bf16x8* __restrict__ x,
    const bf16x8* __restrict__ w, const float* __restrict__ rstd,
    bf16x8* __restrict__ dx, float* __restrict__ dw_scratch, int64_t T,
    int64_t Hv, float invH) {
  int wave = threadIdx.x / kWave;
  int lane = threadIdx.x & (kWave - 1);
  int waves_per_block = blockDim.x / kWave;

  float dw_acc[CHUNKS][8];
#pragma unroll
  for (int k = 0; k < CHUNKS; ++k)
#pragma unroll
    for (int j = 0; j < 8; ++j) dw_acc[k][j] = 0.f;

  for (int64_t r = blockIdx.x * waves_per_block + wave; r < T;
       r += (int64_t)gridDim.x * waves_per_block) {
    const bf16x8* dyr = dy + r * Hv;
    const bf16x8* xr = x + r * Hv;
    float rs = rstd[r];
    // the whole row in registers: 2*CHUNKS b128 loads in flight (~4 KiB of
    // MLP per wave); reused for the dx pass, so dy/x are read from HBM once.
    bf16x8 dv[CHUNKS], xv[CHUNKS];
#pragma unroll
    for (int k = 0; k < CHUNKS; ++k) dv[k] = dyr[lane + k * kWave];
#pragma unroll
    for (int k = 0; k < CHUNKS; ++k) xv[k] = xr[lane + k * kWave];
    float dot = 0.f;
#pragma unroll
    for (int k = 0; k < CHUNKS; ++k) {
      bf16x8 wv = w[lane + k * kWave];  // L2-resident, shared across waves
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += bf2f(dv[k].v[j]) * bf2f(wv.v[j]) * bf2f(xv[k].v[j]);
    }
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1) dot += __shfl_down(dot, off, kWave);
    dot = __shfl(dot, 0, kWave);
    float kf = dot * rs * rs * invH;
    bf16x8* dxr = dx + r * Hv;
#pragma unroll
    for (int k = 0; k < CHUNKS; ++k) {
      bf16x8 wv = w[lane + k * kWave];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bf2f(xv[k].v[j]);
        float g = bf2f(dv[k].v[j]) * bf2f(wv.v[j]);
        o.v[j] = f2bf(rs * (g - xf * kf));
        float xhat = bf2f(f2bf(xf * rs));  // the value forward multiplied by w
        dw_acc[k][j] += bf2f(dv[k].v[j]) * xhat;
      }
      dxr[lane + k * kWave] = o;
    }
  }
  // per-wave partial row (plain coalesced stores; a tiny reduce kernel sums
  // them — same-address fp32 atomic chains from ~2k waves measured ~400 us)
  int64_t wid = (int64_t)blockIdx.x * waves_per_block + wave;
  float* out = dw_scratch + wid * (Hv * 8);
#pragma unroll
  for (int k = 0; k < CHUNKS; ++k) {
    float4* o4 = reinterpret_cast<float4*>(out + (lane + k * kWave) * 8);
    o4[0] = float4{dw_acc[k][0], dw_acc[k][1], dw_acc[k][2], dw_acc[k][3]};
    o4[1] = float4{dw_acc[k][4], dw_acc[k][5], dw_acc[k][6], dw_acc[k][7]};
  }
}

// big-H variant (H = 4096): no row stash — dy/x re-read in the dx pass so
// register pressure stays at ~dw_acc + a few loads (4 waves/SIMD); traffic
// is 5 passes instead of 3 but streams at full rate.
template <int CHUNKS>
__global__ __launch_bounds__(256, 2) void k_rmsnorm_bwd_big(
    const bf16x8* __restrict__ dy, const bf16x8* __restrict__ x,
    const bf16x8* __restrict__ w, const float* __restrict__ rstd,
    bf16x8* __restrict__ dx, float* __restrict__ dw_scratch, int64_t T,
    int64_t Hv, float invH) {
  int wave = threadIdx.x / kWave;
  int lane = threadIdx.x & (kWave - 1);
  int waves_per_block = blockDim.x / kWave;

  float dw_acc[CHUNKS][8];
#pragma unroll
  for (int k = 0; k < CHUNKS; ++k)
#pragma unroll
    for (int j = 0; j < 8; ++j) dw_acc[k][j] = 0.f;

  for (int64_t r = blockIdx.x * waves_per_block + wave; r < T;
       r += (int64_t)gridDim.x * waves_per_block) {
    const bf16x8* dyr = dy + r * Hv;
    const bf16x8* xr = x + r * Hv;
    float rs = rstd[r];
    float dot = 0.f;
#pragma unroll 4
    for (int k = 0; k < CHUNKS; ++k) {
      bf16x8 d = dyr[lane + k * kWave], xv = xr[lane + k * kWave];
      bf16x8 wv = w[lane + k * kWave];
#pragma unroll
      for (int j = 0; j < 8; ++j)
        dot += bf2f(d.v[j]) * bf2f(wv.v[j]) * bf2f(xv.v[j]);
    }
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1) dot += __shfl_down(dot, off, kWave);
    dot = __shfl(dot, 0, kWave);
    float kf = dot * rs * rs * invH;
    bf16x8* dxr = dx + r * Hv;
#pragma unroll 4
    for (int k = 0; k < CHUNKS; ++k) {
      bf16x8 d = dyr[lane + k * kWave], xv = xr[lane + k * kWave];
      bf16x8 wv = w[lane + k * kWave];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bf2f(xv.v[j]);
        float g = bf2f(d.v[j]) * bf2f(wv.v[j]);
        o.v[j] = f2bf(rs * (g - xf * kf));
        float xhat = bf2f(f2bf(xf * rs));
        dw_acc[k][j] += bf2f(d.v[j]) * xhat;
      }
      dxr[lane + k * kWave] = o;
    }
  }
  int64_t wid = (int64_t)blockIdx.x * waves_per_block + wave;
  float* out = dw_scratch + wid * (Hv * 8);
#pragma unroll
  for (int k = 0; k < CHUNKS; ++k) {
    float4* o4 = reinterpret_cast<float4*>(out + (lane + k * kWave) * 8);
    o4[0] = float4{dw_acc[k][0], dw_acc[k][1], dw_acc[k][2], dw_acc[k][3]};
    o4[1] = float4{dw_acc[k][4], dw_acc[k][5], dw_acc[k][6], dw_acc[k][7]};
  }
}

// head-dim-sized rows (H = 128, the per-head q/k norms): 16-lane groups, a
// wave covers 4 rows per iteration; dw per lane column with a cross-subgroup
// shuffle reduce before the scratch store.
__global__ __launch_bounds__(256, 4) void k_rmsnorm_bwd_small128(
    const bf16x8* __restrict__ dy, const bf16x8* __restrict__ x,
    const bf16x8* __restrict__ w, const float* __restrict__ rstd,
    bf16x8* __restrict__ dx, float* __restrict__ dw_scratch, int64_t T,
    float invH) {
  const int lane = threadIdx.x & (kWave - 1);
  const int wave = threadIdx.x / kWave;
  const int sub = lane >> 4;   // row within the wave's quad
  const int cl = lane & 15;    // bf16x8 column
  const int waves_per_block = blockDim.x / kWave;
  const int64_t gw = (int64_t)blockIdx.x * waves_per_block + wave;
  const int64_t nw = (int64_t)gridDim.x * waves_per_block;

  const bf16x8 wv = w[cl];
  float dw_acc[8];
#pragma unroll
  for (int j = 0; j < 8; ++j) dw_acc[j] = 0.f;

  for (int64_t r0 = gw * 4; r0 < T; r0 += nw * 4) {
    int64_t r = r0 + sub;
    bool ok = r < T;
    bf16x8 d = {}, xv = {};
    float rs = 0.f;
    if (ok) {
      d = dy[r * 16 + cl];
      xv = x[r * 16 + cl];
      rs = rstd[r];
    }
    float dot = 0.f;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      dot += bf2f(d.v[j]) * bf2f(wv.v[j]) * bf2f(xv.v[j]);
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) dot += __shfl_xor(dot, off, kWave);
    float kf = dot * rs * rs * invH;
    if (ok) {
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bf2f(xv.v[j]);
        float g = bf2f(d.v[j]) * bf2f(wv.v[j]);
        o.v[j] = f2bf(rs * (g - xf * kf));
        float xhat = bf2f(f2bf(xf * rs));
        dw_acc[j] += bf2f(d.v[j]) * xhat;
      }
      dx[r * 16 + cl] = o;
    }
  }
  // combine the 4 row-subgroups (same columns) before storing
#pragma unroll
  for (int j = 0; j < 8; ++j) {
    dw_acc[j] += __shfl_xor(dw_acc[j], 16, kWave);
    dw_acc[j] += __shfl_xor(dw_acc[j], 32, kWave);
  }
  if (sub == 0) {
    float* out = dw_scratch + gw * 128;
    float4* o4 = reinterpret_cast<float4*>(out + cl * 8);
    o4[0] = float4{dw_acc[0], dw_acc[1], dw_acc[2], dw_acc[3]};
    o4[1] = float4{dw_acc[4], dw_acc[5], dw_acc[6], dw_acc[7]};
  }
}

// dw[i] += sum over W partial rows; grid (H/256, SPLIT) with SPLIT-way
// atomic combine (contention depth SPLIT, not W).
__global__ void k_dw_reduce(const float* __restrict__ scratch,
                            float* __restrict__ dw, int64_t H, int W,
                            int split) {
  int64_t i = (int64_t)blockIdx.x * blockDim.x + threadIdx.x;
  if (i >= H) return;
  int per = (W + split - 1) / split;
  int w0 = blockIdx.y * per;
  int w1 = w0 + per > W ? W : w0 + per;
  float acc = 0.f;
  for (int r = w0; r < w1; ++r) acc += scratch[(int64_t)r * H + i];
  if (acc != 0.f) atomicAdd(&dw[i], acc);
}

// generic-H fallback (LDS fp32 atomics) for shapes outside H % 512 == 0
__global__ void k_rmsnorm_bwd(const bf16x8* __restrict__ dy,
                              const bf16x8* __restrict__ x,
                              const bf16x8* __restrict__ w,
                              const float* __restrict__ rstd,
                              bf16x8* __restrict__ dx, float* __restrict__ dw,
                              int64_t T, int64_t Hv, float invH) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  float* dw_loc = reinterpret_cast<float*>(smem_raw);  // [Hv*8]
  for (int64_t i = threadIdx.x; i < Hv * 8; i += blockDim.x) dw_loc[i] = 0.f;
  __syncthreads();

  int wave = threadIdx.x / kWave;
  int lane = threadIdx.x & (kWave - 1);
  int waves_per_block = blockDim.x / kWave;
  for (int64_t r = blockIdx.x * waves_per_block + wave; r < T;
       r += (int64_t)gridDim.x * waves_per_block) {
    const bf16x8* dyr = dy + r * Hv;
    const bf16x8* xr = x + r * Hv;
    float rs = rstd[r];
    float dot = 0.f;
    for (int64_t c = lane; c < Hv; c += kWave) {
      bf16x8 d = dyr[c], xv = xr[c], wv = w[c];
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = bf2f(d.v[j]) * bf2f(wv.v[j]);
        dot += g * bf2f(xv.v[j]);
      }
    }
#pragma unroll
    for (int off = kWave / 2; off > 0; off >>= 1) dot += __shfl_down(dot, off, kWave);
    dot = __shfl(dot, 0, kWave);
    float k = dot * rs * rs * invH;
    bf16x8* dxr = dx + r * Hv;
    for (int64_t c = lane; c < Hv; c += kWave) {
      bf16x8 d = dyr[c], xv = xr[c], wv = w[c];
      bf16x8 o;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float xf = bf2f(xv.v[j]);
        float g = bf2f(d.v[j]) * bf2f(wv.v[j]);
        o.v[j] = f2bf(rs * (g - xf * k));
        float xhat = bf2f(f2bf(xf * rs));  // the value forward multiplied by w
        atomicAdd(&dw_loc[c * 8 + j], bf2f(d.v[j]) * xhat);
      }
      dxr[c] = o;
    }
  }
  __syncthreads();
  for (int64_t i = threadIdx.x; i < Hv * 8; i += blockDim.x)
    if (dw_loc[i] != 0.f) atomicAdd(&dw[i], dw_loc[i]);
}

extern "C" int vh_rmsnorm_bwd_bf16(const uint16_t* dy, const uint16_t* x,
                                   const uint16_t* w, const float* rstd,
                                   uint16_t* dx, float* dw, int64_t T,
                                   int64_t H, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(H % 8 == 0, "H %% 8 != 0");
  VH_CHECK(H * sizeof(float) <= 64 * 1024, "H too large for LDS dw (%lld)", (long long)H);
  int64_t Hv = H / 8;
  int blocks = (int)((T + 3) / 4);
  if (blocks > 1024) blocks = 1024;
  if (blocks < 1) blocks = 1;
  int blocks_reg = blocks > 512 ? 512 : blocks;
  if (H == 128) {
    int blocks_s = (int)((T + 15) / 16);
    if (blocks_s > 1024) blocks_s = 1024;
    if (blocks_s < 1) blocks_s = 1;
    int waves_total = blocks_s * 4;
    float* scratch = nullptr;
    VH_HIP(hipMallocAsync(&scratch, (size_t)waves_total * 128 * sizeof(float), s));
    hipLaunchKernelGGL(k_rmsnorm_bwd_small128, dim3(blocks_s), dim3(256), 0, s,
                       reinterpret_cast<const bf16x8*>(dy),
                       reinterpret_cast<const bf16x8*>(x),
                       reinterpret_cast<const bf16x8*>(w), rstd,
                       reinterpret_cast<bf16x8*>(dx), scratch, T,
                       1.0f / 128.0f);
    hipLaunchKernelGGL(k_dw_reduce, dim3(1, 16), dim3(128), 0, s, scratch, dw,
                       128, waves_total, 16);
    VH_HIP(hipFreeAsync(scratch, s));
    VH_HIP(hipGetLastError());
    return 0;
  }
  int reg_chunks = 0;
  switch (H) {
    case 512: reg_chunks = 1; break;
    case 1024: reg_chunks = 2; break;
    case 2048: reg_chunks = 4; break;
    case 4096: reg_chunks = 8; break;
  }
  if (reg_chunks) {
    int waves_total = blocks_reg * 4;
    float* scratch = nullptr;
    VH_HIP(hipMallocAsync(&scratch, (size_t)waves_total * H * sizeof(float), s));
#define VH_RMS_BWD_REG(C)                                                     \
  hipLaunchKernelGGL(k_rmsnorm_bwd_reg<C>, dim3(blocks_reg), dim3(256), 0, s, \
                     reinterpret_cast<const bf16x8*>(dy),                     \
                     reinterpret_cast<const bf16x8*>(x),                      \
                     reinterpret_cast<const bf16x8*>(w), rstd,                \
                     reinterpret_cast<bf16x8*>(dx), scratch, T, Hv,           \
                     1.0f / (float)H)
    switch (reg_chunks) {
      case 1: VH_RMS_BWD_REG(1); break;
      case 2: VH_RMS_BWD_REG(2); break;
      case 4: VH_RMS_BWD_REG(4); break;
      case 8:
        hipLaunchKernelGGL(k_rmsnorm_bwd_big<8>, dim3(blocks_reg), dim3(256),
                           0, s, reinterpret_cast<const bf16x8*>(dy),
                           reinterpret_cast<const bf16x8*>(x),
                           reinterpret_cast<const bf16x8*>(w), rstd,
                           reinterpret_cast<bf16x8*>(dx), scratch, T, Hv,
                           1.0f / (float)H);
        break;
    }
#undef VH_RMS_BWD_REG
    int split = 16;
    hipLaunchKernelGGL(k_dw_reduce, dim3((uint32_t)((H + 255) / 256), split),
                       dim3(256), 0, s, scratch, dw, H, waves_total, split);
    VH_HIP(hipFreeAsync(scratch, s));
    VH_HIP(hipGetLastError());
    return 0;
  }
  {
      hipLaunchKernelGGL(k_rmsnorm_bwd, dim3(blocks), dim3(256),
                         (size_t)(H * sizeof(float)), s,
                         reinterpret_cast<const bf16x8*>(dy),
                         reinterpret_cast<const bf16x8*>(x),
                         reinterpret_cast<const bf16x8*>(w), rstd,
                         reinterpret_cast<bf16x8*>(dx), dw, T, Hv,
                         1.0f / (float)H);
  }
  VH_HIP(hipGetLastError());
  return 0;
}

// -------------------------------------------------------------------- RoPE
// x: [B, h, S, D]; cos/sin: [B, S, D] bf16 (halves duplicated). Each thread
// processes one 8-elem chunk of the FIRST half plus its partner chunk in the
// second half:  o1 = x1*c - x2*s ; o2 = x2*c + x1*s.
__global__ void k_rope(const bf16x8* __restrict__ x, const bf16x8* __restrict__ cs,
                       const bf16x8* __restrict__ sn, bf16x8* __restrict__ out,
                       int64_t B, int64_t h, int64_t S, int64_t Dv, float sgn) {
  int64_t half = Dv / 2;                 // vectors per half
  int64_t total = B * h * S * half;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t c = i % half;
    int64_t row = i / half;              // (b, hh, s)
    int64_t s_idx = row % S;
    int64_t b = row / (h * S);
    const bf16x8* xr = x + row * Dv;
    int64_t cs_row = (b * S + s_idx) * Dv;
    bf16x8 x1 = xr[c], x2 = xr[half + c];
    bf16x8 c1 = cs[cs_row + c], s1 = sn[cs_row + c];
    bf16x8 c2 = cs[cs_row + half + c], s2 = sn[cs_row + half + c];
    bf16x8 o1, o2;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float a = bf2f(x1.v[j]), bq = bf2f(x2.v[j]);
      o1.v[j] = f2bf(a * bf2f(c1.v[j]) - sgn * bq * bf2f(s1.v[j]));
      o2.v[j] = f2bf(bq * bf2f(c2.v[j]) + sgn * a * bf2f(s2.v[j]));
    }
    bf16x8* orow = out + row * Dv;
    orow[c] = o1;
    orow[half + c] = o2;
  }
}

extern "C" int vh_rope_bf16(const uint16_t* q, const uint16_t* k,
                            const uint16_t* cos_t, const uint16_t* sin_t,
                            uint16_t* q_out, uint16_t* k_out, int64_t B,
                            int64_t hq, int64_t hk, int64_t S, int64_t D,
                            int negate_sin, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(D % 16 == 0, "D %% 16 != 0 (D=%lld)", (long long)D);
  int64_t Dv = D / 8;
  float sgn = negate_sin ? -1.0f : 1.0f;
  auto launch = [&](const uint16_t* xin, uint16_t* xout, int64_t h) -> int {
    int64_t total = B * h * S * (Dv / 2);
    int blocks = (int)((total + 255) / 256);
    if (blocks > 2048) blocks = 2048;
    if (blocks < 1) blocks = 1;
    hipLaunchKernelGGL(k_rope, dim3(blocks), dim3(256), 0, s,
                       reinterpret_cast<const bf16x8*>(xin),
                       reinterpret_cast<const bf16x8*>(cos_t),
                       reinterpret_cast<const bf16x8*>(sin_t),
                       reinterpret_cast<bf16x8*>(xout), B, h, S, Dv, sgn);
    return 0;
  };
  launch(q, q_out, hq);
  launch(k, k_out, hk);
  VH_HIP(hipGetLastError());
  return 0;
}

// ------------------------------------------------------------------- SwiGLU
__global__ void k_silu_mul(const bf16x8* __restrict__ g,
                           const bf16x8* __restrict__ u, bf16x8* __restrict__ o,
                           int64_t nv) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    bf16x8 gv = g[i], uv = u[i], ov;
#pragma unroll
    for (int j = 0; j < 8; ++j)
      ov.v[j] = f2bf(bf2f(f2bf(siluf(bf2f(gv.v[j])))) * bf2f(uv.v[j]));
    o[i] = ov;
  }
}

extern "C" int vh_silu_mul_bf16(const uint16_t* gate, const uint16_t* up,
                                uint16_t* out, int64_t n, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(n % 8 == 0, "n %% 8 != 0");
  int64_t nv = n / 8;
  int blocks = (int)((nv + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_silu_mul, dim3(blocks), dim3(256), 0, s,
                     reinterpret_cast<const bf16x8*>(gate),
                     reinterpret_cast<const bf16x8*>(up),
                     reinterpret_cast<bf16x8*>(out), nv);
  VH_HIP(hipGetLastError());
  return 0;
}

__global__ void k_silu_mul_bwd(const bf16x8* __restrict__ dy,
                               const bf16x8* __restrict__ g,
                               const bf16x8* __restrict__ u,
                               bf16x8* __restrict__ dg, bf16x8* __restrict__ du,
                               int64_t nv) {
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < nv;
       i += (int64_t)gridDim.x * blockDim.x) {
    bf16x8 dv = dy[i], gv = g[i], uv = u[i], dgv, duv;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      float d = bf2f(dv.v[j]), gf = bf2f(gv.v[j]), uf = bf2f(uv.v[j]);
      dgv.v[j] = f2bf(d * uf * dsiluf(gf));
      duv.v[j] = f2bf(d * siluf(gf));
    }
    dg[i] = dgv;
    du[i] = duv;
  }
}

extern "C" int vh_silu_mul_bwd_bf16(const uint16_t* dy, const uint16_t* gate,
                                    const uint16_t* up, uint16_t* dgate,
                                    uint16_t* dup, int64_t n, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(n % 8 == 0, "n %% 8 != 0");
  int64_t nv = n / 8;
  int blocks = (int)((nv + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_silu_mul_bwd, dim3(blocks), dim3(256), 0, s,
                     reinterpret_cast<const bf16x8*>(dy),
                     reinterpret_cast<const bf16x8*>(gate),
                     reinterpret_cast<const bf16x8*>(up),
                     reinterpret_cast<bf16x8*>(dgate),
                     reinterpret_cast<bf16x8*>(dup), nv);
  VH_HIP(hipGetLastError());
  return 0;
}
