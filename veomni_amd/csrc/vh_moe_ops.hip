// MoE token-bookkeeping + fused-epilogue kernels (HBM-bound, gfx950).
//
// Replaces (semantics, not code):
//   expert_histogram   — ref kernel/moe.py:29-82
//   moe_scatter        — ref kernel/moe.py:253-333
//   moe_gather         — ref kernel/moe.py:87-159 (fp32 accumulation)
//   silu*up*weight     — ref group_gemm.py:105-121 fused (bf16 ops, fp32 math)
//
// Design: bf16 moved as 16-B bf16x8 per lane (guide G13); one workgroup per
// token row for scatter/gather (row = N*2 bytes, threads stride the row).

#include "vh_common.h"

// ---------------------------------------------------------------- histogram
__global__ void k_zero_i32(int32_t* p, int n) {
  int i = blockIdx.x * blockDim.x + threadIdx.x;
  if (i < n) p[i] = 0;
}

__global__ void k_expert_histogram(const int64_t* __restrict__ idx, int64_t n,
                                   int num_experts, int32_t* __restrict__ out) {
  extern __shared__ __attribute__((aligned(16))) char smem_raw[];
  int32_t* smem = reinterpret_cast<int32_t*>(smem_raw);
  for (int e = threadIdx.x; e < num_experts; e += blockDim.x) smem[e] = 0;
  __syncthreads();
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < n;
       i += (int64_t)gridDim.x * blockDim.x) {
    atomicAdd(&smem[(int)idx[i]], 1);
  }
  __syncthreads();
  for (int e = threadIdx.x; e < num_experts; e += blockDim.x)
    if (smem[e] != 0) atomicAdd(&out[e], smem[e]);
}

extern "C" int vh_expert_histogram(const int64_t* expert_index, int64_t n,
                                   int num_experts, int32_t* out, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(num_experts > 0 && num_experts <= 8192, "bad num_experts %d", num_experts);
  int zb = (num_experts + 255) / 256;
  hipLaunchKernelGGL(k_zero_i32, dim3(zb), dim3(256), 0, s, out, num_experts);
  int blocks = (int)((n + 1023) / 1024);
  if (blocks > 1024) blocks = 1024;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_expert_histogram, dim3(blocks), dim3(256),
                     num_experts * sizeof(int32_t), s, expert_index, n,
                     num_experts, out);
  VH_HIP(hipGetLastError());
  return 0;
}

// ------------------------------------------------------------------ scatter
// one block per source row t; each thread copies 16-B chunks of the row to
// the topk destination rows.
__global__ void k_moe_scatter(const bf16x8* __restrict__ x,
                              const int32_t* __restrict__ index,
                              bf16x8* __restrict__ out, int64_t M, int64_t Nv,
                              int topk) {
  int64_t t = blockIdx.x;
  if (t >= M) return;
  const bf16x8* src = x + t * Nv;
  for (int kk = 0; kk < topk; ++kk) {
    int64_t dst_row = index[t * topk + kk];
    bf16x8* dst = out + dst_row * Nv;
    for (int64_t c = threadIdx.x; c < Nv; c += blockDim.x) dst[c] = src[c];
  }
}

extern "C" int vh_moe_scatter_bf16(const uint16_t* x, const int32_t* index,
                                   uint16_t* out, int64_t M, int64_t N,
                                   int topk, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(N % 8 == 0, "N %% 8 != 0 (N=%lld)", (long long)N);
  int64_t Nv = N / 8;
  int threads = Nv < 256 ? (int)Nv : 256;
  hipLaunchKernelGGL(k_moe_scatter, dim3((uint32_t)M), dim3(threads), 0, s,
                     reinterpret_cast<const bf16x8*>(x), index,
                     reinterpret_cast<bf16x8*>(out), M, Nv, topk);
  VH_HIP(hipGetLastError());
  return 0;
}

// ------------------------------------------------------------------- gather
__global__ void k_moe_gather(const bf16x8* __restrict__ x,
                             const int32_t* __restrict__ index,
                             bf16x8* __restrict__ out, int64_t M, int64_t Nv,
                             int topk) {
  int64_t t = blockIdx.x;
  if (t >= M) return;
  bf16x8* dst = out + t * Nv;
  for (int64_t c = threadIdx.x; c < Nv; c += blockDim.x) {
    float acc[8] = {0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f, 0.f};
    for (int kk = 0; kk < topk; ++kk) {
      int64_t src_row = index[t * topk + kk];
      bf16x8 v = x[src_row * Nv + c];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc[j] += bf2f(v.v[j]);
    }
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) o.v[j] = f2bf(acc[j]);
    dst[c] = o;
  }
}

extern "C" int vh_moe_gather_bf16(const uint16_t* x, const int32_t* index,
                                  uint16_t* out, int64_t M, int64_t N,
                                  int topk, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(N % 8 == 0, "N %% 8 != 0 (N=%lld)", (long long)N);
  int64_t Nv = N / 8;
  int threads = Nv < 256 ? (int)Nv : 256;
  hipLaunchKernelGGL(k_moe_gather, dim3((uint32_t)M), dim3(threads), 0, s,
                     reinterpret_cast<const bf16x8*>(x), index,
                     reinterpret_cast<bf16x8*>(out), M, Nv, topk);
  VH_HIP(hipGetLastError());
  return 0;
}

// ------------------------------------------- fused silu(gate)*up*weight
__global__ void k_silu_mul_weighted(const bf16x8* __restrict__ fc1,
                                    const bf16_t* __restrict__ w_row,
                                    bf16x8* __restrict__ out, int64_t rows,
                                    int64_t Iv, int has_w) {
  // fc1 row stride = 2*Iv vectors; gate = [0,Iv), up = [Iv,2Iv)
  int64_t total = rows * Iv;
  for (int64_t i = blockIdx.x * (int64_t)blockDim.x + threadIdx.x; i < total;
       i += (int64_t)gridDim.x * blockDim.x) {
    int64_t r = i / Iv, c = i % Iv;
    bf16x8 g = fc1[r * 2 * Iv + c];
    bf16x8 u = fc1[r * 2 * Iv + Iv + c];
    float w = has_w ? bf2f(w_row[r]) : 1.0f;
    bf16x8 o;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      // match torch bf16 op chain: silu(g) -> bf16, *up -> bf16, *w -> bf16
      float a = bf2f(f2bf(siluf(bf2f(g.v[j]))));
      float b = bf2f(f2bf(a * bf2f(u.v[j])));
      o.v[j] = f2bf(b * w);
    }
    out[r * Iv + c] = o;
  }
}

extern "C" int vh_moe_silu_mul_weighted_bf16(const uint16_t* fc1,
                                             const uint16_t* w_row,
                                             uint16_t* out, int64_t rows,
                                             int64_t I, int has_w,
                                             void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(I % 8 == 0, "I %% 8 != 0");
  int64_t Iv = I / 8;
  int64_t total = rows * Iv;
  int blocks = (int)((total + 255) / 256);
  if (blocks > 2048) blocks = 2048;
  if (blocks < 1) blocks = 1;
  hipLaunchKernelGGL(k_silu_mul_weighted, dim3(blocks), dim3(256), 0, s,
                     reinterpret_cast<const bf16x8*>(fc1),
                     reinterpret_cast<const bf16_t*>(w_row),
                     reinterpret_cast<bf16x8*>(out), rows, Iv, has_w);
  VH_HIP(hipGetLastError());
  return 0;
}

// backward: dgate = dy*w*up*silu'(g); dup = dy*w*silu(g); dw_row = sum dy*silu(g)*up
__global__ void k_silu_mul_weighted_bwd(const bf16x8* __restrict__ dy,
                                        const bf16x8* __restrict__ fc1,
                                        const bf16_t* __restrict__ w_row,
                                        bf16x8* __restrict__ dfc1,
                                        float* __restrict__ dw_row,
                                        int64_t rows, int64_t Iv, int has_w) {
  // one wave per row-chunk; waves stride rows so the per-row dw reduce stays
  // in-wave (one atomic per wave per row).
  int wave = (blockIdx.x * blockDim.x + threadIdx.x) / kWave;
  int lane = threadIdx.x & (kWave - 1);
  int num_waves = (gridDim.x * blockDim.x) / kWave;
  for (int64_t r = wave; r < rows; r += num_waves) {
    float w = has_w ? bf2f(w_row[r]) : 1.0f;
    float dw_acc = 0.f;
    for (int64_t c = lane; c < Iv; c += kWave) {
      bf16x8 g8 = fc1[r * 2 * Iv + c];
      bf16x8 u8 = fc1[r * 2 * Iv + Iv + c];
      bf16x8 d8 = dy[r * Iv + c];
      bf16x8 dg8, du8;
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        float g = bf2f(g8.v[j]), u = bf2f(u8.v[j]), d = bf2f(d8.v[j]);
        float sg = siluf(g);
        float dact = d * w;           // grad wrt silu(g)*up
        dg8.v[j] = f2bf(dact * u * dsiluf(g));
        du8.v[j] = f2bf(dact * sg);
        dw_acc += d * sg * u;
      }
      dfc1[r * 2 * Iv + c] = dg8;
      dfc1[r * 2 * Iv + Iv + c] = du8;
    }
    if (has_w && dw_row != nullptr) {
#pragma unroll
      for (int off = kWave / 2; off > 0; off >>= 1)
        dw_acc += __shfl_down(dw_acc, off, kWave);
      if (lane == 0) atomicAdd(&dw_row[r], dw_acc);
    }
  }
}

extern "C" int vh_moe_silu_mul_weighted_bwd_bf16(
    const uint16_t* dy, const uint16_t* fc1, const uint16_t* w_row,
    uint16_t* dfc1, float* dw_row, int64_t rows, int64_t I, int has_w,
    void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(I % 8 == 0, "I %% 8 != 0");
  int64_t Iv = I / 8;
  int blocks = (int)rows < 2048 ? (int)(rows ? rows : 1) : 2048;
  hipLaunchKernelGGL(k_silu_mul_weighted_bwd, dim3(blocks), dim3(256), 0, s,
                     reinterpret_cast<const bf16x8*>(dy),
                     reinterpret_cast<const bf16x8*>(fc1),
                     reinterpret_cast<const bf16_t*>(w_row),
                     reinterpret_cast<bf16x8*>(dfc1), dw_row, rows, Iv, has_w);
  VH_HIP(hipGetLastError());
  return 0;
}

// ---------------------------------------------------------------------------
// Batched expert-weight transpose: [E, M, N] -> [E, N, M] bf16.
// The dgrad path needs W^T per step; torch's permuted-copy ran at ~1.4 TB/s
// on [128, 1536, 2048] (strided writes). [64][72] LDS tile, both sides b128.
// M % 64 == 0 and N % 64 == 0 (production shapes; host falls back otherwise).
// ---------------------------------------------------------------------------
namespace {
__global__ void k_wtranspose(const bf16_t* __restrict__ src,
                             bf16_t* __restrict__ dst, int64_t M, int64_t N) {
  __shared__ __attribute__((aligned(16))) bf16_t t2[64][72];
  const int64_t e = blockIdx.z;
  const int64_t m0 = (int64_t)blockIdx.x * 64;
  const int64_t n0 = (int64_t)blockIdx.y * 64;
  const bf16_t* s = src + e * M * N;
  bf16_t* d = dst + e * M * N;
  const int tid = threadIdx.x;
  for (int r = tid / 8; r < 64; r += 32) {
    const int cpos = (tid % 8) * 8;
    *reinterpret_cast<bf16x8*>(&t2[r][cpos]) =
        *reinterpret_cast<const bf16x8*>(s + (m0 + r) * N + n0 + cpos);
  }
  __syncthreads();
  for (int c = tid / 8; c < 64; c += 32) {
    const int rpos = (tid % 8) * 8;
    bf16x8 v;
#pragma unroll
    for (int j = 0; j < 8; ++j) v.v[j] = t2[rpos + j][c];
    *reinterpret_cast<bf16x8*>(d + (n0 + c) * M + m0 + rpos) = v;
  }
}
}  // namespace

extern "C" int vh_wtranspose_bf16(const uint16_t* src, uint16_t* dst, int E,
                                  int64_t M, int64_t N, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(M % 64 == 0 && N % 64 == 0, "wtranspose needs 64-multiples");
  dim3 grid((uint32_t)(M / 64), (uint32_t)(N / 64), (uint32_t)E);
  hipLaunchKernelGGL(k_wtranspose, grid, dim3(256), 0, s,
                     reinterpret_cast<const bf16_t*>(src),
                     reinterpret_cast<bf16_t*>(dst), M, N);
  VH_HIP(hipGetLastError());
  return 0;
}
