// 256x256 8-phase grouped GEMM for gfx950 (bf16, fp32 accumulate).
//
// The deep-pipelined variant of vh_group_gemm.hip's same-NK kernel, built on
// the guide's 256-square 8-phase recipe (cdna_hip_programming.md §5 "The 256²
// 8-phase template" + T3/T4/T5): counted vmcnt (loads stay in flight across
// barriers), raw s_barrier (never __syncthreads — it would drain the LDS-DMA
// queue), s_setprio around each MFMA cluster.
//
// Geometry:
//   tile 256x256, K consumed in 32-deep k-subs; 512 threads = 8 waves in a
//   2(M) x 4(N) grid, 128x64 per wave = 8x4 16x16 fragments (128 acc VGPRs).
//   LDS = 4-deep ring of k-sub slots x {A, B}, slot = [256 rows][32 k] bf16
//   (16 KiB) -> 128 KiB total, 1 block/CU, 2 waves/SIMD.
//   Per k-sub: 2 phases x {8 ds_read_b128 + 16 MFMA + raw barrier}.
//   Staging: 4 glds (A 2 + B 2) issued at phase 0 of each k-sub for slot
//   s+3; vmcnt(12) then admits slot s while 3 slots stay in flight.
//
// Dispatched by vh_group_gemm_nk_bf16 (vh_group_gemm.hip) for large shapes.

#include "vh_common.h"

namespace {

constexpr int BM8 = 256, BN8 = 256, KSUB = 32;
constexpr int THREADS8 = 512;

using bf16frag = __attribute__((ext_vector_type(8))) __bf16;

__device__ __forceinline__ void glds16(const bf16_t* g, bf16_t* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)l, 16, 0, 0);
}

// swizzle for the [256][32] k-sub slot (row = 64 B): 4 16-B slots per row;
// bank row = 256 B = 4 tile rows, so rows r, r+4, r+8, r+12 share a bank
// phase -> slot = (row>>2)&3 makes each mod-4 class conflict-free over the
// 16 consecutive rows a ds_read_b128 lane group touches.
__device__ __forceinline__ int swz32(int row, int colb) {
  return colb ^ (((row >> 2) & 3) << 4);
}

__device__ __forceinline__ void raw_barrier() {
  asm volatile("" ::: "memory");
  __builtin_amdgcn_s_barrier();
  asm volatile("" ::: "memory");
}

#define VMCNT(n) asm volatile("s_waitcnt vmcnt(" #n ")" ::: "memory")

// K-contiguous staging of one [256][32] slot via glds (2 instructions of
// 8 KiB; per-lane source carries the inverse swizzle).
struct KStage8 {
  const bf16_t* src[2];
  int lds_base[2];

  template <typename RowFn>
  __device__ __forceinline__ void init(const bf16_t* s, int64_t ld_elems,
                                       RowFn row_of, int tid) {
    const int lane = tid & 63;
    const int wave = tid >> 6;
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int base = i * 8192 + wave * 1024;
      int o = base + lane * 16;
      int row = o >> 6;           // 64 B per row
      int colb = o & 63;
      src[i] = s + row_of(row) * ld_elems + (swz32(row, colb) >> 1);
      lds_base[i] = base;
    }
  }

  __device__ __forceinline__ void stage(bf16_t* slot, int64_t k0) const {
#pragma unroll
    for (int i = 0; i < 2; ++i) glds16(src[i] + k0, slot + (lds_base[i] >> 1));
  }
};

// Transposed staging of one [256 out][32 k] slot from a [k][out] source.
// Thread t: o0 = (t&31)*8, kp = (t>>5)*2 -> 2 x 16-B loads (8 consecutive
// outs, k and k+1; 32 lanes x 16 B = 512 B coalesced), 8 x ds_write_b32
// (the (k, k+1) pair is contiguous in the [out][k] image).
struct TStage8 {
  const bf16_t* base;
  int64_t ld;
  bool out_ok;
  int64_t gout_left;
  int o0, kp;

  __device__ __forceinline__ void init(const bf16_t* s, int64_t ld_elems,
                                       int out0, int out_max, int tid) {
    o0 = (tid & 31) * 8;
    kp = (tid >> 5) * 2;
    int64_t gout = (int64_t)out0 + o0;
    gout_left = out_max - gout;
    out_ok = gout_left >= 8;
    base = s + gout;
    ld = ld_elems;
  }

  __device__ __forceinline__ void stage(bf16_t* slot, int64_t k0) const {
    bf16x8 v[2];
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int64_t k = k0 + kp + i;
      if (out_ok) {
        v[i] = *reinterpret_cast<const bf16x8*>(base + k * ld);
      } else if (gout_left > 0) {
        const bf16_t* p = base + k * ld;
#pragma unroll
        for (int j = 0; j < 8; ++j) v[i].v[j] = (j < gout_left) ? p[j] : bf16_t(0);
      } else {
        v[i] = bf16x8{};
      }
    }
    const int colb = kp * 2;  // 4-B aligned pair position
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int row = o0 + j;
      ushort2 pk = {v[0].v[j], v[1].v[j]};
      *reinterpret_cast<uint32_t*>(&slot[(row * 64 + swz32(row, colb & ~15) + (colb & 15)) >> 1]) =
          *reinterpret_cast<uint32_t*>(&pk);
    }
  }
};

__device__ __forceinline__ bf16frag frag_read8(const bf16_t* slot, int row0,
                                               int lane) {
  int row = row0 + (lane & 15);
  int colb = ((lane >> 4) << 3) * 2;  // k position within the 32-k slot
  int off_b = row * 64 + swz32(row, colb);
  return *reinterpret_cast<const bf16frag*>(
      reinterpret_cast<const char*>(slot) + off_b);
}

template <bool TRANS_B>
__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_nk8(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ cumsum, int G,
    int64_t N, int64_t K, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto slotA = [&](int s) { return sm + (s & 3) * 8192; };          // 4 x 16 KiB
  auto slotB = [&](int s) { return sm + 32768 + (s & 3) * 8192; };  // 4 x 16 KiB

  const int gid = blockIdx.y;
  const int64_t row_start = (gid > 0) ? cumsum[gid - 1] : 0;
  const int64_t m_size = cumsum[gid] - row_start;
  const int bm = blockIdx.x / tiles_n;
  const int bn = blockIdx.x % tiles_n;
  if ((int64_t)bm * BM8 >= m_size) return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;  // 2x4 wave grid; 128x64 per wave

  const bf16_t* Ag = A + row_start * K;
  const bf16_t* Bg = B + (int64_t)gid * N * K;
  bf16_t* Cg = C + row_start * N;

  KStage8 sa;
  sa.init(Ag, K, [&](int r) -> int64_t {
    int64_t gm = (int64_t)bm * BM8 + r;
    return gm % m_size;
  }, tid);
  KStage8 sb_k;
  TStage8 sb_t;
  if (TRANS_B) {
    sb_k.init(Bg, K, [&](int r) -> int64_t {
      int64_t gn = (int64_t)bn * BN8 + r;
      return gn % N;
    }, tid);
  } else {
    sb_t.init(Bg, N, bn * BN8, (int)N, tid);
  }

  auto stage = [&](int s, int64_t k0) {
    sa.stage(slotA(s), k0);
    if (TRANS_B) sb_k.stage(slotB(s), k0);
    else sb_t.stage(slotB(s), k0);
  };

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nsub = (int)(K / KSUB);
  // prologue: slots 0..2 in flight
  stage(0, 0);
  stage(1, KSUB);
  if (nsub > 2) stage(2, 2 * KSUB);
  for (int s = 0; s < nsub; ++s) {
    // phase 0: issue next staging, admit slot s, compute fragment half 0
    if (s + 3 < nsub) stage(s + 3, (int64_t)(s + 3) * KSUB);
    // counted wait: slot s must have landed; younger slots stay in flight.
    // glds per in-flight slot: 4 (A+B) on the TRANS_B path, 2 (A only — B is
    // synchronous ds_writes) otherwise. Tail k-subs have fewer in flight.
    {
      const int rem = nsub - s;  // slots still live: s .. min(s+3, nsub-1)
      if (TRANS_B) {
        if (rem >= 4) VMCNT(12);
        else if (rem == 3) VMCNT(8);
        else if (rem == 2) VMCNT(4);
        else VMCNT(0);
      } else {
        if (rem >= 4) VMCNT(6);
        else if (rem == 3) VMCNT(4);
        else if (rem == 2) VMCNT(2);
        else VMCNT(0);
      }
    }
    raw_barrier();
    {
      const bf16_t* TA = slotA(s);
      const bf16_t* TB = slotB(s);
      bf16frag af[4], bfr[4];
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag_read8(TA, wr * 128 + i * 16, lane);
#pragma unroll
      for (int j = 0; j < 4; ++j) bfr[j] = frag_read8(TB, wc * 64 + j * 16, lane);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      raw_barrier();
      // phase 1: fragment half 1 (rows 64..127 of this wave's panel)
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag_read8(TA, wr * 128 + 64 + i * 16, lane);
#pragma unroll
      for (int j = 0; j < 4; ++j) bfr[j] = frag_read8(TB, wc * 64 + j * 16, lane);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i + 4][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    raw_barrier();
  }

  const int col_in = lane & 15;
  const int row_base_in = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                    row_base_in + rr;
        int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
        if (m < m_size && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
      }
}

}  // namespace

extern "C" int vh_group_gemm_nk8_bf16(const uint16_t* A, const uint16_t* B,
                                      uint16_t* C, const int64_t* cumsum,
                                      int G, int64_t N, int64_t K,
                                      int64_t total_rows, int trans_b,
                                      void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(K % KSUB == 0 && K / KSUB >= 2, "K must be a multiple of 32, >= 64");
  VH_CHECK(N % 16 == 0, "N %% 16 != 0");
  int tiles_m = (int)((total_rows + BM8 - 1) / BM8);
  if (tiles_m < 1) tiles_m = 1;
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  dim3 grid(tiles_m * tiles_n, G);
  if (trans_b)
    hipLaunchKernelGGL((k_group_gemm_nk8<true>), grid, dim3(THREADS8), 131072,
                       s, reinterpret_cast<const bf16_t*>(A),
                       reinterpret_cast<const bf16_t*>(B),
                       reinterpret_cast<bf16_t*>(C), cumsum, G, N, K, tiles_m,
                       tiles_n);
  else
    hipLaunchKernelGGL((k_group_gemm_nk8<false>), grid, dim3(THREADS8), 131072,
                       s, reinterpret_cast<const bf16_t*>(A),
                       reinterpret_cast<const bf16_t*>(B),
                       reinterpret_cast<bf16_t*>(C), cumsum, G, N, K, tiles_m,
                       tiles_n);
  VH_HIP(hipGetLastError());
  return 0;
}

// ============================================================================
// Register-staged 256x256 ring kernels (no glds — every load is
// compiler-visible, so hipcc emits counted waits and the T14 split pipelines
// cleanly; mixing glds with ordinary loads makes the compiler drain vmcnt(0)
// at every ds_write — the §5 "Three .s-level traps" (b) case measured as 61%
// SQ_WAIT_ANY on the earlier dgrad variant).
//
//   k_group_gemm_dgrad8: C_g = A_g @ B_g, A rows K-contiguous (wrapped),
//       B [K,N] transposed-staged. Replaces the !trans_b nk path.
//   k_group_gemm_mn8:    C[g] = A_g^T @ B_g, both operands transposed-staged,
//       per-group ragged K (row count).
//
// Shared ring: 4 k-sub slots x {A, B} (16 KiB each, 128 KiB LDS, 1 block/CU,
// 8 waves). Per k-sub: load slot s+2's registers, two 16-MFMA phases on slot
// s, flush slot s+2's ds_writes, one barrier.
// ============================================================================

namespace {

// K-contiguous register staging of a [256 row][32 k] slot with row wrap.
struct KRegStage8 {
  const bf16_t* src[2];
  int lds_off[2];  // bf16-element offsets

  template <typename RowFn>
  __device__ __forceinline__ void init(const bf16_t* s, int64_t ld_elems,
                                       RowFn row_of, int tid) {
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      int unit = tid + u * 512;          // 1024 units of [1 row][8 k]
      int row = unit & 255;
      int ko = (unit >> 8) * 8;          // 0,8,16,24
      src[u] = s + row_of(row) * ld_elems + ko;
      lds_off[u] = (row * 64 + swz32(row, ko * 2)) >> 1;
    }
  }

  __device__ __forceinline__ void load(bf16x8 (&v)[2], int64_t k0) const {
    v[0] = *reinterpret_cast<const bf16x8*>(src[0] + k0);
    v[1] = *reinterpret_cast<const bf16x8*>(src[1] + k0);
  }

  __device__ __forceinline__ void flush(bf16_t* slot, const bf16x8 (&v)[2]) const {
    *reinterpret_cast<bf16x8*>(&slot[lds_off[0]]) = v[0];
    *reinterpret_cast<bf16x8*>(&slot[lds_off[1]]) = v[1];
  }
};

// Transposed register staging with split load/flush and optional k clamp.
template <bool CLAMP>
struct TRegStage8 {
  const bf16_t* base;
  int64_t ld, kmax;
  bool out_ok;
  int64_t gout_left;
  int o0, kp;

  __device__ __forceinline__ void init(const bf16_t* s, int64_t ld_elems,
                                       int64_t kmax_, int out0, int out_max,
                                       int tid) {
    o0 = (tid & 31) * 8;
    kp = (tid >> 5) * 2;
    int64_t gout = (int64_t)out0 + o0;
    gout_left = out_max - gout;
    out_ok = gout_left >= 8;
    base = s + gout;
    ld = ld_elems;
    kmax = kmax_;
  }

  __device__ __forceinline__ void load(bf16x8 (&v)[2], int64_t k0) const {
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int64_t k = k0 + kp + i;
      bool kok = !CLAMP || k < kmax;
      if (kok && out_ok) {
        v[i] = *reinterpret_cast<const bf16x8*>(base + k * ld);
      } else if (kok && gout_left > 0) {
        const bf16_t* p = base + k * ld;
#pragma unroll
        for (int j = 0; j < 8; ++j) v[i].v[j] = (j < gout_left) ? p[j] : bf16_t(0);
      } else {
        v[i] = bf16x8{};
      }
    }
  }

  __device__ __forceinline__ void flush(bf16_t* slot, const bf16x8 (&v)[2]) const {
    const int colb = kp * 2;
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int row = o0 + j;
      ushort2 pk = {v[0].v[j], v[1].v[j]};
      *reinterpret_cast<uint32_t*>(&slot[(row * 64 + swz32(row, colb & ~15) + (colb & 15)) >> 1]) =
          *reinterpret_cast<uint32_t*>(&pk);
    }
  }
};

#define VH_MFMA_PHASE8(TA_, TB_, ROWOFF, ACCOFF)                               \
  {                                                                            \
    bf16frag af[4], bfr[4];                                                    \
    _Pragma("unroll") for (int i = 0; i < 4; ++i)                              \
        af[i] = frag_read8(TA_, wr * 128 + (ROWOFF) + i * 16, lane);           \
    _Pragma("unroll") for (int j = 0; j < 4; ++j)                              \
        bfr[j] = frag_read8(TB_, wc * 64 + j * 16, lane);                      \
    __builtin_amdgcn_s_setprio(1);                                             \
    _Pragma("unroll") for (int i = 0; i < 4; ++i)                              \
        _Pragma("unroll") for (int j = 0; j < 4; ++j)                          \
            acc[i + (ACCOFF)][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(    \
                af[i], bfr[j], acc[i + (ACCOFF)][j], 0, 0, 0);                 \
    __builtin_amdgcn_s_setprio(0);                                             \
  }

// ring skeleton shared by both kernels. Two pending register sets: slot
// s+3's loads issue at k-sub s and flush at the END of k-sub s+1 — the
// load->flush gap spans TWO 32-MFMA blocks (~2k cycles) so HBM/L3 latency
// is fully covered.
#define VH_RING_LOOP(SA, SB, NSUB)                                             \
  {                                                                            \
    bf16x8 va0[2], vb0[2];                                                     \
    SA.load(va0, 0);                                                           \
    SB.load(vb0, 0);                                                           \
    SA.flush(slotA(0), va0);                                                   \
    SB.flush(slotB(0), vb0);                                                   \
    if ((NSUB) > 1) {                                                          \
      SA.load(va0, KSUB);                                                      \
      SB.load(vb0, KSUB);                                                      \
      SA.flush(slotA(1), va0);                                                 \
      SB.flush(slotB(1), vb0);                                                 \
    }                                                                          \
    bf16x8 vap[2], vbp[2], van[2], vbn[2];                                     \
    if ((NSUB) > 2) {                                                          \
      SA.load(vap, 2 * KSUB);                                                  \
      SB.load(vbp, 2 * KSUB);                                                  \
    }                                                                          \
    __syncthreads();                                                           \
    for (int s = 0; s < (NSUB); ++s) {                                         \
      const bool more = s + 3 < (NSUB);                                        \
      if (more) {                                                              \
        SA.load(van, (int64_t)(s + 3) * KSUB);                                 \
        SB.load(vbn, (int64_t)(s + 3) * KSUB);                                 \
      }                                                                        \
      {                                                                        \
        const bf16_t* TA = slotA(s);                                           \
        const bf16_t* TB = slotB(s);                                           \
        VH_MFMA_PHASE8(TA, TB, 0, 0)                                           \
        VH_MFMA_PHASE8(TA, TB, 64, 4)                                          \
      }                                                                        \
      if (s + 2 < (NSUB)) {                                                    \
        SA.flush(slotA(s + 2), vap);                                           \
        SB.flush(slotB(s + 2), vbp);                                           \
      }                                                                        \
      _Pragma("unroll") for (int q = 0; q < 2; ++q) {                          \
        vap[q] = van[q];                                                       \
        vbp[q] = vbn[q];                                                       \
      }                                                                        \
      __syncthreads();                                                         \
    }                                                                          \
  }

__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_dgrad8(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ cumsum, int G,
    int64_t N, int64_t K, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto slotA = [&](int s) { return sm + (s & 3) * 8192; };
  auto slotB = [&](int s) { return sm + 32768 + (s & 3) * 8192; };

  const int gid = blockIdx.y;
  const int64_t row_start = (gid > 0) ? cumsum[gid - 1] : 0;
  const int64_t m_size = cumsum[gid] - row_start;
  const int bm = blockIdx.x / tiles_n;
  const int bn = blockIdx.x % tiles_n;
  if ((int64_t)bm * BM8 >= m_size) return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  const bf16_t* Ag = A + row_start * K;
  const bf16_t* Bg = B + (int64_t)gid * K * N;  // [K, N]
  bf16_t* Cg = C + row_start * N;

  KRegStage8 sa;
  sa.init(Ag, K, [&](int r) -> int64_t {
    int64_t gm = (int64_t)bm * BM8 + r;
    return gm % m_size;
  }, tid);
  TRegStage8<false> sb;
  sb.init(Bg, N, K, bn * BN8, (int)N, tid);

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nsub = (int)(K / KSUB);
  VH_RING_LOOP(sa, sb, nsub)

  const int col_in = lane & 15;
  const int row_base_in = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                    row_base_in + rr;
        int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
        if (m < m_size && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
      }
}

__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_mn8(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ cumsum, int G,
    int64_t M, int64_t N, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto slotA = [&](int s) { return sm + (s & 3) * 8192; };
  auto slotB = [&](int s) { return sm + 32768 + (s & 3) * 8192; };

  const int gid = blockIdx.y;
  const int64_t row_start = (gid > 0) ? cumsum[gid - 1] : 0;
  const int64_t kcount = cumsum[gid] - row_start;
  const int bm = blockIdx.x / tiles_n;
  const int bn = blockIdx.x % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  bf16_t* Cg = C + (int64_t)gid * M * N;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  if (kcount > 0) {
    const bf16_t* Ag = A + row_start * M;
    const bf16_t* Bg = B + row_start * N;
    TRegStage8<true> sa, sb;
    sa.init(Ag, M, kcount, bm * BM8, (int)M, tid);
    sb.init(Bg, N, kcount, bn * BN8, (int)N, tid);
    const int nsub = (int)((kcount + KSUB - 1) / KSUB);
    VH_RING_LOOP(sa, sb, nsub)
  }

  const int col_in = lane & 15;
  const int row_base_in = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                    row_base_in + rr;
        int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
        if (m < M && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
      }
}

}  // namespace

extern "C" int vh_group_gemm_dgrad8_bf16(const uint16_t* A, const uint16_t* B,
                                         uint16_t* C, const int64_t* cumsum,
                                         int G, int64_t N, int64_t K,
                                         int64_t total_rows, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(K % KSUB == 0, "K %% 32 != 0");
  int tiles_m = (int)((total_rows + BM8 - 1) / BM8);
  if (tiles_m < 1) tiles_m = 1;
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  dim3 grid(tiles_m * tiles_n, G);
  hipLaunchKernelGGL(k_group_gemm_dgrad8, grid, dim3(THREADS8), 131072, s,
                     reinterpret_cast<const bf16_t*>(A),
                     reinterpret_cast<const bf16_t*>(B),
                     reinterpret_cast<bf16_t*>(C), cumsum, G, N, K, tiles_m,
                     tiles_n);
  VH_HIP(hipGetLastError());
  return 0;
}

extern "C" int vh_group_gemm_mn8_bf16(const uint16_t* A, const uint16_t* B,
                                      uint16_t* C, const int64_t* cumsum,
                                      int G, int64_t M, int64_t N,
                                      void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  int tiles_m = (int)((M + BM8 - 1) / BM8);
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  dim3 grid(tiles_m * tiles_n, G);
  hipLaunchKernelGGL(k_group_gemm_mn8, grid, dim3(THREADS8), 131072, s,
                     reinterpret_cast<const bf16_t*>(A),
                     reinterpret_cast<const bf16_t*>(B),
                     reinterpret_cast<bf16_t*>(C), cumsum, G, M, N, tiles_m,
                     tiles_n);
  VH_HIP(hipGetLastError());
  return 0;
}

// ============================================================================
// 256x256 x BK=64 double-buffered glds kernel ("nk256"): the 128x128 2-phase
// structure scaled to a 256-square tile — 2x the MFMA work per staged byte
// and per barrier drain, with the proven conflict-free 128-B-row swizzle.
// 8 waves (2x4), per wave 128x64; LDS = 2 x (A 32K + B 32K) = 128 KiB.
// ============================================================================

namespace {

constexpr int BK64 = 64;

// conflict-free slot map for 128-B rows (see vh_group_gemm.hip swz comment)
__device__ __forceinline__ int swz64(int row, int colb) {
  return colb ^ ((((row >> 1) ^ (row >> 3)) & 7) << 4);
}

struct KStage64 {  // [256][64] tile via 4 glds of 8 KiB (512 threads)
  const bf16_t* src[4];
  int lds_base[4];

  template <typename RowFn>
  __device__ __forceinline__ void init(const bf16_t* s, int64_t ld_elems,
                                       RowFn row_of, int tid) {
    const int lane = tid & 63;
    const int wave = tid >> 6;
#pragma unroll
    for (int i = 0; i < 4; ++i) {
      int base = i * 8192 + wave * 1024;
      int o = base + lane * 16;
      int row = o >> 7;  // 128 B per row
      int colb = o & 127;
      src[i] = s + row_of(row) * ld_elems + (swz64(row, colb) >> 1);
      lds_base[i] = base;
    }
  }

  __device__ __forceinline__ void stage(bf16_t* tile, int64_t k0) const {
#pragma unroll
    for (int i = 0; i < 4; ++i) glds16(src[i] + k0, tile + (lds_base[i] >> 1));
  }
};

__device__ __forceinline__ bf16frag frag_read64(const bf16_t* tile, int row0,
                                                int ks, int lane) {
  int row = row0 + (lane & 15);
  int colb = (ks * 32 + ((lane >> 4) << 3)) * 2;
  int off_b = row * 128 + swz64(row, colb);
  return *reinterpret_cast<const bf16frag*>(
      reinterpret_cast<const char*>(tile) + off_b);
}

__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_nk256(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ cumsum, int G,
    int64_t N, int64_t K, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto ta = [&](int buf) { return sm + buf * 32768; };          // 2 x 32 KiB
  auto tb = [&](int buf) { return sm + 16384 + buf * 32768; };  // 2 x 32 KiB

  const int gid = blockIdx.y;
  const int64_t row_start = (gid > 0) ? cumsum[gid - 1] : 0;
  const int64_t m_size = cumsum[gid] - row_start;
  const int bm = blockIdx.x / tiles_n;
  const int bn = blockIdx.x % tiles_n;
  if ((int64_t)bm * BM8 >= m_size) return;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  const bf16_t* Ag = A + row_start * K;
  const bf16_t* Bg = B + (int64_t)gid * N * K;
  bf16_t* Cg = C + row_start * N;

  KStage64 sa, sb;
  sa.init(Ag, K, [&](int r) -> int64_t {
    int64_t gm = (int64_t)bm * BM8 + r;
    return gm % m_size;
  }, tid);
  sb.init(Bg, K, [&](int r) -> int64_t {
    int64_t gn = (int64_t)bn * BN8 + r;
    return gn % N;
  }, tid);

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  const int nk = (int)(K / BK64);
  sa.stage(ta(0), 0);
  sb.stage(tb(0), 0);
  __syncthreads();
  int cur = 0;
  for (int t = 0; t < nk; ++t) {
    if (t + 1 < nk) {
      sa.stage(ta(cur ^ 1), (int64_t)(t + 1) * BK64);
      sb.stage(tb(cur ^ 1), (int64_t)(t + 1) * BK64);
    }
#pragma unroll
    for (int ks = 0; ks < 2; ++ks) {
      bf16frag af[4], bfr[4];
#pragma unroll
      for (int j = 0; j < 4; ++j) bfr[j] = frag_read64(tb(cur), wc * 64 + j * 16, ks, lane);
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag_read64(ta(cur), wr * 128 + i * 16, ks, lane);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag_read64(ta(cur), wr * 128 + 64 + i * 16, ks, lane);
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i + 4][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
    }
    __syncthreads();
    cur ^= 1;
  }

  const int col_in = lane & 15;
  const int row_base_in = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                    row_base_in + rr;
        int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
        if (m < m_size && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
      }
}

// ============================================================================
// nk256s: the nk256 inner loop under a DEVICE-BUILT tile schedule with
// XCD-clustered persistent blocks.
//
// Why (PMC, profiles/r02_groupgemm_pmc.txt): nk256's grid (tiles x G)
// dispatches consecutive blocks round-robin across the 8 XCDs, so blocks
// sharing a group's A/B tiles land on DIFFERENT L2s — zero reuse. At the
// bench shapes every 256x256 tile streams ~2 MB from HBM (~6.5 TB/s:
// HBM-bound at 20% MFMA-busy, vs the Tensile reference's 82%). The grid is
// also sized for worst-case skew (ceil(total/256) m-tiles for EVERY group):
// at G=128 that is ~786k blocks, 99% of which exit immediately.
//
// Structure:
//   - k_gg_sched (1 tiny launch) turns cumsum into a per-group tile prefix
//     sum in a device workspace: group g owns tiles_m_g x tiles_n tiles,
//     enumerated bm-inner (consecutive tiles share the B n-slab).
//   - the GEMM launches a FIXED grid of 256 persistent blocks (1/CU at
//     128 KiB LDS). Block b runs on XCD b%8 (dispatch affinity);
//     it walks the CONTIGUOUS tile range [X*L, (X+1)*L) of the global list
//     with stride 32, so the ~32 concurrently-resident blocks of an XCD
//     process consecutive tiles of the SAME group — the group's B n-slab
//     (~1 MB) and the shared A m-tiles stay hot in that XCD's 4 MB L2.
//   - zero-count groups occupy no schedule slots; per-tile group lookup is
//     a binary search over the prefix (G<=4096).
// ============================================================================

__global__ void k_gg_sched(const int64_t* __restrict__ cumsum, int G,
                           int tiles_n, int* __restrict__ ws) {
  if (threadIdx.x == 0 && blockIdx.x == 0) {
    int total = 0;
    int64_t prev = 0;
    for (int g = 0; g < G; ++g) {
      int64_t m = cumsum[g] - prev;
      prev = cumsum[g];
      ws[1 + g] = total;
      total += (int)((m + BM8 - 1) / BM8) * tiles_n;
    }
    ws[1 + G] = total;
    ws[0] = total;
  }
}

__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_nk256s(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ cumsum, int G,
    int64_t N, int64_t K, int tiles_n, const int* __restrict__ ws) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto ta = [&](int buf) { return sm + buf * 32768; };          // 2 x 32 KiB
  auto tb = [&](int buf) { return sm + 16384 + buf * 32768; };  // 2 x 32 KiB

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  const int total_tiles = ws[0];
  const int X = blockIdx.x & 7;        // this block's XCD (dispatch affinity)
  const int slot = blockIdx.x >> 3;    // 0..31 within the XCD
  const int L = (total_tiles + 7) / 8; // contiguous tile range per XCD
  const int t_end = min((X + 1) * L, total_tiles);
  const int nk = (int)(K / BK64);

  for (int t = X * L + slot; t < t_end; t += 32) {
    // group lookup: largest g with ws[1+g] <= t
    int lo = 0, hi = G - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (ws[1 + mid] <= t) lo = mid;
      else hi = mid - 1;
    }
    const int gid = lo;
    const int local = t - ws[1 + gid];
    const int tiles_m_g = (ws[2 + gid] - ws[1 + gid]) / tiles_n;
    const int bm = local % tiles_m_g;   // bm-inner: consecutive t share bn
    const int bn = local / tiles_m_g;

    const int64_t row_start = (gid > 0) ? cumsum[gid - 1] : 0;
    const int64_t m_size = cumsum[gid] - row_start;
    const bf16_t* Ag = A + row_start * K;
    const bf16_t* Bg = B + (int64_t)gid * N * K;
    bf16_t* Cg = C + row_start * N;

    KStage64 sa, sb;
    sa.init(Ag, K, [&](int r) -> int64_t {
      int64_t gm = (int64_t)bm * BM8 + r;
      return gm % m_size;
    }, tid);
    sb.init(Bg, K, [&](int r) -> int64_t {
      int64_t gn = (int64_t)bn * BN8 + r;
      return gn % N;
    }, tid);

    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    sa.stage(ta(0), 0);
    sb.stage(tb(0), 0);
    __syncthreads();
    int cur = 0;
    for (int kt = 0; kt < nk; ++kt) {
      if (kt + 1 < nk) {
        sa.stage(ta(cur ^ 1), (int64_t)(kt + 1) * BK64);
        sb.stage(tb(cur ^ 1), (int64_t)(kt + 1) * BK64);
      }
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16frag af[4], bfr[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) bfr[j] = frag_read64(tb(cur), wc * 64 + j * 16, ks, lane);
#pragma unroll
        for (int i = 0; i < 4; ++i) af[i] = frag_read64(ta(cur), wr * 128 + i * 16, ks, lane);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int i = 0; i < 4; ++i) af[i] = frag_read64(ta(cur), wr * 128 + 64 + i * 16, ks, lane);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i + 4][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();
      cur ^= 1;
    }

    const int col_in = lane & 15;
    const int row_base_in = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                      row_base_in + rr;
          int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
          if (m < m_size && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
        }
  }
}


// nk256s32: the nk256s persistent/XCD-clustered wrapper with the inner loop
// on v_mfma_f32_32x32x16_bf16 instead of 16x16x32. PMC on nk256s
// (profiles/r02_gg_pmc_sq.csv) shows the kernel ISSUE-bound
// (SQ_WAIT_INST_ANY 42%, MFMA pipe 47% busy, zero LDS conflicts); the
// 32x32 shape halves MFMA instruction count per FLOP (and is 1024 vs 964
// FLOP/cyc-SIMD dense), attacking the issue stall directly. Fragment
// mappings: A/B frag = 8 contiguous k at (row0 + lane%32,
// ks*16 + (lane>>5)*8); C element (m = (r&3)+8*(r>>2)+4*(lane>>5),
// n = lane&31) — the attention kernels' 32x32 convention.
typedef __attribute__((ext_vector_type(16))) float f32x16gg;

__device__ __forceinline__ bf16frag frag_read64_32(const bf16_t* tile,
                                                   int row0, int ks,
                                                   int lane) {
  int row = row0 + (lane & 31);
  int colb = (ks * 16 + ((lane >> 5) << 3)) * 2;
  int off_b = row * 128 + swz64(row, colb);
  return *reinterpret_cast<const bf16frag*>(
      reinterpret_cast<const char*>(tile) + off_b);
}

__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_nk256s32(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ cumsum, int G,
    int64_t N, int64_t K, int tiles_n, const int* __restrict__ ws) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto ta = [&](int buf) { return sm + buf * 32768; };          // 2 x 32 KiB
  auto tb = [&](int buf) { return sm + 16384 + buf * 32768; };  // 2 x 32 KiB

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  const int total_tiles = ws[0];
  const int X = blockIdx.x & 7;
  const int slot = blockIdx.x >> 3;
  const int L = (total_tiles + 7) / 8;
  const int t_end = min((X + 1) * L, total_tiles);
  const int nk = (int)(K / BK64);

  for (int t = X * L + slot; t < t_end; t += 32) {
    int lo = 0, hi = G - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (ws[1 + mid] <= t) lo = mid;
      else hi = mid - 1;
    }
    const int gid = lo;
    const int local = t - ws[1 + gid];
    const int tiles_m_g = (ws[2 + gid] - ws[1 + gid]) / tiles_n;
    const int bm = local % tiles_m_g;
    const int bn = local / tiles_m_g;

    const int64_t row_start = (gid > 0) ? cumsum[gid - 1] : 0;
    const int64_t m_size = cumsum[gid] - row_start;
    const bf16_t* Ag = A + row_start * K;
    const bf16_t* Bg = B + (int64_t)gid * N * K;
    bf16_t* Cg = C + row_start * N;

    KStage64 sa, sb;
    sa.init(Ag, K, [&](int r) -> int64_t {
      int64_t gm = (int64_t)bm * BM8 + r;
      return gm % m_size;
    }, tid);
    sb.init(Bg, K, [&](int r) -> int64_t {
      int64_t gn = (int64_t)bn * BN8 + r;
      return gn % N;
    }, tid);

    f32x16gg acc[4][2];
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j) acc[i][j] = f32x16gg{};

    sa.stage(ta(0), 0);
    sb.stage(tb(0), 0);
    __syncthreads();
    int cur = 0;
    for (int kt = 0; kt < nk; ++kt) {
      if (kt + 1 < nk) {
        sa.stage(ta(cur ^ 1), (int64_t)(kt + 1) * BK64);
        sb.stage(tb(cur ^ 1), (int64_t)(kt + 1) * BK64);
      }
#pragma unroll
      for (int ks = 0; ks < 4; ++ks) {
        bf16frag af[2], bfr[2];
#pragma unroll
        for (int j = 0; j < 2; ++j)
          bfr[j] = frag_read64_32(tb(cur), wc * 64 + j * 32, ks, lane);
#pragma unroll
        for (int i = 0; i < 2; ++i)
          af[i] = frag_read64_32(ta(cur), wr * 128 + i * 32, ks, lane);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int j = 0; j < 2; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af[i], bfr[j], acc[i][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int i = 0; i < 2; ++i)
          af[i] = frag_read64_32(ta(cur), wr * 128 + 64 + i * 32, ks, lane);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 2; ++i)
#pragma unroll
          for (int j = 0; j < 2; ++j)
            acc[i + 2][j] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(
                af[i], bfr[j], acc[i + 2][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();
      cur ^= 1;
    }

    const int col32 = lane & 31;
    const int half32 = lane >> 5;
#pragma unroll
    for (int i = 0; i < 4; ++i)
#pragma unroll
      for (int j = 0; j < 2; ++j)
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int mrow = (r & 3) + 8 * (r >> 2) + 4 * half32;
          int64_t m = (int64_t)bm * BM8 + wr * 128 + i * 32 + mrow;
          int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 32 + col32;
          if (m < m_size && n < N) Cg[m * N + n] = f2bf(acc[i][j][r]);
        }
  }
}


// nk8's 4-deep KSUB=32 counted-vmcnt ring under the same XCD-clustered
// persistent schedule (trans_b path only — its transposed-write staging for
// !trans_b has a 8-16-way LDS write conflict, and dgrad routes through the
// weight transpose + trans_b anyway). With the schedule removing the HBM
// wall, the deeper pipeline's shorter barrier drains can show.
__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_nk8s(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ cumsum, int G,
    int64_t N, int64_t K, int tiles_n, const int* __restrict__ ws) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto slotA = [&](int s) { return sm + (s & 3) * 8192; };          // 4 x 16 KiB
  auto slotB = [&](int s) { return sm + 32768 + (s & 3) * 8192; };  // 4 x 16 KiB

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  const int total_tiles = ws[0];
  const int X = blockIdx.x & 7;
  const int slot_id = blockIdx.x >> 3;
  const int L = (total_tiles + 7) / 8;
  const int t_end = min((X + 1) * L, total_tiles);
  const int nsub = (int)(K / KSUB);

  for (int t = X * L + slot_id; t < t_end; t += 32) {
    int lo = 0, hi = G - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (ws[1 + mid] <= t) lo = mid;
      else hi = mid - 1;
    }
    const int gid = lo;
    const int local = t - ws[1 + gid];
    const int tiles_m_g = (ws[2 + gid] - ws[1 + gid]) / tiles_n;
    const int bm = local % tiles_m_g;
    const int bn = local / tiles_m_g;

    const int64_t row_start = (gid > 0) ? cumsum[gid - 1] : 0;
    const int64_t m_size = cumsum[gid] - row_start;
    const bf16_t* Ag = A + row_start * K;
    const bf16_t* Bg = B + (int64_t)gid * N * K;
    bf16_t* Cg = C + row_start * N;

    KStage8 sa, sb;
    sa.init(Ag, K, [&](int r) -> int64_t {
      int64_t gm = (int64_t)bm * BM8 + r;
      return gm % m_size;
    }, tid);
    sb.init(Bg, K, [&](int r) -> int64_t {
      int64_t gn = (int64_t)bn * BN8 + r;
      return gn % N;
    }, tid);

    auto stage = [&](int s, int64_t k0) {
      sa.stage(slotA(s), k0);
      sb.stage(slotB(s), k0);
    };

    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    stage(0, 0);
    stage(1, KSUB);
    if (nsub > 2) stage(2, 2 * KSUB);
    for (int s = 0; s < nsub; ++s) {
      if (s + 3 < nsub) stage(s + 3, (int64_t)(s + 3) * KSUB);
      {
        const int rem = nsub - s;
        if (rem >= 4) VMCNT(12);
        else if (rem == 3) VMCNT(8);
        else if (rem == 2) VMCNT(4);
        else VMCNT(0);
      }
      raw_barrier();
      {
        const bf16_t* TA = slotA(s);
        const bf16_t* TB = slotB(s);
        bf16frag af[4], bfr[4];
#pragma unroll
        for (int i = 0; i < 4; ++i) af[i] = frag_read8(TA, wr * 128 + i * 16, lane);
#pragma unroll
        for (int j = 0; j < 4; ++j) bfr[j] = frag_read8(TB, wc * 64 + j * 16, lane);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
        raw_barrier();
#pragma unroll
        for (int i = 0; i < 4; ++i) af[i] = frag_read8(TA, wr * 128 + 64 + i * 16, lane);
#pragma unroll
        for (int j = 0; j < 4; ++j) bfr[j] = frag_read8(TB, wc * 64 + j * 16, lane);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i + 4][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
      raw_barrier();
    }

    const int col_in = lane & 15;
    const int row_base_in = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                      row_base_in + rr;
          int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
          if (m < m_size && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
        }
    // drain this tile's in-flight glds before the ring is re-staged for the
    // next tile (slots would otherwise be overwritten while still pending)
    VMCNT(0);
    raw_barrier();
  }
}


// ============================================================================
// nkp: the guide's 256-square 8-phase pipeline (cdna_hip_programming.md §5
// "The 256² 8-phase template"), reconstructed for the GROUPED case and run
// under the XCD-clustered persistent schedule.
//
// Reading of the template made self-consistent here (the example file is
// not shipped): a K-tile (BK=64) is staged as FOUR K-SPLIT half-tiles
// (A-k0, B-k0, A-k1, B-k1; each [256 rows][32 k] = 16 KiB, the proven
// conflict-free KStage8/swz32 layout), double-buffered = 8 ring slots =
// 128 KiB. Four phases per K-tile, each = one k-chunk x one m-half of the
// wave's 128x64 panel (16 MFMA); B-frags of a k-chunk are read once (ph0/
// ph2) and reused by the mh=1 phase. Each phase: [ds_reads; stage ONE
// half-tile of tile kt+1; (vmcnt clearing at ph1/ph3); raw barrier;
// lgkmcnt(0) implicit via frag use; setprio around the MFMA; raw barrier].
// Clearing schedule (per-wave glds = 2 per half): ph1 clears k1(kt)
// [vmcnt(4): outstanding = k1(kt) + k0(kt+1) = 8 loads], ph3 clears
// k0(kt+1) [vmcnt(4)]; last tile uses vmcnt(0) at ph1. ds_reads lead each
// phase: they target slots cleared >= 1 phase earlier, so they overlap the
// barrier wait.
// ============================================================================
__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_nkp(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ cumsum, int G,
    int64_t N, int64_t K, int tiles_n, const int* __restrict__ ws) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  // slot(t, h): h = 0 A-k0, 1 B-k0, 2 A-k1, 3 B-k1
  auto slot = [&](int t, int h) { return sm + (((t & 1) << 2) + h) * 8192; };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  const int total_tiles = ws[0];
  const int X = blockIdx.x & 7;
  const int slot_id = blockIdx.x >> 3;
  const int L = (total_tiles + 7) / 8;
  const int t_end = min((X + 1) * L, total_tiles);
  const int ntk = (int)(K / 64);

  for (int t = X * L + slot_id; t < t_end; t += 32) {
    int lo = 0, hi = G - 1;
    while (lo < hi) {
      int mid = (lo + hi + 1) >> 1;
      if (ws[1 + mid] <= t) lo = mid;
      else hi = mid - 1;
    }
    const int gid = lo;
    const int local = t - ws[1 + gid];
    const int tiles_m_g = (ws[2 + gid] - ws[1 + gid]) / tiles_n;
    const int bm = local % tiles_m_g;
    const int bn = local / tiles_m_g;

    const int64_t row_start = (gid > 0) ? cumsum[gid - 1] : 0;
    const int64_t m_size = cumsum[gid] - row_start;
    const bf16_t* Ag = A + row_start * K;
    const bf16_t* Bg = B + (int64_t)gid * N * K;
    bf16_t* Cg = C + row_start * N;

    KStage8 sa, sb;
    sa.init(Ag, K, [&](int r) -> int64_t {
      int64_t gm = (int64_t)bm * BM8 + r;
      return gm % m_size;
    }, tid);
    sb.init(Bg, K, [&](int r) -> int64_t {
      int64_t gn = (int64_t)bn * BN8 + r;
      return gn % N;
    }, tid);

    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    // prologue: tile 0's four halves, then clear its k0 pair
    sa.stage(slot(0, 0), 0);
    sb.stage(slot(0, 1), 0);
    sa.stage(slot(0, 2), KSUB);
    sb.stage(slot(0, 3), KSUB);
    VMCNT(4);
    raw_barrier();

    for (int kt = 0; kt < ntk; ++kt) {
      const bool more = kt + 1 < ntk;
      const int64_t nk0 = (int64_t)(kt + 1) * 64;
      bf16frag af[4], bfr[4];

      // ---- ph0: (k0, mh0); stage A-k0(kt+1)
#pragma unroll
      for (int j = 0; j < 4; ++j) bfr[j] = frag_read8(slot(kt, 1), wc * 64 + j * 16, lane);
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag_read8(slot(kt, 0), wr * 128 + i * 16, lane);
      if (more) sa.stage(slot(kt + 1, 0), nk0);
      raw_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      raw_barrier();

      // ---- ph1: (k0, mh1); stage B-k0(kt+1); clear k1(kt)
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag_read8(slot(kt, 0), wr * 128 + 64 + i * 16, lane);
      if (more) {
        sb.stage(slot(kt + 1, 1), nk0);
        VMCNT(4);
      } else {
        VMCNT(0);
      }
      raw_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i + 4][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      raw_barrier();

      // ---- ph2: (k1, mh0); stage A-k1(kt+1)
#pragma unroll
      for (int j = 0; j < 4; ++j) bfr[j] = frag_read8(slot(kt, 3), wc * 64 + j * 16, lane);
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag_read8(slot(kt, 2), wr * 128 + i * 16, lane);
      if (more) sa.stage(slot(kt + 1, 2), nk0 + KSUB);
      raw_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      raw_barrier();

      // ---- ph3: (k1, mh1); stage B-k1(kt+1); clear k0(kt+1)
#pragma unroll
      for (int i = 0; i < 4; ++i) af[i] = frag_read8(slot(kt, 2), wr * 128 + 64 + i * 16, lane);
      if (more) {
        sb.stage(slot(kt + 1, 3), nk0 + KSUB);
        VMCNT(4);
      }
      raw_barrier();
      __builtin_amdgcn_s_setprio(1);
#pragma unroll
      for (int i = 0; i < 4; ++i)
#pragma unroll
        for (int j = 0; j < 4; ++j)
          acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i + 4][j], 0, 0, 0);
      __builtin_amdgcn_s_setprio(0);
      raw_barrier();
    }

    const int col_in = lane & 15;
    const int row_base_in = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                      row_base_in + rr;
          int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
          if (m < m_size && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
        }
    VMCNT(0);
    raw_barrier();
  }
}

}  // namespace

// lazily-allocated device workspace for the tile schedule (ws[0] = total,
// ws[1..G+1] = per-group tile prefix). Single compute stream per process;
// calls on one stream serialize, so one buffer suffices.
static int* vh_gg_sched_ws = nullptr;
static constexpr int VH_GG_SCHED_MAX_G = 4096;

extern "C" int vh_group_gemm_nk256s_bf16(const uint16_t* A, const uint16_t* B,
                                         uint16_t* C, const int64_t* cumsum,
                                         int G, int64_t N, int64_t K,
                                         int64_t total_rows, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(K % BK64 == 0, "K %% 64 != 0");
  VH_CHECK(N % 16 == 0, "N %% 16 != 0");
  VH_CHECK(G <= VH_GG_SCHED_MAX_G, "G > %d", VH_GG_SCHED_MAX_G);
  if (vh_gg_sched_ws == nullptr) {
    VH_HIP(hipMalloc(&vh_gg_sched_ws, (VH_GG_SCHED_MAX_G + 2) * sizeof(int)));
  }
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  hipLaunchKernelGGL(k_gg_sched, dim3(1), dim3(64), 0, s, cumsum, G, tiles_n,
                     vh_gg_sched_ws);
  VH_HIP(hipGetLastError());
  hipLaunchKernelGGL(k_group_gemm_nk256s, dim3(256), dim3(THREADS8), 131072, s,
                     reinterpret_cast<const bf16_t*>(A),
                     reinterpret_cast<const bf16_t*>(B),
                     reinterpret_cast<bf16_t*>(C), cumsum, G, N, K, tiles_n,
                     vh_gg_sched_ws);
  VH_HIP(hipGetLastError());
  return 0;
}

/* probe: nk256s with the 32x32x16 MFMA inner loop. */
extern "C" int vh_group_gemm_nk256s32_bf16(const uint16_t* A, const uint16_t* B,
                                           uint16_t* C, const int64_t* cumsum,
                                           int G, int64_t N, int64_t K,
                                           int64_t total_rows, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(K % BK64 == 0, "K %% 64 != 0");
  VH_CHECK(N % 16 == 0, "N %% 16 != 0");
  VH_CHECK(G <= VH_GG_SCHED_MAX_G, "G > %d", VH_GG_SCHED_MAX_G);
  if (vh_gg_sched_ws == nullptr) {
    VH_HIP(hipMalloc(&vh_gg_sched_ws, (VH_GG_SCHED_MAX_G + 2) * sizeof(int)));
  }
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  hipLaunchKernelGGL(k_gg_sched, dim3(1), dim3(64), 0, s, cumsum, G, tiles_n,
                     vh_gg_sched_ws);
  VH_HIP(hipGetLastError());
  hipLaunchKernelGGL(k_group_gemm_nk256s32, dim3(256), dim3(THREADS8), 131072,
                     s, reinterpret_cast<const bf16_t*>(A),
                     reinterpret_cast<const bf16_t*>(B),
                     reinterpret_cast<bf16_t*>(C), cumsum, G, N, K, tiles_n,
                     vh_gg_sched_ws);
  VH_HIP(hipGetLastError());
  return 0;
}

/* probe: the 8-phase K-split pipeline + XCD schedule (trans_b only). */
extern "C" int vh_group_gemm_nkp_bf16(const uint16_t* A, const uint16_t* B,
                                      uint16_t* C, const int64_t* cumsum,
                                      int G, int64_t N, int64_t K,
                                      int64_t total_rows, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(K % 64 == 0, "K %% 64 != 0");
  VH_CHECK(N % 16 == 0, "N %% 16 != 0");
  VH_CHECK(G <= VH_GG_SCHED_MAX_G, "G > max");
  if (vh_gg_sched_ws == nullptr) {
    VH_HIP(hipMalloc(&vh_gg_sched_ws, (VH_GG_SCHED_MAX_G + 2) * sizeof(int)));
  }
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  hipLaunchKernelGGL(k_gg_sched, dim3(1), dim3(64), 0, s, cumsum, G, tiles_n,
                     vh_gg_sched_ws);
  VH_HIP(hipGetLastError());
  hipLaunchKernelGGL(k_group_gemm_nkp, dim3(256), dim3(THREADS8), 131072, s,
                     reinterpret_cast<const bf16_t*>(A),
                     reinterpret_cast<const bf16_t*>(B),
                     reinterpret_cast<bf16_t*>(C), cumsum, G, N, K, tiles_n,
                     vh_gg_sched_ws);
  VH_HIP(hipGetLastError());
  return 0;
}

/* probe: nk8's counted-vmcnt ring + the XCD schedule (trans_b only). */
extern "C" int vh_group_gemm_nk8s_bf16(const uint16_t* A, const uint16_t* B,
                                       uint16_t* C, const int64_t* cumsum,
                                       int G, int64_t N, int64_t K,
                                       int64_t total_rows, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(K % KSUB == 0 && K / KSUB >= 2, "K must be a multiple of 32, >= 64");
  VH_CHECK(N % 16 == 0, "N %% 16 != 0");
  VH_CHECK(G <= VH_GG_SCHED_MAX_G, "G > max");
  if (vh_gg_sched_ws == nullptr) {
    VH_HIP(hipMalloc(&vh_gg_sched_ws, (VH_GG_SCHED_MAX_G + 2) * sizeof(int)));
  }
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  hipLaunchKernelGGL(k_gg_sched, dim3(1), dim3(64), 0, s, cumsum, G, tiles_n,
                     vh_gg_sched_ws);
  VH_HIP(hipGetLastError());
  hipLaunchKernelGGL(k_group_gemm_nk8s, dim3(256), dim3(THREADS8), 131072, s,
                     reinterpret_cast<const bf16_t*>(A),
                     reinterpret_cast<const bf16_t*>(B),
                     reinterpret_cast<bf16_t*>(C), cumsum, G, N, K, tiles_n,
                     vh_gg_sched_ws);
  VH_HIP(hipGetLastError());
  return 0;
}

extern "C" int vh_group_gemm_nk256_bf16(const uint16_t* A, const uint16_t* B,
                                        uint16_t* C, const int64_t* cumsum,
                                        int G, int64_t N, int64_t K,
                                        int64_t total_rows, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(K % BK64 == 0, "K %% 64 != 0");
  VH_CHECK(N % 16 == 0, "N %% 16 != 0");
  int tiles_m = (int)((total_rows + BM8 - 1) / BM8);
  if (tiles_m < 1) tiles_m = 1;
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  dim3 grid(tiles_m * tiles_n, G);
  hipLaunchKernelGGL(k_group_gemm_nk256, grid, dim3(THREADS8), 131072, s,
                     reinterpret_cast<const bf16_t*>(A),
                     reinterpret_cast<const bf16_t*>(B),
                     reinterpret_cast<bf16_t*>(C), cumsum, G, N, K, tiles_m,
                     tiles_n);
  VH_HIP(hipGetLastError());
  return 0;
}

// ============================================================================
// Transpose-pad + wg256: the fast wgrad path.
//
// vh_transpose_pad_bf16 turns [rows, C] (rows grouped by expert via cumsum)
// into [C, padded_rows] where each group's row-range is zero-padded to a
// 64-multiple (padded_cumsum, host-computed). One memory-bound LDS-tiled
// transpose (~2x tensor bytes) buys both wgrad operands a K-contiguous
// layout, so the wgrad itself runs the nk256 glds structure (wg256) instead
// of the 2x-slower transposed-register staging.
// ============================================================================

namespace {

__global__ void k_transpose_pad(const bf16_t* __restrict__ src,
                                bf16_t* __restrict__ dst,
                                const int64_t* __restrict__ cumsum,
                                const int64_t* __restrict__ padded_cumsum,
                                int G, int64_t C, int64_t padded_total) {
  // block: 64 padded-rows x 64 cols tile; 256 threads. [64][72] image:
  // 144-B rows, 16-B aligned, odd 16-B slot count breaks conflicts.
  __shared__ __attribute__((aligned(16))) bf16_t t2[64][72];

  int64_t chunk = blockIdx.x;          // which 64-row padded chunk
  int64_t col0 = (int64_t)blockIdx.y * 64;
  int64_t p0 = chunk * 64;
  // find group (binary search — a linear walk was ~G dependent global loads
  // per block): padded chunks never straddle groups (64-multiples)
  int lo = 0, hi = G;  // first g with p0 < padded_cumsum[g]
  while (lo < hi) {
    int mid = (lo + hi) >> 1;
    if (p0 >= padded_cumsum[mid]) lo = mid + 1; else hi = mid;
  }
  int g = lo;
  if (g >= G) return;
  int64_t pstart = (g > 0) ? padded_cumsum[g - 1] : 0;
  int64_t sstart = (g > 0) ? cumsum[g - 1] : 0;
  int64_t kcount = cumsum[g] - sstart;
  int64_t local0 = p0 - pstart;        // offset within the group

  // read 64 src rows (coalesced: 8 threads x 16 B per row)
  const int tid = threadIdx.x;
  for (int r = tid / 8; r < 64; r += 32) {
    int64_t lr = local0 + r;
    const int cpos = (tid % 8) * 8;
    bf16x8 v = {};
    if (lr < kcount && col0 + cpos < C) {
      v = *reinterpret_cast<const bf16x8*>(src + (sstart + lr) * C + col0 + cpos);
    }
    *reinterpret_cast<bf16x8*>(&t2[r][cpos]) = v;
  }
  __syncthreads();
  // write transposed: dst[col][p0 + r], 16 B per thread along padded rows
  for (int c = tid / 8; c < 64; c += 32) {
    int rpos = (tid % 8) * 8;
    if (col0 + c < C) {
      bf16x8 v;
#pragma unroll
      for (int j = 0; j < 8; ++j) v.v[j] = t2[rpos + j][c];
      *reinterpret_cast<bf16x8*>(dst + (col0 + c) * padded_total + p0 + rpos) = v;
    }
  }
}

// wg256: C[g][M][N] = A'[:, pg] @ B'[:, pg]^T where A' [M, PR], B' [N, PR]
// are the transpose-padded operands; group g owns padded-row range pg.
// Same staging/compute structure as k_group_gemm_nk256.
__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_wg256(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ padded_cumsum, int G,
    int64_t M, int64_t N, int64_t PR, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto ta = [&](int buf) { return sm + buf * 32768; };
  auto tb = [&](int buf) { return sm + 16384 + buf * 32768; };

  const int gid = blockIdx.y;
  const int64_t p_start = (gid > 0) ? padded_cumsum[gid - 1] : 0;
  const int64_t klen = padded_cumsum[gid] - p_start;  // 64-multiple
  const int bm = blockIdx.x / tiles_n;
  const int bn = blockIdx.x % tiles_n;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  bf16_t* Cg = C + (int64_t)gid * M * N;

  f32x4 acc[8][4];
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

  if (klen > 0) {
    const bf16_t* Ag = A + p_start;
    const bf16_t* Bg = B + p_start;
    KStage64 sa, sb;
    sa.init(Ag, PR, [&](int r) -> int64_t {
      int64_t gm = (int64_t)bm * BM8 + r;
      return gm % M;
    }, tid);
    sb.init(Bg, PR, [&](int r) -> int64_t {
      int64_t gn = (int64_t)bn * BN8 + r;
      return gn % N;
    }, tid);

    const int nk = (int)(klen / BK64);
    sa.stage(ta(0), 0);
    sb.stage(tb(0), 0);
    __syncthreads();
    int cur = 0;
    for (int t = 0; t < nk; ++t) {
      if (t + 1 < nk) {
        sa.stage(ta(cur ^ 1), (int64_t)(t + 1) * BK64);
        sb.stage(tb(cur ^ 1), (int64_t)(t + 1) * BK64);
      }
#pragma unroll
      for (int ks = 0; ks < 2; ++ks) {
        bf16frag af[4], bfr[4];
#pragma unroll
        for (int j = 0; j < 4; ++j) bfr[j] = frag_read64(tb(cur), wc * 64 + j * 16, ks, lane);
#pragma unroll
        for (int i = 0; i < 4; ++i) af[i] = frag_read64(ta(cur), wr * 128 + i * 16, ks, lane);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
#pragma unroll
        for (int i = 0; i < 4; ++i) af[i] = frag_read64(ta(cur), wr * 128 + 64 + i * 16, ks, lane);
        __builtin_amdgcn_s_setprio(1);
#pragma unroll
        for (int i = 0; i < 4; ++i)
#pragma unroll
          for (int j = 0; j < 4; ++j)
            acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i + 4][j], 0, 0, 0);
        __builtin_amdgcn_s_setprio(0);
      }
      __syncthreads();
      cur ^= 1;
    }
  }

  const int col_in = lane & 15;
  const int row_base_in = (lane >> 4) * 4;
#pragma unroll
  for (int i = 0; i < 8; ++i)
#pragma unroll
    for (int j = 0; j < 4; ++j)
#pragma unroll
      for (int rr = 0; rr < 4; ++rr) {
        int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                    row_base_in + rr;
        int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
        if (m < M && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
      }
}


// wg256 under the XCD-clustered persistent schedule (see nk256s): every
// group owns the same tiles_m x tiles_n output tiles (weight-shaped C), so
// the schedule needs no device prefix — block b (XCD b%8) walks the
// contiguous tile range [X*L, (X+1)*L) with stride 32, bm-inner, keeping a
// group's B' k-slab hot in its XCD's L2.
__global__ __launch_bounds__(THREADS8, 2) void k_group_gemm_wg256s(
    const bf16_t* __restrict__ A, const bf16_t* __restrict__ B,
    bf16_t* __restrict__ C, const int64_t* __restrict__ padded_cumsum, int G,
    int64_t M, int64_t N, int64_t PR, int tiles_m, int tiles_n) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* sm = reinterpret_cast<bf16_t*>(smem);
  auto ta = [&](int buf) { return sm + buf * 32768; };
  auto tb = [&](int buf) { return sm + 16384 + buf * 32768; };

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int wr = wave >> 2, wc = wave & 3;

  const int tpg = tiles_m * tiles_n;
  const int total = G * tpg;
  const int X = blockIdx.x & 7;
  const int slot = blockIdx.x >> 3;
  const int L = (total + 7) / 8;
  const int t_end = min((X + 1) * L, total);

  for (int t = X * L + slot; t < t_end; t += 32) {
    const int gid = t / tpg;
    const int local = t - gid * tpg;
    const int bm = local % tiles_m;   // bm-inner: consecutive t share bn
    const int bn = local / tiles_m;

    const int64_t p_start = (gid > 0) ? padded_cumsum[gid - 1] : 0;
    const int64_t klen = padded_cumsum[gid] - p_start;
    bf16_t* Cg = C + (int64_t)gid * M * N;

    f32x4 acc[8][4];
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j) acc[i][j] = {0.f, 0.f, 0.f, 0.f};

    if (klen > 0) {
      const bf16_t* Ag = A + p_start;
      const bf16_t* Bg = B + p_start;
      KStage64 sa, sb;
      sa.init(Ag, PR, [&](int r) -> int64_t {
        int64_t gm = (int64_t)bm * BM8 + r;
        return gm % M;
      }, tid);
      sb.init(Bg, PR, [&](int r) -> int64_t {
        int64_t gn = (int64_t)bn * BN8 + r;
        return gn % N;
      }, tid);

      const int nk = (int)(klen / BK64);
      sa.stage(ta(0), 0);
      sb.stage(tb(0), 0);
      __syncthreads();
      int cur = 0;
      for (int kt = 0; kt < nk; ++kt) {
        if (kt + 1 < nk) {
          sa.stage(ta(cur ^ 1), (int64_t)(kt + 1) * BK64);
          sb.stage(tb(cur ^ 1), (int64_t)(kt + 1) * BK64);
        }
#pragma unroll
        for (int ks = 0; ks < 2; ++ks) {
          bf16frag af[4], bfr[4];
#pragma unroll
          for (int j = 0; j < 4; ++j) bfr[j] = frag_read64(tb(cur), wc * 64 + j * 16, ks, lane);
#pragma unroll
          for (int i = 0; i < 4; ++i) af[i] = frag_read64(ta(cur), wr * 128 + i * 16, ks, lane);
          __builtin_amdgcn_s_setprio(1);
#pragma unroll
          for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
              acc[i][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i][j], 0, 0, 0);
          __builtin_amdgcn_s_setprio(0);
#pragma unroll
          for (int i = 0; i < 4; ++i) af[i] = frag_read64(ta(cur), wr * 128 + 64 + i * 16, ks, lane);
          __builtin_amdgcn_s_setprio(1);
#pragma unroll
          for (int i = 0; i < 4; ++i)
#pragma unroll
            for (int j = 0; j < 4; ++j)
              acc[i + 4][j] = __builtin_amdgcn_mfma_f32_16x16x32_bf16(af[i], bfr[j], acc[i + 4][j], 0, 0, 0);
          __builtin_amdgcn_s_setprio(0);
        }
        __syncthreads();
        cur ^= 1;
      }
    } else {
      // keep barrier parity with peer waves (none needed: no LDS touched)
    }

    const int col_in = lane & 15;
    const int row_base_in = (lane >> 4) * 4;
#pragma unroll
    for (int i = 0; i < 8; ++i)
#pragma unroll
      for (int j = 0; j < 4; ++j)
#pragma unroll
        for (int rr = 0; rr < 4; ++rr) {
          int64_t m = (int64_t)bm * BM8 + wr * 128 + (i & 3) * 16 + (i >> 2) * 64 +
                      row_base_in + rr;
          int64_t n = (int64_t)bn * BN8 + wc * 64 + j * 16 + col_in;
          if (m < M && n < N) Cg[m * N + n] = f2bf(acc[i][j][rr]);
        }
  }
}

}  // namespace

extern "C" int vh_transpose_pad_bf16(const uint16_t* src, uint16_t* dst,
                                     const int64_t* cumsum,
                                     const int64_t* padded_cumsum, int G,
                                     int64_t C, int64_t padded_total,
                                     void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(C % 8 == 0, "C %% 8 != 0");
  VH_CHECK(padded_total % 64 == 0, "padded_total %% 64 != 0");
  dim3 grid((uint32_t)(padded_total / 64), (uint32_t)((C + 63) / 64));
  hipLaunchKernelGGL(k_transpose_pad, grid, dim3(256), 0, s,
                     reinterpret_cast<const bf16_t*>(src),
                     reinterpret_cast<bf16_t*>(dst), cumsum, padded_cumsum, G,
                     C, padded_total);
  VH_HIP(hipGetLastError());
  return 0;
}

extern "C" int vh_group_gemm_wg256_bf16(const uint16_t* A, const uint16_t* B,
                                        uint16_t* C,
                                        const int64_t* padded_cumsum, int G,
                                        int64_t M, int64_t N, int64_t PR,
                                        void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(PR % 64 == 0, "PR %% 64 != 0");
  int tiles_m = (int)((M + BM8 - 1) / BM8);
  int tiles_n = (int)((N + BN8 - 1) / BN8);
  hipLaunchKernelGGL(k_group_gemm_wg256s, dim3(256), dim3(THREADS8), 131072, s,
                     reinterpret_cast<const bf16_t*>(A),
                     reinterpret_cast<const bf16_t*>(B),
                     reinterpret_cast<bf16_t*>(C), padded_cumsum, G, M, N, PR,
                     tiles_m, tiles_n);
  VH_HIP(hipGetLastError());
  return 0;
}
