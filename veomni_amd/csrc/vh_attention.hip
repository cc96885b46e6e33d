// Flash attention forward for gfx950 (bf16, causal, GQA, D=128).
//
// Replaces the external flash_attn wheel the reference delegates to
// (ref ops/kernels/attention/flash.py:153-301) for the packed causal path.
//
// Structure (guide cdna_hip_programming.md Appendix B, swapped-QK^T design):
//   grid (S/256, B*Hq); block = 8 waves (512 thr), wave w owns 32 q rows.
//   Q staged per-wave in LDS once; K tiles ([64][128]) and TRANSPOSED V
//   tiles ([128][64]) double-buffered: K/Q via glds (lane-swizzled source),
//   V via T14-split register staging (loads before the MFMA phase, LDS
//   writes after). Per 32-kv subtile:
//     S^T = mfma_32x32x16(A=K, B=Q): lane owns col q = lane&31, 16 f32
//       scores (the kv rows split across the lane pair);
//     online softmax per q-col (running m, l lane scalars; cross-half
//       reduce = one shfl_xor(32)); defer-max (T13, THR=8) skips the
//       O-rescale when the wave's max is stable;
//     P packed to bf16 quads, partner-quad exchange -> A-fragment;
//     O += mfma(A=P^T, B=V^T).
//   Epilogue: O / l via lane broadcasts; LSE = m + log(l) saved for bwd.
//
// LDS: 2x(K 16 + V^T 16) = 64 KiB; Q lives in registers (tile-invariant).

#include "vh_common.h"

namespace {

constexpr int QB = 256;   // q rows per block
constexpr int WQ = 32;    // q rows per wave
constexpr int KB = 64;    // kv rows per tile
constexpr int DH = 128;   // head dim
constexpr float DEFER_THR = 8.0f;  // T13 defer-max threshold

using bf16frag = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;

__device__ __forceinline__ void glds16a(const bf16_t* g, bf16_t* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)l, 16, 0, 0);
}

// 256-B rows (K, Q tiles): conflict-free slot map for b128 reads
__device__ __forceinline__ int kswz(int row, int colb) {
  return colb ^ ((row & 15) << 4);
}
// 128-B rows (V^T tile): the gemm swizzle
__device__ __forceinline__ int vswz(int row, int colb) {
  return colb ^ ((((row >> 1) ^ (row >> 3)) & 7) << 4);
}

// 32-lane-distance exchange via v_permlane32_swap (1 VALU op) instead of
// __shfl_xor's ds_bpermute (~50-cycle LDS round trip) — guide T12.
__device__ __forceinline__ uint32_t swap32_u(uint32_t v, int half) {
  auto r = __builtin_amdgcn_permlane32_swap(v, v, false, false);
  return half ? (uint32_t)r[0] : (uint32_t)r[1];
}

__device__ __forceinline__ float xor32h(float v, int half) {
  return __builtin_bit_cast(float, swap32_u(__builtin_bit_cast(uint32_t, v), half));
}

template <int MODE>
__global__ __launch_bounds__(512, 2) void k_attn_fwd(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, bf16_t* __restrict__ O,
    float* __restrict__ LSE, int B, int Hq, int Hkv, int64_t S, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  auto kt = [&](int buf) {                                   // 2 x 16 KiB
    return reinterpret_cast<bf16_t*>(smem + buf * 16384);
  };
  auto vt = [&](int buf) {                                   // 2 x 16 KiB
    return reinterpret_cast<bf16_t*>(smem + 32768 + buf * 16384);
  };

  const int qb = blockIdx.x;
  const int bh = blockIdx.y;           // b * Hq + hq
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int half = lane >> 5;
  const int col = lane & 31;           // q within the wave for S; d%32 for O

  const bf16_t* Qb = Q + (((int64_t)b * Hq + hq) * S) * DH;
  const bf16_t* Kb = K + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* Vb = V + (((int64_t)b * Hkv + hkv) * S) * DH;
  bf16_t* Ob = O + (((int64_t)b * Hq + hq) * S) * DH;
  float* Lb = LSE + ((int64_t)b * Hq + hq) * S;

  const int64_t q_global = (int64_t)qb * QB + wave * WQ + col;
  const float scale2 = scale * 1.4426950408889634f;  // log2(e)

  // ---- Q fragments straight into registers (tile-invariant: chunk c =
  // Q[q = col][c*16 + half*8 .. +8))
  bf16frag qreg[8];
#pragma unroll
  for (int c = 0; c < 8; ++c) {
    qreg[c] = *reinterpret_cast<const bf16frag*>(
        Qb + q_global * DH + c * 16 + half * 8);
  }

  // ---- persistent staging addresses
  const bf16_t* ksrc[2];
  int klds[2];
#pragma unroll
  for (int i = 0; i < 2; ++i) {
    int base = i * 8192 + wave * 1024;
    int o = base + lane * 16;
    int row = o >> 8;
    int colb = o & 255;
    ksrc[i] = Kb + (int64_t)row * DH + (kswz(row, colb) >> 1);
    klds[i] = base >> 1;
  }
  const int v_kv = tid & 63;
  const int v_d0 = (tid >> 6) * 8;   // 2 units: rows v_d0 and v_d0+64
  const bf16_t* vsrc = Vb + (int64_t)v_kv * DH + v_d0;

  f32x16 oacc[4];
#pragma unroll
  for (int d = 0; d < 4; ++d) oacc[d] = f32x16{};
  float m_run = -1e30f;
  float l_run = 0.f;

  const int t_max = (int)(((int64_t)qb * QB + QB - 1) / KB);  // inclusive

  // prologue: stage tile 0 synchronously
  {
#pragma unroll
    for (int i = 0; i < 2; ++i) glds16a(ksrc[i], kt(0) + klds[i]);
    bf16x8 v0 = *reinterpret_cast<const bf16x8*>(vsrc);
    bf16x8 v1 = *reinterpret_cast<const bf16x8*>(vsrc + 64);
#pragma unroll
    for (int j = 0; j < 8; ++j) {
      int r0 = v_d0 + j, r1 = v_d0 + 64 + j;
      vt(0)[(r0 * 128 + vswz(r0, v_kv * 2)) >> 1] = v0.v[j];
      vt(0)[(r1 * 128 + vswz(r1, v_kv * 2)) >> 1] = v1.v[j];
    }
  }
  __syncthreads();

  int cur = 0;
  for (int t = 0; t <= t_max; ++t) {
    // ---- T14 split: issue next tile's loads before this tile's MFMAs
    bf16x8 vn0, vn1;
    const bool more = t < t_max;
    if (more) {
#pragma unroll
      for (int i = 0; i < 2; ++i)
        glds16a(ksrc[i] + (int64_t)(t + 1) * KB * DH, kt(cur ^ 1) + klds[i]);
      vn0 = *reinterpret_cast<const bf16x8*>(vsrc + (int64_t)(t + 1) * KB * DH);
      vn1 = *reinterpret_cast<const bf16x8*>(vsrc + (int64_t)(t + 1) * KB * DH + 64);
    }

    const bool diag = ((int64_t)(t + 1) * KB) > ((int64_t)qb * QB + wave * WQ);
    // causal skip: every kv in this tile is beyond every q of this wave
    const bool live = (int64_t)t * KB <= (int64_t)qb * QB + wave * WQ + (WQ - 1);
    const bf16_t* ktc = kt(cur);
    const bf16_t* vtc = vt(cur);

#pragma unroll
    for (int sub = 0; sub < 2 && live; ++sub) {
      // ---- S^T = K·Q^T over 8 d-chunks
      f32x16 sacc = f32x16{};
      if (MODE != 3) {
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          int krow = sub * 32 + col;
          int colb = (c * 16 + half * 8) * 2;
          bf16frag kf = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(ktc) + krow * 256 + kswz(krow, colb));
          sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qreg[c], sacc, 0, 0, 0);
        }
      }

      if (MODE == 3) {
        // keep sacc live without QK cost accounting (probe only)
        asm volatile("" :: "v"(sacc[0]));
      }
      // ---- scale + causal mask in the exp2 domain (scale2 = scale*log2e
      // folded into the score multiply; v_exp_f32 IS 2^x, so exp2-domain
      // bookkeeping drops one multiply per element)
      float p[16];
      const int kv_lim = (int)(q_global - (int64_t)t * KB) - sub * 32 - 4 * half;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float sc = sacc[r] * scale2;
        if (diag && ((r & 3) + 8 * (r >> 2)) > kv_lim) sc = -INFINITY;
        p[r] = sc;
      }
      float m8[8], m4[4];
#pragma unroll
      for (int i = 0; i < 8; ++i) m8[i] = fmaxf(p[2 * i], p[2 * i + 1]);
#pragma unroll
      for (int i = 0; i < 4; ++i) m4[i] = fmaxf(m8[2 * i], m8[2 * i + 1]);
      float mt = fmaxf(fmaxf(m4[0], m4[1]), fmaxf(m4[2], m4[3]));
      mt = fmaxf(mt, xor32h(mt, half));

      if (MODE == 1) {
        // probe: skip ALL softmax VALU/shuffles; fabricate pa from sacc bits
        bf16frag pa1[2];
        uint4 u1{__builtin_bit_cast(uint32_t, sacc[0]), __builtin_bit_cast(uint32_t, sacc[1]),
                 __builtin_bit_cast(uint32_t, sacc[2]), __builtin_bit_cast(uint32_t, sacc[3])};
        pa1[0] = __builtin_bit_cast(bf16frag, u1);
        pa1[1] = pa1[0];
#pragma unroll
        for (int mch = 0; mch < 2; ++mch)
#pragma unroll
          for (int d = 0; d < 4; ++d) {
            int vrow = d * 32 + col;
            int colb2 = (sub * 32 + mch * 16 + half * 8) * 2;
            bf16frag vf = *reinterpret_cast<const bf16frag*>(
                reinterpret_cast<const char*>(vtc) + vrow * 128 + vswz(vrow, colb2));
            oacc[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa1[mch], vf, oacc[d], 0, 0, 0);
          }
        continue;
      }
      // ---- defer-max (T13): rescale only when some lane's max moved past
      // the threshold (wave-uniform decision via ballot)
      bool need = mt > m_run + DEFER_THR;
      if (__builtin_amdgcn_ballot_w64(need) != 0ull) {
        float m_new = fmaxf(m_run, mt);
        float alpha = __builtin_exp2f(m_run - m_new);
        m_run = m_new;
        l_run *= alpha;
#pragma unroll
        for (int r = 0; r < 16; ++r) {
          int qr = (r & 3) + 8 * (r >> 2) + 4 * half;
          float a_r = __shfl(alpha, qr, 32);
#pragma unroll
          for (int d = 0; d < 4; ++d) oacc[d][r] *= a_r;
        }
      }

      // ---- exponentiate + pack pairs + tree row sum
      float s8[8];
      uint32_t pk[8];
#pragma unroll
      for (int r = 0; r < 16; r += 2) {
        float e0 = __builtin_exp2f(p[r] - m_run);
        float e1 = __builtin_exp2f(p[r + 1] - m_run);
        s8[r >> 1] = e0 + e1;
        // native bf16 converts (v_cvt) — the integer-emulated RNE rounding
        // was ~6 VALU ops per element on the hot path
        uint16_t b0 = __builtin_bit_cast(uint16_t, (__bf16)e0);
        uint16_t b1 = __builtin_bit_cast(uint16_t, (__bf16)e1);
        pk[r >> 1] = (uint32_t)b0 | ((uint32_t)b1 << 16);
      }
      float s4a = (s8[0] + s8[1]) + (s8[2] + s8[3]);
      float s4b = (s8[4] + s8[5]) + (s8[6] + s8[7]);
      float psum = s4a + s4b;
      psum += xor32h(psum, half);
      l_run += psum;

      // ---- partner-quad exchange -> P^T A-fragments
      bf16frag pa[2];
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
        uint32_t s0 = half ? pk[4 * mch] : pk[4 * mch + 2];
        uint32_t s1 = half ? pk[4 * mch + 1] : pk[4 * mch + 3];
        uint32_t o0 = swap32_u(s0, half);
        uint32_t o1 = swap32_u(s1, half);
        uint32_t w0 = half ? o0 : pk[4 * mch];
        uint32_t w1 = half ? o1 : pk[4 * mch + 1];
        uint32_t w2 = half ? pk[4 * mch + 2] : o0;
        uint32_t w3 = half ? pk[4 * mch + 3] : o1;
        uint4 u{w0, w1, w2, w3};
        pa[mch] = __builtin_bit_cast(bf16frag, u);
      }

      // ---- O += P^T · V
      if (MODE != 2) {
#pragma unroll
        for (int mch = 0; mch < 2; ++mch) {
#pragma unroll
          for (int d = 0; d < 4; ++d) {
            int vrow = d * 32 + col;
            int colb = (sub * 32 + mch * 16 + half * 8) * 2;
            bf16frag vf = *reinterpret_cast<const bf16frag*>(
                reinterpret_cast<const char*>(vtc) + vrow * 128 + vswz(vrow, colb));
            oacc[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[mch], vf, oacc[d], 0, 0, 0);
          }
        }
      } else {
        asm volatile("" :: "v"(pa[0]), "v"(pa[1]));
      }
    }

    // ---- T14 write-late: flush the next V tile, then the tile barrier
    if (more) {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int r0 = v_d0 + j, r1 = v_d0 + 64 + j;
        vt(cur ^ 1)[(r0 * 128 + vswz(r0, v_kv * 2)) >> 1] = vn0.v[j];
        vt(cur ^ 1)[(r1 * 128 + vswz(r1, v_kv * 2)) >> 1] = vn1.v[j];
      }
    }
    __syncthreads();
    cur ^= 1;
  }

  // ---- epilogue: O / l; bf16 store; LSE
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int qr = (r & 3) + 8 * (r >> 2) + 4 * half;
    float l_r = __shfl(l_run, qr, 32);
    float inv = 1.0f / l_r;
    int64_t qg = (int64_t)qb * QB + wave * WQ + qr;
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      Ob[qg * DH + d * 32 + col] = f2bf(oacc[d][r] * inv);
    }
  }
  if (half == 0) {
    // back to the natural-log domain: LSE = ln2 * (m2 + log2(l))
    Lb[q_global] = 0.6931471805599453f * (m_run + __log2f(l_run));
  }
}

}  // namespace

extern "C" int vh_attn_fwd_bf16(const uint16_t* Q, const uint16_t* K,
                                const uint16_t* V, uint16_t* O, float* LSE,
                                int B, int Hq, int Hkv, int64_t S, float scale,
                                void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % QB == 0, "S %% 256 != 0 (pad the sequence)");
  VH_CHECK(Hq % Hkv == 0, "Hq %% Hkv != 0");
  dim3 grid((uint32_t)(S / QB), (uint32_t)(B * Hq));
  hipLaunchKernelGGL(k_attn_fwd<0>, grid, dim3(512), 65536, s,
                     reinterpret_cast<const bf16_t*>(Q),
                     reinterpret_cast<const bf16_t*>(K),
                     reinterpret_cast<const bf16_t*>(V),
                     reinterpret_cast<bf16_t*>(O), LSE, B, Hq, Hkv, S, scale);
  VH_HIP(hipGetLastError());
  return 0;
}

/* ablation probe: mode 1 = no softmax VALU, 2 = no PV, 3 = no QK (timing only) */
extern "C" int vh_attn_fwd_probe_bf16(const uint16_t* Q, const uint16_t* K,
                                      const uint16_t* V, uint16_t* O,
                                      float* LSE, int B, int Hq, int Hkv,
                                      int64_t S, float scale, int mode,
                                      void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  dim3 grid((uint32_t)(S / QB), (uint32_t)(B * Hq));
#define VH_AM(M_)                                                                hipLaunchKernelGGL(k_attn_fwd<M_>, grid, dim3(512), 65536, s,                                     reinterpret_cast<const bf16_t*>(Q),                                            reinterpret_cast<const bf16_t*>(K),                                            reinterpret_cast<const bf16_t*>(V),                                            reinterpret_cast<bf16_t*>(O), LSE, B, Hq, Hkv, S, scale)
  if (mode == 1) VH_AM(1);
  else if (mode == 2) VH_AM(2);
  else if (mode == 3) VH_AM(3);
  else VH_AM(0);
#undef VH_AM
  VH_HIP(hipGetLastError());
  return 0;
}
