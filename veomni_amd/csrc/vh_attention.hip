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

// DOC: packed-varlen (block-diagonal causal) masking via per-token document
// start indices (doc_start[t] = cu_seqlens[i] for t in document i; B == 1).
// Replaces the reference's flash-attn varlen cu_seqlens path
// (ops/kernels/attention/flash.py:61-91, kwargs from data_collator.py:50).
// TRF: natural-layout V image + tr16 hardware-transpose PV reads (same
// scheme as the dkv TR variant; image off(kv,d) = kv*256 + g*8 + (d&3)*2,
// g = (c&7)|(((kv&3)^(c>>3))&3)<<3, c = d>>2).
// TRF 2: V image = L16 latin square filled by glds direct-to-LDS (per-lane
// permuted sources), PV B-frags via tr16 — the dkv TR2G scheme on the
// forward's V tile (drops the vn register carry + all V ds_writes).
template <int MODE, bool DOC, int TRF = 0>
__global__ __launch_bounds__(512, 2) void k_attn_fwd(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, bf16_t* __restrict__ O,
    float* __restrict__ LSE, const int* __restrict__ doc_start, int B, int Hq,
    int Hkv, int64_t S, float scale) {
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

  // varlen: this lane's document start, and the wave/block minima (doc_start
  // is monotone non-decreasing, so the first row's value is the minimum)
  int ds_l = 0, ds_wave = 0, t0 = 0;
  if (DOC) {
    ds_l = doc_start[q_global];
    ds_wave = doc_start[(int64_t)qb * QB + wave * WQ];
    t0 = doc_start[(int64_t)qb * QB] / KB;
  }

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
  // TRF store assignment keeps every 16-lane phase on 16 distinct 16-B
  // slots of the permuted image (conflict-free)
  const int v_kv = (TRF == 1) ? (((tid >> 2) & 15) | (((tid >> 6) & 3) << 4))
                              : (tid & 63);
  const int v_d0 = (TRF == 1) ? ((tid & 3) * 8 + ((tid >> 8) & 1) * 32)
                              : ((tid >> 6) * 8);  // rows v_d0, v_d0+64
  const bf16_t* vsrc = Vb + (int64_t)v_kv * DH + v_d0;

  f32x16 oacc[4];
#pragma unroll
  for (int d = 0; d < 4; ++d) oacc[d] = f32x16{};
  float m_run = -1e30f;
  float l_run = 0.f;

  const int t_max = (int)(((int64_t)qb * QB + QB - 1) / KB);  // inclusive

  // prologue: stage tile t0 synchronously
  {
#pragma unroll
    for (int i = 0; i < 2; ++i)
      glds16a(ksrc[i] + (int64_t)t0 * KB * DH, kt(0) + klds[i]);
    if constexpr (TRF == 2) {
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        const int win = wave * 2 + u;
        const int kvr = win * 4 + (lane >> 4);
        const int pp0 = (lane & 15) ^ (kvr & 15);
        const int pg = ((pp0 & 3) << 2) | (pp0 >> 2);
        glds16a(Vb + ((int64_t)t0 * KB + kvr) * DH + pg * 8, vt(0) + win * 512);
      }
    } else {
    bf16x8 v0 = *reinterpret_cast<const bf16x8*>(vsrc + (int64_t)t0 * KB * DH);
    bf16x8 v1 = *reinterpret_cast<const bf16x8*>(vsrc + (int64_t)t0 * KB * DH + 64);
    if constexpr (TRF == 1) {
      int c0 = v_d0 >> 2, c1 = (v_d0 + 64) >> 2;
      int g0 = (c0 & 7) | ((((v_kv & 3) ^ (c0 >> 3)) & 3) << 3);
      int g1 = (c1 & 7) | ((((v_kv & 3) ^ (c1 >> 3)) & 3) << 3);
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(vt(0)) + v_kv * 256 + g0 * 8) = v0;
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(vt(0)) + v_kv * 256 + g1 * 8) = v1;
    } else {
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int r0 = v_d0 + j, r1 = v_d0 + 64 + j;
        vt(0)[(r0 * 128 + vswz(r0, v_kv * 2)) >> 1] = v0.v[j];
        vt(0)[(r1 * 128 + vswz(r1, v_kv * 2)) >> 1] = v1.v[j];
      }
    }
    }
  }
  __syncthreads();

  int cur = 0;
  for (int t = t0; t <= t_max; ++t) {
    // ---- T14 split: issue next tile's loads before this tile's MFMAs
    bf16x8 vn0, vn1;
    const bool more = t < t_max;
    if (more) {
#pragma unroll
      for (int i = 0; i < 2; ++i)
        glds16a(ksrc[i] + (int64_t)(t + 1) * KB * DH, kt(cur ^ 1) + klds[i]);
      if constexpr (TRF == 2) {
#pragma unroll
        for (int u = 0; u < 2; ++u) {
          const int win = wave * 2 + u;
          const int kvr = win * 4 + (lane >> 4);
          const int pp0 = (lane & 15) ^ (kvr & 15);
          const int pg = ((pp0 & 3) << 2) | (pp0 >> 2);
          glds16a(Vb + ((int64_t)(t + 1) * KB + kvr) * DH + pg * 8,
                  vt(cur ^ 1) + win * 512);
        }
      } else {
        vn0 = *reinterpret_cast<const bf16x8*>(vsrc + (int64_t)(t + 1) * KB * DH);
        vn1 = *reinterpret_cast<const bf16x8*>(vsrc + (int64_t)(t + 1) * KB * DH + 64);
      }
    }

    const bool diag = ((int64_t)(t + 1) * KB) > ((int64_t)qb * QB + wave * WQ);
    // causal skip: every kv in this tile is beyond every q of this wave
    // (varlen adds: or before every document of this wave)
    const bool live =
        ((int64_t)t * KB <= (int64_t)qb * QB + wave * WQ + (WQ - 1)) &&
        (!DOC || (int64_t)(t + 1) * KB > ds_wave);
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
      const int lo_lim = DOC ? ds_l - (int)((int64_t)t * KB) - sub * 32 - 4 * half
                             : -2147483647;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float sc = sacc[r] * scale2;
        int idx = (r & 3) + 8 * (r >> 2);
        if (diag && idx > kv_lim) sc = -INFINITY;
        if (DOC && idx < lo_lim) sc = -INFINITY;
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
            // probe path: non-TR image only (probe launcher instantiates
            // TRF=false)
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
            bf16frag vf;
            if constexpr (TRF == 2) {
              const int m_ = lane & 15;
              const int colhi_ = (lane >> 4) & 1;
              const int kvb0 = sub * 32 + mch * 16 + half * 8 + (m_ >> 2);
              const int c_r = d * 8 + colhi_ * 4 + (m_ & 3);
              const int perm_ = (((c_r >> 1) & 3) << 2) | (c_r >> 3);
              const int a0 =
                  kvb0 * 256 + ((kvb0 & 15) ^ perm_) * 16 + (c_r & 1) * 8;
              const int a1 = (kvb0 + 4) * 256 +
                             (((kvb0 + 4) & 15) ^ perm_) * 16 + (c_r & 1) * 8;
              auto* vb3 = (__attribute__((address_space(3))) char*)vtc;
              typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4t;
              typedef __attribute__((address_space(3))) bf16x4t as3b4;
              bf16x4t r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(vb3 + a0));
              bf16x4t r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(vb3 + a1));
              vf = __builtin_shufflevector(r0, r1, 0, 1, 2, 3, 4, 5, 6, 7);
            } else if constexpr (TRF == 1) {
              const int m_ = lane & 15;
              const int colhi_ = (lane >> 4) & 1;
              const int kvb0 = sub * 32 + mch * 16 + half * 8 + (m_ >> 2);
              const int c_r = d * 8 + colhi_ * 4 + (m_ & 3);
              const int g_r =
                  (c_r & 7) | ((((m_ >> 2) ^ (c_r >> 3)) & 3) << 3);
              auto* vb3 = (__attribute__((address_space(3))) char*)vtc;
              typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4t;
              typedef __attribute__((address_space(3))) bf16x4t as3b4;
              bf16x4t r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(vb3 + kvb0 * 256 + g_r * 8));
              bf16x4t r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(vb3 + kvb0 * 256 + g_r * 8 + 1024));
              vf = __builtin_shufflevector(r0, r1, 0, 1, 2, 3, 4, 5, 6, 7);
            } else {
              int vrow = d * 32 + col;
              int colb = (sub * 32 + mch * 16 + half * 8) * 2;
              vf = *reinterpret_cast<const bf16frag*>(
                  reinterpret_cast<const char*>(vtc) + vrow * 128 + vswz(vrow, colb));
            }
            oacc[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[mch], vf, oacc[d], 0, 0, 0);
          }
        }
      } else {
        asm volatile("" :: "v"(pa[0]), "v"(pa[1]));
      }
    }

    // ---- T14 write-late: flush the next V tile, then the tile barrier
    if (more && TRF != 2) {
      if constexpr (TRF == 1) {
        int c0 = v_d0 >> 2, c1 = (v_d0 + 64) >> 2;
        int g0 = (c0 & 7) | ((((v_kv & 3) ^ (c0 >> 3)) & 3) << 3);
        int g1 = (c1 & 7) | ((((v_kv & 3) ^ (c1 >> 3)) & 3) << 3);
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(vt(cur ^ 1)) + v_kv * 256 + g0 * 8) = vn0;
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(vt(cur ^ 1)) + v_kv * 256 + g1 * 8) = vn1;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int r0 = v_d0 + j, r1 = v_d0 + 64 + j;
          vt(cur ^ 1)[(r0 * 128 + vswz(r0, v_kv * 2)) >> 1] = vn0.v[j];
          vt(cur ^ 1)[(r1 * 128 + vswz(r1, v_kv * 2)) >> 1] = vn1.v[j];
        }
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
                                const int32_t* doc_start, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % QB == 0, "S %% 256 != 0 (pad the sequence)");
  VH_CHECK(Hq % Hkv == 0, "Hq %% Hkv != 0");
  VH_CHECK(doc_start == nullptr || B == 1, "varlen requires packed B == 1");
  dim3 grid((uint32_t)(S / QB), (uint32_t)(B * Hq));
  if (doc_start) {
    hipLaunchKernelGGL((k_attn_fwd<0, true, 2>), grid, dim3(512), 65536, s,
                       reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<bf16_t*>(O), LSE, doc_start, B, Hq,
                       Hkv, S, scale);
  } else {
    hipLaunchKernelGGL((k_attn_fwd<0, false, 2>), grid, dim3(512), 65536, s,
                       reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<bf16_t*>(O), LSE, nullptr, B, Hq, Hkv,
                       S, scale);
  }
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
#define VH_AM(M_)                                                                hipLaunchKernelGGL((k_attn_fwd<M_, false>), grid, dim3(512), 65536, s,                                     reinterpret_cast<const bf16_t*>(Q),                                            reinterpret_cast<const bf16_t*>(K),                                            reinterpret_cast<const bf16_t*>(V),                                            reinterpret_cast<bf16_t*>(O), LSE, nullptr, B, Hq, Hkv, S, scale)
  if (mode == 1) VH_AM(1);
  else if (mode == 2) VH_AM(2);
  else if (mode == 3) VH_AM(3);
  else if (mode == 21)  // TRF2: glds-filled L16 V image A/B
    hipLaunchKernelGGL((k_attn_fwd<0, false, 2>), grid, dim3(512), 65536,
                       s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<bf16_t*>(O), LSE, nullptr, B, Hq, Hkv,
                       S, scale);
  else if (mode == 20)  // TRF: tr16 V image A/B
    hipLaunchKernelGGL((k_attn_fwd<0, false, 1>), grid, dim3(512), 65536,
                       s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<bf16_t*>(O), LSE, nullptr, B, Hq, Hkv,
                       S, scale);
  else VH_AM(0);
#undef VH_AM
  VH_HIP(hipGetLastError());
  return 0;
}

// ============================================================================
// Flash attention backward, monolithic variant (bf16, causal, GQA, D = 128).
//
// NOTE: this kernel is PROBE INFRASTRUCTURE (vh_attn_bwd_probe_bf16, ablation
// modes 0-8 — the measurements in DESIGN.md came from it). The DISPATCHED
// backward is vh_attn_bwd2_bf16 below: two split kernels (k_attn_bwd_dkv /
// k_attn_bwd_dq) that keep all accumulators in registers with no cross-wave
// reduction; the monolith's LDS dQ exchange is what they replaced.
//
// Structure: grid (S/128, B*Hq); a block (4 waves, 256 thr) owns 128 kv rows
// (wave w the 32-row slice w) and loops 32-row q tiles from the causal
// diagonal. Both score orientations are recomputed so no cross-lane
// transpose is needed (the MFMA A-operand always comes from the forward's
// reg->chunk pack-exchange):
//   or1 (C = [kv regs][q lanes]): S1 = mfma(K, Q), dP1 = mfma(V, dO)
//       -> pack(dS1) = A[q][kv] feeds  dQ += mfma(dS1^T-pack, K^T-tile)
//   or2 (C = [q regs][kv lanes]): S2 = mfma(Q, K), dP2 = mfma(dO, V)
//       -> pack(P2), pack(dS2) = A[kv][q] feed dV/dK += mfma(pack, dO^T/Q^T)
// P = exp2(S*scale2 - lse2[q]); dS = P * (dP - delta[q]) * scale.
// dQ partials go through per-wave bf16 LDS quarters + fp32 global atomics;
// dK/dV are written per Q-head and the host sums GQA groups.
// delta = rowsum(dO*O), lse2 = LSE*log2e from vh_attn_bwd_pre_bf16.
// ============================================================================

namespace {

// swizzle for 64-B rows ([128][32] q-side transposed tiles): 4 16-B slots
// per row; bank row = 256 B = 4 tile rows -> slot = (row>>2)&3 (the vswz
// 8-slot map would overflow a 64-B row and alias across rows).
__device__ __forceinline__ int qswz(int row, int colb) {
  return colb ^ (((row >> 2) & 3) << 4);
}

__global__ void k_attn_bwd_pre(const bf16_t* __restrict__ dO,
                               const bf16_t* __restrict__ O,
                               const float* __restrict__ LSE,
                               float* __restrict__ delta,
                               float* __restrict__ lse2, int64_t rows) {
  int wave = (blockIdx.x * blockDim.x + threadIdx.x) / kWave;
  int lane = threadIdx.x & (kWave - 1);
  int num_waves = (gridDim.x * blockDim.x) / kWave;
  for (int64_t r = wave; r < rows; r += num_waves) {
    const bf16x8* d8 = reinterpret_cast<const bf16x8*>(dO + r * DH);
    const bf16x8* o8 = reinterpret_cast<const bf16x8*>(O + r * DH);
    float acc = 0.f;
    if (lane < 16) {
      bf16x8 a = d8[lane], b = o8[lane];
#pragma unroll
      for (int j = 0; j < 8; ++j) acc += bf2f(a.v[j]) * bf2f(b.v[j]);
    }
#pragma unroll
    for (int off = 8; off > 0; off >>= 1) acc += __shfl_down(acc, off, kWave);
    if (lane == 0) {
      delta[r] = acc;
      lse2[r] = LSE[r] * 1.4426950408889634f;
    }
  }
}

// MODE (probe ablation, product path uses 0): 1 = skip the dQ path (LDS
// tree + global flush), 2 = skip phase 2 (dV/dK), 3 = skip softmax VALU
// (fabricated packs), 4 = skip per-tile Q^T/dO^T staging, 5 = skip qrow/dorow
// global loads. Non-zero modes produce garbage results.
template <int MODE>
__global__ __launch_bounds__(256, 1) void k_attn_bwd(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, const bf16_t* __restrict__ dO,
    const float* __restrict__ delta, const float* __restrict__ lse2,
    float* __restrict__ dQacc, bf16_t* __restrict__ dK,
    bf16_t* __restrict__ dV, int B, int Hq, int Hkv, int64_t S, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* krow = reinterpret_cast<bf16_t*>(smem);            // [128][128] 32 K
  bf16_t* vrow = reinterpret_cast<bf16_t*>(smem + 32768);    // [128][128] 32 K
  bf16_t* ktr = reinterpret_cast<bf16_t*>(smem + 65536);     // [128][128] 32 K
  bf16_t* qtr = reinterpret_cast<bf16_t*>(smem + 98304);     // [128][32] 8 K
  bf16_t* dotr = reinterpret_cast<bf16_t*>(smem + 106496);   // [128][32] 8 K
  // per-wave bf16 dQ partials: each wave owns a quarter, written with plain
  // stores (LDS fp32 atomics measured ~600 cycles/op; a fp32 pairwise tree
  // doubled register spans). One extra bf16 rounding per 32-kv partial.
  bf16_t* dqred = reinterpret_cast<bf16_t*>(smem + 114688);  // [4][32][128] 32 K
  float* lsed = reinterpret_cast<float*>(smem + 147456);     // [32]
  float* deld = reinterpret_cast<float*>(smem + 147584);     // [32]

  const int kvb = blockIdx.x;          // kv block of 128 rows
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;           // 0..3 = kv slice
  const int half = lane >> 5;
  const int col = lane & 31;

  const bf16_t* Qb = Q + (((int64_t)b * Hq + hq) * S) * DH;
  const bf16_t* Kb = K + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* Vb = V + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* dOb = dO + (((int64_t)b * Hq + hq) * S) * DH;
  const float* delb = delta + ((int64_t)b * Hq + hq) * S;
  const float* lseb = lse2 + ((int64_t)b * Hq + hq) * S;
  float* dQb = dQacc + (((int64_t)b * Hq + hq) * S) * DH;
  bf16_t* dKb = dK + (((int64_t)b * Hq + hq) * S) * DH;  // per-HQ; host sums GQA
  bf16_t* dVb = dV + (((int64_t)b * Hq + hq) * S) * DH;

  const int64_t kv0 = (int64_t)kvb * 128;
  const int kvrow_l = wave * 32 + col;   // this lane's kv row (or1 / K,V reads)

  // ---- stage K/V rows + K^T once per block (256 thr: 8 passes of 4 KiB)
  {
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      int o = i * 4096 + tid * 16;
      int row = o >> 8;
      int colb = o & 255;
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(krow) + row * 256 + kswz(row, colb)) =
          *reinterpret_cast<const bf16x8*>(Kb + (kv0 + row) * DH + (colb >> 1));
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(vrow) + row * 256 + kswz(row, colb)) =
          *reinterpret_cast<const bf16x8*>(Vb + (kv0 + row) * DH + (colb >> 1));
    }
    // K^T [128 d][128 kv]: 2048 units of [1 kv][8 d] / 256 thr = 8 each
#pragma unroll
    for (int u = 0; u < 8; ++u) {
      int unit = tid + u * 256;
      int kv = unit & 127;
      int d0 = (unit >> 7) * 8;
      bf16x8 v = *reinterpret_cast<const bf16x8*>(Kb + (kv0 + kv) * DH + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int row = d0 + j;
        ktr[(row * 256 + kswz(row, kv * 2)) >> 1] = v.v[j];
      }
    }
  }

  f32x16 dv_acc[4], dk_acc[4];
#pragma unroll
  for (int d = 0; d < 4; ++d) {
    dv_acc[d] = f32x16{};
    dk_acc[d] = f32x16{};
  }

  const float scale2 = scale * 1.4426950408889634f;
  const int qt0 = (int)(kv0 / 32);
  const int qtn = (int)(S / 32);

  for (int qt = qt0; qt < qtn; ++qt) {
    const int64_t q0 = (int64_t)qt * 32;
    // ---- stage Q^T / dO^T (512 units of [1 q][8 d] / 256 thr = 2 each)
    if (MODE != 4) {
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        int unit = tid + u * 256;
        int q = unit & 31;
        int d0 = (unit >> 5) * 8;
        bf16x8 vq = *reinterpret_cast<const bf16x8*>(Qb + (q0 + q) * DH + d0);
        bf16x8 vd = *reinterpret_cast<const bf16x8*>(dOb + (q0 + q) * DH + d0);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int row = d0 + j;
          qtr[(row * 64 + qswz(row, q * 2)) >> 1] = vq.v[j];
          dotr[(row * 64 + qswz(row, q * 2)) >> 1] = vd.v[j];
        }
      }
      if (tid < 32) {
        lsed[tid] = lseb[q0 + tid];
        deld[tid] = delb[q0 + tid];
      }
    }
    __syncthreads();

    // this wave has work only when some of its kv rows are <= some q
    const bool live = (q0 + 31) >= (kv0 + wave * 32);
    const bool diag = (q0 < kv0 + 128);

    // per-wave Q/dO row fragments at q = q0+col (or1 B-operands and or2
    // A-operands: either way lane l holds row q0 + (l&31)); phase 2 reloads
    // them from L2 so their 64 registers do not span the reduction section.
    bf16_t* myq = dqred + wave * (32 * DH);
    if (live) {
      bf16frag qrow[8], dorow[8];
      if (MODE != 5) {
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          qrow[c] = *reinterpret_cast<const bf16frag*>(Qb + (q0 + col) * DH + c * 16 + half * 8);
          dorow[c] = *reinterpret_cast<const bf16frag*>(dOb + (q0 + col) * DH + c * 16 + half * 8);
        }
      } else {
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          uint4 z{(uint32_t)(c + col), (uint32_t)half, 3u, 4u};
          qrow[c] = __builtin_bit_cast(bf16frag, z);
          dorow[c] = qrow[c];
        }
      }
      const float lse_l = lsed[col];
      const float del_l = deld[col];

      // ---- phase 1 (or1): C = [kv regs][q lanes] -> dS1 -> dQ.
      // s1 = mfma(K, Q): rows = this wave's 32 kv, cols = q (lane scalars
      // lse/delta, the forward's own orientation).
      f32x16 s1 = f32x16{}, dp1 = f32x16{};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        int colb = (c * 16 + half * 8) * 2;
        bf16frag kf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(krow) + kvrow_l * 256 + kswz(kvrow_l, colb));
        bf16frag vf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(vrow) + kvrow_l * 256 + kswz(kvrow_l, colb));
        s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qrow[c], s1, 0, 0, 0);
        dp1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dorow[c], dp1, 0, 0, 0);
      }
      uint32_t dg1[8];
      if (MODE == 3) {
#pragma unroll
        for (int i = 0; i < 8; ++i) dg1[i] = __builtin_bit_cast(uint32_t, s1[i]);
      } else
#pragma unroll
      for (int r = 0; r < 16; r += 2) {
        float g[2];
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
          int r2 = r + rr;
          int kvl = (r2 & 3) + 8 * (r2 >> 2) + 4 * half + wave * 32;
          bool masked = diag && ((int64_t)kv0 + kvl > q0 + col);
          float pp = masked ? 0.f : __builtin_exp2f(s1[r2] * scale2 - lse_l);
          g[rr] = pp * (dp1[r2] - del_l) * scale;
        }
        dg1[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)g[0]) |
                      ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)g[1]) << 16);
      }
      bf16frag da1[2];
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
        uint32_t a0 = half ? dg1[4 * mch] : dg1[4 * mch + 2];
        uint32_t a1 = half ? dg1[4 * mch + 1] : dg1[4 * mch + 3];
        uint32_t b0 = swap32_u(a0, half);
        uint32_t b1 = swap32_u(a1, half);
        uint4 u{half ? b0 : dg1[4 * mch], half ? b1 : dg1[4 * mch + 1],
                half ? dg1[4 * mch + 2] : b0, half ? dg1[4 * mch + 3] : b1};
        da1[mch] = __builtin_bit_cast(bf16frag, u);
      }
      // dQ[q][d] += dS1^T(pack: A[q][kv-chunk]) x K^T-tile(B[kv][d]);
      // each dblk's 16 values go straight to this wave's LDS quarter so the
      // registers die immediately.
#pragma unroll
      for (int dblk = 0; dblk < 4 && MODE != 1; ++dblk) {
        f32x16 dq = f32x16{};
        if (MODE != 8) {
#pragma unroll
          for (int mch = 0; mch < 2; ++mch) {
            int trow = dblk * 32 + col;
            int colb = (wave * 32 + mch * 16 + half * 8) * 2;
            bf16frag ktf = *reinterpret_cast<const bf16frag*>(
                reinterpret_cast<const char*>(ktr) + trow * 256 + kswz(trow, colb));
            dq = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da1[mch], ktf, dq, 0, 0, 0);
          }
        } else {
#pragma unroll
          for (int r = 0; r < 16; ++r)
            dq[r] = __builtin_bit_cast(float, dg1[r >> 1]) + (float)dblk;
        }
        if (MODE == 7) {
          asm volatile("" :: "v"(dq[0]), "v"(dq[15]));
        } else {
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            int qr = (r & 3) + 8 * (r >> 2) + 4 * half;
            myq[qr * DH + dblk * 32 + col] = f2bf(dq[r]);
          }
        }
      }
    }

    // ---- cross-wave dQ flush: non-live waves zero their quarter, then the
    // block sums the 4 bf16 quarters in fp32 and atomically adds to global.
    if (MODE != 1 && MODE != 7) {
      if (!live) {
#pragma unroll
        for (int dblk = 0; dblk < 4; ++dblk)
#pragma unroll
          for (int r = 0; r < 16; ++r) {
            int qr = (r & 3) + 8 * (r >> 2) + 4 * half;
            myq[qr * DH + dblk * 32 + col] = (bf16_t)0;
          }
      }
      __syncthreads();
      if (MODE != 6) {
        for (int i = tid; i < 32 * DH; i += 256) {
          float vsum = bf2f(dqred[i]) + bf2f(dqred[32 * DH + i]) +
                       bf2f(dqred[2 * 32 * DH + i]) + bf2f(dqred[3 * 32 * DH + i]);
          if (vsum != 0.f) atomicAdd(&dQb[q0 * DH + i], vsum);
        }
      }
    } else {
      __syncthreads();
    }

    // ---- phase 2 (or2): C = [q regs][kv lanes] -> P2, dS2 -> dV, dK.
    // s2 = mfma(Q, K): rows = q tile, cols = this wave's 32 kv.
    if (live && MODE != 2) {
      bf16frag qrow[8], dorow[8];
      if (MODE != 5) {
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          qrow[c] = *reinterpret_cast<const bf16frag*>(Qb + (q0 + col) * DH + c * 16 + half * 8);
          dorow[c] = *reinterpret_cast<const bf16frag*>(dOb + (q0 + col) * DH + c * 16 + half * 8);
        }
      } else {
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          uint4 z{(uint32_t)(c + col), (uint32_t)half, 3u, 4u};
          qrow[c] = __builtin_bit_cast(bf16frag, z);
          dorow[c] = qrow[c];
        }
      }
      f32x16 s2 = f32x16{}, dp2 = f32x16{};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        int colb = (c * 16 + half * 8) * 2;
        bf16frag kf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(krow) + kvrow_l * 256 + kswz(kvrow_l, colb));
        bf16frag vf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(vrow) + kvrow_l * 256 + kswz(kvrow_l, colb));
        s2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qrow[c], kf, s2, 0, 0, 0);
        dp2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dorow[c], vf, dp2, 0, 0, 0);
      }
      uint32_t pk2[8], dg2[8];
      if (MODE == 3) {
#pragma unroll
        for (int i = 0; i < 8; ++i) {
          pk2[i] = __builtin_bit_cast(uint32_t, s2[i]);
          dg2[i] = __builtin_bit_cast(uint32_t, dp2[i]);
        }
      } else
#pragma unroll
      for (int r = 0; r < 16; r += 2) {
        float pv[2], gv[2];
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
          int r2 = r + rr;
          int qrm = (r2 & 3) + 8 * (r2 >> 2) + 4 * half;
          bool masked = diag && ((int64_t)kv0 + wave * 32 + col > q0 + qrm);
          float pp = masked ? 0.f : __builtin_exp2f(s2[r2] * scale2 - lsed[qrm]);
          pv[rr] = pp;
          gv[rr] = pp * (dp2[r2] - deld[qrm]) * scale;
        }
        pk2[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)pv[0]) |
                      ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)pv[1]) << 16);
        dg2[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)gv[0]) |
                      ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)gv[1]) << 16);
      }
      bf16frag pa2[2], da2[2];
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
        uint32_t a0 = half ? pk2[4 * mch] : pk2[4 * mch + 2];
        uint32_t a1 = half ? pk2[4 * mch + 1] : pk2[4 * mch + 3];
        uint32_t b0 = swap32_u(a0, half);
        uint32_t b1 = swap32_u(a1, half);
        uint4 u{half ? b0 : pk2[4 * mch], half ? b1 : pk2[4 * mch + 1],
                half ? pk2[4 * mch + 2] : b0, half ? pk2[4 * mch + 3] : b1};
        pa2[mch] = __builtin_bit_cast(bf16frag, u);
        uint32_t c0 = half ? dg2[4 * mch] : dg2[4 * mch + 2];
        uint32_t c1 = half ? dg2[4 * mch + 1] : dg2[4 * mch + 3];
        uint32_t e0 = swap32_u(c0, half);
        uint32_t e1 = swap32_u(c1, half);
        uint4 u2{half ? e0 : dg2[4 * mch], half ? e1 : dg2[4 * mch + 1],
                 half ? dg2[4 * mch + 2] : e0, half ? dg2[4 * mch + 3] : e1};
        da2[mch] = __builtin_bit_cast(bf16frag, u2);
      }
      // dV[kv][d] += P2^T(pack: A[kv][q-chunk]) x dO^T-tile(B[q][d]);
      // dK[kv][d] += dS2^T x Q^T-tile
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
#pragma unroll
        for (int dblk = 0; dblk < 4; ++dblk) {
          int trow = dblk * 32 + col;
          int colb = (mch * 16 + half * 8) * 2;
          bf16frag dof = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(dotr) + trow * 64 + qswz(trow, colb));
          bf16frag qf = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(qtr) + trow * 64 + qswz(trow, colb));
          dv_acc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa2[mch], dof, dv_acc[dblk], 0, 0, 0);
          dk_acc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da2[mch], qf, dk_acc[dblk], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: dK/dV bf16 for this wave's kv rows (C layout: kv in regs,
  // d = lane col + 32*dblk)
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int kvr = (r & 3) + 8 * (r >> 2) + 4 * half;
    int64_t kvg = kv0 + wave * 32 + kvr;
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      dKb[kvg * DH + d * 32 + col] = f2bf(dk_acc[d][r]);
      dVb[kvg * DH + d * 32 + col] = f2bf(dv_acc[d][r]);
    }
  }
}

}  // namespace

extern "C" int vh_attn_bwd_pre_bf16(const uint16_t* dO, const uint16_t* O,
                                    const float* LSE, float* delta,
                                    float* lse2, int64_t rows, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  int blocks = (int)((rows + 3) / 4);
  if (blocks > 2048) blocks = 2048;
  hipLaunchKernelGGL(k_attn_bwd_pre, dim3(blocks), dim3(256), 0, s,
                     reinterpret_cast<const bf16_t*>(dO),
                     reinterpret_cast<const bf16_t*>(O), LSE, delta, lse2, rows);
  VH_HIP(hipGetLastError());
  return 0;
}

extern "C" int vh_attn_bwd_bf16(const uint16_t* Q, const uint16_t* K,
                                const uint16_t* V, const uint16_t* dO,
                                const float* delta, const float* lse2,
                                float* dQacc, uint16_t* dK, uint16_t* dV,
                                int B, int Hq, int Hkv, int64_t S, float scale,
                                void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % 128 == 0, "S %% 128 != 0");
  dim3 grid((uint32_t)(S / 128), (uint32_t)(B * Hq));
  hipLaunchKernelGGL(k_attn_bwd<0>, grid, dim3(256), 147712, s,
                     reinterpret_cast<const bf16_t*>(Q),
                     reinterpret_cast<const bf16_t*>(K),
                     reinterpret_cast<const bf16_t*>(V),
                     reinterpret_cast<const bf16_t*>(dO), delta, lse2, dQacc,
                     reinterpret_cast<bf16_t*>(dK),
                     reinterpret_cast<bf16_t*>(dV), B, Hq, Hkv, S, scale);
  VH_HIP(hipGetLastError());
  return 0;
}

extern "C" int vh_attn_bwd_probe_bf16(const uint16_t* Q, const uint16_t* K,
                                      const uint16_t* V, const uint16_t* dO,
                                      const float* delta, const float* lse2,
                                      float* dQacc, uint16_t* dK, uint16_t* dV,
                                      int B, int Hq, int Hkv, int64_t S,
                                      float scale, int mode, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  dim3 grid((uint32_t)(S / 128), (uint32_t)(B * Hq));
#define VH_BWD_LAUNCH(M)                                                      \
  hipLaunchKernelGGL(k_attn_bwd<M>, grid, dim3(256), 147712, s,               \
                     reinterpret_cast<const bf16_t*>(Q),                      \
                     reinterpret_cast<const bf16_t*>(K),                      \
                     reinterpret_cast<const bf16_t*>(V),                      \
                     reinterpret_cast<const bf16_t*>(dO), delta, lse2, dQacc, \
                     reinterpret_cast<bf16_t*>(dK),                           \
                     reinterpret_cast<bf16_t*>(dV), B, Hq, Hkv, S, scale)
  switch (mode) {
    case 0: VH_BWD_LAUNCH(0); break;
    case 1: VH_BWD_LAUNCH(1); break;
    case 2: VH_BWD_LAUNCH(2); break;
    case 3: VH_BWD_LAUNCH(3); break;
    case 4: VH_BWD_LAUNCH(4); break;
    case 5: VH_BWD_LAUNCH(5); break;
    case 6: VH_BWD_LAUNCH(6); break;
    case 7: VH_BWD_LAUNCH(7); break;
    case 8: VH_BWD_LAUNCH(8); break;
    default: return 1;
  }
#undef VH_BWD_LAUNCH
  VH_HIP(hipGetLastError());
  return 0;
}

// ============================================================================
// Split backward (v2): two kernels, registers only, no atomics.
//   k_attn_bwd_dkv — block owns 128 kv rows (wave = 32-kv slice), loops q
//     tiles from the diagonal; single or2 softmax; dK/dV in registers.
//     80 KB LDS -> 2 blocks/CU (2 waves/SIMD).
//   k_attn_bwd_dq — block owns 128 q rows (wave = 32-q slice), loops kv
//     tiles up to the diagonal; single or1 softmax; dQ in registers, written
//     once as bf16 (this block is the only contributor). 24 KB LDS.
// Scores are recomputed in both kernels (the monolithic variant's cross-wave
// dQ reduction measured 2-4 ms of pure LDS/atomic overhead per call).
// ============================================================================

namespace {

__global__ __launch_bounds__(256, 2) void k_attn_bwd_dkv(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, const bf16_t* __restrict__ dO,
    const float* __restrict__ delta, const float* __restrict__ lse2,
    bf16_t* __restrict__ dK, bf16_t* __restrict__ dV, int B, int Hq, int Hkv,
    int64_t S, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* krow = reinterpret_cast<bf16_t*>(smem);            // [128][128] 32 K
  bf16_t* vrow = reinterpret_cast<bf16_t*>(smem + 32768);    // [128][128] 32 K
  bf16_t* qtr = reinterpret_cast<bf16_t*>(smem + 65536);     // [128][32] 8 K
  bf16_t* dotr = reinterpret_cast<bf16_t*>(smem + 73728);    // [128][32] 8 K

  const int kvb = blockIdx.x;
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;           // 32-row kv slice
  const int half = lane >> 5;
  const int col = lane & 31;

  const bf16_t* Qb = Q + (((int64_t)b * Hq + hq) * S) * DH;
  const bf16_t* Kb = K + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* Vb = V + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* dOb = dO + (((int64_t)b * Hq + hq) * S) * DH;
  const float* delb = delta + ((int64_t)b * Hq + hq) * S;
  const float* lseb = lse2 + ((int64_t)b * Hq + hq) * S;
  bf16_t* dKb = dK + (((int64_t)b * Hq + hq) * S) * DH;  // per-Hq; host sums
  bf16_t* dVb = dV + (((int64_t)b * Hq + hq) * S) * DH;

  const int64_t kv0 = (int64_t)kvb * 128;
  const int kvrow_l = wave * 32 + col;

  // stage K/V rows once (256 thr: 8 passes of 4 KiB)
#pragma unroll
  for (int i = 0; i < 8; ++i) {
    int o = i * 4096 + tid * 16;
    int row = o >> 8;
    int colb = o & 255;
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(krow) + row * 256 + kswz(row, colb)) =
        *reinterpret_cast<const bf16x8*>(Kb + (kv0 + row) * DH + (colb >> 1));
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(vrow) + row * 256 + kswz(row, colb)) =
        *reinterpret_cast<const bf16x8*>(Vb + (kv0 + row) * DH + (colb >> 1));
  }

  f32x16 dv_acc[4], dk_acc[4];
#pragma unroll
  for (int d = 0; d < 4; ++d) {
    dv_acc[d] = f32x16{};
    dk_acc[d] = f32x16{};
  }

  const float scale2 = scale * 1.4426950408889634f;
  const int qt0 = (int)(kv0 / 32);
  const int qtn = (int)(S / 32);

  for (int qt = qt0; qt < qtn; ++qt) {
    const int64_t q0 = (int64_t)qt * 32;
    // stage Q^T / dO^T (512 units / 256 thr = 2 each)
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      int unit = tid + u * 256;
      int q = unit & 31;
      int d0 = (unit >> 5) * 8;
      bf16x8 vq = *reinterpret_cast<const bf16x8*>(Qb + (q0 + q) * DH + d0);
      bf16x8 vd = *reinterpret_cast<const bf16x8*>(dOb + (q0 + q) * DH + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int row = d0 + j;
        qtr[(row * 64 + qswz(row, q * 2)) >> 1] = vq.v[j];
        dotr[(row * 64 + qswz(row, q * 2)) >> 1] = vd.v[j];
      }
    }
    __syncthreads();

    const bool live = (q0 + 31) >= (kv0 + wave * 32);
    const bool diag = (q0 < kv0 + 128);

    if (live) {
      // or2: C = [q regs][kv lanes]; A-operands are the lane's Q/dO rows,
      // loaded per chunk (k = q0+col) so only 8 regs are live per iteration.
      // lse/delta: per-lane own-row loads + cross-lane shuffles (see dkv_g).
      const float lse_own = lseb[q0 + col];
      const float del_own = delb[q0 + col];
      f32x16 s2 = f32x16{}, dp2 = f32x16{};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        int colb = (c * 16 + half * 8) * 2;
        bf16frag qc = *reinterpret_cast<const bf16frag*>(
            Qb + (q0 + col) * DH + c * 16 + half * 8);
        bf16frag dc = *reinterpret_cast<const bf16frag*>(
            dOb + (q0 + col) * DH + c * 16 + half * 8);
        bf16frag kf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(krow) + kvrow_l * 256 + kswz(kvrow_l, colb));
        bf16frag vf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(vrow) + kvrow_l * 256 + kswz(kvrow_l, colb));
        s2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qc, kf, s2, 0, 0, 0);
        dp2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dc, vf, dp2, 0, 0, 0);
      }
      uint32_t pk2[8], dg2[8];
#pragma unroll
      for (int r = 0; r < 16; r += 2) {
        float pv[2], gv[2];
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
          int r2 = r + rr;
          int qrm = (r2 & 3) + 8 * (r2 >> 2) + 4 * half;
          bool masked = diag && ((int64_t)kv0 + wave * 32 + col > q0 + qrm);
          float lse_q = __shfl(lse_own, qrm, 32);
          float del_q = __shfl(del_own, qrm, 32);
          float pp = masked ? 0.f : __builtin_exp2f(s2[r2] * scale2 - lse_q);
          pv[rr] = pp;
          gv[rr] = pp * (dp2[r2] - del_q) * scale;
        }
        pk2[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)pv[0]) |
                      ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)pv[1]) << 16);
        dg2[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)gv[0]) |
                      ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)gv[1]) << 16);
      }
      bf16frag pa2[2], da2[2];
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
        uint32_t a0 = half ? pk2[4 * mch] : pk2[4 * mch + 2];
        uint32_t a1 = half ? pk2[4 * mch + 1] : pk2[4 * mch + 3];
        uint32_t b0 = swap32_u(a0, half);
        uint32_t b1 = swap32_u(a1, half);
        uint4 u{half ? b0 : pk2[4 * mch], half ? b1 : pk2[4 * mch + 1],
                half ? pk2[4 * mch + 2] : b0, half ? pk2[4 * mch + 3] : b1};
        pa2[mch] = __builtin_bit_cast(bf16frag, u);
        uint32_t c0 = half ? dg2[4 * mch] : dg2[4 * mch + 2];
        uint32_t c1 = half ? dg2[4 * mch + 1] : dg2[4 * mch + 3];
        uint32_t e0 = swap32_u(c0, half);
        uint32_t e1 = swap32_u(c1, half);
        uint4 u2{half ? e0 : dg2[4 * mch], half ? e1 : dg2[4 * mch + 1],
                 half ? dg2[4 * mch + 2] : e0, half ? dg2[4 * mch + 3] : e1};
        da2[mch] = __builtin_bit_cast(bf16frag, u2);
      }
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
#pragma unroll
        for (int dblk = 0; dblk < 4; ++dblk) {
          int trow = dblk * 32 + col;
          int colb = (mch * 16 + half * 8) * 2;
          bf16frag dof = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(dotr) + trow * 64 + qswz(trow, colb));
          bf16frag qf = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(qtr) + trow * 64 + qswz(trow, colb));
          dv_acc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa2[mch], dof, dv_acc[dblk], 0, 0, 0);
          dk_acc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da2[mch], qf, dk_acc[dblk], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int kvr = (r & 3) + 8 * (r >> 2) + 4 * half;
    int64_t kvg = kv0 + wave * 32 + kvr;
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      dKb[kvg * DH + d * 32 + col] = f2bf(dk_acc[d][r]);
      dVb[kvg * DH + d * 32 + col] = f2bf(dv_acc[d][r]);
    }
  }
}






// dkv v4: 64-kv strips, 64-row q tiles, 4 waves = (2 kv slices) x (2 q
// subtiles) — each wave owns a distinct 32x32 quadrant, so per-wave work is
// unchanged but the barrier/staging cadence HALVES per unit of work, the
// grid doubles (better fill/balance near the diagonal), and LDS drops to
// 64 KB (2 blocks/CU, 2 waves/SIMD). dK/dV partials: wave (s, u) holds the
// sum over its q subtiles; the two subtile-waves of a slice combine through
// LDS once at kernel end.
__global__ __launch_bounds__(256, 2) void k_attn_bwd_dkv4(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, const bf16_t* __restrict__ dO,
    const float* __restrict__ delta, const float* __restrict__ lse2,
    bf16_t* __restrict__ dK, bf16_t* __restrict__ dV, int B, int Hq, int Hkv,
    int64_t S, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* krow = reinterpret_cast<bf16_t*>(smem);            // [64][128] 16 K
  bf16_t* vrow = reinterpret_cast<bf16_t*>(smem + 16384);    // [64][128] 16 K
  bf16_t* qtr = reinterpret_cast<bf16_t*>(smem + 32768);     // [128][64] 16 K
  bf16_t* dotr = reinterpret_cast<bf16_t*>(smem + 49152);    // [128][64] 16 K
  float* red = reinterpret_cast<float*>(smem);               // epilogue reuse

  const int kvb = blockIdx.x;          // 64-row kv strip
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int kvslice = wave & 1;        // 32-kv slice within the strip
  const int qsub = wave >> 1;          // 32-q subtile within the 64-q tile
  const int half = lane >> 5;
  const int col = lane & 31;

  const bf16_t* Qb = Q + (((int64_t)b * Hq + hq) * S) * DH;
  const bf16_t* Kb = K + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* Vb = V + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* dOb = dO + (((int64_t)b * Hq + hq) * S) * DH;
  const float* delb = delta + ((int64_t)b * Hq + hq) * S;
  const float* lseb = lse2 + ((int64_t)b * Hq + hq) * S;
  bf16_t* dKb = dK + (((int64_t)b * Hq + hq) * S) * DH;  // per-Hq; host sums
  bf16_t* dVb = dV + (((int64_t)b * Hq + hq) * S) * DH;

  const int64_t kv0 = (int64_t)kvb * 64;
  const int kvrow_l = kvslice * 32 + col;

  // stage the strip's 64 K/V rows once (256 thr: 4 passes of 4 KiB each
  // matrix — NOT 8: an 8-pass loop here once overwrote vrow with K rows
  // 64..127 and shipped a dK-only parity bug)
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int o = i * 4096 + tid * 16;
    int row = o >> 8;
    int colb = o & 255;
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(krow) + row * 256 + kswz(row, colb)) =
        *reinterpret_cast<const bf16x8*>(Kb + (kv0 + row) * DH + (colb >> 1));
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(vrow) + row * 256 + kswz(row, colb)) =
        *reinterpret_cast<const bf16x8*>(Vb + (kv0 + row) * DH + (colb >> 1));
  }

  f32x16 dv_acc[4], dk_acc[4];
#pragma unroll
  for (int d = 0; d < 4; ++d) {
    dv_acc[d] = f32x16{};
    dk_acc[d] = f32x16{};
  }

  const float scale2 = scale * 1.4426950408889634f;
  const int qt0 = (int)(kv0 / 64);     // 64-row q tiles
  const int qtn = (int)(S / 64);

  for (int qt = qt0; qt < qtn; ++qt) {
    const int64_t q0t = (int64_t)qt * 64;
    // stage Q^T / dO^T [128][64] (1024 units of [1 q][8 d] / 256 thr = 4)
#pragma unroll
    for (int u = 0; u < 4; ++u) {
      int unit = tid + u * 256;
      int q = unit & 63;
      int d0 = (unit >> 6) * 8;
      bf16x8 vq = *reinterpret_cast<const bf16x8*>(Qb + (q0t + q) * DH + d0);
      bf16x8 vd = *reinterpret_cast<const bf16x8*>(dOb + (q0t + q) * DH + d0);
#pragma unroll
      for (int j = 0; j < 8; ++j) {
        int row = d0 + j;
        qtr[(row * 128 + vswz(row, q * 2)) >> 1] = vq.v[j];
        dotr[(row * 128 + vswz(row, q * 2)) >> 1] = vd.v[j];
      }
    }
    __syncthreads();

    const int64_t q0 = q0t + qsub * 32;   // this wave's 32-q subtile
    const bool live = (q0 + 31) >= (kv0 + kvslice * 32);
    const bool diag = (q0 < kv0 + 64);
    if (live) {
      f32x16 s2 = f32x16{}, dp2 = f32x16{};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        int colb = (c * 16 + half * 8) * 2;
        bf16frag qc = *reinterpret_cast<const bf16frag*>(
            Qb + (q0 + col) * DH + c * 16 + half * 8);
        bf16frag dc = *reinterpret_cast<const bf16frag*>(
            dOb + (q0 + col) * DH + c * 16 + half * 8);
        bf16frag kf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(krow) + kvrow_l * 256 + kswz(kvrow_l, colb));
        bf16frag vf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(vrow) + kvrow_l * 256 + kswz(kvrow_l, colb));
        s2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qc, kf, s2, 0, 0, 0);
        dp2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dc, vf, dp2, 0, 0, 0);
      }
      uint32_t pk2[8], dg2[8];
#pragma unroll
      for (int r = 0; r < 16; r += 2) {
        float pv[2], gv[2];
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
          int r2 = r + rr;
          int qrm = (r2 & 3) + 8 * (r2 >> 2) + 4 * half;
          bool masked = diag && ((int64_t)kv0 + kvslice * 32 + col > q0 + qrm);
          float pp = masked ? 0.f : __builtin_exp2f(s2[r2] * scale2 - lseb[q0 + qrm]);
          pv[rr] = pp;
          gv[rr] = pp * (dp2[r2] - delb[q0 + qrm]) * scale;
        }
        pk2[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)pv[0]) |
                      ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)pv[1]) << 16);
        dg2[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)gv[0]) |
                      ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)gv[1]) << 16);
      }
      bf16frag pa2[2], da2[2];
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
        uint32_t a0 = half ? pk2[4 * mch] : pk2[4 * mch + 2];
        uint32_t a1 = half ? pk2[4 * mch + 1] : pk2[4 * mch + 3];
        uint32_t b0 = swap32_u(a0, half);
        uint32_t b1 = swap32_u(a1, half);
        uint4 u{half ? b0 : pk2[4 * mch], half ? b1 : pk2[4 * mch + 1],
                half ? pk2[4 * mch + 2] : b0, half ? pk2[4 * mch + 3] : b1};
        pa2[mch] = __builtin_bit_cast(bf16frag, u);
        uint32_t c0 = half ? dg2[4 * mch] : dg2[4 * mch + 2];
        uint32_t c1 = half ? dg2[4 * mch + 1] : dg2[4 * mch + 3];
        uint32_t e0 = swap32_u(c0, half);
        uint32_t e1 = swap32_u(c1, half);
        uint4 u2{half ? e0 : dg2[4 * mch], half ? e1 : dg2[4 * mch + 1],
                 half ? dg2[4 * mch + 2] : e0, half ? dg2[4 * mch + 3] : e1};
        da2[mch] = __builtin_bit_cast(bf16frag, u2);
      }
      // B-frag q-chunks live in this wave's half of the 64-wide tiles
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
#pragma unroll
        for (int dblk = 0; dblk < 4; ++dblk) {
          int trow = dblk * 32 + col;
          int colb = (qsub * 32 + mch * 16 + half * 8) * 2;
          bf16frag dof = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(dotr) + trow * 128 + vswz(trow, colb));
          bf16frag qf = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(qtr) + trow * 128 + vswz(trow, colb));
          dv_acc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa2[mch], dof, dv_acc[dblk], 0, 0, 0);
          dk_acc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da2[mch], qf, dk_acc[dblk], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // combine the two q-subtile partials per kv slice through LDS, then store.
  // red layout: [slice 2][lane 64][64 floats] = 32 KB (reuses K/V space).
#pragma unroll
  for (int m = 0; m < 2; ++m) {   // 0: dv, 1: dk
    f32x16* acc = m ? dk_acc : dv_acc;
    if (qsub == 1) {
      float* out = red + (kvslice * 64 + lane) * 64;
#pragma unroll
      for (int d = 0; d < 4; ++d)
#pragma unroll
        for (int r = 0; r < 16; r += 4)
          *reinterpret_cast<float4*>(out + d * 16 + r) =
              float4{acc[d][r], acc[d][r + 1], acc[d][r + 2], acc[d][r + 3]};
    }
    __syncthreads();
    if (qsub == 0) {
      const float* in = red + (kvslice * 64 + lane) * 64;
#pragma unroll
      for (int d = 0; d < 4; ++d)
#pragma unroll
        for (int r = 0; r < 16; ++r) acc[d][r] += in[d * 16 + r];
      bf16_t* dst = m ? dKb : dVb;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int kvr = (r & 3) + 8 * (r >> 2) + 4 * half;
        int64_t kvg = kv0 + kvslice * 32 + kvr;
#pragma unroll
        for (int d = 0; d < 4; ++d)
          dst[kvg * DH + d * 32 + col] = f2bf(acc[d][r]);
      }
    }
    __syncthreads();
  }
}

// dkv v6 (the dispatched variant): GQA-folded on the 64-kv-strip geometry.
//   - grid (S/64, B*Hkv): a block owns a 64-row kv strip of ONE KV head and
//     loops the whole GQA head group (rep = Hq/Hkv) — dK/dV are written ONCE
//     to the [B,Hkv,S,D] buffers (no per-Q-head intermediates, no host group
//     sum) and the K/V staging amortizes over the group. The strip split
//     keeps the grid at >= 2 blocks/CU even at B = 1: the first fold attempt
//     on 128-kv blocks (grid S/128 x B*Hkv = 256 blocks at the llama
//     microbench shape) measured ~2.45 ms from half-empty wave slots vs
//     1.78 ms for the unfolded v2 — occupancy beat traffic.
//   - 4 waves = (2 kv slices) x (2 q subtiles of a 64-row staged q tile):
//     each wave owns a distinct 32x32 quadrant; barrier cadence halves per
//     unit of work vs 32-row q tiles (the dkv4 probe geometry).
//   - PREF of the or2 A-operands (the lane's Q/dO rows): the first PREF
//     chunks are prefetched into registers BEFORE the transposed-LDS staging
//     stores (their L2 latency hides under the staging writes), the rest
//     load in-loop as dkv v2/v4 did. PREF=8 costs 64 VGPRs and spills
//     (~250 B/lane) against the 128 accumulator regs at the unified-file
//     2-waves/SIMD budget of 256; PREF=4 fits. All three measured on-box
//     via vh_attn_bwd2_dkv6probe_bf16.
//   - DOC: packed-varlen block-diagonal causal via doc_start/doc_end
//     (per-token document bounds; see k_attn_fwd).
//   - DB (probe g2): register double-buffer of the Q^T/dO^T staging. PMC on
//     v6 (profiles/r02_dkv_pmc.txt): SQ_WAIT_ANY = 66% of wave cycles —
//     waves parked at the per-tile barrier pair draining the staging
//     GLOBAL loads. DB issues the NEXT tile's staging loads right after
//     the stage barrier (they drain under the MFMA chain) so only the
//     ds_writes + lgkm sit between barriers. 32 VGPRs (4x bf16x8 x 2
//     tensors), paid for by PREF=0 (the dispatched depth).
// LDS 64 KB: K/V strips 2x16 K + Q^T/dO^T tiles 2x16 K -> 2 blocks/CU.
//   - TR (probe g3): natural-layout Q/dO staging image + CDNA4
//     ds_read_b64_tr_b16 hardware transpose reads for the dv/dk B-frags
//     (guide T10). Replaces the 64 scalar b16 transpose stores per thread
//     per tile with 8 b128 stores into a bank-permuted natural image
//     (off(q,d) = q*256 + g*8 + (d&3)*2, g = (c&7)|(((q&3)^(c>>3))&3)<<3,
//     c = d>>2: reads 32 distinct 8-B slots/phase, stores 16 distinct
//     16-B slots/phase -> conflict-free both ways; mapping derived from
//     the on-box dump tests/gpu_tr16_probe.hip -> gpurun_out/tr16map.txt).
//   - TR2 (probe g4): like TR but over the L16 latin-square image
//     off(q,d) = q*256 + L16*16 + (d&7)*2, L16 = (q&15) ^ (((p&3)<<2)|(p>>2)),
//     p = d>>3 — conflict-free for the b128 stores, the tr16 B-frag reads
//     AND plain b128 A-frag reads, so the per-iteration Q/dO A-fragments
//     come from LDS instead of re-reading global (halves Q/dO traffic).
template <bool DOC, int PREF, int DBN = 0, bool DBLATE = false, int TRMODE = 0>
__global__ __launch_bounds__(256, 2) void k_attn_bwd_dkv_g(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, const bf16_t* __restrict__ dO,
    const float* __restrict__ delta, const float* __restrict__ lse2,
    bf16_t* __restrict__ dK, bf16_t* __restrict__ dV,
    const int* __restrict__ doc_start, const int* __restrict__ doc_end, int B,
    int Hq, int Hkv, int64_t S, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* krow = reinterpret_cast<bf16_t*>(smem);            // [64][128] 16 K
  bf16_t* vrow = reinterpret_cast<bf16_t*>(smem + 16384);    // [64][128] 16 K
  bf16_t* qtr = reinterpret_cast<bf16_t*>(smem + 32768);     // [128][64] 16 K
  bf16_t* dotr = reinterpret_cast<bf16_t*>(smem + 49152);    // [128][64] 16 K
  float* red = reinterpret_cast<float*>(smem);               // epilogue reuse

  const int kvb = blockIdx.x;          // 64-row kv strip
  const int bkh = blockIdx.y;          // b * Hkv + hkv
  const int b = bkh / Hkv;
  const int hkv = bkh % Hkv;
  const int rep = Hq / Hkv;

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int kvslice = wave & 1;        // 32-kv slice within the strip
  const int qsub = wave >> 1;          // 32-q subtile within the 64-q tile
  const int half = lane >> 5;
  const int col = lane & 31;

  const bf16_t* Kb = K + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* Vb = V + (((int64_t)b * Hkv + hkv) * S) * DH;
  bf16_t* dKb = dK + (((int64_t)b * Hkv + hkv) * S) * DH;
  bf16_t* dVb = dV + (((int64_t)b * Hkv + hkv) * S) * DH;

  const int64_t kv0 = (int64_t)kvb * 64;
  const int kvrow_l = kvslice * 32 + col;

  // stage the strip's 64 K/V rows once (256 thr: 4 passes of 4 KiB each)
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int o = i * 4096 + tid * 16;
    int row = o >> 8;
    int colb = o & 255;
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(krow) + row * 256 + kswz(row, colb)) =
        *reinterpret_cast<const bf16x8*>(Kb + (kv0 + row) * DH + (colb >> 1));
    *reinterpret_cast<bf16x8*>(
        reinterpret_cast<char*>(vrow) + row * 256 + kswz(row, colb)) =
        *reinterpret_cast<const bf16x8*>(Vb + (kv0 + row) * DH + (colb >> 1));
  }

  f32x16 dv_acc[4], dk_acc[4];
#pragma unroll
  for (int d = 0; d < 4; ++d) {
    dv_acc[d] = f32x16{};
    dk_acc[d] = f32x16{};
  }

  const float scale2 = scale * 1.4426950408889634f;
  const int qt0 = (int)(kv0 / 64);     // 64-row q tiles
  int qtn = (int)(S / 64);
  int de_wave = 0;
  if (DOC) {
    qtn = (doc_end[kv0 + 63] + 63) / 64;          // block-uniform end
    de_wave = doc_end[kv0 + kvslice * 32 + 31];   // wave-live bound
  }

  // single flattened loop over (head g, q tile): one live address chain —
  // a nested g/qt loop form kept per-head pointer sets alive across the
  // whole accumulator section and spilled ~190 extra bytes/lane
  const int ntiles = qtn - qt0;
  const int itn = rep * ntiles;
  bf16x8 svq[DBN > 0 ? DBN : 1], svd[DBN > 0 ? DBN : 1];  // DB staging regs
#define VH_DKV_LDNEXT(IT2)                                                     \
  {                                                                            \
    const int g2_ = (IT2) / ntiles;                                            \
    const bf16_t* Qb2_ = Q + (((int64_t)b * Hq + hkv * rep + g2_) * S) * DH;   \
    const bf16_t* dOb2_ = dO + (((int64_t)b * Hq + hkv * rep + g2_) * S) * DH; \
    const int64_t q0t2_ = (int64_t)(qt0 + ((IT2) - g2_ * ntiles)) * 64;        \
    _Pragma("unroll") for (int u = 0; u < DBN; ++u) {                          \
      int unit_ = tid + u * 256;                                               \
      int qq_ = unit_ & 63;                                                    \
      int d0_ = (unit_ >> 6) * 8;                                              \
      svq[u] = *reinterpret_cast<const bf16x8*>(Qb2_ + (q0t2_ + qq_) * DH + d0_); \
      svd[u] = *reinterpret_cast<const bf16x8*>(dOb2_ + (q0t2_ + qq_) * DH + d0_); \
    }                                                                          \
  }
  if constexpr (DBN > 0) {
    if (itn > 0) VH_DKV_LDNEXT(0);
  }
  for (int it = 0; it < itn; ++it) {
    const int g = it / ntiles;
    const int qt = qt0 + (it - g * ntiles);
    {
      const int hq = hkv * rep + g;
      const bf16_t* Qb = Q + (((int64_t)b * Hq + hq) * S) * DH;
      const bf16_t* dOb = dO + (((int64_t)b * Hq + hq) * S) * DH;
      const float* delb = delta + ((int64_t)b * Hq + hq) * S;
      const float* lseb = lse2 + ((int64_t)b * Hq + hq) * S;
      const int64_t q0t = (int64_t)qt * 64;
      const int64_t q0 = q0t + qsub * 32;   // this wave's 32-q subtile
      // prefetch this wave's first PREF or2 A-operand chunks (q = q0+col)
      // BEFORE the staging stores so their load latency hides under them
      bf16frag qrow[PREF > 0 ? PREF : 1], dorow[PREF > 0 ? PREF : 1];
#pragma unroll
      for (int c = 0; c < PREF; ++c) {
        qrow[c] = *reinterpret_cast<const bf16frag*>(
            Qb + (q0 + col) * DH + c * 16 + half * 8);
        dorow[c] = *reinterpret_cast<const bf16frag*>(
            dOb + (q0 + col) * DH + c * 16 + half * 8);
      }
      // stage Q^T / dO^T [128][64] (1024 units of [1 q][8 d] / 256 thr = 4)
      if constexpr (TRMODE == 3) {
        // glds fill of the L16 image: window win = wave*4+u covers 4 q rows
        // (1024 B, lane-linear dest); lane sources pair p = P(slot ^ (q&15))
        // — zero VGPR staging, zero ds_writes
#pragma unroll
        for (int u = 0; u < 4; ++u) {
          const int win = wave * 4 + u;
          const int qrow = win * 4 + (lane >> 4);
          const int pp0 = (lane & 15) ^ (qrow & 15);
          const int pg = ((pp0 & 3) << 2) | (pp0 >> 2);
          glds16a(Qb + (q0t + qrow) * DH + pg * 8, qtr + win * 512);
          glds16a(dOb + (q0t + qrow) * DH + pg * 8, dotr + win * 512);
        }
      } else
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        int unit = tid + u * 256;
        int q, d0;
        if constexpr (TRMODE == 1) {
          // phase-conflict-free assignment for the permuted natural image
          q = ((unit >> 2) & 15) | (((unit >> 6) & 3) << 4);
          d0 = (unit & 3) * 8 + ((unit >> 8) & 3) * 32;
        } else {
          // TRMODE 2: the L16 image is conflict-free under the simple
          // 64-consecutive-q assignment
          q = unit & 63;
          d0 = (unit >> 6) * 8;
        }
        bf16x8 vq, vd;
        if (DBN > 0 && u < DBN) {
          vq = svq[u < DBN ? u : 0];
          vd = svd[u < DBN ? u : 0];
        } else {
          vq = *reinterpret_cast<const bf16x8*>(Qb + (q0t + q) * DH + d0);
          vd = *reinterpret_cast<const bf16x8*>(dOb + (q0t + q) * DH + d0);
        }
        if constexpr (TRMODE == 1) {
          int c = d0 >> 2;
          int g = (c & 7) | ((((q & 3) ^ (c >> 3)) & 3) << 3);
          *reinterpret_cast<bf16x8*>(
              reinterpret_cast<char*>(qtr) + q * 256 + g * 8) = vq;
          *reinterpret_cast<bf16x8*>(
              reinterpret_cast<char*>(dotr) + q * 256 + g * 8) = vd;
        } else if constexpr (TRMODE == 2) {
          int pp = d0 >> 3;
          int l16 = (q & 15) ^ (((pp & 3) << 2) | (pp >> 2));
          *reinterpret_cast<bf16x8*>(
              reinterpret_cast<char*>(qtr) + q * 256 + l16 * 16) = vq;
          *reinterpret_cast<bf16x8*>(
              reinterpret_cast<char*>(dotr) + q * 256 + l16 * 16) = vd;
        } else {
#pragma unroll
          for (int j = 0; j < 8; ++j) {
            int row = d0 + j;
            qtr[(row * 128 + vswz(row, q * 2)) >> 1] = vq.v[j];
            dotr[(row * 128 + vswz(row, q * 2)) >> 1] = vd.v[j];
          }
        }
      }
      __syncthreads();
      if constexpr (DBN > 0 && !DBLATE) {
        // next tile's staging loads drain under the MFMA chain below
        if (it + 1 < itn) VH_DKV_LDNEXT(it + 1);
      }

      const bool live = ((q0 + 31) >= (kv0 + kvslice * 32)) &&
                        (!DOC || q0 < de_wave);
      const bool diag = (q0 < kv0 + 64);
      if (live) {
        // per-lane softmax scalars for the lane's own q row; the per-element
        // values come from cross-lane shuffles below. Reading
        // lseb[q0+qrm]/delb[q0+qrm] straight from global per element was a
        // SCATTERED gather (qrm depends on lane>>5) on the critical path
        // between the s2 MFMA chain and the pack; these two coalesced loads
        // issue before the MFMA chain and hide under it.
        const float lse_own = lseb[q0 + col];
        const float del_own = delb[q0 + col];
        const int ds_own = DOC ? doc_start[q0 + col] : 0;
        f32x16 s2 = f32x16{}, dp2 = f32x16{};
#pragma unroll
        for (int c = 0; c < 8; ++c) {
          int colb = (c * 16 + half * 8) * 2;
          bf16frag qc, dc;
          if (c < PREF) {
            qc = qrow[c < PREF ? c : 0];
            dc = dorow[c < PREF ? c : 0];
          } else if constexpr (TRMODE >= 2) {
            // A-frags from the staged L16 image (conflict-free b128):
            // chunk pair p = c*2 + half, row q0+col
            const int pp = c * 2 + half;
            const int rl = qsub * 32 + col;  // local image row
            const int l16 = (rl & 15) ^ (((pp & 3) << 2) | (pp >> 2));
            qc = *reinterpret_cast<const bf16frag*>(
                reinterpret_cast<const char*>(qtr) + rl * 256 + l16 * 16);
            dc = *reinterpret_cast<const bf16frag*>(
                reinterpret_cast<const char*>(dotr) + rl * 256 + l16 * 16);
          } else {
            qc = *reinterpret_cast<const bf16frag*>(
                Qb + (q0 + col) * DH + c * 16 + half * 8);
            dc = *reinterpret_cast<const bf16frag*>(
                dOb + (q0 + col) * DH + c * 16 + half * 8);
          }
          bf16frag kf = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(krow) + kvrow_l * 256 + kswz(kvrow_l, colb));
          bf16frag vf = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(vrow) + kvrow_l * 256 + kswz(kvrow_l, colb));
          s2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(qc, kf, s2, 0, 0, 0);
          dp2 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(dc, vf, dp2, 0, 0, 0);
        }
        uint32_t pk2[8], dg2[8];
#pragma unroll
        for (int r = 0; r < 16; r += 2) {
          float pv[2], gv[2];
#pragma unroll
          for (int rr = 0; rr < 2; ++rr) {
            int r2 = r + rr;
            int qrm = (r2 & 3) + 8 * (r2 >> 2) + 4 * half;
            float lse_q = __shfl(lse_own, qrm, 32);
            float del_q = __shfl(del_own, qrm, 32);
            bool masked = diag && ((int64_t)kv0 + kvslice * 32 + col > q0 + qrm);
            if (DOC)
              masked = masked ||
                       ((int64_t)kv0 + kvslice * 32 + col < __shfl(ds_own, qrm, 32));
            float pp = masked ? 0.f : __builtin_exp2f(s2[r2] * scale2 - lse_q);
            pv[rr] = pp;
            gv[rr] = pp * (dp2[r2] - del_q) * scale;
          }
          pk2[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)pv[0]) |
                        ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)pv[1]) << 16);
          dg2[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)gv[0]) |
                        ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)gv[1]) << 16);
        }
        bf16frag pa2[2], da2[2];
#pragma unroll
        for (int mch = 0; mch < 2; ++mch) {
          uint32_t a0 = half ? pk2[4 * mch] : pk2[4 * mch + 2];
          uint32_t a1 = half ? pk2[4 * mch + 1] : pk2[4 * mch + 3];
          uint32_t b0 = swap32_u(a0, half);
          uint32_t b1 = swap32_u(a1, half);
          uint4 u{half ? b0 : pk2[4 * mch], half ? b1 : pk2[4 * mch + 1],
                  half ? pk2[4 * mch + 2] : b0, half ? pk2[4 * mch + 3] : b1};
          pa2[mch] = __builtin_bit_cast(bf16frag, u);
          uint32_t c0 = half ? dg2[4 * mch] : dg2[4 * mch + 2];
          uint32_t c1 = half ? dg2[4 * mch + 1] : dg2[4 * mch + 3];
          uint32_t e0 = swap32_u(c0, half);
          uint32_t e1 = swap32_u(c1, half);
          uint4 u2{half ? e0 : dg2[4 * mch], half ? e1 : dg2[4 * mch + 1],
                   half ? dg2[4 * mch + 2] : e0, half ? dg2[4 * mch + 3] : e1};
          da2[mch] = __builtin_bit_cast(bf16frag, u2);
        }
        // B-frag q-chunks live in this wave's half of the 64-wide tiles
#pragma unroll
        for (int mch = 0; mch < 2; ++mch) {
#pragma unroll
          for (int dblk = 0; dblk < 4; ++dblk) {
            bf16frag dof, qf;
            if constexpr (TRMODE >= 2) {
              const int m_ = lane & 15;
              const int colhi_ = (lane >> 4) & 1;
              const int qr_ = qsub * 32 + mch * 16 + half * 8 + (m_ >> 2);
              const int c_r = dblk * 8 + colhi_ * 4 + (m_ & 3);
              const int perm_c = (((c_r >> 1) & 3) << 2) | (c_r >> 3);
              const int l16a = (qr_ & 15) ^ perm_c;
              const int l16b = ((qr_ + 4) & 15) ^ perm_c;  // q&15 changes at +4
              const int a0 = qr_ * 256 + l16a * 16 + (c_r & 1) * 8;
              const int a1 = (qr_ + 4) * 256 + l16b * 16 + (c_r & 1) * 8;
              auto* dob3 = (__attribute__((address_space(3))) char*)dotr;
              auto* qb3 = (__attribute__((address_space(3))) char*)qtr;
              typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4t;
              typedef __attribute__((address_space(3))) bf16x4t as3b4;
              bf16x4t d0v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(dob3 + a0));
              bf16x4t d1v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(dob3 + a1));
              bf16x4t q0v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(qb3 + a0));
              bf16x4t q1v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(qb3 + a1));
              dof = __builtin_shufflevector(d0v, d1v, 0, 1, 2, 3, 4, 5, 6, 7);
              qf = __builtin_shufflevector(q0v, q1v, 0, 1, 2, 3, 4, 5, 6, 7);
            } else if constexpr (TRMODE == 1) {
              // two 4-q hardware-transpose reads per frag; lane m of each
              // 16-lane group supplies (q = qb+(m>>2), chunk (m&3)) and
              // receives (q = qb+0..3, d = db + lane%16)
              const int m_ = lane & 15;
              const int colhi_ = (lane >> 4) & 1;
              const int qb0 = qsub * 32 + mch * 16 + half * 8 + (m_ >> 2);
              const int c_r = dblk * 8 + colhi_ * 4 + (m_ & 3);
              const int g_r =
                  (c_r & 7) | ((((m_ >> 2) ^ (c_r >> 3)) & 3) << 3);
              const int a0 = qb0 * 256 + g_r * 8;
              auto* dob3 = (__attribute__((address_space(3))) char*)dotr;
              auto* qb3 = (__attribute__((address_space(3))) char*)qtr;
              typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4t;
              typedef __attribute__((address_space(3))) bf16x4t as3b4;
              bf16x4t d0v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(dob3 + a0));
              bf16x4t d1v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(dob3 + a0 + 1024));
              bf16x4t q0v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(qb3 + a0));
              bf16x4t q1v = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                  (as3b4*)(qb3 + a0 + 1024));
              dof = __builtin_shufflevector(d0v, d1v, 0, 1, 2, 3, 4, 5, 6, 7);
              qf = __builtin_shufflevector(q0v, q1v, 0, 1, 2, 3, 4, 5, 6, 7);
            } else {
              int trow = dblk * 32 + col;
              int colb = (qsub * 32 + mch * 16 + half * 8) * 2;
              dof = *reinterpret_cast<const bf16frag*>(
                  reinterpret_cast<const char*>(dotr) + trow * 128 + vswz(trow, colb));
              qf = *reinterpret_cast<const bf16frag*>(
                  reinterpret_cast<const char*>(qtr) + trow * 128 + vswz(trow, colb));
            }
            dv_acc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa2[mch], dof, dv_acc[dblk], 0, 0, 0);
            dk_acc[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da2[mch], qf, dk_acc[dblk], 0, 0, 0);
          }
        }
      }
      if constexpr (DBN > 0 && DBLATE) {
        // late placement: loads drain under the final barrier + next store
        if (it + 1 < itn) VH_DKV_LDNEXT(it + 1);
      }
      __syncthreads();
    }
  }
#undef VH_DKV_LDNEXT

  // combine the two q-subtile partials per kv slice through LDS, then store.
  // red layout: [slice 2][lane 64][64 floats] = 32 KB (reuses K/V space —
  // every wave has passed the loop-final barrier). Written out directly per
  // accumulator (no pointer indirection: an f32x16* into the accumulator
  // arrays sent all 128 accumulator VGPRs to scratch).
#define VH_DKVG_COMBINE(ACC, DST)                                             \
  {                                                                           \
    if (qsub == 1) {                                                          \
      float* out = red + (kvslice * 64 + lane) * 64;                          \
      _Pragma("unroll") for (int d = 0; d < 4; ++d)                           \
          _Pragma("unroll") for (int r = 0; r < 16; r += 4)                   \
              *reinterpret_cast<float4*>(out + d * 16 + r) = float4{          \
                  ACC[d][r], ACC[d][r + 1], ACC[d][r + 2], ACC[d][r + 3]};    \
    }                                                                         \
    __syncthreads();                                                          \
    if (qsub == 0) {                                                          \
      const float* in = red + (kvslice * 64 + lane) * 64;                     \
      _Pragma("unroll") for (int d = 0; d < 4; ++d)                           \
          _Pragma("unroll") for (int r = 0; r < 16; ++r) ACC[d][r] +=         \
          in[d * 16 + r];                                                     \
      _Pragma("unroll") for (int r = 0; r < 16; ++r) {                        \
        int kvr = (r & 3) + 8 * (r >> 2) + 4 * half;                          \
        int64_t kvg = kv0 + kvslice * 32 + kvr;                               \
        _Pragma("unroll") for (int d = 0; d < 4; ++d) DST[kvg * DH + d * 32 + \
                                                          col] =             \
            f2bf(ACC[d][r]);                                                  \
      }                                                                       \
    }                                                                         \
    __syncthreads();                                                          \
  }
  VH_DKVG_COMBINE(dv_acc, dVb)
  VH_DKVG_COMBINE(dk_acc, dKb)
#undef VH_DKVG_COMBINE
}

// TRQ: tr16 hardware-transpose K^T reads over the bank-permuted natural
// [32 kv][128 d] image (same scheme as dkv TR; see that comment).
// TRQ 2: the L16 latin-square image filled by glds (dkv TR2G scheme).
// TRQ 3: TRQ 2 + the WHOLE stage (krow/vrow kswz images + ktr L16) goes
// glds into DOUBLE buffers (24 KB spare LDS), pipelined one kv tile
// ahead with a single barrier per tile — the staging drain leaves the
// barrier path entirely.
template <bool DOC, int TRQ = 0>
__global__ __launch_bounds__(256, 2) void k_attn_bwd_dq(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, const bf16_t* __restrict__ dO,
    const float* __restrict__ delta, const float* __restrict__ lse2,
    bf16_t* __restrict__ dQ, const int* __restrict__ doc_start, int B, int Hq,
    int Hkv, int64_t S, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* krow = reinterpret_cast<bf16_t*>(smem);            // [32][128] 8 K
  bf16_t* vrow = reinterpret_cast<bf16_t*>(smem + 8192);     // [32][128] 8 K
  bf16_t* ktr = reinterpret_cast<bf16_t*>(smem + 16384);     // [128][32] 8 K
  // TRQ 3 double buffers: second set at +24 KiB (launch carves 48 KiB)

  const int qb = blockIdx.x;           // q block of 128 rows
  const int bh = blockIdx.y;
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;           // 32-row q slice
  const int half = lane >> 5;
  const int col = lane & 31;

  const bf16_t* Qb = Q + (((int64_t)b * Hq + hq) * S) * DH;
  const bf16_t* Kb = K + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* Vb = V + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* dOb = dO + (((int64_t)b * Hq + hq) * S) * DH;
  const float* delb = delta + ((int64_t)b * Hq + hq) * S;
  const float* lseb = lse2 + ((int64_t)b * Hq + hq) * S;
  bf16_t* dQb = dQ + (((int64_t)b * Hq + hq) * S) * DH;

  const int64_t q0b = (int64_t)qb * 128;
  const int64_t q_l = q0b + wave * 32 + col;   // this lane's q row

  // the wave's Q/dO rows and lse/delta are fixed for the whole kernel
  bf16frag qrow[8], dorow[8];
#pragma unroll
  for (int c = 0; c < 8; ++c) {
    qrow[c] = *reinterpret_cast<const bf16frag*>(Qb + q_l * DH + c * 16 + half * 8);
    dorow[c] = *reinterpret_cast<const bf16frag*>(dOb + q_l * DH + c * 16 + half * 8);
  }
  const float lse_l = lseb[q_l];
  const float del_l = delb[q_l];
  const float scale2 = scale * 1.4426950408889634f;

  int ds_l = 0, ds_wave = 0, kvt_begin = 0;
  if (DOC) {
    ds_l = doc_start[q_l];
    ds_wave = doc_start[q0b + wave * 32];
    kvt_begin = doc_start[q0b] / 32;
  }

  f32x16 dq4[4];
#pragma unroll
  for (int d = 0; d < 4; ++d) dq4[d] = f32x16{};

  const int kvtn = (int)((q0b + 128) / 32);
  // TRQ 3: glds one whole tile set (krow/vrow kswz + ktr L16) into buffer Bf
#define VH_DQ_STAGE(KVT2, BF)                                                  \
  {                                                                            \
    const int64_t kv0_ = (int64_t)(KVT2) * 32;                                 \
    char* kb_ = reinterpret_cast<char*>(smem) + (BF) * 24576;                  \
    _Pragma("unroll") for (int u = 0; u < 2; ++u) {                            \
      const int win_ = wave * 2 + u;                                           \
      const int row_ = win_ * 4 + (lane >> 4);                                 \
      const int c16_ = (lane & 15) ^ (row_ & 15);                              \
      const int pg_ = (((c16_ & 3) << 2) | (c16_ >> 2));                       \
      glds16a(Kb + (kv0_ + row_) * DH + c16_ * 8,                              \
              reinterpret_cast<bf16_t*>(kb_) + win_ * 512);                    \
      glds16a(Vb + (kv0_ + row_) * DH + c16_ * 8,                              \
              reinterpret_cast<bf16_t*>(kb_ + 8192) + win_ * 512);             \
      glds16a(Kb + (kv0_ + row_) * DH + pg_ * 8,                               \
              reinterpret_cast<bf16_t*>(kb_ + 16384) + win_ * 512);            \
    }                                                                          \
  }
  int cur3 = 0;
  if constexpr (TRQ == 3) {
    if (kvt_begin < kvtn) VH_DQ_STAGE(kvt_begin, 0);
    __syncthreads();
  }
  for (int kvt = kvt_begin; kvt < kvtn; ++kvt) {
    const int64_t kvt0 = (int64_t)kvt * 32;
    if constexpr (TRQ == 3) {
      char* base_ = reinterpret_cast<char*>(smem) + cur3 * 24576;
      krow = reinterpret_cast<bf16_t*>(base_);
      vrow = reinterpret_cast<bf16_t*>(base_ + 8192);
      ktr = reinterpret_cast<bf16_t*>(base_ + 16384);
      if (kvt + 1 < kvtn) VH_DQ_STAGE(kvt + 1, cur3 ^ 1);
    } else {
    // stage K/V rows [32][128] (2 passes) + K^T [128][32] (2 units each)
#pragma unroll
    for (int i = 0; i < 2; ++i) {
      int o = i * 4096 + tid * 16;
      int row = o >> 8;
      int colb = o & 255;
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(krow) + row * 256 + kswz(row, colb)) =
          *reinterpret_cast<const bf16x8*>(Kb + (kvt0 + row) * DH + (colb >> 1));
      *reinterpret_cast<bf16x8*>(
          reinterpret_cast<char*>(vrow) + row * 256 + kswz(row, colb)) =
          *reinterpret_cast<const bf16x8*>(Vb + (kvt0 + row) * DH + (colb >> 1));
    }
    if constexpr (TRQ == 2) {
      // glds fill of the L16 K^T image: 8 windows of 4 kv rows, 4 waves x 2
#pragma unroll
      for (int u = 0; u < 2; ++u) {
        const int win = wave * 2 + u;
        const int kvr = win * 4 + (lane >> 4);
        const int pp0 = (lane & 15) ^ (kvr & 15);
        const int pg = ((pp0 & 3) << 2) | (pp0 >> 2);
        glds16a(Kb + (kvt0 + kvr) * DH + pg * 8, ktr + win * 512);
      }
    } else
#pragma unroll
    for (int u = 0; u < 2; ++u) {
      int unit = tid + u * 256;
      int kv, d0;
      if constexpr (TRQ == 1) {
        kv = ((unit >> 2) & 7) | (((unit >> 5) & 3) << 3);
        d0 = (unit & 3) * 8 + ((unit >> 7) & 3) * 32;
      } else {
        kv = unit & 31;
        d0 = (unit >> 5) * 8;
      }
      bf16x8 v = *reinterpret_cast<const bf16x8*>(Kb + (kvt0 + kv) * DH + d0);
      if constexpr (TRQ == 1) {
        int c = d0 >> 2;
        int g = (c & 7) | ((((kv & 3) ^ (c >> 3)) & 3) << 3);
        *reinterpret_cast<bf16x8*>(
            reinterpret_cast<char*>(ktr) + kv * 256 + g * 8) = v;
      } else {
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int row = d0 + j;
          ktr[(row * 64 + qswz(row, kv * 2)) >> 1] = v.v[j];
        }
      }
    }
    __syncthreads();
    }

    const bool live = (kvt0 <= q0b + wave * 32 + 31) &&
                      (!DOC || kvt0 + 31 >= ds_wave);
    const bool diag = (kvt0 + 31 >= q0b + wave * 32);

    if (live) {
      // or1: C = [kv regs][q lanes]; s1 = mfma(K, Q)
      f32x16 s1 = f32x16{}, dp1 = f32x16{};
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        int colb = (c * 16 + half * 8) * 2;
        bf16frag kf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(krow) + col * 256 + kswz(col, colb));
        bf16frag vf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(vrow) + col * 256 + kswz(col, colb));
        s1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qrow[c], s1, 0, 0, 0);
        dp1 = __builtin_amdgcn_mfma_f32_32x32x16_bf16(vf, dorow[c], dp1, 0, 0, 0);
      }
      uint32_t dg1[8];
#pragma unroll
      for (int r = 0; r < 16; r += 2) {
        float g[2];
#pragma unroll
        for (int rr = 0; rr < 2; ++rr) {
          int r2 = r + rr;
          int kvl = (r2 & 3) + 8 * (r2 >> 2) + 4 * half;
          bool masked = diag && (kvt0 + kvl > q_l);
          if (DOC) masked = masked || (kvt0 + kvl < ds_l);
          float pp = masked ? 0.f : __builtin_exp2f(s1[r2] * scale2 - lse_l);
          g[rr] = pp * (dp1[r2] - del_l) * scale;
        }
        dg1[r >> 1] = (uint32_t)__builtin_bit_cast(uint16_t, (__bf16)g[0]) |
                      ((uint32_t)__builtin_bit_cast(uint16_t, (__bf16)g[1]) << 16);
      }
      bf16frag da1[2];
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
        uint32_t a0 = half ? dg1[4 * mch] : dg1[4 * mch + 2];
        uint32_t a1 = half ? dg1[4 * mch + 1] : dg1[4 * mch + 3];
        uint32_t b0 = swap32_u(a0, half);
        uint32_t b1 = swap32_u(a1, half);
        uint4 u{half ? b0 : dg1[4 * mch], half ? b1 : dg1[4 * mch + 1],
                half ? dg1[4 * mch + 2] : b0, half ? dg1[4 * mch + 3] : b1};
        da1[mch] = __builtin_bit_cast(bf16frag, u);
      }
      // dQ[q][d] += dS1^T(pack) x K^T-tile
#pragma unroll
      for (int dblk = 0; dblk < 4; ++dblk) {
#pragma unroll
        for (int mch = 0; mch < 2; ++mch) {
          bf16frag ktf;
          if constexpr (TRQ >= 2) {
            const int m_ = lane & 15;
            const int colhi_ = (lane >> 4) & 1;
            const int kvb0 = mch * 16 + half * 8 + (m_ >> 2);
            const int c_r = dblk * 8 + colhi_ * 4 + (m_ & 3);
            const int perm_ = (((c_r >> 1) & 3) << 2) | (c_r >> 3);
            const int a0 =
                kvb0 * 256 + ((kvb0 & 15) ^ perm_) * 16 + (c_r & 1) * 8;
            const int a1 = (kvb0 + 4) * 256 +
                           (((kvb0 + 4) & 15) ^ perm_) * 16 + (c_r & 1) * 8;
            auto* kb3 = (__attribute__((address_space(3))) char*)ktr;
            typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4t;
            typedef __attribute__((address_space(3))) bf16x4t as3b4;
            bf16x4t r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (as3b4*)(kb3 + a0));
            bf16x4t r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (as3b4*)(kb3 + a1));
            ktf = __builtin_shufflevector(r0, r1, 0, 1, 2, 3, 4, 5, 6, 7);
          } else if constexpr (TRQ == 1) {
            const int m_ = lane & 15;
            const int colhi_ = (lane >> 4) & 1;
            const int kvb0 = mch * 16 + half * 8 + (m_ >> 2);
            const int c_r = dblk * 8 + colhi_ * 4 + (m_ & 3);
            const int g_r = (c_r & 7) | ((((m_ >> 2) ^ (c_r >> 3)) & 3) << 3);
            auto* kb3 = (__attribute__((address_space(3))) char*)ktr;
            typedef __attribute__((ext_vector_type(4))) __bf16 bf16x4t;
            typedef __attribute__((address_space(3))) bf16x4t as3b4;
            bf16x4t r0 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (as3b4*)(kb3 + kvb0 * 256 + g_r * 8));
            bf16x4t r1 = __builtin_amdgcn_ds_read_tr16_b64_v4bf16(
                (as3b4*)(kb3 + kvb0 * 256 + g_r * 8 + 1024));
            ktf = __builtin_shufflevector(r0, r1, 0, 1, 2, 3, 4, 5, 6, 7);
          } else {
            int trow = dblk * 32 + col;
            int colb = (mch * 16 + half * 8) * 2;
            ktf = *reinterpret_cast<const bf16frag*>(
                reinterpret_cast<const char*>(ktr) + trow * 64 + qswz(trow, colb));
          }
          dq4[dblk] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(da1[mch], ktf, dq4[dblk], 0, 0, 0);
        }
      }
    }
    __syncthreads();
    cur3 ^= 1;
  }
#undef VH_DQ_STAGE

  // single-contributor store: this block covers every kv for its q rows
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int qr = (r & 3) + 8 * (r >> 2) + 4 * half;
    int64_t qg = q0b + wave * 32 + qr;
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      dQb[qg * DH + d * 32 + col] = f2bf(dq4[d][r]);
    }
  }
}

}  // namespace

extern "C" int vh_attn_bwd2_dkv4probe_bf16(const uint16_t* Q, const uint16_t* K,
                                           const uint16_t* V, const uint16_t* dO,
                                           const float* delta, const float* lse2,
                                           uint16_t* dK, uint16_t* dV, int B,
                                           int Hq, int Hkv, int64_t S,
                                           float scale, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % 128 == 0, "S %% 128 != 0");
  dim3 grid_kv((uint32_t)(S / 64), (uint32_t)(B * Hq));
  hipLaunchKernelGGL(k_attn_bwd_dkv4, grid_kv, dim3(256), 65536, s,
                     reinterpret_cast<const bf16_t*>(Q),
                     reinterpret_cast<const bf16_t*>(K),
                     reinterpret_cast<const bf16_t*>(V),
                     reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                     reinterpret_cast<bf16_t*>(dK),
                     reinterpret_cast<bf16_t*>(dV), B, Hq, Hkv, S, scale);
  VH_HIP(hipGetLastError());
  return 0;
}

/* Dispatched split backward: GQA-folded dkv (dK/dV [B,Hkv,S,D] written
 * once — no per-Q-head intermediates, no host group sum) with the TR2G
 * staging (tr16 hardware-transpose reads over the L16 latin-square image,
 * A-frags from LDS, image filled by glds direct-to-LDS with per-lane
 * permuted sources) — bit-identical to v6, measured -47% cumulative
 * (tests/gpu_dkv_tr.py) — + per-Q-head dq.
 * doc_start/doc_end (nullable, B == 1) select the packed-varlen
 * block-diagonal causal mask. */
extern "C" int vh_attn_bwd2_bf16(const uint16_t* Q, const uint16_t* K,
                                 const uint16_t* V, const uint16_t* dO,
                                 const float* delta, const float* lse2,
                                 uint16_t* dQ, uint16_t* dK, uint16_t* dV,
                                 int B, int Hq, int Hkv, int64_t S, float scale,
                                 const int32_t* doc_start,
                                 const int32_t* doc_end, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % 128 == 0, "S %% 128 != 0");
  VH_CHECK(Hq % Hkv == 0, "Hq %% Hkv != 0");
  VH_CHECK((doc_start == nullptr) == (doc_end == nullptr),
           "doc_start/doc_end must be passed together");
  VH_CHECK(doc_start == nullptr || B == 1, "varlen requires packed B == 1");
  dim3 grid_kv((uint32_t)(S / 64), (uint32_t)(B * Hkv));
  dim3 grid_q((uint32_t)(S / 128), (uint32_t)(B * Hq));
  if (doc_start) {
    hipLaunchKernelGGL((k_attn_bwd_dkv_g<true, 0, 0, false, 3>), grid_kv,
                       dim3(256), 65536, s,
                       reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dK),
                       reinterpret_cast<bf16_t*>(dV), doc_start, doc_end, B,
                       Hq, Hkv, S, scale);
    VH_HIP(hipGetLastError());
    hipLaunchKernelGGL((k_attn_bwd_dq<true, 3>), grid_q, dim3(256), 49152, s,
                       reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dQ), doc_start, B, Hq, Hkv,
                       S, scale);
  } else {
    hipLaunchKernelGGL((k_attn_bwd_dkv_g<false, 0, 0, false, 3>), grid_kv,
                       dim3(256), 65536, s,
                       reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dK),
                       reinterpret_cast<bf16_t*>(dV), nullptr, nullptr, B, Hq,
                       Hkv, S, scale);
    VH_HIP(hipGetLastError());
    hipLaunchKernelGGL((k_attn_bwd_dq<false, 3>), grid_q, dim3(256), 49152, s,
                       reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dQ), nullptr, B, Hq, Hkv, S,
                       scale);
  }
  VH_HIP(hipGetLastError());
  return 0;
}

/* probe: the dispatched dkv v6 at prefetch depths 0/4/8 for on-box A/B
 * (dK/dV [B,Hkv,S,D], no varlen). */
extern "C" int vh_attn_bwd2_dkv6probe_bf16(const uint16_t* Q, const uint16_t* K,
                                           const uint16_t* V, const uint16_t* dO,
                                           const float* delta, const float* lse2,
                                           uint16_t* dK, uint16_t* dV, int B,
                                           int Hq, int Hkv, int64_t S,
                                           float scale, int pref, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % 128 == 0, "S %% 128 != 0");
  dim3 grid((uint32_t)(S / 64), (uint32_t)(B * Hkv));
#define VH_DKV6(P_)                                                           \
  hipLaunchKernelGGL((k_attn_bwd_dkv_g<false, P_>), grid, dim3(256), 65536,   \
                     s, reinterpret_cast<const bf16_t*>(Q),                   \
                     reinterpret_cast<const bf16_t*>(K),                      \
                     reinterpret_cast<const bf16_t*>(V),                      \
                     reinterpret_cast<const bf16_t*>(dO), delta, lse2,        \
                     reinterpret_cast<bf16_t*>(dK),                           \
                     reinterpret_cast<bf16_t*>(dV), nullptr, nullptr, B, Hq,  \
                     Hkv, S, scale)
#define VH_DKVDB(N_, L_)                                                      \
  hipLaunchKernelGGL((k_attn_bwd_dkv_g<false, 0, N_, L_>), grid, dim3(256),   \
                     65536, s, reinterpret_cast<const bf16_t*>(Q),            \
                     reinterpret_cast<const bf16_t*>(K),                      \
                     reinterpret_cast<const bf16_t*>(V),                      \
                     reinterpret_cast<const bf16_t*>(dO), delta, lse2,        \
                     reinterpret_cast<bf16_t*>(dK),                           \
                     reinterpret_cast<bf16_t*>(dV), nullptr, nullptr, B, Hq,  \
                     Hkv, S, scale)
  if (pref == 22)
    hipLaunchKernelGGL((k_attn_bwd_dkv_g<false, 0, 0, false, 3>), grid,
                       dim3(256), 65536, s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dK),
                       reinterpret_cast<bf16_t*>(dV), nullptr, nullptr, B, Hq,
                       Hkv, S, scale);
  else if (pref == 21)
    hipLaunchKernelGGL((k_attn_bwd_dkv_g<false, 0, 0, false, 2>), grid,
                       dim3(256), 65536, s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dK),
                       reinterpret_cast<bf16_t*>(dV), nullptr, nullptr, B, Hq,
                       Hkv, S, scale);
  else if (pref == 20)
    hipLaunchKernelGGL((k_attn_bwd_dkv_g<false, 0, 0, false, 1>), grid,
                       dim3(256), 65536, s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dK),
                       reinterpret_cast<bf16_t*>(dV), nullptr, nullptr, B, Hq,
                       Hkv, S, scale);
  else if (pref == 16) VH_DKVDB(2, false);
  else if (pref == 17) VH_DKVDB(4, false);
  else if (pref == 18) VH_DKVDB(4, true);
  else if (pref == 19) VH_DKVDB(2, true);
  else if (pref == 0) VH_DKV6(0);
  else if (pref == 8) VH_DKV6(8);
  else VH_DKV6(4);
#undef VH_DKV6
#undef VH_DKVDB
  VH_HIP(hipGetLastError());
  return 0;
}

/* probe: dq with the TRQ (tr16) K^T image vs the dispatched layout. */
extern "C" int vh_attn_bwd2_dqprobe_bf16(const uint16_t* Q, const uint16_t* K,
                                         const uint16_t* V, const uint16_t* dO,
                                         const float* delta, const float* lse2,
                                         uint16_t* dQ, int B, int Hq, int Hkv,
                                         int64_t S, float scale, int mode,
                                         void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % 128 == 0, "S %% 128 != 0");
  dim3 grid((uint32_t)(S / 128), (uint32_t)(B * Hq));
  if (mode == 22)
    hipLaunchKernelGGL((k_attn_bwd_dq<false, 3>), grid, dim3(256), 49152,
                       s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dQ), nullptr, B, Hq, Hkv, S,
                       scale);
  else if (mode == 21)
    hipLaunchKernelGGL((k_attn_bwd_dq<false, 2>), grid, dim3(256), 24576,
                       s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dQ), nullptr, B, Hq, Hkv, S,
                       scale);
  else if (mode == 20)
    hipLaunchKernelGGL((k_attn_bwd_dq<false, 1>), grid, dim3(256), 24576,
                       s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dQ), nullptr, B, Hq, Hkv, S,
                       scale);
  else
    hipLaunchKernelGGL((k_attn_bwd_dq<false, 0>), grid, dim3(256), 24576,
                       s, reinterpret_cast<const bf16_t*>(Q),
                       reinterpret_cast<const bf16_t*>(K),
                       reinterpret_cast<const bf16_t*>(V),
                       reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                       reinterpret_cast<bf16_t*>(dQ), nullptr, B, Hq, Hkv, S,
                       scale);
  VH_HIP(hipGetLastError());
  return 0;
}

/* probe: the round-1 per-Q-head dkv (v2) kept for on-box A/B against the
 * GQA-folded v6 (dK/dV here are per-Q-head [B,Hq,S,D]). */
extern "C" int vh_attn_bwd2_dkv2probe_bf16(const uint16_t* Q, const uint16_t* K,
                                           const uint16_t* V, const uint16_t* dO,
                                           const float* delta, const float* lse2,
                                           uint16_t* dK, uint16_t* dV, int B,
                                           int Hq, int Hkv, int64_t S,
                                           float scale, void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % 128 == 0, "S %% 128 != 0");
  dim3 grid((uint32_t)(S / 128), (uint32_t)(B * Hq));
  hipLaunchKernelGGL(k_attn_bwd_dkv, grid, dim3(256), 81920, s,
                     reinterpret_cast<const bf16_t*>(Q),
                     reinterpret_cast<const bf16_t*>(K),
                     reinterpret_cast<const bf16_t*>(V),
                     reinterpret_cast<const bf16_t*>(dO), delta, lse2,
                     reinterpret_cast<bf16_t*>(dK),
                     reinterpret_cast<bf16_t*>(dV), B, Hq, Hkv, S, scale);
  VH_HIP(hipGetLastError());
  return 0;
}
