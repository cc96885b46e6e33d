// Flash attention forward for gfx950 (bf16, causal, GQA, D=128).
//
// Replaces the external flash_attn wheel the reference delegates to
// (ref ops/kernels/attention/flash.py:153-301) for the packed causal path.
//
// Structure (guide cdna_hip_programming.md Appendix B, swapped-QK^T design):
//   grid (S/128, B*Hq); block = 4 waves, wave w owns 32 q rows.
//   Per wave: Q held in registers; per 64-kv tile (staged in LDS):
//     two 32-kv subtiles, each:
//       S^T = mfma_32x32x16(A=K, B=Q)  -> lane owns col q = lane&31,
//             16 f32 scores over the subtile's kv rows (half per lane pair);
//       online softmax per q-col (running m, l as lane scalars; the
//             cross-half reduce is one shfl_xor(32));
//       P packed to bf16 (quad exchange via shfl_xor) ->
//       O += mfma(A=P^T, B=V^T) with V staged TRANSPOSED in LDS.
//   Epilogue: O rows divided by l via 16 lane-broadcasts, bf16 store;
//   LSE = m + log(l) saved fp32 for the backward.
//
// K tile LDS rows are 256 B -> slot map (row&15)<<4 is conflict-free for the
// b128 fragment reads; V^T rows are 128 B -> the gemm swizzle applies.

#include "vh_common.h"

namespace {

constexpr int QB = 128;   // q rows per block
constexpr int WQ = 32;    // q rows per wave
constexpr int KB = 64;    // kv rows per tile
constexpr int DH = 128;   // head dim

using bf16frag = __attribute__((ext_vector_type(8))) __bf16;
using f32x16 = __attribute__((ext_vector_type(16))) float;

__device__ __forceinline__ void glds16a(const bf16_t* g, bf16_t* l) {
  __builtin_amdgcn_global_load_lds(
      (const __attribute__((address_space(1))) unsigned int*)g,
      (__attribute__((address_space(3))) unsigned int*)l, 16, 0, 0);
}

// K tile [64][128] bf16 (256-B rows): conflict-free slot map
__device__ __forceinline__ int kswz(int row, int colb) {
  return colb ^ ((row & 15) << 4);
}
// V^T tile [128][64] bf16 (128-B rows): the gemm swizzle
__device__ __forceinline__ int vswz(int row, int colb) {
  return colb ^ ((((row >> 1) ^ (row >> 3)) & 7) << 4);
}

__device__ __forceinline__ float xor32(float v) {
  return __shfl_xor(v, 32, 64);
}

__global__ __launch_bounds__(256, 2) void k_attn_fwd(
    const bf16_t* __restrict__ Q, const bf16_t* __restrict__ K,
    const bf16_t* __restrict__ V, bf16_t* __restrict__ O,
    float* __restrict__ LSE, int B, int Hq, int Hkv, int64_t S, float scale) {
  extern __shared__ __attribute__((aligned(16))) char smem[];
  bf16_t* kt = reinterpret_cast<bf16_t*>(smem);            // [64][128] 16 KiB
  bf16_t* vt = reinterpret_cast<bf16_t*>(smem + 16384);    // [128][64] 16 KiB
  bf16_t* qt = reinterpret_cast<bf16_t*>(smem + 32768);    // 4 x [32][128] 32 KiB

  const int qb = blockIdx.x;
  const int bh = blockIdx.y;           // b * Hq + hq
  const int b = bh / Hq;
  const int hq = bh % Hq;
  const int hkv = hq / (Hq / Hkv);

  const int tid = threadIdx.x;
  const int lane = tid & 63;
  const int wave = tid >> 6;
  const int half = lane >> 5;          // 0 | 1
  const int col = lane & 31;           // this lane's q (within the wave) for S,
                                       // and its d (within a 32-block) for O

  const bf16_t* Qb = Q + (((int64_t)b * Hq + hq) * S) * DH;
  const bf16_t* Kb = K + (((int64_t)b * Hkv + hkv) * S) * DH;
  const bf16_t* Vb = V + (((int64_t)b * Hkv + hkv) * S) * DH;
  bf16_t* Ob = O + (((int64_t)b * Hq + hq) * S) * DH;
  float* Lb = LSE + ((int64_t)b * Hq + hq) * S;

  const int64_t q_global = (int64_t)qb * QB + wave * WQ + col;

  // ---- stage this wave's 32 Q rows into LDS once (glds, kswz image)
  {
    bf16_t* qw = qt + wave * 4096;  // 8 KiB per wave
#pragma unroll
    for (int i = 0; i < 8; ++i) {
      int base = i * 1024;
      int o = base + lane * 16;
      int row = o >> 8;
      int colb = o & 255;
      const bf16_t* g = Qb + ((int64_t)qb * QB + wave * WQ + row) * DH +
                        (kswz(row, colb) >> 1);
      glds16a(g, qw + (base >> 1));
    }
  }

  f32x16 oacc[4];
#pragma unroll
  for (int d = 0; d < 4; ++d) oacc[d] = f32x16{};
  float m_run = -1e30f;
  float l_run = 0.f;

  // persistent per-lane staging addresses (advance by KB*DH per tile)
  const bf16_t* ksrc[4];
  int klds[4];
#pragma unroll
  for (int i = 0; i < 4; ++i) {
    int base = i * 4096 + wave * 1024;
    int o = base + lane * 16;
    int row = o >> 8;
    int colb = o & 255;
    ksrc[i] = Kb + (int64_t)row * DH + (kswz(row, colb) >> 1);
    klds[i] = base >> 1;
  }
  const int v_kv = tid & 63;
  const int v_d0 = (tid >> 6) * 8;     // 4 d-chunks per thread (stride 32 rows)
  const bf16_t* vsrc = Vb + (int64_t)v_kv * DH + v_d0;

  const int t_max = (int)(((int64_t)qb * QB + QB - 1) / KB);  // inclusive
  for (int t = 0; t <= t_max; ++t) {
    // ---- stage K tile via glds (4 instructions; swizzled source)
    {
#pragma unroll
      for (int i = 0; i < 4; ++i) glds16a(ksrc[i], kt + klds[i]);
      // ---- stage V transposed: thread t loads 16 B (8 d at one kv) and
      // scatters 8 2-B writes into [d][kv]; 4 units serialised to bound
      // register liveness
#pragma unroll
      for (int u = 0; u < 4; ++u) {
        int d0 = v_d0 + u * 32;
        bf16x8 v = *reinterpret_cast<const bf16x8*>(vsrc + u * 32);
#pragma unroll
        for (int j = 0; j < 8; ++j) {
          int row = d0 + j;
          vt[(row * 128 + vswz(row, v_kv * 2)) >> 1] = v.v[j];
        }
      }
#pragma unroll
      for (int i = 0; i < 4; ++i) ksrc[i] += KB * DH;
      vsrc += KB * DH;
    }
    __syncthreads();

    // diagonal tile for this wave: some kv in the tile can exceed some q
    const bool diag = ((int64_t)(t + 1) * KB) > ((int64_t)qb * QB + wave * WQ);

#pragma unroll
    for (int sub = 0; sub < 2; ++sub) {
      // ---- S^T = K·Q^T over 8 d-chunks
      f32x16 sacc = f32x16{};
      const char* qw = reinterpret_cast<const char*>(qt) + wave * 8192;
#pragma unroll
      for (int c = 0; c < 8; ++c) {
        int krow = sub * 32 + col;
        int colb = (c * 16 + half * 8) * 2;
        bf16frag kf = *reinterpret_cast<const bf16frag*>(
            reinterpret_cast<const char*>(kt) + krow * 256 + kswz(krow, colb));
        bf16frag qf = *reinterpret_cast<const bf16frag*>(
            qw + col * 256 + kswz(col, colb));
        sacc = __builtin_amdgcn_mfma_f32_32x32x16_bf16(kf, qf, sacc, 0, 0, 0);
      }

      // ---- scale + causal mask; per-lane row max over its 16 scores
      float p[16];
      float mt = -INFINITY;
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        float s = sacc[r] * scale;
        if (diag) {
          int kv_in = sub * 32 + (r & 3) + 8 * (r >> 2) + 4 * half;
          int64_t kv_global = (int64_t)t * KB + kv_in;
          if (kv_global > q_global) s = -INFINITY;
        }
        p[r] = s;
        mt = fmaxf(mt, s);
      }
      mt = fmaxf(mt, xor32(mt));
      float m_new = fmaxf(m_run, mt);
      float alpha = __expf(m_run - m_new);
      // ---- exponentiate + partial row sum, packing pairs to bf16 on the
      // fly (frees the fp32 score registers early)
      float psum = 0.f;
      uint32_t pk[8];  // 4 quads x 2 dwords of packed bf16 pairs
#pragma unroll
      for (int r = 0; r < 16; r += 2) {
        float e0 = __expf(p[r] - m_new);
        float e1 = __expf(p[r + 1] - m_new);
        psum += e0 + e1;
        pk[r >> 1] = (uint32_t)f2bf(e0) | ((uint32_t)f2bf(e1) << 16);
      }
      psum += xor32(psum);
      l_run = l_run * alpha + psum;
      m_run = m_new;

      // ---- rescale O by alpha (alpha is per-q = per-lane col: O's q lives
      // in REGS, so fetch alpha per reg-row via lane broadcast)
      // alpha for q-row qr is held by lanes with col == qr (both halves).
#pragma unroll
      for (int r = 0; r < 16; ++r) {
        int qr = (r & 3) + 8 * (r >> 2) + 4 * half;
        // alpha is symmetric across halves; width-32 shfl broadcasts within
        // this lane's half from the lane whose col == qr
        float a_r = __shfl(alpha, qr, 32);
#pragma unroll
        for (int d = 0; d < 4; ++d) oacc[d][r] *= a_r;
      }

      // ---- exchange packed quads for the P^T A-fragment. Chunk m covers
      // kv 16m..16m+15: half0 = [own quad 2m | partner quad 2m],
      // half1 = [partner quad 2m+1 | own quad 2m+1]. Each lane only needs
      // the partner's MATCHING quad: exchange its own counterpart (the
      // shuffle is symmetric, both sides send the quad the other needs:
      // half0 sends quad 2m+1's slot? no — both halves hold quads 0..3 of
      // DIFFERENT kv sets; the partner's quad with the SAME index is the
      // one required, so a plain xor-32 shuffle of quad pairs suffices).
      bf16frag pa[2];
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
        // convergent exchange: each half sends the quad the partner needs
        // (half0 sends quad 2m, half1 sends quad 2m+1); select BEFORE the
        // collective so every lane executes the same shuffles.
        // half0 needs the partner's quad 2m (so half1 sends quad 2m);
        // half1 needs the partner's quad 2m+1 (so half0 sends quad 2m+1)
        uint32_t s0 = half ? pk[4 * mch] : pk[4 * mch + 2];
        uint32_t s1 = half ? pk[4 * mch + 1] : pk[4 * mch + 3];
        uint32_t o0 = (uint32_t)__shfl_xor((int)s0, 32, 64);
        uint32_t o1 = (uint32_t)__shfl_xor((int)s1, 32, 64);
        // half0 frag = [own 2m | partner 2m]; half1 = [partner 2m+1 | own 2m+1]
        uint32_t w0 = half ? o0 : pk[4 * mch];
        uint32_t w1 = half ? o1 : pk[4 * mch + 1];
        uint32_t w2 = half ? pk[4 * mch + 2] : o0;
        uint32_t w3 = half ? pk[4 * mch + 3] : o1;
        uint4 u{w0, w1, w2, w3};
        pa[mch] = __builtin_bit_cast(bf16frag, u);
      }

      // ---- O += P^T · V  (A = P^T fragments, B = V^T tile reads)
#pragma unroll
      for (int mch = 0; mch < 2; ++mch) {
#pragma unroll
        for (int d = 0; d < 4; ++d) {
          int vrow = d * 32 + col;
          int colb = (sub * 32 + mch * 16 + half * 8) * 2;
          bf16frag vf = *reinterpret_cast<const bf16frag*>(
              reinterpret_cast<const char*>(vt) + vrow * 128 + vswz(vrow, colb));
          oacc[d] = __builtin_amdgcn_mfma_f32_32x32x16_bf16(pa[mch], vf, oacc[d], 0, 0, 0);
        }
      }
    }
    __syncthreads();
  }

  // ---- epilogue: O rows / l; store bf16; LSE
  // l for q-row qr is on lanes with col == qr; O reg r belongs to q-row
  // qr = (r&3)+8*(r>>2)+4*half, col d = col + 32*dblk.
#pragma unroll
  for (int r = 0; r < 16; ++r) {
    int qr = (r & 3) + 8 * (r >> 2) + 4 * half;
    float l_r = __shfl(l_run, qr, 32);
    float inv = 1.0f / l_r;
#pragma unroll
    for (int d = 0; d < 4; ++d) {
      int64_t qg = (int64_t)qb * QB + wave * WQ + qr;
      Ob[qg * DH + d * 32 + col] = f2bf(oacc[d][r] * inv);
    }
  }
  if (half == 0) {
    Lb[q_global] = m_run + __logf(l_run);
  }
}

}  // namespace

extern "C" int vh_attn_fwd_bf16(const uint16_t* Q, const uint16_t* K,
                                const uint16_t* V, uint16_t* O, float* LSE,
                                int B, int Hq, int Hkv, int64_t S, float scale,
                                void* stream) {
  hipStream_t s = reinterpret_cast<hipStream_t>(stream);
  VH_CHECK(S % QB == 0, "S %% 128 != 0 (pad the sequence)");
  VH_CHECK(Hq % Hkv == 0, "Hq %% Hkv != 0");
  dim3 grid((uint32_t)(S / QB), (uint32_t)(B * Hq));
  hipLaunchKernelGGL(k_attn_fwd, grid, dim3(256), 65536, s,
                     reinterpret_cast<const bf16_t*>(Q),
                     reinterpret_cast<const bf16_t*>(K),
                     reinterpret_cast<const bf16_t*>(V),
                     reinterpret_cast<bf16_t*>(O), LSE, B, Hq, Hkv, S, scale);
  VH_HIP(hipGetLastError());
  return 0;
}
