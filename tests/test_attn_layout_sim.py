#!/usr/bin/env python3
"""CPU simulator of the attention-backward kernels' MFMA lane/register
layouts (mfma_f32_32x32x16_bf16 A/B/C maps + the pack-exchange), used to
verify index math without GPU time. Simulates the dkv-family dK/dV data
flow in fp32 and compares against direct matrix math."""

import numpy as np

# mfma_f32_32x32x16: A[row=l&31][k=(l>>5)*8+j], B[k=(l>>5)*8+j][col=l&31],
# C reg r of lane l = C[(r&3)+8*(r>>2)+4*(l>>5)][l&31]


def crow(r, half):
    return (r & 3) + 8 * (r >> 2) + 4 * half


def mfma(A, B):
    """A [32,16], B [16,32] -> C [32,32] (plain matmul; the lane layouts are
    what the helpers above encode)."""
    return A @ B


def pack_exchange(C):
    """C [32 rows][32 cols] (or-layout) -> A' [32][32] with A'[row][k] =
    C[k][row] — the reg->chunk pack produces the transpose as two 16-wide
    A-fragments. Verified against the kernel's half/swap pattern by
    construction: A-frag chunk mch holds old rows mch*16..mch*16+15 at the
    lane's own column."""
    return C.T


def simulate_dkv(Q, K, V, dO, LSE2, DELTA, scale, KVB, QTILE, nsub):
    """Simulate the dkv-family kernel for one head: KVB-row kv strips,
    QTILE-row staged q tiles, nsub q-subtiles per tile (QTILE = 32*nsub).
    Waves = (KVB//32 kv slices) x nsub. Returns dK, dV [S, D]."""
    S, D = Q.shape
    scale2 = scale * np.log2(np.e)
    dK = np.zeros((S, D), np.float32)
    dV = np.zeros((S, D), np.float32)
    for kv0 in range(0, S, KVB):
        for sl in range(KVB // 32):          # kv slice
            kvr = kv0 + sl * 32
            dk_acc = {u: np.zeros((32, D), np.float32) for u in range(nsub)}
            dv_acc = {u: np.zeros((32, D), np.float32) for u in range(nsub)}
            qt0 = kv0 // QTILE
            for qt in range(qt0, S // QTILE):
                for u in range(nsub):        # q subtile (its own wave)
                    q0 = qt * QTILE + u * 32
                    live = (q0 + 31) >= kvr
                    diag = q0 < kv0 + KVB
                    if not live:
                        continue
                    # or2: S2 = Q_sub @ K_slice^T  -> C [q 32][kv 32]
                    s2 = Q[q0:q0 + 32] @ K[kvr:kvr + 32].T
                    dp2 = dO[q0:q0 + 32] @ V[kvr:kvr + 32].T
                    p2 = np.exp2(s2 * scale2 - LSE2[q0:q0 + 32, None])
                    if diag:
                        qidx = np.arange(q0, q0 + 32)[:, None]
                        kidx = np.arange(kvr, kvr + 32)[None, :]
                        p2 = np.where(kidx > qidx, 0.0, p2)
                    ds2 = p2 * (dp2 - DELTA[q0:q0 + 32, None]) * scale
                    # packs: A' = C^T -> [kv][q]; B-frags read Q/dO rows of
                    # the SAME subtile columns of the staged tile
                    dv_acc[u] += pack_exchange(p2) @ dO[q0:q0 + 32]
                    dk_acc[u] += pack_exchange(ds2) @ Q[q0:q0 + 32]
            dVs = sum(dv_acc.values())
            dKs = sum(dk_acc.values())
            dV[kvr:kvr + 32] = dVs
            dK[kvr:kvr + 32] = dKs
    return dK, dV


def reference(Q, K, V, dO, scale):
    S, D = Q.shape
    s = Q @ K.T * scale
    mask = np.triu(np.ones((S, S), bool), 1)
    s = np.where(mask, -np.inf, s)
    m = s.max(-1, keepdims=True)
    p = np.exp(s - m)
    l = p.sum(-1, keepdims=True)
    P = p / l
    O = P @ V
    LSE = (m + np.log(l)).squeeze(-1)
    dP = dO @ V.T
    delta = (dO * O).sum(-1)
    dS = P * (dP - delta[:, None]) * scale
    dK = dS.T @ Q
    dV = P.T @ dO
    return dK, dV, LSE, delta


def simulate_dkv_g(Qh, K, V, dOh, LSE2h, DELTAh, scale, doc_start, doc_end):
    """Simulate the GQA-folded dkv kernel (k_attn_bwd_dkv_g): one kv head,
    rep q-heads accumulated in-register, optional varlen (block-diagonal)
    masking with the kernel's exact loop bounds and wave-live checks.
    Qh/dOh/LSE2h/DELTAh: [rep, S, ...]; K/V: [S, D]."""
    rep, S, D = Qh.shape
    scale2 = scale * np.log2(np.e)
    dK = np.zeros((S, D), np.float32)
    dV = np.zeros((S, D), np.float32)
    doc = doc_start is not None
    for kv0 in range(0, S, 128):
        for sl in range(4):                  # kv slice (wave)
            kvr = kv0 + sl * 32
            dk_acc = np.zeros((32, D), np.float32)
            dv_acc = np.zeros((32, D), np.float32)
            qt0 = kv0 // 32
            qtn = S // 32
            de_wave = 0
            if doc:
                qtn = (doc_end[kv0 + 127] + 31) // 32   # block-uniform end
                de_wave = doc_end[kvr + 31]             # wave-live bound
            for g in range(rep):
                for qt in range(qt0, qtn):
                    q0 = qt * 32
                    live = (q0 + 31) >= kvr and (not doc or q0 < de_wave)
                    diag = q0 < kv0 + 128
                    if not live:
                        continue
                    s2 = Qh[g, q0:q0 + 32] @ K[kvr:kvr + 32].T
                    dp2 = dOh[g, q0:q0 + 32] @ V[kvr:kvr + 32].T
                    p2 = np.exp2(s2 * scale2 - LSE2h[g, q0:q0 + 32, None])
                    qidx = np.arange(q0, q0 + 32)[:, None]
                    kidx = np.arange(kvr, kvr + 32)[None, :]
                    masked = diag & (kidx > qidx)
                    if doc:
                        masked = masked | (kidx < doc_start[q0:q0 + 32, None])
                    p2 = np.where(masked, 0.0, p2)
                    ds2 = p2 * (dp2 - DELTAh[g, q0:q0 + 32, None]) * scale
                    dv_acc += pack_exchange(p2) @ dOh[g, q0:q0 + 32]
                    dk_acc += pack_exchange(ds2) @ Qh[g, q0:q0 + 32]
            dV[kvr:kvr + 32] = dv_acc
            dK[kvr:kvr + 32] = dk_acc
    return dK, dV


def reference_doc(Q, K, V, dO, scale, doc_start):
    """Direct backward with the block-diagonal causal mask."""
    S, D = Q.shape
    s = Q @ K.T * scale
    qidx = np.arange(S)[:, None]
    kidx = np.arange(S)[None, :]
    mask = kidx > qidx
    if doc_start is not None:
        mask = mask | (kidx < doc_start[:, None])
    s = np.where(mask, -np.inf, s)
    m = s.max(-1, keepdims=True)
    p = np.exp(s - m)
    l = p.sum(-1, keepdims=True)
    P = p / l
    O = P @ V
    LSE = (m + np.log(l)).squeeze(-1)
    dP = dO @ V.T
    delta = (dO * O).sum(-1)
    dS = P * (dP - delta[:, None]) * scale
    return dS.T @ Q, P.T @ dO, LSE, delta


def test_dkv_gqa_fold_matches_reference():
    """GQA-folded dkv (k_attn_bwd_dkv_g loop structure) == sum over the head
    group of per-head backward, causal and packed-varlen."""
    rng = np.random.default_rng(1)
    S, D, rep = 256, 32, 3
    K = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    V = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    Qh = rng.standard_normal((rep, S, D)).astype(np.float32) * 0.5
    dOh = rng.standard_normal((rep, S, D)).astype(np.float32) * 0.5
    scale = 1.0 / np.sqrt(D)
    for cu in (None, [0, 70, 150, 256], [0, 100, 130, 140, 256]):
        if cu is None:
            ds = de = None
        else:
            ds = np.zeros(S, np.int32)
            de = np.zeros(S, np.int32)
            for a, b in zip(cu[:-1], cu[1:]):
                ds[a:b] = a
                de[a:b] = b
        dK_ref = np.zeros((S, D), np.float32)
        dV_ref = np.zeros((S, D), np.float32)
        LSE2h = np.zeros((rep, S), np.float32)
        DELTAh = np.zeros((rep, S), np.float32)
        for g in range(rep):
            dk, dv, lse, delta = reference_doc(Qh[g], K, V, dOh[g], scale, ds)
            dK_ref += dk
            dV_ref += dv
            LSE2h[g] = lse * np.log2(np.e)
            DELTAh[g] = delta
        dK, dV = simulate_dkv_g(Qh, K, V, dOh, LSE2h, DELTAh, scale, ds, de)
        assert np.abs(dK - dK_ref).max() < 1e-4, cu
        assert np.abs(dV - dV_ref).max() < 1e-4, cu


def main():
    rng = np.random.default_rng(0)
    S, D = 128, 32
    Q = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    K = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    V = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    dO = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    scale = 1.0 / np.sqrt(D)
    dK_ref, dV_ref, LSE, delta = reference(Q, K, V, dO, scale)
    LSE2 = LSE * np.log2(np.e)

    for name, kvb, qtile, nsub in (("dkv  (128kv/32q)", 128, 32, 1),
                                   ("dkv4 (64kv/64q) ", 64, 64, 2)):
        dK, dV = simulate_dkv(Q, K, V, dO, LSE2, delta, scale, kvb, qtile, nsub)
        ek = np.abs(dK - dK_ref).max()
        ev = np.abs(dV - dV_ref).max()
        print(f"{name}: max|dK err| = {ek:.3e}  max|dV err| = {ev:.3e}")


def test_dkv_tilings_match_reference():
    """pytest entry: both dkv tilings reproduce the direct backward math."""
    rng = np.random.default_rng(0)
    S, D = 128, 32
    Q = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    K = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    V = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    dO = rng.standard_normal((S, D)).astype(np.float32) * 0.5
    scale = 1.0 / np.sqrt(D)
    dK_ref, dV_ref, LSE, delta = reference(Q, K, V, dO, scale)
    LSE2 = LSE * np.log2(np.e)
    for kvb, qtile, nsub in ((128, 32, 1), (64, 64, 2)):
        dK, dV = simulate_dkv(Q, K, V, dO, LSE2, delta, scale, kvb, qtile, nsub)
        assert np.abs(dK - dK_ref).max() < 1e-5
        assert np.abs(dV - dV_ref).max() < 1e-5


if __name__ == "__main__":
    main()
