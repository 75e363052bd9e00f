"""Lane-exact NumPy simulation of the round-2 attention forward
(`attn_fwd2` in ops/csrc/attention.hip): 8 waves x 32 q-rows, MFMA
32x32x16 with swapped QK^T (S^T = K Q^T), per-lane online softmax,
P^T fragments assembled in-register via cvt_pk + permlane32_swap, and
V consumed from a row-major LDS image through ds_read_b64_tr_b16.

The tr16 / permlane semantics simulated here are the MEASURED gfx950
behavior (tools/probe_tr16.hip, profiles/r02 notes), not guessed docs:

* ds_read_b64_tr_b16: per 16-lane group g, out lane l element j is the
  (l&3)-th element of the 4-element read issued by lane
  16g + 4j + ((l>>2)&3).
* permlane32_swap(a, b) -> (a.lo | b.lo, a.hi | b.hi) by half-waves.
* mfma_f32_32x32x16_bf16 layouts: A[m][k] lane l holds
  A[l&31][(l>>5)*8 + j]; B[k][n] lane l holds B[(l>>5)*8 + j][l&31];
  C[m][n] lane l reg r holds C[(r&3) + 8*(r>>2) + 4*(l>>5)][l&31].

Every helper mirrors one hardware primitive so the HIP kernel can be a
line-for-line transcription.
"""
import numpy as np
import pytest

WAVE = 64
QBLK2 = 32     # q rows per wave
KVBLK = 64
D = 128
VRS = 160      # V LDS row stride (elements) — conflict-free for tr reads


def to_bf16(x):
    """Round f32 -> bf16 (round-to-nearest-even) represented as f32."""
    x = np.asarray(x, dtype=np.float32)
    u = x.view(np.uint32)
    rounded = ((u + 0x7FFF + ((u >> 16) & 1)) & 0xFFFF0000).view(np.float32)
    return rounded


def mfma32_AB(A, B, C):
    """D = A @ B + C for one 32x32x16 tile given full matrices (the lane
    fragment maps are exercised by the callers below)."""
    return A.astype(np.float64) @ B.astype(np.float64) + C


class LaneSim:
    """Per-lane register state for one wave."""

    def __init__(self):
        self.regs = {}


def a_frag_rows(lane):
    """MFMA A[m][k]: lane holds row m = lane&31, k = (lane>>5)*8 + j."""
    return lane & 31, (lane >> 5) * 8


def c_rows(lane, r):
    """MFMA C[m][n]: reg r -> m = (r&3) + 8*(r>>2) + 4*(lane>>5), n=lane&31."""
    return (r & 3) + 8 * (r >> 2) + 4 * (lane >> 5)


def tr16_read(lds, addr_elem_by_lane):
    """Measured ds_read_b64_tr_b16: returns out[lane][j] (4 elements)."""
    out = np.zeros((WAVE, 4), dtype=np.float32)
    for l in range(WAVE):
        g = l >> 4
        for j in range(4):
            src_lane = 16 * g + 4 * j + ((l >> 2) & 3)
            base = addr_elem_by_lane[src_lane]
            out[l, j] = lds[base + (l & 3)]
    return out


def permlane32_swap(a, b):
    """a,b: arrays[WAVE]. Returns (a.lo|b.lo, a.hi|b.hi)."""
    ra = np.concatenate([a[:32], b[:32]])
    rb = np.concatenate([a[32:], b[32:]])
    return ra, rb


def simulate_wave_attention(q, k, v, scale, causal, q0):
    """Simulate one wave's QBLK2=32 q rows against all of k/v using the
    exact fragment pipeline of attn_fwd2.  q:[sq,D] (this head), k/v:
    [sk,D].  Returns O[32,D] and lse[32]."""
    sq, sk = q.shape[0], k.shape[0]
    n_tiles = (min(sk, q0 + QBLK2 + (sk - sq)) + KVBLK - 1) // KVBLK \
        if causal else (sk + KVBLK - 1) // KVBLK

    # per-lane state
    m_run = np.full(WAVE, -np.inf)
    l_run = np.zeros(WAVE)
    # oacc[dsub][r][lane] — O^T C-layout per 32-d m-tile
    oacc = np.zeros((D // 32, 16, WAVE))

    # Q B-fragments: qfrag[f][lane][j] = Q[q0 + (l&31)][16f + (l>>5)*8 + j]
    qfrag = np.zeros((D // 16, WAVE, 8), dtype=np.float32)
    for f in range(D // 16):
        for l in range(WAVE):
            qrow = q0 + (l & 31)
            if qrow < sq:
                d0 = 16 * f + (l >> 5) * 8
                qfrag[f, l] = to_bf16(q[qrow, d0:d0 + 8])

    for t in range(n_tiles):
        kv0 = t * KVBLK
        # K tile in "LDS" (row-major; swizzle is an addressing detail)
        k_tile = np.zeros((KVBLK, D), dtype=np.float32)
        v_tile = np.zeros((KVBLK, VRS), dtype=np.float32)
        for r in range(KVBLK):
            if kv0 + r < sk:
                k_tile[r] = to_bf16(k[kv0 + r])
                v_tile[r, :D] = to_bf16(v[kv0 + r])

        # ---- S^T = K Q^T per 32-kv sub-block (C[m=kv][n=q])
        st = np.zeros((2, 16, WAVE))  # [ksub][reg][lane]
        for ksub in range(2):
            # matrix-level check of the fragment contraction:
            # A = K[32ksub:32ksub+32, :], B = Q^T -> C = A @ B^T... B[k][n]
            # with B[d][q] = Q[q][d]
            A = k_tile[32 * ksub:32 * ksub + 32, :]           # [32 kv, D]
            Qm = np.zeros((32, D), dtype=np.float32)
            for l in range(32):
                qrow = q0 + l
                if qrow < sq:
                    Qm[l] = to_bf16(q[qrow])
            Cm = A.astype(np.float64) @ Qm.T.astype(np.float64)  # [kv, q]
            for r in range(16):
                for l in range(WAVE):
                    st[ksub, r, l] = Cm[c_rows(l, r), l & 31]

        # ---- mask + per-lane online softmax (lane owns q = q0 + (l&31))
        pvals = np.zeros((2, 16, WAVE))
        mtile = np.full(WAVE, -np.inf)
        sv = np.zeros((2, 16, WAVE))
        for ksub in range(2):
            for r in range(16):
                for l in range(WAVE):
                    kv = kv0 + 32 * ksub + c_rows(l, r)
                    qrow = q0 + (l & 31)
                    s = st[ksub, r, l] * scale
                    ok = kv < sk and qrow < sq
                    if causal:
                        ok = ok and kv <= qrow + (sk - sq)
                    sv[ksub, r, l] = s if ok else -np.inf
                    if ok:
                        mtile[l] = max(mtile[l], s)
        # cross-half max (shfl_xor 32)
        for l in range(WAVE):
            mtile[l] = max(mtile[l], mtile[l ^ 32])
        mn = np.maximum(m_run, mtile)
        alpha = np.where(np.isinf(m_run), 0.0, np.exp(m_run - mn))
        m_run = mn
        lt = np.zeros(WAVE)
        for ksub in range(2):
            for r in range(16):
                for l in range(WAVE):
                    p = 0.0 if np.isinf(sv[ksub, r, l]) else \
                        np.exp(sv[ksub, r, l] - m_run[l])
                    pvals[ksub, r, l] = p
                    lt[l] += p
        lt = lt + lt[np.arange(WAVE) ^ 32]
        l_run = l_run * alpha + lt
        oacc *= alpha[None, None, :]

        # ---- P^T fragment assembly: cvt_pk + permlane32_swap
        # own packed words u[ksub][i4][t2][lane] = (p[rr=2t2], p[rr=2t2+1])
        # as a bf16 pair; we keep them as float pairs.
        pfrag = np.zeros((4, WAVE, 8))  # [ks 0..3][lane][jj]
        for ksub in range(2):
            u = np.zeros((4, 2, WAVE, 2))
            for i4 in range(4):
                for t2 in range(2):
                    for l in range(WAVE):
                        u[i4, t2, l, 0] = to_bf16(pvals[ksub, 4 * i4 + 2 * t2, l])
                        u[i4, t2, l, 1] = to_bf16(pvals[ksub, 4 * i4 + 2 * t2 + 1, l])
            for K in range(2):  # 16-kv step within the 32-kv sub-block
                s0a, s0b = permlane32_swap(u[2 * K, 0], u[2 * K + 1, 0])
                s1a, s1b = permlane32_swap(u[2 * K, 1], u[2 * K + 1, 1])
                ks = 2 * ksub + K
                pfrag[ks, :, 0:2] = s0a
                pfrag[ks, :, 2:4] = s1a
                pfrag[ks, :, 4:6] = s0b
                pfrag[ks, :, 6:8] = s1b

        # verify pfrag against the B-fragment definition of P^T:
        # B[k=16ks + (l>>5)*8 + jj][n=l&31] = P^T[kv][q] = pvals at kv,q
        for ks in range(4):
            for l in range(WAVE):
                for jj in range(8):
                    kvl = 16 * ks + (l >> 5) * 8 + jj
                    # locate owner of (kv=kvl, q=l&31) in C layout
                    ksub_o, within = divmod(kvl, 32)
                    # within = (r&3) + 8*(r>>2) + 4*h
                    h_o = (within >> 2) & 1
                    r_o = (within & 3) + 4 * ((within >> 3))
                    owner = (l & 31) + 32 * h_o
                    expect = to_bf16(pvals[ksub_o, r_o, owner])
                    assert pfrag[ks, l, jj] == expect, (ks, l, jj)

        # ---- O^T += V^T P^T via tr-read A-fragments
        # A[m=d(32dsub)][k=kv]: lane holds V[16ks + (l>>5)*8 + jj][
        #    32dsub + (l&31)] — from the row-major v_tile via 2 tr reads.
        for dsub in range(D // 32):
            for ks in range(4):
                vfrag = np.zeros((WAVE, 8))
                for rr in range(2):
                    addr = np.zeros(WAVE, dtype=np.int64)
                    for L in range(WAVE):
                        row = 16 * ks + 8 * (L >> 5) + 4 * rr + ((L >> 2) & 3)
                        col = 32 * dsub + 16 * ((L >> 4) & 1) + 4 * (L & 3)
                        addr[L] = row * VRS + col
                    out = tr16_read(v_tile.ravel(), addr)
                    vfrag[:, 4 * rr:4 * rr + 4] = out
                # verify fragment contents
                for l in range(WAVE):
                    for jj in range(8):
                        kvl = 16 * ks + (l >> 5) * 8 + jj
                        d_ = 32 * dsub + (l & 31)
                        assert vfrag[l, jj] == v_tile[kvl, d_], (dsub, ks, l, jj)
                # MFMA: C[m=d][n=q] += A(V^T) x B(P^T), matrix-level
                Vt = v_tile[16 * ks:16 * ks + 16, 32 * dsub:32 * dsub + 32].T
                Pm = np.zeros((16, 32))
                for l in range(32):
                    for kk in range(16):
                        ksub_o, within = divmod(16 * ks + kk, 32)
                        h_o = (within >> 2) & 1
                        r_o = (within & 3) + 4 * (within >> 3)
                        Pm[kk, l] = to_bf16(pvals[ksub_o, r_o, l + 32 * h_o])
                Cm = Vt.astype(np.float64) @ Pm.astype(np.float64)
                for r in range(16):
                    for l in range(WAVE):
                        oacc[dsub, r, l] += Cm[c_rows(l, r), l & 31]

    # ---- epilogue: O^T -> O rows, normalize per lane's q
    O = np.zeros((QBLK2, D))
    lse = np.zeros(QBLK2)
    for l in range(WAVE):
        qq = l & 31
        denom = 1.0 / l_run[l] if l_run[l] > 0 else 0.0
        if l < 32:
            lse[qq] = m_run[l] + np.log(max(l_run[l], 1e-30))
        for dsub in range(D // 32):
            for r in range(16):
                d_ = 32 * dsub + c_rows(l, r)
                O[qq, d_] = oacc[dsub, r, l] * denom
    return O, lse


def ref_attention(q, k, v, scale, causal):
    s = to_bf16(q) @ to_bf16(k).T * scale
    sq, sk = s.shape
    if causal:
        mask = np.triu(np.ones((sq, sk), dtype=bool), k=1 + (sk - sq))
        s = np.where(mask, -np.inf, s)
    m = s.max(axis=1, keepdims=True)
    p = np.exp(s - m)
    l = p.sum(axis=1, keepdims=True)
    return (p / l) @ to_bf16(v), (m[:, 0] + np.log(l[:, 0]))


@pytest.mark.parametrize("causal", [False, True])
def test_fwd2_wave_pipeline_matches_reference(causal):
    rng = np.random.default_rng(11)
    sq = sk = 128
    q = rng.standard_normal((sq, D)).astype(np.float32)
    k = rng.standard_normal((sk, D)).astype(np.float32)
    v = rng.standard_normal((sk, D)).astype(np.float32)
    scale = D ** -0.5
    for q0 in (0, 32, 96):
        O, lse = simulate_wave_attention(q, k, v, scale, causal, q0)
        O_ref, lse_ref = ref_attention(q, k, v, scale, causal)
        np.testing.assert_allclose(O, O_ref[q0:q0 + 32], atol=3e-2)
        np.testing.assert_allclose(lse, lse_ref[q0:q0 + 32], atol=1e-3)


def test_tr16_probe_matches_measured_table():
    """Mode-0 of the hardware probe: lane l address = element 4l."""
    lds = np.arange(4096, dtype=np.float32)
    addr = np.arange(WAVE, dtype=np.int64) * 4
    out = tr16_read(lds, addr)
    # measured: out(l, j) = 64*(l>>4) + (l&15) + 16*j
    for l in range(WAVE):
        for j in range(4):
            assert out[l, j] == 64 * (l >> 4) + (l & 15) + 16 * j
