"""Lane-exact NumPy simulation for the round-2 attention backward
(`attn_bwd2` = dq2 + dkv2 kernels in ops/csrc/attention.hip).

dq2 (grid over q): lane owns q = l&31; S^T and dP^T come out of swapped
MFMAs in C[m=kv][n=q] layout so lse/Drow are per-lane scalars; dS^T is
packed to B-fragments with the fwd2 swap recipe and contracted against
K^T fragments (tr16 reads of a row-major K image) into dQ^T.

dkv2 (grid over kv): lane owns kv = l&31 as the MFMA n; S and dP come
out in C[m=q][n=kv] layout; P and dS are assembled into A-fragments
with the SAME swap recipe (A and B fragment maps are mutual transposes)
and contracted against Q^T/dO^T tr16 fragments into dK/dV accumulated
over the q loop and the GQA head group.

Shares the hardware-verified primitives of test_attn_fwd2_sim.py.
"""
import numpy as np
import pytest

from tests.test_attn_fwd2_sim import (
    WAVE, KVBLK, D, VRS, to_bf16, c_rows, tr16_read, permlane32_swap)

QW2 = 32  # rows owned per wave (q rows in dq2, kv rows in dkv2)


def swap_assemble(vals):
    """The fwd2 in-register transpose: vals[reg 0..15][lane] in C layout
    (rows crow(l, r) of one 32-row tile, col l&31) -> frags[K 0..1][lane]
    [jj 0..7] where frag K holds row 16K + (l>>5)*8 + jj at col l&31.
    Works identically for B-fragments (fwd2 P^T) and A-fragments (dkv2
    P / dS) since the A and B lane maps are mutual transposes."""
    u = np.zeros((4, 2, WAVE, 2))
    for i4 in range(4):
        for t2 in range(2):
            u[i4, t2, :, 0] = to_bf16(vals[4 * i4 + 2 * t2])
            u[i4, t2, :, 1] = to_bf16(vals[4 * i4 + 2 * t2 + 1])
    frags = np.zeros((2, WAVE, 8))
    for K in range(2):
        s0a, s0b = permlane32_swap(u[2 * K, 0], u[2 * K + 1, 0])
        s1a, s1b = permlane32_swap(u[2 * K, 1], u[2 * K + 1, 1])
        frags[K, :, 0:2] = s0a
        frags[K, :, 2:4] = s1a
        frags[K, :, 4:6] = s0b
        frags[K, :, 6:8] = s1b
    return frags


def test_swap_assemble_is_a_transpose():
    """frag[K][l][jj] must equal vals at row 16K + 8*(l>>5) + jj, col l&31."""
    rng = np.random.default_rng(5)
    vals = to_bf16(rng.standard_normal((16, WAVE)))
    frags = swap_assemble(vals)
    for K in range(2):
        for l in range(WAVE):
            for jj in range(8):
                row = 16 * K + 8 * (l >> 5) + jj
                h_o = (row >> 2) & 1
                r_o = ((row % 32) & 3) + 4 * ((row % 32) >> 3)
                owner = (l & 31) + 32 * h_o
                assert frags[K, l, jj] == vals[r_o, owner]


def tr_column_frag(img, base_row, dsub):
    """B[k][n] / A[m][k] column fragment from a row-major [rows][VRS]
    image: lane l gets img[base_row + 8*(l>>5) + jj][32*dsub + (l&31)]
    via 2 tr16 reads (the fwd2 V pattern)."""
    frag = np.zeros((WAVE, 8))
    for rr in range(2):
        addr = np.zeros(WAVE, dtype=np.int64)
        for L in range(WAVE):
            row = base_row + 8 * (L >> 5) + 4 * rr + ((L >> 2) & 3)
            col = 32 * dsub + 16 * ((L >> 4) & 1) + 4 * (L & 3)
            addr[L] = row * VRS + col
        out = tr16_read(img.ravel(), addr)
        frag[:, 4 * rr:4 * rr + 4] = out
    return frag


def ref_grads(q, k, v, do, scale, causal):
    qb, kb, vb = to_bf16(q), to_bf16(k), to_bf16(v)
    s = qb @ kb.T * scale
    sq, sk = s.shape
    if causal:
        mask = np.triu(np.ones((sq, sk), dtype=bool), k=1 + (sk - sq))
        s = np.where(mask, -np.inf, s)
    m = s.max(axis=1, keepdims=True)
    p = np.exp(s - m)
    l = p.sum(axis=1, keepdims=True)
    P = p / l
    o = P @ vb
    lse = m[:, 0] + np.log(l[:, 0])
    dP = do @ vb.T
    drow = (do * o).sum(axis=1)
    dS = P * (dP - drow[:, None]) * scale
    return (to_bf16(dS) @ kb, to_bf16(dS).T @ qb, to_bf16(P).T @ do,
            lse, drow)


def sim_dq2_wave(q, k, v, do, lse, drow, scale, causal, q0):
    """One wave of dq2: 32 q rows vs all kv tiles."""
    sq, sk = q.shape[0], k.shape[0]
    dq_acc = np.zeros((D // 32, 16, WAVE))  # dQ^T C layout per d m-tile
    lse_l = np.array([lse[q0 + (l & 31)] for l in range(WAVE)])
    dr_l = np.array([drow[q0 + (l & 31)] for l in range(WAVE)])
    kv_end = min(sk, q0 + QW2 + (sk - sq)) if causal else sk
    for t in range((kv_end + KVBLK - 1) // KVBLK):
        kv0 = t * KVBLK
        k_tile = np.zeros((KVBLK, VRS))
        k_tile[:, :D] = to_bf16(k[kv0:kv0 + KVBLK])
        v_tile = to_bf16(v[kv0:kv0 + KVBLK])
        # S^T, dP^T in C[m=kv][n=q] per 32-kv sub-block
        dsT = np.zeros((2, 16, WAVE))
        for ksub in range(2):
            Km = k_tile[32 * ksub:32 * ksub + 32, :D]
            Vm = v_tile[32 * ksub:32 * ksub + 32]
            Qm = to_bf16(q[q0:q0 + 32])
            Dm = to_bf16(do[q0:q0 + 32])
            stC = Km @ Qm.T
            dpC = Vm @ Dm.T
            for r in range(16):
                for l in range(WAVE):
                    kv = kv0 + 32 * ksub + c_rows(l, r)
                    qrow = q0 + (l & 31)
                    ok = kv < sk
                    if causal:
                        ok = ok and kv <= qrow + (sk - sq)
                    p = np.exp(stC[c_rows(l, r), l & 31] * scale - lse_l[l]) \
                        if ok else 0.0
                    dsT[ksub, r, l] = p * (dpC[c_rows(l, r), l & 31]
                                           - dr_l[l]) * scale
        # dS^T B-fragments + K^T tr fragments -> dQ^T
        for ksub in range(2):
            frags = swap_assemble(dsT[ksub])
            for K in range(2):
                ks = 2 * ksub + K
                for dsub in range(D // 32):
                    kfrag = tr_column_frag(k_tile, 16 * ks, dsub)
                    # matrix check: C[m=d][n=q] += K^T[d][kv16] @ dS^T[kv16][q]
                    Kt = k_tile[16 * ks:16 * ks + 16,
                                32 * dsub:32 * dsub + 32].T
                    dSm = np.zeros((16, 32))
                    for l in range(WAVE):
                        for jj in range(8):
                            kk = 8 * (l >> 5) + jj
                            dSm[kk, l & 31] = frags[K, l, jj]
                    Cm = Kt @ dSm
                    for r in range(16):
                        for l in range(WAVE):
                            dq_acc[dsub, r, l] += Cm[c_rows(l, r), l & 31]
                    # fragment-level: kfrag must be K[16ks+8h+jj][32dsub+l&31]
                    for l in range(WAVE):
                        for jj in range(8):
                            assert kfrag[l, jj] == k_tile[
                                16 * ks + 8 * (l >> 5) + jj,
                                32 * dsub + (l & 31)]
    dq = np.zeros((QW2, D))
    for l in range(32):
        for dsub in range(D // 32):
            for r in range(16):
                dq[l, 32 * dsub + c_rows(l, r)] = dq_acc[dsub, r, l]
                dq[l, 32 * dsub + c_rows(l + 32, r)] = dq_acc[dsub, r, l + 32]
    return dq


def sim_dkv2_wave(q, k, v, do, lse, drow, scale, causal, kv0w):
    """One wave of dkv2: 32 kv rows vs all q tiles (single head)."""
    sq, sk = q.shape[0], k.shape[0]
    dk_acc = np.zeros((D // 32, 16, WAVE))
    dv_acc = np.zeros((D // 32, 16, WAVE))
    Km = to_bf16(k[kv0w:kv0w + 32])
    Vm = to_bf16(v[kv0w:kv0w + 32])
    q_start = max(0, (kv0w - (sk - sq)) // QW2 * QW2) if causal else 0
    for qt in range(q_start, sq, QW2):
        q_img = np.zeros((QW2, VRS))
        do_img = np.zeros((QW2, VRS))
        q_img[:, :D] = to_bf16(q[qt:qt + QW2])
        do_img[:, :D] = to_bf16(do[qt:qt + QW2])
        # S, dP in C[m=q][n=kv]
        sC = to_bf16(q[qt:qt + 32]) @ Km.T
        dpC = to_bf16(do[qt:qt + 32]) @ Vm.T
        pv = np.zeros((16, WAVE))
        dsv = np.zeros((16, WAVE))
        for r in range(16):
            for l in range(WAVE):
                qrow = qt + c_rows(l, r)
                kv = kv0w + (l & 31)
                ok = qrow < sq
                if causal:
                    ok = ok and kv <= qrow + (sk - sq)
                p = np.exp(sC[c_rows(l, r), l & 31] * scale - lse[qrow]) \
                    if ok else 0.0
                pv[r, l] = p
                dsv[r, l] = p * (dpC[c_rows(l, r), l & 31]
                                 - drow[qrow]) * scale
        pfr = swap_assemble(pv)
        dsfr = swap_assemble(dsv)
        for K in range(2):
            for dsub in range(D // 32):
                dof = tr_column_frag(do_img, 16 * K, dsub)
                qf = tr_column_frag(q_img, 16 * K, dsub)
                # dV[kv][d] += P^T[kv][q16] @ dO[q16][d]
                Pm = np.zeros((32, 16))
                dSm = np.zeros((32, 16))
                for l in range(WAVE):
                    for jj in range(8):
                        qq = 8 * (l >> 5) + jj
                        Pm[l & 31, qq] = pfr[K, l, jj]
                        dSm[l & 31, qq] = dsfr[K, l, jj]
                dOm = do_img[16 * K:16 * K + 16, 32 * dsub:32 * dsub + 32]
                Qm2 = q_img[16 * K:16 * K + 16, 32 * dsub:32 * dsub + 32]
                Cv = Pm @ dOm
                Ck = dSm @ Qm2
                for r in range(16):
                    for l in range(WAVE):
                        dv_acc[dsub, r, l] += Cv[c_rows(l, r), l & 31]
                        dk_acc[dsub, r, l] += Ck[c_rows(l, r), l & 31]
    dk = np.zeros((QW2, D))
    dv = np.zeros((QW2, D))
    for l in range(WAVE):
        for dsub in range(D // 32):
            for r in range(16):
                dk[c_rows(l, r), 32 * dsub + (l & 31)] = dk_acc[dsub, r, l]
                dv[c_rows(l, r), 32 * dsub + (l & 31)] = dv_acc[dsub, r, l]
    return dk, dv


@pytest.mark.parametrize("causal", [False, True])
def test_dq2_wave_matches_reference(causal):
    rng = np.random.default_rng(17)
    sq = sk = 128
    q = rng.standard_normal((sq, D)).astype(np.float32)
    k = rng.standard_normal((sk, D)).astype(np.float32)
    v = rng.standard_normal((sk, D)).astype(np.float32)
    do = rng.standard_normal((sq, D)).astype(np.float32)
    scale = D ** -0.5
    dq_ref, _, _, lse, drow = ref_grads(q, k, v, do, scale, causal)
    for q0 in (0, 96):
        dq = sim_dq2_wave(q, k, v, do, lse, drow, scale, causal, q0)
        np.testing.assert_allclose(dq, dq_ref[q0:q0 + 32], atol=5e-2)


@pytest.mark.parametrize("causal", [False, True])
def test_dkv2_wave_matches_reference(causal):
    rng = np.random.default_rng(19)
    sq = sk = 128
    q = rng.standard_normal((sq, D)).astype(np.float32)
    k = rng.standard_normal((sk, D)).astype(np.float32)
    v = rng.standard_normal((sk, D)).astype(np.float32)
    do = rng.standard_normal((sq, D)).astype(np.float32)
    scale = D ** -0.5
    _, dk_ref, dv_ref, lse, drow = ref_grads(q, k, v, do, scale, causal)
    for kv0 in (0, 64):
        dk, dv = sim_dkv2_wave(q, k, v, do, lse, drow, scale, causal, kv0)
        np.testing.assert_allclose(dk, dk_ref[kv0:kv0 + 32], atol=5e-2)
        np.testing.assert_allclose(dv, dv_ref[kv0:kv0 + 32], atol=5e-2)
