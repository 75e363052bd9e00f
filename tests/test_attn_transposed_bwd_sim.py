"""Design simulation for the transposed-S flash BACKWARD (round-2 kernel):
lane-accurate NumPy model of the dQ pass with per-lane query state.

Layout plan being locked here:
  * recompute P^T = exp(S^T*scale - lse_q) exactly as the forward
    (S^T = K.Q^T; lane owns one q column; lse per-lane scalar)
  * dP^T = V . dO^T via mfma(A = V fragments [kv][d],
    B = dO-as-Q-layout fragments [d][q]) -> C [kv][q]
  * dS^T = P^T * (dP^T - D_q) * scale with D_q = rowsum(dO * O) a
    per-lane scalar
  * dQ^T[d][q] += K^T[d][kv] . dS^T[kv][q]: A = K^T fragments read from a
    transposed-K LDS image (same padded-stride layout as V^T), B = dS^T
    routed through the SAME ds_bpermute pattern the forward uses for P^T
  * dQ accumulates in registers (lane's q column), written once at the end
dK/dV stay with the kv-block-owner kernel (symmetric roles).
"""
import numpy as np

WAVE = 64
D = 32
KVBLK = 64

lanes = np.arange(WAVE)
lrow = lanes >> 4
lcol = lanes & 15


def mfma(A_frag, B_frag, C_frag):
    A = np.zeros((16, 32)); B = np.zeros((32, 16))
    for l in range(WAVE):
        for j in range(8):
            A[l & 15, (l >> 4) * 8 + j] = A_frag[l, j]
            B[(l >> 4) * 8 + j, l & 15] = B_frag[l, j]
    Dm = A @ B
    out = C_frag.copy()
    for l in range(WAVE):
        for r in range(4):
            out[l, r] += Dm[(l >> 4) * 4 + r, l & 15]
    return out


def route_bpermute(pf):
    """fwd's P^T C-layout -> B-fragment routing (pf: [4][WAVE][4])."""
    def one_ks(ks):
        pk = np.zeros((2, 2, WAVE, 2))
        for tsub in range(2):
            ksub = ks * 2 + tsub
            pk[tsub][0][:, 0] = pf[ksub][:, 0]
            pk[tsub][0][:, 1] = pf[ksub][:, 1]
            pk[tsub][1][:, 0] = pf[ksub][:, 2]
            pk[tsub][1][:, 1] = pf[ksub][:, 3]
        frag = np.zeros((WAVE, 8))
        for pp in range(4):
            src = ((lrow & 1) * 2 + (pp >> 1)) * 16 + lcol
            v0 = pk[0][pp & 1][src]
            v1 = pk[1][pp & 1][src]
            vv = np.where((lrow >= 2)[:, None], v1, v0)
            frag[:, 2 * pp] = vv[:, 0]
            frag[:, 2 * pp + 1] = vv[:, 1]
        return frag
    return [one_ks(0), one_ks(1)]


def test_transposed_flash_bwd_dq_simulation():
    rng = np.random.RandomState(1)
    sq, sk = 16, 128
    Q = rng.randn(sq, D).astype(np.float32) * 0.5
    K = rng.randn(sk, D).astype(np.float32) * 0.5
    V = rng.randn(sk, D).astype(np.float32) * 0.5
    dO = rng.randn(sq, D).astype(np.float32) * 0.5
    scale = D ** -0.5

    # reference backward (dQ)
    S = (Q @ K.T) * scale
    mask = np.triu(np.ones((sq, sk)), 1 + sk - sq).astype(bool)
    S[mask] = -np.inf
    m = S.max(-1, keepdims=True)
    P = np.exp(S - m)
    l = P.sum(-1, keepdims=True)
    Pn = P / l
    O = Pn @ V
    lse = (m + np.log(l)).squeeze(-1)
    Drow = (dO * O).sum(-1)
    dP = dO @ V.T
    dS = Pn * (dP - Drow[:, None]) * scale
    dQ_ref = dS @ K

    # lane model
    q0 = 0
    qrow = q0 + lcol
    qfrag = np.zeros((WAVE, 8))
    dofrag = np.zeros((WAVE, 8))
    for lane in range(WAVE):
        qfrag[lane] = Q[qrow[lane], (lane >> 4) * 8:(lane >> 4) * 8 + 8]
        dofrag[lane] = dO[qrow[lane], (lane >> 4) * 8:(lane >> 4) * 8 + 8]
    lse_l = lse[qrow]
    Drow_l = Drow[qrow]
    DS = D // 16
    dqacc = np.zeros((DS, WAVE, 4))

    for t in range(sk // KVBLK):
        kv0 = t * KVBLK
        # recompute P^T (normalized by l via lse)
        pf = np.zeros((4, WAVE, 4))
        for ksub in range(4):
            afrag = np.zeros((WAVE, 8))
            for lane in range(WAVE):
                krow = kv0 + ksub * 16 + (lane & 15)
                afrag[lane] = K[krow, (lane >> 4) * 8:(lane >> 4) * 8 + 8]
            stc = mfma(afrag, qfrag, np.zeros((WAVE, 4)))
            for r in range(4):
                kvcol = kv0 + ksub * 16 + lrow * 4 + r
                valid = kvcol <= qrow + (sk - sq)
                pf[ksub][:, r] = np.where(
                    valid, np.exp(stc[:, r] * scale - lse_l), 0.0)
        # dP^T = V . dO^T
        dpt = np.zeros((4, WAVE, 4))
        for ksub in range(4):
            afrag = np.zeros((WAVE, 8))
            for lane in range(WAVE):
                krow = kv0 + ksub * 16 + (lane & 15)
                afrag[lane] = V[krow, (lane >> 4) * 8:(lane >> 4) * 8 + 8]
            dpt[ksub] = mfma(afrag, dofrag, np.zeros((WAVE, 4)))
        # dS^T in place
        dsf = np.zeros((4, WAVE, 4))
        for ksub in range(4):
            for r in range(4):
                dsf[ksub][:, r] = pf[ksub][:, r] * \
                    (dpt[ksub][:, r] - Drow_l) * scale
        # dQ^T[d][q] += K^T . dS^T  (dS^T routed like the fwd's P^T)
        ds_frags = route_bpermute(dsf)
        for ks in range(2):
            for dsub in range(DS):
                ktfrag = np.zeros((WAVE, 8))
                for lane in range(WAVE):
                    drow = dsub * 16 + (lane & 15)
                    kvs = kv0 + ks * 32 + (lane >> 4) * 8
                    ktfrag[lane] = K[kvs:kvs + 8, drow]
                dqacc[dsub] = mfma(ktfrag, ds_frags[ks], dqacc[dsub])

    dQ = np.zeros((sq, D))
    for lane in range(WAVE):
        for dsub in range(DS):
            for r in range(4):
                dQ[qrow[lane], dsub * 16 + (lane >> 4) * 4 + r] = \
                    dqacc[dsub][lane, r]
    err = np.abs(dQ - dQ_ref).max()
    assert err < 1e-4, err


def test_transposed_flash_bwd_dkv_simulation():
    """dK/dV kernel: the workgroup owns a KV block of 16 columns-as-lanes
    (kv = lane column), streams Q/dO tiles; dK^T/dV^T accumulate in the
    lane's registers, one global write at the end.
      * S = Q.K^T via mfma(A = Q fragments [q][d], B = K-as-Q-layout
        [d][kv]) -> C [q][kv]: per-lane kv column, per-ROW q (lse/D are
        per-q, read from LDS-staged vectors, not per-lane scalars)
      * P = exp(S*scale - lse_q) in C layout [q][kv]
      * dP = dO.V^T likewise -> dS [q][kv] in regs
      * dV^T[d][kv] += dO^T[d][q] . P[q][kv]: A = dO^T fragments from a
        transposed-dO LDS image, B = P routed through the SAME bpermute
        pattern (C layout [q][kv] -> B fragment [q-contraction][kv])
      * dK^T[d][kv] += Q^T[d][q] . dS[q][kv]: identical routing with dS
    """
    rng = np.random.RandomState(7)
    sq, sk = 128, 16
    Q = rng.randn(sq, D).astype(np.float32) * 0.5
    K = rng.randn(sk, D).astype(np.float32) * 0.5
    V = rng.randn(sk, D).astype(np.float32) * 0.5
    dO = rng.randn(sq, D).astype(np.float32) * 0.5
    scale = D ** -0.5

    S = (Q @ K.T) * scale
    # this kv tile is the FIRST 16 keys of the sequence: diag offset 0
    mask = np.triu(np.ones((sq, sk)), 1).astype(bool)
    S[mask] = -np.inf
    m = S.max(-1, keepdims=True)
    P = np.exp(S - m)
    l = P.sum(-1, keepdims=True)
    Pn = P / l
    O = Pn @ V
    lse = (m + np.log(l)).squeeze(-1)
    Drow = (dO * O).sum(-1)
    dP = dO @ V.T
    dS = Pn * (dP - Drow[:, None]) * scale
    dK_ref = dS.T @ Q
    dV_ref = Pn.T @ dO

    kv0 = 0
    kvcol = kv0 + lcol
    kfrag = np.zeros((WAVE, 8))
    vfrag = np.zeros((WAVE, 8))
    for lane in range(WAVE):
        kfrag[lane] = K[kvcol[lane], (lane >> 4) * 8:(lane >> 4) * 8 + 8]
        vfrag[lane] = V[kvcol[lane], (lane >> 4) * 8:(lane >> 4) * 8 + 8]
    DS = D // 16
    dkacc = np.zeros((DS, WAVE, 4))
    dvacc = np.zeros((DS, WAVE, 4))
    QBLK = 64

    for t in range(sq // QBLK):
        q0 = t * QBLK
        pc = np.zeros((4, WAVE, 4))   # P   C-layout [q][kv]
        dsc = np.zeros((4, WAVE, 4))  # dS  C-layout [q][kv]
        for qsub in range(4):
            afrag = np.zeros((WAVE, 8))
            dofragA = np.zeros((WAVE, 8))
            for lane in range(WAVE):
                qr = q0 + qsub * 16 + (lane & 15)
                afrag[lane] = Q[qr, (lane >> 4) * 8:(lane >> 4) * 8 + 8]
                dofragA[lane] = dO[qr, (lane >> 4) * 8:(lane >> 4) * 8 + 8]
            sc = mfma(afrag, kfrag, np.zeros((WAVE, 4)))
            dpc = mfma(dofragA, vfrag, np.zeros((WAVE, 4)))
            for r in range(4):
                qr = q0 + qsub * 16 + lrow * 4 + r
                valid = kvcol <= qr
                pv = np.where(valid,
                              np.exp(sc[:, r] * scale - lse[qr]), 0.0)
                pc[qsub][:, r] = pv
                dsc[qsub][:, r] = pv * (dpc[:, r] - Drow[qr]) * scale
        p_frags = route_bpermute(pc)
        ds_frags = route_bpermute(dsc)
        for ks in range(2):
            for dsub in range(DS):
                dotfrag = np.zeros((WAVE, 8))
                qtfrag = np.zeros((WAVE, 8))
                for lane in range(WAVE):
                    drow = dsub * 16 + (lane & 15)
                    qs = q0 + ks * 32 + (lane >> 4) * 8
                    dotfrag[lane] = dO[qs:qs + 8, drow]
                    qtfrag[lane] = Q[qs:qs + 8, drow]
                dvacc[dsub] = mfma(dotfrag, p_frags[ks], dvacc[dsub])
                dkacc[dsub] = mfma(qtfrag, ds_frags[ks], dkacc[dsub])

    dK = np.zeros((sk, D)); dV = np.zeros((sk, D))
    for lane in range(WAVE):
        for dsub in range(DS):
            for r in range(4):
                dcol = dsub * 16 + (lane >> 4) * 4 + r
                dK[kvcol[lane], dcol] = dkacc[dsub][lane, r]
                dV[kvcol[lane], dcol] = dvacc[dsub][lane, r]
    assert np.abs(dV - dV_ref).max() < 1e-4
    assert np.abs(dK - dK_ref).max() < 1e-4
