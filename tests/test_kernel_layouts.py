"""Symbolic (host-side) verification of HIP kernel LDS layouts.

These tests transcribe the exact integer index arithmetic of the staged
kernels and check, cell by cell, that (a) staging writes never collide or
leave the allocated region and (b) every compute-phase read sees exactly
the (row, column) element the math requires. This is how the round-1
bwd_dkv 64-row staging bug (32-row column stride kept for 64 staged rows —
see docs/ROADMAP_R2.md) would have been caught without hardware.
"""

import pytest


def _c_row(r, h):
    return (r & 3) + 8 * (r >> 2) + 4 * h


@pytest.mark.parametrize("HD", [64, 128])
@pytest.mark.parametrize("N", [65, 128, 197, 300])
def test_dkv64_staging_layout(HD, N):
    NT = 256
    QB, SQB = 32, 64
    QT_STRIDE = SQB + 8
    HALF = HD // 2
    PAIRS_PER_ROW = HALF // 8
    PER_ROW = HD // 8
    QK_ITEMS = SQB * PAIRS_PER_ROW
    DO_ITEMS = SQB * PER_ROW
    QK_IPT = (QK_ITEMS + NT - 1) // NT
    DO_IPT = (DO_ITEMS + NT - 1) // NT
    qt_size = HD * QT_STRIDE

    for qbase0 in range(0, N, SQB):
        qt, dot, lse, dl = {}, {}, {}, {}
        writes = set()
        for tid in range(NT):
            for j in range(QK_IPT):
                idx = tid + j * NT
                if idx < QK_ITEMS:
                    lrow = idx // PAIRS_PER_ROW
                    c0 = (idx % PAIRS_PER_ROW) * 8
                    for e in range(8):
                        for c in (c0 + e, c0 + HALF + e):
                            addr = c * QT_STRIDE + lrow
                            assert addr < qt_size
                            assert ("q", addr) not in writes
                            writes.add(("q", addr))
                            qt[addr] = (qbase0 + lrow, c)
            for j in range(DO_IPT):
                idx = tid + j * NT
                if idx < DO_ITEMS:
                    lrow = idx // PER_ROW
                    c8 = (idx % PER_ROW) * 8
                    for e in range(8):
                        addr = (c8 + e) * QT_STRIDE + lrow
                        assert addr < qt_size
                        assert ("do", addr) not in writes
                        writes.add(("do", addr))
                        dot[addr] = (qbase0 + lrow, c8 + e)
            if tid < SQB:
                lse[tid] = qbase0 + tid
                dl[tid] = qbase0 + tid

        for tid in range(NT):
            lane = tid & 63
            hhalf = lane >> 5
            l31 = lane & 31
            for sub in range(2):
                if qbase0 + sub * QB >= N:
                    continue
                qrow_off = sub * QB
                qbase = qbase0 + sub * QB
                for s in range(HD // 16):
                    for e in range(8):
                        d = s * 16 + hhalf * 8 + e
                        addr = d * QT_STRIDE + qrow_off + l31
                        assert qt[addr] == (qbase + l31, d)
                        assert dot[addr] == (qbase + l31, d)
                for r in range(16):
                    want = qbase + _c_row(r, hhalf)
                    assert lse[qrow_off + _c_row(r, hhalf)] == want
                    assert dl[qrow_off + _c_row(r, hhalf)] == want
                for t in range(HD // 32):
                    for half16 in (0, 16):
                        for e in range(8):
                            row = qrow_off + half16 + hhalf * 8 + e
                            addr = (t * 32 + l31) * QT_STRIDE + row
                            want = (qbase0 + row, t * 32 + l31)
                            assert dot[addr] == want
                            assert qt[addr] == want


@pytest.mark.parametrize("HD,NT", [(64, 128), (64, 256), (128, 128), (128, 256)])
def test_dkv32_staging_layout(HD, NT):
    """Same check for the production 32-row kernel (and its buggy-history
    counterexample: with 64 staged rows this stride would fail)."""
    QB = 32
    QT_STRIDE = QB + 8
    HALF = HD // 2
    PAIRS_PER_ROW = HALF // 8
    PER_ROW = HD // 8
    QK_ITEMS = QB * PAIRS_PER_ROW
    DO_ITEMS = QB * PER_ROW
    QK_IPT = (QK_ITEMS + NT - 1) // NT
    DO_IPT = (DO_ITEMS + NT - 1) // NT
    qt_size = HD * QT_STRIDE
    qt, dot = {}, {}
    for tid in range(NT):
        for j in range(QK_IPT):
            idx = tid + j * NT
            if idx < QK_ITEMS:
                lrow = idx // PAIRS_PER_ROW
                c0 = (idx % PAIRS_PER_ROW) * 8
                for e in range(8):
                    for c in (c0 + e, c0 + HALF + e):
                        addr = c * QT_STRIDE + lrow
                        assert addr < qt_size
                        assert addr not in qt
                        qt[addr] = (lrow, c)
        for j in range(DO_IPT):
            idx = tid + j * NT
            if idx < DO_ITEMS:
                lrow = idx // PER_ROW
                c8 = (idx % PER_ROW) * 8
                for e in range(8):
                    addr = (c8 + e) * QT_STRIDE + lrow
                    assert addr < qt_size
                    assert addr not in dot
                    dot[addr] = (lrow, c8 + e)
    for lane in range(64):
        hhalf = lane >> 5
        l31 = lane & 31
        for s in range(HD // 16):
            for e in range(8):
                d = s * 16 + hhalf * 8 + e
                assert qt[d * QT_STRIDE + l31] == (l31, d)
                assert dot[d * QT_STRIDE + l31] == (l31, d)
        for t in range(HD // 32):
            for half16 in (0, 16):
                for e in range(8):
                    row = half16 + hhalf * 8 + e
                    addr = (t * 32 + l31) * QT_STRIDE + row
                    assert dot[addr] == (row, t * 32 + l31)
                    assert qt[addr] == (row, t * 32 + l31)


@pytest.mark.parametrize("HD,NT", [(64, 128), (64, 256), (128, 128), (128, 256)])
def test_fwd_kv_staging_layout(HD, NT):
    """fwd_kernel 64-key super-tile: K rows into k_lds [2*KVB][HD+8], V^T
    into vt_lds [HD][2*KVB+8]; compute reads must see the staged element."""
    KVB = 32
    LDS_STRIDE = HD + 8
    VT_STRIDE = 2 * KVB + 8
    HALF = HD // 2
    PAIRS_PER_ROW = HALF // 8
    PER_ROW = HD // 8
    K_ITEMS = 2 * KVB * PAIRS_PER_ROW
    K_IPT = (K_ITEMS + NT - 1) // NT
    V_ITEMS = 2 * KVB * PER_ROW
    V_IPT = (V_ITEMS + NT - 1) // NT
    k_size = 2 * KVB * LDS_STRIDE
    v_size = HD * VT_STRIDE

    k_lds, vt_lds = {}, {}
    kw, vw = set(), set()
    for tid in range(NT):
        for j in range(K_IPT):
            item = tid + j * NT
            if item < K_ITEMS:
                lrow = item // PAIRS_PER_ROW
                c0 = (item % PAIRS_PER_ROW) * 8
                for e in range(8):
                    for c in (c0 + e, HALF + c0 + e):
                        addr = lrow * LDS_STRIDE + c
                        assert addr < k_size
                        assert addr not in kw
                        kw.add(addr)
                        k_lds[addr] = (lrow, c)
        for j in range(V_IPT):
            idx = tid + j * NT
            if idx < V_ITEMS:
                row = idx // PER_ROW
                c8 = (idx % PER_ROW) * 8
                for e in range(8):
                    addr = (c8 + e) * VT_STRIDE + row
                    assert addr < v_size
                    assert addr not in vw
                    vw.add(addr)
                    vt_lds[addr] = (row, c8 + e)

    for lane in range(64):
        hhalf = lane >> 5
        l31 = lane & 31
        for sub in range(2):
            krow_off = sub * KVB
            # S^T A-fragment: K[krow_off + l31][k = s*16 + hhalf*8 + e]
            for s in range(HD // 16):
                for e in range(8):
                    c = s * 16 + hhalf * 8 + e
                    addr = (krow_off + l31) * LDS_STRIDE + c
                    assert k_lds[addr] == (krow_off + l31, c)
            # O accumulation: V^T[d = t*32+l31][k-rows]
            for t in range(HD // 32):
                for half16 in (0, 16):
                    for e in range(8):
                        row = krow_off + half16 + hhalf * 8 + e
                        addr = (t * 32 + l31) * VT_STRIDE + row
                        assert vt_lds[addr] == (row, t * 32 + l31)


@pytest.mark.parametrize("HD,NT", [(64, 128), (64, 256), (128, 128), (128, 256)])
def test_dq_kv_staging_layout(HD, NT):
    """bwd_dq_kernel: K rows into k_lds/[2KVB][HD+8] AND transposed into
    kt_lds/[HD][2KVB+8]; V rows into v_lds; all compute reads verified."""
    KVB = 32
    LDS_STRIDE = HD + 8
    KT_STRIDE = 2 * KVB + 8
    HALF = HD // 2
    PAIRS_PER_ROW = HALF // 8
    PER_ROW = HD // 8
    K_ITEMS = 2 * KVB * PAIRS_PER_ROW
    K_IPT = (K_ITEMS + NT - 1) // NT
    V_ITEMS = 2 * KVB * PER_ROW
    V_IPT = (V_ITEMS + NT - 1) // NT
    k_size = 2 * KVB * LDS_STRIDE
    kt_size = HD * KT_STRIDE

    k_lds, v_lds, kt_lds = {}, {}, {}
    for tid in range(NT):
        for j in range(K_IPT):
            item = tid + j * NT
            if item < K_ITEMS:
                lrow = item // PAIRS_PER_ROW
                c0 = (item % PAIRS_PER_ROW) * 8
                for e in range(8):
                    for c in (c0 + e, HALF + c0 + e):
                        addr = lrow * LDS_STRIDE + c
                        assert addr < k_size and addr not in k_lds
                        k_lds[addr] = (lrow, c)
                        taddr = c * KT_STRIDE + lrow
                        assert taddr < kt_size and taddr not in kt_lds
                        kt_lds[taddr] = (lrow, c)
        for j in range(V_IPT):
            idx = tid + j * NT
            if idx < V_ITEMS:
                row = idx // PER_ROW
                c8 = (idx % PER_ROW) * 8
                for e in range(8):
                    addr = row * LDS_STRIDE + c8 + e
                    assert addr < k_size and addr not in v_lds
                    v_lds[addr] = (row, c8 + e)

    for lane in range(64):
        hhalf = lane >> 5
        l31 = lane & 31
        for sub in range(2):
            krow_off = sub * KVB
            for s in range(HD // 16):
                for e in range(8):
                    c = s * 16 + hhalf * 8 + e
                    addr = (krow_off + l31) * LDS_STRIDE + c
                    assert k_lds[addr] == (krow_off + l31, c)
                    assert v_lds[addr] == (krow_off + l31, c)
            for t in range(HD // 32):
                for half16 in (0, 16):
                    for e in range(8):
                        row = krow_off + half16 + hhalf * 8 + e
                        addr = (t * 32 + l31) * KT_STRIDE + row
                        assert kt_lds[addr] == (row, t * 32 + l31)


def test_patch_embed_staging_layout():
    """patch_embed::fwd_kernel: 256 threads stage a 128x32 A tile and a
    128x32 W tile per K-step ([row][40] layout); MFMA reads must line up."""
    A_STRIDE = 40
    a_lds, b_lds = {}, {}
    for tid in range(256):
        s_row = tid // 2
        s_half = (tid % 2) * 16
        for e in range(16):
            addr = s_row * A_STRIDE + s_half + e
            assert addr < 128 * A_STRIDE
            assert addr not in a_lds
            a_lds[addr] = (s_row, s_half + e)   # (tile row, k within step)
            b_lds[addr] = (s_row, s_half + e)
    for wave in range(4):
        for lane in range(64):
            hhalf = lane >> 5
            l31 = lane & 31
            for ks in range(2):
                for e in range(8):
                    k = ks * 16 + hhalf * 8 + e
                    addr = (wave * 32 + l31) * A_STRIDE + k
                    assert a_lds[addr] == (wave * 32 + l31, k)
                for t in range(4):
                    addr = (t * 32 + l31) * A_STRIDE + ks * 16 + hhalf * 8
                    for e in range(8):
                        assert b_lds[addr + e] == (t * 32 + l31, ks * 16 + hhalf * 8 + e)


def test_multi_tensor_plan_chunk_coverage():
    """MultiTensorPlan chunk table: every element of every tensor covered by
    exactly one (tensor, offset) chunk of <= CHUNK elements."""
    import torch

    from dinov3_amd.ops.mt_plan import CHUNK, MultiTensorPlan

    sizes = [1, CHUNK - 1, CHUNK, CHUNK + 1, 3 * CHUNK + 17]
    tensors = [torch.zeros(s) for s in sizes]
    plan = MultiTensorPlan([tensors])
    ct = plan.ct.tolist()
    co = plan.co.tolist()
    covered = {i: [] for i in range(len(sizes))}
    for t, off in zip(ct, co):
        n = min(CHUNK, sizes[t] - off)
        assert n > 0
        covered[t].append((off, off + n))
    for i, s in enumerate(sizes):
        spans = sorted(covered[i])
        assert spans[0][0] == 0 and spans[-1][1] == s
        for (a0, a1), (b0, b1) in zip(spans, spans[1:]):
            assert a1 == b0, f"gap/overlap in tensor {i}: {spans}"
    # pointer-change detection triggers a table refresh
    old = plan.ptrs.clone()
    tensors[2].data = torch.zeros(CHUNK)
    assert plan.check_pointers() is False
    assert not torch.equal(plan.ptrs, old)
