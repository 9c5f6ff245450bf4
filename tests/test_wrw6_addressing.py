"""Lane-level machine verification of the wrw v6 addressing (CPU).

Models the device-probed ds_read_b64_tr_b16 semantics
(scripts/probe_tr16.hip, r2 probes): a 16-lane group's addresses define
16 four-element runs; lane i receives element (i&3) of the runs at
lanes {4j + (i>>2)} for j = 0..3. Verifies end-to-end that

  glds deposit (per-lane source permutation, csrc/conv_wrw.hip v6)
  -> blocked LDS image -> tr_b16 fragment reads

delivers exactly the MFMA B/A operand fragments: lane l of group g
holds gy/x values for (k = opix g*8..g*8+7, col = l&15) in ascending-k
dword order. Any change to the deposit mapping or fragment addressing
that breaks MFMA-operand correctness fails here without a GPU.
"""

import pytest


def tr_read(lds, group_addrs):
    """Probed semantics for one 16-lane group: returns out[i][j]."""
    assert len(group_addrs) == 16
    runs = []
    for a in group_addrs:
        assert a % 8 == 0, "8-byte alignment required"
        e0 = a // 2  # element index (2-byte elems)
        runs.append([lds[e0 + t] for t in range(4)])
    out = [[None] * 4 for _ in range(16)]
    for i in range(16):
        for j in range(4):
            out[i][j] = runs[4 * j + (i >> 2)][i & 3]
    return out


def simulate_v6_B(Cin=64, chunks=9):
    """Deposit a symbolic B tile and read every fragment back."""
    BK = 32
    # symbolic "global" value for (opix, tapci)
    def val(opix, tapci):
        return ("B", opix, tapci)

    # --- deposit: instr (c, og) writes 1 KiB at (c*32 + og*8)*128 ----
    lds = {}
    for c in range(chunks):
        for og in range(4):
            base_elem = (c * 32 + og * 8) * 128 // 2
            for lane in range(64):
                q = (lane >> 5) & 1
                cs = (lane >> 3) & 3
                r2 = (lane & 7) >> 1
                h = lane & 1
                opix = og * 8 + q * 4 + r2
                ci8 = cs * 2 + h  # ci octet within the 64-ci chunk
                for e in range(8):
                    tapci = c * 64 + ci8 * 8 + e
                    lds[base_elem + lane * 8 + e] = val(opix, tapci)

    # --- fragment reads: wave wid, group g, subtile cs_abs -----------
    for wid in range(4):
        for g in range(4):
            for ni in range(9):
                cs_abs = wid * 9 + ni
                c = cs_abs >> 2
                cs = cs_abs & 3
                frag = [[None] * 8 for _ in range(16)]  # [col][k]
                for h2 in range(2):
                    base = ((c * 8 + 2 * g + h2) * 4 + cs) * 128
                    addrs = [base + l15 * 8 for l15 in range(16)]
                    out = tr_read(lds, addrs)
                    for i in range(16):
                        for j in range(4):
                            frag[i][h2 * 4 + j] = out[i][j]
                # MFMA operand contract: lane col i holds
                # (opix = g*8 + k, tapci = cs_abs*16 + i), k ascending
                for i in range(16):
                    for k in range(8):
                        expect = val(g * 8 + k, cs_abs * 16 + i)
                        assert frag[i][k] == expect, (
                            wid, g, ni, i, k, frag[i][k], expect)


def simulate_v6_A(Cout=64):
    def val(opix, co):
        return ("A", opix, co)

    lds = {}
    for og in range(4):  # wave og deposits its 8-opix group
        base_elem = og * 1024 // 2
        for lane in range(64):
            q = (lane >> 5) & 1
            cs = (lane >> 3) & 3
            r2 = (lane & 7) >> 1
            h = lane & 1
            opix = og * 8 + q * 4 + r2
            co8 = cs * 2 + h
            for e in range(8):
                lds[base_elem + lane * 8 + e] = val(opix, co8 * 8 + e)

    for g in range(4):
        for mi in range(4):
            frag = [[None] * 8 for _ in range(16)]
            for h2 in range(2):
                base = ((2 * g + h2) * 4 + mi) * 128
                addrs = [base + l15 * 8 for l15 in range(16)]
                out = tr_read(lds, addrs)
                for i in range(16):
                    for j in range(4):
                        frag[i][h2 * 4 + j] = out[i][j]
            for i in range(16):
                for k in range(8):
                    expect = val(g * 8 + k, mi * 16 + i)
                    assert frag[i][k] == expect, (g, mi, i, k)


def test_v6_b_operand_addressing():
    simulate_v6_B()


def test_v6_a_operand_addressing():
    simulate_v6_A()


def test_tr_semantics_match_probe():
    """The Python model reproduces the device probe's pattern-3 output
    (scripts/probe_tr16.hip): lanes 0..15 with addrs l*8 over a flat
    image receive columns of the [4][16] block."""
    lds = {i: i for i in range(256)}
    out = tr_read(lds, [l * 8 for l in range(16)])
    for i in range(16):
        assert out[i] == [i, i + 16, i + 32, i + 48]


def simulate_gemm_tn_operand(which="B"):
    """gemm_tn (csrc/conv_wrw.hip): stage instr (wave wid, ii) deposits
    1 KiB at mq*1024 (mq = wid*2+ii); lane -> (ns = l>>3, r = (l&7)>>1,
    h = l&1) loading col-octet ns*16 + h*8 of m-row mq*4 + r. Fragment:
    block (mq*8 + ns)*128, k-halves at +0/+1024."""
    def val(m, col):
        return (which, m, col)

    lds = {}
    for mq in range(8):
        base_elem = mq * 1024 // 2
        for lane in range(64):
            ns = lane >> 3
            r = (lane & 7) >> 1
            h = lane & 1
            m = mq * 4 + r
            c0 = ns * 16 + h * 8
            for e in range(8):
                lds[base_elem + lane * 8 + e] = val(m, c0 + e)

    for g in range(4):          # lane group = k octet
        for ns in range(8):     # 16-col subtile
            frag = [[None] * 8 for _ in range(16)]
            for h2 in range(2):
                base = ((2 * g + h2) * 8 + ns) * 128
                addrs = [base + l15 * 8 for l15 in range(16)]
                out = tr_read(lds, addrs)
                for i in range(16):
                    for j in range(4):
                        frag[i][h2 * 4 + j] = out[i][j]
            for i in range(16):
                for k in range(8):
                    expect = val(g * 8 + k, ns * 16 + i)
                    assert frag[i][k] == expect, (g, ns, i, k)


def test_gemm_tn_operand_addressing():
    simulate_gemm_tn_operand()
