"""Full-dataflow emulation of csrc/gemm_256_8phase.hip at matrix level:
LDS slots are mutated at each staging's ISSUE phase (adversarial-early
completion), ds_reads take whatever the slot holds at their phase, and
operands persist in 'registers' exactly where the kernel keeps them
(A m-sub for 2 phases, the wave's whole B span for the 4-phase tile).
If the staging rotation, buffer parity, register residency or quadrant
bookkeeping were wrong, the final C would not equal A @ B^T.

(Timing legality — writes not landing before prior reads finish — is
verified separately in test_gemm256_schedule.py; addressing in
test_gemm256_addressing.py. This file checks the DATA.)"""

import torch

BK = 64


def run_workgroup(A, B, total_kt):
    """A: (256, K), B: (256, K); returns emulated C (256, 256)."""
    # slots[(which, parity)] = current (128, 64) half-tile payload
    # which: 0=A0 1=A1 2=B0 3=B1
    slots = {}

    def stage(t, which):
        src = A if which < 2 else B
        rows = (which % 2) * 128
        slots[(which, t & 1)] = \
            src[rows:rows + 128, t * BK:(t + 1) * BK].clone()

    # prologue: B0(0) B1(0) A0(0) A1(0) B0(1) B1(1)
    for (t, h) in [(0, 2), (0, 3), (0, 0), (0, 1), (1, 2), (1, 3)]:
        stage(t, h)

    # per-wave accumulators: wave (wr in 2, wc in 4) owns C rows
    # wr*128..+128, cols wc*64..+64
    acc = {(wr, wc): torch.zeros(128, 64) for wr in range(2)
           for wc in range(4)}

    for t in range(total_kt):
        buf = t & 1
        a_reg = {}
        b_reg = {}
        for p in range(4):
            # ds_reads of this phase (before staging mutates anything
            # in the same phase would be illegal; schedule test proves
            # staging of phase p never clobbers content read at p, so
            # order within the phase does not matter for legal data)
            for wr in range(2):
                for wc in range(4):
                    if p == 0:
                        a_reg[(wr, wc)] = \
                            slots[(wr, buf)][0:64].clone()      # m-sub 0
                        bh = 2 + (wc >> 1)
                        c0 = (wc & 1) * 64
                        b_reg[(wr, wc)] = \
                            slots[(bh, buf)][c0:c0 + 64].clone()
                    elif p == 2:
                        a_reg[(wr, wc)] = \
                            slots[(wr, buf)][64:128].clone()    # m-sub 1
            # staging rotation (issue == adversarial-immediate landing)
            if p < 2:
                if t + 1 < total_kt:
                    stage(t + 1, p)              # A0/A1 of t+1
            else:
                if t + 2 < total_kt:
                    stage(t + 2, p)              # B0/B1 of t+2
            # MFMA quadrant: p0:(m0,n0) p1:(m0,n1) p2:(m1,n1) p3:(m1,n0)
            msub = 1 if p >= 2 else 0
            nsub = 1 if p in (1, 2) else 0
            for wr in range(2):
                for wc in range(4):
                    a = a_reg[(wr, wc)]                       # (64, 64k)
                    b = b_reg[(wr, wc)][nsub * 32:nsub * 32 + 32]
                    acc[(wr, wc)][msub * 64:msub * 64 + 64,
                                  nsub * 32:nsub * 32 + 32] += a @ b.t()

    C = torch.zeros(256, 256)
    for (wr, wc), blockacc in acc.items():
        C[wr * 128:wr * 128 + 128, wc * 64:wc * 64 + 64] = blockacc
    return C


def test_dataflow_produces_the_product():
    torch.manual_seed(0)
    for total_kt in (4, 6, 9):
        K = total_kt * BK
        A = torch.randn(256, K)
        B = torch.randn(256, K)
        C = run_workgroup(A, B, total_kt)
        torch.testing.assert_close(C, A @ B.t(), rtol=1e-4, atol=1e-3)
