"""CPU simulation of csrc/gemm_256_8phase.hip's LDS addressing — the
part of a blind-written kernel most likely to be wrong and the part that
CAN be verified without a GPU. Replicates the exact formulas:

  swz(rel)   = rel ^ (((rel>>9)&1)<<5)           (st_16x32 swizzle)
  stage_half: glds g, wave w, lane L writes LDS bytes
              d = ((g*8+w)*8)*128 + L*16 .. +16  (lane-linear dest)
              from logical half-tile bytes swz(d) .. +16
  read_a/b:   16B fragment of logical (row, kf) read at swz(row*128+kf*2)

and checks (a) staging covers every LDS byte exactly once, (b) the
swizzle keeps each 16B load contiguous in the logical image, and
(c) every fragment read returns exactly the logical operand bytes.
If these pass, only the SYNC schedule remains to be validated on device
(scripts/validate_gemm256.py race screen)."""

BK = 64
HALF_BYTES = 128 * BK * 2  # 16 KiB half-tile image


def swz(rel):
    return rel ^ (((rel >> 9) & 1) << 5)


def test_swizzle_is_involution_and_16B_contiguous():
    for rel in range(0, HALF_BYTES, 16):
        assert swz(swz(rel)) == rel
        base = swz(rel)
        assert base % 16 == 0
        # bytes of a 16B-aligned block move as one contiguous block
        # (the swizzle only flips bit 5, a 32B granule)
        for b in range(16):
            assert swz(rel + b) == base + b


def test_staging_covers_lds_once_and_roundtrips():
    logical = list(range(0, HALF_BYTES))  # logical byte id = its offset
    lds = [None] * HALF_BYTES
    for g in range(2):
        for wid in range(8):
            for lane in range(64):
                d = ((g * 8 + wid) * 8) * 128 + lane * 16
                lg = swz(d)
                for b in range(16):
                    assert lds[d + b] is None, "double write"
                    lds[d + b] = logical[lg + b]
    assert all(v is not None for v in lds), "uncovered LDS bytes"

    # fragment reads: every (row, kf) 16B read returns logical bytes
    for row in range(128):
        for kf in range(0, BK, 8):
            rel = row * 128 + kf * 2
            phys = swz(rel)
            got = lds[phys:phys + 16]
            assert got == logical[rel:rel + 16], (row, kf)


def test_stage_source_rows_are_16B_runs():
    """Each lane's glds source (row, kb) must be 16B-aligned inside a
    row (global memory is read 16B-contiguous per lane)."""
    for g in range(2):
        for wid in range(8):
            for lane in range(64):
                d = ((g * 8 + wid) * 8) * 128 + lane * 16
                lg = swz(d)
                row, kb = lg >> 7, lg & 127
                assert 0 <= row < 128
                assert kb % 16 == 0 and kb + 16 <= 128


def test_c_write_matches_read_layout():
    """The epilogue's (row, col) per acc[mi][ni][j] must tile the wave's
    128x64 output exactly once (WARPS_M=2 x WARPS_N=4 waves -> 256x256)."""
    seen = set()
    for wid in range(8):
        wr, wc = wid // 4, wid % 4
        for lane in range(64):
            for mi in range(8):
                for ni in range(4):
                    for j in range(4):
                        row = wr * 128 + mi * 16 + (lane >> 4) * 4 + j
                        col = wc * 64 + ni * 16 + (lane & 15)
                        assert (row, col) not in seen
                        seen.add((row, col))
    assert len(seen) == 256 * 256
