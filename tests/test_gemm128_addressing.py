"""CPU replication of the device-VALIDATED 128² GEMM's (gemm_masked.hip)
and conv-wrw's (conv_wrw.hip) LDS addressing. These kernels already pass
their on-device numerics suites; these tests pin the index math so
round-2 tuning edits (8-phase conversion, double-buffered wrw staging)
can be checked here before spending GPU budget."""

BK = 64
TILE_BYTES = 128 * BK * 2


def lds_byte(row, k):
    blk = (k >> 3) ^ (row & 7)
    return row * (BK * 2) + blk * 16 + (k & 7) * 2


def test_gemm_bt_stage_roundtrips_and_covers():
    """stage(): wave wid, iter i, lane -> glds dest (wid*32+i*8)*128 +
    lane*16 from source row `row`, 16B block `lblk ^ (row&7)`; fragment
    reads use lds_byte(row, kf). Staging must cover the 16 KiB image
    once and every fragment read must see logical (row, k) data."""
    logical = lambda row, byte: row * 1000_000 + byte  # unique ids
    lds = [None] * TILE_BYTES
    for wid in range(4):
        for i in range(4):
            for lane in range(64):
                lrow, lblk = lane >> 3, lane & 7
                row = wid * 32 + i * 8 + lrow
                src_blk = lblk ^ (row & 7)
                dest = (wid * 32 + i * 8) * (BK * 2) + lane * 16
                for b in range(16):
                    assert lds[dest + b] is None
                    lds[dest + b] = logical(row, src_blk * 16 + b)
    assert all(v is not None for v in lds)
    for row in range(128):
        for kf in range(0, BK, 8):
            base = lds_byte(row, kf)
            for b in range(16):
                assert lds[base + b] == logical(row, kf * 2 + b), (row, kf)


def test_gemm_bt_c_layout_tiles_once():
    seen = set()
    for wid in range(4):
        wr, wc = wid // 2, wid % 2
        for lane in range(64):
            for mi in range(4):
                for ni in range(4):
                    for j in range(4):
                        row = wr * 64 + mi * 16 + (lane >> 4) * 4 + j
                        col = wc * 64 + ni * 16 + (lane & 15)
                        assert (row, col) not in seen
                        seen.add((row, col))
    assert len(seen) == 128 * 128


def test_wrw_transpose_scatter_roundtrips():
    """conv_wrw stage A: thread (op = tid&63, co8 = tid>>6), iters it:
    loads gy[opix][co0..co0+8] and scatters 8 ds_write_b16 to
    lds_byte(co0+j, op). Fragment reads lds_byte(row, kf) must then see
    [co][opix]-major data."""
    img = {}
    for tid in range(256):
        op = tid & 63
        co8 = tid >> 6
        for it in range(4):
            co0 = (co8 + it * 4) * 8
            for j in range(8):
                addr = lds_byte(co0 + j, op)
                assert addr not in img or img[addr] == (co0 + j, op)
                img[addr] = (co0 + j, op)   # logical (co, opix)
    # every element written exactly once across 128 x 64
    assert len(img) == 128 * 64
    for row in range(128):
        for kf in range(0, BK, 8):
            for j in range(8):
                # element (row, kf+j) lives at lds_byte(row, kf+j)
                assert img[lds_byte(row, kf + j)] == (row, kf + j)
