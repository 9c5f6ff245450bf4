"""Adversarial-timing verification of csrc/gemm_256_8phase.hip's
prefetch/compute SCHEDULE — the other half of the blind kernel (the
addressing half is test_gemm256_addressing.py).

Model: phases are globally synchronized (every phase ends with an
s_barrier all waves pass). A glds to a half-tile slot may complete at
ANY time from its issue phase onward; the only lower bound on completion
comes from `s_waitcnt vmcnt(4)` (all but the newest 2 half-tile stagings
= 4 loads are complete in every wave, and a following barrier makes that
global). A ds_read of slot S expecting tile t's data at phase p is
correct iff BOTH:

  (1) retire:   the staging of (t, S) is vmcnt-retired before p, and
  (2) overwrite: any LATER staging to slot S (which could complete
      immediately under adversarial timing) is issued at a phase
      strictly after the last ds_read of the previous content —
      equivalently, after phase p if it would clobber what p reads.

The test replays the kernel's exact issue program (prologue + in-loop
rotation + per-phase ds_reads with register residency) and asserts both
conditions for every read over many K-tile counts."""

import pytest

# half-tile ids: 0=A0 1=A1 2=B0 3=B1 ; slot = (half, tile % 2)


def issue_program(total_kt):
    """Reproduce the kernel's staging issue order as a list of
    (phase_index, tile, half). Phase index: prologue stagings get
    negative indices (they all precede phase 0's barrier); in-loop phase
    p of tile t is global index t*4 + p."""
    prog = []
    # prologue: B0(0) B1(0) A0(0) A1(0) B0(1) B1(1)
    pro = [(0, 2), (0, 3), (0, 0), (0, 1), (1, 2), (1, 3)]
    for i, (t, h) in enumerate(pro):
        prog.append((-(len(pro)) + i, t, h))
    for t in range(total_kt):
        for p in range(4):
            if p < 2:
                if t + 1 < total_kt:
                    prog.append((t * 4 + p, t + 1, p))      # A0/A1 of t+1
            else:
                if t + 2 < total_kt:
                    prog.append((t * 4 + p, t + 2, p))      # B0/B1 of t+2
    return prog


def read_program(total_kt):
    """ds_reads per phase with the kernel's register residency:
    p0 of tile t reads A0(t) and ALL of B0(t)+B1(t)... no: p0 reads
    A-m0 (half A0's rows 0-63 + A1's rows... both A halves are read at
    p0 (different waves) and BOTH B halves at p0 (bfr preload).
    p2 re-reads both A halves (m-sub 1 rows). Expressed per half-tile:
      p0: A0, A1, B0, B1 (first rows / all n-frags)
      p2: A0, A1 (second m-sub rows)
    Returns (phase_index, tile, half, is_last_read_of_tile)."""
    reads = []
    for t in range(total_kt):
        base = t * 4
        for h in (0, 1, 2, 3):
            reads.append((base + 0, t, h, h >= 2))  # B: only read at p0
        for h in (0, 1):
            reads.append((base + 2, t, h, True))    # A: last read at p2
    return reads


@pytest.mark.parametrize("total_kt", [4, 6, 8, 20])
def test_retire_and_overwrite_hazards(total_kt):
    prog = issue_program(total_kt)
    reads = read_program(total_kt)

    # vmcnt + barrier at the END of phase t*4+3 (and after the prologue,
    # modeled as end of phase -1): vmcnt(4) keeps the newest 2 stagings
    # in flight, EXCEPT at the tail (no B stagings followed the A's:
    # t+2 >= total_kt) where the kernel drains with vmcnt(0).
    def retired_before(phase):
        """Return set of stagings globally known complete before `phase`
        begins, under the weakest guarantee (only vmcnt waits count)."""
        done = set()
        waits = [(-1, 2)] + [(4 * t + 3, 2 if t + 2 < total_kt else 0)
                             for t in range(total_kt)]
        for wphase, keep in waits:
            if wphase >= phase:
                break
            issued = [e for e in prog if e[0] <= wphase]
            for e in (issued[:-keep] if keep else issued):
                done.add(e)
        return done

    # (1) retire: every read's staging must be retired before its phase
    for (p, t, h, _last) in reads:
        done = retired_before(p)
        staged = [e for e in prog if e[1] == t and e[2] == h]
        assert staged, (t, h)
        assert staged[0] in done, (
            f"read of tile {t} half {h} at phase {p} not retired")

    # (2) overwrite: a staging to slot (h, t%2) may complete the moment
    # it issues; every read of the PREVIOUS tenant (tile t-2... the
    # prior tile with same parity) must be at a phase strictly before
    # the staging's issue phase.
    for (ip, t, h) in prog:
        prev_t = t - 2
        if prev_t < 0:
            continue
        prev_reads = [r for r in reads if r[1] == prev_t and r[2] == h]
        for (rp, _, _, _) in prev_reads:
            assert rp < ip, (
                f"staging of tile {t} half {h} issued at phase {ip} "
                f"can clobber tile {prev_t} read at phase {rp}")


def test_all_tiles_fully_staged(total_kt=10):
    prog = issue_program(total_kt)
    for t in range(total_kt):
        halves = sorted(h for (_, tt, h) in prog if tt == t)
        assert halves == [0, 1, 2, 3], (t, halves)
