"""Executable proof of the 8-phase conv/GEMM schedule invariants
(csrc/conv_igemm8.hip / tools/gemm8.hip — derivation in
docs/IGEMM8_PLAN.md).

Simulates the staging automaton — half-image glds issue order, the two
LDS buffers, the vmcnt(6) checkpoints — and asserts, for every K-tile
count:

  1. every consumed half-image was staged (right region, right K-tile);
  2. its glds pair is RETIRED (covered by a vmcnt checkpoint) before the
     first phase that reads it;
  3. no region is overwritten while a phase that reads its previous
     content can still be running (writers must start strictly after the
     last reader's phase).

A schedule edit that silently breaks the race-freedom argument fails
here on CPU, long before it corrupts tensors on a GPU.
"""


def simulate(nkt: int):
    """Returns (stages, events): stages[i] = (phase, kt, op, kh) in issue
    order (phase -1 = prologue); events = per-phase consumed regions and
    the retire horizon after each checkpoint."""
    stages = []
    # prologue: kt0 all 4 halves + kt1.(A,k0),(B,k0); vmcnt(4) afterwards
    for (op, kh) in [(0, 0), (1, 0), (0, 1), (1, 1)]:
        stages.append((-1, 0, op, kh))
    for (op, kh) in [(0, 0), (1, 0)]:
        stages.append((-1, 1, op, kh))
    prologue_retired = len(stages) - 2   # vmcnt(4): all but newest 2 pairs

    phases = []
    for a in range(nkt * 4):
        hp = (a + 2) & 3
        kt_s = (a + 6) >> 2
        stages.append((a, kt_s, hp & 1, hp >> 1))
        kt_c, kh_c = a >> 2, (a >> 1) & 1
        phases.append({"a": a, "consumes": [(kt_c, 0, kh_c), (kt_c, 1, kh_c)],
                       "checkpoint": a % 4 == 0})
    return stages, phases, prologue_retired


def retire_horizon(stages, upto_phase):
    """Index below which every staging pair is GUARANTEED retired when
    phase `upto_phase` starts reading (after its checkpoint if it has
    one): the most recent vmcnt(6) (or the prologue vmcnt(4)) that
    executed at or before that point."""
    # checkpoints execute at phases a % 4 == 0, AFTER that phase's own
    # glds; they leave the newest 3 pairs unretired.
    best = 0
    # prologue vmcnt(4): leaves newest 2 of the 6 prologue pairs
    best = 4
    for i, (ph, *_rest) in enumerate(stages):
        if ph == -1:
            continue
        if ph % 4 == 0 and ph <= upto_phase:
            # vmcnt(6) after the glds issued AT phase ph: everything up
            # to (index of that glds) - 3 is retired
            best = max(best, i + 1 - 3)
    return best


def test_schedule_invariants():
    for nkt in range(1, 11):
        stages, phases, _ = simulate(nkt)
        # region content: (buf, op, kh) -> list of (kt, issue_index)
        for ph in phases:
            a = ph["a"]
            for (kt, op, kh) in ph["consumes"]:
                buf = kt & 1
                # 1. staged at all?
                cands = [i for i, (p, k, o, h) in enumerate(stages)
                         if k == kt and o == op and h == kh]
                assert cands, f"nkt={nkt} phase {a}: ({kt},{op},{kh}) never staged"
                idx = cands[0]
                # 2. retired before this phase reads it
                horizon = retire_horizon(stages, a)
                assert idx < horizon, (
                    f"nkt={nkt} phase {a}: half (kt={kt},op={op},kh={kh}) "
                    f"staged at issue {idx} but retire horizon is {horizon}")

        # 3. write-after-read: a staging into region (buf,op,kh) for K-tile
        # kt2 must start strictly after the LAST phase reading the region's
        # previous occupant kt2-2
        for (p, kt, op, kh) in stages:
            prev_kt = kt - 2
            if prev_kt < 0 or prev_kt >= nkt:
                continue
            last_read = max(a for a in range(nkt * 4)
                            if a >> 2 == prev_kt and ((a >> 1) & 1) == kh)
            assert p > last_read, (
                f"nkt={nkt}: staging ({kt},{op},{kh}) at phase {p} "
                f"overwrites content still readable until phase {last_read}")


def test_dummy_stages_target_dead_or_safe_regions():
    """Clamped (kt >= NKT) stagings still write SOME region; they must obey
    the same write-after-read rule for that region's live content."""
    for nkt in range(1, 9):
        stages, _, _ = simulate(nkt)
        for (p, kt, op, kh) in stages:
            if kt < nkt:
                continue
            buf = kt & 1
            # live tiles that share this buffer and k-half
            live = [t for t in range(nkt) if (t & 1) == buf]
            for t in live:
                last_read = max(a for a in range(nkt * 4)
                                if a >> 2 == t and ((a >> 1) & 1) == kh)
                first_next_write = p
                # dummy write at phase p must come after t's last reader
                # OR t's content must already have been replaced earlier
                replaced_earlier = any(
                    (pp < p and k2 != t and k2 > t and (k2 & 1) == buf and
                     o2 == op and h2 == kh)
                    for (pp, k2, o2, h2) in stages if k2 < nkt + 2)
                assert p > last_read or replaced_earlier, (
                    f"nkt={nkt}: dummy staging ({kt},{op},{kh}) at phase {p}"
                    f" could race tile {t} read until {last_read}")


def test_pipe_prefetch_legality_rule():
    """The PIPE=1 interleave (tools/conv8.hip) issues phase p+1's fragment
    reads inside phase p's MFMA region.  Legal iff every image phase p+1
    consumes is retired by a checkpoint that executed at or before phase p.
    The derived rule — prefetch on every phase except the last of each
    4-phase group (p % 4 == 3) — must be exactly what the automaton
    permits."""
    for nkt in range(1, 9):
        stages, phases, _ = simulate(nkt)
        for p in range(nkt * 4 - 1):
            horizon = retire_horizon(stages, p)
            legal = True
            for (kt, op, kh) in phases[p + 1]["consumes"]:
                cands = [i for i, (ph, k, o, h) in enumerate(stages)
                         if k == kt and o == op and h == kh]
                if not cands or cands[0] >= horizon:
                    legal = False
            assert legal == (p % 4 != 3), (
                f"nkt={nkt} phase {p}: prefetch legality {legal} != rule "
                f"{(p % 4 != 3)}")
