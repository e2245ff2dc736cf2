"""Ping-pong GEMM stage-schedule verifier (the tool that caught the r2
deep-schedule certificate bug the hand derivation missed).

Models, per ping-pong group, the VMEM FIFO (4 glds per half-tile unit),
the per-phase stage targets, the end-of-phase vmcnt waits (wait+barrier =
certificate), LDS slot contents by buffer parity, and every fragment read.
Flags any read of an uncertified or overwritten slot. Mirrors the schedule
compiled into src/hip/gemm.hip gemm_bf16_nt_pingpong_k / _pers_k — keep in
sync when editing the kernel (two-lane discipline: sync-structure edits are
new templates).
"""

STAGES = {  # phase -> (group, (kind, half, tile_offset))
    0: (1, ('A', 0, +2)), 1: (0, ('A', 1, +2)),
    2: (1, ('B', 0, +2)), 3: (0, ('B', 1, +2)),
    4: (1, ('A', 0, +3)), 5: (0, ('A', 1, +3)),
    6: (1, ('B', 0, +3)), 7: (0, ('B', 1, +3)),
}
WAITS = {  # end-of-phase -> {group: vmcnt N (4 glds per unit)}
    0: {0: 8}, 2: {1: 8}, 3: {0: 12}, 4: {0: 8}, 6: {1: 8}, 7: {0: 12, 1: 8},
}
READS = {  # phase -> (group, [(kind, half, tile_offset)])
    0: (1, [('A', 1, 0), ('B', 0, 0)]),
    1: (0, [('B', 1, 0)]),
    2: (1, [('B', 1, 0)]),
    3: (0, [('A', 0, 1), ('B', 0, 1)]),
    4: (1, [('A', 1, 1), ('B', 0, 1)]),
    5: (0, [('B', 1, 1)]),
    6: (1, [('B', 1, 1)]),
    7: (0, [('A', 0, 2), ('B', 0, 2)]),
}
PROLOGUE = {
    1: [('A', 0, 0), ('B', 0, 0), ('A', 0, 1), ('B', 0, 1)],
    0: [('A', 1, 0), ('B', 1, 0), ('A', 1, 1), ('B', 1, 1)],
}
PROLOGUE_KEEP = {0: 3, 1: 2}  # units allowed in flight at the prologue wait


def verify(ntiles):
    fifo = {0: [], 1: []}
    certified = set()
    slot = {}
    errors = []
    for g, units in PROLOGUE.items():
        for u in units:
            fifo[g].append(u)
            slot[(u[0], u[1], u[2] & 1)] = u
    for g, keep in PROLOGUE_KEEP.items():
        for u in fifo[g][:len(fifo[g]) - keep]:
            certified.add(u)
    for u in [('A', 0, 0), ('B', 0, 0)]:  # G0 pre-read
        if u not in certified:
            errors.append(f"preread uncertified {u}")
    for t in range(0, ntiles, 2):
        for p in range(8):
            g, rl = READS[p]
            for kind, half, dtt in rl:
                tile = t + dtt
                if tile >= ntiles:
                    continue  # guarded in the kernel
                u = (kind, half, tile)
                cur = slot.get((kind, half, tile & 1))
                if cur != u:
                    errors.append(f"t={t} p{p}: slot holds {cur}, want {u}")
                elif u not in certified:
                    errors.append(f"t={t} p{p}: UNCERTIFIED read {u}")
            sg, (kind, half, dtt) = STAGES[p]
            tile = t + dtt
            if tile < ntiles:
                u = (kind, half, tile)
                fifo[sg].append(u)
                slot[(kind, half, tile & 1)] = u
            tail = (t + 2 >= ntiles)
            if tail:
                for g2 in (0, 1):
                    certified.update(fifo[g2])
            elif p in WAITS:
                for g2, n in WAITS[p].items():
                    keep = n // 4
                    for u in fifo[g2][:max(0, len(fifo[g2]) - keep)]:
                        certified.add(u)
    return errors


if __name__ == "__main__":
    ok = True
    for nt in (4, 6, 8, 12, 24, 64, 512):
        errs = verify(nt)
        print(f"ntiles={nt}: {'OK' if not errs else 'FAIL'}")
        for e in errs[:6]:
            print("  ", e)
        ok = ok and not errs
    raise SystemExit(0 if ok else 1)
