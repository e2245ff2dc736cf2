"""Certificate/slot-lifetime verifier for the 4-phase pp64q variant
(gemm_bf16_pp64q4_k): 4 fat phases per 2 k-iters, A0/B0 double-buffered
(G1-staged), A1/B1 TRIPLE-buffered (G0-staged), vmcnt<=4 waits for both
groups at end-P0 and end-P2 only. Mirrors src/hip/gemm.hip — keep in sync.

Counts are in glds units of 4 KB (one stage() call = 2 glds = one 8 KB
64x64 half-unit; a pair like "A0B0(t)" = 4 glds).

Checks per phase, in program order:
  1. reads: slot must hold the wanted unit AND the unit must be certified
  2. stage legality: every read of the slot's current occupant must have
     been ISSUED in an earlier phase, or earlier in this phase (the kernel
     drains lgkm between its reads and its stage issues)
  3. end-of-phase waits certify own-FIFO prefixes; barrier publishes
"""

NBUF = {('A', 0): 2, ('B', 0): 2, ('A', 1): 3, ('B', 1): 3}

# phase -> (computing group, reading group, [(kind,half,dt)], [stages (kind,half,dt)])
# reading group == staging group (reads then lgkm-drains then stages).
PHASES = [
    (0, 1, [('A', 1, 0), ('B', 0, 0), ('B', 1, 0)], [('A', 0, +2), ('B', 0, +2)]),
    (1, 0, [('A', 0, 1), ('B', 0, 1), ('B', 1, 1)], [('A', 1, +3), ('B', 1, +3)]),
    (0, 1, [('A', 1, 1), ('B', 0, 1), ('B', 1, 1)], [('A', 0, +3), ('B', 0, +3)]),
    (1, 0, [('A', 0, 2), ('B', 0, 2), ('B', 1, 2)], [('A', 1, +4), ('B', 1, +4)]),
]
WAITS = {0: {0: 4, 1: 4}, 2: {0: 4, 1: 4}}  # end-of-phase -> {group: vmcnt}

PROLOGUE = {
    1: [('A', 0, 0), ('B', 0, 0), ('A', 0, 1), ('B', 0, 1)],
    0: [('A', 1, 0), ('B', 1, 0), ('A', 1, 1), ('B', 1, 1),
        ('A', 1, 2), ('B', 1, 2)],
}
PREREAD = [('A', 0, 0), ('B', 0, 0), ('B', 1, 0)]  # G0, before the loop


def verify(ntiles, verbose=False):
    fifo = {0: [], 1: []}           # own-FIFO of staged units, in issue order
    certified = set()
    slot = {}                        # (kind,half,buf) -> unit
    last_read_phase = {}             # unit -> global phase index of last read
    errors = []
    gp = -1                          # global phase counter; prologue = -1

    def place(g, u):
        kind, half, tile = u
        key = (kind, half, tile % NBUF[(kind, half)])
        old = slot.get(key)
        if old is not None and old in last_read_phase and \
           last_read_phase[old] >= gp and old != u:
            # a read of the occupant was issued in THIS phase after... the
            # kernel orders reads before stages with an lgkm drain, so
            # same-phase reads are legal; only future reads are races —
            # checked below when that read happens.
            pass
        slot[key] = u
        fifo[g].append(u)

    for g, units in PROLOGUE.items():
        for u in units:
            place(g, u)
    for g in (0, 1):                 # full drain
        certified.update(fifo[g])
        fifo[g] = []
    for u in PREREAD:
        if u not in certified:
            errors.append(f"preread uncertified {u}")
        last_read_phase[u] = gp

    for t in range(0, ntiles, 2):
        for p, (cg, rg, reads, stages) in enumerate(PHASES):
            gp = t * 2 + p
            for kind, half, dt in reads:
                tile = t + dt
                if tile >= ntiles:
                    continue
                u = (kind, half, tile)
                key = (kind, half, tile % NBUF[(kind, half)])
                cur = slot.get(key)
                if cur != u:
                    errors.append(f"t={t} P{p}: slot {key} holds {cur}, want {u}")
                elif u not in certified:
                    errors.append(f"t={t} P{p}: UNCERTIFIED read {u}")
                last_read_phase[u] = gp
            for kind, half, dt in stages:
                tile = t + dt
                if tile >= ntiles:
                    continue
                u = (kind, half, tile)
                key = (kind, half, tile % NBUF[(kind, half)])
                old = slot.get(key)
                if old is not None and old != u:
                    # every read of the occupant must already be issued
                    # (same phase ok: kernel lgkm-drains reads first). A
                    # FUTURE read of old would be a DMA-overwrite race.
                    for (fp, (fcg, frg, freads, _)) in enumerate(PHASES):
                        for k2, h2, dt2 in freads:
                            if (k2, h2) != (kind, half):
                                continue
                            for t2 in range(t, ntiles, 2):
                                gp2 = t2 * 2 + fp
                                if gp2 <= gp:
                                    continue
                                if t2 + dt2 == old[2] and t2 + dt2 < ntiles:
                                    errors.append(
                                        f"t={t} P{p}: stage {u} overwrites "
                                        f"{old} still read at t={t2} P{fp}")
                place(rg, u)
            tail = t + 2 >= ntiles   # kernel: tail => waits become vmcnt(0)
            if p in WAITS:
                for g2, n in WAITS[p].items():
                    # fifo entries are UNITS of 2 glds (one stage call);
                    # vmcnt<=4 glds = keep 2 stage-call units in flight.
                    keep = 0 if tail else n // 2
                    for u in fifo[g2][:max(0, len(fifo[g2]) - keep)]:
                        certified.add(u)
                    fifo[g2] = fifo[g2][-keep:] if keep else []
    return errors


if __name__ == "__main__":
    import sys
    bad = 0
    for nt in (4, 6, 8, 12, 24, 48, 64):
        errs = verify(nt)
        if errs:
            bad += 1
            print(f"ntiles={nt}: {len(errs)} errors")
            for e in errs[:8]:
                print("   ", e)
        else:
            print(f"ntiles={nt}: OK")
    sys.exit(1 if bad else 0)
