#!/usr/bin/env python3
"""Min-conflicts searcher for LDS tile layouts, driven by the bank model
(lds_bank_model.py).  This is the tool that produced the Q_V2 table
(`RELORA_AMD_ROT_V2`): give it the wave-instruction access patterns of a
planned kernel, and it searches the family of 8-element-block rotations
  elem(major, minor) = major*64 + f(major, minor>>3)*8 + (minor&7)
for a table f that minimizes (ideally zeroes) bank conflicts across ALL
the patterns simultaneously.  Any f is numerically valid (the layout is a
bijection as long as f(major,·) is a permutation) — so a solution can be
dropped into a kernel as a constant LUT without re-deriving correctness.

Usage (library):
    from search_lds_layout import search
    ok, table = search(patterns=[(gen_factory, kind), ...], n_major=64)
    # gen_factory(f) -> iterable of {lane: byte_addr} dicts, where f is
    # the candidate rotation function f(major, minor_grp) -> 0..7

Usage (CLI, reproduces the Q_V2 search for the attention transposed tile):
    python tools/search_lds_layout.py
"""

import random
import sys

sys.path.insert(0, __file__.rsplit("/", 1)[0])

from lds_bank_model import access_cycles  # noqa: E402


def total_conflicts(f, patterns):
    cost = 0
    for gen_factory, kind in patterns:
        for addrs in gen_factory(f):
            c, m = access_cycles(addrs, kind)
            cost += c - m
    return cost


def search(patterns, n_major, seed_f=None, trials=8, iters=30000, rng_seed=0):
    """Min-conflicts local search over rotation tables.

    patterns: list of (gen_factory, kind); gen_factory takes f(major, grp)
    and yields {lane: byte_addr} wave instructions.
    n_major: number of major-axis lines (channels/rows) the table covers.
    seed_f: optional callable to seed the table (e.g. the current layout).
    Returns (cost, table) with cost==0 meaning conflict-free everywhere.
    """
    rng = random.Random(rng_seed)
    best_cost, best_table = None, None
    for trial in range(trials):
        table = [[seed_f(c, g) if seed_f else rng.randrange(8)
                  for g in range(8)] for c in range(n_major)]
        if trial and seed_f:
            for c in range(n_major):
                if rng.random() < 0.3:
                    rng.shuffle(table[c])

        def f(major, grp):
            return table[major % n_major][grp & 7]

        # bijectivity per major line is part of the cost
        def cost_fn():
            c = total_conflicts(f, patterns)
            for row in table:
                c += 8 - len(set(row))
            return c

        cost = cost_fn()
        stall = 0
        for it in range(iters):
            if cost == 0:
                break
            ci, gi = rng.randrange(n_major), rng.randrange(8)
            old = table[ci][gi]
            table[ci][gi] = rng.randrange(8)
            c2 = cost_fn()
            if c2 <= cost:
                stall = stall + 1 if c2 == cost else 0
                cost = c2
            else:
                table[ci][gi] = old
                stall += 1
            if stall > 2000:
                for _ in range(6):
                    table[rng.randrange(n_major)][rng.randrange(8)] = rng.randrange(8)
                cost = cost_fn()
                stall = 0
        if best_cost is None or cost < best_cost:
            best_cost, best_table = cost, [row[:] for row in table]
        if cost == 0:
            break
    return best_cost, best_table


def main():
    """Demo: verify the shipped Q_V2 table is a zero-conflict solution for
    the attention transposed-tile patterns (and report the baseline)."""
    from lds_bank_model import (Q_V2, attn_ldsT_frag_instructions,
                                attn_tile_write_t_instructions)

    def make_elem(f):
        def elem(c, kv):
            return c * 64 + (f(c, kv >> 3) & 7) * 8 + (kv & 7)
        return elem

    def writes(f):
        return attn_tile_write_t_instructions(64, make_elem(f))

    def reads(f):
        return attn_ldsT_frag_instructions(64, make_elem(f))

    patterns = [(writes, "write_b16"), (reads, "read_b128")]

    def current(c, g):
        return (g + (c >> 3) + (c & 7)) & 7

    def q_v2(c, g):
        return (Q_V2[c & 15][g] + 2 * (c >> 4)) & 7

    print("hd64 transposed tile, conflict cost (0 = conflict-free):")
    print("  current t_rot :", total_conflicts(current, patterns))
    print("  Q_V2          :", total_conflicts(q_v2, patterns))
    # NOTE for from-scratch searches: a flat 64x8 table is a rough
    # landscape for local search.  The Q_V2 table was found by exploiting
    # structure: search a 16x8 pattern Q with a per-16-block shift
    # r(t)=2t (f(c,g) = (Q[c&15][g] + 2*(c>>4)) & 7) — the shift turns the
    # cross-block write constraint into a local odd-difference condition,
    # after which min-conflicts solves it in seconds.
    print("search seeded from Q_V2 (verifies the zero-cost solution):")
    cost, table = search(patterns, n_major=64, seed_f=q_v2, trials=1,
                         iters=1000)
    print("  search result cost:", cost)
    return 0 if cost == 0 else 1


if __name__ == "__main__":
    sys.exit(main())
