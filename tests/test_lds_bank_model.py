"""LDS bank-conflict verification of the rotated tile layouts used by the
HIP kernels — the in-tree model behind docs/DESIGN.md rule 2.  Pure CPU:
models the CDNA4 per-instruction lane-group banking rules and replays the
exact address patterns of attention.hip (t_rot) and lora_gemm.hip (tr64).

Measured ground truth these tests encode: the naive transposed layouts
were 8/16-way conflicted on writes (16% of dkdv wave cycles, 44% in
skinny_grad — profiles/README.md r1.8); the rotated layouts are
conflict-free on the hd64 transpose writes and cap every other pattern at
2-way."""

import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))), "tools"))

from lds_bank_model import (  # noqa: E402
    access_cycles, attn_ldsT_frag_instructions, attn_t_elem,
    attn_t_elem_naive, attn_tile_write_t_instructions,
    lora_frag_read_instructions, lora_stage_write_instructions, verify)


def test_model_reproduces_guide_t2_case():
    """Self-check of the model: the textbook row-major [R][128B-stride]
    b128 read with row=lane and fixed column is 16-way per group, and the
    (row&15)<<4 XOR swizzle makes it conflict-free — the documented T2
    behavior the PMC counters confirmed on hardware."""
    stride = 256  # bytes: a [R][128] bf16 row-major tile
    linear = {lane: lane * stride for lane in range(64)}
    cycles, min_cycles = access_cycles(linear, "read_b128")
    assert cycles // min_cycles == 16

    swizzled = {lane: lane * stride ^ ((lane & 15) << 4) for lane in range(64)}
    cycles, min_cycles = access_cycles(swizzled, "read_b128")
    assert cycles == min_cycles


def test_attention_rotated_writes_hd64_conflict_free():
    ok, ways = verify(attn_tile_write_t_instructions(64), "write_b16")
    assert ok and ways == 1, f"hd64 rotated transpose writes are {ways}-way"


def test_attention_rotated_patterns_cap_at_2way():
    for hd in (64, 128):
        _, w_ways = verify(attn_tile_write_t_instructions(hd), "write_b16")
        _, r_ways = verify(attn_ldsT_frag_instructions(hd), "read_b128")
        assert w_ways <= 2, (hd, w_ways)
        assert r_ways <= 2, (hd, r_ways)


def test_attention_naive_layout_is_heavily_conflicted():
    """The regression the rotation fixed: every column write of one
    instruction lands on one bank."""
    _, w64 = verify(attn_tile_write_t_instructions(64, attn_t_elem_naive),
                    "write_b16")
    _, w128 = verify(attn_tile_write_t_instructions(128, attn_t_elem_naive),
                     "write_b16")
    assert w64 >= 8 and w128 >= 16, (w64, w128)
    _, r_ways = verify(attn_ldsT_frag_instructions(64, attn_t_elem_naive),
                       "read_b128")
    assert r_ways >= 4, r_ways


def test_lora_tr64_patterns_cap_at_2way():
    # X^T staging: thread t -> (row block (t%16)*8, column t//16)
    _, xt_ways = verify(lora_stage_write_instructions(
        64 * 16, lambda t: (t % 16) * 8, lambda t: t // 16), "write_b16")
    # P^T staging at rtile=128: thread t -> (row (t%16)*8, col t//16)
    _, pt_ways = verify(lora_stage_write_instructions(
        64 * 16, lambda t: (t % 16) * 8, lambda t: t // 16), "write_b16")
    _, rd_ways = verify(lora_frag_read_instructions(
        row0s=range(0, 128, 16), k0s=(0, 32)), "read_b128")
    assert xt_ways <= 2 and pt_ways <= 2 and rd_ways <= 2, \
        (xt_ways, pt_ways, rd_ways)


def test_v2_layout_fully_conflict_free_hd64():
    """The round-2 candidate layout (Q_V2 table) removes the residual
    2-way read class of the current rotation: conflict-free on BOTH
    writes and reads at hd64."""
    from lds_bank_model import attn_t_elem_v2
    wok, w_ways = verify(attn_tile_write_t_instructions(64, attn_t_elem_v2),
                         "write_b16")
    rok, r_ways = verify(attn_ldsT_frag_instructions(64, attn_t_elem_v2),
                         "read_b128")
    assert wok and w_ways == 1, w_ways
    assert rok and r_ways == 1, r_ways


def test_v2_layout_hd128_reads_conflict_free_writes_at_floor():
    from lds_bank_model import attn_t_elem_v2
    _, w_ways = verify(attn_tile_write_t_instructions(128, attn_t_elem_v2),
                       "write_b16")
    rok, r_ways = verify(attn_ldsT_frag_instructions(128, attn_t_elem_v2),
                         "read_b128")
    assert w_ways == 2, w_ways  # provable floor of the stride-64 family
    assert rok and r_ways == 1, r_ways


def test_v2_layout_bijective():
    from lds_bank_model import attn_t_elem_v2
    for hd in (64, 128):
        seen = {attn_t_elem_v2(c, kv) for c in range(hd) for kv in range(64)}
        assert len(seen) == hd * 64


def test_v2_layout_lora_orientation():
    """Q_V2 on the lora tr64 orientation: reads conflict-free, writes at
    the 2-way floor, bijective over the 128x64 tile."""
    from lds_bank_model import tr64_v2

    def stage_writes():
        for it in range(2):
            for w in range(8):
                for j in range(8):
                    addrs = {}
                    for lane in range(64):
                        t = w * 64 + lane + it * 512
                        if t >= 64 * 16:
                            continue
                        addrs[lane] = 2 * tr64_v2((t % 16) * 8 + j, t // 16)
                    if addrs:
                        yield addrs

    def frag_reads():
        for row0 in range(0, 128, 16):
            for k0 in (0, 32):
                yield {lane: 2 * tr64_v2(row0 + (lane & 15),
                                         k0 + (lane >> 4) * 8)
                       for lane in range(64)}

    _, w_ways = verify(stage_writes(), "write_b16")
    ok, r_ways = verify(frag_reads(), "read_b128")
    assert w_ways == 2, w_ways
    assert ok and r_ways == 1, r_ways
    assert len({tr64_v2(r, c) for r in range(128) for c in range(64)}) == 128 * 64


def test_rotation_preserves_layout_bijectivity():
    """Sanity: the rotated mapping is a bijection on the 64x64 tile (no
    element aliasing) for both layout families."""
    seen = {attn_t_elem(c, kv) for c in range(64) for kv in range(64)}
    assert len(seen) == 64 * 64
    from lds_bank_model import tr64
    seen = {tr64(r, c) for r in range(128) for c in range(64)}
    assert len(seen) == 128 * 64


def test_rowmajor_default_padding_writes_cf_reads_2way():
    """The shipped default (8-element padding, no swizzle): staging writes
    conflict-free, fragment reads 2-way."""
    from lds_bank_model import (attn_k_tile_reads, attn_k_tile_writes,
                                attn_p_tile_reads, attn_p_tile_writes)
    for hd in (64, 128):
        ok, w = verify(attn_k_tile_writes(hd, 8, False), "write_b128")
        assert ok and w == 1, (hd, w)
        _, r = verify(attn_k_tile_reads(hd, 8, False), "read_b128")
        assert r == 2, (hd, r)
    ok, w = verify(attn_p_tile_writes(8, False), "write_b16")
    assert ok and w == 1, w
    _, r = verify(attn_p_tile_reads(8, False), "read_b128")
    assert r == 2, r


def test_rowmajor_v2_swizzle_fully_conflict_free():
    """RELORA_AMD_ROT_V2 row-major path (zero pad + per-row XOR): every K/P
    write and read conflict-free at hd64 AND hd128 — and it saves the
    padding LDS."""
    from lds_bank_model import (attn_k_tile_reads, attn_k_tile_writes,
                                attn_p_tile_reads, attn_p_tile_writes)
    for hd in (64, 128):
        ok, w = verify(attn_k_tile_writes(hd, 0, True), "write_b128")
        assert ok and w == 1, (hd, "write", w)
        ok, r = verify(attn_k_tile_reads(hd, 0, True), "read_b128")
        assert ok and r == 1, (hd, "read", r)
    ok, w = verify(attn_p_tile_writes(0, True), "write_b16")
    assert ok and w == 1, w
    ok, r = verify(attn_p_tile_reads(0, True), "read_b128")
    assert ok and r == 1, r


def test_rowmajor_swizzle_bijective():
    from lds_bank_model import rm_swz
    for ld in (64, 128):
        seen = {(row * ld + col) ^ rm_swz(row, ld, True)
                for row in range(64) for col in range(ld)}
        assert len(seen) == 64 * ld


def test_lora_rowmajor_v2_swizzle_conflict_free_at_r_multiple_64():
    """lora_add (lora_skinny_kernel) row-major tiles under ROT_V2 (pad0 +
    per-row XOR): staging writes, A-fragment reads, epilogue C-dump and
    re-read all conflict-free at the flagship r=128 (and every r%64==0)."""
    from lds_bank_model import access_cycles

    def swz(row, ld):
        return ((row & 15) if (ld & 127) == 0 else (row & 7)) << 3

    for r in (64, 128, 192, 256):
        ldt, r8, OLD = r, r // 8, 128

        def p_stage():
            total = 128 * r8
            for it in range((total + 255) // 256):
                for w in range(4):
                    addrs = {}
                    for lane in range(64):
                        t = w * 64 + lane + it * 256
                        if t >= total:
                            continue
                        row, c = t // r8, (t % r8) * 8
                        addrs[lane] = 2 * ((row * ldt + c) ^ swz(row, ldt))
                    if addrs:
                        yield addrs

        def a_reads():
            for w in range(4):
                wr = (w >> 1) * 64
                for mi in range(4):
                    for kk in range(0, r, 32):
                        yield {l: 2 * (((wr + mi * 16 + (l & 15)) * ldt
                                        + kk + (l >> 4) * 8)
                                       ^ swz(wr + mi * 16 + (l & 15), ldt))
                               for l in range(64)}

        def o_reads():
            total = 128 * 16
            for it in range((total + 255) // 256):
                for w in range(4):
                    addrs = {}
                    for lane in range(64):
                        t = w * 64 + lane + it * 256
                        if t >= total:
                            continue
                        row, c8 = t // 16, (t % 16) * 8
                        addrs[lane] = 2 * ((row * OLD + c8) ^ swz(row, OLD))
                    if addrs:
                        yield addrs

        for gen, kind in ((p_stage(), "write_b128"), (a_reads(), "read_b128"),
                          (o_reads(), "read_b128")):
            for addrs in gen:
                c, m = access_cycles(addrs, kind)
                assert c == m, (r, kind, c, m)


def test_v2_dominates_current_at_all_rtiles():
    """skinny_grad stages rtile=min(r-r0,128) rows; Q_V2's conflict profile
    is better-or-equal to the current rotation at every rtile: fragment
    reads conflict-free everywhere (currently 2-way), staging writes at the
    same level."""
    from lds_bank_model import access_cycles, tr64, tr64_v2

    def worst(gen, kind):
        w = 1
        for addrs in gen:
            c, m = access_cycles(addrs, kind)
            w = max(w, c // m if m else 1)
        return w

    for rtile in (32, 64, 96, 128):
        def pt_writes(fn):
            total = 64 * (rtile // 8)
            for it in range((total + 255) // 256):
                for w in range(4):
                    for j in range(8):
                        addrs = {}
                        for lane in range(64):
                            t = w * 64 + lane + it * 256
                            if t >= total:
                                continue
                            mm = t // (rtile // 8)
                            j8 = (t % (rtile // 8)) * 8
                            addrs[lane] = 2 * fn(j8 + j, mm)
                        if addrs:
                            yield addrs

        def a_reads(fn):
            for w in range(4):
                for i in range(2):
                    for ks in (0, 1):
                        addrs = {}
                        for l in range(64):
                            jrow = w * 32 + i * 16 + (l & 15)
                            if jrow >= rtile:
                                continue
                            addrs[l] = 2 * fn(jrow, ks * 32 + (l >> 4) * 8)
                        if addrs:
                            yield addrs

        assert worst(a_reads(tr64_v2), "read_b128") == 1, rtile
        assert (worst(pt_writes(tr64_v2), "write_b16")
                <= worst(pt_writes(tr64), "write_b16")), rtile


def test_v2_transq_never_worse_than_current():
    """lora_add_nt's TRANSQ q_im tile (rot8 inside ldt-strided rows):
    ROT_V2 (pad0 + Q_V2 rot8) is conflict-free on reads at r=64 and equal
    to the current layout elsewhere.  (r>=128 reads are pigeonhole-bound
    at 2-way for any 8-deep rotation in a 256-byte row — a 16-deep
    whole-row permutation is the round-2 escape hatch.)"""
    from lds_bank_model import Q_V2, access_cycles

    def rot8_cur(row, c64):
        return ((((c64 >> 3) + (row >> 3) + (row & 7)) & 7) << 3) + (c64 & 7)

    def rot8_v2(row, c64):
        return (((Q_V2[row & 15][c64 >> 3] + 2 * (row >> 4)) & 7) << 3) + (c64 & 7)

    def worst(gen, kind):
        w = 1
        for addrs in gen:
            c, m = access_cycles(addrs, kind)
            w = max(w, c // m if m else 1)
        return w

    def patterns(rot8, lpad, r):
        ldt = r + lpad

        def tw():
            total = r * 16
            for it in range((total + 255) // 256):
                for w in range(4):
                    for j in range(8):
                        addrs = {}
                        for lane in range(64):
                            t = w * 64 + lane + it * 256
                            if t >= total:
                                continue
                            k, nb = t // 16, (t % 16) * 8
                            addrs[lane] = 2 * ((nb + j) * ldt + (k & ~63)
                                               + rot8(nb + j, k & 63))
                        if addrs:
                            yield addrs

        def tr():
            for w in range(4):
                wc = (w & 1) * 64
                for ni in range(4):
                    for kk in range(0, r, 32):
                        yield {l: 2 * ((wc + ni * 16 + (l & 15)) * ldt
                                       + ((kk + (l >> 4) * 8) & ~63)
                                       + rot8(wc + ni * 16 + (l & 15),
                                              (kk + (l >> 4) * 8) & 63))
                               for l in range(64)}
        return tw, tr

    for r in (64, 128, 256):
        tw_c, tr_c = patterns(rot8_cur, 8, r)
        tw_v, tr_v = patterns(rot8_v2, 0, r)
        assert worst(tw_v(), "write_b16") <= worst(tw_c(), "write_b16"), r
        assert worst(tr_v(), "read_b128") <= worst(tr_c(), "read_b128"), r
    # r=64 reads fully conflict-free under v2
    _, tr_v64 = patterns(rot8_v2, 0, 64)
    assert worst(tr_v64(), "read_b128") == 1


def test_16deep_transq_layout_reads_conflict_free():
    """The closed-form 16-deep permutation solves the one pattern Q_V2
    cannot (TRANSQ q_im at r=128, 256-byte rows): reads conflict-free,
    writes at the 2-way floor, bijective."""
    from lds_bank_model import access_cycles, perm16, transq_elem_16deep

    r = 128

    def tw():
        total = r * 16
        for it in range((total + 255) // 256):
            for w in range(4):
                for j in range(8):
                    addrs = {}
                    for lane in range(64):
                        t = w * 64 + lane + it * 256
                        if t >= total:
                            continue
                        k, nb = t // 16, (t % 16) * 8
                        addrs[lane] = 2 * transq_elem_16deep(nb + j, k)
                    if addrs:
                        yield addrs

    def tr():
        for w in range(4):
            wc = (w & 1) * 64
            for ni in range(4):
                for kk in range(0, r, 32):
                    yield {l: 2 * transq_elem_16deep(wc + ni * 16 + (l & 15),
                                                    kk + (l >> 4) * 8)
                           for l in range(64)}

    for addrs in tr():
        c, m = access_cycles(addrs, "read_b128")
        assert c == m, (c, m)
    worst = 1
    for addrs in tw():
        c, m = access_cycles(addrs, "write_b16")
        worst = max(worst, c // m)
    assert worst == 2
    assert len({transq_elem_16deep(row, k)
                for row in range(128) for k in range(128)}) == 128 * 128
    for row in (0, 5, 127):
        assert len({perm16(row, g) for g in range(16)}) == 16
