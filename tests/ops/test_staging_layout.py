"""Pin the attention staging-packet layout (csrc/attention.hip pkt_rc /
sub_idx) in pure Python: the conflict-free-staging claims of
profiles/attn_pmc_r02.md are geometric facts about these index maps, so
a CPU test can guard them against regressions.

Checks, per head dim D in {64, 128}:
- pkt_rc is a bijection packets -> (row, 16B column chunk);
- every 8-lane b128 store group covers all 32 LDS write banks exactly
  once (store bank = (byte_addr/4) % 32, 4 consecutive banks per 16-B
  store) — the measured fwd 24.3% -> 0.0% CONF property;
- each load instruction's lanes cover whole contiguous 256-B row slices
  (global coalescing preserved);
- sub_idx is a bijection onto the image and tiles stay 128-B contiguous
  (the ds_read_b64_tr_b16 requirement).
"""
import pytest


def pkt_rc(p, D):
    dlog = 3 if D // 16 == 8 else 2
    rw = (p >> 1) & 3
    dd = (p >> 3) & (D // 16 - 1)
    row = (p >> (3 + dlog)) * 4 + rw
    col = dd * 16 + (p & 1) * 8
    return row, col


def sub_idx(row, col, D):
    dblk_n = D // 16
    g = row >> 2
    dblk = (col >> 4) ^ (g & (dblk_n - 1))
    return g * (dblk_n * 64) + dblk * 64 + (row & 3) * 16 + (col & 15)


@pytest.mark.parametrize("D", [64, 128])
def test_packet_map_bijection_and_banks(D):
    QT = 64
    NP = QT * (D // 8)
    seen = set()
    for p in range(NP):
        row, col = pkt_rc(p, D)
        assert 0 <= row < QT and 0 <= col < D and col % 8 == 0
        seen.add((row, col))
    assert len(seen) == NP, "pkt_rc must be a bijection"

    # b128 stores: 8 contiguous lanes form one store group; each lane
    # writes 16 B at element index sub_idx(row, col) -> 4 banks
    for group_start in range(0, NP, 8):
        banks = []
        for p in range(group_start, group_start + 8):
            row, col = pkt_rc(p, D)
            byte = 2 * sub_idx(row, col, D)
            assert byte % 16 == 0
            word = byte // 4
            banks.extend((word + j) % 32 for j in range(4))
        assert sorted(banks) == list(range(32)), (
            f"store group at {group_start} does not cover all 32 banks: "
            f"{sorted(banks)}")

    # load coalescing: each 64-lane instruction (p = w*64 + l) covers
    # whole contiguous 256-B slices of rows
    NT = 512
    for i in range(NP // NT):
        for w in range(8):
            by_row = {}
            for lane in range(64):
                p = i * NT + w * 64 + lane
                row, col = pkt_rc(p, D)
                by_row.setdefault(row, []).append(col)
            for row, cols in by_row.items():
                cols = sorted(cols)
                assert cols[0] == 0 or cols == list(
                    range(cols[0], cols[0] + 8 * len(cols), 8))
                # contiguous 16-B chunks: stride exactly 8 elements
                assert all(b - a == 8 for a, b in zip(cols, cols[1:])), (
                    f"row {row} load not contiguous: {cols}")


@pytest.mark.parametrize("D", [64, 128])
def test_sub_idx_bijection_and_tile_contiguity(D):
    QT = 64
    seen = {}
    for row in range(QT):
        for col in range(0, D, 1):
            idx = sub_idx(row, col & ~15, D) + (col & 15)
            assert idx not in seen or seen[idx] == (row, col)
            seen[idx] = (row, col)
    assert len(seen) == QT * D, "sub_idx must cover the image exactly"

    # each [4][16] tile must be 128 B contiguous (tr16 requirement):
    # rows 4g..4g+3 of one logical d-block live at consecutive elements
    dblk_n = D // 16
    for g in range(QT // 4):
        for dlog in range(dblk_n):
            base = None
            elems = []
            for r in range(4):
                for c in range(16):
                    elems.append(sub_idx(4 * g + r, dlog * 16 + c, D))
            elems.sort()
            base = elems[0]
            assert elems == list(range(base, base + 64)), (
                f"tile (g={g}, d={dlog}) not contiguous")
