"""Byte-level model of the attention LDS image swizzles (CPU, no GPU).

Models the V-transposed image staging/read addressing of
vilbert_multi_task_amd/ops/csrc/attention.hip (transpose scatter at
~line 167, B-fragment reads at the PV loop) and asserts the row-masked
swizzle (SWZR) keeps the map collision-free and read-exact for every
LK_PAD. This is the exact model that caught the round-2 LK_PAD=32 bug:
the unmasked swizzle (up to 112 B) escapes a 64-byte row, collides
writes across rows, and misroutes 224/2048 B-fragment reads
(profiles/r09_final_round2.md §4).
"""

import pytest


def swz(row: int) -> int:
    # attention.hip SWZ: st_16x32 XOR pattern over a 128-byte-row image
    return ((row & 7) ^ ((row >> 3) & 7)) << 4


def swzr(row: int, row_bytes: int) -> int:
    # attention.hip SWZR: masked to the image's actual row length
    return swz(row) & (row_bytes - 1)


def v_image_map(lk_pad: int, d_dim: int, masked: bool):
    """Returns (collisions, bad_reads) for the VT [D][LK_PAD] image."""
    sw = (lambda r: swzr(r, lk_pad * 2)) if masked else swz
    mem = {}
    collisions = 0
    # staging: 256 threads, KCH = D/8 column chunks, transpose scatter
    kch = d_dim // 8
    rows_per_pass = 256 // kch
    npass = (lk_pad + rows_per_pass - 1) // rows_per_pass
    for tid in range(256):
        r0, c = tid // kch, tid % kch
        for pi in range(npass):
            r = r0 + pi * rows_per_pass
            if r >= lk_pad:
                break
            for j in range(8):
                d = c * 8 + j
                addr = d * (lk_pad * 2) + ((r * 2) ^ sw(d))
                if addr in mem:
                    collisions += 1
                mem[addr] = (r, d)
    # B-fragment reads: lane l holds V[key = kk*32 + (l>>4)*8 + j][col]
    bad = 0
    for kk in range(lk_pad // 32):
        for lane in range(64):
            for nt in range(d_dim // 16):
                d = nt * 16 + (lane & 15)
                keyoff = kk * 64 + (lane >> 4) * 16
                base = d * (lk_pad * 2) + (keyoff ^ sw(d))
                for j in range(8):
                    want = (kk * 32 + (lane >> 4) * 8 + j, d)
                    if mem.get(base + 2 * j) != want:
                        bad += 1
    return collisions, bad


@pytest.mark.parametrize("lk_pad", [32, 64, 128])
@pytest.mark.parametrize("d_dim", [64, 128])
def test_masked_swizzle_is_collision_free_and_read_exact(lk_pad, d_dim):
    collisions, bad = v_image_map(lk_pad, d_dim, masked=True)
    assert collisions == 0, f"LK_PAD={lk_pad} D={d_dim}: {collisions} write collisions"
    assert bad == 0, f"LK_PAD={lk_pad} D={d_dim}: {bad} misrouted B-frag reads"


def test_unmasked_swizzle_collides_at_lk_pad_32():
    """The r2 bug: documents WHY the mask exists (guards against 'simplifying'
    SWZR back to SWZ)."""
    collisions, bad = v_image_map(32, 64, masked=False)
    assert collisions > 0 and bad > 0


def test_mask_is_identity_at_128_byte_rows():
    # LK_PAD >= 64 rows are 128+ bytes: SWZ <= 112 already fits — the masked
    # and unmasked forms must agree (no behavior change for serving shapes)
    for lk_pad in (64, 128):
        for row in range(128):
            assert swzr(row, lk_pad * 2) == swz(row)
