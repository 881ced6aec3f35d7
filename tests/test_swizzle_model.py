"""Python models of the LDS swizzles in ops/attest_kernels.hip, locking
the invariants that were enumerated by hand during kernel bring-up
(docs/KERNELS.md findings 2 and 5). If someone edits a swizzle in the
.hip file, these tests force the same analysis to be redone.

LDS bank model (CDNA4): 64 banks x 4 bytes; a ds_read_b128 processes 16
lanes per group; a group is conflict-free iff its 16 lanes' 4-byte-bank
quads are disjoint, i.e. lane -> (addr >> 4) & 15 is a bijection when
each lane reads 16 contiguous bytes.
"""


def swz(off: int) -> int:
    """bf16 3-bit swizzle (attest_kernels.hip:111): row bits 1-3 of the
    128-B-row image into byte bits 4-6."""
    return off ^ (((off >> 8) & 7) << 4)


def swz8(off: int) -> int:
    """fp8 2-bit swizzle (attest_kernels.hip:578): byte bits 5-6 only,
    keeping every 32-B fragment contiguous."""
    return off ^ (((off >> 8) & 3) << 5)


def bank_quad(addr: int) -> int:
    return (addr >> 4) & 15


def test_swizzles_are_involutions():
    """glds stages lds[x] = global[f(x)] and readers use f again, so f
    must be self-inverse."""
    for off in range(0, 1 << 15, 16):
        assert swz(swz(off)) == off
        assert swz8(swz8(off)) == off


def test_swz_bijective_image():
    """The swizzle must permute the tile image (no two logical offsets
    landing on one physical slot)."""
    size = 1 << 14  # one 16 KiB tile
    seen = {swz(o) for o in range(0, size, 16)}
    assert len(seen) == size // 16
    seen8 = {swz8(o) for o in range(0, size, 32)}
    assert len(seen8) == size // 32


def test_swz8_preserves_32B_fragment_contiguity():
    """fp8 fragments are single v8i (32-B) loads: the swizzle may only
    move whole 32-B blocks (XOR bits >= 5). Finding 5: an XOR below
    fragment size keyed on a row bit silently breaks MFMA byte pairing."""
    for off in range(0, 1 << 14, 32):
        base = swz8(off)
        for b in range(32):
            assert swz8(off + b) == base + b


def test_bf16_swizzle_conflict_free_for_fragment_pattern():
    """Finding 2: for the 16x16 and 32x32 fragment patterns
    (row-per-lane, 16-B half-K per lane) every 16-lane ds_read_b128
    group must hit 16 distinct bank quads. Enumerate all groups of both
    MFMA fragment shapes over a 128-B-row image."""
    ROW_B = 128
    # 16x16 shape: lanes l=0..63 read row=(l&15), khalf=(l>>4)*16
    for k_step in range(0, ROW_B, 64):  # 32-B k-window per op, 2 halves
        for group in range(4):  # lanes 16g..16g+15 share one cycle
            quads = set()
            for lane in range(16 * group, 16 * group + 16):
                row = lane & 15
                khalf = (lane >> 4) * 16
                addr = swz(row * ROW_B + k_step + khalf)
                quads.add(bank_quad(addr))
            assert len(quads) == 16, (k_step, group, sorted(quads))
    # 32x32 shape: row=(l&31), khalf=(l>>5)*16
    for k_step in range(0, ROW_B, 32):
        for group in range(4):
            quads = set()
            for lane in range(16 * group, 16 * group + 16):
                row = lane & 31
                khalf = (lane >> 5) * 16
                addr = swz(row * ROW_B + k_step + khalf)
                quads.add(bank_quad(addr))
            assert len(quads) == 16, (k_step, group, sorted(quads))


def test_fp8_swizzle_residual_conflicts_are_exactly_2way():
    """The fp8 trade (finding in BASELINE.md): swz8 keeps fragments
    contiguous at the cost of bounded 2-way conflicts — rows r and r+8
    alias. Assert the conflict degree never exceeds 2 for the 32x32
    fp8 fragment pattern (degree 4+ would be a regression; fully
    conflict-free was measured SLOWER, profiles/fp8_swz16.log)."""
    ROW_B = 128
    for k_step in (0, 32, 64, 96):
        for group in range(4):
            quad_counts = {}
            for lane in range(16 * group, 16 * group + 16):
                row = lane & 31
                kq = (lane >> 5) * 32
                # v8i = 2 consecutive 16-B quads; count the first
                addr = swz8(row * ROW_B + k_step + kq)
                q = bank_quad(addr)
                quad_counts[q] = quad_counts.get(q, 0) + 1
            assert max(quad_counts.values()) <= 2, (k_step, group, quad_counts)


def test_models_match_hip_source():
    """The Python models must stay textually in sync with the .hip
    definitions (cheap tripwire: the exact XOR expressions)."""
    src = open("k8s_cc_manager_amd/ops/attest_kernels.hip").read()
    assert "byte_off ^ (((byte_off >> 8) & 7) << 4)" in src  # swz
    assert "byte_off ^ (((byte_off >> 8) & 3) << 5)" in src  # swz8
