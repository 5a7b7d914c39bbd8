"""CPU guards for the GEMM LDS-skew bijections (native/gemm.hip).

The skews are pure integer maps; these tests replicate them and assert
(a) skew/unskew are inverse bijections on the whole image, and (b) the
bank-window distinctness property each skew was derived for — so a future
edit that silently breaks the conflict-freedom fails CI without a GPU.
"""


def bf16_skew(e):  # [128][64] bf16 elements
    row, k = e >> 6, e & 63
    return (row << 6) | ((k + 16 * ((row >> 1) & 3)) & 63)


def bf16_unskew(y):
    row, k = y >> 6, y & 63
    return (row << 6) | ((k - 16 * ((row >> 1) & 3)) & 63)


def fp8_skew(e):  # [128][64] bytes
    row, k = e >> 6, e & 63
    return (row << 6) | ((k + 16 * ((row >> 2) & 3)) & 63)


def mx_skew(e):  # [128][128] bytes
    row, k = e >> 7, e & 127
    return (row << 7) | ((k + 16 * ((row >> 1) & 5)) & 127)


def i8_skew(e):  # [128][64] bytes (plain i8)
    row, k = e >> 6, e & 63
    return (row << 6) | ((k + 32 * ((row >> 3) & 1)) & 63)


def i8s_skew(e):  # [128][128] bytes (8-phase i8)
    row, k = e >> 7, e & 127
    return (row << 7) | ((k + 32 * ((row >> 1) & 3)) & 127)


def test_skews_are_row_preserving_bijections():
    for skew, size, rowshift in ((bf16_skew, 128 * 64, 6),
                                 (fp8_skew, 128 * 64, 6),
                                 (mx_skew, 128 * 128, 7),
                                 (i8_skew, 128 * 64, 6),
                                 (i8s_skew, 128 * 128, 7)):
        img = [skew(e) for e in range(size)]
        assert sorted(img) == list(range(size)), skew.__name__
        assert all((skew(e) >> rowshift) == (e >> rowshift)
                   for e in range(size)), skew.__name__


def test_bf16_unskew_inverts_skew():
    for e in range(128 * 64):
        assert bf16_unskew(bf16_skew(e)) == e


def _bank_window(byte_addr, width_dwords=4):
    return (byte_addr // 4) % 64 // width_dwords


def test_bf16_true_lane_groups_conflict_free():
    """The gfx950 ds_read_b128 lane groups (MI355X_MICROARCH.md): all 16
    lanes of each group must land on 16 distinct 4-dword windows."""
    groups = [
        [0, 1, 2, 3, 12, 13, 14, 15, 20, 21, 22, 23, 24, 25, 26, 27],
        [4, 5, 6, 7, 8, 9, 10, 11, 16, 17, 18, 19, 28, 29, 30, 31],
        [32, 33, 34, 35, 44, 45, 46, 47, 52, 53, 54, 55, 56, 57, 58, 59],
        [36, 37, 38, 39, 40, 41, 42, 43, 48, 49, 50, 51, 60, 61, 62, 63],
    ]
    for kk in (0, 32):  # both MFMA K-steps of a 64-element tile row
        for grp in groups:
            windows = set()
            for lane in grp:
                row = lane & 15
                kfrag = kk + 8 * (lane >> 4)
                e = bf16_skew(row * 64 + kfrag)  # element index
                windows.add(_bank_window((e & 63) * 2 + (e >> 6) * 128))
            assert len(windows) == 16, (kk, grp, sorted(windows))


def test_i8s_lane_groups_conflict_free():
    groups = [
        [0, 1, 2, 3, 12, 13, 14, 15, 20, 21, 22, 23, 24, 25, 26, 27],
        [4, 5, 6, 7, 8, 9, 10, 11, 16, 17, 18, 19, 28, 29, 30, 31],
        [32, 33, 34, 35, 44, 45, 46, 47, 52, 53, 54, 55, 56, 57, 58, 59],
        [36, 37, 38, 39, 40, 41, 42, 43, 48, 49, 50, 51, 60, 61, 62, 63],
    ]
    for kk in (0, 64):
        for grp in groups:
            windows = set()
            for lane in grp:
                row = lane & 15
                kfrag = kk + 16 * (lane >> 4)
                y = i8s_skew(row * 128 + kfrag)
                windows.add(_bank_window((y & 127) + (y >> 7) * 128))
            assert len(windows) == 16, (kk, grp, sorted(windows))


# ---------------------------------------------------------------------------
# matmul_nt padding math (ops.gemm_pad_shapes) — pure CPU
# ---------------------------------------------------------------------------

def test_gemm_pad_shapes():
    from hpc_patterns_amd.ops import gemm_pad_shapes

    # already-aligned shapes are untouched
    assert gemm_pad_shapes("bf16", 256, 512, 128) == (256, 512, 128)
    assert gemm_pad_shapes("i8", 256, 256, 256) == (256, 256, 256)
    assert gemm_pad_shapes("mxfp8", 128, 128, 128) == (128, 128, 128)
    # odd shapes round UP to the fast-path multiples
    assert gemm_pad_shapes("bf16", 300, 520, 130) == (512, 768, 256)
    assert gemm_pad_shapes("fp8", 1, 1, 1) == (256, 256, 128)
    assert gemm_pad_shapes("i8", 100, 300, 700) == (256, 512, 768)
    assert gemm_pad_shapes("mxfp8", 129, 127, 257) == (256, 128, 384)
    # padded result always satisfies the kernel's divisibility contract
    for kind, (qm, qk) in (("bf16", (256, 128)), ("fp8", (256, 128)),
                           ("i8", (256, 256)), ("mxfp8", (128, 128))):
        for m, n, k in ((1, 1, 1), (300, 520, 736), (511, 513, 257)):
            mp, np_, kp = gemm_pad_shapes(kind, m, n, k)
            assert mp % (128 if kind == "mxfp8" else qm) == 0
            assert np_ % (128 if kind == "mxfp8" else qm) == 0
            assert kp % qk == 0
            assert mp >= m and np_ >= n and kp >= k
    import pytest
    with pytest.raises(ValueError):
        gemm_pad_shapes("fp16", 128, 128, 128)


def test_group_remap_bijective():
    """Replicates hpk_group_remap (gemm.hip): the grouped tile order must
    be a bijection on [0, nwg) for every band width, including grids whose
    tiles_n is not a multiple of the group."""
    def remap(wg, tiles_n, nwg, group):
        if group <= 1:
            return wg
        tiles_m = nwg // tiles_n
        band = group * tiles_m
        b = wg // band
        within = wg - b * band
        gw = min(group, tiles_n - b * group)
        return (within // gw) * tiles_n + b * group + within % gw

    for tiles_m, tiles_n in ((32, 32), (4, 7), (7, 4), (1, 5), (5, 1),
                             (16, 48), (3, 3)):
        nwg = tiles_m * tiles_n
        for group in (1, 2, 3, 4, 8, 16, 64):
            out = {remap(w, tiles_n, nwg, group) for w in range(nwg)}
            assert out == set(range(nwg)), (tiles_m, tiles_n, group)
    # grouped order covers a band of `group` columns before moving on:
    # the first tiles_m remapped ids all live in columns [0, group)
    first = {remap(w, 32, 32 * 32, 8) % 32 for w in range(8 * 32)}
    assert first == set(range(8))


def test_fp4_natural_layout_conflict_free():
    """The MX-fp4 tile ([128 rows][64 packed bytes], fragment = 16
    contiguous bytes at (row, 16*g)) needs NO skew: the 64-byte row
    stride naturally rotates rows across the b128 bank windows. Enumerate
    the true gfx950 b128 lane groups (as for bf16) and assert all 16
    lanes of each group hit 16 distinct 16-byte windows."""
    groups = [
        [0, 1, 2, 3, 12, 13, 14, 15, 20, 21, 22, 23, 24, 25, 26, 27],
        [4, 5, 6, 7, 8, 9, 10, 11, 16, 17, 18, 19, 28, 29, 30, 31],
        [32, 33, 34, 35, 44, 45, 46, 47, 52, 53, 54, 55, 56, 57, 58, 59],
        [36, 37, 38, 39, 40, 41, 42, 43, 48, 49, 50, 51, 60, 61, 62, 63],
    ]
    for mrow_base in (0, 16, 32):  # several fragment row offsets
        for grp in groups:
            windows = set()
            for lane in grp:
                row = mrow_base + (lane & 15)
                g = lane >> 4
                byte = row * 64 + 16 * g
                windows.add((byte // 16) % 32)
            assert len(windows) == 16, (mrow_base, grp, sorted(windows))


def test_e2m1_pack_decode_roundtrip():
    import torch

    from hpc_patterns_amd import ops

    vals = torch.tensor([0., 0.5, 1., 1.5, 2., 3., 4., 6.])
    t = torch.cat([vals, -vals]).repeat(4).view(2, -1)  # [2, 32]
    p = ops.e2m1_pack(t)
    assert p.dtype == torch.uint8 and p.shape == (2, 16)
    d = ops.e2m1_decode(p)
    assert torch.equal(d.abs(), t.abs())
    assert bool((d[t != 0] == t[t != 0]).all())
    import pytest
    with pytest.raises(ValueError):
        ops.e2m1_pack(torch.tensor([[5.0, 1.0]]))
    with pytest.raises(ValueError):
        ops.e2m1_pack(torch.tensor([[1.0, 2.0, 3.0]]))  # odd last dim


def test_mx4_32x32_chunk_rotation_conflict_free():
    """The 32x32x64 kernel's 64-byte-stride rows 8-way-conflict without a
    skew (two rows per (4*row mod 32) window band in every true lane
    group); the (row>>3)&3 chunk rotation separates them. Enumerate the
    b128 lane groups over all fragment windows and k-steps."""
    groups = [
        [0, 1, 2, 3, 12, 13, 14, 15, 20, 21, 22, 23, 24, 25, 26, 27],
        [4, 5, 6, 7, 8, 9, 10, 11, 16, 17, 18, 19, 28, 29, 30, 31],
        [32, 33, 34, 35, 44, 45, 46, 47, 52, 53, 54, 55, 56, 57, 58, 59],
        [36, 37, 38, 39, 40, 41, 42, 43, 48, 49, 50, 51, 60, 61, 62, 63],
    ]
    for base in (0, 32, 64, 96, 128, 224):  # fragment 32-row windows
        for kk in (0, 1):
            for grp in groups:
                windows = set()
                for lane in grp:
                    row = base + (lane & 31)
                    g = lane >> 5
                    ch = (g + 2 * kk + (row >> 3)) & 3
                    byte = row * 64 + 16 * ch
                    windows.add((byte // 16) % 32)
                assert len(windows) == 16, (base, kk, grp, sorted(windows))
    # and WITHOUT the rotation it is genuinely 8-way broken
    bad = set()
    for lane in groups[0]:
        row = lane & 31
        byte = row * 64 + 16 * (lane >> 5)
        bad.add((byte // 16) % 32)
    assert len(bad) < 16


def test_gemm_pad_shapes_mxfp4():
    from hpc_patterns_amd.ops import gemm_pad_shapes

    assert gemm_pad_shapes("mxfp4", 256, 512, 128) == (256, 512, 128)
    assert gemm_pad_shapes("mxfp4", 200, 136, 192) == (256, 256, 256)
    mp, np_, kp = gemm_pad_shapes("mxfp4", 1, 1, 32)
    assert mp % 256 == 0 and np_ % 256 == 0 and kp % 128 == 0


def test_mx8_32x32_chunk_rotation_conflict_free():
    """The fp8 32x32x64 kernel's 128-byte rows put FOUR rows in each
    (8*row mod 32) window band per true lane group; the (row>>2)&7 chunk
    rotation separates them (their row>>2 values are distinct mod 8)."""
    groups = [
        [0, 1, 2, 3, 12, 13, 14, 15, 20, 21, 22, 23, 24, 25, 26, 27],
        [4, 5, 6, 7, 8, 9, 10, 11, 16, 17, 18, 19, 28, 29, 30, 31],
        [32, 33, 34, 35, 44, 45, 46, 47, 52, 53, 54, 55, 56, 57, 58, 59],
        [36, 37, 38, 39, 40, 41, 42, 43, 48, 49, 50, 51, 60, 61, 62, 63],
    ]
    for base in (0, 32, 96, 224):
        for kk in (0, 1):
            for h in (0, 1):
                for grp in groups:
                    windows = set()
                    for lane in grp:
                        row = base + (lane & 31)
                        g = lane >> 5
                        ch = (4 * kk + 2 * h + g + (row >> 2)) & 7
                        byte = row * 128 + 16 * ch
                        windows.add((byte // 16) % 32)
                    assert len(windows) == 16, (base, kk, h, sorted(windows))
    # without the rotation: 4-way broken
    bad = set()
    for lane in groups[0]:
        byte = (lane & 31) * 128 + 16 * (lane >> 5)
        bad.add((byte // 16) % 32)
    assert len(bad) <= 8
