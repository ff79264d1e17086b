"""Wave-transposed layout builder (CPU): de-transposing every slot
must reproduce the multiset of original records, pad slots are
sentineled, and out-of-record bytes are newline fill (the device
cursor may peek past a record's end)."""

import numpy as np

from dragnet_amd.engine.gpu import _build_xpose_layout


def detranspose(xb, wave_base, slot_len, r, gran=64):
    length = int(slot_len[r])
    w, lane = r // 64, r % 64
    out = bytearray()
    for p in range(length):
        out.append(xb[int(wave_base[w]) + (p // gran) * (64 * gran)
                      + lane * gran + p % gran])
    return bytes(out)


def test_roundtrip_mixed_lengths():
    recs = [b"{}", b'{"a": 1}', b"x" * 200, b"", b'{"b": [1,2,3]}',
            b"y" * 70] * 30 + [b"z" * 513]
    buf = b"\n".join(recs) + b"\n"
    xb, wb, sl, nslots, n = _build_xpose_layout(buf)
    assert n == len(recs)
    assert nslots % 64 == 0 and nslots >= n
    got = sorted(detranspose(xb, wb, sl, r) for r in range(n))
    assert got == sorted(recs)
    # pad slots carry the sentinel
    for r in range(n, nslots):
        assert sl[r] == 0xFFFFFFFF
    # bytes past a record's end within its granule row are newline
    r0 = next(r for r in range(n) if sl[r] == 0)
    w, lane = r0 // 64, r0 % 64
    assert xb[int(wb[w]) + lane * 64] == 10


def test_skewed_lengths_compact():
    """Per-wave granule counts: one 8KB record must not inflate every
    wave (was a global-max prototype limitation)."""
    recs = [b"s" * 100] * 1000 + [b"L" * 8000]
    buf = b"\n".join(recs) + b"\n"
    xb, wb, sl, nslots, n = _build_xpose_layout(buf)
    # compact bound: ~2 granules per short record + the one big wave
    assert xb.size < 1000 * 2 * 64 * 2 + 64 * 8064 + 8192
    got = sorted(detranspose(xb, wb, sl, r) for r in range(n))
    assert got == sorted(recs)


def test_single_wave_and_exact_multiple():
    for count in (1, 64, 65, 128):
        recs = [b"abc"] * count
        buf = b"\n".join(recs) + b"\n"
        xb, wb, sl, nslots, n = _build_xpose_layout(buf)
        assert n == count and nslots == (count + 63) & ~63
        for r in range(n):
            assert detranspose(xb, wb, sl, r) == b"abc"


def test_granule_sizes():
    recs = [b"g" * 40, b"h" * 100, b"i" * 200] * 50
    buf = b"\n".join(recs) + b"\n"
    for gran in (32, 64, 128):
        xb, wb, sl, nslots, n = _build_xpose_layout(buf, gran)
        got = sorted(detranspose(xb, wb, sl, r, gran)
                     for r in range(n))
        assert got == sorted(recs), gran
