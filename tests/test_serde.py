"""SerializedPage wire-format tests (CPU-only — the node-boundary seam is
host-side).  Pins the restatement byte-for-byte against a hand-built wire
image constructed directly from the cited format
(PagesSerdeUtil.java:64-88, BlockEncodingManager.java:96-99,
LongArrayBlockEncoding.java:26-48, EncoderUtil.java:31-63), plus
round-trips incl. nulls and error paths."""
import ctypes as C
import pathlib
import struct
import zlib

import numpy as np
import pytest

REPO = pathlib.Path(__file__).resolve().parent.parent
SO = REPO / "presto_amd" / "libpresto_gpu.so"


@pytest.fixture(scope="module")
def L():
    import sys
    sys.path.insert(0, str(REPO))
    from presto_amd.engine import PgPage, PgCol  # noqa
    lib = C.CDLL(str(SO))
    lib.pg_last_error.restype = C.c_char_p
    lib.pg_page_serialize.argtypes = [C.c_void_p, C.c_void_p, C.c_int64,
                                      C.POINTER(C.c_int64)]
    lib.pg_page_deserialize.argtypes = [C.c_void_p, C.c_int64, C.c_void_p]
    lib.pg_page_free.argtypes = [C.c_void_p]
    return lib


def _page(cols, nulls=None):
    from presto_amd.engine import PgPage, PgCol, _NP_TAG
    pg = PgPage()
    arrs = list(cols.values())
    pg.n_rows = len(arrs[0])
    pg.n_cols = len(arrs)
    for i, a in enumerate(arrs):
        pg.cols[i].tag = _NP_TAG[a.dtype]
        pg.cols[i].on_device = 0
        pg.cols[i].data = a.ctypes.data
        if nulls is not None and nulls[i] is not None:
            pg.cols[i].null_mask = nulls[i].ctypes.data
        else:
            pg.cols[i].null_mask = None
    return pg


def _serialize(L, pg):
    buf = C.create_string_buffer(1 << 20)
    out_len = C.c_int64()
    st = L.pg_page_serialize(C.byref(pg), buf, len(buf), C.byref(out_len))
    assert st == 0, L.pg_last_error()
    return bytes(buf[:out_len.value])


def _expected_wire(n_rows, blocks):
    """Hand-built wire image per the cited format. blocks: list of
    (encoding_name, values_bytes, nullbits_bytes_or_None)."""
    body = struct.pack("<i", len(blocks))
    for name, vals, nulls in blocks:
        nb = name.encode()
        body += struct.pack("<i", len(nb)) + nb
        body += struct.pack("<i", n_rows)
        if nulls is None:
            body += b"\x00"
        else:
            body += b"\x01" + nulls
        body += vals
    crc = zlib.crc32(body)
    crc = zlib.crc32(bytes([0]), crc)
    crc = zlib.crc32(struct.pack("<i", n_rows), crc)
    crc = zlib.crc32(struct.pack("<i", len(body)), crc)
    meta = struct.pack("<iBiiq", n_rows, 0, len(body), len(body), crc)
    return meta + body


def test_wire_bytes_pinned(L):
    a = np.array([1, -2, 3], np.int64)
    b = np.array([10, 20, 30], np.int32)
    u = np.array([7, 8, 9], np.uint8)
    got = _serialize(L, _page({"a": a, "b": b, "u": u}))
    exp = _expected_wire(3, [
        ("LONG_ARRAY", a.tobytes(), None),
        ("INT_ARRAY", b.tobytes(), None),
        ("BYTE_ARRAY", u.tobytes(), None),
    ])
    assert got == exp


def test_wire_bytes_with_nulls(L):
    # 10 rows, nulls at 0, 3, 9 -> bits MSB-first: rows 0-7 = 0b10010000,
    # rows 8-9 = 0b01000000 (EncoderUtil.java:31-63); only non-null values
    a = np.arange(100, 110, dtype=np.int64)
    mask = np.zeros(10, np.uint8)
    mask[[0, 3, 9]] = 1
    got = _serialize(L, _page({"a": a}, nulls=[mask]))
    vals = a[mask == 0].tobytes()
    exp = _expected_wire(10, [
        ("LONG_ARRAY", vals, bytes([0b10010000, 0b01000000])),
    ])
    assert got == exp


def test_roundtrip(L):
    from presto_amd.engine import PgPage
    rng = np.random.default_rng(5)
    a = rng.integers(-2**62, 2**62, 1000)
    f = rng.random(1000)
    b = rng.integers(-2**31, 2**31 - 1, 1000).astype(np.int32)
    u = rng.integers(0, 256, 1000).astype(np.uint8)
    mask = (rng.random(1000) < 0.1).astype(np.uint8)
    wire = _serialize(L, _page({"a": a, "f": f, "b": b, "u": u},
                               nulls=[mask, None, None, mask]))
    out = PgPage()
    st = L.pg_page_deserialize(wire, len(wire), C.byref(out))
    assert st == 0, L.pg_last_error()
    assert out.n_rows == 1000 and out.n_cols == 4
    got_a = np.ctypeslib.as_array(
        C.cast(out.cols[0].data, C.POINTER(C.c_int64)), (1000,))
    got_f = np.ctypeslib.as_array(
        C.cast(out.cols[1].data, C.POINTER(C.c_double)), (1000,))
    got_b = np.ctypeslib.as_array(
        C.cast(out.cols[2].data, C.POINTER(C.c_int32)), (1000,))
    got_u = np.ctypeslib.as_array(
        C.cast(out.cols[3].data, C.POINTER(C.c_uint8)), (1000,))
    keep = mask == 0
    assert np.array_equal(got_a[keep], a[keep])
    # F64 travels as LONG_ARRAY bits; reinterpret
    assert np.array_equal(got_f, f)
    assert np.array_equal(got_b, b)
    assert np.array_equal(got_u[keep], u[keep])
    gm = np.ctypeslib.as_array(
        C.cast(out.cols[0].null_mask, C.POINTER(C.c_uint8)), (1000,))
    assert np.array_equal(gm, mask)
    L.pg_page_free(C.byref(out))


def test_corruption_detected(L):
    a = np.arange(64, dtype=np.int64)
    wire = bytearray(_serialize(L, _page({"a": a})))
    wire[40] ^= 0xFF  # flip a data byte
    out = C.create_string_buffer(1024)
    st = L.pg_page_deserialize(bytes(wire), len(wire), out)
    assert st != 0
    assert b"checksum" in L.pg_last_error()


def test_compressed_rejected(L):
    a = np.arange(8, dtype=np.int64)
    wire = bytearray(_serialize(L, _page({"a": a})))
    wire[4] = 1  # set COMPRESSED codec marker bit
    out = C.create_string_buffer(1024)
    st = L.pg_page_deserialize(bytes(wire), len(wire), out)
    assert st != 0


def test_varbin_wire_bytes_pinned(L):
    """VARIABLE_WIDTH block encoding (VariableWidthBlockEncoding.java:
    37-58): positionCount cumulative lengths, null bits, total length,
    raw bytes."""
    from presto_amd.engine import PgPage, PgCol
    strings = [b"abc", b"", b"defgh"]
    offs = np.array([0, 3, 3, 8], np.int32)
    data = np.frombuffer(b"abcdefgh", np.uint8).copy()
    ids = np.array([1, 2, 3], np.int64)
    pg = PgPage()
    pg.n_rows = 3
    pg.n_cols = 2
    pg.cols[0].tag = 4  # VARBIN
    pg.cols[0].on_device = 0
    pg.cols[0].data = data.ctypes.data
    pg.cols[0].offsets = offs.ctypes.data
    pg.cols[1].tag = 2  # I64
    pg.cols[1].on_device = 0
    pg.cols[1].data = ids.ctypes.data
    got = _serialize(L, pg)
    vb = (struct.pack("<iii", 3, 3, 8)  # cumulative lengths per position
          + b"\x00"                      # no nulls
          + struct.pack("<i", 8) + b"abcdefgh")
    # VARIABLE_WIDTH embeds its null byte mid-block, so build the body
    # fully by hand:
    body = struct.pack("<i", 2)
    nb = b"VARIABLE_WIDTH"
    body += struct.pack("<i", len(nb)) + nb + struct.pack("<i", 3) + vb
    nb = b"LONG_ARRAY"
    body += struct.pack("<i", len(nb)) + nb + struct.pack("<i", 3)
    body += b"\x00" + ids.tobytes()
    crc = zlib.crc32(body)
    crc = zlib.crc32(bytes([0]), crc)
    crc = zlib.crc32(struct.pack("<i", 3), crc)
    crc = zlib.crc32(struct.pack("<i", len(body)), crc)
    exp = struct.pack("<iBiiq", 3, 0, len(body), len(body), crc) + body
    assert got == exp
    # round trip
    out = PgPage()
    st = L.pg_page_deserialize(got, len(got), C.byref(out))
    assert st == 0, L.pg_last_error()
    assert out.n_rows == 3 and out.n_cols == 2
    assert out.cols[0].tag == 4
    roffs = C.cast(out.cols[0].offsets,
                   C.POINTER(C.c_int32))[:4]
    assert roffs == [0, 3, 3, 8]
    rdata = C.cast(out.cols[0].data, C.POINTER(C.c_uint8))[:8]
    assert bytes(rdata) == b"abcdefgh"
    L.pg_page_free(C.byref(out))


def _l2(L):
    L.pg_page_serialize2.argtypes = [C.c_void_p, C.c_int32, C.c_void_p,
                                     C.c_int64, C.POINTER(C.c_int64)]
    return L


def _serialize2(L, pg, compress):
    _l2(L)
    buf = C.create_string_buffer(1 << 22)
    out_len = C.c_int64()
    st = L.pg_page_serialize2(C.byref(pg), compress, buf, len(buf),
                              C.byref(out_len))
    assert st == 0, L.pg_last_error()
    return bytes(buf[:out_len.value])


def test_lz4_roundtrip(L):
    """Round-2 scope: LZ4-compressed pages (PagesSerde.java:67-95,
    PageCodecMarker.COMPRESSED, MINIMUM_COMPRESSION_RATIO=0.9)."""
    from presto_amd.engine import PgPage
    # compressible data: long runs
    a = np.repeat(np.arange(64, dtype=np.int64), 128)
    b = np.zeros(64 * 128, np.int32)
    pg = _page({"a": a, "b": b})
    wire = _serialize2(L, pg, 1)
    plain = _serialize2(L, pg, 0)
    assert len(wire) < len(plain) * 0.5
    assert wire[4] == 1  # COMPRESSED marker
    out = PgPage()
    st = L.pg_page_deserialize(wire, len(wire), C.byref(out))
    assert st == 0, L.pg_last_error()
    got = np.ctypeslib.as_array(
        C.cast(out.cols[0].data, C.POINTER(C.c_int64)),
        shape=(len(a),)).copy()
    gotb = np.ctypeslib.as_array(
        C.cast(out.cols[1].data, C.POINTER(C.c_int32)),
        shape=(len(b),)).copy()
    assert np.array_equal(got, a) and np.array_equal(gotb, b)
    L.pg_page_free(C.byref(out))
    # incompressible random data falls back to the uncompressed marker
    rng = np.random.RandomState(3)
    r = rng.randint(-2**62, 2**62, 4096).astype(np.int64)
    pg2 = _page({"r": r})
    wire2 = _serialize2(L, pg2, 1)
    assert wire2[4] == 0
    out2 = PgPage()
    assert L.pg_page_deserialize(wire2, len(wire2), C.byref(out2)) == 0
    got2 = np.ctypeslib.as_array(
        C.cast(out2.cols[0].data, C.POINTER(C.c_int64)),
        shape=(len(r),)).copy()
    assert np.array_equal(got2, r)
    L.pg_page_free(C.byref(out2))


def test_int128_roundtrip(L):
    """INT128_ARRAY encoding (Int128ArrayBlockEncoding.java:36-50):
    two longs per non-null position after the null bits."""
    from presto_amd.engine import PgPage
    n = 37
    vals = np.arange(2 * n, dtype=np.int64)  # (lo, hi) interleaved
    nulls = np.zeros(n, np.uint8)
    nulls[5] = 1
    pg = PgPage()
    pg.n_rows = n
    pg.n_cols = 1
    pg.cols[0].tag = 5  # PG_T_I128
    pg.cols[0].on_device = 0
    pg.cols[0].data = vals.ctypes.data
    pg.cols[0].null_mask = nulls.ctypes.data
    wire = _serialize2(L, pg, 0)
    assert b"INT128_ARRAY" in wire
    out = PgPage()
    st = L.pg_page_deserialize(wire, len(wire), C.byref(out))
    assert st == 0, L.pg_last_error()
    got = np.ctypeslib.as_array(
        C.cast(out.cols[0].data, C.POINTER(C.c_int64)),
        shape=(2 * n,)).copy()
    mask = np.ctypeslib.as_array(
        C.cast(out.cols[0].null_mask, C.POINTER(C.c_uint8)),
        shape=(n,)).copy()
    assert mask[5] == 1 and mask.sum() == 1
    keep = np.ones(n, bool)
    keep[5] = False
    assert np.array_equal(got.reshape(n, 2)[keep],
                          vals.reshape(n, 2)[keep])
    L.pg_page_free(C.byref(out))


def test_dictionary_on_the_wire(L):
    """DICTIONARY encoding (DictionaryBlockEncoding.java:32-55):
    positionCount, nested dictionary block, raw ids, 3-long
    DictionaryId; varbin dictionaries come back in dictionary form."""
    from presto_amd.engine import PgPage
    strings = [b"aa", b"bbb", b"c"]
    offs = np.array([0, 2, 5, 6], np.int32)
    data = np.frombuffer(b"aabbbc", np.uint8).copy()
    ids = np.array([2, 0, 1, 1, 0], np.int32)
    pg = PgPage()
    pg.n_rows = 5
    pg.n_cols = 1
    pg.cols[0].tag = 4
    pg.cols[0].on_device = 0
    pg.cols[0].data = data.ctypes.data
    pg.cols[0].offsets = offs.ctypes.data
    pg.cols[0].dict_ids = ids.ctypes.data
    pg.cols[0].dict_n = 3
    wire = _serialize2(L, pg, 0)
    assert b"DICTIONARY" in wire and b"VARIABLE_WIDTH" in wire
    out = PgPage()
    st = L.pg_page_deserialize(wire, len(wire), C.byref(out))
    assert st == 0, L.pg_last_error()
    assert out.cols[0].dict_n == 3
    gids = np.ctypeslib.as_array(
        C.cast(out.cols[0].dict_ids, C.POINTER(C.c_int32)),
        shape=(5,)).copy()
    assert np.array_equal(gids, ids)
    goffs = np.ctypeslib.as_array(
        C.cast(out.cols[0].offsets, C.POINTER(C.c_int32)),
        shape=(4,)).copy()
    assert np.array_equal(goffs, offs)
    L.pg_page_free(C.byref(out))


def test_rle_expanded_on_read(L):
    """RLE encoding (RunLengthBlockEncoding.java:31-51): run length then
    the single-position value block; expanded to a flat column on read."""
    import struct as S
    # hand-build: metadata + body with one RLE block of 6 x LONG 42
    inner = (S.pack("<i", 10) + b"LONG_ARRAY" + S.pack("<i", 1) + b"\x00"
             + S.pack("<q", 42))
    body = S.pack("<i", 1) + S.pack("<i", 3) + b"RLE" + S.pack("<i", 6) \
        + inner
    crc = zlib.crc32(body)
    tail = bytes([0]) + S.pack("<i", 6) + S.pack("<i", len(body))
    crc = zlib.crc32(tail, crc)
    wire = S.pack("<i", 6) + bytes([0]) + S.pack("<ii", len(body),
                                                 len(body)) \
        + S.pack("<q", crc) + body
    from presto_amd.engine import PgPage
    out = PgPage()
    st = L.pg_page_deserialize(wire, len(wire), C.byref(out))
    assert st == 0, L.pg_last_error()
    got = np.ctypeslib.as_array(
        C.cast(out.cols[0].data, C.POINTER(C.c_int64)), shape=(6,)).copy()
    assert np.array_equal(got, np.full(6, 42, np.int64))
    L.pg_page_free(C.byref(out))


def test_serde_fuzz_roundtrip(L):
    """Property fuzz: random pages over every block type, with and
    without nulls, plain and LZ4, must round-trip exactly."""
    from presto_amd.engine import PgPage
    rng = np.random.RandomState(77)
    for trial in range(25):
        n = int(rng.randint(0, 700))
        pg = PgPage()
        pg.n_rows = n
        keep = []
        ncols = int(rng.randint(1, 6))
        pg.n_cols = ncols
        spec = []
        for c in range(ncols):
            kind = rng.randint(0, 5)
            nulls = None
            if rng.randint(0, 2) and n:
                nulls = (rng.randint(0, 4, n) == 0).astype(np.uint8)
            if kind == 0:
                a = rng.randint(-2**60, 2**60, n).astype(np.int64)
                tag = 2
            elif kind == 1:
                a = rng.randint(-2**31, 2**31, n).astype(np.int32)
                tag = 1
            elif kind == 2:
                a = rng.randint(0, 256, n).astype(np.uint8)
                tag = 0
            elif kind == 3:
                a = rng.randint(-2**60, 2**60, 2 * n).astype(np.int64)
                tag = 5  # i128 (lo, hi) pairs
            else:
                # VARBIN: random lengths incl. empty strings
                lens = rng.randint(0, 12, n)
                offs = np.zeros(n + 1, np.int32)
                np.cumsum(lens, out=offs[1:])
                data = rng.randint(0, 256,
                                   int(offs[-1])).astype(np.uint8)
                a = (data, offs)
                tag = 4
            keep.append((a, nulls))
            spec.append((tag, a, nulls))
            pg.cols[c].tag = tag
            pg.cols[c].on_device = 0
            if tag == 4:
                pg.cols[c].data = a[0].ctypes.data
                pg.cols[c].offsets = a[1].ctypes.data
            else:
                pg.cols[c].data = a.ctypes.data
            pg.cols[c].null_mask = (nulls.ctypes.data
                                    if nulls is not None else None)
        wire = _serialize2(L, pg, int(rng.randint(0, 2)))
        out = PgPage()
        st = L.pg_page_deserialize(wire, len(wire), C.byref(out))
        assert st == 0, L.pg_last_error()
        assert out.n_rows == n and out.n_cols == ncols
        for c, (tag, a, nulls) in enumerate(spec):
            if tag == 4:
                data, offs = a
                roffs = np.ctypeslib.as_array(
                    C.cast(C.c_void_p(out.cols[c].offsets),
                           C.POINTER(C.c_int32)),
                    shape=(n + 1,)).copy() if n else offs[:1]
                if n:
                    m = (nulls.astype(bool) if nulls is not None
                         else np.zeros(n, bool))
                    for i in range(n):
                        if m[i]:
                            continue
                        exp = data[offs[i]:offs[i + 1]].tobytes()
                        got_b = bytes(np.ctypeslib.as_array(
                            C.cast(out.cols[c].data,
                                   C.POINTER(C.c_uint8)),
                            shape=(max(int(roffs[-1]), 1),))
                            [roffs[i]:roffs[i + 1]]) \
                            if roffs[-1] else b""
                        assert got_b == exp, (trial, c, i)
                continue
            esz = {0: 1, 1: 4, 2: 8, 5: 16}[tag]
            cnt = n * (2 if tag == 5 else 1)
            dt = {0: np.uint8, 1: np.int32, 2: np.int64,
                  5: np.int64}[tag]
            if cnt:
                got = np.ctypeslib.as_array(
                    C.cast(out.cols[c].data,
                           C.POINTER(np.ctypeslib.as_ctypes_type(dt))),
                    shape=(cnt,)).copy()
                if nulls is None:
                    assert np.array_equal(got, a), (trial, c)
                else:
                    m = nulls.astype(bool)
                    if tag == 5:
                        m2 = np.repeat(m, 2)
                        assert np.array_equal(got[~m2], a[~m2])
                    else:
                        assert np.array_equal(got[~m], a[~m])
        L.pg_page_free(C.byref(out))
