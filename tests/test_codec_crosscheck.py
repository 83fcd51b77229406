"""Independent third-implementation cross-checks of the codec formats.

VERDICT r1 (weak #5): the golden vectors are generated from the oracle
itself, so a matched encoder+decoder transliteration bug — the same
format misreading on both sides — would pass every round-trip test and
every golden comparison. These tests close that hole with from-spec
pure-Python implementations written against the PUBLISHED formats:

- tsm1 Gorilla (Pelkonen et al., "Gorilla: A Fast, Scalable, In-Memory
  Time Series Database", VLDB 2015 §4.1.2, with tsm1's framing: one tag
  byte 1<<4, first value as raw big-endian u64, 2 control bits,
  5-bit leading-zero count, 6-bit meaningful-bit count where 0 means 64,
  and the 0x7FF8000000000001 NaN terminator);
- google/snappy block format (format_description.txt: varint preamble,
  2-bit element tags, literals and 1/2-byte-offset copies);
- simple8b (Lemire's 16-selector variant used by InfluxDB: 4-bit
  selector in the top bits, packings 240x0 .. 1x60).

Each test decodes oracle/product-writer output with the Python
implementation (and feeds Python-encoded streams back through the C
decoders), so any drift between the C restatement and the published
format fails here even when enc+dec agree with each other.
"""

import struct

import numpy as np
import pytest

import binding as orc

UVNAN = 0x7FF8000000000001
M64 = (1 << 64) - 1


# ---------------------------------------------------------------- gorilla

class _BitReader:
    """MSB-first bit reader over bytes."""

    def __init__(self, data):
        self.d = data
        self.pos = 0
        self.n = len(data) * 8

    def take(self, k):
        out = 0
        for _ in range(k):
            if self.pos >= self.n:
                raise EOFError("gorilla stream truncated")
            byte = self.d[self.pos >> 3]
            out = (out << 1) | ((byte >> (7 - (self.pos & 7))) & 1)
            self.pos += 1
        return out


class _BitWriter:
    def __init__(self):
        self.bits = []

    def put(self, v, k):
        for i in range(k - 1, -1, -1):
            self.bits.append((v >> i) & 1)

    def bytes(self):
        out = bytearray()
        for i in range(0, len(self.bits), 8):
            b = 0
            for j, bit in enumerate(self.bits[i:i + 8]):
                b |= bit << (7 - j)
            out.append(b)
        return bytes(out)


def py_gorilla_decode(buf):
    """From-spec tsm1 Gorilla decoder (paper §4.1.2 + tsm1 framing)."""
    assert buf[0] >> 4 == 1, "tsm1 tag byte"
    first = int.from_bytes(buf[1:9], "big")
    if first == UVNAN:
        return []
    vals = [first]
    br = _BitReader(buf[9:])
    val = first
    meaningful, trailing = 64, 0
    while True:
        if br.take(1):  # value changed
            if br.take(1):  # new leading/meaningful window
                lead = br.take(5)
                meaningful = br.take(6)
                if meaningful == 0:
                    meaningful = 64
                    trailing = 0
                else:
                    trailing = 64 - lead - meaningful
            sbits = br.take(meaningful)
            val ^= (sbits << trailing) & M64
            if val == UVNAN:
                break
        vals.append(val)
    return [struct.unpack("<d", struct.pack("<Q", v))[0] for v in vals]


def py_gorilla_encode(values):
    """From-spec encoder. Always emits a NEW window per changed value
    (valid per the format; exercises decoder paths the oracle's
    window-reusing encoder rarely hits)."""
    out = bytearray([1 << 4])
    bits = [struct.unpack("<Q", struct.pack("<d", v))[0] for v in values]
    first = bits[0] if bits else UVNAN
    out += first.to_bytes(8, "big")
    bw = _BitWriter()
    prev = first
    for v in bits[1:] + [UVNAN]:
        x = prev ^ v
        prev = v
        if x == 0:
            bw.put(0, 1)
            continue
        lead = min(64 - x.bit_length(), 31)
        trail = (x & -x).bit_length() - 1
        meaningful = 64 - lead - trail
        bw.put(0b11, 2)
        bw.put(lead, 5)
        bw.put(0 if meaningful == 64 else meaningful, 6)
        bw.put(x >> trail, meaningful)
    return bytes(out + bw.bytes())


class TestGorillaCrossCheck:
    def test_decode_oracle_streams(self):
        rng = np.random.default_rng(7)
        for n in (1, 2, 5, 100, 997):
            vals = np.round(np.cumsum(rng.normal(0, 1, n)) * 128) / 128
            enc = orc.gorilla_encode(vals)
            got = py_gorilla_decode(bytes(enc))
            assert np.array_equal(
                np.array(got).view(np.uint64), vals.view(np.uint64))

    def test_decode_oracle_random_mantissas(self):
        rng = np.random.default_rng(8)
        vals = rng.normal(0, 1e6, 300)
        got = py_gorilla_decode(bytes(orc.gorilla_encode(vals)))
        assert np.array_equal(np.array(got).view(np.uint64),
                              vals.view(np.uint64))

    def test_decode_oracle_empty(self):
        assert py_gorilla_decode(bytes(orc.gorilla_encode(
            np.array([], dtype=np.float64)))) == []

    def test_oracle_decodes_python_streams(self):
        # the always-new-window Python encoder exercises the 11-bit
        # header path on every changed value
        rng = np.random.default_rng(9)
        for vals in (
            np.array([1.5]),
            np.array([2.0, 2.0, 2.0]),
            np.round(np.cumsum(rng.normal(0, 1, 500)) * 64) / 64,
            rng.normal(0, 1e9, 200),
            np.array([0.0, -0.0, 1e-308, 1e308]),
        ):
            enc = py_gorilla_encode(list(vals))
            got = orc.gorilla_decode(np.frombuffer(enc, dtype=np.uint8))
            assert np.array_equal(np.asarray(got).view(np.uint64),
                                  vals.view(np.uint64))

    def test_meaningful64_special_case(self):
        # meaningful == 0 encodes "all 64 bits" (tsm1 quirk): force a
        # full-width XOR via sign flip of a full-mantissa value
        vals = np.array([1.7976931348623157e308, -1.7976931348623157e308])
        enc = py_gorilla_encode(list(vals))
        got = orc.gorilla_decode(np.frombuffer(enc, dtype=np.uint8))
        assert np.array_equal(np.asarray(got).view(np.uint64),
                              vals.view(np.uint64))
        assert py_gorilla_decode(bytes(orc.gorilla_encode(vals))) == list(vals)


# ---------------------------------------------------------------- snappy

def py_snappy_decode(data):
    """From-spec snappy block decoder (format_description.txt)."""
    # varint uncompressed length
    ulen, shift, pos = 0, 0, 0
    while True:
        b = data[pos]
        pos += 1
        ulen |= (b & 0x7F) << shift
        shift += 7
        if not b & 0x80:
            break
    out = bytearray()
    while pos < len(data):
        tag = data[pos]
        pos += 1
        t = tag & 3
        if t == 0:  # literal
            ln = tag >> 2
            if ln >= 60:
                nb = ln - 59
                ln = int.from_bytes(data[pos:pos + nb], "little")
                pos += nb
            ln += 1
            out += data[pos:pos + ln]
            pos += ln
        elif t == 1:  # copy, 1-byte offset
            ln = ((tag >> 2) & 7) + 4
            off = ((tag & 0xE0) << 3) | data[pos]
            pos += 1
            for _ in range(ln):
                out.append(out[-off])
        elif t == 2:  # copy, 2-byte offset
            ln = (tag >> 2) + 1
            off = int.from_bytes(data[pos:pos + 2], "little")
            pos += 2
            for _ in range(ln):
                out.append(out[-off])
        else:
            raise ValueError("copy4 not emitted by go/snappy encoders")
    assert len(out) == ulen, (len(out), ulen)
    return bytes(out)


def py_snappy_encode_with_copies(data):
    """Valid snappy stream using copy elements (naive greedy matcher) —
    exercises the C decoder's copy paths, which the oracle's all-literal
    encoder never emits."""
    out = bytearray()
    ln = len(data)
    v = ln
    while v >= 0x80:
        out.append((v & 0x7F) | 0x80)
        v >>= 7
    out.append(v)
    pos = 0
    lit_start = 0

    def flush_literal(end):
        nonlocal lit_start
        while lit_start < end:
            chunk = min(end - lit_start, 60)
            out.append((chunk - 1) << 2)
            out.extend(data[lit_start:lit_start + chunk])
            lit_start += chunk

    while pos < ln:
        # look for a >=4-byte match within the last 255 bytes
        best_len, best_off = 0, 0
        if pos >= 4:
            lo = max(0, pos - 255)
            for cand in range(lo, pos):
                m = 0
                while (pos + m < ln and m < 64 and
                       data[cand + m] == data[pos + m]):
                    m += 1
                if m > best_len:
                    best_len, best_off = m, pos - cand
        if best_len >= 4:
            flush_literal(pos)
            take = min(best_len, 11)
            out.append(1 | ((take - 4) << 2) | ((best_off >> 8) << 5))
            out.append(best_off & 0xFF)
            pos += take
            lit_start = pos
        else:
            pos += 1
    flush_literal(ln)
    return bytes(out)


class TestSnappyCrossCheck:
    def test_python_decodes_oracle_streams(self):
        rng = np.random.default_rng(11)
        for n in (1, 10, 100, 5000):
            raw = rng.integers(0, 256, n).astype(np.uint8).tobytes()
            enc = orc.snappy_encode(np.frombuffer(raw, dtype=np.uint8))
            assert py_snappy_decode(bytes(enc)) == raw

    def test_oracle_decodes_python_copy_streams(self):
        rng = np.random.default_rng(12)
        for raw in (
            b"abcdabcdabcdabcd" * 10,
            b"\x00" * 300,
            bytes(rng.integers(0, 4, 2000).astype(np.uint8)),  # many matches
            b"x",
        ):
            enc = py_snappy_encode_with_copies(raw)
            # stream really contains copy elements (unless too short)
            got = orc.snappy_decode(np.frombuffer(enc, dtype=np.uint8),
                                    len(raw))
            assert bytes(got) == raw

    def test_python_roundtrip_sanity(self):
        raw = b"the quick brown fox " * 50
        assert py_snappy_decode(py_snappy_encode_with_copies(raw)) == raw


# ---------------------------------------------------------------- simple8b

_S8B = [(240, 0), (120, 0), (60, 1), (30, 2), (20, 3), (15, 4), (12, 5),
        (10, 6), (8, 7), (7, 8), (6, 10), (5, 12), (4, 15), (3, 20),
        (2, 30), (1, 60)]


def py_simple8b_decode_word(word):
    """From-spec selector unpack (selector in bits 60..63; selectors 0/1
    encode runs of 240/120 ones)."""
    sel = word >> 60
    n, bits = _S8B[sel]
    if bits == 0:
        return [1] * n
    mask = (1 << bits) - 1
    return [(word >> (i * bits)) & mask for i in range(n)]


class TestSimple8bCrossCheck:
    def test_python_decodes_oracle_words(self):
        rng = np.random.default_rng(13)
        for hi in (2, 8, 2**7, 2**15, 2**29, 2**59):
            vals = rng.integers(0, hi, 400).astype(np.uint64)
            words = orc.simple8b_encode(vals)
            got = []
            for wd in np.asarray(words):
                got.extend(py_simple8b_decode_word(int(wd)))
            assert got[:len(vals)] == list(vals)

    def test_ones_runs(self):
        vals = np.ones(360, dtype=np.uint64)
        words = orc.simple8b_encode(vals)
        got = []
        for wd in np.asarray(words):
            got.extend(py_simple8b_decode_word(int(wd)))
        assert got[:360] == [1] * 360

    def test_oracle_decodes_python_packed_word(self):
        # hand-pack 20 3-bit values with selector 4
        vals = [(i * 3) % 8 for i in range(20)]
        word = 4 << 60
        for i, v in enumerate(vals):
            word |= v << (i * 3)
        got = orc.simple8b_decode_word(word)
        assert list(got[:20]) == vals


# ------------------------------------------------------------- time codec

def py_time_decode(buf):
    """From-spec timestamp block decoder (lib/encoding/timestamp.go):
    tag nibble in byte 0: 1 = const-delta ([first u64be][delta uvarint]
    [count uvarint], :190-225), 2 = simple8b x scale ([scale u64be]
    [encCount u32be][srcCount u32be][first u64be][words u64be...],
    :227-272), 4 = uncompressed zigzag u64be each (:299-308)."""
    tag = buf[0] >> 4
    p = 1

    def uvarint(pos):
        v, sh = 0, 0
        while True:
            b = buf[pos]
            pos += 1
            v |= (b & 0x7F) << sh
            sh += 7
            if not b & 0x80:
                return v, pos

    if tag == 1:
        first = int.from_bytes(buf[p:p + 8], "big", signed=False)
        p += 8
        delta, p = uvarint(p)
        count, p = uvarint(p)
        out = [first + i * delta for i in range(count + 1)]
        return [v - 2**64 if v >= 2**63 else v for v in out]
    if tag == 2:
        scale = int.from_bytes(buf[p:p + 8], "big")
        enc_count = int.from_bytes(buf[p + 8:p + 12], "big")
        src_count = int.from_bytes(buf[p + 12:p + 16], "big")
        p += 16
        cur = int.from_bytes(buf[p:p + 8], "big")
        p += 8
        out = [cur]
        for w in range(enc_count - 1):
            word = int.from_bytes(buf[p:p + 8], "big")
            p += 8
            for v in py_simple8b_decode_word(word):
                cur = (cur + v * scale) & M64
                out.append(cur)
        assert len(out) >= src_count
        return out[:src_count]
    if tag == 4:
        n = int.from_bytes(buf[p:p + 4], "big") // 8
        p += 4
        out = []
        for i in range(n):
            u = int.from_bytes(buf[p + i * 8:p + i * 8 + 8], "big")
            out.append((u >> 1) ^ -(u & 1))
        return out
    raise ValueError(f"tag {tag}")


class TestTimeCodecCrossCheck:
    def test_const_delta(self):
        times = np.arange(500, dtype=np.int64) * 60 * 10**9 + 17
        enc = orc.time_encode(times)
        assert enc[0] >> 4 == 1
        assert py_time_decode(bytes(enc)) == list(times)

    def test_simple8b_scale(self):
        rng = np.random.default_rng(19)
        deltas = rng.integers(1, 10, 400) * 10**9  # common 1e9 scale
        times = np.cumsum(deltas).astype(np.int64)
        enc = orc.time_encode(times)
        assert enc[0] >> 4 == 2
        assert py_time_decode(bytes(enc)) == list(times)

    def test_uncompressed(self):
        rng = np.random.default_rng(20)
        deltas = rng.integers(1, 10**6, 50)
        deltas[25] = 2**61  # above simple8b's bound -> raw form
        times = np.cumsum(deltas).astype(np.int64)
        enc = orc.time_encode(times)
        assert enc[0] >> 4 == 4
        assert py_time_decode(bytes(enc)) == list(times)

    def test_negative_first_time(self):
        times = -300 * 10**9 + np.arange(100, dtype=np.int64) * 10**9
        enc = orc.time_encode(times)
        assert py_time_decode(bytes(enc)) == list(times)
