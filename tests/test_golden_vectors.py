"""Byte-level format stability: the committed golden vectors
(tests/golden/codec_vectors.json, see make_golden.py for how parity with
the reference is pinned) must match what the oracle encoders and the
product writer produce today, and must decode back to the inputs."""
import json
import os

import numpy as np

import binding as orc
from opengemini_amd import engine as gxe

HERE = os.path.dirname(os.path.abspath(__file__))
VECS = json.load(open(os.path.join(HERE, "golden", "codec_vectors.json")))


def test_gorilla_bytes_stable():
    v = VECS["gorilla_walk64"]
    walk = np.array(v["input"])
    assert bytes(orc.gorilla_encode(walk)).hex() == v["oracle"]
    # and the stream decodes back bit-exact
    dec = orc.gorilla_decode(bytes.fromhex(v["oracle"]), len(walk))
    assert np.array_equal(dec, walk)


def test_segment_bytes_stable_and_roundtrip():
    for name in ("int_const_delta", "int_simple8b", "int_raw",
                 "float_same", "float_rle"):
        v = VECS[name]
        is_int = name.startswith("int")
        vals = np.array(v["input"],
                        dtype=np.int64 if is_int else np.float64)
        ct = gxe.GEMX_TYPE_INT if is_int else gxe.GEMX_TYPE_FLOAT
        n = len(vals)
        sids = np.full(n, 1, dtype=np.uint64)
        times = np.arange(n, dtype=np.int64) * 10**9
        wblob, _ = gxe.encode_shard(ct, sids, times, vals)
        assert wblob.hex() == v["writer_blob"], name
        oct_ = orc.ORC_TYPE_INT if is_int else orc.ORC_TYPE_FLOAT
        oseg = orc.encode_data_segment(oct_, vals, None, n, 0)
        assert bytes(oseg).hex() == v["oracle_data_segment"], name
        got, bm, rows, nil = orc.decode_data_segment(
            oct_, bytes.fromhex(v["oracle_data_segment"]), n)
        assert rows == n and nil == 0
        assert np.array_equal(np.asarray(got)[:n].view(vals.dtype), vals), name


def test_time_segment_bytes_stable():
    for name in ("time_const_delta", "time_s8b_scale"):
        v = VECS[name]
        times = np.array(v["input"], dtype=np.int64)
        assert bytes(orc.encode_time_segment(times)).hex() == \
            v["oracle_time_segment"], name
        dec = orc.decode_time_segment(
            bytes.fromhex(v["oracle_time_segment"]), len(times))
        assert np.array_equal(dec, times), name
