"""Generate the committed golden codec vectors.

The reference is Go and cannot run in this container, so byte-level
format parity is pinned two ways: (1) the oracle's codecs were validated
against the reference's own test shapes (tests/test_oracle_codecs.py,
transcribed expected records in tests/test_oracle_golden_agg.py /
test_oracle_rate.py), and (2) THIS file freezes the oracle's and the
product writer's encoded bytes for fixed inputs into
tests/golden/codec_vectors.json, so any later drift in either
implementation fails tests/test_golden_vectors.py.

Run from the repo root: python tests/golden/make_golden.py
"""
import json
import os
import sys

import numpy as np

REPO = os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
sys.path.insert(0, REPO)
sys.path.insert(0, os.path.join(REPO, "oracle"))

import binding as orc  # noqa: E402
from opengemini_amd import engine as gxe  # noqa: E402


def b2h(b):
    return bytes(b).hex()


def main():
    out = {}
    # gorilla: walk-like quantized values (reference batch_float.go format)
    rng = np.random.default_rng(424242)
    walk = np.round(np.cumsum(rng.normal(0, 1, 64)) * 128) / 128
    out["gorilla_walk64"] = {
        "input": [float(x) for x in walk],
        "oracle": b2h(orc.gorilla_encode(walk)),
    }
    # int codecs through the segment layer: const-delta / simple8b / raw
    cases = {
        "int_const_delta": np.arange(50, dtype=np.int64) * 7 + 3,
        "int_simple8b": (np.arange(50, dtype=np.int64) ** 2) % 997,
        "int_raw": np.array([0, 2**62, -(2**62), 5], dtype=np.int64),
        "float_same": np.full(20, 2.5),
        "float_rle": np.repeat(np.array([1.5, 0.0, -2.0]), [8, 6, 6]),
    }
    for name, vals in cases.items():
        ct = gxe.GEMX_TYPE_INT if vals.dtype == np.int64 else gxe.GEMX_TYPE_FLOAT
        n = len(vals)
        sids = np.full(n, 1, dtype=np.uint64)
        times = np.arange(n, dtype=np.int64) * 10**9
        wblob, wdescs = gxe.encode_shard(ct, sids, times, vals)
        oct_ = orc.ORC_TYPE_INT if ct == 1 else orc.ORC_TYPE_FLOAT
        if ct == 1:
            oseg = orc.encode_data_segment(oct_, vals, None, n, 0)
        else:
            oseg = orc.encode_data_segment(oct_, vals, None, n, 0)
        out[name] = {
            "input": [int(x) if vals.dtype == np.int64 else float(x)
                      for x in vals],
            "writer_blob": b2h(wblob),
            "oracle_data_segment": b2h(oseg),
        }
    # time codec: const-delta and simple8b x scale
    tcases = {
        "time_const_delta": np.arange(40, dtype=np.int64) * 60 * 10**9 + 17,
        "time_s8b_scale": np.cumsum(
            np.array([1, 2, 3, 1, 5, 2] * 6, dtype=np.int64) * 10**9),
    }
    for name, times in tcases.items():
        out[name] = {
            "input": [int(t) for t in times],
            "oracle_time_segment": b2h(orc.encode_time_segment(times)),
        }
    path = os.path.join(REPO, "tests", "golden", "codec_vectors.json")
    with open(path, "w") as f:
        json.dump(out, f, indent=1, sort_keys=True)
    print("wrote", path, os.path.getsize(path), "bytes")


if __name__ == "__main__":
    main()
