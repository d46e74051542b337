"""CPU tests: SQL residual-predicate filter evaluation (moann_preds.cpp)
vs a numpy restatement of eval_filter_bitmap_cpu's semantics
(cgo/cuvs/filter.hpp:855-960): conjunctive AND, three-valued NULL logic,
virtual PK column (col == -1), per-type literal narrowing, fail-closed
parse errors."""

import os

import numpy as np
import pytest

from matrixone_amd import engine


def _mk(n=5000, seed=1):
    rng = np.random.Generator(np.random.PCG64(seed))
    cols_meta = [("a", engine.COL_I32), ("b", engine.COL_I64),
                 ("c", engine.COL_F32), ("d", engine.COL_F64),
                 ("h", engine.COL_U64)]
    cols = [rng.integers(-100, 100, n).astype(np.int32),
            rng.integers(-10**12, 10**12, n).astype(np.int64),
            rng.standard_normal(n).astype(np.float32) * 10,
            rng.standard_normal(n).astype(np.float64) * 10,
            rng.integers(0, 2**63, n).astype(np.uint64)]
    ids = rng.permutation(n).astype(np.int64) * 3 + 7
    return cols_meta, cols, ids, rng


def _null_words(mask):
    n = len(mask)
    w = np.zeros((n + 31) // 32, dtype=np.uint32)
    idx = np.nonzero(mask)[0]
    np.bitwise_or.at(w, idx // 32, (np.uint32(1) << (idx % 32)
                                    .astype(np.uint32)))
    return w


def test_ops_all_types():
    cols_meta, cols, ids, rng = _mk()
    n = len(cols[0])
    cases = [
        ('[{"col":0,"op":">=","val":5}]', cols[0] >= 5),
        ('[{"col":0,"op":"<","val":-20}]', cols[0] < -20),
        ('[{"col":1,"op":"!=","val":0}]', cols[1] != 0),
        ('[{"col":2,"op":"<=","val":1.5}]', cols[2] <= np.float32(1.5)),
        ('[{"col":3,"op":"between","lo":-2.5,"hi":4.0}]',
         (cols[3] >= -2.5) & (cols[3] <= 4.0)),
        ('[{"col":0,"op":"in","vals":[1,2,3,50]}]',
         np.isin(cols[0], [1, 2, 3, 50])),
        # conjunction
        ('[{"col":0,"op":">","val":0},{"col":2,"op":"<","val":0.0}]',
         (cols[0] > 0) & (cols[2] < np.float32(0.0))),
        # empty list passes everything
        ('[]', np.ones(n, dtype=bool)),
    ]
    for js, exp in cases:
        got = engine.preds_eval(cols_meta, cols, js, n)
        np.testing.assert_array_equal(got, exp, err_msg=js)


def test_u64_hash_column():
    cols_meta, cols, ids, rng = _mk()
    n = len(cols[0])
    v = int(cols[4][17])
    js = f'[{{"col":4,"op":"=","val":{v}}}]'
    got = engine.preds_eval(cols_meta, cols, js, n)
    np.testing.assert_array_equal(got, cols[4] == np.uint64(v))


def test_null_semantics():
    cols_meta, cols, ids, rng = _mk()
    n = len(cols[0])
    null_mask = rng.random(n) < 0.2
    nulls = [None, None, _null_words(null_mask), None, None]
    # NULL cells fail value comparisons (3-valued logic)
    got = engine.preds_eval(cols_meta, cols, '[{"col":2,"op":">","val":-100}]',
                            n, nulls=nulls)
    np.testing.assert_array_equal(got, ~null_mask)
    # is_null / is_not_null consult only the bitmap
    got = engine.preds_eval(cols_meta, cols, '[{"col":2,"op":"is_null"}]',
                            n, nulls=nulls)
    np.testing.assert_array_equal(got, null_mask)
    got = engine.preds_eval(cols_meta, cols,
                            '[{"col":2,"op":"is_not_null"}]', n, nulls=nulls)
    np.testing.assert_array_equal(got, ~null_mask)
    # dense column: is_null matches nothing
    got = engine.preds_eval(cols_meta, cols, '[{"col":0,"op":"is_null"}]', n)
    assert not got.any()


def test_virtual_pk_column():
    cols_meta, cols, ids, rng = _mk()
    n = len(cols[0])
    sel = ids[[3, 8, 44]]
    js = ('[{"col":-1,"op":"in","vals":[%d,%d,%d]}]'
          % (sel[0], sel[1], sel[2]))
    got = engine.preds_eval(cols_meta, cols, js, n, host_ids=ids)
    np.testing.assert_array_equal(got, np.isin(ids, sel))
    # PKs are non-nullable
    got = engine.preds_eval(cols_meta, cols, '[{"col":-1,"op":"is_null"}]',
                            n, host_ids=ids)
    assert not got.any()
    got = engine.preds_eval(cols_meta, cols,
                            '[{"col":-1,"op":"is_not_null"}]', n,
                            host_ids=ids)
    assert got.all()
    # absent host-id view: value predicates pass through (planner residual
    # filter stays authoritative)
    got = engine.preds_eval(cols_meta, cols,
                            '[{"col":-1,"op":"=","val":123}]', n)
    assert got.all()


def test_literal_narrowing():
    """Literals re-narrow per column type like pred_value_as: an int
    literal compares as float on float columns and vice versa."""
    cols_meta = [("x", engine.COL_I32), ("y", engine.COL_F64)]
    cols = [np.array([1, 2, 3], dtype=np.int32),
            np.array([1.0, 2.5, 3.0])]
    got = engine.preds_eval(cols_meta, cols, '[{"col":0,"op":"=","val":2.0}]',
                            3)
    np.testing.assert_array_equal(got, [False, True, False])
    got = engine.preds_eval(cols_meta, cols, '[{"col":1,"op":">","val":2}]',
                            3)
    np.testing.assert_array_equal(got, [False, True, True])


def test_errors_fail_closed():
    cols_meta, cols, ids, rng = _mk(n=64)
    for js in ['[{"col":9,"op":"=","val":1}]',      # column out of range
               '[{"col":0,"op":"~","val":1}]',      # unknown op
               '[{"op":"=","val":1}]',              # missing col
               '[{"col":0,"op":"="',                # truncated JSON
               ]:
        with pytest.raises(engine.MoannError):
            engine.preds_eval(cols_meta, cols, js, 64)


def test_tail_bits_zero():
    cols_meta = [("x", engine.COL_I32)]
    cols = [np.arange(37, dtype=np.int32)]
    got = engine.preds_eval(cols_meta, cols, "[]", 37)
    assert got.all() and len(got) == 37


_REF = os.path.join(os.path.dirname(os.path.abspath(__file__)), "..",
                    "oracle", "_ref", "libref_preds.so")


@pytest.mark.skipif(not os.path.exists(_REF),
                    reason="reference filter build unavailable")
def test_preds_vs_reference_fuzz():
    """Randomized parity vs the REFERENCE's own eval_filter_bitmap_cpu
    (oracle/_ref/libref_preds.so — the real cgo/cuvs/filter.hpp compiled):
    500 random predicate sets across every op, type, null-density and the
    virtual PK column must produce IDENTICAL bitsets."""
    import ctypes as ct
    import json
    L = ct.CDLL(_REF)
    L.ref_preds_eval.restype = ct.c_int
    L.ref_preds_eval.argtypes = [
        ct.c_char_p, ct.c_char_p, ct.POINTER(ct.c_void_p),
        ct.POINTER(ct.c_void_p), ct.c_void_p, ct.c_uint64, ct.c_void_p]

    rng = np.random.Generator(np.random.PCG64(777))
    n = 2000
    cols_meta, cols, ids, _ = _mk(n=n, seed=5)
    null_masks = [rng.random(n) < p for p in (0.0, 0.3, 0.1, 0.0, 0.05)]
    nulls = [None if not m.any() else _null_words(m) for m in null_masks]

    ops = ["=", "!=", "<", "<=", ">", ">=", "between", "in",
           "is_null", "is_not_null"]
    for trial in range(500):
        preds = []
        for _ in range(int(rng.integers(0, 4))):
            col = int(rng.integers(-1, 5))
            op = ops[int(rng.integers(0, len(ops) - (2 if col == -1 else 0)))]
            p = {"col": col, "op": op}
            pick = (float(rng.standard_normal() * 50)
                    if rng.random() < 0.5 else int(rng.integers(-100, 100)))
            if op == "between":
                lo, hi = sorted([pick, pick + float(rng.random() * 50)])
                p["lo"], p["hi"] = lo, hi
            elif op == "in":
                p["vals"] = [int(rng.integers(-100, 100)) for _ in range(3)]
            elif op not in ("is_null", "is_not_null"):
                p["val"] = pick
            preds.append(p)
        js = json.dumps(preds)

        mine = engine.preds_eval(cols_meta, cols, js, n, nulls=nulls,
                                 host_ids=ids)

        cols_json, data_arr, null_arr, keep = engine._col_arrays(
            cols_meta, cols, nulls)
        out = np.zeros((n + 31) // 32, dtype=np.uint32)
        hid = np.ascontiguousarray(ids, dtype=np.int64)
        rc = L.ref_preds_eval(cols_json, js.encode(), data_arr, null_arr,
                              hid.ctypes.data, n, out.ctypes.data)
        assert rc == 0, js
        idx = np.arange(n)
        ref = ((out[idx >> 5] >> (idx & 31)) & 1).astype(bool)
        np.testing.assert_array_equal(mine, ref, err_msg=js)
