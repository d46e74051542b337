"""CPU tests: SQL residual-predicate filter evaluation (moann_preds.cpp)
vs a numpy restatement of eval_filter_bitmap_cpu's semantics
(cgo/cuvs/filter.hpp:855-960): conjunctive AND, three-valued NULL logic,
virtual PK column (col == -1), per-type literal narrowing, fail-closed
parse errors."""

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
