"""CPU tier: host-side logic that needs no device — chunking parity with the
reference, config knobs, operator-template plumbing and error surfaces."""

import numpy as np
import pandas
import pytest

import modin_amd.config as config
from modin_amd.core import lib
from modin_amd.core.partition_manager import compute_chunksize
import oracle


def test_compute_chunksize_matches_reference_rule():
    """modin/core/storage_formats/pandas/utils.py:28."""
    for n, splits, mn in [(100, 4, 32), (1000, 3, 32), (7, 2, 32),
                          (10**6, 8, 32), (55, 55, 1)]:
        chunk = compute_chunksize(n, splits, mn)
        assert chunk == max(-(-n // splits), mn)
        # and the oracle's splitter uses the same rule
        counts = oracle.split_row_counts(n, splits, mn)
        assert all(c == chunk for c in counts[:-1])
        assert sum(counts) == n


def test_config_env_roundtrip(monkeypatch):
    monkeypatch.setenv("MODIN_AMD_NPARTITIONS", "6")
    config.NPartitions._value = None
    assert config.NPartitions.get() == 6
    config.NPartitions.put(3)
    assert config.NPartitions.get() == 3
    config.NPartitions._value = None
    monkeypatch.delenv("MODIN_AMD_NPARTITIONS")
    assert config.NPartitions.get() == config.NPartitions.default


def test_api_surface_raises_outside_scope():
    import modin_amd.pandas as mpd
    # constructing needs a GPU; but attribute surface checks are host-side
    assert hasattr(mpd.DataFrame, "groupby")
    assert hasattr(mpd.DataFrame, "sum")
    with pytest.raises(lib.HfError):
        mpd.DataFrame(data=[1, 2, 3])  # unsupported ctor form


def test_groupby_reduce_agg_table():
    """The algebra layer's agg table mirrors GroupbyReduceImpl's supported
    map/reduce pairs (storage_formats/pandas/groupby.py:237-248 subset)."""
    from modin_amd.algebra import GroupByReduce
    assert set(GroupByReduce.SUPPORTED) == {"sum", "count", "mean", "min", "max"}
    with pytest.raises(lib.HfError, match="not implemented"):
        GroupByReduce.register("median")


def test_partition_call_queue_semantics():
    """add_to_apply_calls is lazy and non-mutating; drain runs in order
    (partition.py:140/:174 semantics) — exercised with a host-side stub block."""
    from modin_amd.core.partition import HipDataframePartition

    class FakeBlock:
        def __init__(self, v):
            self.v = v
            self.length = 1
            self.width = 1

    log = []

    def op(tag):
        def fn(block):
            log.append(tag)
            return FakeBlock(block.v + [tag])
        return fn

    p0 = HipDataframePartition(FakeBlock([]))
    p1 = p0.add_to_apply_calls(op("a"))
    p2 = p1.add_to_apply_calls(op("b"))
    assert p0.call_queue == [] and len(p1.call_queue) == 1
    assert log == []  # nothing ran yet
    p2.drain_call_queue()
    assert log == ["a", "b"]
    assert p2._block.v == ["a", "b"]
    # p1 still lazily holds only "a"
    p1.drain_call_queue()
    assert p1._block.v == ["a"]


def test_shuffle_dest_rule():
    """oracle.shuffle_dest: searchsorted-right binning — boundary keys go
    WITH the right bin, equal keys always share a destination, full-span
    int64 exact."""
    spl = np.array([-2**62, 0, 2**62], dtype=np.int64)
    keys = np.array([-2**63, -2**62 - 1, -2**62, -1, 0, 1,
                     2**62 - 1, 2**62, 2**63 - 1], dtype=np.int64)
    expect = np.array([0, 0, 1, 1, 2, 2, 2, 3, 3])
    np.testing.assert_array_equal(oracle.shuffle_dest(keys, spl), expect)
    # no splitters -> everything to rank 0
    np.testing.assert_array_equal(
        oracle.shuffle_dest(keys, np.empty(0, dtype=np.int64)),
        np.zeros(len(keys), dtype=np.int64))
    # equal keys -> one destination (groups never straddle ranks)
    dup = np.full(100, 7, dtype=np.int64)
    assert len(np.unique(oracle.shuffle_dest(dup, np.array([3, 7, 11])))) == 1


def test_pick_splitters_quantiles():
    s = oracle.pick_splitters(np.arange(100, dtype=np.int64), 4)
    np.testing.assert_array_equal(s, [25, 50, 75])
    assert oracle.pick_splitters(np.empty(0, dtype=np.int64), 4).size == 0
    assert oracle.pick_splitters(np.arange(10), 1).size == 0
    # splitters are sorted and dests cover [0, P)
    rng = np.random.default_rng(5)
    sample = rng.integers(-2**62, 2**62, 1000).astype(np.int64)
    spl = oracle.pick_splitters(sample, 8)
    assert np.all(np.diff(spl) >= 0)
    d = oracle.shuffle_dest(sample, spl)
    assert d.min() >= 0 and d.max() <= 7


def test_parquet_column_conversion():
    """io._column_to_device_ready: dictionary remap onto the sorted-cats
    invariant, null semantics (float NaN, nullable-int -> f64, string -1),
    narrow int widening — all host-side, no GPU."""
    import pyarrow as pa
    from modin_amd.io import _column_to_device_ready
    arr = pa.array(["pear", "apple", None, "pear", "zebra"]).dictionary_encode()
    codes, cats = _column_to_device_ready(pa.chunked_array([arr]), "s")
    assert list(cats) == ["apple", "pear", "zebra"]
    assert codes.tolist() == [1, 0, -1, 1, 2]
    # plain (non-dictionary) string column dictionary-encodes on the fly
    codes2, cats2 = _column_to_device_ready(
        pa.chunked_array([pa.array(["b", "a", "b", None])]), "t")
    assert list(cats2) == ["a", "b"] and codes2.tolist() == [1, 0, 1, -1]
    v, c = _column_to_device_ready(
        pa.chunked_array([pa.array([1.5, None, 2.5])]), "v")
    assert c is None and np.isnan(v[1]) and v[0] == 1.5
    v2, _ = _column_to_device_ready(
        pa.chunked_array([pa.array([1, 2, None], type=pa.int64())]), "i")
    assert v2.dtype == np.float64 and np.isnan(v2[2])
    v3, _ = _column_to_device_ready(
        pa.chunked_array([pa.array([7, 8], type=pa.int32())]), "j")
    assert v3.dtype == np.int64 and v3.tolist() == [7, 8]


def test_ordered_inverse_host_roundtrip():
    """ordered_to_f64_np (the host inverse of the device f64->i64 ordered
    bit transform) inverts a numpy restatement of the forward rule on
    every special value — the same rule k_f64_ordered implements
    (negatives bit-reversed below zero, -0.0 -> +0.0, every NaN
    canonicalized to one key)."""
    import numpy as np
    from modin_amd.core.lib import ordered_to_f64_np

    def forward(x):
        x = np.asarray(x, dtype=np.float64) + 0.0  # -0.0 -> +0.0
        v = x.view(np.int64).copy()
        v[np.isnan(x)] = 0x7FF8000000000000
        neg = v < 0
        v[neg] = ~v[neg] ^ np.int64(-2**63)
        return v

    xs = np.array([0.0, -0.0, 1.5, -1.5, np.inf, -np.inf, np.nan,
                   5e-324, -5e-324, 1e308, -1e308, 2.0, -2.0])
    back = ordered_to_f64_np(forward(xs))
    np.testing.assert_array_equal(back, np.where(xs == -0.0, 0.0, xs))
    # the forward rule is order-preserving on ordinary values
    rng = np.random.default_rng(7)
    r = rng.standard_normal(1000) * 1e6
    f = forward(np.sort(r))
    # element compare, not np.diff: the key span is nearly the full int64
    # range, so a sign-crossing difference overflows int64
    assert (f[1:] > f[:-1]).all()
    # all NaNs (any payload/sign) map to ONE key above ordered(+inf)
    nans = np.array([np.nan, -np.nan, np.float64("nan")])
    fk = forward(nans)
    assert len(set(fk.tolist())) == 1 and fk[0] > forward([np.inf])[0]


def test_decode_encode_dict_roundtrip():
    """Host dictionary encode/decode round-trips values incl. NaN."""
    import numpy as np
    import pandas
    from modin_amd.core.partition import decode_dict, encode_dict
    s = pandas.Series(["b", None, "a", "b", np.nan, "cc"])
    codes, cats = encode_dict(s)
    assert list(cats) == ["a", "b", "cc"]  # sorted-cats invariant
    assert codes.dtype == np.int64 and codes[1] == -1 and codes[4] == -1
    back = decode_dict(np.asarray(codes), cats)
    for got, exp in zip(back, s):
        if isinstance(exp, str):
            assert got == exp
        else:
            assert isinstance(got, float) and np.isnan(got)


def test_ffill_bfill_composition_prototype():
    """Round-2 de-risk (DESIGN roadmap 6): the groupby ffill/bfill device
    composition — stable key sort, run heads, segmented MAX over
    (valid ? position : −1), gather, inverse scatter, NaN-key fixup —
    restated in numpy and pinned against pandas groupby.ffill/bfill.
    Every step below maps 1:1 onto an existing hf_* primitive."""
    import numpy as np
    import pandas

    def seg_ffill(keys, vals, reverse=False):
        n = len(keys)
        valid_key = ~np.isnan(keys)
        # effective key: canonical-NaN sentinel sorts last (ordered_i64)
        eff = np.where(valid_key, keys, np.inf)
        if reverse:
            # bfill = ffill over reversed rows (hf_gather by reversed iota)
            rev = np.arange(n - 1, 0 - 1, -1)
            out = seg_ffill(keys[rev], vals[rev], False)
            return out[rev]
        perm = np.argsort(eff, kind="stable")      # hf_sort_perm
        sk, sv = eff[perm], vals[perm]
        head = np.empty(n, dtype=np.int64)         # run_head compose
        head[0] = 1
        head[1:] = (sk[1:] != sk[:-1]).astype(np.int64)
        rid = np.cumsum(head) - 1                  # hf_cumsum (i64)
        posv = np.where(~np.isnan(sv), np.arange(n), -1)  # compare+mul
        # segmented max scan == hf_seg_cumsum(posv, head, AGG_MAX)
        segmax = np.empty(n, dtype=np.int64)
        cur = -1
        for i in range(n):                         # (device: 3-phase scan)
            if head[i]:
                cur = -1
            cur = max(cur, posv[i])
            segmax[i] = cur
        filled = np.where(segmax >= 0, sv[np.clip(segmax, 0, n - 1)],
                          np.nan)                  # hf_gather + fixup
        out = np.empty(n)
        out[perm] = filled                         # hf_scatter
        out[~valid_key] = np.nan                   # valid-key fixup
        return out

    rng = np.random.default_rng(11)
    n = 4000
    keys = rng.integers(0, 40, n).astype(np.float64)
    keys[rng.random(n) < 0.05] = np.nan
    vals = rng.standard_normal(n)
    vals[rng.random(n) < 0.3] = np.nan
    pdf = pandas.DataFrame({"k": keys, "v": vals})
    exp_f = pdf.groupby("k")["v"].ffill().to_numpy()
    got_f = seg_ffill(keys, vals)
    np.testing.assert_allclose(got_f, exp_f, rtol=0, equal_nan=True)
    exp_b = pdf.groupby("k")["v"].bfill().to_numpy()
    got_b = seg_ffill(keys, vals, reverse=True)
    np.testing.assert_allclose(got_b, exp_b, rtol=0, equal_nan=True)


def test_coshuffled_merge_composition_prototype():
    """Round-2 de-risk (DESIGN roadmap 2): the co-shuffled merge for
    giant right tables — both sides range-binned by the same splitters
    (hf_shuffle_dest), per-bin dense CSR join, and pandas' left-major
    match order restored by a stable sort on the global left row id.
    Restated in numpy and pinned against pandas.merge; every step maps
    onto an existing primitive (shuffle_dest / filter / join_probe /
    sort_perm / gather)."""
    import numpy as np
    import pandas

    rng = np.random.default_rng(13)
    nl, nr = 30_000, 8_000
    lk = rng.integers(-10**12, 10**12, nl) // 10**7  # clustered keys
    rk = rng.integers(-10**12, 10**12, nr) // 10**7
    lv = rng.standard_normal(nl)
    rv = rng.standard_normal(nr)
    P = 8
    spl = oracle.pick_splitters(np.sort(rk)[:: max(nr // 512, 1)], P)
    ld = oracle.shuffle_dest(lk, spl)
    rd = oracle.shuffle_dest(rk, spl)
    out_keys, out_lidx, out_a, out_b = [], [], [], []
    for p in range(P):                       # per-bin local CSR join
        lsel = np.nonzero(ld == p)[0]
        rsel = np.nonzero(rd == p)[0]
        k, lidx, la, rb = oracle.inner_join(
            lk[lsel], {"a": lv[lsel]}, rk[rsel], {"b": rv[rsel]})
        out_keys.append(k)
        out_lidx.append(lsel[lidx])          # back to GLOBAL left rows
        out_a.append(la["a"])
        out_b.append(rb["b"])
    keys = np.concatenate(out_keys)
    glidx = np.concatenate(out_lidx)
    a = np.concatenate(out_a)
    b = np.concatenate(out_b)
    # pandas order: left-row-major, right matches in right-row order —
    # a STABLE sort by global left id restores it because each bin's
    # per-left matches are already right-row ordered
    order = np.argsort(glidx, kind="stable")
    keys, a, b = keys[order], a[order], b[order]
    exp = pandas.DataFrame({"k": lk, "a": lv}).merge(
        pandas.DataFrame({"k": rk, "b": rv}), on="k")
    np.testing.assert_array_equal(keys, exp["k"].to_numpy())
    np.testing.assert_allclose(a, exp["a"].to_numpy(), rtol=0)
    np.testing.assert_allclose(b, exp["b"].to_numpy(), rtol=0)


def test_distributed_median_composition_prototype():
    """Round-2 de-risk: distributed median/quantile rides the existing
    range shuffle — after shuffling VALUES (ordered-transformed) by
    sampled splitters, rank r holds a contiguous value range; the global
    middle positions land on one rank, found by an exclusive prefix of
    per-rank non-NaN counts (one tiny allgather).  numpy restatement vs
    numpy median."""
    import numpy as np

    rng = np.random.default_rng(17)
    n = 50_001
    v = rng.standard_normal(n) * 100
    v[rng.random(n) < 0.1] = np.nan
    P = 4
    vv = v[~np.isnan(v)]                    # NOTNA filter per rank
    spl = oracle.pick_splitters(
        np.sort(vv)[:: max(len(vv) // 512, 1)].astype(np.float64), P)
    # the real path bins the ordered-i64 form via hf_shuffle_dest; float
    # compares are equivalent for finite values here
    dest = np.searchsorted(spl, vv, side="right")
    shards = [np.sort(vv[dest == p], kind="stable") for p in range(P)]
    counts = np.array([len(s) for s in shards])
    total = counts.sum()
    mids = [(total - 1) // 2, total // 2]   # lower/upper middle
    got = []
    base = np.concatenate([[0], np.cumsum(counts)])
    for m in mids:
        r = int(np.searchsorted(base, m, side="right")) - 1
        got.append(shards[r][m - base[r]])
    np.testing.assert_allclose(np.mean(got), np.nanmedian(v), rtol=1e-15)


def test_multikey_idx_nan_keys_prototype():
    """Round-2 de-risk: NaN keys in MULTI-key idxmax — filter invalid
    rows first (keeping original positions via filter_iota), run the
    existing composition on the filtered frame, and report the kept
    positions; pinned vs pandas."""
    import numpy as np
    import pandas

    rng = np.random.default_rng(19)
    n = 5000
    a = rng.integers(0, 6, n).astype(np.float64)
    a[rng.random(n) < 0.05] = np.nan
    b = rng.integers(0, 4, n).astype(np.float64)
    b[rng.random(n) < 0.05] = np.nan
    v = rng.integers(-5, 5, n).astype(np.float64)  # heavy ties
    pdf = pandas.DataFrame({"a": a, "b": b, "v": v})
    keep = ~(np.isnan(a) | np.isnan(b))            # filter_plan
    pos = np.nonzero(keep)[0]                      # filter_iota
    fa, fb, fv = a[keep], b[keep], v[keep]
    # existing composition on the filtered rows: stable sort by
    # (a, b, v), key-run ends via counts, tie-block start = first max
    perm = np.lexsort((fv, fb, fa))                # device: 3 stable passes
    sa, sb, sv = fa[perm], fb[perm], fv[perm]
    head = np.ones(len(sa), dtype=bool)
    head[1:] = (sa[1:] != sa[:-1]) | (sb[1:] != sb[:-1])
    starts = np.nonzero(head)[0]
    ends = np.append(starts[1:], len(sa))
    got = {}
    for s, e in zip(starts, ends):
        mx = sv[e - 1]                             # NaN-free values here
        ts = s + int(np.searchsorted(sv[s:e], mx, side="left"))
        got[(sa[s], sb[s])] = pos[perm[ts]]
    exp = pdf.groupby(["a", "b"]).idxmax()
    for (ka, kb), row in zip(exp.index, exp["v"]):
        assert got[(ka, kb)] == row, (ka, kb)


def test_string_minmax_first_composition_prototype():
    """Round-2 de-risk: groupby min/max over STRING value columns — the
    sorted-cats invariant makes lexicographic order == code order, so
    min/max = the existing groupby over codes; decode at the end.
    Pinned vs pandas on NaN-free strings.  (pandas 2.3 itself RAISES on
    object min/max when a group contains NaN — TypeError in its py
    fallback — so the NaN-string case must raise loudly, not improvise.)
    """
    import numpy as np
    import pandas

    rng = np.random.default_rng(23)
    n = 6000
    cats = pandas.Index(["apple", "fig", "pear", "zebra"])  # sorted
    codes = rng.integers(0, len(cats), n).astype(np.int64)
    keys = rng.integers(0, 30, n).astype(np.int64)
    for agg in ("min", "max"):
        gk, gv = oracle.groupby_agg(keys, {"s": codes.astype(np.float64)},
                                    agg)
        got = cats.to_numpy(dtype=object)[
            np.asarray(gv["s"]).astype(np.int64)]
        pdf = pandas.DataFrame(
            {"k": keys, "s": cats.to_numpy(dtype=object)[codes]})
        exp = getattr(pdf.groupby("k")["s"], agg)()
        np.testing.assert_array_equal(gk, exp.index.to_numpy())
        np.testing.assert_array_equal(got, exp.to_numpy())


def test_unbounded_multikey_groupby_prototype():
    """Round-2 de-risk: multi-key groupby BEYOND the 2^62 combined-span
    fold needs no tuple hash — per-column stable LSD sort (any spans),
    OR'd per-column run heads delimit the tuple groups exactly, and the
    sorted/segmented aggregation path consumes the head flags.  numpy
    restatement vs pandas; collision-free by construction."""
    import numpy as np
    import pandas

    rng = np.random.default_rng(29)
    n = 20_000
    a = rng.integers(-2**62, 2**62, n)          # full-span keys
    a = a - (a % 10**15)                        # ~9e3 distinct
    b = rng.integers(-2**62, 2**62, n)
    b = b - (b % (3 * 10**17))
    v = rng.standard_normal(n)
    perm = np.lexsort((b, a))                   # device: stable LSD passes
    sa, sb, sv = a[perm], b[perm], v[perm]
    head = np.ones(n, dtype=bool)
    head[1:] = (sa[1:] != sa[:-1]) | (sb[1:] != sb[:-1])
    starts = np.nonzero(head)[0]
    ends = np.append(starts[1:], n)
    sums = np.add.reduceat(sv, starts)
    exp = pandas.DataFrame({"a": a, "b": b, "v": v}).groupby(
        ["a", "b"])["v"].sum()
    np.testing.assert_array_equal(sa[starts],
                                  exp.index.get_level_values(0).to_numpy())
    np.testing.assert_array_equal(sb[starts],
                                  exp.index.get_level_values(1).to_numpy())
    np.testing.assert_allclose(sums, exp.to_numpy(), rtol=1e-12)
    assert len(starts) == len(exp) and ends[-1] == n


def test_rolling_composition_prototype():
    """Round-2 design pin: pandas rolling(w) without new kernel families.
    sum/mean: windowed difference of NaN-zero-filled prefix sums + a
    windowed non-NaN count against min_periods (pandas default
    min_periods=w -> any NaN in the window yields NaN).  min/max: the
    van Herk/Gil-Werman two-scan trick — tile the rows at width w, take
    suffix-max within each tile (a segmented scan with heads at tile
    starts, exactly hf_seg_cumsum AGG_MAX) and prefix-max (the existing
    hf_cumsum-per-tile shape); window max at i = comb(suffix[i-w+1],
    prefix[i]).  Pinned vs pandas."""
    import numpy as np
    import pandas

    rng = np.random.default_rng(31)
    n = 5000
    v = rng.standard_normal(n)
    v[rng.random(n) < 0.1] = np.nan
    s = pandas.Series(v)
    for w in (1, 3, 16, 100):
        zf = np.where(np.isnan(v), 0.0, v)
        cs = np.concatenate([[0.0], np.cumsum(zf)])
        cc = np.concatenate([[0], np.cumsum(~np.isnan(v))])
        wsum = cs[w:] - cs[:-w]                # rows w-1..n-1
        wcnt = cc[w:] - cc[:-w]
        out = np.full(n, np.nan)
        ok = wcnt == w                         # min_periods = w
        out[w - 1:][ok] = wsum[ok]
        np.testing.assert_allclose(out, s.rolling(w).sum().to_numpy(),
                                   rtol=1e-12, atol=1e-12, equal_nan=True,
                                   err_msg=f"sum/w={w}")
        outm = np.full(n, np.nan)
        outm[w - 1:][ok] = wsum[ok] / w
        np.testing.assert_allclose(outm, s.rolling(w).mean().to_numpy(),
                                   rtol=1e-12, atol=1e-12, equal_nan=True,
                                   err_msg=f"mean/w={w}")
        # van Herk max: NaN rows poison their windows under min_periods=w,
        # so compute over zero... no — compute over v with NaN->-inf and
        # mask windows containing NaN afterwards
        vv = np.where(np.isnan(v), -np.inf, v)
        ntiles = -(-n // w)
        pad = ntiles * w
        vp = np.full(pad, -np.inf)
        vp[:n] = vv
        tiles = vp.reshape(ntiles, w)
        pref = np.maximum.accumulate(tiles, axis=1).reshape(pad)
        suff = np.maximum.accumulate(tiles[:, ::-1], axis=1)[:, ::-1] \
            .reshape(pad)
        wmax = np.full(n, np.nan)
        for i in range(w - 1, n):
            lo = i - w + 1
            wmax[i] = max(suff[lo], pref[i]) if (lo // w) != (i // w) \
                else pref[i]
        res = np.full(n, np.nan)
        res[w - 1:][ok] = wmax[w - 1:][ok]
        np.testing.assert_allclose(res, s.rolling(w).max().to_numpy(),
                                   rtol=0, equal_nan=True,
                                   err_msg=f"max/w={w}")
