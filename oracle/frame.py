"""Oracle restatement of the reference hot-path operator semantics (pandas).

Each function cites the reference file:line it follows. TEST INFRASTRUCTURE
ONLY — see oracle/__init__.py header.
"""
from __future__ import annotations

import numpy as np
import pandas as pd

# ---------------------------------------------------------------------------
# Filter — reference filter.py:20-45 (filter_or_scalar)
# ---------------------------------------------------------------------------


def oracle_filter(df: pd.DataFrame, condition) -> pd.DataFrame:
    """SQL WHERE semantics.

    Reference: dask_sql/physical/rel/logical/filter.py:20-45 —
    scalar condition short-circuits (:31-36); otherwise SQL NULL-in-boolean
    is False on filtering (`fillna(False)`, :39) then boolean mask take
    (`df[filter_condition]`, :40). Row order preserved.
    """
    if np.isscalar(condition):
        if not condition:
            return df.head(0)
        return df
    condition = pd.Series(condition, index=df.index)
    if condition.dtype == object or str(condition.dtype).startswith(("boolean", "Boolean")):
        condition = condition.astype("boolean")
    condition = condition.fillna(False).astype(bool)
    return df[condition]


# ---------------------------------------------------------------------------
# Join — reference join.py:50-322
# ---------------------------------------------------------------------------

_JOIN_TYPE_MAPPING = {
    # reference join.py:41-48
    "INNER": "inner",
    "LEFT": "left",
    "RIGHT": "right",
    "FULL": "outer",
    "LEFTSEMI": "leftsemi",
    "LEFTANTI": "leftanti",
}


def oracle_join(
    lhs: pd.DataFrame,
    rhs: pd.DataFrame,
    lhs_on: list,
    rhs_on: list,
    how: str = "INNER",
    residual=None,
) -> pd.DataFrame:
    """Equijoin + optional residual filter, reference join.py semantics.

    - column namespace: lhs columns renamed lhs_i, rhs columns rhs_i by
      POSITION (reference join.py:65-72 make_unique), output = lhs cols then
      rhs cols (join.py:146-166).
    - NULL-key drop (join.py:202-213): inner/right drop NULL keys on lhs;
      inner/left/leftanti/leftsemi drop NULL keys on rhs.
    - merge on temp key columns (join.py:215-246): pandas merge, how mapped
      per JOIN_TYPE_MAPPING (join.py:41-48); leftsemi falls back to inner on
      CPU (join.py:78-79); leftanti via left-merge + indicator
      (join.py:229-239).
    - FULL OUTER fills with NaN not NA (reference test_join.py:55-65).
    - residual: callable(df)->mask applied via filter_or_scalar
      (join.py:169-181).

    lhs_on / rhs_on are column POSITIONS (integer indices) as the plan
    provides them (join.py:250-322 extracts InputRef indices).
    """
    how_pd = _JOIN_TYPE_MAPPING[how]
    if how_pd == "leftsemi":
        how_pd = "inner"  # reference join.py:78-79 (CPU)

    lhs_r = lhs.copy()
    rhs_r = rhs.copy()
    lhs_r.columns = [f"lhs_{i}" for i in range(len(lhs.columns))]
    rhs_r.columns = [f"rhs_{i}" for i in range(len(rhs.columns))]

    if lhs_on:
        # NULL-key drop, reference join.py:202-213
        if how_pd in ("inner", "right"):
            keep = np.ones(len(lhs_r), dtype=bool)
            for idx in lhs_on:
                keep &= ~lhs_r.iloc[:, idx].isna().to_numpy()
            lhs_r = lhs_r[keep]
        if how_pd in ("inner", "left", "leftanti"):
            keep = np.ones(len(rhs_r), dtype=bool)
            for idx in rhs_on:
                keep &= ~rhs_r.iloc[:, idx].isna().to_numpy()
            rhs_r = rhs_r[keep]

        # temp common_i key columns, reference join.py:215-226
        for i, (li, ri) in enumerate(zip(lhs_on, rhs_on)):
            lhs_r = lhs_r.assign(**{f"common_{i}": lhs_r.iloc[:, li]})
            rhs_r = rhs_r.assign(**{f"common_{i}": rhs_r.iloc[:, ri]})
        added = [f"common_{i}" for i in range(len(lhs_on))]

        if how_pd == "leftanti":
            # reference join.py:229-239
            df = lhs_r.merge(rhs_r, on=added, how="left", indicator=True).drop(
                columns=added
            )
            df = df[df["_merge"] == "left_only"].drop(
                columns=["_merge"] + list(rhs_r.drop(columns=added).columns),
                errors="ignore",
            )
        else:
            df = lhs_r.merge(rhs_r, on=added, how=how_pd).drop(columns=added)
    else:
        # cross join via constant key, reference join.py:133-140
        df = lhs_r.assign(common=1).merge(rhs_r.assign(common=1), on="common").drop(
            columns="common"
        )

    # column order: lhs then rhs (join.py:146-152); leftanti keeps lhs only
    if how in ("LEFTSEMI", "LEFTANTI") and how_pd != "inner":
        order = list(lhs_r.columns[: len(lhs.columns)])
    else:
        order = [f"lhs_{i}" for i in range(len(lhs.columns))] + [
            f"rhs_{i}" for i in range(len(rhs.columns))
        ]
        order = [c for c in order if c in df.columns]
    df = df[order]

    if residual is not None:
        mask = residual(df)
        df = oracle_filter(df, mask)  # join.py:169-181 via filter_or_scalar
    return df


# ---------------------------------------------------------------------------
# Groupby-aggregate — reference aggregate.py:117-589
# ---------------------------------------------------------------------------


def _custom_sum(s: pd.Series):
    """SUM with SQL NULL semantics: sum of all-NULL group is NULL, not 0.

    Reference aggregate.py:486-493 (custom_sum, min_count=1).
    """
    return s.sum(min_count=1)


_AGG_FUNCS = {
    # reference AGGREGATION_MAPPING aggregate.py:117-231 (in-scope subset) +
    # custom_sum aggregate.py:486-493; count skips NULLs (pandas "count");
    # avg = mean.
    "sum": _custom_sum,
    "count": "count",
    "avg": "mean",
    "min": "min",
    "max": "max",
    "any_value": "first",
    "single_value": "first",
    # stddev family: pandas std/var, ddof=1 for sample forms (the
    # reference's dask "std" aggregation default), ddof=0 for *_POP
    "stddev": "std",
    "stddev_samp": "std",
    "stddev_pop": lambda s: s.std(ddof=0),
    "var_samp": "var",
    "variance": "var",
    "var_pop": lambda s: s.var(ddof=0),
}


def oracle_groupby(
    df: pd.DataFrame,
    group_cols: list,
    aggs: list,
) -> pd.DataFrame:
    """GROUP BY + aggregations, reference aggregate.py semantics.

    aggs: list of (input_col, output_name, func_name, filter_col_or_None,
    distinct_bool). func_name in _AGG_FUNCS; COUNT(*) is count over a
    constant-1 column the caller adds (reference aggregate.py:305-306).

    Semantics restated:
    - groupby(..., dropna=False): NULL groups kept (aggregate.py:575-577).
    - aggregations bucketed by (filter_col, distinct): each bucket is its own
      groupby pass, non-filtered bucket FIRST so no groups are lost; others
      joined onto its index (aggregate.py:336-374).
    - filtered bucket: pre-filter rows (aggregate.py:558-561); groups with no
      qualifying rows → NaN/NULL (test_groupby.py:136).
    - distinct: drop_duplicates over group_cols+input before aggregating
      (aggregate.py:562-565).
    - no group_cols: full-table aggregation → single row (aggregate.py:251).
    """
    from collections import defaultdict

    buckets = defaultdict(list)
    output_order = list(group_cols)
    for input_col, out_name, func, filter_col, distinct in aggs:
        buckets[(filter_col, distinct)].append((input_col, out_name, func))
        output_order.append(out_name)

    const_col = "__const_1__"
    work = df.assign(**{const_col: 1})

    def one_bucket(filter_col, distinct, items):
        tmp = work
        if filter_col is not None:
            tmp = tmp[pd.Series(tmp[filter_col]).fillna(False).astype(bool)]
        if distinct:
            subset = list(group_cols) + sorted({ic for ic, _, _ in items})
            tmp = tmp.drop_duplicates(subset=subset)
        named = {
            out_name: pd.NamedAgg(column=input_col, aggfunc=_AGG_FUNCS[func])
            for input_col, out_name, func in items
        }
        if group_cols:
            g = tmp.groupby(list(group_cols), dropna=False)
            return g.agg(**named)
        g = tmp.groupby([const_col])
        res = g.agg(**named)
        if len(res) == 0:
            # global aggregate over zero rows: SQL yields ONE row — count 0,
            # everything else NULL (reference aggregate.py:251 whole-frame agg)
            def _apply(ic, f):
                fn = _AGG_FUNCS[f]
                s = pd.Series(tmp[ic])
                return fn(s) if callable(fn) else s.agg(fn)

            row = {out_name: [_apply(ic, f)] for ic, out_name, f in items}
            res = pd.DataFrame(row, index=pd.Index([1], name=const_col))
        return res

    # non-filtered, non-distinct bucket first (aggregate.py:336-350)
    df_result = None
    key0 = (None, False)
    keys = sorted(buckets.keys(), key=lambda k: (k != key0, str(k)))
    for key in keys:
        filter_col, distinct = key
        res = one_bucket(filter_col, distinct, buckets[key])
        if df_result is None:
            df_result = res
        else:
            df_result = df_result.join(res, how="left")

    if df_result is None:
        # pure DISTINCT node: drop_duplicates (aggregate.py:329-332)
        out = work[list(group_cols)].drop_duplicates()
        return out.reset_index(drop=True)

    df_result = df_result.reset_index(drop=not group_cols)
    if not group_cols:
        df_result = df_result.drop(columns=[const_col], errors="ignore")
    cols = [c for c in output_order if c in df_result.columns]
    return df_result[cols]
