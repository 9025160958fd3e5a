"""Extract SQL strings from the reference's integration tests and try to
PLAN each against approximated fixtures. Reports planning failures by
category — a coverage probe, not a test (OUT-scope constructs expected
to fail)."""
import re
import sys
from pathlib import Path

sys.path.insert(0, "/root/repo")
import numpy as np
import pandas as pd

from dask_sql_amd.context import Context

REF = Path("/root/reference/tests/integration")
FILES = ["test_select.py", "test_filter.py", "test_groupby.py",
         "test_join.py", "test_sort.py", "test_union.py", "test_rex.py",
         "test_complex.py", "test_distributeby.py",
         "test_over.py",
         "test_function.py"]

c = Context()
np.random.seed(42)
# fixtures.py approximations
c.create_table("df", pd.DataFrame(
    {"a": [1.0] * 100 + [2.0] * 200 + [3.0] * 400,
     "b": 10 * np.random.rand(700),
     # extra columns several test files add to their local `df` frames
     "c": np.arange(700, dtype="int64"),
     "d": pd.date_range("2021-01-01", periods=700, freq="7h"),
     "t": pd.date_range("2020-06-01", periods=700, freq="13h"),
     "x": np.random.rand(700),
     "y": np.random.rand(700)}))
c.create_table("df_simple", pd.DataFrame(
    {"a": [1, 2, 3], "b": [1.1, 2.2, 3.3]}))
c.create_table("df_wide", pd.DataFrame(
    {"a": [1, 2], "b": [3, 4], "c": [5, 6]}))
c.create_table("user_table_1", pd.DataFrame(
    {"user_id": [2, 1, 2, 3], "b": [3, 3, 1, 3]}))
c.create_table("user_table_2", pd.DataFrame(
    {"user_id": [1, 1, 2, 4], "c": [1, 2, 3, 4]}))
c.create_table("long_table", pd.DataFrame({"a": [0] * 100}))
c.create_table("user_table_nan", pd.DataFrame(
    {"c": pd.array([3, pd.NA, 1], dtype="Int8")}))
c.create_table("user_table_inf", pd.DataFrame(
    {"c": [3, float("inf"), 1]}))
c.create_table("string_table", pd.DataFrame(
    {"a": ["a normal string", "%_%", "^|()-*[]$"]}))
c.create_table("datetime_table", pd.DataFrame({
    "timezone": pd.date_range("2014-08-01", periods=3, freq="h"),
    "no_timezone": pd.date_range("2014-08-01", periods=3, freq="h"),
    "utc_timezone": pd.date_range("2014-08-01", periods=3, freq="h"),
}))
c.create_table("df1", pd.DataFrame(
    {"id": [1, 2], "a": [1, 2], "b": pd.Series(["w", "x"]
                                               ).astype("category")}))
c.create_table("df2", pd.DataFrame(
    {"id": [1, 2], "c": [2, 3], "d": pd.Series(["h", "i"]
                                               ).astype("category")}))
c.create_table("df_1", pd.DataFrame({"id": [1, 2, 3]}))
c.create_table("df_2", pd.DataFrame({"id": [2, 3, 4]}))
c.create_table("dates", pd.DataFrame(
    {"d": pd.date_range("2021-01-01", periods=5, freq="D")}))
c.create_table("datetime_test", pd.DataFrame(
    {"a": pd.date_range("2021-01-01", periods=5, freq="D"),
     "dt": pd.date_range("2014-01-01", periods=5, freq="250D"),
     "b": np.arange(5)}))
# dask.datasets.timeseries schema: id/name/x/y
c.create_table("timeseries", pd.DataFrame(
    {"id": np.arange(30), "name": pd.Series(["Alice", "Bob", "Xavier"] * 10
                                            ).astype("category"),
     "x": np.random.rand(30) * 2 - 1, "y": np.random.rand(30) * 2 - 1}))
c.create_table("department_table", pd.DataFrame(
    {"department_name": ["English", "Math", "Science"]}))
c.create_table("string_table2", pd.DataFrame(
    {"b": pd.Series(["a", "b", None]).astype("category")}))
c.create_table("d_table", pd.DataFrame(
    {"d_date": pd.to_datetime(["2023-07-01", "2023-07-05"]),
     "x": [1, 2]}))
c.create_table("sales", pd.DataFrame(
    {"sales_hdemo_sk": [1], "sales_page_sk": [1], "sold_time_sk": [1]}))
c.create_table("demos", pd.DataFrame(
    {"demo_sku": [1], "hd_dep_count": [1]}))
c.create_table("site_page", pd.DataFrame(
    {"site_page_sk": [1], "site_char_count": [1]}))
c.create_table("t_dim", pd.DataFrame({"t_time_sk": [1], "t_hour": [1]}))
c.create_table("many_partitions", pd.DataFrame(
    {"a": [1, 2], "b": [3, 4], "c": [5, 6]}))
c.create_table("parquet_ddf", pd.DataFrame(
    {"a": [1, 2, 3], "b": [0, 1, 2],
     "c": pd.Series(["A"] * 3).astype("category"),
     "d": pd.to_datetime(["2013-08-01 23:00:00"] * 3),
     "index": [0, 1, 2]}))
c.create_table("my_csv_table", pd.DataFrame(
    {"a": [1, 2], "b": [1.0, 2.0], "c": [2, 3]}))
c.create_table("gpu_df", pd.DataFrame({"a": [1.0], "b": [1.0]}))
c.create_table("gpu_user_table_1", pd.DataFrame(
    {"user_id": [2], "b": [3]}))
c.create_table("gpu_long_table", pd.DataFrame({"a": [0]}))
c.create_table("gpu_string_table", pd.DataFrame({"a": ["x"]}))
c.create_table("gpu_datetime_table", pd.DataFrame(
    {"timezone": pd.date_range("2014-08-01", periods=3, freq="h")}))

sql_rx = re.compile(r'(?:c|context)\.sql\(\s*(?:f?"""(.*?)"""|f?"([^"]+)")',
                    re.S)
ok = bad = skipped = 0
fails = {}
for fn in FILES:
    text = (REF / fn).read_text()
    for m in sql_rx.finditer(text):
        q = (m.group(1) or m.group(2)).strip()
        if not q.upper().startswith(("SELECT", "WITH")):
            skipped += 1
            continue
        if "{" in q:  # f-string templates — substitute common params
            q2 = (q.replace("{input_table_1}", "user_table_1")
                   .replace("{input_df}", "df")
                   .replace("{input_table}", "user_table_1")
                   .replace("{table}", "user_table_1")
                   .replace("{gpu_t}", "df"))
            if "{" in q2:
                skipped += 1
                continue
            q = q2
        try:
            rel = c._get_ral(q)
            ok += 1
        except Exception as e:
            bad += 1
            key = f"{type(e).__name__}: {str(e)[:90]}"
            fails.setdefault(key, []).append((fn, q[:100].replace("\n", " ")))
            continue
        if "--compile" in sys.argv:
            # rex-compile every Projection/Filter/Join expression in the
            # plan against fake device columns of the planned dtypes —
            # catches EXECUTION-time compile gaps without a GPU. String
            # exprs legitimately route through the dict LUT paths
            # (dict_string_fn/dict_int_fn) instead of the VM; exprs whose
            # compile fails but whose dict path succeeds are NOT gaps.
            import types as _t

            from dask_sql_amd import runtime as rt
            from dask_sql_amd.physical import rex as R

            _SQL_DT = {"BIGINT": rt.I64, "INTEGER": rt.I64,
                       "SMALLINT": rt.I64, "TINYINT": rt.I64,
                       "DATE": rt.I32, "TIMESTAMP": rt.I64,
                       "VARCHAR": rt.I32, "DOUBLE": rt.F64,
                       "FLOAT": rt.F64, "DECIMAL": rt.F64,
                       "BOOLEAN": rt.BOOL8, "NULL": rt.I64}

            def cols_of(node):
                fs = node.getRowType().getFieldList()
                cols = [_t.SimpleNamespace(
                    dtype=_SQL_DT.get(f.getType().getSqlType(), rt.I64))
                    for f in fs]
                dicts = [["aa", "bb", "cc"]
                         if f.getType().getSqlType() == "VARCHAR" else None
                         for f in fs]
                return cols, dicts

            def walk(node):
                t = node.get_current_node_type()
                exprs = []
                if t == "Projection":
                    exprs = [e for e, _ in
                             node.projection().getNamedProjects()]
                elif t == "Filter":
                    exprs = [node.filter().getCondition()]
                for ins in node.get_inputs():
                    walk(ins)
                if not exprs or not node.get_inputs():
                    return
                cols, dicts = cols_of(node.get_inputs()[0])
                for e in exprs:
                    try:
                        R.RexCompiler(cols, dicts).compile(e)
                    except R.RexCompileError as ex:
                        if "UDF:" in str(ex):
                            continue  # host-evaluated (_eval_udf_nodes)
                        if R.dict_string_fn(e, dicts) is not None:
                            continue
                        if R.dict_int_fn(e, dicts) is not None:
                            continue
                        if R.fold_string_literal(e) is not None:
                            continue
                        try:
                            from dask_sql_amd.physical.rel_plugins import                                 _case_string_rewrite
                            if _case_string_rewrite(e) is not None:
                                continue
                        except Exception:
                            pass
                        key = f"COMPILE {str(ex)[:80]}"
                        fails.setdefault(key, []).append(
                            (fn, q[:100].replace("\n", " ")))
                    except Exception as ex:
                        key = f"COMPILE-{type(ex).__name__}: {str(ex)[:70]}"
                        fails.setdefault(key, []).append(
                            (fn, q[:100].replace("\n", " ")))
            try:
                walk(rel)
            except Exception as ex:
                fails.setdefault(f"WALK-{type(ex).__name__}: "
                                 f"{str(ex)[:70]}", []).append((fn, q[:80]))
print(f"planned OK: {ok}  failed: {bad}  skipped(non-select/f-str): {skipped}")
for k, v in sorted(fails.items(), key=lambda kv: -len(kv[1]))[:25]:
    print(f"\n[{len(v)}x] {k}")
    for fn, q in v[:2]:
        print(f"    {fn}: {q}")
