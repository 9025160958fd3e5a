"""Context — the user-facing API, kept verbatim-shaped with the reference
(dask_sql/context.py): Context(), create_table (:168), sql (:482),
drop_table, explain. The planner is ours (thin); execution is
the HIP physical layer (no CPU fallback — DsxUnavailable propagates)."""
from __future__ import annotations

import logging
import re

import numpy as np
import pandas as pd

from dask_sql_amd import runtime as rt
from dask_sql_amd.datacontainer import DeviceTable, HostDataContainer
from dask_sql_amd.planner.builder import Builder, Catalog
from dask_sql_amd.physical.convert import RelConverter
from dask_sql_amd.physical.rel_plugins import register_defaults

logger = logging.getLogger(__name__)

register_defaults()


class _HostColumn:
    def __init__(self, arr, validity, sql_type, dtype, dictionary=None,
                 tz=None):
        self.arr = arr
        self.validity = validity
        self.sql_type = sql_type
        self.dtype = dtype
        self.dictionary = dictionary
        self.tz = tz  # original timezone of a tz-aware datetime column


def _from_pandas(df: pd.DataFrame, date_columns=(), dictionaries=None):
    """pandas → host column set (reference input_utils/convert.py:40-60
    functionally: anything the user hands in becomes our columnar form)."""
    dictionaries = dictionaries or {}
    cols = {}
    for name in df.columns:
        s = df[name]
        validity = None
        dictionary = None
        dt = s.dtype
        if isinstance(dt, pd.CategoricalDtype):
            codes = s.cat.codes.to_numpy()
            validity = (codes >= 0).astype(np.uint8)
            arr = np.where(codes >= 0, codes, 0).astype(np.int32)
            dtype, sqlt = rt.I32, "VARCHAR"
            dictionary = list(s.cat.categories)
            if validity.all():
                validity = None
        elif isinstance(dt, pd.DatetimeTZDtype):
            # tz-aware datetimes: store UTC ns, reattach the zone at
            # materialization (reference keeps the dtype through pandas)
            nat = s.isna().to_numpy()
            validity = (~nat).astype(np.uint8) if nat.any() else None
            ns = s.dt.tz_convert("UTC").dt.tz_localize(None) \
                .to_numpy().astype("datetime64[ns]").astype(np.int64)
            arr = np.where(nat, 0, ns)
            dtype, sqlt = rt.I64, "TIMESTAMP"
            col = _HostColumn(np.ascontiguousarray(arr), validity, sqlt,
                              dtype, None, tz=str(dt.tz))
            cols[str(name)] = col
            continue
        elif isinstance(dt, pd.StringDtype):
            # pandas "string" dtype → dictionary encoding like object
            # strings (factorize maps pd.NA to code -1)
            codes, uniques = pd.factorize(s)
            validity = (codes >= 0).astype(np.uint8)
            if validity.all():
                validity = None
            arr = np.where(codes >= 0, codes, 0).astype(np.int32)
            dtype, sqlt = rt.I32, "VARCHAR"
            dictionary = list(uniques)
        elif pd.api.types.is_extension_array_dtype(dt):
            # nullable Int*/UInt*/Float*/boolean
            mask = s.isna().to_numpy()
            base = s.fillna(False if dt == pd.BooleanDtype() else 0) \
                .to_numpy()
            if base.dtype == object:
                base = base.astype(np.float64)
            validity = (~mask).astype(np.uint8) if mask.any() else None
            arr, dtype, sqlt = _np_map(np.asarray(base))
        elif dt == object or pd.api.types.is_string_dtype(dt):
            codes, uniques = pd.factorize(s)
            validity = (codes >= 0).astype(np.uint8)
            if validity.all():
                validity = None
            arr = np.where(codes >= 0, codes, 0).astype(np.int32)
            dtype, sqlt = rt.I32, "VARCHAR"
            dictionary = list(uniques)
        elif np.issubdtype(dt, np.datetime64):
            nat = s.isna().to_numpy()
            validity = (~nat).astype(np.uint8) if nat.any() else None
            ns = s.to_numpy().astype("datetime64[ns]").astype(np.int64)
            day_ns = 86_400_000_000_000
            if ((ns % day_ns == 0) | nat).all():
                # midnight-only → day-int DATE (compact; reference
                # mappings.py:78-80 semantics are identical)
                vals = s.to_numpy().astype("datetime64[D]").astype(np.int32)
                arr, dtype, sqlt = vals, rt.I32, "DATE"
            else:
                # sub-day precision → ns-resolution TIMESTAMP (i64, exactly
                # the reference's datetime64[ns])
                arr = np.where(nat, 0, ns)
                dtype, sqlt = rt.I64, "TIMESTAMP"
        elif np.issubdtype(dt, np.floating):
            arr = s.to_numpy()
            # float NaN stays a VALUE (pandas semantics treat it as missing
            # in aggregations; we mirror by mapping NaN → NULL validity)
            nan = np.isnan(arr)
            validity = (~nan).astype(np.uint8) if nan.any() else None
            arr, dtype, sqlt = _np_map(arr)
        else:
            arr, dtype, sqlt = _np_map(s.to_numpy())
        if name in date_columns:
            sqlt = "DATE"
        if name in dictionaries:
            dictionary = dictionaries[name]
            sqlt = "VARCHAR"
        cols[str(name)] = _HostColumn(np.ascontiguousarray(arr), validity,
                                      sqlt, dtype, dictionary)
    return cols


def _from_arrow(table, date_columns=(), dictionaries=None):
    """pyarrow.Table → host column set with NO pandas round-trip (SURVEY
    §8f3 direct parquet ingest): primitive buffers are zero-copy numpy
    views, validity bitmaps unpack to byte masks, arrow dictionary columns
    map 1:1 onto our code+dictionary form, strings dictionary-encode in
    arrow. Upload then streams pinned→HBM (dsx_upload_pinned)."""
    import pyarrow as pa
    import pyarrow.compute as pc

    dictionaries = dictionaries or {}
    cols = {}
    for name, chunked in zip(table.column_names, table.columns):
        arr = chunked.combine_chunks() if chunked.num_chunks != 1             else chunked.chunk(0)
        if isinstance(arr, pa.ChunkedArray):
            arr = arr.combine_chunks()
        t = arr.type
        validity = None
        dictionary = None
        if arr.null_count:
            validity = pc.is_valid(arr).to_numpy(
                zero_copy_only=False).astype(np.uint8)
        if pa.types.is_string(t) or pa.types.is_large_string(t):
            arr = arr.dictionary_encode()
            t = arr.type
        if pa.types.is_dictionary(t):
            dictionary = arr.dictionary.to_pylist()
            codes = arr.indices
            vals = codes.fill_null(0).to_numpy(
                zero_copy_only=False).astype(np.int32)
            data, dtype, sqlt = vals, rt.I32, "VARCHAR"
        elif pa.types.is_date32(t):
            data = arr.view(pa.int32()).to_numpy(zero_copy_only=False)
            dtype, sqlt = rt.I32, "DATE"
        elif pa.types.is_timestamp(t):
            ns = arr.cast(pa.timestamp("ns")).view(pa.int64()).to_numpy(
                zero_copy_only=False)
            data, dtype, sqlt = ns, rt.I64, "TIMESTAMP"
        elif pa.types.is_boolean(t):
            data = arr.fill_null(False).to_numpy(
                zero_copy_only=False).astype(np.uint8)
            dtype, sqlt = rt.BOOL8, "BOOLEAN"
        else:
            if arr.null_count:
                arr = arr.fill_null(0)
            npv = arr.to_numpy(zero_copy_only=arr.null_count == 0)
            data, dtype, sqlt = _np_map(np.ascontiguousarray(npv))
        if name in date_columns:
            sqlt = "DATE"
        if name in dictionaries:
            dictionary = dictionaries[name]
            sqlt = "VARCHAR"
        cols[str(name)] = _HostColumn(np.ascontiguousarray(data), validity,
                                      sqlt, dtype, dictionary)
    return cols


def _np_map(arr):
    m = {
        np.dtype("int64"): (rt.I64, "BIGINT"),
        np.dtype("int32"): (rt.I32, "INTEGER"),
        np.dtype("int16"): (rt.I32, "INTEGER"),
        np.dtype("int8"): (rt.I8, "TINYINT"),
        np.dtype("uint8"): (rt.I32, "INTEGER"),
        np.dtype("uint16"): (rt.I32, "INTEGER"),
        np.dtype("uint32"): (rt.I64, "BIGINT"),
        np.dtype("uint64"): (rt.I64, "BIGINT"),
        np.dtype("float64"): (rt.F64, "DOUBLE"),
        np.dtype("float32"): (rt.F32, "FLOAT"),
        np.dtype("bool"): (rt.BOOL8, "BOOLEAN"),
    }
    if arr.dtype in (np.dtype("int16"), np.dtype("uint16")):
        arr = arr.astype(np.int32)
    elif arr.dtype in (np.dtype("uint8"),):
        arr = arr.astype(np.int32)
    elif arr.dtype in (np.dtype("uint32"), np.dtype("uint64")):
        arr = arr.astype(np.int64)
    if arr.dtype not in m:
        raise NotImplementedError(f"dtype {arr.dtype} not supported")
    dtype, sqlt = m[arr.dtype]
    return arr, dtype, sqlt


class RegisteredTable:
    def __init__(self, host_cols: dict):
        self.host_cols = host_cols
        self.device_table: DeviceTable | None = None

    def fields(self):
        return [(n, c.sql_type) for n, c in self.host_cols.items()]

    def upload(self, runtime) -> DeviceTable:
        if self.device_table is None:
            cols = {}
            for n, h in self.host_cols.items():
                col = runtime.upload_column(h.arr, h.validity, h.dtype)
                if h.dictionary is not None:
                    col.dictionary = h.dictionary
                if getattr(h, "tz", None) is not None:
                    col.tz = h.tz
                cols[n] = col
            self.device_table = DeviceTable(cols)
        return self.device_table


_DSX_SQLT = {rt.I64: "BIGINT", rt.F64: "DOUBLE", rt.I32: "INTEGER",
             rt.F32: "FLOAT", rt.I8: "TINYINT", rt.BOOL8: "BOOLEAN"}


class DeviceRegisteredTable:
    """A table registered directly from device-resident columns (e.g. a
    shuffle-received intermediate in the distributed pipeline)."""

    def __init__(self, table: DeviceTable, sql_types: dict | None = None):
        self.device_table = table
        self._sql_types = sql_types or {}

    def fields(self):
        return [(n, self._sql_types.get(n, _DSX_SQLT[c.dtype]))
                for n, c in self.device_table.columns.items()]

    def upload(self, runtime) -> DeviceTable:
        return self.device_table


class ResultFrame:
    """Shaped like the reference's lazy return of Context.sql (a dataframe
    you .compute()); here execution already happened on the GPU and compute()
    is the device→host materialization."""

    def __init__(self, dc, rel, context):
        self._dc = dc
        self._rel = rel
        self._context = context

    def compute(self):
        from dask_sql_amd.materialize import to_pandas
        if isinstance(self._dc, HostDataContainer):
            return self._dc.pdf
        return to_pandas(self._dc, self._context, self._rel.getRowType())

    # parity alias
    def to_pandas(self):
        return self.compute()

    @property
    def dc(self):
        return self._dc


class Context:
    """reference dask_sql/context.py:90+ — same API surface for the hot path.

    c = Context(); c.create_table("t", df); c.sql("SELECT ...").compute()
    """

    DEFAULT_SCHEMA_NAME = "root"

    def __init__(self, device_id: int = 0):
        self.schema_name = self.DEFAULT_SCHEMA_NAME
        self.catalog = Catalog()
        self.tables: dict[str, RegisteredTable] = {}
        self._runtime = None
        self._device_id = device_id
        self._plan_cache: dict = {}
        self._schema_version = 0

    # -- reference context.py:168 create_table ----------------------------
    def create_table(self, table_name: str, input_table, persist: bool = False,
                     date_columns=(), dictionaries=None, gpu: bool = True,
                     **kwargs):
        if isinstance(input_table, dict):
            input_table = pd.DataFrame(input_table)
        if isinstance(input_table, ResultFrame):
            # register a query result device-resident (CTAS; reference
            # context.py create_table accepts dask frames — ours are device
            # tables, no host round-trip)
            dc = input_table.dc
            if isinstance(dc, HostDataContainer):
                input_table = dc.pdf
            else:
                cc = dc.column_container
                cols = {}
                sqlts = {}
                fields = input_table._rel.getRowType().getFieldList()
                for i, f in enumerate(cc.columns):
                    col = dc.table.col(cc.get_backend_by_frontend_name(f))
                    cols[f] = col
                    if i < len(fields):
                        sqlts[f] = fields[i].getType().getSqlType()
                self.create_table_from_device(
                    table_name, DeviceTable(cols,
                                            num_rows=dc.table.num_rows),
                    sql_types=sqlts)
                return
        if isinstance(input_table, str):
            # file-path inputs (reference input_utils/location.py:22-60:
            # format inferred from the extension; parquet via pyarrow,
            # csv via pandas — the same engines the reference delegates to)
            fmt = kwargs.get("format")
            low = input_table.lower()
            if fmt == "parquet" or low.endswith(".parquet") \
                    or low.endswith(".parq"):
                # direct arrow → HBM, no pandas round-trip
                import pyarrow.parquet as pq
                host_cols = _from_arrow(pq.read_table(input_table),
                                        date_columns, dictionaries)
                t = RegisteredTable(host_cols)
                self.tables[table_name.lower()] = t
                self.catalog.add(table_name, t.fields())
                self._schema_version += 1
                if persist:
                    t.upload(self._get_runtime())
                return
            elif fmt == "csv" or low.endswith(".csv"):
                input_table = pd.read_csv(input_table)
            elif fmt == "json" or low.endswith(".json"):
                input_table = pd.read_json(input_table)
            else:
                raise NotImplementedError(
                    f"cannot infer input format of {input_table!r} "
                    "(pass format='parquet'|'csv'|'json')")
        if not isinstance(input_table, pd.DataFrame):
            raise NotImplementedError(
                "only pandas/dict/path inputs (other input plugins are out "
                "of scope, SURVEY §2)")
        host_cols = _from_pandas(input_table, date_columns, dictionaries)
        t = RegisteredTable(host_cols)
        self.tables[table_name.lower()] = t
        self.catalog.add(table_name, t.fields())
        self._schema_version += 1
        if persist:
            t.upload(self._get_runtime())

    def create_table_from_device(self, table_name: str, table: DeviceTable,
                                 sql_types: dict | None = None):
        """Register device-resident columns as a table (distributed
        intermediates; no host round-trip). Re-registering with an identical
        schema keeps cached plans valid (plans are data-independent), so the
        per-step re-registration in the distributed pipeline stays cheap."""
        t = DeviceRegisteredTable(table, sql_types)
        key = table_name.lower()
        old = self.tables.get(key)
        self.tables[key] = t
        if old is None or old.fields() != t.fields():
            self.catalog.add(table_name, t.fields())
            self._schema_version += 1

    # -- reference context.py:324/:415 UDF registration --------------------
    @staticmethod
    def _np_to_sql(t) -> str:
        m = {"int64": "BIGINT", "int32": "INTEGER", "int16": "INTEGER",
             "int8": "TINYINT", "uint64": "BIGINT", "uint32": "BIGINT",
             "uint16": "INTEGER", "uint8": "INTEGER",
             "float64": "DOUBLE", "float32": "FLOAT", "bool": "BOOLEAN",
             "object": "VARCHAR", "str": "VARCHAR"}
        name = np.dtype(t).name if t is not str else "str"
        if name not in m:
            raise NotImplementedError(f"UDF return type {t!r} unsupported")
        return m[name]

    def register_function(self, f, name: str, parameters, return_type,
                          replace: bool = False, schema_name: str = None,
                          row_udf: bool = False):
        """reference context.py:324 — register a scalar function usable in
        SQL. Executed exactly as on the reference: the Python callable runs
        on host column data (a UDF IS Python — the operand columns round-
        trip device→host→device on the explicit slow path)."""
        self._register_callable(f, name, False, parameters, return_type,
                                replace, row_udf)

    def register_aggregation(self, f, name: str, parameters, return_type,
                             replace: bool = False, schema_name: str = None):
        """reference context.py:415 — register a custom aggregation (a
        dask.dataframe.Aggregation-like object with .chunk/.agg[/.finalize],
        or a plain callable Series→scalar)."""
        self._register_callable(f, name, True, parameters, return_type,
                                replace, False)

    def _register_callable(self, f, name, aggregation, parameters,
                           return_type, replace, row_udf):
        key = name.lower()
        cat = self.catalog
        existing = cat.functions.get(key) or cat.aggregations.get(key)
        if existing is not None and not replace and existing[0] is not f:
            # reference _register_callable: one namespace for both kinds;
            # re-registering the SAME callable (type overloads,
            # test_function.py:180-188) is fine, a different one needs
            # replace=True
            raise ValueError(
                f"A function with the name {name} is already present; "
                "use replace=True to overwrite it")
        cat.functions.pop(key, None)
        cat.aggregations.pop(key, None)
        ret = self._np_to_sql(return_type)
        if aggregation:
            cat.aggregations[key] = (f, ret)
        else:
            cat.functions[key] = (f, ret, bool(row_udf),
                                  list(parameters or []))
        self._schema_version += 1

    def alter_table(self, old_table_name, new_table_name,
                    schema_name=None):
        """reference context.py:599 — rename a registered table."""
        key = old_table_name.lower()
        t = self.tables.pop(key)  # KeyError like the reference
        self.tables[new_table_name.lower()] = t
        self.catalog.drop(key)
        self.catalog.add(new_table_name, [(n, ty) for n, ty in t.fields()])
        self._schema_version += 1

    def drop_table(self, table_name: str):
        self.tables.pop(table_name.lower(), None)
        self.catalog.drop(table_name)
        self._schema_version += 1

    # -- reference context.py:482 sql --------------------------------------
    def sql(self, sql, return_futures: bool = True,
            config_options=None, dataframes=None, gpu: bool = True):
        # `gpu` is accepted for signature parity (context.py:482-489);
        # every table is device-resident here
        if dataframes:
            # reference context.py sql(dataframes=...): register inline
            for name, frame in dataframes.items():
                self.create_table(name, frame)
        if not isinstance(sql, str):
            # a LogicalPlan built earlier (reference accepts those too)
            dc = RelConverter.convert(sql, context=self)
            res = ResultFrame(dc, sql, self)
            return res.compute() if not return_futures else res
        em = re.match(r"\s*EXPLAIN\s+(.*)$", sql,
                      re.IGNORECASE | re.DOTALL)
        if em:
            # EXPLAIN <select> returns the plan STRING
            # (reference test_explain.py:13-23)
            return self.explain(em.group(1))
        if config_options:
            # per-query overrides, scoped like the reference's
            # dask.config.set(config_options) (context.py:519); unknown or
            # unsupported keys raise (round-1 silently ignored them)
            from dask_sql_amd import config
            with config.set(config_options):
                return self.sql(sql, return_futures=return_futures)
        # SHOW SCHEMAS / TABLES / COLUMNS (reference rel/custom/*.py,
        # expected frames pinned by tests/integration/test_show.py)
        sm = re.match(r'\s*SHOW\s+(SCHEMAS|TABLES|COLUMNS)'
                      r'(?:\s+FROM\s+([\w".]+))?'
                      r"(?:\s+LIKE\s+'([^']*)')?\s*;?\s*$", sql,
                      re.IGNORECASE)
        if sm:
            what = sm.group(1).upper()
            arg = (sm.group(2) or "").replace('"', "")
            like = sm.group(3)
            if what == "SCHEMAS":
                pdf = pd.DataFrame({"Schema": [self.schema_name,
                                               "information_schema"]})
            elif what == "TABLES":
                pdf = pd.DataFrame({"Table": sorted(self.tables)})
            else:
                tname = arg.split(".")[-1]
                t = self.tables[tname.lower()]  # KeyError like the reference
                sqlt_low = {n: ty.lower() for n, ty in t.fields()}
                pdf = pd.DataFrame({
                    "Column": list(sqlt_low),
                    "Type": list(sqlt_low.values()),
                    "Extra": [""] * len(sqlt_low),
                    "Comment": [""] * len(sqlt_low),
                })
            if like is not None:
                # SHOW ... LIKE '<pattern>' filters the first column by
                # the LIKE pattern (reference show_schemas.py)
                from dask_sql_amd.physical.rex import _like_regex
                rx = _like_regex(like)
                first = pdf.columns[0]
                pdf = pdf[[bool(rx.fullmatch(str(v)))
                           for v in pdf[first]]].reset_index(drop=True)
            from dask_sql_amd.datacontainer import HostDataContainer
            return ResultFrame(HostDataContainer(pdf), None, self)
        # ALTER TABLE [IF EXISTS] x RENAME TO y (reference DDL alter.py)
        alm = re.match(r"\s*ALTER\s+TABLE\s+(IF\s+EXISTS\s+)?(\w+)\s+"
                       r"RENAME\s+TO\s+(\w+)\s*;?\s*$", sql,
                       re.IGNORECASE)
        if alm:
            from dask_sql_amd.datacontainer import HostDataContainer
            try:
                self.alter_table(alm.group(2), alm.group(3))
            except KeyError:
                if not alm.group(1):
                    raise
            return ResultFrame(HostDataContainer(pd.DataFrame()), None,
                               self)
        # DROP TABLE (reference rel/custom/drop.py)
        dvm = re.match(r"\s*DROP\s+VIEW\s+(?:IF\s+EXISTS\s+)?(\w+)"
                       r"\s*;?\s*$", sql, re.IGNORECASE)
        if dvm:
            from dask_sql_amd.datacontainer import HostDataContainer
            self.catalog.views.pop(dvm.group(1).lower(), None)
            self._schema_version += 1
            return ResultFrame(HostDataContainer(pd.DataFrame()), None,
                               self)
        dm = re.match(r"\s*DROP\s+TABLE\s+(?:IF\s+EXISTS\s+)?(\w+)\s*;?\s*$",
                      sql, re.IGNORECASE)
        if dm:
            self.drop_table(dm.group(1))
            from dask_sql_amd.datacontainer import HostDataContainer
            return ResultFrame(HostDataContainer(pd.DataFrame()), None, self)
        # ANALYZE TABLE ... COMPUTE STATISTICS (reference rel/custom/
        # analyze_table.py; frame shape pinned by test_analyze.py:8-33:
        # describe() rows + data_type + col_name)
        am = re.match(r"\s*ANALYZE\s+TABLE\s+(\w+)\s+COMPUTE\s+STATISTICS"
                      r"\s+FOR\s+(ALL\s+COLUMNS|COLUMNS\s+(.+?))\s*;?\s*$",
                      sql, re.IGNORECASE)
        if am:
            pdf = self.sql(f"SELECT * FROM {am.group(1)}").compute()
            if am.group(3):
                cols = [c.strip() for c in am.group(3).split(",")]
                pdf = pdf[cols]
            stats = pdf.describe()
            extra = pd.DataFrame(
                {c: [str(t).lower(), c] for c, t in zip(
                    pdf.columns,
                    [dict(self.tables[am.group(1).lower()].fields()).get(
                        c, "double") for c in pdf.columns])},
                index=["data_type", "col_name"])
            out = pd.concat([stats, extra])
            from dask_sql_amd.datacontainer import HostDataContainer
            return ResultFrame(HostDataContainer(out), None, self)
        # CREATE [OR REPLACE] VIEW <name> AS <select> — a view re-plans
        # its SELECT at every use (reference rel/custom/
        # create_memory_table.py CreateView, persist=False)
        vm = re.match(r"\s*CREATE\s+(?:OR\s+REPLACE\s+)?VIEW\s+(\w+)"
                      r"\s+AS\s*\(?\s*(SELECT.*?)\)?\s*;?\s*$", sql,
                      re.IGNORECASE | re.DOTALL)
        if vm:
            from dask_sql_amd.datacontainer import HostDataContainer
            self._get_ral(vm.group(2))  # validate eagerly
            self.catalog.views[vm.group(1).lower()] = vm.group(2)
            self._schema_version += 1
            return ResultFrame(HostDataContainer(pd.DataFrame()), None,
                               self)
        # CREATE TABLE <name> WITH (location=..., format=..., ...) —
        # reference rel/custom/create_table.py (test_create.py:14-40)
        cwm = re.match(r"\s*CREATE\s+(?:OR\s+REPLACE\s+)?TABLE\s+(\w+)"
                       r"\s+WITH\s*\((.*)\)\s*;?\s*$", sql,
                       re.IGNORECASE | re.DOTALL)
        if cwm:
            from dask_sql_amd.datacontainer import HostDataContainer
            kv = {}
            for m_ in re.finditer(r"(\w+)\s*=\s*(?:'([^']*)'|(\w+))",
                                  cwm.group(2)):
                kv[m_.group(1).lower()] = m_.group(2) \
                    if m_.group(2) is not None else m_.group(3)
            loc = kv.get("location")
            if loc is None:
                raise ValueError(
                    "CREATE TABLE ... WITH needs a location")
            self.create_table(cwm.group(1), loc,
                              format=kv.get("format"),
                              persist=str(kv.get("persist", "false")
                                          ).lower() == "true")
            return ResultFrame(HostDataContainer(pd.DataFrame()), None,
                               self)
        # CREATE TABLE <name> AS <select> (reference DDL create_table.py)
        m = re.match(r"\s*CREATE\s+(?:OR\s+REPLACE\s+)?TABLE\s+(\w+)\s+AS\s*"
                     r"\(?\s*(SELECT.*?)\)?\s*;?\s*$", sql,
                     re.IGNORECASE | re.DOTALL)
        if m:
            res = self.sql(m.group(2))
            self.create_table(m.group(1), res)
            return res
        rel = self._get_ral(sql)
        logger.debug("plan:\n%s", rel.explain())
        dc = RelConverter.convert(rel, context=self)
        res = ResultFrame(dc, rel, self)
        return res.compute() if not return_futures else res

    def explain(self, sql: str) -> str:
        return self._get_ral(sql).explain()

    # -- internals ----------------------------------------------------------
    def _get_ral(self, sql: str):
        """reference context.py:819 _get_ral (planner entry). Plans are
        immutable → cached per (sql, schema version)."""
        from dask_sql_amd import config
        up = sql.upper()
        if "CURRENT_" in up or "LOCALTIME" in up or "NOW(" in up \
                or "RAND" in up:
            # now()-style and random expressions fold at build — never
            # serve them from the plan cache
            return Builder(self.catalog, self.schema_name).build(sql)
        key = (sql, self._schema_version, config.plan_fingerprint())
        plan = self._plan_cache.get(key)
        if plan is None:
            plan = Builder(self.catalog, self.schema_name).build(sql)
            if len(self._plan_cache) > 256:
                self._plan_cache.clear()
            self._plan_cache[key] = plan
        return plan

    def _get_runtime(self):
        if self._runtime is None:
            self._runtime = rt.Runtime(self._device_id)
        return self._runtime

    def _device_table(self, table_name: str) -> DeviceTable:
        t = self.tables[table_name.lower()]
        return t.upload(self._get_runtime())
