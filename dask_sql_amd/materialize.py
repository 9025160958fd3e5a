"""Device result → pandas, with the reference's output dtype semantics
(mappings.py:67-110 _SQL_TO_PYTHON_FRAMES + fix_dtype_to_row_type,
rel/base.py:89-124): DATE columns come back as datetime64[ns]
(mappings.py:78-80), dict-encoded VARCHAR decodes to object strings, NULLs in
integer columns upcast to float64 NaN exactly as pandas does on the reference
path."""
from __future__ import annotations

import numpy as np
import pandas as pd

from dask_sql_amd import runtime as rt
from dask_sql_amd.datacontainer import DataContainer, HostDataContainer

_SQL_NP = {
    "BIGINT": np.int64, "INTEGER": np.int32, "SMALLINT": np.int16,
    "TINYINT": np.int8, "DOUBLE": np.float64, "FLOAT": np.float32,
    "BOOLEAN": np.bool_,
}


def to_pandas(dc, context=None, row_type=None) -> pd.DataFrame:
    if isinstance(dc, HostDataContainer):
        return dc.pdf
    assert isinstance(dc, DataContainer)
    cc = dc.column_container
    fields = row_type.getFieldList() if row_type is not None else None
    data = {}
    for i, frontend in enumerate(cc.columns):
        backend = cc.get_backend_by_frontend_name(frontend)
        col = dc.table.col(backend)
        arr, valid = col.to_numpy()
        sql_t = None
        if fields is not None and i < len(fields):
            sql_t = fields[i].getType().getSqlType()
        data[frontend] = _convert(arr, valid, col, sql_t)
    return pd.DataFrame(data)


def _convert(arr, valid, col, sql_t):
    d = getattr(col, "dictionary", None)
    if d is not None:
        out = np.array([None] * len(arr), dtype=object)
        ok = valid if valid is not None else np.ones(len(arr), bool)
        codes = arr.astype(np.int64)
        in_range = ok & (codes >= 0) & (codes < len(d))
        lut = np.array(d, dtype=object)
        out[in_range] = lut[codes[in_range]]
        return pd.Series(out)
    if sql_t == "DATE":
        # date32 day-ints → datetime64[ns] (mappings.py:78-80: DATE columns
        # are datetime64[ns] on the reference path)
        s = pd.Series(pd.to_datetime(arr.astype("int64"), unit="D",
                                     errors="coerce"))
        if valid is not None:
            s[~valid] = pd.NaT
        return s
    if sql_t == "TIMESTAMP":
        s = pd.Series(pd.to_datetime(arr.astype("int64"), unit="ns",
                                     errors="coerce"))
        if valid is not None:
            s[~valid] = pd.NaT
        tz = getattr(col, "tz", None)
        if tz is not None:
            s = s.dt.tz_localize("UTC").dt.tz_convert(tz)
        return s
    if valid is not None and not valid.all():
        # NULL-bearing numeric → float64 with NaN (pandas upcast semantics)
        out = arr.astype(np.float64)
        out[~valid] = np.nan
        return pd.Series(out)
    if sql_t in _SQL_NP:
        return pd.Series(arr.astype(_SQL_NP[sql_t]))
    if col.dtype == rt.BOOL8 and sql_t == "BOOLEAN":
        return pd.Series(arr.astype(bool))
    return pd.Series(arr)
