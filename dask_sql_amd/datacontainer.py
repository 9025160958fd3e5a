"""DataContainer / ColumnContainer — the plugin output contract.

Mirrors the reference's dask_sql/datacontainer.py:19-231 conceptually: a
ColumnContainer maps SQL (frontend) field order/names onto backend column
handles; DataContainer pairs it with the backing table. Here the backing
table is a DeviceTable of HBM-resident columns instead of a dask frame.
"""
from __future__ import annotations

from dask_sql_amd.runtime import DeviceColumn


class DeviceTable:
    """Ordered named device columns (all the same length). num_rows must be
    given explicitly when the table has no columns (e.g. a fully pruned join
    feeding COUNT(*))."""

    def __init__(self, columns: dict[str, DeviceColumn], num_rows=None):
        self.columns = dict(columns)
        lens = {c.len for c in self.columns.values()}
        assert len(lens) <= 1, f"ragged table: {lens}"
        if lens:
            self.num_rows = lens.pop()
            assert num_rows is None or num_rows == self.num_rows
        else:
            self.num_rows = num_rows if num_rows is not None else 0

    def col(self, name) -> DeviceColumn:
        return self.columns[name]

    def names(self):
        return list(self.columns.keys())


class ColumnContainer:
    """reference datacontainer.py:19-171 (frontend↔backend mapping)."""

    def __init__(self, frontend_columns, mapping=None):
        self.columns = list(frontend_columns)
        if mapping is None:
            self._frontend_backend_mapping = {c: c for c in self.columns}
        else:
            self._frontend_backend_mapping = dict(mapping)

    def get_backend_by_frontend_name(self, name):
        return self._frontend_backend_mapping[name]

    def get_backend_by_frontend_index(self, i):
        return self._frontend_backend_mapping[self.columns[i]]

    def make_unique(self, prefix="col"):
        # reference datacontainer.py make_unique: rename frontend to
        # f"{prefix}_{i}" keeping backend mapping
        new_cols = [f"{prefix}_{i}" for i in range(len(self.columns))]
        mapping = {
            new: self._frontend_backend_mapping[old]
            for new, old in zip(new_cols, self.columns)
        }
        return ColumnContainer(new_cols, mapping)

    def rename(self, mapping: dict):
        new_cols = [mapping.get(c, c) for c in self.columns]
        new_map = {}
        for old, new in zip(self.columns, new_cols):
            new_map[new] = self._frontend_backend_mapping[old]
        return ColumnContainer(new_cols, new_map)

    def limit_to(self, cols):
        return ColumnContainer(
            list(cols),
            {c: self._frontend_backend_mapping[c] for c in cols},
        )

    def add(self, frontend, backend=None):
        cc = ColumnContainer(self.columns, self._frontend_backend_mapping)
        if frontend not in cc.columns:
            cc.columns.append(frontend)
        cc._frontend_backend_mapping[frontend] = backend or frontend
        return cc


class DataContainer:
    """reference datacontainer.py:190-231."""

    def __init__(self, table: DeviceTable, column_container: ColumnContainer):
        self.table = table
        self.column_container = column_container

    @property
    def df(self):
        return self.table

    def backend_cols(self):
        """Device columns in FRONTEND order (what InputRef indices mean)."""
        cc = self.column_container
        return [self.table.col(cc.get_backend_by_frontend_name(c))
                for c in cc.columns]

    def assign(self) -> DeviceTable:
        """Materialize frontend view: rename backend → frontend names in
        frontend order (reference datacontainer.py:217-231)."""
        cc = self.column_container
        return DeviceTable({
            c: self.table.col(cc.get_backend_by_frontend_name(c))
            for c in cc.columns
        })


class HostDataContainer:
    """Post-Sort host-side result (≤G rows; SURVEY §8f1 — top-k on host)."""

    def __init__(self, pdf):
        self.pdf = pdf
