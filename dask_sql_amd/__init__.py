"""dask_sql_amd — MI355X-native physical execution layer for dask-sql's hot
path (filter / hash-join / hash-groupby-aggregate), built from scratch.

Keeps the reference's user API (Context.sql / create_table — reference
dask_sql/context.py) and plugin boundary (RelConverter.add_plugin_class —
reference physical/rel/convert.py:32-36) while the execution layer is
hand-written HIP for gfx950 behind a C ABI (include/dsxhip.h).

There is NO CPU fallback: on a machine without a GPU or without the built
extension, queries raise DsxUnavailable."""

from dask_sql_amd.context import Context  # noqa: F401
from dask_sql_amd.physical.convert import RelConverter  # noqa: F401
from dask_sql_amd.runtime import DsxError, DsxUnavailable  # noqa: F401

__version__ = "0.1.0"
