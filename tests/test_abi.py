"""CPU tests: the C-ABI library loads and exports every symbol
include/dsxhip.h declares (no compute without a GPU)."""
import ctypes as ct
import re
from pathlib import Path

import pytest

REPO = Path(__file__).resolve().parent.parent
SO = REPO / "dask_sql_amd" / "libdsxhip.so"
HEADER = REPO / "include" / "dsxhip.h"


def _declared_symbols():
    text = HEADER.read_text()
    return sorted(set(re.findall(r"\b(dsx_\w+)\s*\(", text)))


def test_library_builds_and_loads():
    if not SO.exists():
        import subprocess
        subprocess.run(["make", "-C", str(REPO / "dask_sql_amd" / "csrc")],
                       check=True)
    lib = ct.CDLL(str(SO))
    assert lib is not None


def test_all_header_symbols_exported():
    lib = ct.CDLL(str(SO))
    missing = [s for s in _declared_symbols() if not hasattr(lib, s)]
    assert not missing, f"header declares but .so does not export: {missing}"
    assert len(_declared_symbols()) >= 20


def test_last_error_callable_without_gpu():
    lib = ct.CDLL(str(SO))
    lib.dsx_last_error.restype = ct.c_char_p
    assert isinstance(lib.dsx_last_error(), (bytes, type(None)))


def test_runtime_fails_loudly_without_gpu():
    """Product path must raise, never silently fall back to CPU."""
    import torch

    from dask_sql_amd.runtime import DsxUnavailable, Runtime
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    with pytest.raises(DsxUnavailable):
        Runtime(0)


def test_context_sql_fails_loudly_without_gpu(user_table_1):
    import torch

    from dask_sql_amd.context import Context
    from dask_sql_amd.runtime import DsxUnavailable
    if torch.cuda.is_available():
        pytest.skip("GPU present")
    c = Context()
    c.create_table("user_table_1", user_table_1)
    with pytest.raises(DsxUnavailable):
        c.sql("SELECT user_id FROM user_table_1 WHERE b > 1").compute()
