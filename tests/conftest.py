import json
import os
import sys
from pathlib import Path

# GPU tests run with device-side bounds validation active (runtime reads the
# env at context creation; production default is off)
os.environ.setdefault("DSX_DEBUG", "1")

import numpy as np
import pandas as pd
import pytest

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def pytest_configure(config):
    config.addinivalue_line("markers", "gpu: needs a real MI355X GPU (run via gpurun)")


GOLDEN = json.loads((REPO / "tests" / "golden" / "golden_cases.json").read_text())


def golden_fixture_df(name: str) -> pd.DataFrame:
    fx = GOLDEN["fixtures"]
    if name == "user_table_nan":
        return pd.DataFrame({"c": pd.array(fx["user_table_nan_c"], dtype="UInt8")})
    return pd.DataFrame(fx[name])


def golden_expected(case: str) -> pd.DataFrame:
    data = GOLDEN["cases"][case]["expected"]
    return pd.DataFrame({k: [np.nan if v is None else v for v in vals] for k, vals in data.items()})


@pytest.fixture
def user_table_1():
    return golden_fixture_df("user_table_1")


@pytest.fixture
def user_table_2():
    return golden_fixture_df("user_table_2")


@pytest.fixture
def df_simple():
    return golden_fixture_df("df_simple")


@pytest.fixture
def df700():
    # reference tests/integration/fixtures.py:50-57 (seed-42 700-row frame)
    np.random.seed(42)
    return pd.DataFrame(
        {"a": [1.0] * 100 + [2.0] * 200 + [3.0] * 400, "b": 10 * np.random.rand(700)}
    )


def assert_frame_close(got: pd.DataFrame, exp: pd.DataFrame, rel=1e-6, sort_by=None):
    """Value comparison in the spirit of reference tests/utils.py assert_eq:
    index ignored, float columns to tolerance, ints/keys exact."""
    got = got.reset_index(drop=True)
    exp = exp.reset_index(drop=True)
    assert list(got.columns) == list(exp.columns), (list(got.columns), list(exp.columns))
    if sort_by:
        got = got.sort_values(sort_by, kind="mergesort").reset_index(drop=True)
        exp = exp.sort_values(sort_by, kind="mergesort").reset_index(drop=True)
    assert len(got) == len(exp), (len(got), len(exp))
    for col in exp.columns:
        g = got[col].to_numpy()
        e = exp[col].to_numpy()
        if np.issubdtype(np.asarray(e).dtype, np.floating) or np.issubdtype(
            np.asarray(g).dtype, np.floating
        ):
            g = np.asarray(g, dtype=np.float64)
            e = np.asarray(e, dtype=np.float64)
            nan_g, nan_e = np.isnan(g), np.isnan(e)
            assert (nan_g == nan_e).all(), f"{col}: NULL mismatch"
            ok = np.isclose(g[~nan_g], e[~nan_e], rtol=rel, atol=1e-12)
            assert ok.all(), f"{col}: {g[~nan_g][~ok][:5]} vs {e[~nan_e][~ok][:5]}"
        else:
            assert (np.asarray(g) == np.asarray(e)).all(), f"{col} mismatch"
