"""CPU tests for the config namespace (reference dask_sql/sql.yaml keys,
config.py:1-12; per-query overrides context.py:519). VERDICT r1 weak#7:
config_options must be honored or raise."""
import pandas as pd
import pytest

from dask_sql_amd import config
from dask_sql_amd.context import Context


def test_defaults_match_reference_yaml():
    assert config.get("sql.aggregate.split_out") == 1
    assert config.get("sql.join.broadcast") is None
    assert config.get("sql.predicate_pushdown") is True
    assert config.get("sql.sort.topk-nelem-limit") == 1_000_000
    assert config.get("sql.identifier.case_sensitive") is True


def test_unknown_key_raises():
    with pytest.raises(KeyError):
        config.set({"sql.nonsense.key": 1})
    with pytest.raises(KeyError):
        Context().sql("SELECT 1", config_options={"sql.bogus": True})


def test_pinned_key_rejects_unsupported_value():
    with pytest.raises(NotImplementedError):
        config.set({"sql.identifier.case_sensitive": False})
    with pytest.raises(NotImplementedError):
        config.set({"sql.mappings.decimal_support": "cudf"})


def test_set_scoping_and_nested_form():
    assert config.get("sql.join.broadcast") is None
    with config.set({"sql": {"join": {"broadcast": True}}}):
        assert config.get("sql.join.broadcast") is True
        with config.set({"sql.join.broadcast": 0.5}):
            assert config.get("sql.join.broadcast") == 0.5
        assert config.get("sql.join.broadcast") is True
    assert config.get("sql.join.broadcast") is None


def _filter_depths(rel):
    """Walk the plan; return node types below the first Join."""
    node = rel
    above = []
    while node.get_current_node_type() != "Join":
        above.append(node.get_current_node_type())
        node = node.get_inputs()[0]
    return node, above


def test_predicate_pushdown_off_is_honored():
    c = Context()
    c.create_table("pa", pd.DataFrame({"k": [1, 2], "x": [1, 2]}))
    c.create_table("pb", pd.DataFrame({"k": [1, 2], "y": [3, 4]}))
    q = "SELECT pa.k FROM pa JOIN pb ON pa.k = pb.k WHERE pb.y = 3"
    # default: pushed below the scan
    join, above = _filter_depths(c._get_ral(q))
    assert "Filter" not in above
    assert join.get_inputs()[1].get_current_node_type() == "Filter"
    # off: one post-join Filter, bare scans (plan cache must not leak the
    # default-config plan — the fingerprint is part of the cache key)
    with config.set({"sql.predicate_pushdown": False}):
        join, above = _filter_depths(c._get_ral(q))
        assert "Filter" in above
        assert join.get_inputs()[1].get_current_node_type() == "TableScan"


def test_config_options_accepted_key_plans():
    c = Context()
    c.create_table("t", pd.DataFrame({"k": [1, 1, 2], "x": [1.0, 2.0, 3.0]}))
    # accepted reference key with no observable analog: planning must not
    # raise (execution needs a GPU — covered by -m gpu)
    with config.set({"sql.aggregate.split_out": 4}):
        rel = c._get_ral("SELECT k, SUM(x) AS s FROM t GROUP BY k")
    assert rel is not None
