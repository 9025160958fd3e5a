"""CPU unit tests for the column-pruning pass (planner/prune.py)."""
import numpy as np
import pandas as pd

from dask_sql_amd.context import Context


def _ctx():
    c = Context()
    c.create_table("a", pd.DataFrame({
        "k": np.arange(3, dtype=np.int64), "x": [1.0, 2.0, 3.0],
        "unused1": [0, 0, 0], "unused2": [1, 1, 1]}))
    c.create_table("b", pd.DataFrame({
        "k": np.arange(3, dtype=np.int64), "y": [9.0, 8.0, 7.0],
        "junk": [5, 5, 5]}))
    return c


def test_scan_pruned_to_used_columns():
    c = _ctx()
    rel = c._get_ral("SELECT x FROM a WHERE k = 1")
    node = rel
    while node.get_current_node_type() != "TableScan":
        node = node.get_inputs()[0]
    assert sorted(node.getRowType().getFieldNames()) == ["k", "x"]
    assert node.table_scan().containsProjections()


def test_join_output_pruned():
    c = _ctx()
    rel = c._get_ral(
        "SELECT a.x, b.y FROM a JOIN b ON a.k = b.k")
    node = rel
    while node.get_current_node_type() != "Join":
        node = node.get_inputs()[0]
    # join output carries only consumed columns (equi keys excluded)
    assert sorted(node.getRowType().getFieldNames()) == ["x", "y"]
    oi = node.join().output_indices
    assert oi is not None and len(oi) == 2
    # children pruned to keys + outputs
    lhs, rhs = node.get_inputs()
    assert sorted(lhs.getRowType().getFieldNames()) == ["k", "x"]
    assert sorted(rhs.getRowType().getFieldNames()) == ["k", "y"]


def test_aggregate_input_pruned():
    c = _ctx()
    rel = c._get_ral("SELECT k, SUM(x) AS s FROM a GROUP BY k")
    node = rel
    while node.get_current_node_type() != "TableScan":
        node = node.get_inputs()[0]
    assert sorted(node.getRowType().getFieldNames()) == ["k", "x"]
