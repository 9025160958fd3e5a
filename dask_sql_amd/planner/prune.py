"""Column pruning (projection pushdown) over the built plan.

The functional analog of DataFusion's PushDownProjection on the reference
path (src/sql/optimizer.rs stock rules): scans read only referenced columns,
join outputs carry only columns some ancestor consumes — which on this
backend directly removes per-column gather kernels.
"""
from __future__ import annotations

from dask_sql_amd.planner.plan import (AggCall, AggregateNode, Call,
                                       Expression, FilterNode, InputRef,
                                       JoinNode, LogicalPlan, ProjectionNode,
                                       RelDataType, SortNode, TableScanNode)


def _expr_refs(e: Expression, out: set):
    if isinstance(e, InputRef):
        out.add(e.getIndex())
    elif isinstance(e, Call):
        for o in e.getOperands():
            _expr_refs(o, out)
    elif isinstance(e, AggCall):
        for a in e.args:
            _expr_refs(a, out)
        if e.filter_expr is not None:
            _expr_refs(e.filter_expr, out)


def _remap(e: Expression, m: dict):
    if isinstance(e, InputRef):
        return InputRef(m[e.getIndex()], e.getType())
    if isinstance(e, Call):
        return Call(e.op, [_remap(o, m) for o in e.operands], e._type)
    if isinstance(e, AggCall):
        return AggCall(e.func_name, [_remap(a, m) for a in e.args],
                       e.output_name,
                       _remap(e.filter_expr, m) if e.filter_expr is not None
                       else None, e.distinct)
    return e  # Literal


def prune_plan(plan: LogicalPlan) -> LogicalPlan:
    new_plan, _ = _prune(plan, None)
    return new_plan


def _prune(plan: LogicalPlan, needed):
    """needed: sorted list of output indices the parent consumes, or None
    for all. Returns (new_plan, mapping old_out_idx → new_out_idx)."""
    t = plan.get_current_node_type()
    fields = plan.getRowType().getFieldList()
    all_idx = list(range(len(fields)))
    if needed is None:
        needed = all_idx
    needed = sorted(set(needed))

    if t == "TableScan":
        node = plan.table_scan()
        if needed == all_idx:
            return plan, {i: i for i in all_idx}
        new_fields = [fields[i] for i in needed]
        new_node = TableScanNode(node.schema_name, node.table_name,
                                 projects=[fields[i].getName()
                                           for i in needed])
        m = {old: new for new, old in enumerate(needed)}
        return LogicalPlan("TableScan", [], RelDataType(new_fields),
                           new_node), m

    if t == "Filter":
        cond = plan.filter().getCondition()
        refs = set(needed)
        _expr_refs(cond, refs)
        child, cm = _prune(plan.get_inputs()[0], sorted(refs))
        new_cond = _remap(cond, cm)
        # filter output = its (pruned) input row
        new = LogicalPlan("Filter", [child], child.getRowType(),
                          FilterNode(new_cond))
        return new, dict(cm)

    if t == "Projection":
        named = plan.projection().getNamedProjects()
        kept = [(i, named[i]) for i in needed]
        child_refs = set()
        for _, (e, _n) in kept:
            _expr_refs(e, child_refs)
        child, cm = _prune(plan.get_inputs()[0], sorted(child_refs))
        new_named = [(_remap(e, cm), n) for _, (e, n) in kept]
        new_fields = [fields[i] for i in needed]
        m = {old: new for new, (old, _) in enumerate(kept)}
        return LogicalPlan("Projection", [child], RelDataType(new_fields),
                           ProjectionNode(new_named)), m

    if t == "Join":
        node = plan.join()
        lhs, rhs = plan.get_inputs()
        n_l = len(lhs.getRowType().getFieldList())
        refs = set(needed)
        if node.getCondition() is not None:
            _expr_refs(node.getCondition(), refs)
        l_refs = sorted(r for r in refs if r < n_l)
        r_refs = sorted(r - n_l for r in refs if r >= n_l)
        new_l, lm = _prune(lhs, l_refs)
        new_r, rm = _prune(rhs, r_refs)
        n_l_new = len(new_l.getRowType().getFieldList())
        m = {}
        for old in l_refs:
            m[old] = lm[old]
        for old0 in r_refs:
            m[old0 + n_l] = rm[old0] + n_l_new
        new_cond = _remap(node.getCondition(), m) \
            if node.getCondition() is not None else None
        out_is_lhs_only = str(node.getJoinType()) in ("LEFTSEMI", "LEFTANTI")
        out_old = [i for i in needed if (i < n_l or not out_is_lhs_only)]
        new_fields = [fields[i] for i in out_old]
        output_indices = [m[i] for i in out_old]  # child-combined positions
        new = LogicalPlan("Join", [new_l, new_r], RelDataType(new_fields),
                          JoinNode(node.getJoinType(), new_cond,
                                   output_indices=output_indices))
        out_m = {old: pos for pos, old in enumerate(out_old)}
        return new, out_m

    if t in ("Aggregate", "Distinct"):
        agg = plan.aggregate()
        refs = set()
        for g in agg.getGroupSets():
            _expr_refs(g, refs)
        for call in agg.getNamedAggCalls():
            _expr_refs(call, refs)
        child, cm = _prune(plan.get_inputs()[0], sorted(refs))
        new_groups = [_remap(g, cm) for g in agg.getGroupSets()]
        new_calls = [_remap(call, cm) for call in agg.getNamedAggCalls()]
        node = AggregateNode(new_groups, new_calls,
                             distinct_node=agg.isDistinctNode(),
                             distinct_columns=agg.getDistinctColumns())
        new = LogicalPlan(t, [child], plan.getRowType(), node)
        return new, {i: i for i in all_idx}

    if t == "Sort":
        keys = plan.sort().getCollation()
        refs = set(needed) | {k[0] for k in keys}
        child, cm = _prune(plan.get_inputs()[0], sorted(refs))
        new_keys = [(cm[i], asc, nf) for (i, asc, nf) in keys]
        new = LogicalPlan("Sort", [child], child.getRowType(),
                          SortNode(new_keys))
        return new, dict(cm)

    if t == "Limit":
        child, cm = _prune(plan.get_inputs()[0], needed)
        new = LogicalPlan("Limit", [child], child.getRowType(),
                          plan.limit())
        return new, dict(cm)

    # unknown node: keep as-is, require all
    return plan, {i: i for i in all_idx}
