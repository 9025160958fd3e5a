"""LogicalPlan / Expression — the duck-typed plan API the physical plugins
consume.

Mirrors the *interface shape* of the reference's PyO3 wrappers:
  - LogicalPlan.get_current_node_type   (src/sql/logical.rs:300)
  - LogicalPlan.get_inputs              (src/sql/logical.rs:256)
  - LogicalPlan.getRowType              (src/sql/logical.rs:389-390)
  - node accessors .filter()/.join()/.aggregate()/.table_scan()/.projection()
    (src/sql/logical.rs:102-152)
  - Expression.getRexType/getOperatorName/getOperands/getIndex/column_name/
    getType (src/expression.rs:194-512), AggCall accessors
    (src/sql/logical/aggregate.rs:24-68).

The plan *semantics* are pinned end-to-end only: no reference
test pins plan structure, so these are a fresh design carrying just what the
converters need.
"""
from __future__ import annotations


class SqlType:
    def __init__(self, name: str):
        self._name = name  # "BIGINT", "DOUBLE", "INTEGER", "DATE", "BOOLEAN",
                           # "VARCHAR", "TINYINT", "FLOAT"

    def getSqlType(self):
        return self._name

    def __str__(self):
        return f"SqlTypeName.{self._name}"

    def __eq__(self, other):
        return str(self) == str(other)


class Field:
    def __init__(self, name: str, sql_type: SqlType, qualifier: str | None = None):
        self._name = name
        self._type = sql_type
        self.qualifier = qualifier  # table alias this field came from

    def getName(self):
        return self._name

    def getQualifiedName(self):
        return f"{self.qualifier}.{self._name}" if self.qualifier else self._name

    def getType(self):
        return self._type

    def __repr__(self):
        return f"Field({self.getQualifiedName()}: {self._type.getSqlType()})"


class RelDataType:
    def __init__(self, fields: list[Field]):
        self._fields = fields

    def getFieldList(self):
        return list(self._fields)

    def getFieldNames(self):
        return [f.getName() for f in self._fields]


# ---------------------------------------------------------------------------
# Expressions (RexType.Reference / Literal / Call — expression.rs:319)
# ---------------------------------------------------------------------------
class Expression:
    def getRexType(self) -> str:
        raise NotImplementedError

    def getOperands(self):
        return []


class InputRef(Expression):
    def __init__(self, index: int, sql_type: SqlType | None = None):
        self.index = index
        self._type = sql_type

    def getRexType(self):
        return "RexType.Reference"

    def getIndex(self):
        return self.index

    def getType(self):
        return self._type

    def column_name(self, rel) -> str:
        return rel.getRowType().getFieldNames()[self.index]

    def __repr__(self):
        return f"InputRef({self.index})"


class ScalarSub(Expression):
    """Uncorrelated scalar subquery — resolved to a Literal at convert time
    (execution is eager; DataFusion folds these on the reference side)."""

    def __init__(self, plan):
        self.plan = plan

    def getRexType(self):
        return "RexType.ScalarSubquery"

    def getType(self):
        return self.plan.getRowType().getFieldList()[0].getType()


class Literal(Expression):
    def __init__(self, value, sql_type: SqlType):
        self.value = value
        self._type = sql_type

    def getRexType(self):
        return "RexType.Literal"

    def getType(self):
        return self._type

    def getValue(self):
        return self.value

    def __repr__(self):
        return f"Literal({self.value!r})"


class Call(Expression):
    def __init__(self, op: str, operands: list[Expression],
                 sql_type: SqlType | None = None):
        self.op = op  # "=", "<", "AND", "+", "CAST", "IS NULL", ...
        self.operands = operands
        self._type = sql_type

    def getRexType(self):
        return "RexType.Call"

    def getOperatorName(self):
        return self.op

    def getOperands(self):
        return list(self.operands)

    def getType(self):
        return self._type

    def __repr__(self):
        return f"Call({self.op}, {self.operands})"


class AggCall(Expression):
    """One named aggregate call of an Aggregate node (aggregate.rs:24-68)."""

    def __init__(self, func_name: str, args: list[Expression], output_name: str,
                 filter_expr: Expression | None = None, distinct: bool = False):
        self.func_name = func_name
        self.args = args
        self.output_name = output_name
        self.filter_expr = filter_expr
        self.distinct = distinct

    def getRexType(self):
        return "RexType.Call"

    def getExprType(self):
        return "AggregateFunction"

    def getFilterExpr(self):
        return self.filter_expr

    def isDistinctAgg(self):
        return self.distinct

    def toString(self):  # output column name (aggregate.py:512)
        return self.output_name

    def __repr__(self):
        return f"AggCall({self.func_name}, {self.args}, as={self.output_name})"


# ---------------------------------------------------------------------------
# Plan nodes
# ---------------------------------------------------------------------------
class LogicalPlan:
    def __init__(self, node_type: str, inputs: list["LogicalPlan"],
                 row_type: RelDataType, payload):
        self._node_type = node_type
        self._inputs = inputs
        self._row_type = row_type
        self._payload = payload

    def get_current_node_type(self) -> str:
        return self._node_type

    def get_inputs(self):
        return list(self._inputs)

    def getRowType(self) -> RelDataType:
        return self._row_type

    def _get(self, expect):
        assert self._node_type == expect, (self._node_type, expect)
        return self._payload

    def table_scan(self):
        return self._get("TableScan")

    def filter(self):
        return self._get("Filter")

    def join(self):
        return self._get("Join")

    def aggregate(self):
        p = self._payload
        assert self._node_type in ("Aggregate", "Distinct")
        return p

    def projection(self):
        return self._get("Projection")

    def sort(self):
        return self._get("Sort")

    def window(self):
        return self._get("Window")

    def limit(self):
        return self._get("Limit")

    def explain(self, indent=0) -> str:
        s = "  " * indent + f"{self._node_type}: " \
            f"{[f.getName() for f in self._row_type.getFieldList()]}\n"
        for i in self._inputs:
            s += i.explain(indent + 1)
        return s


class TableScanNode:
    def __init__(self, schema_name: str, table_name: str,
                 projects: list[str] | None = None, filters=None):
        self.schema_name = schema_name
        self.table_name = table_name
        self._projects = projects
        self._filters = filters or []

    def getTableName(self):
        return self.table_name

    # reference table_scan.py:61-111 accessors (table_scan.rs:186-212)
    def containsProjections(self):
        return self._projects is not None

    def getTableScanProjects(self):
        return list(self._projects or [])

    def getFilters(self):
        return list(self._filters)

    def getDNFFilters(self):
        class _F:
            filtered_exprs = []
            io_unfilterable_exprs = []
        _f = _F()
        _f.io_unfilterable_exprs = list(self._filters)
        return _f


class FilterNode:
    def __init__(self, condition: Expression):
        self._condition = condition

    def getCondition(self):
        return self._condition


class JoinNode:
    def __init__(self, join_type: str, condition: Expression | None,
                 output_indices=None):
        self._join_type = join_type  # INNER/LEFT/RIGHT/FULL/LEFTSEMI/LEFTANTI
        self._condition = condition
        # positions into the combined (lhs ++ rhs) row this join outputs;
        # None = all (set by the pruning pass, planner/prune.py)
        self.output_indices = output_indices

    def getJoinType(self):
        return self._join_type

    def getCondition(self):
        return self._condition


class AggregateNode:
    def __init__(self, group_exprs: list[Expression], agg_calls: list[AggCall],
                 distinct_node: bool = False,
                 distinct_columns: list[str] | None = None):
        self._group_exprs = group_exprs
        self._agg_calls = agg_calls
        self._distinct_node = distinct_node
        self._distinct_columns = distinct_columns or []

    def getGroupSets(self):
        return list(self._group_exprs)

    def getNamedAggCalls(self):
        return list(self._agg_calls)

    def getArgs(self, expr: AggCall):
        return list(expr.args)

    def getAggregationFuncName(self, expr: AggCall):
        return expr.func_name

    def isDistinctNode(self):
        return self._distinct_node

    def getDistinctColumns(self):
        return list(self._distinct_columns)


class UnionNode:
    """Positional UNION ALL of the inputs (reference: the Union rel lowered
    to dd.concat of the branch frames)."""


class WindowSpec:
    """One window column (reference rel/logical/window.py:212-428 lowering).
    arg/partition/order are input field indices; out column appended after
    the input fields."""

    def __init__(self, func: str, arg_idx, part_idx: list,
                 order_idx: list, out_name: str, out_type,
                 offset: int = 1, default=None, frame=None):
        self.func = func                  # row_number|rank|...|lag|lead
        self.arg_idx = arg_idx            # int | None (ranking / COUNT(*))
        self.part_idx = list(part_idx)
        self.order_idx = list(order_idx)  # [(field index, desc bool)]
        self.out_name = out_name
        self.out_type = out_type
        self.offset = offset              # LAG/LEAD row offset
        self.default = default            # LAG/LEAD boundary default
        # explicit frame ("rows"|"range", lo, hi) with bounds
        # ("unbounded_preceding"|"current"|"preceding"|"following", k);
        # None = the reference's defaults (window.py:280-300)
        self.frame = frame


class WindowNode:
    def __init__(self, specs: list):
        self.specs = list(specs)

    def getWindowSpecs(self):
        return list(self.specs)


class ProjectionNode:
    def __init__(self, named_projects: list[tuple[Expression, str]]):
        self._named = named_projects

    def getNamedProjects(self):
        return [(e, n) for (e, n) in self._named]


class SortNode:
    def __init__(self, keys: list[tuple[int, bool, bool]]):
        # (input field index, ascending, nulls_first)
        self.keys = keys

    def getCollation(self):
        return list(self.keys)


class LimitNode:
    def __init__(self, fetch: int | None, offset: int = 0):
        self.fetch = fetch
        self.offset = offset
