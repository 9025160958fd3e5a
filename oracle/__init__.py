"""CPU oracle for the MI355X hot path — TEST INFRASTRUCTURE ONLY.

This package is a CPU restatement (pandas 2.3.3 — the same compute substrate
the reference delegates to) of the reference dask-sql physical layer's
semantics for filter / hash-join / hash-groupby-aggregate:

  - filter:    /root/reference/dask_sql/physical/rel/logical/filter.py:20-45
  - join:      /root/reference/dask_sql/physical/rel/logical/join.py:50-322
  - aggregate: /root/reference/dask_sql/physical/rel/logical/aggregate.py:117-589

Parity pinning: the oracle is checked against golden vectors restated from the
reference's own integration tests (tests/integration/test_groupby.py,
test_join.py, test_filter.py — literal expected frames committed under
/root/repo/tests/golden/). See tests/test_oracle.py.

ONLY tests/, __graft_entry__.smoke() and bench.py's cpu_baseline leg may
import this package, and only as the checker / reported CPU baseline — never
as the shipped compute path. The product path (dask_sql_amd) fails loudly when
the HIP extension is missing; it never falls back to this code.
"""

from oracle.frame import (  # noqa: F401
    oracle_filter,
    oracle_join,
    oracle_groupby,
)
