import sys
sys.path.insert(0, "/root/repo")
from dask_sql_amd.context import Context
from datagen import gen_q3

cust, orders, li = gen_q3()
c = Context()
c.create_table("customer", cust, persist=True)
c.create_table("orders", orders, persist=True)
c.create_table("lineitem", li, persist=True)

stage = sys.argv[1]
Q = {
    "filter": "SELECT COUNT(*) AS c, SUM(l_extendedprice) AS s FROM lineitem "
              "WHERE l_shipdate > 9204",
    "join1": "SELECT COUNT(*) AS c FROM customer, orders "
             "WHERE c_mktsegment = 0 AND c_custkey = o_custkey "
             "AND o_orderdate < 9204",
    "join2": "SELECT COUNT(*) AS c FROM customer, orders, lineitem "
             "WHERE c_mktsegment = 0 AND c_custkey = o_custkey "
             "AND l_orderkey = o_orderkey AND o_orderdate < 9204 "
             "AND l_shipdate > 9204",
    "agg": "SELECT l_orderkey, SUM(l_extendedprice*(1-l_discount)) AS rev, "
           "o_orderdate, o_shippriority FROM customer, orders, lineitem "
           "WHERE c_mktsegment = 0 AND c_custkey = o_custkey "
           "AND l_orderkey = o_orderkey AND o_orderdate < 9204 "
           "AND l_shipdate > 9204 "
           "GROUP BY l_orderkey, o_orderdate, o_shippriority",
}[stage]
for i in range(int(sys.argv[2]) if len(sys.argv) > 2 else 3):
    out = c.sql(Q).compute()
    print(stage, i, "OK", len(out), out.iloc[0].tolist()[:2] if len(out) else [])
