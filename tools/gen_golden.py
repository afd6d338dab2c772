#!/usr/bin/env python3
"""Generate golden parity fixtures from the reference's own regression data.

Runs in the AUTHORING container where /root/reference (OpenTenBase) exists;
the committed fixtures under tests/golden/ are what travels to the GPU box.

Sources (reference citations):
  - src/test/regress/data/onek.data  — the onek table (COPY format), column
    order per src/test/regress/sql/create_table.sql:18-35
  - src/test/regress/data/tenk.data  — the tenk1 table, same column order
  - src/test/regress/sql/select_having.sql — the 10-row test_having table
Expected outputs pinned by the fixtures' metadata:
  - expected/aggregates.out:6   avg(four)  over onek = 1.5
  - expected/aggregates.out:32  sum(four)  over onek = 1500
  - expected/aggregates.out:258 count(four) over onek = 1000
  - expected/join.out:2475ff    count(*) tenk1 a ⋈ tenk1 b
                                on a.hundred = b.thousand
                                and (b.fivethous % 10) < 10  = 100000
  - expected/select_having.out:16-22  GROUP BY b,c HAVING count(*)=1
                                → {(1,'XXXX'),(3,'bbbb')}
"""
import json
import os

import numpy as np

REF = "/root/reference/src/test/regress"
OUT = os.path.join(os.path.dirname(__file__), "..", "tests", "golden")

COLS = ["unique1", "unique2", "two", "four", "ten", "twenty", "hundred",
        "thousand", "twothousand", "fivethous", "tenthous", "odd", "even"]


def load_table(path):
    rows = []
    with open(path) as f:
        for line in f:
            line = line.rstrip("\n")
            if not line:
                continue
            parts = line.split("\t")
            rows.append([int(x) for x in parts[:13]])
    a = np.array(rows, dtype=np.int64)
    return {c: a[:, i] for i, c in enumerate(COLS)}


def main():
    os.makedirs(OUT, exist_ok=True)

    onek = load_table(os.path.join(REF, "data", "onek.data"))
    tenk = load_table(os.path.join(REF, "data", "tenk.data"))

    np.savez_compressed(
        os.path.join(OUT, "regress_tables.npz"),
        onek_four=onek["four"],
        tenk_hundred=tenk["hundred"],
        tenk_thousand=tenk["thousand"],
        tenk_fivethous=tenk["fivethous"],
    )

    # aggtest (4 rows): src/test/regress/data/agg.data — a int2, b float4
    agg_rows = []
    with open(os.path.join(REF, "data", "agg.data")) as f:
        for line in f:
            a, b = line.split("\t")
            agg_rows.append([int(a), float(b)])

    # test_having rows transcribed from sql/select_having.sql INSERTs
    test_having = [
        (0, 1, "XXXX", "A"), (1, 2, "AAAA", "b"), (2, 2, "AAAA", "c"),
        (3, 3, "BBBB", "D"), (4, 3, "BBBB", "e"), (5, 3, "bbbb", "F"),
        (6, 4, "cccc", "g"), (7, 4, "cccc", "h"), (8, 4, "CCCC", "I"),
        (9, 4, "CCCC", "j"),
    ]

    meta = {
        "aggtest": {
            "rows": agg_rows,
            # aggregates.out:12 avg(a) where a<100 = 32.666...; :38 sum(a)=198
            # :44 sum(b)=431.773 (float4 display); :24 avg(b)≈107.943
            "avg_a_lt100": 32.666666666666666,
            "sum_a": 198,
            "sum_b_3dp": 431.773,
            "avg_b_3dp": 107.943,
        },
        "onek": {
            "nrows": int(len(onek["four"])),
            "avg_four": 1.5,          # aggregates.out:6
            "sum_four": 1500,         # aggregates.out:32
            "count_four": 1000,       # aggregates.out:258
        },
        "tenk_selfjoin": {
            # join.out:2475ff: count(*) tenk1 a, tenk1 b where
            # a.hundred = b.thousand and (b.fivethous % 10) < 10
            "count": 100000,
        },
        "test_having": {
            "rows": test_having,
            # select_having.out:16-22: GROUP BY b,c HAVING count(*)=1
            "groups_count1": [[1, "XXXX"], [3, "bbbb"]],
        },
        "xc_fqs_join": {
            # xc_FQS_join.out — tab1_rep/tab2_rep = the 5x5 grid
            # (val, val2) in 1..5 x 1..5 (generate_series cross join, :51)
            # :89-105  multi-key self-join (val, val2) with 1<val<4
            #          -> exactly 10 matched pairs (grid keys are unique)
            # :257-263 avg(val) over the natural 3-way self-join with
            #          val>0 and val<3 -> 1.5
            "join_filtered_pairs": 10,
            "avg_val_filtered": 1.5,
        },
        "xl_join": {
            # xl_join.out:20-27 — chained inner join through the
            # "Distribute results by H" repartition exchange plan:
            # t1.val1 in {1,2} joined on t2.val2 in {30,40} -> 0 rows
            "t1": [[1, 10], [2, 20]],
            "t2": [[3, 30], [4, 40]],
            "t3": [[5, 50], [6, 60]],
            "chained_rows": 0,
        },
        "test_vec": {
            # opentenbase_c_aggregation.out:1-139 — the distributed
            # two-phase (Partial on DN -> Finalize on CN) aggregate plans
            # over test_vec = generate_series(-10,15), int2/int4/int8 and
            # float8 variants all agreeing: sum=65, avg=2.5, 26 rows.
            # Negative values included — exercises signed accumulation.
            "values": list(range(-10, 16)),
            "sum": 65,
            "avg": 2.5,
            "count": 26,
        },
    }
    with open(os.path.join(OUT, "expected.json"), "w") as f:
        json.dump(meta, f, indent=1)
    print("wrote", OUT)


if __name__ == "__main__":
    main()
