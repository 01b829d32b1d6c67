#!/usr/bin/env python3
"""
make_golden.py — builds the committed golden fixtures under tests/golden/
from the reference's own regression test data (runs only in the build
container where /root/reference is mounted; the fixtures travel with the
repo — /root/reference does NOT exist on the GPU box).

Inputs (reference, read-only):
  src/test/regress/data/lineitem.{1,2}.data      12k-row sample lineitem
  expected/multi_tpch_query6.out                 Q6 revenue 243277.7858
  expected/multi_tpch_query1.out                 Q1 result table

Outputs (committed):
  tests/golden/lineitem12k_lz4.cs    product-written stripes (lz4, default knobs)
  tests/golden/lineitem12k_none.cs   same rows, no compression
  tests/golden/lineitem12k_zstd.cs   same rows, zstd
  tests/golden/expected.json         known answers + schema/date mapping notes

Physical schema mirrors BASELINE configs: decimal(15,2) measures as
fixed-point int64 scale 2; dates as int64 days since 1970-01-01; char(1)
flags as i8 codes (A=0,N=1,R=2; O=0,F=1).
"""
import datetime
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import numpy as np
import citus_amd as ca

REF = "/root/reference/src/test/regress"
OUT = os.path.join(os.path.dirname(__file__), "..", "tests", "golden")

RF = {"A": 0, "N": 1, "R": 2}
LS = {"O": 0, "F": 1}
EPOCH = datetime.date(1970, 1, 1)


def cents(s):
    """exact decimal(15,2) -> int64 cents from text"""
    neg = s.startswith("-")
    if neg:
        s = s[1:]
    if "." in s:
        w, f = s.split(".")
        f = (f + "00")[:2]
    else:
        w, f = s, "00"
    v = int(w) * 100 + int(f)
    return -v if neg else v


def days(s):
    y, m, d = map(int, s.split("-"))
    return (datetime.date(y, m, d) - EPOCH).days


def load_lineitem():
    rows = []
    for part in (1, 2):
        with open(f"{REF}/data/lineitem.{part}.data") as f:
            for line in f:
                c = line.rstrip("\n").split("|")
                rows.append((
                    int(c[0]),            # l_orderkey
                    cents(c[4]),          # l_quantity
                    cents(c[5]),          # l_extendedprice
                    cents(c[6]),          # l_discount
                    cents(c[7]),          # l_tax
                    days(c[10]),          # l_shipdate
                    RF[c[8]],             # l_returnflag
                    LS[c[9]],             # l_linestatus
                ))
    return rows


def main():
    os.makedirs(OUT, exist_ok=True)
    rows = load_lineitem()
    n = len(rows)
    cols = list(zip(*rows))
    arrs = [np.array(cols[i], dtype=np.int64) for i in range(6)] + \
           [np.array(cols[i], dtype=np.int8) for i in (6, 7)]
    defs = [("l_orderkey", ca.I64, 0), ("l_quantity", ca.I64, 2),
            ("l_extendedprice", ca.I64, 2), ("l_discount", ca.I64, 2),
            ("l_tax", ca.I64, 2), ("l_shipdate", ca.I64, 0),
            ("l_returnflag", ca.I8, 0), ("l_linestatus", ca.I8, 0)]

    for name, comp in (("lz4", ca.COMP_LZ4), ("none", ca.COMP_NONE), ("zstd", ca.COMP_ZSTD)):
        path = os.path.join(OUT, f"lineitem12k_{name}.cs")
        ca.write_table(path, defs, list(arrs), compression=comp,
                       stripe_row_limit=10000, chunk_group_row_limit=1000)
        print(name, os.path.getsize(path), "bytes")

    # known answers from the reference's expected outputs
    # (multi_tpch_query6.out:14-17; multi_tpch_query1.out)
    expected = {
        "n_rows": n,
        "q6": {
            "revenue_scale4": 2432777858,       # 243277.7858
            "pred": "shipdate >= 1994-01-01 (8766) AND < 1995-01-01 (9131) "
                    "AND discount in [5,7] AND quantity < 2400",
            "shipdate_ge": days("1994-01-01"),
            "shipdate_lt": days("1995-01-01"),
        },
        "q1": {
            "shipdate_le": days("1998-09-02"),  # 1998-12-01 - 90 days
            # key (returnflag, linestatus) -> [sum_qty(s2), sum_base_price(s2),
            #   sum_disc_price(s4), sum_charge(s6), count]
            "groups": {
                "A,F": [7546500, 11361987363, 1078412870728, 112171153245923, 2944],
                "N,F": [202200, 310255145, 29525407118, 3072642770652, 76],
                "N,O": [14977800, 22470694816, 2136348576854, 222134071929801, 5883],
                "R,F": [7315600, 10893797973, 1035166236698, 107743533784328, 2901],
            },
        },
        "schema": [d[0] for d in defs],
        "flag_codes": {"returnflag": RF, "linestatus": LS},
    }
    with open(os.path.join(OUT, "expected.json"), "w") as f:
        json.dump(expected, f, indent=1)
    print("wrote expected.json; rows:", n)


if __name__ == "__main__":
    main()
