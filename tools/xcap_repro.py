"""Repro for the xcap hash mismatch: 3 ANDed conditions over distinct
columns + hash agg, engine vs oracle, printing differing groups."""
import ctypes as C
import importlib.util
import os
import random
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)
import tikv_amd
from tikv_amd import _ffi as F
sys.path.insert(0, os.path.join(ROOT, "tests"))
from test_topn_stream import (_four_col_region, _orc, cell_int, row_key,
                              split_rows)


def main():
    n = int(sys.argv[1]) if len(sys.argv) > 1 else 2000
    k, ko, v, vo, nn, keep = _four_col_region(n)
    cols = [tikv_amd.Col(i) for i in range(1, 5)]
    sels = [tikv_amd.cmp_col_const(1, F.SIG_GT_INT, -800),
            tikv_amd.cmp_col_const(2, F.SIG_LT_INT, 800),
            tikv_amd.cmp_col_const(3, F.SIG_NE_INT, 7)]
    req = (tikv_amd.DagSelect(cols).where(*sels)
           .hash_agg([tikv_amd.count_star(), tikv_amd.max_col(1)],
                     tikv_amd.Expr().col(0)).build())
    orc = _orc()
    od, orows = orc.dag_run(req, k, ko, v, vo, nn)
    eng = tikv_amd.Engine(0)
    rgn = eng.region_raw(k, ko, v, vo, nn)
    gd, gr, _ = eng.dag_run(req, [rgn])
    print("groups oracle", orows, "gpu", gr)
    o = sorted(split_rows(od, 3))
    g = sorted(split_rows(gd, 3))
    diff = 0
    for a, b in zip(o, g):
        if a != b:
            print("O", a.hex())
            print("G", b.hex())
            diff += 1
            if diff > 5:
                break
    only_o = set(o) - set(g)
    only_g = set(g) - set(o)
    print("only-oracle", len(only_o), "only-gpu", len(only_g))
    for x in list(only_o)[:3]:
        print("Oonly", x.hex())
    for x in list(only_g)[:3]:
        print("Gonly", x.hex())
    rgn.close(); eng.close()


if __name__ == "__main__":
    main()
