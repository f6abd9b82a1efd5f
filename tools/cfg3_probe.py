"""Attribute cfg3's per-row instruction cost: run hash-agg variants over one
resident 50M-row cfg3 region and print per-step kernel ms for each shape.
Dev tool (GPU box), not part of the product or bench contract."""
import os
import sys
import time

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
sys.path.insert(0, ROOT)

import tikv_amd
from tikv_amd import _ffi as F


def run(eng, rgn, req, tag, steps=3):
    eng.dag_run(req, [rgn])
    t0 = time.perf_counter()
    kns = 0
    for _ in range(steps):
        _, n, k = eng.dag_run(req, [rgn])
        kns += k
    dt = (time.perf_counter() - t0) / steps
    print("%-28s wall %.2f ms  kernel %.2f ms  (groups %d)"
          % (tag, dt * 1e3, kns / steps / 1e6, n))


def main():
    rows = int(sys.argv[1]) if len(sys.argv) > 1 else 50_000_000
    gen = tikv_amd.GenRegion(config_index=2, n_rows=rows, table_id=1, n_cols=64)
    eng = tikv_amd.Engine(0)
    rgn = eng.region(gen)
    cols = [tikv_amd.Col(1),
            tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=2),
            tikv_amd.Col(3, tp=F.TP_VARCHAR)]

    def dag(aggs):
        return tikv_amd.DagSelect(cols).hash_agg(aggs, tikv_amd.Expr().col(0)).build()

    run(eng, rgn, dag([tikv_amd.count_star(), tikv_amd.sum_col(1, decimal=2),
                       tikv_amd.avg_col(0)]), "full (count+sumdec+avg)")
    run(eng, rgn, dag([tikv_amd.count_star(), tikv_amd.avg_col(0)]),
        "no-decimal (count+avg)")
    run(eng, rgn, dag([tikv_amd.count_star()]), "count-only BY col0")
    run(eng, rgn, dag([tikv_amd.count_star(), tikv_amd.sum_col(1, decimal=2)]),
        "count+sumdec")
    # simple agg over the decimal col: isolates decimal parse without hash
    req = tikv_amd.DagSelect(cols).simple_agg(
        [tikv_amd.count_star(), tikv_amd.sum_col(1, decimal=2)]).build()
    run(eng, rgn, req, "simple count+sumdec")
    req = tikv_amd.DagSelect(cols).simple_agg([tikv_amd.count_star()]).build()
    run(eng, rgn, req, "simple count-only")
    rgn.close()
    eng.close()
    gen.close()


if __name__ == "__main__":
    main()
