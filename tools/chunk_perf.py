"""Device vs host TypeChunk encode wall time (whole copr_dag_run call).

Same request twice: default (device chunk encoder) and COPR_DEV_CHUNK=0
(host datum->chunk re-encode). cfg2 region, int column output, filtered
project. Run on a GPU box; appends a line to gpurun_out/chunk_perf.log.
"""
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import tikv_amd


def timed(engine, req, rgn, iters=5):
    engine.dag_run(req, [rgn])                       # warmup
    t0 = time.perf_counter()
    for _ in range(iters):
        data, nrows, _ = engine.dag_run(req, [rgn])
    dt = (time.perf_counter() - t0) / iters
    return dt, nrows, len(data)


def main():
    n_rows = int(sys.argv[1]) if len(sys.argv) > 1 else 20_000_000
    from tikv_amd import _ffi as F
    eng = tikv_amd.Engine()
    g = tikv_amd.GenRegion(config_index=2, n_rows=n_rows, table_id=5)
    rgn = eng.region(g)
    cols = [tikv_amd.Col(1),
            tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=2),
            tikv_amd.Col(3, tp=F.TP_VARCHAR)]
    out = []
    # config_index=2 col1 is the group key, uniform over [0, 64)
    for name, thr in (("sel~5pct", 3), ("sel~50pct", 32)):
        sel = tikv_amd.cmp_col_const(0, F.SIG_LT_INT, thr)
        req = (tikv_amd.DagSelect(cols).where(sel).output([0])
               .chunked().build())
        os.environ.pop("COPR_DEV_CHUNK", None)
        dev_t, dev_n, dev_b = timed(eng, req, rgn)
        os.environ["COPR_DEV_CHUNK"] = "0"
        host_t, host_n, host_b = timed(eng, req, rgn)
        del os.environ["COPR_DEV_CHUNK"]
        assert (dev_n, dev_b) == (host_n, host_b), (dev_n, dev_b, host_n, host_b)
        line = ("%s rows=%d out_rows=%d resp=%.1fMB dev=%.1fms host=%.1fms "
                "speedup=%.2fx" % (name, n_rows, dev_n, dev_b / 1e6,
                                   dev_t * 1e3, host_t * 1e3, host_t / dev_t))
        print(line, flush=True)
        out.append(line)
    rgn.close()
    g.close()
    os.makedirs("gpurun_out", exist_ok=True)
    with open("gpurun_out/chunk_perf.log", "a") as f:
        f.write("\n".join(out) + "\n")


if __name__ == "__main__":
    main()
