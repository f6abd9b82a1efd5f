#!/usr/bin/env python3
"""bench.py — coprocessor rows/sec on the BASELINE workload (contract bench).

N=1 workload = BASELINE.json configs[1]: 100 M-row x 16-i64-column synthetic
region, TableScan + Selection(col3 < k @10% selectivity) + count(*), on one
MI355X. A "step" is one full pass of the fused scan/filter/agg hot path over
the HBM-resident region. Multi-GPU: one process per GPU (torchrun), each rank
owns its own Region shard (weak scaling — Regions are disjoint key ranges,
exactly TiDB's per-Region fan-out, endpoint.rs:238-248); the only exchange is
the final count merge (one u64 all_reduce).

Output: ONE JSON line from rank 0 (see the repo task contract).
"""
import argparse
import ctypes as C
import importlib.util
import json
import os
import sys
import time

ROOT = os.path.dirname(os.path.abspath(__file__))
sys.path.insert(0, ROOT)

N_ROWS_DEFAULT = 100_000_000
FILTER_K = -800_000_000          # 10% selectivity over uniform ±1e9
HBM_PEAK_GBS = 8000.0            # 8 TB/s spec (MI355X_MICROARCH.md)


def log(msg):
    sys.stderr.write("[bench] %s\n" % msg)
    sys.stderr.flush()


def load_oracle():
    spec = importlib.util.spec_from_file_location(
        "orc_ffi", os.path.join(ROOT, "oracle", "orc_ffi.py"))
    mod = importlib.util.module_from_spec(spec)
    spec.loader.exec_module(mod)
    return mod


def build_request(tikv_amd, F, filter_offset=3):
    cols = [tikv_amd.Col(i) for i in range(1, 17)]
    sel = tikv_amd.cmp_col_const(filter_offset, F.SIG_LT_INT, FILTER_K)
    return (tikv_amd.DagSelect(cols).where(sel)
            .simple_agg([tikv_amd.count_star()]).build())


def parse_count(data):
    assert len(data) == 9 and data[0] == 3
    return int.from_bytes(data[1:9], "big") ^ (1 << 63)


def cpu_baseline_leg(gen, req, target_seconds=12.0):
    """Time the ORACLE (kind 'port') on a bounded sample of the same workload
    on this host's cores (single thread). Returns (rows_per_sec, sample_desc)."""
    orc = load_oracle()
    # probe on 100k rows, then size the sample for ~target_seconds
    probe_rows = min(100_000, gen.n_kv)
    t0 = time.perf_counter()
    orc.dag_run(req, gen.keys, gen.key_offs, gen.vals, gen.val_offs, probe_rows)
    dt = time.perf_counter() - t0
    rps = probe_rows / dt
    sample = int(min(gen.n_kv, max(probe_rows, rps * target_seconds)))
    t0 = time.perf_counter()
    orc.dag_run(req, gen.keys, gen.key_offs, gen.vals, gen.val_offs, sample)
    dt = time.perf_counter() - t0
    return sample / dt, "%d rows of the same region, 1 thread" % sample


def cpu_checksum_leg(gen, target_seconds=12.0):
    orc = load_oracle()
    probe = min(200_000, gen.n_kv)
    t0 = time.perf_counter()
    orc.checksum(gen.keys, gen.key_offs, gen.vals, gen.val_offs, probe)
    dt = time.perf_counter() - t0
    sample = int(min(gen.n_kv, max(probe, probe / dt * target_seconds)))
    t0 = time.perf_counter()
    orc.checksum(gen.keys, gen.key_offs, gen.vals, gen.val_offs, sample)
    dt = time.perf_counter() - t0
    return sample / dt, "%d KV pairs of the same region, 1 thread" % sample


def read_traffic():
    """Per-launch HBM bytes measured offline with rocprofv3 --pmc (FETCH_SIZE
    x2 gfx950 correction + WRITE_SIZE, per MI355X_MICROARCH.md §HBM), stored
    by tools/roofline.py into profiles/pmc_traffic.json. None if absent."""
    p = os.path.join(ROOT, "profiles", "pmc_traffic.json")
    try:
        with open(p) as f:
            d = json.load(f)
        return d.get("cfg2_scan_hbm_bytes_per_launch")
    except Exception:
        return None


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--gpus", type=int, default=1)
    ap.add_argument("--steps", type=int, default=20)
    ap.add_argument("--warmup", type=int, default=5)
    ap.add_argument("--rows", type=int, default=N_ROWS_DEFAULT,
                    help="rows per GPU (dev override; BASELINE value default)")
    ap.add_argument("--filter-offset", type=int, default=3,
                    help="dev: which column offset the predicate filters")
    ap.add_argument("--no-cpu-baseline", action="store_true")
    ap.add_argument("--workload", default="cfg2",
                    choices=["cfg2", "cfg3", "cfg4", "cfg5"],
                    help="cfg2 is the contract workload; others are secondary lines")
    args = ap.parse_args()

    # torch first: initialize HIP device discovery before the engine's own
    # runtime use in this process
    import torch
    have_cuda = torch.cuda.is_available()

    import tikv_amd
    from tikv_amd import _ffi as F

    world = int(os.environ.get("WORLD_SIZE", "1"))
    rank = int(os.environ.get("RANK", "0"))
    local_rank = int(os.environ.get("LOCAL_RANK", "0"))
    dist = None
    if world > 1:
        import torch.distributed as tdist
        # RCCL needs one device per rank; oversubscribed test boxes use gloo
        n_dev0 = torch.cuda.device_count() if have_cuda else 0
        backend = "nccl" if have_cuda and world <= n_dev0 else "gloo"
        tdist.init_process_group(backend=backend)
        dist = tdist

    n_rows = args.rows
    cfg_index = {"cfg2": 1, "cfg3": 2, "cfg4": 3, "cfg5": 4}[args.workload]
    log("rank %d/%d: generating %d rows (%s shape)" % (rank, world, n_rows, args.workload))
    t0 = time.perf_counter()
    gen = tikv_amd.GenRegion(config_index=cfg_index, n_rows=n_rows, table_id=1,
                             first_handle=rank * n_rows,
                             n_cols=64 if args.workload == "cfg3" else 0)
    log("generated in %.1fs (%.2f GB values)" %
        (time.perf_counter() - t0, gen.val_bytes() / 1e9))

    if args.workload == "cfg2":
        req = build_request(tikv_amd, F, args.filter_offset)
    elif args.workload == "cfg3":
        cols = [tikv_amd.Col(1),
                tikv_amd.Col(2, tp=F.TP_NEWDECIMAL, decimal=2),
                tikv_amd.Col(3, tp=F.TP_VARCHAR)]
        req = tikv_amd.DagSelect(cols).hash_agg(
            [tikv_amd.count_star(), tikv_amd.sum_col(1, decimal=2),
             tikv_amd.avg_col(0)], tikv_amd.Expr().col(0)).build()
    elif args.workload == "cfg5":
        cols = [tikv_amd.Col(1), tikv_amd.Col(2),
                tikv_amd.Col(-1, pk_handle=True)]
        sel = tikv_amd.cmp_col_const(1, F.SIG_GE_INT, 0)
        req = (tikv_amd.DagSelect(cols, index=True).where(sel)
               .hash_agg([tikv_amd.count_star(), tikv_amd.sum_col(1)],
                         tikv_amd.Expr().col(0)).build())
    else:
        req = None
    n_dev = torch.cuda.device_count() if have_cuda else 1
    dev = local_rank % max(n_dev, 1)
    eng = tikv_amd.Engine(dev)
    rgn = eng.region(gen)
    if have_cuda:
        torch.cuda.set_device(dev)

    # algorithmic bytes per pass: every encoded value byte once + the
    # val_offs the kernel reads (8 B per row boundary) + the filter column's
    # cell-directory plane (1 B per row, built at ingest; COPR_NO_DIR drops
    # it and the kernel's read of it together). Keys are only read by the
    # checksum (cfg4).
    algo_bytes = gen.val_bytes() + 8 * (gen.n_kv + 1)
    if not os.environ.get("COPR_NO_DIR") and args.workload == "cfg2":
        algo_bytes += gen.n_kv
    if args.workload == "cfg4":
        algo_bytes += gen.key_bytes() + 8 * (gen.n_kv + 1)
    elif args.workload == "cfg5":
        # index scans parse the KEY stream
        algo_bytes = gen.key_bytes() + 8 * (gen.n_kv + 1)

    def step():
        if args.workload == "cfg4":
            cs, kvs, byts = eng.checksum([rgn])
            return cs, 0
        data, n, kns = eng.dag_run(req, [rgn])
        if args.workload in ("cfg3", "cfg5"):
            return n, kns
        return parse_count(data), kns

    # warmup
    cnt = None
    for _ in range(args.warmup):
        cnt, _ = step()
    if dist:
        t = torch.tensor([cnt], dtype=torch.long,
                         device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t)
        cnt = int(t.item())

    # timed region (each step already ends with a hipStreamSynchronize inside
    # copr_dag_run; torch sync covers any torch-side stream)
    if dist:
        dist.barrier()
    if have_cuda:
        torch.cuda.synchronize()
    kern_ns_total = 0
    t0 = time.perf_counter()
    for _ in range(args.steps):
        _, kns = step()
        kern_ns_total += kns
    if have_cuda:
        torch.cuda.synchronize()
    if dist:
        dist.barrier()
    elapsed = time.perf_counter() - t0
    if dist:
        t = torch.tensor([elapsed],
                         device="cuda" if dist.get_backend() == "nccl" else "cpu")
        dist.all_reduce(t, op=dist.ReduceOp.MAX)
        elapsed = float(t.item())

    total_rows = n_rows * world * args.steps
    rows_per_sec = total_rows / elapsed
    ms_per_step = elapsed / args.steps * 1e3
    kern_s = kern_ns_total / 1e9 / args.steps
    if args.workload == "cfg4":
        kern_s = elapsed / args.steps     # checksum call is synchronous
    achieved_gbs = algo_bytes / kern_s / 1e9 if kern_s > 0 else 0.0

    result = None
    if rank == 0:
        # the PMC-measured traffic applies to the cfg2 kernel; scale by rows
        traffic = None
        if args.workload == "cfg2":
            t = read_traffic()
            if t is not None:
                traffic = t * (n_rows / N_ROWS_DEFAULT)
        cpu = None
        if world == 1 and not args.no_cpu_baseline:
            log("cpu baseline (oracle) ...")
            if args.workload == "cfg4":
                cpu_rps, sample = cpu_checksum_leg(gen)
            else:
                cpu_rps, sample = cpu_baseline_leg(gen, req)
            cpu = {"value": cpu_rps, "unit": "rows/s", "cores": 1,
                   "kind": "port", "sample": sample}
        result = {
            "metric": "coprocessor rows/sec (scan+filter+count)",
            "value": rows_per_sec,
            "unit": "rows/s",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": None,
            "dtype": "int64",
            "data": "synthetic",
            "config": {
                "workload": {
                    "cfg2": "cfg2: 100M-row i64x16 TableScan + Selection(col3<k, 10%) + count(*)",
                    "cfg3": "cfg3: mixed i64/Decimal/VarBytes TableScan + HashAgg(count,sum,avg BY col0, K=64)",
                    "cfg4": "cfg4: CRC64-XOR checksum over KV pairs",
                    "cfg5": "cfg5: secondary IndexScan + Selection + HashAgg",
                }[args.workload],
                "rows_per_gpu": n_rows,
                "selectivity": 0.1,
                "parallelism": "region-sharded dp%d" % world,
            },
            "roofline": {
                "bound": "hbm",
                "achieved": achieved_gbs,
                "peak": HBM_PEAK_GBS,
                "unit": "GB/s",
                "frac": achieved_gbs / HBM_PEAK_GBS,
                "traffic": traffic,
            },
            "cpu_baseline": cpu,
        }
        print(json.dumps(result))
    rgn.close()
    eng.close()
    gen.close()
    if dist:
        dist.destroy_process_group()
    return result


if __name__ == "__main__":
    main()
