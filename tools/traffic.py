"""Fold rocprofv3 --pmc FETCH_SIZE / WRITE_SIZE counter CSVs into
profiles/pmc_traffic.json (per-launch HBM bytes for bench.py's
roofline.traffic field).

Per MI355X_MICROARCH.md §HBM: FETCH_SIZE on gfx950 reports half the bytes
of a wide coalesced streaming read -> x1024 (KB) x2; WRITE_SIZE x1024
(uncalibrated but small here). Usage:
  python tools/traffic.py <workload> <kernel-substr> <rows> fetch.csv write.csv
"""
import csv
import json
import os
import sys

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def per_launch(path, counter, kernel_substr):
    total = 0.0
    n = 0
    with open(path) as f:
        for row in csv.DictReader(f):
            if row["Counter_Name"] != counter:
                continue
            if kernel_substr not in row["Kernel_Name"]:
                continue
            total += float(row["Counter_Value"])
            n += 1
    if n == 0:
        raise SystemExit("no %s rows for kernel %r in %s"
                         % (counter, kernel_substr, path))
    return total / n, n


def main():
    workload, kern, rows = sys.argv[1], sys.argv[2], int(sys.argv[3])
    fetch_csv, write_csv = sys.argv[4], sys.argv[5]
    # gfx950 FETCH_SIZE reports HALF the bytes of a WIDE (16 B/lane)
    # coalesced streaming read; other widths are uncalibrated
    # (MI355X_MICROARCH §HBM). x2 for glds-DMA kernels (cfg2/cfg3); x1 for
    # the CRC kernel's 8 B-granule register loads (calibrated against the
    # known algorithmic byte count: x1 lands at 0.92x algorithmic, x2 at
    # 1.8x which would exceed what the kernel can touch).
    fetch_mult = float(sys.argv[6]) if len(sys.argv) > 6 else 2.0
    fetch_kb, nf = per_launch(fetch_csv, "FETCH_SIZE", kern)
    write_kb, nw = per_launch(write_csv, "WRITE_SIZE", kern)
    # WRITE_SIZE's unit is uncalibrated on gfx950 (MI355X_MICROARCH §HBM);
    # interpreting it as KB gives impossible numbers for the atomic-only
    # write sets here while BYTES gives sane ones (cfg3 ~5.7 MB of table
    # atomics, cfg4 ~1 KB of XOR atomics) — counted as bytes, and dwarfed
    # by the read stream either way
    bytes_per_launch = fetch_kb * 1024 * fetch_mult + write_kb
    p = os.path.join(ROOT, "profiles", "pmc_traffic.json")
    with open(p) as f:
        d = json.load(f)
    d["%s_hbm_bytes_per_launch" % workload] = bytes_per_launch
    d["%s_rows" % workload] = rows
    d["%s_fetch_kb_per_launch" % workload] = fetch_kb
    d["%s_write_raw_per_launch" % workload] = write_kb
    d["%s_kernel" % workload] = kern
    d["%s_fetch_mult" % workload] = fetch_mult
    with open(p, "w") as f:
        json.dump(d, f, indent=1)
    print(workload, "bytes/launch %.3e (fetch x2 %.3e + write %.3e), "
          "%d/%d dispatches" % (bytes_per_launch, fetch_kb * 1024 * fetch_mult,
                                write_kb * 1024, nf, nw))


if __name__ == "__main__":
    main()
