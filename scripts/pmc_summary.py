"""Summarize rocprofv3 counter CSVs for the mfma gemm kernel by grid size."""
import csv, glob, sys, collections
rows = collections.defaultdict(dict)
files = glob.glob(sys.argv[1] + "/**/*.csv", recursive=True)
for f in files:
    try:
        for r in csv.DictReader(open(f)):
            if "mfma" not in r.get("Kernel_Name", ""):
                continue
            key = (r["Grid_Size"], r["Dispatch_Id"])
            rows[key][r["Counter_Name"]] = float(r["Counter_Value"])
    except Exception:
        pass
agg = collections.defaultdict(lambda: collections.defaultdict(float))
cnt = collections.defaultdict(int)
for (grid, _d), cs in rows.items():
    for k, v in cs.items():
        agg[grid][k] += v
    cnt[grid] += 1
for grid, cs in sorted(agg.items(), key=lambda x: int(x[0])):
    n = max(cnt[grid], 1)
    hit, miss = cs.get("TCC_HIT_sum", 0), cs.get("TCC_MISS_sum", 0)
    fetch = cs.get("FETCH_SIZE", 0) / n
    print("grid=%s dispatches=%d L2hit=%.1f%% fetch=%.0f MB/disp"
          % (grid, n, 100 * hit / max(hit + miss, 1), fetch / 1024))
