"""Aggregate rocprofv3 counter CSVs for the L2-NN engine A/B."""
import collections
import csv
import glob

for eng in ("1", "0"):
    files = glob.glob(f"gpurun_out/prof256_{eng}/**/*counter_collection.csv",
                      recursive=True)
    agg = collections.defaultdict(lambda: collections.defaultdict(float))
    for f in files:
        for row in csv.DictReader(open(f)):
            k = row.get("Kernel_Name", "?")
            agg[k][row.get("Counter_Name")] += float(row.get("Counter_Value", 0))
    print(f"=== engine {'256' if eng == '1' else '2d'} ({len(files)} csv) ===",
          flush=True)
    for k, c in agg.items():
        if "l2nn" in k:
            print(" ", k[:80])
            for n, v in sorted(c.items()):
                print(f"    {n:24s} {v:.4e}")
