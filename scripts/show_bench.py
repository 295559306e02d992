"""Parse bench output (first JSON line) and optionally kernel stats csv."""
import csv
import json
import sys

path = sys.argv[1]
for line in open(path):
    line = line.strip()
    if line.startswith("{"):
        d = json.loads(line)
        print(f"{d['value']:.2f} rounds/s  {d['ms_per_step']:.1f} ms/round  N={d['n_gpus']}")
        break

if len(sys.argv) > 2:
    rows = [r for r in csv.DictReader(open(sys.argv[2])) if "bn_" in r["Name"]]
    for r in rows:
        print(f"{float(r['AverageNs'])/1e3:7.1f}us x{r['Calls']:>6} {r['Name'][:60]}")
