"""Extend the DMTT 3-condition study with additional seeds."""
import json
import os
import pathlib
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from experiments.dmtt_study_r2 import run_condition  # noqa: E402


def main():
    res = pathlib.Path(__file__).parent.parent / "experiments/results/dmtt_study_r2.json"
    out = json.loads(res.read_text())
    port = 29880
    for cond in ["static", "mobility", "dmtt"]:
        for seed in (45, 46):
            if str(seed) in out.get(cond, {}):
                continue
            h = run_condition(cond, 40, 10, port, seed)
            port += 1
            out.setdefault(cond, {})[str(seed)] = h
            last5 = sum(h["honest_accuracy"][-5:]) / 5
            print(f"{cond} seed={seed} last5={last5:.4f}", flush=True)
            res.write_text(json.dumps(out, indent=1))
    print("done")


if __name__ == "__main__":
    main()
