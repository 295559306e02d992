#!/usr/bin/env python3
"""Run every experiment config and collect results into one JSON
(the reference's run_comprehensive.py shells out per config and parses stdout,
experiments/paper/run_comprehensive.py:44-131; here experiments run in-process
through the Python API and histories are saved directly).

Usage:
  python experiments/run_all.py [--configs experiments/configs]
      [--pattern "attacks/*"] [--out experiments/results.json]
      [--timeout 1800]
"""

from __future__ import annotations

import argparse
import json
import os
import sys
import time
import traceback
from pathlib import Path

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def run_one(path: Path, timeout_s: float) -> dict:
    from murmura_amd.cli import _run_simulation
    from murmura_amd.config.loader import load_config
    from murmura_amd.parallel.runner import DistributedRunner

    cfg = load_config(path)
    t0 = time.time()
    if cfg.backend == "simulation":
        history = _run_simulation(cfg, verbose=False)
    else:
        history = DistributedRunner(cfg).run()
    elapsed = time.time() - t0
    final = {k: (v[-1] if v else None) for k, v in history.items()
             if isinstance(v, list)}
    return {
        "name": cfg.experiment.name,
        "status": "ok",
        "elapsed_s": elapsed,
        "final_accuracy": final.get("mean_accuracy"),
        "final_honest_accuracy": final.get("honest_accuracy"),
        "history": history,
    }


def main() -> None:
    ap = argparse.ArgumentParser()
    ap.add_argument("--configs", default="experiments/configs")
    ap.add_argument("--pattern", default="**/*.yaml")
    ap.add_argument("--out", default="experiments/results.json")
    ap.add_argument("--timeout", type=float, default=1800.0)
    ap.add_argument("--limit", type=int, default=None)
    args = ap.parse_args()

    files = sorted(Path(args.configs).glob(args.pattern))
    if args.limit:
        files = files[: args.limit]
    results = {}
    # resume: keep prior successful results, re-run errors/missing
    out_path = Path(args.out)
    if out_path.exists():
        try:
            prior = json.loads(out_path.read_text())
            results = {k: v for k, v in prior.items() if v.get("status") == "ok"}
            if results:
                print(f"resuming: {len(results)} prior results kept")
        except Exception:
            pass
    for i, f in enumerate(files):
        key = str(f.relative_to(args.configs)).replace("/", "__").removesuffix(".yaml")
        if key in results:
            continue
        print(f"[{i + 1}/{len(files)}] {key}", flush=True)
        try:
            results[key] = run_one(f, args.timeout)
        except Exception as e:  # noqa: BLE001
            results[key] = {"status": "error", "error": f"{type(e).__name__}: {e}",
                            "trace": traceback.format_exc()}
        Path(args.out).parent.mkdir(parents=True, exist_ok=True)
        Path(args.out).write_text(json.dumps(results, indent=1))
    ok = sum(1 for r in results.values() if r.get("status") == "ok")
    print(f"done: {ok}/{len(results)} ok -> {args.out}")


if __name__ == "__main__":
    main()
