#!/usr/bin/env python3
"""Plot a QPS sweep produced by run.sh (output tokens/s + TTFT vs QPS).
Writes CSV always; PNG when matplotlib is importable."""
import argparse
import csv
import glob
import json
import re


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--prefix", default="results")
    ap.add_argument("--output", default="sweep")
    a = ap.parse_args()
    rows = []
    for path in sorted(glob.glob(f"{a.prefix}_qps*.json")):
        m = re.search(r"qps([0-9.]+)\.json$", path)
        if not m:
            continue
        with open(path) as f:
            s = json.load(f)
        rows.append(
            {
                "qps": float(m.group(1)),
                "output_tokens_per_s": s.get("output_tokens_per_s"),
                "ttft_p50_s": s.get("ttft_p50_s"),
                "ttft_p90_s": s.get("ttft_p90_s"),
                "errors": s.get("errors"),
            }
        )
    rows.sort(key=lambda r: r["qps"])
    with open(a.output + ".csv", "w", newline="") as f:
        w = csv.DictWriter(f, fieldnames=list(rows[0].keys()) if rows else
                           ["qps"])
        w.writeheader()
        w.writerows(rows)
    print(f"wrote {a.output}.csv ({len(rows)} points)")
    try:
        import matplotlib

        matplotlib.use("Agg")
        import matplotlib.pyplot as plt

        fig, (ax1, ax2) = plt.subplots(1, 2, figsize=(10, 4))
        ax1.plot([r["qps"] for r in rows],
                 [r["output_tokens_per_s"] for r in rows], "o-")
        ax1.set_xlabel("QPS")
        ax1.set_ylabel("output tokens/s")
        ax2.plot([r["qps"] for r in rows],
                 [r["ttft_p50_s"] for r in rows], "o-", label="p50")
        ax2.plot([r["qps"] for r in rows],
                 [r["ttft_p90_s"] for r in rows], "s--", label="p90")
        ax2.set_xlabel("QPS")
        ax2.set_ylabel("TTFT (s)")
        ax2.legend()
        fig.tight_layout()
        fig.savefig(a.output + ".png", dpi=120)
        print(f"wrote {a.output}.png")
    except ImportError:
        print("matplotlib not available; CSV only")


if __name__ == "__main__":
    main()
