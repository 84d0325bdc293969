"""Print a per-kernel time table from a rocpd_summarize.py JSON.

Usage: python tools/ksum_print.py KSUM.json [topN]
"""

import json
import sys


def main():
    d = json.load(open(sys.argv[1]))
    top = int(sys.argv[2]) if len(sys.argv) > 2 else 20
    ks = d["kernels"] if "kernels" in d else list(d.values())[0]["kernels"]
    rows = sorted(ks.items(), key=lambda kv: -kv[1]["total_ns"])
    tot = sum(v["total_ns"] for _, v in rows)
    print(f"total gpu kernel time {tot / 1e6:.1f} ms over "
          f"{sum(v['count'] for _, v in rows)} dispatches")
    for name, v in rows[:top]:
        avg_us = v["total_ns"] / v["count"] / 1e3
        print(f'{v["total_ns"] / 1e6:9.2f} ms  n={v["count"]:6d} '
              f'avg={avg_us:9.1f} us  {name[:72]}')


if __name__ == "__main__":
    main()
