"""Summarize a rocprofv3 kernel_stats CSV: top-N kernels by total time."""

import csv
import sys


def short(name: str, width: int = 70) -> str:
    name = name.replace("(anonymous namespace)::", "")
    if "(" in name:
        name = name.split("(")[0]
    return name[-width:] if len(name) > width else name


def main(path: str, top: int = 24) -> None:
    rows = []
    with open(path) as f:
        for row in csv.DictReader(f):
            rows.append((row["Name"], int(row["Calls"]), float(row["TotalDurationNs"]),
                         float(row["AverageNs"]), float(row["Percentage"])))
    rows.sort(key=lambda r: -r[2])
    total_ms = sum(r[2] for r in rows) / 1e6
    print(f"total GPU kernel time: {total_ms:.0f} ms")
    print(f"{'kernel':70s} {'calls':>6s} {'total ms':>9s} {'avg us':>9s} {'%':>6s}")
    for name, calls, tot, avg, pct in rows[:top]:
        print(f"{short(name):70s} {calls:6d} {tot / 1e6:9.1f} {avg / 1e3:9.1f} {pct:6.2f}")


if __name__ == "__main__":
    main(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 24)
