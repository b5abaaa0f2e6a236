"""Scrape sweep outputs into CSV (the reference's scripts/scrape.py
analog): each result file's last line is a python dict printed by
kernel_benchmark.py; parse (ast.literal_eval, not eval) and emit CSV.

Usage: python benchmarks/scrape.py <dir-or-files...> [> sweep.csv]
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import ast
import csv
import os
import sys


def parse_file(path):
    last = None
    with open(path) as f:
        for line in f:
            line = line.strip()
            if line.startswith("{") and line.endswith("}"):
                last = line
    if last is None:
        return None
    try:
        d = ast.literal_eval(last)
        return d if isinstance(d, dict) else None
    except (ValueError, SyntaxError):
        return None


def main(argv):
    paths = []
    for a in argv:
        if os.path.isdir(a):
            paths += [os.path.join(a, f) for f in sorted(os.listdir(a))
                      if f.endswith(".txt")]
        else:
            paths.append(a)
    rows = [d for d in (parse_file(p) for p in paths) if d]
    if not rows:
        print("no results", file=sys.stderr)
        return 1
    cols = sorted({k for d in rows for k in d})
    w = csv.DictWriter(sys.stdout, fieldnames=cols)
    w.writeheader()
    for d in rows:
        w.writerow(d)
    return 0


if __name__ == "__main__":
    sys.exit(main(sys.argv[1:]))
