#!/usr/bin/env python3
"""Merge TunableOp CSVs into the shipped gfx950 table by BEST MEASURED
TIME per (op, key): a retuned entry replaces the shipped one only when
its measured time is lower (the whole-table quick retune regressed e2e
4.5% — see ROADMAP — so best-of-key is the only safe merge).

    python benchmarks/merge_tunableop.py table.csv new1.csv [new2.csv ...]
"""

import sys


def read(path):
    validators, rows = [], {}
    with open(path) as f:
        for line in f:
            line = line.rstrip("\n")
            if not line:
                continue
            if line.startswith("Validator,"):
                validators.append(line)
                continue
            parts = line.split(",")
            if len(parts) < 4:
                continue
            key = (parts[0], parts[1])
            try:
                t = float(parts[3])
            except ValueError:
                continue
            if key not in rows or t < rows[key][1]:
                rows[key] = (line, t)
    return validators, rows


def main():
    table, news = sys.argv[1], sys.argv[2:]
    validators, rows = read(table)
    replaced = added = 0
    for path in news:
        _, new_rows = read(path)
        for key, (line, t) in new_rows.items():
            if key not in rows:
                rows[key] = (line, t)
                added += 1
            elif t < rows[key][1]:
                rows[key] = (line, t)
                replaced += 1
    with open(table, "w") as f:
        for v in validators:
            f.write(v + "\n")
        for line, _ in sorted(rows.values()):
            f.write(line + "\n")
    print(f"merged: {added} new keys, {replaced} faster keys, "
          f"{len(rows)} total")


if __name__ == "__main__":
    main()
