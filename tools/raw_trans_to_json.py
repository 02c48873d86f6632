#!/usr/bin/env python3
"""Raw text -> jsonl (reference data_tools/gpt/raw_trans_to_json.py).

    python tools/raw_trans_to_json.py --input_path dir_or_file \
        --output_path corpus.jsonl [--json_key text]
Each non-empty line (or blank-line-separated paragraph with
--mode paragraph) becomes one {"text": ...} record.
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from paddlefleetx_amd.utils.log import logger


def iter_files(path):
    if os.path.isfile(path):
        yield path
        return
    for root, _, files in os.walk(path):
        for f in sorted(files):
            if f.endswith((".txt", ".text")):
                yield os.path.join(root, f)


def main():
    p = argparse.ArgumentParser("raw_trans_to_json")
    p.add_argument("--input_path", required=True)
    p.add_argument("--output_path", required=True)
    p.add_argument("--json_key", default="text")
    p.add_argument("--mode", choices=["line", "paragraph"], default="line")
    args = p.parse_args()

    n = 0
    with open(args.output_path, "w", encoding="utf-8") as out:
        for path in iter_files(args.input_path):
            with open(path, encoding="utf-8", errors="replace") as f:
                if args.mode == "line":
                    for line in f:
                        line = line.strip()
                        if line:
                            out.write(json.dumps({args.json_key: line},
                                                 ensure_ascii=False) + "\n")
                            n += 1
                else:
                    para = []
                    for line in list(f) + [""]:
                        line = line.strip()
                        if line:
                            para.append(line)
                        elif para:
                            out.write(json.dumps(
                                {args.json_key: " ".join(para)},
                                ensure_ascii=False) + "\n")
                            para = []
                            n += 1
    logger.info(f"wrote {n} records to {args.output_path}")


if __name__ == "__main__":
    main()
