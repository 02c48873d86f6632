#!/usr/bin/env python3
"""Offline preprocessing: jsonl text -> token bins (`_ids.npy` + `_idx.npz`).

Reference: ppfleetx/data/data_tools/gpt/preprocess_data.py (multiprocess
tokenization writing the same two files GPTDataset mmaps).

    python tools/preprocess_data.py --input_path corpus.jsonl \
        --output_prefix ./data/corpus --json_key text \
        --vocab_dir /path/with/vocab.json+merges.txt [--workers 8]
"""

import argparse
import json
import multiprocessing as mp
import os
import sys

import numpy as np

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

from paddlefleetx_amd.utils.log import logger

_tok = None
_seg = None
_wwm = False


def _init_worker(vocab_dir, family="gpt", wwm=False, seg_backend="auto",
                 lexicon_path=None):
    global _tok, _seg, _wwm
    _wwm = wwm
    if family == "ernie":
        from paddlefleetx_amd.data.tokenizers.ernie_tokenizer import \
            ErnieTokenizer
        _tok = ErnieTokenizer.from_pretrained(vocab_dir)
        if wwm:
            from paddlefleetx_amd.data.ernie_preprocess import \
                ChineseWordSegmenter
            lex = None
            if lexicon_path:
                with open(lexicon_path, encoding="utf-8") as f:
                    lex = [w.strip() for w in f if w.strip()]
            _seg = ChineseWordSegmenter(seg_backend, lexicon=lex)
    else:
        from paddlefleetx_amd.data.tokenizers import GPTTokenizer
        if vocab_dir:
            _tok = GPTTokenizer.from_pretrained(vocab_dir)
        else:
            _tok = GPTTokenizer.gpt2_tokenizer()


def _encode(line):
    line = line.strip()
    if not line:
        return None
    try:
        obj = json.loads(line)
        text = obj[_encode.json_key] if isinstance(obj, dict) else str(obj)
    except json.JSONDecodeError:
        text = line
    if _wwm:
        # ERNIE Chinese whole-word masking: persist '##' continuation
        # marks as a parallel stream (reference words_segmentation.py +
        # create_pretraining_data.py:161)
        from paddlefleetx_amd.data.ernie_preprocess import \
            create_wwm_ids_and_marks
        ids, cont = create_wwm_ids_and_marks(text, _tok, _seg)
        return ids, cont
    ids = _tok.encode(text)
    if _tok.eos_token_id is not None:
        ids.append(_tok.eos_token_id)
    return ids


def main():
    p = argparse.ArgumentParser("preprocess_data")
    p.add_argument("--input_path", required=True)
    p.add_argument("--output_prefix", required=True)
    p.add_argument("--json_key", default="text")
    p.add_argument("--vocab_dir", default=None,
                   help="dir holding vocab.json + merges.txt")
    p.add_argument("--workers", type=int, default=max(1, os.cpu_count() // 2))
    p.add_argument("--model_family", default="gpt", choices=["gpt", "ernie"])
    p.add_argument("--whole_word_mask", action="store_true",
                   help="ERNIE Chinese whole-word masking marks")
    p.add_argument("--seg_backend", default="auto",
                   choices=["auto", "jieba", "lexicon", "char"])
    p.add_argument("--lexicon_path", default=None,
                   help="one word per line, for --seg_backend lexicon")
    args = p.parse_args()
    _encode.json_key = args.json_key

    with open(args.input_path, encoding="utf-8") as f:
        lines = f.readlines()
    logger.info(f"tokenizing {len(lines)} documents with "
                f"{args.workers} workers")
    initargs = (args.vocab_dir, args.model_family, args.whole_word_mask,
                args.seg_backend, args.lexicon_path)
    if args.workers > 1:
        with mp.Pool(args.workers, initializer=_init_worker,
                     initargs=initargs) as pool:
            docs = pool.map(_encode, lines, chunksize=64)
    else:
        _init_worker(*initargs)
        docs = [_encode(l) for l in lines]
    docs = [d for d in docs if d]

    conts = None
    if args.whole_word_mask:
        conts = [d[1] for d in docs]
        docs = [d[0] for d in docs]
    lens = np.array([len(d) for d in docs], dtype=np.int64)
    ids = np.concatenate([np.asarray(d, dtype=np.int32) for d in docs])
    os.makedirs(os.path.dirname(os.path.abspath(args.output_prefix)),
                exist_ok=True)
    np.save(args.output_prefix + "_ids.npy", ids)
    np.savez(args.output_prefix + "_idx.npz", lens=lens)
    if conts is not None:
        np.save(args.output_prefix + "_wwm.npy",
                np.concatenate([np.asarray(c, dtype=np.int8)
                                for c in conts]))
    logger.info(f"wrote {args.output_prefix}_ids.npy ({ids.nbytes/1e6:.1f} MB,"
                f" {len(docs)} docs, {int(lens.sum())} tokens) and _idx.npz")


if __name__ == "__main__":
    main()
