from paddlefleetx_amd.data.tokenizers.gpt_tokenizer import GPTTokenizer


def t5_tokenizer(*args, **kw):
    from paddlefleetx_amd.data.tokenizers.t5_tokenizer import T5Tokenizer
    return T5Tokenizer(*args, **kw)


def debertav2_tokenizer(*args, **kw):
    from paddlefleetx_amd.data.tokenizers.debertav2_tokenizer import \
        DebertaV2Tokenizer
    return DebertaV2Tokenizer(*args, **kw)


__all__ = ["GPTTokenizer", "t5_tokenizer", "debertav2_tokenizer"]
