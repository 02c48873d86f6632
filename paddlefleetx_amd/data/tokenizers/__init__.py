from paddlefleetx_amd.data.tokenizers.gpt_tokenizer import GPTTokenizer

__all__ = ["GPTTokenizer"]
