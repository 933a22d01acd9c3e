from .clip_tokenizer import SimpleTokenizer

__all__ = ["SimpleTokenizer"]
