from .tokenizers import build_tokenizer  # noqa: F401
