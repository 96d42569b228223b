from .tokenizer import ByteTokenizer, ChatTemplater, HFTokenizer, make_tokenizer
from .service import ModelEntry, ModelManager

__all__ = ["ByteTokenizer", "ChatTemplater", "HFTokenizer", "make_tokenizer",
           "ModelEntry", "ModelManager"]
