from .t5_tokenizer import AutoTokenizer, BatchEncoding, T5Tokenizer
from .unigram import UnigramModel, train_unigram

__all__ = ["T5Tokenizer", "AutoTokenizer", "BatchEncoding", "UnigramModel",
           "train_unigram"]
