from .dictionary import build_dictionary, load_dictionary, invert_dictionary
from .iterator import TextIterator
from .prepare import prepare_data

__all__ = [
    "build_dictionary",
    "load_dictionary",
    "invert_dictionary",
    "TextIterator",
    "prepare_data",
]
