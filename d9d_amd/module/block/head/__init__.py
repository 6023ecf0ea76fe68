from .language_modelling import SplitLanguageModellingHead, LM_IGNORE_INDEX
from .classification import ClassificationHead
from .embedding import EmbeddingHead

__all__ = [
    "SplitLanguageModellingHead",
    "LM_IGNORE_INDEX",
    "ClassificationHead",
    "EmbeddingHead",
]
