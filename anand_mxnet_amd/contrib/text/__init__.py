"""contrib.text: vocabulary and token embeddings (reference parity:
python/mxnet/contrib/text/{vocab.py, embedding.py, utils.py})."""
from . import utils
from .vocab import Vocabulary
from .embedding import TokenEmbedding, CustomEmbedding

__all__ = ["Vocabulary", "TokenEmbedding", "CustomEmbedding", "utils"]
