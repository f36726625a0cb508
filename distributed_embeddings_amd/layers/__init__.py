from .embedding import ConcatOneHotEmbedding, Embedding, scaled_uniform_init
from .integer_lookup import IntegerLookup
