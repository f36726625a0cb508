from .embedding_lookup import Ragged, embedding_lookup, row_to_split
