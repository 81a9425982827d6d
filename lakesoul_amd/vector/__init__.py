from .index import VectorIndex, build_vector_index, vector_search  # noqa: F401
