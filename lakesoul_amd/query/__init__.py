from .engine import (  # noqa: F401
    distinct_indices, factorize, groupby_agg, hash_join, sort_indices)
