from .asp import ASP
from .sparse_masklib import create_mask
from .permutation_search import search_for_good_permutation, exhaustive_search

__all__ = ["ASP", "create_mask", "search_for_good_permutation", "exhaustive_search"]
