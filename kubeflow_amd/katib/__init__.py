from .suggestion import (RandomSuggestion, GridSuggestion, BayesOptSuggestion,
                         make_suggestion)

__all__ = ["RandomSuggestion", "GridSuggestion", "BayesOptSuggestion",
           "make_suggestion"]
