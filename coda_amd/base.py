"""ModelSelector interface.

The three-method contract every selection algorithm implements, plus the
`stochastic` attribute the harness reads to decide whether further seeds are
needed (reference: coda/base.py:1-16, main.py:105,164-168).
"""
from __future__ import annotations


class ModelSelector:
    """Abstract active-model-selection algorithm.

    Attributes:
        stochastic: True if the algorithm made any random choice so far, in
            which case the harness runs additional seeds.
    """

    stochastic: bool = False

    def get_next_item_to_label(self):
        """Pick the next unlabeled point to send to the oracle.

        Returns:
            (index, selection_probability_or_score)
        """
        raise NotImplementedError

    def add_label(self, chosen_idx, true_class, selection_prob):
        """Incorporate an oracle label for a point."""
        raise NotImplementedError

    def get_best_model_prediction(self):
        """Return the index of the currently-believed-best model."""
        raise NotImplementedError
