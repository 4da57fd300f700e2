"""Evaluation metrics — same semantics as the reference's Python-side metric
classes (tensorflow_model.py:450-516): top-k exact-match accuracy over
normalized names, and subtoken precision/recall/F1 of the FIRST legal
prediction (legal = non-OOV and matching ^[a-zA-Z|]+$)."""

from collections import Counter

import numpy as np

from ..common import common


class SubtokensEvaluationMetric:
    def __init__(self, filter_impossible_names_fn):
        self.nr_true_positives = 0
        self.nr_false_positives = 0
        self.nr_false_negatives = 0
        self.nr_predictions = 0
        self.filter_impossible_names_fn = filter_impossible_names_fn

    def update_batch(self, results):
        for original_name, top_words in results:
            legal = self.filter_impossible_names_fn(top_words)
            if not legal:
                # No legal word in the top-k: count the whole target as missed
                # (the reference indexes [0] and would raise; real vocabularies
                # always contain legal words, but be robust for tiny tests).
                self.nr_false_negatives += sum(
                    Counter(common.get_subtokens(original_name)).values())
                self.nr_predictions += 1
                continue
            prediction = legal[0]
            original_subtokens = Counter(common.get_subtokens(original_name))
            predicted_subtokens = Counter(common.get_subtokens(prediction))
            self.nr_true_positives += sum(
                c for tok, c in predicted_subtokens.items() if tok in original_subtokens)
            self.nr_false_positives += sum(
                c for tok, c in predicted_subtokens.items() if tok not in original_subtokens)
            self.nr_false_negatives += sum(
                c for tok, c in original_subtokens.items() if tok not in predicted_subtokens)
            self.nr_predictions += 1

    @property
    def precision(self):
        denom = self.nr_true_positives + self.nr_false_positives
        return self.nr_true_positives / denom if denom else 0.0

    @property
    def recall(self):
        denom = self.nr_true_positives + self.nr_false_negatives
        return self.nr_true_positives / denom if denom else 0.0

    @property
    def f1(self):
        p, r = self.precision, self.recall
        return 2 * p * r / (p + r) if (p + r) else 0.0


class TopKAccuracyEvaluationMetric:
    def __init__(self, top_k: int, get_first_match_word_from_top_predictions_fn):
        self.top_k = top_k
        self.nr_correct_predictions = np.zeros(self.top_k)
        self.nr_predictions = 0
        self.get_first_match_word_from_top_predictions_fn = \
            get_first_match_word_from_top_predictions_fn

    def update_batch(self, results):
        for original_name, top_predicted_words in results:
            self.nr_predictions += 1
            found_match = self.get_first_match_word_from_top_predictions_fn(
                original_name, top_predicted_words)
            if found_match is not None:
                suggestion_idx, _ = found_match
                # a hit at rank i counts as a hit for every k ≥ i
                self.nr_correct_predictions[suggestion_idx:self.top_k] += 1

    @property
    def topk_correct_predictions(self):
        if not self.nr_predictions:
            return np.zeros(self.top_k)
        return self.nr_correct_predictions / self.nr_predictions
