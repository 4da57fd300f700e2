"""Metric semantics vs hand-computed values (reference
tensorflow_model.py:450-516 and common.py:122-187)."""

import numpy as np

from code2vec_amd.common import common
from code2vec_amd.utils.metrics import (SubtokensEvaluationMetric,
                                        TopKAccuracyEvaluationMetric)
from code2vec_amd.vocabularies import _SpecialVocabWords_JoinedOovPad as SP


def _filter(top):
    return common.filter_impossible_names(SP, top)


def _first_match(orig, top):
    return common.get_first_match_word_from_top_predictions(SP, orig, top)


def test_normalize_word():
    assert common.normalize_word('getX123') == 'getx'
    assert common.normalize_word('123') == '123'
    assert common.normalize_word('Get|Value') == 'getvalue'


def test_legal_name_filter():
    top = ['<PAD_OR_OOV>', 'get|value', 'bad name', 'x1y', 'ok']
    assert _filter(top) == ['get|value', 'ok']


def test_subtoken_metric_hand_computed():
    m = SubtokensEvaluationMetric(_filter)
    # prediction 'get|value' vs original 'get|item':
    # TP: get (1); FP: value (1); FN: item (1)
    m.update_batch([('get|item', ['<PAD_OR_OOV>', 'get|value'])])
    assert (m.nr_true_positives, m.nr_false_positives, m.nr_false_negatives) == (1, 1, 1)
    assert m.precision == 0.5 and m.recall == 0.5 and m.f1 == 0.5
    # duplicates count with multiplicity
    m2 = SubtokensEvaluationMetric(_filter)
    m2.update_batch([('a|a|b', ['a|a|a'])])
    assert (m2.nr_true_positives, m2.nr_false_positives, m2.nr_false_negatives) == (3, 0, 1)


def test_topk_metric_rank_accumulation():
    m = TopKAccuracyEvaluationMetric(3, _first_match)
    # match at (post-filter) rank 1 → counts for k=2,3 but not k=1
    m.update_batch([('value', ['<PAD_OR_OOV>', 'other', 'value'])])
    # the OOV word is filtered before ranking: legal list = [other, value]
    np.testing.assert_array_equal(m.nr_correct_predictions, [0, 1, 1])
    m.update_batch([('get', ['get', 'x', 'y'])])
    np.testing.assert_array_equal(m.nr_correct_predictions, [1, 2, 2])
    np.testing.assert_allclose(m.topk_correct_predictions, [0.5, 1.0, 1.0])


def test_topk_metric_normalized_match():
    m = TopKAccuracyEvaluationMetric(1, _first_match)
    # normalization strips non-alpha and lowercases before comparison
    m.update_batch([('getValue123', ['getvalue'])])
    np.testing.assert_array_equal(m.nr_correct_predictions, [1])
