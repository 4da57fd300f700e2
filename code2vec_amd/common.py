"""Shared string/metric utilities.

Behavior matches the reference's `common.py` semantics (normalization
common.py:12-18, legal-name filter :122-129, subtoken split :131-133,
top-k first-match scan :180-187, word2vec export :82-91, prediction-result
parsing :135-158) without any TF dependency.
"""

import re
from collections import OrderedDict
from datetime import datetime
from itertools import repeat, takewhile
from typing import Iterable, List, Optional, Tuple

import numpy as np

_NON_ALPHA_RE = re.compile(r'[^a-zA-Z]')
_LEGAL_NAME_RE = re.compile(r'^[a-zA-Z|]+$')


class common:
    @staticmethod
    def normalize_word(word: str) -> str:
        """Strip non-alphabetic chars and lowercase; if nothing remains, just lowercase."""
        stripped = _NON_ALPHA_RE.sub('', word)
        return word.lower() if not stripped else stripped.lower()

    @staticmethod
    def legal_method_names_checker(special_words, name: str) -> bool:
        return name != special_words.OOV and bool(_LEGAL_NAME_RE.match(name))

    @staticmethod
    def filter_impossible_names(special_words, top_words: Iterable[str]) -> List[str]:
        return [w for w in top_words if common.legal_method_names_checker(special_words, w)]

    @staticmethod
    def get_subtokens(name: str) -> List[str]:
        return name.split('|')

    @staticmethod
    def get_first_match_word_from_top_predictions(
            special_words, original_name: str,
            top_predicted_words: Iterable[str]) -> Optional[Tuple[int, str]]:
        """Index (within the legal-filtered list) of the first prediction whose
        normalized form equals the normalized original name."""
        normalized_original = common.normalize_word(original_name)
        for idx, predicted in enumerate(
                common.filter_impossible_names(special_words, top_predicted_words)):
            if common.normalize_word(predicted) == normalized_original:
                return idx, predicted
        return None

    @staticmethod
    def save_word2vec_file(output_file, index_to_word, vocab_embedding_matrix: np.ndarray):
        assert len(vocab_embedding_matrix.shape) == 2
        vocab_size, dim = vocab_embedding_matrix.shape
        output_file.write('%d %d\n' % (vocab_size, dim))
        for word_idx in range(vocab_size):
            assert word_idx in index_to_word
            output_file.write(index_to_word[word_idx] + ' ')
            output_file.write(' '.join(map(str, vocab_embedding_matrix[word_idx])) + '\n')

    @staticmethod
    def count_lines_in_file(file_path: str) -> int:
        with open(file_path, 'rb') as f:
            bufgen = takewhile(lambda x: x, (f.raw.read(1024 * 1024) for _ in repeat(None)))
            return sum(buf.count(b'\n') for buf in bufgen)

    @staticmethod
    def load_file_lines(path: str) -> List[str]:
        with open(path, 'r') as f:
            return f.read().splitlines()

    @staticmethod
    def split_to_batches(data_lines, batch_size):
        for i in range(0, len(data_lines), batch_size):
            yield data_lines[i:i + batch_size]

    @staticmethod
    def chunks(lst, n):
        for i in range(0, len(lst), n):
            yield lst[i:i + n]

    @staticmethod
    def get_unique_list(lst: Iterable) -> list:
        return list(OrderedDict((item, 0) for item in lst).keys())

    @staticmethod
    def now_str() -> str:
        return datetime.now().strftime("%Y%m%d-%H%M%S: ")

    @staticmethod
    def parse_prediction_results(raw_prediction_results, unhash_dict, special_words,
                                 topk: int = 5) -> List['MethodPredictionResults']:
        """Turn raw model predictions into display-ready results: drop OOV
        suggestions, split into subtokens, sort attention descending and keep
        the top-k attended contexts (with unhashed path strings when known)."""
        out = []
        for single in raw_prediction_results:
            res = MethodPredictionResults(single.original_name)
            for i, predicted in enumerate(single.topk_predicted_words):
                if predicted == special_words.OOV:
                    continue
                score = single.topk_predicted_words_scores[i]
                res.append_prediction(common.get_subtokens(predicted),
                                      float(getattr(score, 'item', lambda: score)()))
            ranked_contexts = sorted(single.attention_per_context,
                                     key=single.attention_per_context.get,
                                     reverse=True)[:topk]
            for context in ranked_contexts:
                attention = single.attention_per_context[context]
                token1, hashed_path, token2 = context
                if hashed_path in unhash_dict:
                    res.append_attention_path(
                        float(getattr(attention, 'item', lambda: attention)()),
                        token1=token1, path=unhash_dict[hashed_path], token2=token2)
            out.append(res)
        return out


class MethodPredictionResults:
    def __init__(self, original_name: str):
        self.original_name = original_name
        self.predictions = []
        self.attention_paths = []

    def append_prediction(self, name, probability):
        self.predictions.append({'name': name, 'probability': probability})

    def append_attention_path(self, attention_score, token1, path, token2):
        self.attention_paths.append({'score': attention_score, 'path': path,
                                     'token1': token1, 'token2': token2})
