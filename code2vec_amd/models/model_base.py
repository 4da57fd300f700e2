"""Abstract model API + shared result types.

Mirrors the reference's model_base.py:37-182 surface: construction order
(verify → count examples with a `.num_examples` cache → vocabs → load-or-create
→ initialize), `train()/evaluate()/predict()/save()`, word2vec export, and the
attention-weight→context-triple zipper used by the predict path."""

import abc
import os
from typing import Dict, Iterable, NamedTuple, Optional, Tuple

import numpy as np

from ..common import common
from ..config import Config
from ..vocabularies import Code2VecVocabs, VocabType


class ModelEvaluationResults(NamedTuple):
    topk_acc: float
    subtoken_precision: float
    subtoken_recall: float
    subtoken_f1: float
    loss: Optional[float] = None

    def __str__(self):
        s = 'topk_acc: {}, precision: {}, recall: {}, F1: {}'.format(
            self.topk_acc, self.subtoken_precision, self.subtoken_recall,
            self.subtoken_f1)
        if self.loss is not None:
            s = 'loss: {}, '.format(self.loss) + s
        return s


class ModelPredictionResults(NamedTuple):
    original_name: str
    topk_predicted_words: np.ndarray
    topk_predicted_words_scores: np.ndarray
    attention_per_context: Dict[Tuple[str, str, str], float]
    code_vector: Optional[np.ndarray] = None


class Code2VecModelBase(abc.ABC):
    def __init__(self, config: Config):
        self.config = config
        self.config.verify()
        self._log_creating_model()
        if not config.RELEASE:
            self._init_num_of_examples()
        self._log_model_configuration()
        self.vocabs = Code2VecVocabs(config)
        self._load_or_create_inner_model()
        self._initialize()

    def _log_creating_model(self):
        self.log('')
        self.log('---------------------------------------------------------------------')
        self.log('---------------------- Creating code2vec model ----------------------')
        self.log('---------------------------------------------------------------------')

    def _log_model_configuration(self):
        self.log('---------------------------------------------------------------------')
        self.log('----------------- Configuration - Hyper Parameters ------------------')
        longest = max(len(name) for name, _ in self.config)
        for name, val in self.config:
            self.log('{name: <{pad}}{val}'.format(name=name, val=val, pad=longest + 2))
        self.log('---------------------------------------------------------------------')

    @property
    def logger(self):
        return self.config.get_logger()

    def log(self, msg):
        self.logger.info(msg)

    def _init_num_of_examples(self):
        self.log('Checking number of examples ...')
        if self.config.is_training:
            self.config.NUM_TRAIN_EXAMPLES = self._get_num_of_examples_for_dataset(
                self.config.train_data_path)
            self.log('    Number of train examples: {}'.format(self.config.NUM_TRAIN_EXAMPLES))
        if self.config.is_testing:
            self.config.NUM_TEST_EXAMPLES = self._get_num_of_examples_for_dataset(
                self.config.TEST_DATA_PATH)
            self.log('    Number of test examples: {}'.format(self.config.NUM_TEST_EXAMPLES))

    @staticmethod
    def _get_num_of_examples_for_dataset(dataset_path: str) -> int:
        cache_path = dataset_path + '.num_examples'
        if os.path.isfile(cache_path):
            with open(cache_path, 'r') as f:
                return int(f.readline())
        n = common.count_lines_in_file(dataset_path)
        with open(cache_path, 'w') as f:
            f.write(str(n))
        return n

    def save(self, model_save_path=None):
        if model_save_path is None:
            model_save_path = self.config.MODEL_SAVE_PATH
        model_save_dir = '/'.join(model_save_path.split('/')[:-1])
        if model_save_dir and not os.path.isdir(model_save_dir):
            os.makedirs(model_save_dir, exist_ok=True)
        self.vocabs.save(self.config.get_vocabularies_path_from_model_path(model_save_path))
        self._save_inner_model(model_save_path)

    def _write_code_vectors(self, file, code_vectors):
        for vec in code_vectors:
            file.write(' '.join(map(str, vec)) + '\n')

    def _get_attention_weight_per_context(
            self, path_source_strings: Iterable[str], path_strings: Iterable[str],
            path_target_strings: Iterable[str],
            attention_weights: Iterable[float]) -> Dict[Tuple[str, str, str], float]:
        attention_weights = np.reshape(np.asarray(attention_weights), (-1,))
        attention_per_context: Dict[Tuple[str, str, str], float] = {}
        for source, path, target, weight in zip(
                path_source_strings, path_strings, path_target_strings, attention_weights):
            attention_per_context[(source, path, target)] = weight
        return attention_per_context

    def close_session(self):
        pass

    @abc.abstractmethod
    def train(self):
        ...

    @abc.abstractmethod
    def evaluate(self) -> Optional[ModelEvaluationResults]:
        ...

    @abc.abstractmethod
    def predict(self, predict_data_lines):
        ...

    @abc.abstractmethod
    def _save_inner_model(self, path):
        ...

    def save_word2vec_format(self, dest_save_path: str, vocab_type: VocabType):
        if vocab_type not in VocabType:
            raise ValueError('vocab_type must be a VocabType member.')
        vocab_embedding_matrix = self._get_vocab_embedding_as_np_array(vocab_type)
        index_to_word = self.vocabs.get(vocab_type).index_to_word
        with open(dest_save_path, 'w') as f:
            common.save_word2vec_file(f, index_to_word, vocab_embedding_matrix)

    @abc.abstractmethod
    def _get_vocab_embedding_as_np_array(self, vocab_type: VocabType) -> np.ndarray:
        ...

    @abc.abstractmethod
    def _load_or_create_inner_model(self):
        ...

    def _initialize(self):
        pass
