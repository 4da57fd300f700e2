"""Vocabularies — pickle-compatible with the reference's `dictionaries.bin`.

Semantics reproduced from /root/reference/vocabularies.py:
- special-word regimes (:22-35): joined `<PAD_OR_OOV>` by default; when
  SEPARATE_OOV_AND_PAD, token/path vocabs get `<PAD>`+`<OOV>` and the target
  vocab gets OOV only (:204-209).
- a saved vocab excludes special words; they are re-added on load and the
  minimum stored index must equal the number of special words (:57-97).
- `dictionaries.bin` frames three vocabs in order token, target, path, each
  as three consecutive pickles (word_to_index, index_to_word, size) (:211-218).
- freq-dict construction takes the top-N words by count (:99-106).

Lookup tables here are plain Python dicts plus numpy-vectorizable helpers —
the string→index work is CPU-side in the reader; there is
no TF hash-table equivalent needed on the GPU path.
"""

import os
import pickle
from argparse import Namespace
from enum import Enum
from typing import Dict, Iterable, NamedTuple, Optional, Set

from .common import common
from .config import Config


class VocabType(Enum):
    Token = 1
    Target = 2
    Path = 3


SpecialVocabWordsType = Namespace

_SpecialVocabWords_OnlyOov = Namespace(OOV='<OOV>')

_SpecialVocabWords_SeparateOovPad = Namespace(PAD='<PAD>', OOV='<OOV>')

_SpecialVocabWords_JoinedOovPad = Namespace(
    PAD_OR_OOV='<PAD_OR_OOV>', PAD='<PAD_OR_OOV>', OOV='<PAD_OR_OOV>')


class Vocab:
    def __init__(self, vocab_type: VocabType, words: Iterable[str],
                 special_words: Optional[SpecialVocabWordsType] = None):
        if special_words is None:
            special_words = Namespace()
        self.vocab_type = vocab_type
        self.word_to_index: Dict[str, int] = {}
        self.index_to_word: Dict[int, str] = {}
        self.special_words: SpecialVocabWordsType = special_words

        index = 0
        for word in common.get_unique_list(special_words.__dict__.values()):
            self.word_to_index[word] = index
            self.index_to_word[index] = word
            index += 1
        for word in words:
            if word in self.word_to_index:
                continue
            self.word_to_index[word] = index
            self.index_to_word[index] = word
            index += 1
        self.size = len(self.word_to_index)

    @property
    def nr_special_words(self) -> int:
        return len(common.get_unique_list(self.special_words.__dict__.values()))

    @property
    def pad_index(self) -> int:
        return self.word_to_index[self.special_words.PAD] \
            if hasattr(self.special_words, 'PAD') else -1

    @property
    def oov_index(self) -> int:
        return self.word_to_index[self.special_words.OOV]

    def lookup_index(self, word: str) -> int:
        """string → index with OOV default (reference lookup-table semantics,
        vocabularies.py:108-125)."""
        return self.word_to_index.get(word, self.oov_index)

    def lookup_word(self, index: int) -> str:
        return self.index_to_word.get(index, self.special_words.OOV)

    def save_to_file(self, file):
        # Stored WITHOUT special words; consumers re-add them on load.
        nr_specials = self.nr_special_words
        w2i = {w: i for w, i in self.word_to_index.items() if i >= nr_specials}
        i2w = {i: w for i, w in self.index_to_word.items() if i >= nr_specials}
        pickle.dump(w2i, file)
        pickle.dump(i2w, file)
        pickle.dump(self.size - nr_specials, file)

    @classmethod
    def load_from_file(cls, vocab_type: VocabType, file,
                       special_words: SpecialVocabWordsType) -> 'Vocab':
        specials = common.get_unique_list(special_words.__dict__.values())
        w2i_wo_specials = pickle.load(file)
        i2w_wo_specials = pickle.load(file)
        size_wo_specials = pickle.load(file)
        assert len(i2w_wo_specials) == len(w2i_wo_specials) == size_wo_specials
        min_idx = min(i2w_wo_specials.keys())
        if min_idx != len(specials):
            raise ValueError(
                "Vocabulary `{}` in `{}` stores minimum word index {} but {} special "
                "words ({}) are expected to occupy the low indices. Check "
                "config.SEPARATE_OOV_AND_PAD.".format(
                    vocab_type, getattr(file, 'name', '<stream>'), min_idx,
                    len(specials), specials))
        vocab = cls(vocab_type, [], special_words)
        vocab.word_to_index = {**w2i_wo_specials,
                               **{w: i for i, w in enumerate(specials)}}
        vocab.index_to_word = {**i2w_wo_specials,
                               **{i: w for i, w in enumerate(specials)}}
        vocab.size = size_wo_specials + len(specials)
        return vocab

    @classmethod
    def create_from_freq_dict(cls, vocab_type: VocabType,
                              word_to_count: Dict[str, int], max_size: int,
                              special_words: Optional[SpecialVocabWordsType] = None):
        if special_words is None:
            special_words = Namespace()
        words_by_count = sorted(word_to_count, key=word_to_count.get, reverse=True)
        return cls(vocab_type, words_by_count[:max_size], special_words)


WordFreqDictType = Dict[str, int]


class Code2VecWordFreqDicts(NamedTuple):
    token_to_count: WordFreqDictType
    path_to_count: WordFreqDictType
    target_to_count: WordFreqDictType


class Code2VecVocabs:
    """The three vocabularies, loaded from `dictionaries.bin` next to a model
    or created from the `.dict.c2v` pickle (reference: vocabularies.py:151-230)."""

    def __init__(self, config: Config):
        self.config = config
        self.token_vocab: Optional[Vocab] = None
        self.path_vocab: Optional[Vocab] = None
        self.target_vocab: Optional[Vocab] = None
        self._already_saved_in_paths: Set[str] = set()
        self._load_or_create()

    def _load_or_create(self):
        assert self.config.is_training or self.config.is_loading
        if self.config.is_loading:
            path = self.config.get_vocabularies_path_from_model_path(
                self.config.MODEL_LOAD_PATH)
            if not os.path.isfile(path):
                raise ValueError(
                    "Model dictionaries file not found in model load dir; "
                    "expected `{}`.".format(path))
            self._load_from_path(path)
        else:
            self._create_from_word_freq_dict()

    def _load_from_path(self, vocabularies_load_path: str):
        self.config.log('Loading model vocabularies from: `%s` ...' % vocabularies_load_path)
        with open(vocabularies_load_path, 'rb') as file:
            # Framing order in the file is token, target, path.
            self.token_vocab = Vocab.load_from_file(
                VocabType.Token, file, self._get_special_words_by_vocab_type(VocabType.Token))
            self.target_vocab = Vocab.load_from_file(
                VocabType.Target, file, self._get_special_words_by_vocab_type(VocabType.Target))
            self.path_vocab = Vocab.load_from_file(
                VocabType.Path, file, self._get_special_words_by_vocab_type(VocabType.Path))
        self.config.log('Done loading model vocabularies.')
        self._already_saved_in_paths.add(vocabularies_load_path)

    def _create_from_word_freq_dict(self):
        freq = self._load_word_freq_dict()
        self.config.log('Word frequency dictionaries loaded; creating vocabularies.')
        self.token_vocab = Vocab.create_from_freq_dict(
            VocabType.Token, freq.token_to_count, self.config.MAX_TOKEN_VOCAB_SIZE,
            special_words=self._get_special_words_by_vocab_type(VocabType.Token))
        self.config.log('Created token vocab. size: %d' % self.token_vocab.size)
        self.path_vocab = Vocab.create_from_freq_dict(
            VocabType.Path, freq.path_to_count, self.config.MAX_PATH_VOCAB_SIZE,
            special_words=self._get_special_words_by_vocab_type(VocabType.Path))
        self.config.log('Created path vocab. size: %d' % self.path_vocab.size)
        self.target_vocab = Vocab.create_from_freq_dict(
            VocabType.Target, freq.target_to_count, self.config.MAX_TARGET_VOCAB_SIZE,
            special_words=self._get_special_words_by_vocab_type(VocabType.Target))
        self.config.log('Created target vocab. size: %d' % self.target_vocab.size)

    def _get_special_words_by_vocab_type(self, vocab_type: VocabType) -> SpecialVocabWordsType:
        if not self.config.SEPARATE_OOV_AND_PAD:
            return _SpecialVocabWords_JoinedOovPad
        if vocab_type == VocabType.Target:
            return _SpecialVocabWords_OnlyOov
        return _SpecialVocabWords_SeparateOovPad

    def save(self, vocabularies_save_path: str):
        if vocabularies_save_path in self._already_saved_in_paths:
            return
        with open(vocabularies_save_path, 'wb') as file:
            self.token_vocab.save_to_file(file)
            self.target_vocab.save_to_file(file)
            self.path_vocab.save_to_file(file)
        self._already_saved_in_paths.add(vocabularies_save_path)

    def _load_word_freq_dict(self) -> Code2VecWordFreqDicts:
        assert self.config.is_training
        self.config.log('Loading word frequency dictionaries from: %s ...'
                        % self.config.word_freq_dict_path)
        with open(self.config.word_freq_dict_path, 'rb') as file:
            token_to_count = pickle.load(file)
            path_to_count = pickle.load(file)
            target_to_count = pickle.load(file)
        self.config.log('Done loading word frequency dictionaries.')
        return Code2VecWordFreqDicts(token_to_count=token_to_count,
                                     path_to_count=path_to_count,
                                     target_to_count=target_to_count)

    def get(self, vocab_type: VocabType) -> Vocab:
        if not isinstance(vocab_type, VocabType):
            raise ValueError('`vocab_type` must be a VocabType member.')
        return {VocabType.Token: self.token_vocab,
                VocabType.Target: self.target_vocab,
                VocabType.Path: self.path_vocab}[vocab_type]
