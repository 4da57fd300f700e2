"""Vocabulary semantics: special-word regimes, pickle round-trip without
specials (reference vocabularies.py:57-97), freq-dict top-N creation, and
dictionaries.bin framing order (token, target, path)."""

import io
import pickle

import pytest

from code2vec_amd.vocabularies import (Vocab, VocabType,
                                       _SpecialVocabWords_JoinedOovPad,
                                       _SpecialVocabWords_SeparateOovPad)


def make_vocab(words=('foo', 'bar', 'baz'), specials=_SpecialVocabWords_JoinedOovPad):
    return Vocab(VocabType.Token, list(words), specials)


def test_joined_pad_oov_single_special():
    v = make_vocab()
    assert v.word_to_index['<PAD_OR_OOV>'] == 0
    assert v.pad_index == 0 and v.oov_index == 0
    assert v.word_to_index['foo'] == 1
    assert v.size == 4


def test_separate_pad_oov():
    v = make_vocab(specials=_SpecialVocabWords_SeparateOovPad)
    assert v.word_to_index['<PAD>'] == 0
    assert v.word_to_index['<OOV>'] == 1
    assert v.word_to_index['foo'] == 2
    assert v.size == 5


def test_lookup_defaults_to_oov():
    v = make_vocab()
    assert v.lookup_index('nope') == v.oov_index
    assert v.lookup_word(9999) == v.special_words.OOV
    assert v.lookup_index('bar') == 2
    assert v.lookup_word(2) == 'bar'


def test_save_load_roundtrip_excludes_specials():
    v = make_vocab()
    buf = io.BytesIO()
    v.save_to_file(buf)
    buf.seek(0)
    # stored framing: three pickles, no special words, min index == nr specials
    w2i = pickle.load(buf)
    i2w = pickle.load(buf)
    size = pickle.load(buf)
    assert '<PAD_OR_OOV>' not in w2i
    assert min(i2w.keys()) == 1
    assert size == 3
    buf.seek(0)
    v2 = Vocab.load_from_file(VocabType.Token, buf, _SpecialVocabWords_JoinedOovPad)
    assert v2.word_to_index == v.word_to_index
    assert v2.index_to_word == v.index_to_word
    assert v2.size == v.size


def test_load_wrong_special_regime_raises():
    v = make_vocab()
    buf = io.BytesIO()
    v.save_to_file(buf)
    buf.seek(0)
    with pytest.raises(ValueError):
        Vocab.load_from_file(VocabType.Token, buf, _SpecialVocabWords_SeparateOovPad)


def test_create_from_freq_dict_top_n():
    counts = {'a': 5, 'b': 9, 'c': 1, 'd': 7}
    v = Vocab.create_from_freq_dict(VocabType.Target, counts, max_size=2,
                                    special_words=_SpecialVocabWords_JoinedOovPad)
    # top-2 by count: b, d
    assert set(v.word_to_index) == {'<PAD_OR_OOV>', 'b', 'd'}
    assert v.word_to_index['b'] == 1


def test_code2vec_vocabs_dictionaries_bin_roundtrip(tmp_path):
    from code2vec_amd.config import Config
    from code2vec_amd.vocabularies import Code2VecVocabs

    # create from a .dict.c2v freq pickle
    prefix = str(tmp_path / 'ds')
    with open(prefix + '.dict.c2v', 'wb') as f:
        pickle.dump({'tokA': 3, 'tokB': 1}, f)
        pickle.dump({'p1': 2}, f)
        pickle.dump({'targ|one': 5, 'targ|two': 2}, f)
        pickle.dump(7, f)  # num_training_examples (extra frame, must be tolerated)
    cfg = Config(set_defaults=True)
    cfg.TRAIN_DATA_PATH_PREFIX = prefix
    vocabs = Code2VecVocabs(cfg)
    assert vocabs.token_vocab.size == 3
    assert vocabs.path_vocab.size == 2
    assert vocabs.target_vocab.size == 3

    # save dictionaries.bin next to a model and load it back via MODEL_LOAD_PATH
    model_dir = tmp_path / 'model'
    model_dir.mkdir()
    dict_path = str(model_dir / 'dictionaries.bin')
    vocabs.save(dict_path)
    cfg2 = Config(set_defaults=True)
    cfg2.MODEL_LOAD_PATH = str(model_dir / 'saved_model')
    vocabs2 = Code2VecVocabs(cfg2)
    assert vocabs2.token_vocab.word_to_index == vocabs.token_vocab.word_to_index
    assert vocabs2.target_vocab.word_to_index == vocabs.target_vocab.word_to_index
    assert vocabs2.path_vocab.word_to_index == vocabs.path_vocab.word_to_index
