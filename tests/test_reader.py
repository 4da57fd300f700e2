"""Reader parsing/filter semantics (reference path_context_reader.py:153-228):
padding defaults, OOV lookups, the any-part-non-PAD validity mask, the
train/evaluate row filters, and the unfiltered predict path."""

import pickle

import numpy as np
import pytest

from code2vec_amd.config import Config
from code2vec_amd.data.reader import EstimatorAction, PathContextReader
from code2vec_amd.vocabularies import Code2VecVocabs


@pytest.fixture()
def vocabs_and_cfg(tmp_path):
    prefix = str(tmp_path / 'ds')
    with open(prefix + '.dict.c2v', 'wb') as f:
        pickle.dump({'s1': 3, 's2': 2, 't1': 2}, f)     # tokens
        pickle.dump({'p1': 2, 'p2': 1}, f)              # paths
        pickle.dump({'name|one': 5, 'other': 2}, f)     # targets
    cfg = Config(set_defaults=True)
    cfg.TRAIN_DATA_PATH_PREFIX = prefix
    cfg.MAX_CONTEXTS = 4
    cfg.TRAIN_BATCH_SIZE = cfg.TEST_BATCH_SIZE = 2
    cfg.NUM_TRAIN_EPOCHS = 1
    cfg.SHUFFLE_BUFFER_SIZE = 0
    return Code2VecVocabs(cfg), cfg


def test_parse_line_basic(vocabs_and_cfg):
    vocabs, cfg = vocabs_and_cfg
    r = PathContextReader(vocabs, cfg, EstimatorAction.Train)
    target, tidx, src, pth, tgt, mask, _ = r._parse_line('name|one s1,p1,t1 s2,p2,s1')
    assert target == 'name|one'
    assert tidx == vocabs.target_vocab.word_to_index['name|one']
    tok = vocabs.token_vocab.word_to_index
    assert src[0] == tok['s1'] and pth[0] == vocabs.path_vocab.word_to_index['p1']
    assert tgt[0] == tok['t1']
    np.testing.assert_array_equal(mask, [1, 1, 0, 0])


def test_parse_line_oov_and_padding(vocabs_and_cfg):
    vocabs, cfg = vocabs_and_cfg
    r = PathContextReader(vocabs, cfg, EstimatorAction.Train)
    # unknown words → OOV; in the joined regime OOV == PAD, so an
    # all-unknown context has mask 0 (same as the reference's PAD-index test)
    _, tidx, src, pth, tgt, mask, _ = r._parse_line('zzz u1,u2,u3 s1,u2,u3')
    assert tidx == vocabs.target_vocab.oov_index
    assert src[0] == vocabs.token_vocab.oov_index
    np.testing.assert_array_equal(mask, [0, 1, 0, 0])


def test_parse_line_trailing_spaces_and_short_contexts(vocabs_and_cfg):
    vocabs, cfg = vocabs_and_cfg
    r = PathContextReader(vocabs, cfg, EstimatorAction.Train)
    # trailing space padding (preprocess pads lines) and a 2-part context
    _, _, src, pth, tgt, mask, _ = r._parse_line('other s1,p1   \n')
    assert mask[0] == 1 and mask[1] == 0


def test_row_filters(vocabs_and_cfg):
    vocabs, cfg = vocabs_and_cfg
    train_r = PathContextReader(vocabs, cfg, EstimatorAction.Train)
    eval_r = PathContextReader(vocabs, cfg, EstimatorAction.Evaluate)
    pred_r = PathContextReader(vocabs, cfg, EstimatorAction.Predict)

    all_pad = train_r._parse_line('name|one')          # no contexts at all
    oov_target = train_r._parse_line('zzz s1,p1,t1')   # target OOV, ctx valid
    good = train_r._parse_line('name|one s1,p1,t1')

    assert not train_r._row_passes_filter(all_pad[1], all_pad[5])
    assert not train_r._row_passes_filter(oov_target[1], oov_target[5])
    assert train_r._row_passes_filter(good[1], good[5])
    # evaluate keeps OOV-target rows (only requires a valid context)
    assert eval_r._row_passes_filter(oov_target[1], oov_target[5])
    assert not eval_r._row_passes_filter(all_pad[1], all_pad[5])
    # predict applies no filter
    assert pred_r._row_passes_filter(all_pad[1], all_pad[5])


def test_iter_batches_and_epoch_repeat(vocabs_and_cfg, tmp_path):
    vocabs, cfg = vocabs_and_cfg
    data = tmp_path / 'ds.train.c2v'
    lines = ['name|one s1,p1,t1 s2,p2,s1\n',
             'other s2,p1,s1\n',
             'zzz s1,p1,t1\n']          # filtered out in training (OOV target)
    data.write_text(''.join(lines) * 2)
    cfg.NUM_TRAIN_EPOCHS = 2
    r = PathContextReader(vocabs, cfg, EstimatorAction.Train)
    batches = list(r.iter_batches(str(data)))
    total_rows = sum(b.source_token_indices.shape[0] for b in batches)
    assert total_rows == 2 * 2 * 2  # 2 valid rows × file dup ×2 × 2 epochs
    b0 = batches[0]
    assert b0.source_token_indices.shape == (2, cfg.MAX_CONTEXTS)
    assert b0.context_valid_mask.dtype.is_floating_point
    assert b0.target_index is not None


def test_process_input_row_predict_unfiltered(vocabs_and_cfg):
    vocabs, cfg = vocabs_and_cfg
    r = PathContextReader(vocabs, cfg, EstimatorAction.Predict, keep_strings=True)
    batch = r.process_input_row('')  # completely empty row must not crash
    assert batch.source_token_indices.shape == (1, cfg.MAX_CONTEXTS)
    assert batch.context_valid_mask.sum() == 0
    assert batch.source_token_strings is not None


def test_dp_sharding(vocabs_and_cfg, tmp_path):
    vocabs, cfg = vocabs_and_cfg
    data = tmp_path / 'shard.c2v'
    data.write_text(''.join('name|one s1,p1,t1\n' for _ in range(10)))
    cfg.NUM_TRAIN_EPOCHS = 1
    cfg.SHUFFLE_BUFFER_SIZE = 0
    r0 = PathContextReader(vocabs, cfg, EstimatorAction.Train, world_size=2, rank=0)
    r1 = PathContextReader(vocabs, cfg, EstimatorAction.Train, world_size=2, rank=1)
    n0 = sum(b.source_token_indices.shape[0] for b in r0.iter_batches(str(data)))
    n1 = sum(b.source_token_indices.shape[0] for b in r1.iter_batches(str(data)))
    assert n0 == 5 and n1 == 5


def test_reader_crlf_long_lines_empty_file(tmp_path):
    """Format robustness: CRLF line endings parse like LF; a line longer
    than the 4 MB IO chunk survives the carry logic; an empty dataset
    yields zero batches without hanging."""
    import os
    from code2vec_amd.config import Config
    from code2vec_amd.data.reader import EstimatorAction, PathContextReader
    from code2vec_amd.vocabularies import (
        Vocab, VocabType, _SpecialVocabWords_JoinedOovPad)
    from types import SimpleNamespace

    toks = ['t%d' % i for i in range(50)]
    paths = ['p%d' % i for i in range(50)]
    tgts = ['alpha', 'beta']
    vocabs = SimpleNamespace(
        token_vocab=Vocab(VocabType.Token, toks,
                          _SpecialVocabWords_JoinedOovPad),
        path_vocab=Vocab(VocabType.Path, paths,
                         _SpecialVocabWords_JoinedOovPad),
        target_vocab=Vocab(VocabType.Target, tgts,
                           _SpecialVocabWords_JoinedOovPad))
    cfg = Config(set_defaults=True)
    cfg.MAX_CONTEXTS = 8
    cfg.TRAIN_BATCH_SIZE = 4
    cfg.NUM_TRAIN_EPOCHS = 1
    cfg.SHUFFLE_BUFFER_SIZE = 0
    cfg.VERBOSE_MODE = 0

    # CRLF + a giant line (> chunk) + normal lines
    os.environ['C2V_READER_CHUNK_BYTES'] = str(1 << 16)  # 64 KB chunks
    try:
        big_ctxs = ' '.join('t1,p1,t2' for _ in range(12000))  # ~100 KB line
        data = tmp_path / 'x.train.c2v'
        with open(data, 'wb') as f:
            f.write(b'alpha t1,p1,t2 t3,p3,t4\r\n')
            f.write(('beta ' + big_ctxs + '\n').encode())
            f.write(b'alpha t5,p5,t6\r\n')
        reader = PathContextReader(vocabs=vocabs, config=cfg,
                                   estimator_action=EstimatorAction.Train)
        rows = 0
        targets = []
        for b in reader.iter_batches(data_path=str(data)):
            rows += b.source_token_indices.shape[0]
            targets.extend(b.target_index.tolist())
        assert rows == 3
        reader.stop_streaming()

        empty = tmp_path / 'e.train.c2v'
        empty.write_bytes(b'')
        reader2 = PathContextReader(vocabs=vocabs, config=cfg,
                                    estimator_action=EstimatorAction.Train)
        assert list(reader2.iter_batches(data_path=str(empty))) == []
        reader2.stop_streaming()
    finally:
        os.environ.pop('C2V_READER_CHUNK_BYTES', None)


def test_malformed_rows_degrade_not_crash(vocabs_and_cfg, tmp_path):
    """Adversarial .c2v input: malformed contexts (2 or 4 comma fields),
    over-long rows, non-UTF8 bytes, empty lines and context-less rows.
    Valid rows stream through; empty/context-less/OOV-target rows are
    dropped by the train filters; malformed contexts degrade to PAD/OOV
    per-field (the reference's decode_csv+split is equally permissive)
    — and nothing crashes the native parser."""
    vocabs, cfg = vocabs_and_cfg
    rows = [
        b'name|one s1,p1,t1 s2,p2,s1',      # valid
        b'other s1,p1',                     # 2-field context -> padded
        b'',                                # dropped
        b'name|one',                        # no contexts -> dropped
        b'other s1,p1,t1,zzz s2,p2,s1',     # 4-field context
        b'name|one ' + b's1,p1,t1 ' * 9,    # > MAX_CONTEXTS, truncated
        b'nope s1,p1,t1',                   # OOV target -> dropped (train)
        b'other \xff\xfe,p1,t1 s2,p2,s1',   # non-UTF8 token bytes
    ]
    data = tmp_path / 'mal.train.c2v'
    data.write_bytes(b'\n'.join(rows) + b'\n')
    cfg.TRAIN_DATA_PATH_PREFIX = str(tmp_path / 'mal')
    cfg.NUM_TRAIN_EPOCHS = 1
    r = PathContextReader(vocabs, cfg, EstimatorAction.Train)
    n = 0
    try:
        for b in r.iter_batches():
            assert b.source_token_indices.shape[1] == cfg.MAX_CONTEXTS
            n += int(b.target_index.shape[0])
    finally:
        r.stop_streaming(join=True)
    assert n == 5, n   # all but the empty, context-less and OOV-target rows
