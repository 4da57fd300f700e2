"""Whole-system integration on CPU: .java sources → c2v-extract → preprocess
(histograms + truncate/pad + dictionaries) → training → evaluation →
interactive-predict bridge. This is the reference's preprocess.sh + train.sh
+ REPL flow end to end on our native components."""

import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EXTRACTOR = os.path.join(ROOT, 'extractor', 'c2v-extract')


@pytest.fixture(scope='module')
def java_corpus(tmp_path_factory):
    if not os.path.isfile(EXTRACTOR):
        r = subprocess.run(['make', '-C', os.path.dirname(EXTRACTOR)],
                           capture_output=True, text=True)
        if r.returncode != 0:
            pytest.skip('extractor unavailable')
    root = tmp_path_factory.mktemp('corpus')
    bodies = [
        ('GetterA', 'int getValue() { return this.value + 1; }'),
        ('GetterB', 'int getCount() { return this.count + 2; }'),
        ('SetterA', 'void setValue(int v) { this.value = v; }'),
        ('SetterB', 'void setCount(int c) { this.count = c; }'),
        ('Checker', 'boolean isEmpty() { return size == 0; }'),
        ('Maker', 'String makeName(String a) { return a + "x"; }'),
    ]
    for split in ('train', 'val', 'test'):
        d = root / split
        d.mkdir()
        for i in range(8):
            for name, body in bodies:
                (d / ('%s%s%d.java' % (name, split, i))).write_text(
                    'class %s%d { %s }' % (name, i, body))
    return root


def test_extract_preprocess_train_predict(java_corpus, tmp_path):
    # 1. extract each split
    raw = {}
    for split in ('train', 'val', 'test'):
        out = subprocess.run(
            [EXTRACTOR, '--dir', str(java_corpus / split),
             '--max_path_length', '8', '--max_path_width', '2',
             '--num_threads', '4'],
            capture_output=True, text=True)
        assert out.returncode == 0
        raw[split] = tmp_path / ('%s.raw.txt' % split)
        raw[split].write_text(out.stdout)
        assert out.stdout.strip()

    # 2. preprocess (self-contained histogram mode)
    outname = str(tmp_path / 'ds')
    r = subprocess.run(
        ['python', '-m', 'code2vec_amd.data.preprocess',
         '--train_data', str(raw['train']), '--val_data', str(raw['val']),
         '--test_data', str(raw['test']), '--max_contexts', '12',
         '--word_vocab_size', '1000', '--path_vocab_size', '1000',
         '--target_vocab_size', '1000', '--output_name', outname],
        capture_output=True, text=True,
        env=dict(os.environ, PYTHONPATH=ROOT), cwd=str(tmp_path))
    assert r.returncode == 0, r.stderr[-1500:]
    for suffix in ('.train.c2v', '.val.c2v', '.test.c2v', '.dict.c2v'):
        assert os.path.isfile(outname + suffix)

    # 3. train + evaluate through the real model stack
    from code2vec_amd.config import Config
    from code2vec_amd.models.torch_model import Code2VecModel
    cfg = Config(set_defaults=True)
    cfg.TRAIN_DATA_PATH_PREFIX = outname
    cfg.TEST_DATA_PATH = outname + '.val.c2v'
    cfg.MODEL_SAVE_PATH = str(tmp_path / 'm' / 'model')
    cfg.MAX_CONTEXTS = 12
    cfg.TOKEN_EMBEDDINGS_SIZE = 16
    cfg.PATH_EMBEDDINGS_SIZE = 16
    cfg.CODE_VECTOR_SIZE = 48
    cfg.TARGET_EMBEDDINGS_SIZE = 48
    cfg.TRAIN_BATCH_SIZE = cfg.TEST_BATCH_SIZE = 16
    cfg.NUM_TRAIN_EPOCHS = 30
    cfg.SAVE_EVERY_EPOCHS = 30
    cfg.COMPUTE_DTYPE = 'fp32'
    cfg.DEVICE = 'cpu'
    cfg.VERBOSE_MODE = 0
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        model = Code2VecModel(cfg)
        model.train()
        results = model.evaluate()
    finally:
        os.chdir(cwd)
    # 6 distinct, highly separable method names: training must learn them
    assert results.topk_acc[-1] > 0.5
    # 4. predict through the extractor bridge (REPL path)
    from code2vec_amd.serving.extractor import Extractor
    ex = Extractor(cfg)
    ex.native_bin = EXTRACTOR
    src = tmp_path / 'Input.java'
    src.write_text('class Q { int getValue() { return this.value + 1; } }')
    lines, unhash = ex.extract_paths(str(src))
    preds = model.predict(lines)
    assert preds and preds[0].original_name == 'get|value'
    words = list(preds[0].topk_predicted_words)
    assert 'get|value' in words[:3], words


def test_interactive_predictor_loop(tmp_path, monkeypatch, capsys):
    """InteractivePredictor's REPL loop with stubbed model/extractor: one
    'ready' keypress predicts Input.java and prints the top-k names,
    attention contexts (paths unhashed) and the code vector; 'q' exits
    (reference behavior: interactive_predict.py:28-57)."""
    import numpy as np

    from code2vec_amd.models.model_base import ModelPredictionResults
    from code2vec_amd.serving import interactive_predict as ip

    raw = ModelPredictionResults(
        original_name='get|value',
        topk_predicted_words=np.array(['get|value', 'set|value']),
        topk_predicted_words_scores=np.array([0.9, 0.1]),
        attention_per_context={('this', '1234', 'value'): 0.75,
                               ('value', '77', 'one'): 0.25},
        code_vector=np.array([0.5, -0.5]))

    class StubVocab:
        class target_vocab:  # noqa: N801 — attribute-shaped stub
            class special_words:  # noqa: N801
                OOV = '<OOV>'

    class StubModel:
        vocabs = StubVocab
        def predict(self, lines):
            return [raw] if lines else []

    class StubExtractor:
        def __init__(self, *a, **k):
            pass
        def extract_paths(self, filename):
            assert filename == ip.INPUT_FILENAME
            return ['get|value this,1234,value'], {'1234': 'unhashed^path'}

    monkeypatch.setattr(ip, 'Extractor', StubExtractor)

    class Cfg:
        EXPORT_CODE_VECTORS = True

    pred = ip.InteractivePredictor(Cfg(), StubModel())
    answers = iter(['', 'q'])
    monkeypatch.setattr('builtins.input', lambda: next(answers))
    pred.predict()
    out = capsys.readouterr().out
    assert 'Original name:\tget|value' in out
    assert 'predicted: get|value' in out or "predicted: ['get', 'value']" in out
    assert 'unhashed^path' in out          # unhash dict applied
    assert 'Code vector:' in out and '0.5' in out
    assert 'Exiting...' in out
