"""Interactive prediction shell.

Behavior contract (reference: interactive_predict.py:12-57): loop forever —
the user edits `Input.java` and hits enter, the extractor turns it into
path-contexts, the model predicts, and the shell prints the top-k candidate
names with probabilities, the top attention-weighted contexts (with paths
shown unhashed via the extractor's reverse map), and optionally the raw code
vector. Typing an exit keyword leaves the loop.
"""

from ..common import common
from .extractor import Extractor

SHOW_TOP_CONTEXTS = 10
MAX_PATH_LENGTH = 8
MAX_PATH_WIDTH = 2
INPUT_FILENAME = 'Input.java'
EXIT_KEYWORDS = frozenset(('exit', 'quit', 'q'))


class InteractivePredictor:
    def __init__(self, config, model):
        model.predict([])  # force lazy predict-path build before first input
        self.model = model
        self.config = config
        self.path_extractor = Extractor(config,
                                        max_path_length=MAX_PATH_LENGTH,
                                        max_path_width=MAX_PATH_WIDTH)

    def _predict_one_file(self, filename: str):
        lines, unhash = self.path_extractor.extract_paths(filename)
        raw_results = self.model.predict(lines)
        parsed = common.parse_prediction_results(
            raw_results, unhash,
            self.model.vocabs.target_vocab.special_words,
            topk=SHOW_TOP_CONTEXTS)
        for raw, method in zip(raw_results, parsed):
            self._print_method(raw, method)

    def _print_method(self, raw, method):
        print('Original name:\t' + method.original_name)
        for candidate in method.predictions:
            print('\t(%f) predicted: %s' % (candidate['probability'],
                                            candidate['name']))
        print('Attention:')
        for ctx in method.attention_paths:
            print('%f\tcontext: %s,%s,%s' % (ctx['score'], ctx['token1'],
                                             ctx['path'], ctx['token2']))
        if self.config.EXPORT_CODE_VECTORS:
            print('Code vector:')
            print(' '.join(str(x) for x in raw.code_vector))

    def predict(self):
        print('Starting interactive prediction...')
        while True:
            print('Modify the file: "%s" and press any key when ready, '
                  'or "q" / "quit" / "exit" to exit' % INPUT_FILENAME)
            if input().lower() in EXIT_KEYWORDS:
                print('Exiting...')
                return
            try:
                self._predict_one_file(INPUT_FILENAME)
            except ValueError as e:
                print(e)
