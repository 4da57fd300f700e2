"""Configuration & logging for code2vec_amd.

Behavior-compatible with the reference's flag surface and hyperparameter set
(see /root/reference/config.py:11-70 for the flag names and defaults we keep),
plus MI355X-native knobs (compute dtype, sampled softmax, distributed bucket
sizing) that the reference has no equivalent for.
"""

import logging
import os
import sys
from argparse import ArgumentParser
from math import ceil
from typing import Optional


class Config:
    """All knobs in one object, passed to every layer (reference: config.py:9)."""

    @classmethod
    def arguments_parser(cls) -> ArgumentParser:
        # Same CLI surface as the reference (config.py:11-44), minus the
        # tensorflow/keras framework switch (one backend here) which is
        # accepted-and-ignored for drop-in compatibility.
        parser = ArgumentParser()
        parser.add_argument("-d", "--data", dest="data_path", required=False,
                            help="path to preprocessed dataset prefix")
        parser.add_argument("-te", "--test", dest="test_path", metavar="FILE",
                            required=False, default='',
                            help="path to test .c2v file")
        parser.add_argument("-s", "--save", dest="save_path", metavar="FILE",
                            required=False, help="path to save the model")
        parser.add_argument("-w2v", "--save_word2v", dest="save_w2v",
                            metavar="FILE", required=False,
                            help="path to save token embeddings (word2vec text format)")
        parser.add_argument("-t2v", "--save_target2v", dest="save_t2v",
                            metavar="FILE", required=False,
                            help="path to save target embeddings (word2vec text format)")
        parser.add_argument("-l", "--load", dest="load_path", metavar="FILE",
                            required=False, help="path to load a model from")
        parser.add_argument('--save_w2v', dest='save_w2v', required=False,
                            help="save word (token) vectors in word2vec format")
        parser.add_argument('--save_t2v', dest='save_t2v', required=False,
                            help="save target vectors in word2vec format")
        parser.add_argument('--export_code_vectors', action='store_true',
                            required=False,
                            help="export code vectors for the given examples")
        parser.add_argument('--release', action='store_true',
                            help="strip optimizer state from a loaded model and "
                                 "re-save it as a smaller release artifact")
        parser.add_argument('--predict', action='store_true',
                            help="run the interactive prediction shell")
        parser.add_argument("-fw", "--framework", dest="dl_framework",
                            choices=['keras', 'tensorflow', 'torch-rocm'],
                            default='torch-rocm',
                            help="accepted for reference CLI compatibility; "
                                 "this framework has a single PyTorch-ROCm backend")
        parser.add_argument("-v", "--verbose", dest="verbose_mode", type=int,
                            required=False, default=1,
                            help="verbosity in {0,1,2}")
        parser.add_argument("-lp", "--logs-path", dest="logs_path",
                            metavar="FILE", required=False,
                            help="optional log file path")
        parser.add_argument('-tb', '--tensorboard', dest='use_tensorboard',
                            action='store_true',
                            help="accepted for compatibility (scalar logs go to the logger)")
        # MI355X-native additions:
        parser.add_argument('--device', dest='device', default=None,
                            help="torch device (default: cuda if available else cpu)")
        parser.add_argument('--dtype', dest='compute_dtype', default='bf16',
                            choices=['bf16', 'fp32'],
                            help="compute dtype for GEMM-shaped work (master weights stay fp32)")
        parser.add_argument('--sampled-softmax', dest='sampled_softmax_size',
                            type=int, default=0,
                            help="if >0, train with sampled softmax over this many negatives "
                                 "(evaluation always uses the full softmax)")
        return parser

    def set_defaults(self):
        # Training schedule (reference defaults: config.py:46-58)
        self.NUM_TRAIN_EPOCHS = 20
        self.SAVE_EVERY_EPOCHS = 1
        self.TRAIN_BATCH_SIZE = 1024
        self.TEST_BATCH_SIZE = self.TRAIN_BATCH_SIZE
        self.TOP_K_WORDS_CONSIDERED_DURING_PREDICTION = 10
        self.NUM_BATCHES_TO_LOG_PROGRESS = 100
        self.NUM_TRAIN_BATCHES_TO_EVALUATE = 1800
        self.READER_NUM_PARALLEL_BATCHES = 6
        self.SHUFFLE_BUFFER_SIZE = 10000
        self.CSV_BUFFER_SIZE = 100 * 1024 * 1024
        self.MAX_TO_KEEP = 10

        # Model hyper-parameters (reference: config.py:60-70)
        self.MAX_CONTEXTS = 200
        self.MAX_TOKEN_VOCAB_SIZE = 1301136
        self.MAX_TARGET_VOCAB_SIZE = 261245
        self.MAX_PATH_VOCAB_SIZE = 911417
        self.DEFAULT_EMBEDDINGS_SIZE = 128
        self.TOKEN_EMBEDDINGS_SIZE = self.DEFAULT_EMBEDDINGS_SIZE
        self.PATH_EMBEDDINGS_SIZE = self.DEFAULT_EMBEDDINGS_SIZE
        self.CODE_VECTOR_SIZE = self.context_vector_size
        self.TARGET_EMBEDDINGS_SIZE = self.CODE_VECTOR_SIZE
        self.DROPOUT_KEEP_RATE = 0.75
        self.SEPARATE_OOV_AND_PAD = False

        # MI355X-native knobs (no reference equivalent)
        self.DEVICE = None              # resolved lazily; 'cuda' is ROCm/HIP on this stack
        self.COMPUTE_DTYPE = 'bf16'     # GEMMs in bf16 (MFMA), masters fp32
        self.SAMPLED_SOFTMAX_SIZE = 0   # 0 = full softmax (reference behavior)
        self.ADAM_LR = 0.001            # TF AdamOptimizer defaults (reference K10)
        self.ADAM_BETA1 = 0.9
        self.ADAM_BETA2 = 0.999
        self.ADAM_EPS = 1e-8
        self.DP_BUCKET_BYTES = 4 << 20  # dense-grad all-reduce bucket size over xGMI
        self.READER_QUEUE_DEPTH = 8     # prefetched batches on the H2D copy stream
        self.READER_WORKERS = 8         # parallel parse_buffer calls (stream path)

    def load_from_args(self):
        args = self.arguments_parser().parse_args()
        self.PREDICT = args.predict
        self.MODEL_SAVE_PATH = args.save_path
        self.MODEL_LOAD_PATH = args.load_path
        self.TRAIN_DATA_PATH_PREFIX = args.data_path
        self.TEST_DATA_PATH = args.test_path
        self.RELEASE = args.release
        self.EXPORT_CODE_VECTORS = args.export_code_vectors
        self.SAVE_W2V = args.save_w2v
        self.SAVE_T2V = args.save_t2v
        self.VERBOSE_MODE = args.verbose_mode
        self.LOGS_PATH = args.logs_path
        self.DL_FRAMEWORK = args.dl_framework or 'torch-rocm'
        self.USE_TENSORBOARD = args.use_tensorboard
        self.DEVICE = args.device
        self.COMPUTE_DTYPE = args.compute_dtype
        self.SAMPLED_SOFTMAX_SIZE = args.sampled_softmax_size

    def __init__(self, set_defaults: bool = False, load_from_args: bool = False,
                 verify: bool = False):
        self.NUM_TRAIN_EPOCHS: int = 0
        self.SAVE_EVERY_EPOCHS: int = 0
        self.TRAIN_BATCH_SIZE: int = 0
        self.TEST_BATCH_SIZE: int = 0
        self.TOP_K_WORDS_CONSIDERED_DURING_PREDICTION: int = 0
        self.NUM_BATCHES_TO_LOG_PROGRESS: int = 0
        self.NUM_TRAIN_BATCHES_TO_EVALUATE: int = 0
        self.READER_NUM_PARALLEL_BATCHES: int = 0
        self.SHUFFLE_BUFFER_SIZE: int = 0
        self.CSV_BUFFER_SIZE: int = 0
        self.MAX_TO_KEEP: int = 0

        self.MAX_CONTEXTS: int = 0
        self.MAX_TOKEN_VOCAB_SIZE: int = 0
        self.MAX_TARGET_VOCAB_SIZE: int = 0
        self.MAX_PATH_VOCAB_SIZE: int = 0
        self.DEFAULT_EMBEDDINGS_SIZE: int = 0
        self.TOKEN_EMBEDDINGS_SIZE: int = 0
        self.PATH_EMBEDDINGS_SIZE: int = 0
        self.CODE_VECTOR_SIZE: int = 0
        self.TARGET_EMBEDDINGS_SIZE: int = 0
        self.DROPOUT_KEEP_RATE: float = 0
        self.SEPARATE_OOV_AND_PAD: bool = False

        self.PREDICT: bool = False
        self.MODEL_SAVE_PATH: Optional[str] = None
        self.MODEL_LOAD_PATH: Optional[str] = None
        self.TRAIN_DATA_PATH_PREFIX: Optional[str] = None
        self.TEST_DATA_PATH: Optional[str] = ''
        self.RELEASE: bool = False
        self.EXPORT_CODE_VECTORS: bool = False
        self.SAVE_W2V: Optional[str] = None
        self.SAVE_T2V: Optional[str] = None
        self.VERBOSE_MODE: int = 0
        self.LOGS_PATH: Optional[str] = None
        self.DL_FRAMEWORK: str = 'torch-rocm'
        self.USE_TENSORBOARD: bool = False

        # MI355X-native knobs
        self.DEVICE: Optional[str] = None
        self.COMPUTE_DTYPE: str = 'bf16'
        self.SAMPLED_SOFTMAX_SIZE: int = 0
        self.ADAM_LR: float = 0.001
        self.ADAM_BETA1: float = 0.9
        self.ADAM_BETA2: float = 0.999
        self.ADAM_EPS: float = 1e-8
        self.DP_BUCKET_BYTES: int = 4 << 20
        self.READER_QUEUE_DEPTH: int = 8
        self.READER_WORKERS: int = 8

        # Filled by the model base when datasets are counted
        self.NUM_TRAIN_EXAMPLES: int = 0
        self.NUM_TEST_EXAMPLES: int = 0

        self.__logger: Optional[logging.Logger] = None

        if set_defaults:
            self.set_defaults()
        if load_from_args:
            self.load_from_args()
        if verify:
            self.verify()

    # ---- derived properties (reference: config.py:143-230) ----

    @property
    def context_vector_size(self) -> int:
        # [source-token ‖ path ‖ target-token] concatenation width
        return self.PATH_EMBEDDINGS_SIZE + 2 * self.TOKEN_EMBEDDINGS_SIZE

    @property
    def is_training(self) -> bool:
        return bool(self.TRAIN_DATA_PATH_PREFIX)

    @property
    def is_loading(self) -> bool:
        return bool(self.MODEL_LOAD_PATH)

    @property
    def is_saving(self) -> bool:
        return bool(self.MODEL_SAVE_PATH)

    @property
    def is_testing(self) -> bool:
        return bool(self.TEST_DATA_PATH)

    @property
    def train_steps_per_epoch(self) -> int:
        return ceil(self.NUM_TRAIN_EXAMPLES / self.TRAIN_BATCH_SIZE) if self.TRAIN_BATCH_SIZE else 0

    @property
    def test_steps(self) -> int:
        return ceil(self.NUM_TEST_EXAMPLES / self.TEST_BATCH_SIZE) if self.TEST_BATCH_SIZE else 0

    def data_path(self, is_evaluating: bool = False):
        return self.TEST_DATA_PATH if is_evaluating else self.train_data_path

    def batch_size(self, is_evaluating: bool = False):
        return self.TEST_BATCH_SIZE if is_evaluating else self.TRAIN_BATCH_SIZE

    @property
    def train_data_path(self) -> Optional[str]:
        if not self.is_training:
            return None
        return '{}.train.c2v'.format(self.TRAIN_DATA_PATH_PREFIX)

    @property
    def word_freq_dict_path(self) -> Optional[str]:
        if not self.is_training:
            return None
        return '{}.dict.c2v'.format(self.TRAIN_DATA_PATH_PREFIX)

    # ---- checkpoint path scheme, kept identical to the reference so released
    # models and tooling interoperate (reference: config.py:191-230) ----

    @classmethod
    def get_vocabularies_path_from_model_path(cls, model_file_path: str) -> str:
        return '/'.join(model_file_path.split('/')[:-1] + ["dictionaries.bin"])

    @classmethod
    def get_entire_model_path(cls, model_path: str) -> str:
        return model_path + '__entire-model'

    @classmethod
    def get_model_weights_path(cls, model_path: str) -> str:
        return model_path + '__only-weights'

    @property
    def model_load_dir(self):
        return '/'.join(self.MODEL_LOAD_PATH.split('/')[:-1])

    @property
    def entire_model_load_path(self) -> Optional[str]:
        return self.get_entire_model_path(self.MODEL_LOAD_PATH) if self.is_loading else None

    @property
    def model_weights_load_path(self) -> Optional[str]:
        return self.get_model_weights_path(self.MODEL_LOAD_PATH) if self.is_loading else None

    @property
    def entire_model_save_path(self) -> Optional[str]:
        return self.get_entire_model_path(self.MODEL_SAVE_PATH) if self.is_saving else None

    @property
    def model_weights_save_path(self) -> Optional[str]:
        return self.get_model_weights_path(self.MODEL_SAVE_PATH) if self.is_saving else None

    def resolve_device(self) -> str:
        if self.DEVICE:
            return self.DEVICE
        try:
            import torch
            return 'cuda' if torch.cuda.is_available() else 'cpu'
        except Exception:
            return 'cpu'

    @property
    def torch_compute_dtype(self):
        import torch
        return torch.bfloat16 if self.COMPUTE_DTYPE == 'bf16' else torch.float32

    def verify(self):
        if not self.is_training and not self.is_loading:
            raise ValueError("Must train or load a model.")
        if self.is_loading and not os.path.isdir(self.model_load_dir):
            raise ValueError("Model load dir `{}` does not exist.".format(self.model_load_dir))

    def __iter__(self):
        for attr_name in dir(self):
            if attr_name.startswith("__"):
                continue
            try:
                attr_value = getattr(self, attr_name, None)
            except Exception:
                attr_value = None
            if callable(attr_value):
                continue
            yield attr_name, attr_value

    def get_logger(self) -> logging.Logger:
        if self.__logger is None:
            self.__logger = logging.getLogger('code2vec')
            self.__logger.setLevel(logging.INFO)
            self.__logger.handlers = []
            self.__logger.propagate = False
            formatter = logging.Formatter('%(asctime)s %(levelname)-8s %(message)s')
            if self.VERBOSE_MODE >= 1:
                ch = logging.StreamHandler(sys.stdout)
                ch.setLevel(logging.INFO)
                ch.setFormatter(formatter)
                self.__logger.addHandler(ch)
            if self.LOGS_PATH:
                fh = logging.FileHandler(self.LOGS_PATH)
                fh.setLevel(logging.INFO)
                fh.setFormatter(formatter)
                self.__logger.addHandler(fh)
        return self.__logger

    def log(self, msg):
        self.get_logger().info(msg)
