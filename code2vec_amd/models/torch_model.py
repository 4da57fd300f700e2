"""The PyTorch-ROCm code2vec model: training loop, evaluation, prediction,
checkpoint I/O and embedding export.

This is the single backend replacing the reference's TF/Keras pair. Behavior
parity anchors (all in /root/reference/):
- train loop cadence: progress log every NUM_BATCHES_TO_LOG_PROGRESS batches,
  save+evaluate every SAVE_EVERY_EPOCHS epochs (tensorflow_model.py:75-101).
- evaluation: top-k words + metrics + per-example `log.txt` dump + optional
  `.vectors` export (tensorflow_model.py:114-195).
- prediction: one row at a time, unfiltered, normalized top-k scores and
  attention per context (tensorflow_model.py:311-368).
- checkpoints: `<path>__entire-model` includes optimizer state, release form
  is weights-only `<path>.release` (tensorflow_model.py:129-136,
  keras_model.py:230-296); `dictionaries.bin` sits next to the weights.
"""

import os
import time
from typing import Iterable, List, Optional

import numpy as np
import torch

from ..config import Config
from ..common import common
from ..data.reader import EstimatorAction, PathContextReader
from ..utils.metrics import SubtokensEvaluationMetric, TopKAccuracyEvaluationMetric
from ..vocabularies import VocabType
from .model_base import (Code2VecModelBase, ModelEvaluationResults,
                         ModelPredictionResults)
from .network import Code2VecNetwork, NullReducer


class Code2VecModel(Code2VecModelBase):
    def __init__(self, config: Config, reducer=None, world_size: int = 1,
                 rank: int = 0):
        self.network: Optional[Code2VecNetwork] = None
        self.reducer = reducer or NullReducer()
        self.world_size = world_size
        self.rank = rank
        self._epochs_trained = 0
        self._throughput_ewma = None
        super().__init__(config)

    # ---- inner model lifecycle ----

    def _create_inner_model(self):
        self.network = Code2VecNetwork(
            self.config,
            token_vocab_size=self.vocabs.token_vocab.size,
            path_vocab_size=self.vocabs.path_vocab.size,
            target_vocab_size=self.vocabs.target_vocab.size)
        self.log('Created model on device: %s (compute dtype: %s)'
                 % (self.network.device, self.config.COMPUTE_DTYPE))
        self.log('Number of trainable params: %d' % self.network.num_trainable_params())

    def _load_or_create_inner_model(self):
        self._create_inner_model()
        if self.config.is_loading:
            self._load_inner_model()

    def _checkpoint_candidates(self, base_path: str):
        return [self.config.get_entire_model_path(base_path),
                self.config.get_model_weights_path(base_path),
                base_path + '.release',
                base_path]

    def _load_inner_model(self):
        from ..utils.tf_bundle import TFCheckpointReader
        for path in self._checkpoint_candidates(self.config.MODEL_LOAD_PATH):
            if os.path.isfile(path):
                payload = torch.load(path, map_location='cpu', weights_only=False)
                state = payload.get('model', payload)
                self.network.load_state_dict(state)
                self._epochs_trained = int(payload.get('epoch', 0))
                self.log('Loaded model weights from: %s (epochs trained: %d)'
                         % (path, self._epochs_trained))
                return
            if TFCheckpointReader.is_tf_checkpoint(path):
                # released reference models (TF Saver V2 bundle,
                # tensorflow_model.py:370-377) load unchanged
                self._load_tf_checkpoint(path)
                self.log('Loaded TF-format model weights from: %s' % path)
                return
        raise ValueError('No checkpoint found for load path: %s'
                         % self.config.MODEL_LOAD_PATH)

    # TF graph variable names (reference tensorflow_model.py:205-220,249-250)
    TF_NAME_MAP = {
        'model/WORDS_VOCAB': 'tok_table',
        'model/PATHS_VOCAB': 'path_table',
        'model/TARGET_WORDS_VOCAB': 'target_table',
        'model/TRANSFORM': 'w',
        'model/ATTENTION': 'a',
    }

    def _save_tf_checkpoint(self, prefix: str):
        """Write the 5 model tensors as a TF Saver V2 bundle (weights-only
        release form; Adam slots deliberately omitted like the reference's
        `--release`)."""
        import numpy as np
        from ..utils.tf_bundle import write_checkpoint
        net = self.network
        tensors = {}
        for tf_name, param_name in self.TF_NAME_MAP.items():
            arr = net.get_param(param_name).detach().cpu().numpy()
            if param_name == 'a':
                arr = arr.reshape(-1, 1)        # TF stores ATTENTION (D, 1)
            tensors[tf_name] = np.ascontiguousarray(arr, dtype=np.float32)
        write_checkpoint(prefix, tensors)

    def _load_tf_checkpoint(self, prefix: str):
        """Load a reference-format TF V2 checkpoint (entire-model with Adam
        slots, or a weights-only `.release`) into the engine."""
        import math
        from ..utils.tf_bundle import TFCheckpointReader
        reader = TFCheckpointReader(prefix)
        net = self.network
        for tf_name, param_name in self.TF_NAME_MAP.items():
            param = net.get_param(param_name)
            arr = reader.get_tensor(tf_name)
            if param_name == 'a':
                arr = arr.reshape(-1)       # stored (D, 1)
            if tuple(arr.shape) != tuple(param.shape):
                raise ValueError(
                    'TF tensor %s has shape %s; this model expects %s '
                    '(is dictionaries.bin next to the checkpoint the one it '
                    'was trained with?)'
                    % (tf_name, tuple(arr.shape), tuple(param.shape)))
            param.copy_(torch.from_numpy(arr).to(param.device))
            for slot, store in (('/Adam', net._adam_m), ('/Adam_1', net._adam_v)):
                if reader.has_tensor(tf_name + slot):
                    s = reader.get_tensor(tf_name + slot)
                    if param_name == 'a':
                        s = s.reshape(-1)
                    store[param_name].copy_(
                        torch.from_numpy(s).to(param.device))
        # Adam step from beta1_power = beta1^t (TF non-slot variable)
        for name in reader.tensor_names():
            if name.endswith('beta1_power'):
                b1p = float(reader.get_tensor(name).reshape(-1)[0])
                if 0.0 < b1p < 1.0:
                    net.adam_step = max(1, round(math.log(b1p) / math.log(0.9)))
                break
        net._refresh_shadows()
        if net._step_t is not None:
            net._step_t.fill_(net.adam_step)

    def _save_inner_model(self, path: str):
        if self.config.RELEASE:
            # weights only, optimizer state stripped (reference release flow)
            torch.save({'model': self.network.weights_state_dict(),
                        'epoch': self._epochs_trained}, path + '.release')
            # also emit the TF Saver V2 form so the ORIGINAL reference code
            # (tf.compat.v1.train.Saver.restore) can load models trained
            # here — the inverse of the released-model import path
            self._save_tf_checkpoint(path + '.release')
            self.log('Released model saved to: %s (+ TF-format .index/.data)'
                     % (path + '.release'))
            return
        torch.save({'model': self.network.state_dict(),
                    'epoch': self._epochs_trained},
                   self.config.get_entire_model_path(path))

    # ---- training ----

    def train(self):
        self.log('Starting training')
        cfg = self.config
        reader = PathContextReader(vocabs=self.vocabs, config=cfg,
                                   estimator_action=EstimatorAction.Train,
                                   world_size=self.world_size, rank=self.rank)
        device = self.network.device
        steps_per_epoch = max(1, cfg.train_steps_per_epoch // max(1, self.world_size))
        save_every = max(1, steps_per_epoch * cfg.SAVE_EVERY_EPOCHS)

        start = time.time()
        scalar_log = None
        if cfg.USE_TENSORBOARD:
            # tensorboard isn't installed in this environment; the flag writes
            # the same scalars to a CSV instead (batch, avg_loss, samples/sec)
            scalar_path = (cfg.MODEL_SAVE_PATH or 'model') + '.scalars.csv'
            os.makedirs(os.path.dirname(scalar_path) or '.', exist_ok=True)
            scalar_log = open(scalar_path, 'a')
            scalar_log.write('batch,avg_loss,samples_per_sec\n')
        from ..data.prefetcher import BatchPrefetcher
        prefetcher = BatchPrefetcher(reader.iter_batches(), device,
                                     depth=cfg.READER_QUEUE_DEPTH)
        batches = iter(prefetcher)
        # DP termination consensus with one-batch lookahead: the shard split
        # is not batch-aligned, so ranks can finish with unequal batch counts
        # — a rank stepping once more than its peers would hang in the
        # gradient collectives. Each rank votes on its NEXT batch's existence
        # asynchronously while the current step runs (Reducer.start_vote on
        # the host-side group), so the consensus costs no per-step latency;
        # longer ranks drop their surplus batches.
        try:
            self._train_loop(cfg, batches, prefetcher, steps_per_epoch,
                             save_every, scalar_log, start)
        finally:
            # tear the pipeline threads down on EVERY exit path: a daemon
            # thread still inside the C++ parser at interpreter shutdown
            # aborts the process
            reader.stop_streaming(join=False)
            prefetcher.stop()
            reader.stop_streaming(join=True)

        elapsed = int(time.time() - start)
        self.log('Done training')
        self.log('Training time: %sH:%sM:%sS'
                 % (elapsed // 3600, (elapsed // 60) % 60, elapsed % 60))

    def _train_loop(self, cfg, batches, prefetcher, steps_per_epoch,
                    save_every, scalar_log, start):
        batch_num = 0
        sum_loss = None
        window_examples = 0
        multi_batch_start = time.time()
        nxt = next(batches, None)
        vote = self.reducer.start_vote(nxt is not None)
        while True:
            if not self.reducer.finish_vote(vote):
                break
            b = nxt
            nxt = next(batches, None)
            vote = self.reducer.start_vote(nxt is not None)
            loss = self.network.train_step(
                b.source_token_indices, b.path_indices, b.target_token_indices,
                b.context_valid_mask, b.target_index, reducer=self.reducer)
            batch_num += 1
            loss = loss.detach()
            sum_loss = loss if sum_loss is None else sum_loss + loss
            window_examples += b.source_token_indices.shape[0] * self.world_size

            if batch_num % cfg.NUM_BATCHES_TO_LOG_PROGRESS == 0:
                avg_loss = float(sum_loss) / cfg.NUM_BATCHES_TO_LOG_PROGRESS
                elapsed = time.time() - multi_batch_start
                throughput = window_examples / max(elapsed, 1e-9)
                # EWMA + epoch ETA (reference P13 progress logger semantics,
                # keras_checkpoint_saver_callback.py:92-127)
                self._throughput_ewma = (
                    throughput if self._throughput_ewma is None
                    else 0.5 * self._throughput_ewma + 0.5 * throughput)
                msg = ('Average loss at batch %d: %f, throughput: %d samples/sec'
                       % (batch_num, avg_loss, throughput))
                if steps_per_epoch > 0 and self._throughput_ewma > 0:
                    remaining = (steps_per_epoch - batch_num % steps_per_epoch)
                    eta_sec = remaining * cfg.TRAIN_BATCH_SIZE * self.world_size \
                        / self._throughput_ewma
                    msg += ', epoch ETA: %dm%02ds' % (eta_sec // 60, eta_sec % 60)
                self.log(msg)
                if scalar_log is not None:
                    scalar_log.write('%d,%f,%f\n' % (batch_num, avg_loss,
                                                     throughput))
                    scalar_log.flush()
                sum_loss = None
                window_examples = 0
                multi_batch_start = time.time()

            if (cfg.is_testing and self.rank == 0
                    and cfg.NUM_TRAIN_BATCHES_TO_EVALUATE > 0
                    and batch_num % cfg.NUM_TRAIN_BATCHES_TO_EVALUATE == 0
                    and batch_num % save_every != 0):
                # mid-epoch evaluation cadence (reference: every 1800 batches
                # via ModelEvaluationCallback, keras_model.py:326-369)
                results = self.evaluate()
                if results is not None:
                    self.log('After %d batches -- %s' % (batch_num, results))

            if batch_num % save_every == 0:
                self._epochs_trained += cfg.SAVE_EVERY_EPOCHS
                epoch_num = self._epochs_trained
                if self.rank == 0 and cfg.MODEL_SAVE_PATH:
                    save_path = cfg.MODEL_SAVE_PATH + '_iter' + str(epoch_num)
                    self.save(save_path)
                    self.log('Saved after %d epochs in: %s' % (epoch_num, save_path))
                if cfg.is_testing and self.rank == 0:
                    results = self.evaluate()
                    if results is not None:
                        self.log('After %d epochs -- %s' % (epoch_num, str(results)))

        if scalar_log is not None:
            scalar_log.close()

    # ---- evaluation ----

    def evaluate(self) -> Optional[ModelEvaluationResults]:
        cfg = self.config
        if cfg.RELEASE and cfg.is_loading:
            # `--release` with a loaded model: strip optimizer state and exit
            # (reference: tensorflow_model.py:129-136).
            self._save_inner_model(cfg.MODEL_LOAD_PATH)
            return None
        self.log('Starting evaluation')
        reader = PathContextReader(vocabs=self.vocabs, config=cfg,
                                   estimator_action=EstimatorAction.Evaluate)
        device = self.network.device
        k = cfg.TOP_K_WORDS_CONSIDERED_DURING_PREDICTION
        index_to_word = self.vocabs.target_vocab.index_to_word
        special_words = self.vocabs.target_vocab.special_words

        topk_metric = TopKAccuracyEvaluationMetric(
            k, lambda orig, top: common.get_first_match_word_from_top_predictions(
                special_words, orig, top))
        subtoken_metric = SubtokensEvaluationMetric(
            lambda top: common.filter_impossible_names(special_words, top))

        export_file = None
        if cfg.EXPORT_CODE_VECTORS:
            export_file = open(cfg.TEST_DATA_PATH + '.vectors', 'w')
        log_file = open('log.txt', 'w')
        total_loss, total_rows = 0.0, 0
        start = time.time()
        # prefetch like training (the reference eval pipeline is the same
        # tf.data graph incl. prefetch, path_context_reader.py:150); batches
        # arrive device-resident with the H2D on the copy stream
        from ..data.prefetcher import BatchPrefetcher
        prefetcher = BatchPrefetcher(reader.iter_batches(), device,
                                     depth=cfg.READER_QUEUE_DEPTH)
        try:
            nr_examples = 0
            for batch in prefetcher:
                b = batch
                if b.target_index is not None:
                    indices, scores, code, _alpha, loss_sum = \
                        self.network.eval_batch(
                            b.source_token_indices, b.path_indices,
                            b.target_token_indices, b.context_valid_mask,
                            b.target_index, top_k=k)
                    total_loss += float(loss_sum)
                    total_rows += int(b.target_index.shape[0])
                else:
                    indices, scores, code, _alpha = self.network.predict_batch(
                        b.source_token_indices, b.path_indices,
                        b.target_token_indices, b.context_valid_mask, top_k=k)
                idx_np = indices.cpu().numpy()
                top_words = [[index_to_word.get(int(i), special_words.OOV)
                              for i in row] for row in idx_np]
                pairs = list(zip(b.target_string, top_words))
                topk_metric.update_batch(pairs)
                subtoken_metric.update_batch(pairs)
                nr_examples += len(pairs)
                for original_name, words in pairs:
                    found = common.get_first_match_word_from_top_predictions(
                        special_words, original_name, words)
                    if found is not None:
                        log_file.write('Original: ' + original_name
                                       + ', predicted %d. most likely: %s\n'
                                       % (found[0] + 1, found[1]))
                    else:
                        log_file.write('No results for predicting: ' + original_name + '\n')
                if export_file is not None:
                    self._write_code_vectors(export_file, code.cpu().numpy())
        finally:
            log_file.close()
            if export_file is not None:
                export_file.close()
            reader.stop_streaming(join=False)
            prefetcher.stop()
            reader.stop_streaming(join=True)

        elapsed = int(time.time() - start)
        self.log('Done evaluating, epoch reached. Evaluated %d examples in %ds'
                 % (nr_examples, elapsed))
        return ModelEvaluationResults(
            topk_acc=topk_metric.topk_correct_predictions,
            subtoken_precision=subtoken_metric.precision,
            subtoken_recall=subtoken_metric.recall,
            subtoken_f1=subtoken_metric.f1,
            loss=(total_loss / total_rows) if total_rows else None)

    # ---- prediction (serving path) ----

    def predict(self, predict_data_lines: Iterable[str]) -> List[ModelPredictionResults]:
        reader = PathContextReader(vocabs=self.vocabs, config=self.config,
                                   estimator_action=EstimatorAction.Predict,
                                   keep_strings=True)
        device = self.network.device
        k = self.config.TOP_K_WORDS_CONSIDERED_DURING_PREDICTION
        index_to_word = self.vocabs.target_vocab.index_to_word
        special_words = self.vocabs.target_vocab.special_words
        results: List[ModelPredictionResults] = []
        for line in predict_data_lines:
            batch = reader.process_input_row(line)
            b = batch.to(device) if device.type != 'cpu' else batch
            indices, scores, code, alpha = self.network.predict_batch(
                b.source_token_indices, b.path_indices, b.target_token_indices,
                b.context_valid_mask, top_k=k, normalize_scores=True)
            words = np.array([index_to_word.get(int(i), special_words.OOV)
                              for i in indices[0].cpu().numpy()])
            attention_per_context = self._get_attention_weight_per_context(
                batch.source_token_strings[0], batch.path_strings[0],
                batch.target_token_strings[0], alpha[0].cpu().numpy())
            results.append(ModelPredictionResults(
                original_name=batch.target_string[0],
                topk_predicted_words=words,
                topk_predicted_words_scores=scores[0].cpu().numpy(),
                attention_per_context=attention_per_context,
                code_vector=code[0].cpu().numpy()))
        return results

    # ---- embedding export ----

    def _get_vocab_embedding_as_np_array(self, vocab_type: VocabType) -> np.ndarray:
        assert vocab_type in VocabType
        table = {VocabType.Token: self.network.tok_table,
                 VocabType.Path: self.network.path_table,
                 VocabType.Target: self.network.target_table}[vocab_type]
        return table.detach().cpu().numpy()
