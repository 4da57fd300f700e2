#!/usr/bin/env python3
"""CLI driver — same dispatch surface as the reference's code2vec.py:16-38:
train / export token-vecs / export target-vecs / test / release / interactive
predict, all driven by flag combinations."""

from code2vec_amd.config import Config
from code2vec_amd.models.torch_model import Code2VecModel
from code2vec_amd.vocabularies import VocabType


def load_model(config: Config) -> Code2VecModel:
    """Builds the model; under torchrun (WORLD_SIZE>1) the data-parallel
    reducer (RCCL over xGMI on GPU, gloo on CPU) is wired automatically."""
    from code2vec_amd.parallel.ddp import Reducer, init_distributed_from_env
    rank, world_size = init_distributed_from_env()
    reducer = Reducer() if world_size > 1 else None
    return Code2VecModel(config, reducer=reducer, world_size=world_size,
                         rank=rank)


if __name__ == '__main__':
    config = Config(set_defaults=True, load_from_args=True, verify=True)
    model = load_model(config)

    if config.is_training:
        model.train()
    if config.SAVE_W2V is not None:
        model.save_word2vec_format(config.SAVE_W2V, VocabType.Token)
        config.log('Origin word vectors saved in word2vec text format in: %s'
                   % config.SAVE_W2V)
    if config.SAVE_T2V is not None:
        model.save_word2vec_format(config.SAVE_T2V, VocabType.Target)
        config.log('Target word vectors saved in word2vec text format in: %s'
                   % config.SAVE_T2V)
    if (config.is_testing and not config.is_training) or config.RELEASE:
        eval_results = model.evaluate()
        if eval_results is not None:
            config.log(str(eval_results).replace('topk', 'top{}'.format(
                config.TOP_K_WORDS_CONSIDERED_DURING_PREDICTION)))
    if config.PREDICT:
        from code2vec_amd.serving.interactive_predict import InteractivePredictor
        predictor = InteractivePredictor(config, model)
        predictor.predict()
    model.close_session()
