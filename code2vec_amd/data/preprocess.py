"""Offline dataset builder — truncation, sampling, padding, dictionary pickles.

Reproduces the reference pipeline's behavior end to end:
- histogram construction over the raw training file (the awk pipeline in
  preprocess.sh:56-58): target histogram from field 1, origin-token histogram
  from context parts 1 and 3, path histogram from part 2.
- histogram→count-dict loading with a max-size count cutoff
  (common.py:46-58 semantics: if more than max_size words, keep only words
  with count strictly greater than the max_size'th count).
- per-method truncation to max_contexts with vocab-aware sampling
  (preprocess.py:41-56): prefer contexts whose three parts are all in-vocab,
  then partially-in-vocab ones, then drop; pad short rows with trailing
  spaces (:64-65); drop empty rows (:58-60).
- `.dict.c2v` with four pickles: word/path/target count dicts +
  num_training_examples (preprocess.py:12-20).
"""

import pickle
import random
from argparse import ArgumentParser
from collections import Counter
from typing import Dict, Tuple


def build_histograms(raw_train_path: str) -> Tuple[Dict[str, int], Dict[str, int], Dict[str, int]]:
    """One pass over the raw training file → (token, path, target) count dicts."""
    target_count: Counter = Counter()
    token_count: Counter = Counter()
    path_count: Counter = Counter()
    with open(raw_train_path, 'r') as f:
        for line in f:
            parts = line.rstrip('\n').split(' ')
            if not parts or not parts[0]:
                continue
            target_count[parts[0]] += 1
            for ctx in parts[1:]:
                if not ctx:
                    continue
                triple = ctx.split(',')
                if len(triple) != 3:
                    continue
                token_count[triple[0]] += 1
                path_count[triple[1]] += 1
                token_count[triple[2]] += 1
    return dict(token_count), dict(path_count), dict(target_count)


def load_histogram(path: str, max_size: int = None, start_from: int = 0) -> Dict[str, int]:
    """Load a `word count` histogram file, applying the reference's max-size
    cutoff rule (count must exceed the max_size'th largest count)."""
    word_to_count: Dict[str, int] = {}
    with open(path, 'r') as f:
        for line in f:
            vals = line.rstrip().split(' ')
            if len(vals) != 2:
                continue
            word, count = vals[0], int(vals[1])
            if word not in word_to_count:
                word_to_count[word] = count
    if max_size is not None and len(word_to_count) > max_size:
        min_count = sorted(word_to_count.values(), reverse=True)[max_size] + 1
        word_to_count = {w: c for w, c in word_to_count.items() if c >= min_count}
    return word_to_count


def apply_max_size_cutoff(word_to_count: Dict[str, int], max_size: int) -> Dict[str, int]:
    if len(word_to_count) <= max_size:
        return dict(word_to_count)
    min_count = sorted(word_to_count.values(), reverse=True)[max_size] + 1
    return {w: c for w, c in word_to_count.items() if c >= min_count}


def context_full_found(parts, word_to_count, path_to_count) -> bool:
    return parts[0] in word_to_count and parts[1] in path_to_count and parts[2] in word_to_count


def context_partial_found(parts, word_to_count, path_to_count) -> bool:
    return parts[0] in word_to_count or parts[1] in path_to_count or parts[2] in word_to_count


def process_file(file_path: str, data_file_role: str, dataset_name: str,
                 word_to_count: Dict[str, int], path_to_count: Dict[str, int],
                 max_contexts: int, rng: random.Random = None) -> int:
    rng = rng or random
    sum_total = sum_sampled = total = empty = max_unfiltered = 0
    output_path = '{}.{}.c2v'.format(dataset_name, data_file_role)
    with open(output_path, 'w') as outfile, open(file_path, 'r') as infile:
        for line in infile:
            parts = line.rstrip('\n').split(' ')
            target_name = parts[0]
            contexts = parts[1:]
            max_unfiltered = max(max_unfiltered, len(contexts))
            sum_total += len(contexts)

            if len(contexts) > max_contexts:
                context_parts = [c.split(',') for c in contexts]
                full = [c for i, c in enumerate(contexts)
                        if context_full_found(context_parts[i], word_to_count, path_to_count)]
                partial = [c for i, c in enumerate(contexts)
                           if context_partial_found(context_parts[i], word_to_count, path_to_count)
                           and not context_full_found(context_parts[i], word_to_count, path_to_count)]
                if len(full) > max_contexts:
                    contexts = rng.sample(full, max_contexts)
                elif len(full) + len(partial) > max_contexts:
                    contexts = full + rng.sample(partial, max_contexts - len(full))
                else:
                    contexts = full + partial

            if len(contexts) == 0:
                empty += 1
                continue
            sum_sampled += len(contexts)
            csv_padding = " " * (max_contexts - len(contexts))
            outfile.write(target_name + ' ' + " ".join(contexts) + csv_padding + '\n')
            total += 1

    print('File: ' + file_path)
    if total:
        print('Average total contexts: ' + str(float(sum_total) / total))
        print('Average final (after sampling) contexts: ' + str(float(sum_sampled) / total))
    print('Total examples: ' + str(total))
    print('Empty examples: ' + str(empty))
    print('Max number of contexts per word: ' + str(max_unfiltered))
    return total


def save_dictionaries(dataset_name: str, word_to_count, path_to_count,
                      target_to_count, num_training_examples: int):
    path = '{}.dict.c2v'.format(dataset_name)
    with open(path, 'wb') as f:
        pickle.dump(word_to_count, f)
        pickle.dump(path_to_count, f)
        pickle.dump(target_to_count, f)
        pickle.dump(num_training_examples, f)
    print('Dictionaries saved to: {}'.format(path))


def main():
    parser = ArgumentParser()
    parser.add_argument("-trd", "--train_data", dest="train_data_path", required=True)
    parser.add_argument("-ted", "--test_data", dest="test_data_path", required=True)
    parser.add_argument("-vd", "--val_data", dest="val_data_path", required=True)
    parser.add_argument("-mc", "--max_contexts", dest="max_contexts", default=200)
    parser.add_argument("-wvs", "--word_vocab_size", dest="word_vocab_size", default=1301136)
    parser.add_argument("-pvs", "--path_vocab_size", dest="path_vocab_size", default=911417)
    parser.add_argument("-tvs", "--target_vocab_size", dest="target_vocab_size", default=261245)
    parser.add_argument("-wh", "--word_histogram", dest="word_histogram", required=False)
    parser.add_argument("-ph", "--path_histogram", dest="path_histogram", required=False)
    parser.add_argument("-th", "--target_histogram", dest="target_histogram", required=False)
    parser.add_argument("-o", "--output_name", dest="output_name", required=True)
    args = parser.parse_args()

    if args.word_histogram and args.path_histogram and args.target_histogram:
        word_to_count = load_histogram(args.word_histogram, max_size=int(args.word_vocab_size))
        path_to_count = load_histogram(args.path_histogram, max_size=int(args.path_vocab_size))
        target_to_count = load_histogram(args.target_histogram, max_size=int(args.target_vocab_size))
    else:
        # Self-contained mode: build histograms from the raw training file
        # (replaces the reference's awk pipeline, preprocess.sh:56-58).
        token_c, path_c, target_c = build_histograms(args.train_data_path)
        word_to_count = apply_max_size_cutoff(token_c, int(args.word_vocab_size))
        path_to_count = apply_max_size_cutoff(path_c, int(args.path_vocab_size))
        target_to_count = apply_max_size_cutoff(target_c, int(args.target_vocab_size))

    num_training_examples = 0
    for data_file_path, data_role in zip(
            [args.test_data_path, args.val_data_path, args.train_data_path],
            ['test', 'val', 'train']):
        n = process_file(file_path=data_file_path, data_file_role=data_role,
                         dataset_name=args.output_name, word_to_count=word_to_count,
                         path_to_count=path_to_count, max_contexts=int(args.max_contexts))
        if data_role == 'train':
            num_training_examples = n

    save_dictionaries(args.output_name, word_to_count, path_to_count,
                      target_to_count, num_training_examples)


if __name__ == '__main__':
    main()
