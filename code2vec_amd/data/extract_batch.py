"""Fault-tolerant batch extraction driver.

Reproduces the failure-handling semantics of the reference's
`JavaExtractor/extract.py:18-62` around our native `c2v-extract` (or any
extractor binary with the same CLI):

- the top-level corpus directory is split into its immediate
  subdirectories, processed by a pool of worker processes in small batches;
- each batch gets a wall-clock timeout — a stuck batch is SKIPPED, not
  retried (reference :50-57);
- each extractor process gets a kill timer (reference :26-32);
- if the extractor fails on a directory, its partial output file is deleted
  and every immediate subdirectory is retried recursively (reference
  :34-41), so one poison file costs only its own subtree;
- surviving per-directory outputs are concatenated to stdout (or --ofile).
"""

import argparse
import multiprocessing
import os
import shutil
import subprocess
import sys
import tempfile
from threading import Timer


def immediate_subdirs(d):
    return sorted(os.path.join(d, name) for name in os.listdir(d)
                  if os.path.isdir(os.path.join(d, name)))


def _dir_tag(dir_):
    return os.path.basename(os.path.normpath(dir_))


def extract_dir(bin_path, opts, out_dir, dir_, prefix=''):
    """Extract one directory into `<out_dir>/<prefix><dirname>`; on failure
    drop the partial output and recurse into immediate subdirectories."""
    out_name = os.path.join(out_dir, prefix + _dir_tag(dir_))
    cmd = [bin_path, '--dir', dir_,
           '--max_path_length', str(opts['max_path_length']),
           '--max_path_width', str(opts['max_path_width']),
           '--num_threads', str(opts['num_threads'])]
    failed = False
    with open(out_name, 'a') as out:
        proc = subprocess.Popen(cmd, stdout=out, stderr=subprocess.DEVNULL)
        kill_timer = Timer(opts['kill_timeout'], proc.kill)
        try:
            kill_timer.start()
            proc.communicate()
        finally:
            kill_timer.cancel()
        failed = proc.poll() != 0
    if failed:
        if os.path.exists(out_name):
            os.remove(out_name)
        for sub in immediate_subdirs(dir_):
            extract_dir(bin_path, opts, out_dir, sub,
                        prefix + _dir_tag(dir_) + '_')


def run(bin_path, corpus_dir, out_stream, max_path_length=8, max_path_width=2,
        num_threads=32, batch_size=3, batch_timeout=60.0,
        kill_timeout=600000.0, pool_size=4):
    opts = {'max_path_length': max_path_length,
            'max_path_width': max_path_width, 'num_threads': num_threads,
            'kill_timeout': kill_timeout}
    dirs = immediate_subdirs(corpus_dir) or [corpus_dir]
    tmp = tempfile.mkdtemp(prefix='c2v_extract_')
    skipped = []
    try:
        for i in range(0, len(dirs), batch_size):
            batch = dirs[i:i + batch_size]
            try:
                with multiprocessing.Pool(pool_size) as pool:
                    result = pool.starmap_async(
                        extract_dir,
                        [(bin_path, opts, tmp, d) for d in batch])
                    result.get(timeout=batch_timeout)
            except multiprocessing.TimeoutError:
                skipped.extend(batch)
                continue
        for name in sorted(os.listdir(tmp)):
            with open(os.path.join(tmp, name)) as f:
                shutil.copyfileobj(f, out_stream)
    finally:
        shutil.rmtree(tmp, ignore_errors=True)
    return skipped


def main():
    here = os.path.dirname(os.path.dirname(os.path.dirname(
        os.path.abspath(__file__))))
    ap = argparse.ArgumentParser(description=__doc__)
    ap.add_argument('--dir', required=True, help='corpus root directory')
    ap.add_argument('--bin', default=os.path.join(here, 'extractor',
                                                  'c2v-extract'))
    ap.add_argument('--max_path_length', type=int, default=8)
    ap.add_argument('--max_path_width', type=int, default=2)
    ap.add_argument('--num_threads', type=int, default=32)
    ap.add_argument('--batch_size', type=int, default=3)
    ap.add_argument('--batch_timeout', type=float, default=60.0,
                    help='seconds before a whole batch of dirs is skipped')
    ap.add_argument('--kill_timeout', type=float, default=600000.0,
                    help='per-extractor-process kill timer, seconds')
    ap.add_argument('--ofile_name', default=None)
    args = ap.parse_args()

    out = open(args.ofile_name, 'w') if args.ofile_name else sys.stdout
    try:
        skipped = run(args.bin, args.dir, out,
                      max_path_length=args.max_path_length,
                      max_path_width=args.max_path_width,
                      num_threads=args.num_threads,
                      batch_size=args.batch_size,
                      batch_timeout=args.batch_timeout,
                      kill_timeout=args.kill_timeout)
    finally:
        if out is not sys.stdout:
            out.close()
    if skipped:
        print('skipped %d timed-out dirs: %s'
              % (len(skipped), ' '.join(skipped)), file=sys.stderr)


if __name__ == '__main__':
    main()
