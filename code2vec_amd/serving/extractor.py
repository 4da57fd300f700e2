"""Extractor bridge for the interactive predict shell.

Behavior contract (reference: extractor.py:11-49): run an AST path extractor
over one source file with hashing disabled, then re-hash every path string
with a java.lang.String#hashCode-compatible hash so the model sees the hashed
vocabulary it was trained on while a reverse map keeps the raw path strings
available for display. Lines are truncated to MAX_CONTEXTS contexts and
space-padded like the offline preprocessor output.

The extractor executable is resolved in order:
1. our native C++ extractor `c2v-extract` (extractor/ build output), or
2. a JavaExtractor jar if EXTRACTOR_JAR points at one (reference layout).
"""

import os
import shutil
import subprocess
from typing import Dict, List, Tuple

_REPO_ROOT = os.path.dirname(os.path.dirname(os.path.dirname(
    os.path.abspath(__file__))))


def java_string_hashcode(s: str) -> int:
    """java.lang.String#hashCode: h = 31*h + c over UTF-16 units, wrapping in
    signed 32-bit. The training data's path hashes were produced with this
    (reference extractor.py:40-49), so serving must match it exactly."""
    h = 0
    for ch in s:
        h = (h * 31 + ord(ch)) & 0xFFFFFFFF
    return h - (1 << 32) if h >= (1 << 31) else h


class Extractor:
    def __init__(self, config, jar_path: str = None, max_path_length: int = 8,
                 max_path_width: int = 2):
        self.config = config
        self.max_path_length = max_path_length
        self.max_path_width = max_path_width
        self.max_contexts = config.MAX_CONTEXTS
        self.jar_path = jar_path or os.environ.get('EXTRACTOR_JAR')
        self.native_bin = shutil.which('c2v-extract') or os.path.join(
            _REPO_ROOT, 'extractor', 'c2v-extract')

    def _command(self, path: str) -> List[str]:
        limits = ['--max_path_length', str(self.max_path_length),
                  '--max_path_width', str(self.max_path_width)]
        if os.path.isfile(self.native_bin) and os.access(self.native_bin, os.X_OK):
            return [self.native_bin, '--file', path, '--no_hash'] + limits
        if self.jar_path:
            return (['java', '-cp', self.jar_path, 'JavaExtractor.App']
                    + limits + ['--file', path, '--no_hash'])
        raise RuntimeError('No extractor available: build '
                           'extractor/c2v-extract or set EXTRACTOR_JAR.')

    def _rehash_line(self, line: str,
                     unhash: Dict[str, str]) -> str:
        """One raw extractor line -> model-input line: contexts truncated to
        MAX_CONTEXTS, path strings replaced by their decimal hash (recorded in
        `unhash`), and the line space-padded by the number of dropped-or-
        missing context slots (matching preprocess.py:64-65 output)."""
        fields = line.rstrip().split(' ')
        method_name, contexts = fields[0], fields[1:]
        kept = [method_name]
        for ctx in contexts[:self.max_contexts]:
            triple = ctx.split(',')
            if len(triple) != 3:
                continue
            left, path_str, right = triple
            hashed = str(java_string_hashcode(path_str))
            unhash[hashed] = path_str
            kept.append(f'{left},{hashed},{right}')
        return ' '.join(kept) + ' ' * (self.max_contexts - len(contexts))

    def extract_paths(self, path: str) -> Tuple[List[str], Dict[str, str]]:
        """Returns (model-input lines, hash->path-string dict for display)."""
        proc = subprocess.run(self._command(path), capture_output=True,
                              text=True)
        if proc.returncode != 0:
            raise ValueError(proc.stderr.strip() or 'extractor failed')
        raw_lines = proc.stdout.splitlines()
        if not raw_lines:
            raise ValueError(proc.stderr.strip()
                             or 'extractor produced no output')
        unhash: Dict[str, str] = {}
        lines = [self._rehash_line(line, unhash) for line in raw_lines]
        return lines, unhash
