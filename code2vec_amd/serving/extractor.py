"""Extractor bridge for the interactive predict shell.

Runs an AST path extractor over a single source file and re-hashes the path
strings with a Java-`String#hashCode`-compatible hash so the displayed paths
stay human-readable while the model sees the hashed vocabulary it was trained
on (reference: extractor.py:12-49).

The extractor executable is resolved in order:
1. our native C++ extractor `c2v-extract` (extractor/ build output), or
2. a JavaExtractor jar if EXTRACTOR_JAR points at one (reference layout).
"""

import os
import shutil
import subprocess
from typing import Dict, List, Tuple




def java_string_hashcode(s: str) -> int:
    """Reimplementation of java.lang.String#hashCode (32-bit wrapping), the
    hash the reference training data was produced with (extractor.py:40-49)."""
    h = 0
    for ch in s:
        h = (31 * h + ord(ch)) & 0xFFFFFFFF
    h = h & 0xFFFFFFFF
    return h - 0x100000000 if h > 0x7FFFFFFF else h


class Extractor:
    def __init__(self, config, jar_path: str = None, max_path_length: int = 8,
                 max_path_width: int = 2):
        self.config = config
        self.max_path_length = max_path_length
        self.max_path_width = max_path_width
        self.max_contexts = config.MAX_CONTEXTS
        self.jar_path = jar_path or os.environ.get('EXTRACTOR_JAR')
        self.native_bin = shutil.which('c2v-extract') or os.path.join(
            os.path.dirname(os.path.dirname(os.path.dirname(os.path.abspath(__file__)))),
            'extractor', 'c2v-extract')

    def _command(self, path: str) -> List[str]:
        if os.path.isfile(self.native_bin) and os.access(self.native_bin, os.X_OK):
            return [self.native_bin, '--file', path, '--no_hash',
                    '--max_path_length', str(self.max_path_length),
                    '--max_path_width', str(self.max_path_width)]
        if self.jar_path:
            return ['java', '-cp', self.jar_path, 'JavaExtractor.App',
                    '--max_path_length', str(self.max_path_length),
                    '--max_path_width', str(self.max_path_width),
                    '--file', path, '--no_hash']
        raise RuntimeError(
            'No extractor available: build extractor/c2v-extract or set EXTRACTOR_JAR.')

    def extract_paths(self, path: str) -> Tuple[List[str], Dict[int, str]]:
        """Returns (model-input lines with hashed paths truncated to
        MAX_CONTEXTS, hash→path-string dict for display)."""
        out = subprocess.run(self._command(path), capture_output=True, text=True)
        if out.returncode != 0:
            raise ValueError(out.stderr.strip() or 'extractor failed')
        output = out.stdout.splitlines()
        if not output:
            raise ValueError(out.stderr.strip() or 'extractor produced no output')
        hash_to_string_dict: Dict[int, str] = {}
        result: List[str] = []
        for line in output:
            parts = line.rstrip().split(' ')
            method_name = parts[0]
            current_result_line_parts = [method_name]
            contexts = parts[1:]
            for context in contexts[:self.max_contexts]:
                context_parts = context.split(',')
                if len(context_parts) != 3:
                    continue
                context_word1, context_path, context_word2 = context_parts
                hashed_path = str(java_string_hashcode(context_path))
                hash_to_string_dict[hashed_path] = context_path
                current_result_line_parts.append(
                    '%s,%s,%s' % (context_word1, hashed_path, context_word2))
            space_padding = ' ' * (self.max_contexts - len(contexts))
            result.append(' '.join(current_result_line_parts) + space_padding)
        return result, hash_to_string_dict
