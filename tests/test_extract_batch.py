"""Fault-tolerance semantics of the batch extraction driver
(code2vec_amd/data/extract_batch.py vs reference JavaExtractor/extract.py):
failed dirs retry per-subdirectory with partial output dropped; stuck
batches are skipped on timeout."""

import io
import os
import stat

from code2vec_amd.data import extract_batch


def make_fake_extractor(tmp_path):
    """Extractor stand-in: emits one line per .java file found directly in
    --dir (non-recursive, so the retry recursion is observable); exits 1
    when the directory name contains 'poison'; sleeps when it contains
    'slow'."""
    script = tmp_path / 'fake-extract'
    script.write_text('''#!/usr/bin/env bash
dir=""
while [ $# -gt 0 ]; do
  if [ "$1" = "--dir" ]; then dir="$2"; shift; fi
  shift
done
base=$(basename "$dir")
case "$base" in
  slow) sleep 60 ;;
esac
for f in "$dir"/*.java; do
  [ -e "$f" ] && echo "method_from_$base ctx,1,ctx"
done
case "$base" in
  poison) exit 1 ;;
esac
exit 0
''')
    script.chmod(script.stat().st_mode | stat.S_IEXEC)
    return str(script)


def make_corpus(tmp_path, with_slow=False):
    root = tmp_path / 'corpus'
    (root / 'good').mkdir(parents=True)
    (root / 'good' / 'A.java').write_text('class A {}')
    (root / 'poison').mkdir()
    (root / 'poison' / 'Bad.java').write_text('class Bad {}')
    (root / 'poison' / 'sub1').mkdir()
    (root / 'poison' / 'sub1' / 'S1.java').write_text('class S1 {}')
    (root / 'poison' / 'sub2').mkdir()
    (root / 'poison' / 'sub2' / 'S2.java').write_text('class S2 {}')
    if with_slow:
        (root / 'slow').mkdir()
        (root / 'slow' / 'Z.java').write_text('class Z {}')
    return str(root)


def test_failed_dir_retries_subdirs_and_drops_partial(tmp_path):
    bin_path = make_fake_extractor(tmp_path)
    root = make_corpus(tmp_path)
    out = io.StringIO()
    skipped = extract_batch.run(bin_path, root, out, batch_timeout=30,
                                pool_size=2)
    lines = out.getvalue().strip().split('\n')
    assert skipped == []
    # good dir extracted; poison's own (partial) output dropped; its two
    # subdirs retried and extracted
    assert 'method_from_good ctx,1,ctx' in lines
    assert 'method_from_sub1 ctx,1,ctx' in lines
    assert 'method_from_sub2 ctx,1,ctx' in lines
    assert not any('from_poison' in l for l in lines)


def test_stuck_batch_is_skipped(tmp_path):
    bin_path = make_fake_extractor(tmp_path)
    root = make_corpus(tmp_path, with_slow=True)
    out = io.StringIO()
    # batch_size=1 so only the slow dir's batch times out
    skipped = extract_batch.run(bin_path, root, out, batch_size=1,
                                batch_timeout=5, pool_size=2)
    lines = out.getvalue()
    assert any('slow' in s for s in skipped)
    assert 'method_from_good' in lines
    assert 'method_from_sub1' in lines
    assert 'method_from_slow' not in lines


def test_single_dir_corpus_without_subdirs(tmp_path):
    bin_path = make_fake_extractor(tmp_path)
    root = tmp_path / 'flat'
    root.mkdir()
    (root / 'F.java').write_text('class F {}')
    out = io.StringIO()
    extract_batch.run(bin_path, str(root), out, batch_timeout=30)
    assert 'method_from_flat' in out.getvalue()
