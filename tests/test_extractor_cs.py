"""c2v-extract-cs contract tests: Roslyn-kind path grammar, variable
grouping, METHOD_NAME masking, comment contexts, normalization
(reference CSharpExtractor semantics; divergences in c2v_extract_cs.cpp
header)."""

import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EXTRACTOR_DIR = os.path.join(ROOT, 'extractor')
BIN = os.path.join(EXTRACTOR_DIR, 'c2v-extract-cs')


@pytest.fixture(scope='module')
def cs_extractor():
    if not os.path.isfile(BIN):
        r = subprocess.run(['make', '-C', EXTRACTOR_DIR, 'c2v-extract-cs'],
                           capture_output=True, text=True)
        if r.returncode != 0:
            pytest.skip('cannot build c2v-extract-cs: ' + r.stderr[-400:])
    return BIN


def run_cs(cs_extractor, code, tmp_path, extra=('--no_hash',)):
    src = tmp_path / 'T.cs'
    src.write_text(code)
    out = subprocess.run([cs_extractor, '--path', str(src), *extra],
                         capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    return [l for l in out.stdout.splitlines() if l]


def test_basic_method(cs_extractor, tmp_path):
    lines = run_cs(cs_extractor, '''
class C {
    public int AddValue(int x) { return x + 1; }
}''', tmp_path)
    assert len(lines) == 1
    parts = lines[0].split(' ')
    assert parts[0] == 'add|value'
    joined = ' '.join(parts[1:])
    assert 'METHOD_NAME' in joined
    assert 'AddExpression' in joined
    assert 'PredefinedType^MethodDeclaration' in joined
    # numeric whitelist keeps 1
    assert ',1' in joined or '1,' in joined


def test_member_access_child_ids(cs_extractor, tmp_path):
    lines = run_cs(cs_extractor, '''
class C {
    void Go(string s) { s.Trim(); }
}''', tmp_path)
    joined = lines[0]
    # member-access / invocation parents add child ids
    assert 'SimpleMemberAccessExpression0' in joined or \
           'IdentifierName0' in joined


def test_comments_become_contexts(cs_extractor, tmp_path):
    lines = run_cs(cs_extractor, '''
class C {
    // compute the running total now
    int F(int a) { return a; }
}''', tmp_path)
    joined = lines[0]
    assert ',COMMENT,' in joined
    assert 'compute|the|running|total|now' in joined


def test_var_not_a_leaf(cs_extractor, tmp_path):
    lines = run_cs(cs_extractor, '''
class C {
    int F() { var q = 3; return q; }
}''', tmp_path)
    joined = lines[0]
    assert ',var' not in joined and 'var,' not in joined


def test_hash_mode_is_deterministic_int(cs_extractor, tmp_path):
    code = 'class C { int F(int a) { return a; } }'
    hashed1 = run_cs(cs_extractor, code, tmp_path, extra=())
    hashed2 = run_cs(cs_extractor, code, tmp_path, extra=())
    assert hashed1 == hashed2
    ctxs = [c for c in hashed1[0].split(' ')[1:] if c and 'COMMENT' not in c]
    for c in ctxs:
        int(c.split(',')[1])  # 32-bit int hash


def test_dir_mode(cs_extractor, tmp_path):
    d = tmp_path / 'proj'
    d.mkdir()
    for i in range(6):
        (d / ('F%d.cs' % i)).write_text(
            'class F%d { int M%d(int v) { return v * %d; } }' % (i, i, i))
    out = subprocess.run([cs_extractor, '--path', str(d), '--no_hash',
                          '--threads', '4'], capture_output=True, text=True)
    lines = [l for l in out.stdout.splitlines() if l]
    assert len(lines) == 6


def test_properties_and_interpolated_strings_dont_break_file(cs_extractor,
                                                             tmp_path):
    """Files with property accessors and interpolated strings (documented
    dialect gaps) must still yield their ordinary methods — a dialect gap
    may cost its construct, never the whole file. (The reference's Roslyn
    extractor enumerates MethodDeclarationSyntax, so property accessors are
    not method examples there either.)"""
    code = '''
using System;
class P {
    private int _x;
    public int X {
        get { return _x + 1; }
        set { _x = value; }
    }
    public string Name { get; set; }
    public int Add(int a, int b) { return a + b; }
    public string Greet(string who) { return $"hello {who} ({_x})"; }
}
'''
    src = tmp_path / 'P.cs'
    src.write_text(code)
    import subprocess
    out = subprocess.run([cs_extractor, '--path', str(src), '--max_length',
                          '8', '--max_width', '2', '--no_hash'],
                         capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    names = [l.split(' ')[0] for l in out.stdout.strip().split('\n') if l]
    assert names == ['add', 'greet']


def test_csharp_idiom_sweep(cs_extractor, tmp_path):
    """Null literals, string switch, collection initializers, lambdas,
    expression-bodied members, generic constraints, null-coalescing and
    nullable types all extract (one example per method)."""
    code = '''
using System;
using System.Collections.Generic;
class Q {
    object N() { return null; }
    int Sw(string k) { switch (k) { case "a": return 1; default: return 0; } }
    List<int> Gen() { var l = new List<int> { 1, 2 }; return l; }
    int Lam(Func<int,int> f) { return f(3); }
    void Use() { Lam(x => x * 2); }
    int Tern(int a) => a > 0 ? 1 : 0;
    T Cast<T>(object o) where T : class { return o as T; }
    int Nullc(int? v) { return v ?? -1; }
}
'''
    src = tmp_path / 'Q.cs'
    src.write_text(code)
    import subprocess
    out = subprocess.run([cs_extractor, '--path', str(src), '--max_length',
                          '8', '--max_width', '2', '--no_hash'],
                         capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    names = [l.split(' ')[0] for l in out.stdout.strip().split('\n') if l]
    assert names == ['n', 'sw', 'gen', 'lam', 'use', 'tern', 'cast', 'nullc']


def test_cs_advanced_constructs(cs_extractor, tmp_path):
    """Generic constraints, expression-bodied methods, using statements,
    try/finally and nested classes all extract; operators, property
    accessors and events produce NO lines — the reference's Roslyn
    walker only takes MethodDeclarationSyntax (Extractor.cs), so those
    members are correctly excluded rather than mis-extracted."""
    lines = run_cs(cs_extractor, '''
using System;
using System.Collections.Generic;
namespace N {
    public class P<T> where T : IComparable<T> {
        public event EventHandler Changed;
        public int Count { get; private set; }
        public static P<T> operator +(P<T> a, P<T> b) { return a; }
        public int ExprBody(int x) => x * 2 + 1;
        public T Max(List<T> items) {
            T best = items[0];
            foreach (var it in items) if (it.CompareTo(best) > 0) best = it;
            return best;
        }
        public void UsingStmt() {
            using (var d = new System.IO.MemoryStream()) { d.WriteByte(1); }
        }
        public int TryFin(int k) {
            try { return 10 / k; } catch (DivideByZeroException) { return 0; }
            finally { Count++; }
        }
        private class Inner { public int Val() { return 7; } }
    }
}
''', tmp_path)
    names = [l.split(' ')[0] for l in lines]
    assert names == ['expr|body', 'max', 'using|stmt', 'try|fin', 'val'], names


def test_cs_unicode_identifier(cs_extractor, tmp_path):
    """Unicode identifiers lex as single tokens (Roslyn permits them);
    without this the C# parser — which has file-level, not per-member,
    error recovery — lost the entire file."""
    lines = run_cs(cs_extractor, '''
public class U {
    public double UniId() { double π = 3.14; return π; }
    public int Plain(int x) { return x + 1; }
}
''', tmp_path)
    names = [l.split(' ')[0] for l in lines]
    assert names == ['uni|id', 'plain'], names
