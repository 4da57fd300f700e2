"""c2v-extract golden tests: output-contract checks for the C++ AST path
extractor against hand-derived expectations for the reference path grammar
(FeatureExtractor.java:120-191, Property.java:23-76)."""

import os
import subprocess

import pytest

ROOT = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
EXTRACTOR_DIR = os.path.join(ROOT, 'extractor')
BIN = os.path.join(EXTRACTOR_DIR, 'c2v-extract')


@pytest.fixture(scope='module')
def extractor():
    if not os.path.isfile(BIN):
        r = subprocess.run(['make', '-C', EXTRACTOR_DIR],
                           capture_output=True, text=True)
        if r.returncode != 0:
            pytest.skip('cannot build c2v-extract: ' + r.stderr[-500:])
    return BIN


def run_extract(extractor, code, tmp_path, extra=()):
    src = tmp_path / 'In.java'
    src.write_text(code)
    cmd = [extractor, '--file', str(src), '--max_path_length', '8',
           '--max_path_width', '2', '--no_hash', *extra]
    out = subprocess.run(cmd, capture_output=True, text=True)
    assert out.returncode == 0, out.stderr
    return out.stdout.strip().split('\n') if out.stdout.strip() else []


def contexts_of(line):
    parts = line.split(' ')
    return parts[0], [c for c in parts[1:] if c]


def test_simple_method_golden(extractor, tmp_path):
    lines = run_extract(extractor, '''
public class T {
    int getX(int y) { return y + 1; }
}''', tmp_path)
    assert len(lines) == 1
    name, ctxs = contexts_of(lines[0])
    assert name == 'get|x'
    # hand-derived expectations
    assert 'int,(PrimitiveType0)^(MethodDeclaration)_(NameExpr1),METHOD_NAME' in ctxs
    assert 'y,(NameExpr0)^(BinaryExpr:plus)_(IntegerLiteralExpr1),1' in ctxs
    # Parameter children order: id BEFORE type (javaparser 3.0.0-alpha.4)
    assert 'y,(VariableDeclaratorId0)^(Parameter)_(PrimitiveType1),int' in ctxs
    # all contexts are triples
    for c in ctxs:
        assert len(c.split(',')) == 3, c


def test_method_name_masked_and_label_split(extractor, tmp_path):
    lines = run_extract(extractor, '''
class A {
    void doSomethingGreat42Now() { int x = 0; }
}''', tmp_path)
    name, ctxs = contexts_of(lines[0])
    # digits split subtokens; parts normalized lowercase
    assert name == 'do|something|great|now'
    assert any('METHOD_NAME' in c for c in ctxs)
    assert not any('dosomething' in c.lower() and 'METHOD_NAME' not in c for c in ctxs)


def test_path_length_and_width_limits(extractor, tmp_path):
    code = '''
class A { int f(int a) { if (a > 0) { if (a > 1) { if (a > 2) { return a; } } } return 0; } }'''
    all_ctxs = set()
    for line in run_extract(extractor, code, tmp_path):
        all_ctxs.update(contexts_of(line)[1])
    short = set()
    for line in run_extract(extractor, code, tmp_path,
                            extra=()):
        short.update(contexts_of(line)[1])
    # with max_path_length 8, deep nesting paths are dropped: fewer contexts
    # than the pair count of leaves
    assert len(all_ctxs) > 0
    # every path respects node count <= 9 (length counts nodes, includes LCA)
    for c in all_ctxs:
        path = c.split(',')[1]
        n_nodes = path.count('(')
        assert n_nodes <= 9, c


def test_operators_and_literals(extractor, tmp_path):
    lines = run_extract(extractor, '''
class A {
  void f() {
    int x = 5;
    x += 2;
    boolean b = x != 64 && !false;
    String s = "hello, world";
    x++;
  }
}''', tmp_path)
    name, ctxs = contexts_of(lines[0])
    joined = ' '.join(ctxs)
    assert 'AssignExpr:plus' in joined
    assert 'BinaryExpr:notEquals' in joined
    assert 'BinaryExpr:and' in joined
    assert 'UnaryExpr:not' in joined
    assert 'UnaryExpr:posIncrement' in joined
    # integer whitelist: 64 kept, 5 and 2 kept as names after normalize?
    # normalizeName("5") -> stripped empty -> careful "5"; Name is "5"
    assert ',64' in joined or '64,' in joined
    # string literal name: quotes and comma stripped, lowercased
    assert 'helloworld' in joined


def test_boxed_types_and_generics(extractor, tmp_path):
    lines = run_extract(extractor, '''
class A {
  Integer f(java.util.List<Integer> xs) { return xs.get(0); }
}''', tmp_path)
    name, ctxs = contexts_of(lines[0])
    joined = ' '.join(ctxs)
    # boxed Integer -> PrimitiveType type with unboxed name
    assert '(PrimitiveType' in joined
    assert 'int,' in joined or ',int' in joined


def test_no_hash_vs_hash(extractor, tmp_path):
    code = 'class A { int f() { return 1; } }'
    unhashed = run_extract(extractor, code, tmp_path)
    src = tmp_path / 'In.java'
    out = subprocess.run([extractor, '--file', str(src), '--max_path_length',
                          '8', '--max_path_width', '2'],
                         capture_output=True, text=True)
    hashed = out.stdout.strip().split('\n')
    _, ctxs_u = contexts_of(unhashed[0])
    _, ctxs_h = contexts_of(hashed[0])
    assert len(ctxs_u) == len(ctxs_h)
    for cu, ch in zip(ctxs_u, ctxs_h):
        pu, ph = cu.split(',')[1], ch.split(',')[1]
        assert pu.startswith('(')
        # hashed path is a 32-bit signed integer string
        int(ph)


def test_java_hashcode_compat(extractor, tmp_path):
    """The hash must equal Java String.hashCode so hashed paths match the
    vocabulary of models trained on the reference pipeline."""
    from code2vec_amd.serving.extractor import java_string_hashcode
    code = 'class A { int f() { return 1; } }'
    unhashed = run_extract(extractor, code, tmp_path)
    src = tmp_path / 'In.java'
    out = subprocess.run([extractor, '--file', str(src), '--max_path_length',
                          '8', '--max_path_width', '2'],
                         capture_output=True, text=True)
    hashed = out.stdout.strip().split('\n')
    _, ctxs_u = contexts_of(unhashed[0])
    _, ctxs_h = contexts_of(hashed[0])
    for cu, ch in zip(ctxs_u, ctxs_h):
        assert str(java_string_hashcode(cu.split(',')[1])) == ch.split(',')[1]


def test_dir_mode_multithreaded(extractor, tmp_path):
    d = tmp_path / 'proj'
    d.mkdir()
    for i in range(10):
        (d / ('F%d.java' % i)).write_text(
            'class F%d { int m%d() { return %d; } }' % (i, i, i))
    out = subprocess.run([extractor, '--dir', str(d), '--max_path_length', '8',
                          '--max_path_width', '2', '--no_hash',
                          '--num_threads', '4'],
                         capture_output=True, text=True)
    lines = [l for l in out.stdout.strip().split('\n') if l]
    assert len(lines) == 10


def test_bare_method_parse_retry(extractor, tmp_path):
    # a bare method body (no class) must parse via the wrap retry
    lines = run_extract(extractor, 'int f(int n) { return n; }', tmp_path)
    assert len(lines) == 1
    assert lines[0].startswith('f ')


def test_extractor_bridge_roundtrip(extractor, tmp_path, monkeypatch):
    """serving.Extractor -> c2v-extract -> model-input lines with re-hashed
    paths and an unhash dict (reference extractor.py:20-49 flow)."""
    from code2vec_amd.config import Config
    from code2vec_amd.serving.extractor import Extractor
    cfg = Config(set_defaults=True)
    cfg.MAX_CONTEXTS = 10
    ex = Extractor(cfg)
    ex.native_bin = extractor
    src = tmp_path / 'Input.java'
    src.write_text('class A { int add(int a, int b) { return a + b; } }')
    lines, unhash = ex.extract_paths(str(src))
    assert len(lines) == 1
    parts = lines[0].split(' ')
    assert parts[0] == 'add'
    ctx = [p for p in parts[1:] if p][0].split(',')
    assert len(ctx) == 3
    assert ctx[1] in unhash           # hashed -> original path string
    assert unhash[ctx[1]].startswith('(')


def test_extractor_thread_pool_tsan(tmp_path):
    """Race detection: the extractor's thread pool under
    ThreadSanitizer must report no data races."""
    r = subprocess.run(['make', '-C', EXTRACTOR_DIR, 'c2v-extract-tsan'],
                       capture_output=True, text=True)
    if r.returncode != 0:
        pytest.skip('tsan build unavailable: ' + r.stderr[-300:])
    d = tmp_path / 'proj'
    d.mkdir()
    for i in range(12):
        (d / ('T%d.java' % i)).write_text(
            'class T%d { int g(int v) { return v * %d; } }' % (i, i))
    out = subprocess.run(
        [os.path.join(EXTRACTOR_DIR, 'c2v-extract-tsan'), '--dir', str(d),
         '--max_path_length', '8', '--max_path_width', '2', '--no_hash',
         '--num_threads', '8'], capture_output=True, text=True)
    assert out.returncode == 0
    assert 'WARNING: ThreadSanitizer' not in out.stderr
    assert len([l for l in out.stdout.splitlines() if l]) == 12


# ---- grammar-edge coverage (round-2: lambdas, anonymous classes, bounded
# generics, nested arrays, switch fallthrough, method refs, varargs, twr) ----

def _paths(lines):
    out = []
    for line in lines:
        _, ctxs = contexts_of(line)
        out.extend(c.split(',')[1] for c in ctxs)
    return out


def test_lambda_positions(extractor, tmp_path):
    """Lambdas in return / initializer / call-arg positions all parse and
    produce LambdaExpr nodes on paths (JavaParser accepts all three)."""
    code = '''
import java.util.function.Function;
import java.util.List;
class In {
    Function<Integer, Integer> fromReturn(int x) { return y -> y * x; }
    void fromInit(int x) {
        Function<Integer, Integer> f = y -> y + x;
        f.apply(3);
    }
    void fromArg(List<Integer> xs) { xs.forEach(v -> System.out.println(v)); }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    names = [contexts_of(l)[0] for l in lines]
    assert names == ['from|return', 'from|init', 'from|arg']
    assert any('LambdaExpr' in p for p in _paths(lines))


def test_anonymous_class_inner_method(extractor, tmp_path):
    """Methods inside anonymous classes are extracted as their own examples
    (JavaParser's visitor descends into ObjectCreationExpr bodies)."""
    code = '''
class In {
    Runnable makeRunner(int x) {
        Runnable r = new Runnable() {
            public void run() { System.out.println(x); }
        };
        return r;
    }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    names = [contexts_of(l)[0] for l in lines]
    assert names == ['make|runner', 'run']
    assert any('ObjectCreationExpr' in p for p in _paths([lines[0]]))


def test_bounded_generics(extractor, tmp_path):
    code = '''
class In {
    <T extends Comparable<T>> T maxOf(T a, T b) {
        return a.compareTo(b) > 0 ? a : b;
    }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    name, ctxs = contexts_of(lines[0])
    assert name == 'max|of'
    assert any('ConditionalExpr' in c for c in ctxs)
    terminals = {p for c in ctxs for p in (c.split(',')[0], c.split(',')[2])}
    assert 't' in terminals  # the type-variable leaf, lowercased


def test_nested_arrays(extractor, tmp_path):
    code = '''
class In {
    int[][] grid = new int[3][4];
    int get(int i, int j) { return grid[i][j]; }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    name, ctxs = contexts_of(lines[0])
    assert name == 'get'
    # nested indexing produces ArrayAccessExpr parents with child ids
    assert any('ArrayAccessExpr' in c for c in ctxs)


def test_switch_fallthrough(extractor, tmp_path):
    code = '''
class In {
    int classify(int v) {
        switch (v) {
            case 0:
            case 1:
                return 10;
            case 2:
                v += 1;
            default:
                return v;
        }
    }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    name, ctxs = contexts_of(lines[0])
    assert name == 'classify'
    assert any('SwitchStmt' in c for c in ctxs)
    assert any('SwitchEntryStmt' in c for c in ctxs)


def test_method_reference_and_varargs(extractor, tmp_path):
    code = '''
import java.util.List;
class In {
    void refs(List<String> xs) { xs.forEach(System.out::println); }
    int sum(int... vals) {
        int s = 0;
        for (int v : vals) s += v;
        return s;
    }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    names = [contexts_of(l)[0] for l in lines]
    assert names == ['refs', 'sum']
    assert any('ForeachStmt' in c for _, l in enumerate(lines)
               for c in contexts_of(l)[1])


def test_try_with_resources_and_string_switch(extractor, tmp_path):
    code = '''
class In {
    String pick(String k) {
        try (java.io.StringReader r = new java.io.StringReader(k)) {
            switch (k) { case "a": return "x"; default: return "y"; }
        }
    }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    name, ctxs = contexts_of(lines[0])
    assert name == 'pick'
    assert any('TryStmt' in c for c in ctxs)


def test_null_literal_everywhere(extractor, tmp_path):
    """Regression: the null-literal parse consumed no token, so ANY file
    containing `null` produced zero output (whole-file loss)."""
    code = '''
class In {
    Object plain() { return null; }
    String concat(Object o) { return "x" + o + null; }
    boolean isNull(Object o) { return o == null; }
    void assign() { Object a = null; a = null; }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    names = [contexts_of(l)[0] for l in lines]
    assert names == ['plain', 'concat', 'is|null', 'assign']
    assert any('NullLiteralExpr' in p for p in _paths(lines))


def test_enum_ctor_inner_class_static_init_labels(extractor, tmp_path):
    """Enums with methods, constructors, static initializers, inner classes
    and labeled break/continue neither crash nor leak non-method members as
    examples (reference extracts MethodDeclarations only)."""
    code = '''
enum Color { RED, GREEN;
    int shade(int base) { return base * 2; } }
class In {
    static int counter;
    static { counter = 5; }
    In(int x) { counter += x; }
    class Inner { int get() { return counter; } }
    void loops() {
        outer:
        for (int i = 0; i < 3; i++) {
            for (int j = 0; j < 3; j++) {
                if (j == 1) continue outer;
                if (i == 2) break outer;
            }
        }
    }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    names = [contexts_of(l)[0] for l in lines]
    assert names == ['shade', 'get', 'loops']
    assert any('LabeledStmt' in p for p in _paths(lines))


def test_instanceof_cast_dowhile_escapes(extractor, tmp_path):
    code = '''
class In {
    boolean check(Object o) { return o instanceof String && ((String) o).isEmpty(); }
    int dw(int n) { int s = 0; do { s += n; n--; } while (n > 0); return s; }
    String esc() { return "a,b\\n\\"c\\""; }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    names = [contexts_of(l)[0] for l in lines]
    assert names == ['check', 'dw', 'esc']
    assert any('InstanceOfExpr' in p for p in _paths(lines))
    assert any('DoStmt' in p for p in _paths(lines))
    # string-literal tokens are normalized: no raw commas can corrupt the
    # ctx1,ctx2,ctx3 wire format
    for _, ctxs in map(contexts_of, lines):
        for c in ctxs:
            assert c.count(',') == 2, c


def test_empty_and_bodyless_methods_skipped(extractor, tmp_path):
    """Reference parity: empty bodies have method length 0 and min_code_len
    is 1 (FunctionVisitor.getMethodLength + ExtractFeaturesTask filter), so
    `void e() {}` and abstract/native declarations yield no example."""
    code = '''
abstract class In {
    void empty() { }
    native void nat();
    abstract int abs(int x);
    int real() { return 1; }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    assert [contexts_of(l)[0] for l in lines] == ['real']


def test_deep_generics_multicatch_array_inits(extractor, tmp_path):
    code = '''
import java.util.*;
class In {
    Map<String, List<Map<Integer, String>>> deep() { return new HashMap<>(); }
    void multicatch() {
        try { deep(); } catch (RuntimeException | Error e) { e.toString(); }
    }
    int[][] init2() { int[][] a = {{1, 2}, {3}}; return a; }
    void bounds(List<? super Integer> l, List<? extends Number> u) { l.add(1); }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    # note: normalizeName/subtoken split drops digits -> init2 reads "init"
    assert [contexts_of(l)[0] for l in lines] == ['deep', 'multicatch',
                                                  'init', 'bounds']
    assert any('ArrayInitializerExpr' in p for p in _paths(lines))


def test_unknown_construct_costs_only_its_method(extractor, tmp_path):
    """Per-member error recovery: a grammar gap (here a Java-14 switch
    expression) drops only the containing method; siblings survive. The
    3-stage retry still prefers stages that parse cleanly."""
    code = '''
class In {
    int good1() { return 1; }
    void weird() { int x = switch (1) { case 1 -> 2; default -> 3; }; }
    int good2() { return 2; }
}
'''
    lines = run_extract(extractor, code, tmp_path)
    assert [contexts_of(l)[0] for l in lines] == ['good', 'good']


def test_advanced_constructs_all_extract(extractor, tmp_path):
    """Default interface methods, enum bodies (ctor/field/method), static
    initializers, try-with-resources, multi-catch rethrow, method
    references (instance and static), string switch and block-body
    lambdas must all parse — every named method in the file produces an
    output line (javaparser-3.0.0-alpha.4 handles all of these, so a
    skip here would be a fidelity gap vs the reference)."""
    code = '''
import java.io.*;
import java.util.function.*;
public class B {
    interface Op { default int apply(int a) { return a + 1; } }
    enum E { ONE(1), TWO(2); final int v; E(int v) { this.v = v; }
             int get() { return v; } }
    static int[] table = new int[]{1, 2, 3};
    static { table[0] = 9; }
    public String twr(File f) throws IOException {
        try (BufferedReader r = new BufferedReader(new FileReader(f))) {
            return r.readLine();
        } catch (IOException | RuntimeException e) { throw e;
        } finally { System.gc(); }
    }
    public IntSupplier mref() { return table.length > 0 ? this::len : B::slen; }
    private int len() { return table.length; }
    private static int slen() { return 3; }
    public int strSwitch(String s) {
        switch (s) { case "a": return 1; default: return 0; }
    }
    public Function<Integer, Integer> lamBlock() {
        return x -> { int y = x * 2; return y + 1; };
    }
}
'''
    names = [contexts_of(l)[0] for l in run_extract(extractor, code, tmp_path)]
    assert names == ['apply', 'get', 'twr', 'mref', 'len', 'slen',
                     'str|switch', 'lam|block'], names


def test_post_java8_method_skipped_others_survive(extractor, tmp_path):
    """Syntax beyond the reference parser's generation (Java 14 switch
    expressions, var) costs only the containing method via per-member
    recovery — sibling methods still extract, matching the reference's
    behavior of failing only what javaparser-alpha.4 cannot parse."""
    code = '''
public class C {
    public int modern(int k) {
        var x = switch (k) { case 1 -> 10; default -> 0; };
        return x;
    }
    public int classic(int k) { return k + 1; }
}
'''
    names = [contexts_of(l)[0] for l in run_extract(extractor, code, tmp_path)]
    assert names == ['classic'], names


def test_labeled_loops_wildcards_unicode(extractor, tmp_path):
    """Labeled break/continue, bounded wildcards with cast, bit-shift
    chains, char literals and UNICODE identifiers (Java permits them;
    the lexer folds multi-byte UTF-8 into the identifier token) all
    extract."""
    code = '''
public class D {
    public int labeled(int[][] m) {
        int s = 0;
        outer:
        for (int i = 0; i < m.length; i++) {
            for (int j = 0; j < m[i].length; j++) {
                if (m[i][j] < 0) continue outer;
                if (m[i][j] == 99) break outer;
                s += m[i][j];
            }
        }
        return s;
    }
    public char charLit(String s) { return s.isEmpty() ? '\\0' : s.charAt(0); }
    public boolean inst(Object o) {
        return o instanceof String && ((String) o).length() > 2;
    }
    @SuppressWarnings({"unchecked", "rawtypes"})
    public java.util.List<? extends Number> wild(
            java.util.List<? super Integer> in) {
        return (java.util.List) in;
    }
    public long bitops(long a, long b) { return (a << 3) | (b >>> 2) ^ ~a & b; }
    public double unicodeId() { double π = 3.14159; return π; }
}
'''
    names = [contexts_of(l)[0] for l in run_extract(extractor, code, tmp_path)]
    assert names == ['labeled', 'char|lit', 'inst', 'wild', 'bitops',
                     'unicode|id'], names


def test_member_kind_exclusions_and_generics(extractor, tmp_path):
    """Instance initializers, generic constructors, interface/abstract/
    native methods (no body) produce NO lines — the reference visits
    only MethodDeclaration nodes with bodies (FunctionVisitor.java:25-55)
    — while class literals, static imports, conditional array
    initializers, generic methods with throws, and strictfp all
    extract."""
    code = '''
import static java.lang.Math.max;
public class E {
    { counter = 1; }
    int counter;
    <T> E(T seed) { counter = seed.hashCode(); }
    interface Cb { void fire(Class<?> cls); }
    public Class<?> classLit() { return String.class; }
    public int statImport(int a, int b) { return max(a, b); }
    public int[] arrInit() { return new int[]{1 > 0 ? 1 : 2, 3}; }
    public <K, V> java.util.Map<K, V> genMethod(K k, V v)
            throws IllegalStateException {
        java.util.Map<K, V> m = new java.util.HashMap<>();
        m.put(k, v);
        return m;
    }
    public strictfp double sfp(double d) { return d * 2.0; }
    public final native int nat(int x);
    public abstract static class Inner { public abstract int go(); }
}
'''
    names = [contexts_of(l)[0] for l in run_extract(extractor, code, tmp_path)]
    assert names == ['class|lit', 'stat|import', 'arr|init', 'gen|method',
                     'sfp'], names
