// c2v-extract-cs — C++ AST path-context extractor for C# sources.
//
// Reimplements the reference CSharpExtractor (Roslyn-based, C#) output
// contract: per method, `sub|tokens name,hash(path),name ...` where paths
// walk Roslyn syntax NODES (tokens excluded) between variable-leaf tokens.
// Semantics reproduced from /root/reference/CSharpExtractor/:
// - leaf tokens: identifiers, numeric/string/char literals, and tokens whose
//   parent is PredefinedType; `var` in local declarations excluded
//   (Tree.cs Leaf.IsLeafToken)
// - variables group leaves by value-text; the method-name identifier maps to
//   METHOD_NAME (Variable.cs:66-104)
// - pairs = Choose2(variables) ++ self-pairs, reservoir-sampled to
//   --max_contexts (default 30000) (Extractor.cs:111-138)
// - path: LCA over token PARENT nodes; length = depth(l.P)+depth(r.P)
//   -2*depth(LCA)+2 <= max_length (default 9); width = |childIdx(left
//   divergent) - childIdx(right divergent)| < max_width (default 2)
//   (PathFinder.cs:82-109)
// - path string: Kind(^Kind)*^Ancestor(_Kind)* with child ids (truncated at
//   3) appended under {SimpleAssignmentExpression, ElementAccessExpression,
//   SimpleMemberAccessExpression, InvocationExpression,
//   BracketedArgumentList, ArgumentList} (Extractor.cs:23-97)
// - names: subtoken split/normalize with NUM whitelist {0,1,2,3,4,5,10},
//   SPACE/BLANK fallbacks (Utilities.cs, Extractor.SplitNameUnlessEmpty)
// - whole-file comments appended per method as `batch,COMMENT,batch` in
//   5-subtoken batches (Extractor.cs:204-218 — including the reference's
//   file-scope behavior)
//
// DOCUMENTED DIVERGENCES: the reference hashes with .NET String.GetHashCode,
// which modern .NET randomizes per process — exact hash compatibility is
// impossible even between two runs of the reference. We use the classic
// deterministic .NET Framework 32-bit string hash. Reservoir sampling uses a
// fixed-seed RNG (the reference is time-seeded).
//
// Build: g++ -O2 -std=c++17 -pthread -o c2v-extract-cs c2v_extract_cs.cpp

#include <algorithm>
#include <atomic>
#include <cstdint>
#include <cstring>
#include <deque>
#include <fstream>
#include <iostream>
#include <map>
#include <random>
#include <set>
#include <sstream>
#include <string>
#include <thread>
#include <vector>

#include <dirent.h>
#include <sys/stat.h>

// ---------------------------------------------------------------------------
// Lexer (C#)
// ---------------------------------------------------------------------------

enum class Tok { End, Ident, Keyword, NumLit, StrLit, CharLit, Punct };

struct Token {
  Tok kind = Tok::End;
  std::string text;   // raw text (identifier name / literal value-text)
};

static const std::set<std::string> kCsKeywords = {
    "abstract", "as", "base", "bool", "break", "byte", "case", "catch",
    "char", "checked", "class", "const", "continue", "decimal", "default",
    "delegate", "do", "double", "else", "enum", "event", "explicit", "extern",
    "false", "finally", "fixed", "float", "for", "foreach", "goto", "if",
    "implicit", "in", "int", "interface", "internal", "is", "lock", "long",
    "namespace", "new", "null", "object", "operator", "out", "override",
    "params", "private", "protected", "public", "readonly", "ref", "return",
    "sbyte", "sealed", "short", "sizeof", "stackalloc", "static", "string",
    "struct", "switch", "this", "throw", "true", "try", "typeof", "uint",
    "ulong", "unchecked", "unsafe", "ushort", "using", "virtual", "void",
    "volatile", "while"};

static const std::set<std::string> kPredefined = {
    "bool", "byte", "sbyte", "char", "decimal", "double", "float", "int",
    "uint", "long", "ulong", "object", "short", "ushort", "string", "void"};

struct CsParseError : std::runtime_error {
  explicit CsParseError(const std::string& m) : std::runtime_error(m) {}
};

struct Comment {
  std::string text;
};

class CsLexer {
 public:
  CsLexer(const std::string& src, std::vector<Comment>* comments)
      : s_(src), comments_(comments) {
    advance();
  }
  const Token& cur() const { return cur_; }
  const Token& peek() {
    if (!has_peek_) {
      peek_ = lex();
      has_peek_ = true;
    }
    return peek_;
  }
  void advance() {
    if (has_peek_) {
      cur_ = peek_;
      has_peek_ = false;
    } else {
      cur_ = lex();
    }
  }
  void split_gt() {
    if (cur_.kind == Tok::Punct && cur_.text.size() > 1 && cur_.text[0] == '>')
      cur_.text.erase(cur_.text.begin());
    else
      advance();
  }

 private:
  Token lex() {
    skip_ws_comments();
    Token t;
    if (i_ >= s_.size()) return t;
    char c = s_[i_];
    // bytes >= 0x80 are UTF-8 sequences: C# (like Roslyn) permits unicode
    // identifiers, and with no per-member recovery here a lex failure
    // would cost the whole file
    if (isalpha((unsigned char)c) || c == '_' || c == '@' ||
        (unsigned char)c >= 0x80) {
      size_t j = i_ + (c == '@' ? 1 : 0);
      size_t b = j;
      while (j < s_.size() && (isalnum((unsigned char)s_[j]) || s_[j] == '_' ||
                               (unsigned char)s_[j] >= 0x80))
        ++j;
      t.text = s_.substr(b, j - b);
      t.kind = (c != '@' && kCsKeywords.count(t.text)) ? Tok::Keyword : Tok::Ident;
      i_ = j;
      return t;
    }
    if (isdigit((unsigned char)c) ||
        (c == '.' && i_ + 1 < s_.size() && isdigit((unsigned char)s_[i_ + 1]))) {
      size_t j = i_;
      while (j < s_.size() && (isalnum((unsigned char)s_[j]) || s_[j] == '.' ||
                               ((s_[j] == '+' || s_[j] == '-') && j > i_ &&
                                (s_[j - 1] == 'e' || s_[j - 1] == 'E'))))
        ++j;
      t.kind = Tok::NumLit;
      t.text = s_.substr(i_, j - i_);
      // strip numeric suffixes for value-text-ish behavior
      while (!t.text.empty() && isalpha((unsigned char)t.text.back()))
        t.text.pop_back();
      i_ = j;
      return t;
    }
    if (c == '"' || (c == '$' && i_ + 1 < s_.size() && s_[i_ + 1] == '"') ||
        (c == '@' && i_ + 1 < s_.size() && s_[i_ + 1] == '"')) {
      if (c != '"') ++i_;  // $ / @ prefix
      return lex_string();
    }
    if (c == '\'') return lex_char();
    static const char* ops[] = {"?\?=", "<<=", ">>=", "=>", "??", "?.", "++",
                                "--", "&&", "||", "==", "!=", "<=", ">=",
                                "+=", "-=", "*=", "/=", "%=", "&=", "|=",
                                "^=", "<<", ">>", "::"};
    for (const char* op : ops) {
      size_t n = strlen(op);
      if (s_.compare(i_, n, op) == 0) {
        t.kind = Tok::Punct;
        t.text = op;
        i_ += n;
        return t;
      }
    }
    t.kind = Tok::Punct;
    t.text = std::string(1, c);
    ++i_;
    return t;
  }

  Token lex_string() {
    size_t j = i_ + 1;
    std::string v;
    while (j < s_.size() && s_[j] != '"') {
      if (s_[j] == '\\' && j + 1 < s_.size()) {
        v += s_[j + 1];  // crude unescape to mimic ValueText
        j += 2;
      } else {
        v += s_[j];
        ++j;
      }
    }
    ++j;
    Token t;
    t.kind = Tok::StrLit;
    t.text = v;
    i_ = j;
    return t;
  }

  Token lex_char() {
    size_t j = i_ + 1;
    std::string v;
    while (j < s_.size() && s_[j] != '\'') {
      if (s_[j] == '\\' && j + 1 < s_.size()) {
        v += s_[j + 1];
        j += 2;
      } else {
        v += s_[j];
        ++j;
      }
    }
    ++j;
    Token t;
    t.kind = Tok::CharLit;
    t.text = v;
    i_ = j;
    return t;
  }

  void skip_ws_comments() {
    for (;;) {
      while (i_ < s_.size() && isspace((unsigned char)s_[i_])) ++i_;
      if (i_ + 1 < s_.size() && s_[i_] == '/' && s_[i_ + 1] == '/') {
        size_t b = i_;
        while (i_ < s_.size() && s_[i_] != '\n') ++i_;
        if (comments_) comments_->push_back({s_.substr(b, i_ - b)});
        continue;
      }
      if (i_ + 1 < s_.size() && s_[i_] == '/' && s_[i_ + 1] == '*') {
        size_t b = i_;
        i_ += 2;
        while (i_ + 1 < s_.size() && !(s_[i_] == '*' && s_[i_ + 1] == '/')) ++i_;
        i_ = std::min(i_ + 2, s_.size());
        if (comments_) comments_->push_back({s_.substr(b, i_ - b)});
        continue;
      }
      if (i_ < s_.size() && s_[i_] == '#') {  // preprocessor directive
        while (i_ < s_.size() && s_[i_] != '\n') ++i_;
        continue;
      }
      break;
    }
  }

  const std::string& s_;
  size_t i_ = 0;
  Token cur_, peek_;
  bool has_peek_ = false;
  std::vector<Comment>* comments_;
};

// ---------------------------------------------------------------------------
// AST (Roslyn-kind nodes + leaf tokens)
// ---------------------------------------------------------------------------

struct CsNode;

struct CsLeaf {         // a Roslyn SyntaxToken that passed IsLeafToken
  std::string text;     // ValueText
  CsNode* parent = nullptr;
  bool is_method_name = false;
};

struct CsNode {
  std::string kind;                  // SyntaxKind.ToString()
  CsNode* parent = nullptr;
  std::vector<CsNode*> kids;         // ChildNodes() (no tokens)
  std::vector<CsLeaf*> tokens;       // leaf ChildTokens()
  int depth = 0;
  std::string method_name;           // MethodDeclaration only
};

struct CsAst {
  std::deque<CsNode> nodes;
  std::deque<CsLeaf> leaves;
  CsNode* mk(const std::string& kind) {
    nodes.emplace_back();
    nodes.back().kind = kind;
    return &nodes.back();
  }
  void add(CsNode* p, CsNode* c) {
    if (!c) return;
    c->parent = p;
    p->kids.push_back(c);
  }
  CsLeaf* tok(CsNode* p, const std::string& text, bool method_name = false) {
    leaves.emplace_back();
    CsLeaf* l = &leaves.back();
    l->text = text;
    l->parent = p;
    l->is_method_name = method_name;
    p->tokens.push_back(l);
    return l;
  }
};

// ---------------------------------------------------------------------------
// Parser (recursive descent over the practical C# subset)
// ---------------------------------------------------------------------------

class CsParser {
 public:
  CsParser(const std::string& src, CsAst& ast, std::vector<Comment>* comments)
      : lx_(src, comments), ast_(ast) {}

  CsNode* parse_compilation_unit() {
    CsNode* cu = ast_.mk("CompilationUnit");
    while (!at_end()) {
      if (is_kw("using")) {  // using directive — skip to ';'
        while (!is_punct(";") && !at_end()) lx_.advance();
        if (is_punct(";")) lx_.advance();
        continue;
      }
      skip_attributes_modifiers();
      if (is_kw("namespace")) {
        lx_.advance();
        CsNode* ns = ast_.mk("NamespaceDeclaration");
        while (!is_punct("{") && !at_end()) lx_.advance();
        expect("{");
        while (!is_punct("}") && !at_end()) {
          skip_attributes_modifiers();
          CsNode* d = parse_type_declaration();
          if (d) ast_.add(ns, d);
        }
        expect("}");
        ast_.add(cu, ns);
        continue;
      }
      CsNode* d = parse_type_declaration();
      if (d) ast_.add(cu, d);
      else if (!at_end()) lx_.advance();  // skip stray tokens robustly
    }
    return cu;
  }

 private:
  bool at_end() const { return lx_.cur().kind == Tok::End; }
  bool is_punct(const char* p) const {
    return lx_.cur().kind == Tok::Punct && lx_.cur().text == p;
  }
  bool is_kw(const char* k) const {
    return lx_.cur().kind == Tok::Keyword && lx_.cur().text == k;
  }
  bool is_ident() const { return lx_.cur().kind == Tok::Ident; }
  std::string take() {
    std::string t = lx_.cur().text;
    lx_.advance();
    return t;
  }
  void expect(const char* p) {
    if (!is_punct(p))
      throw CsParseError("expected " + std::string(p) + " got " + lx_.cur().text);
    lx_.advance();
  }
  void skip_balanced(const char* o, const char* c) {
    int depth = 0;
    if (is_punct(o)) { ++depth; lx_.advance(); }
    while (depth > 0 && !at_end()) {
      if (is_punct(o)) ++depth;
      else if (is_punct(c)) --depth;
      lx_.advance();
    }
  }
  void skip_attributes_modifiers() {
    static const std::set<std::string> mods = {
        "public", "private", "protected", "internal", "static", "readonly",
        "sealed", "abstract", "virtual", "override", "const", "extern",
        "unsafe", "volatile", "async", "partial", "new"};
    for (;;) {
      if (is_punct("[")) { skip_balanced("[", "]"); continue; }
      if ((lx_.cur().kind == Tok::Keyword && mods.count(lx_.cur().text)) ||
          (lx_.cur().kind == Tok::Ident &&
           (lx_.cur().text == "async" || lx_.cur().text == "partial"))) {
        lx_.advance();
        continue;
      }
      break;
    }
  }

  CsNode* parse_type_declaration() {
    if (is_kw("class") || is_kw("struct") || is_kw("interface")) {
      const bool is_iface = is_kw("interface");
      const char* kind = is_kw("class") ? "ClassDeclaration"
                         : is_kw("struct") ? "StructDeclaration"
                                           : "InterfaceDeclaration";
      lx_.advance();
      CsNode* cd = ast_.mk(kind);
      take();  // name (class name is a token, not a leaf per IsLeafToken? it
               // IS an IdentifierToken → leaf; Roslyn: identifier token of
               // the declaration)
      // NOTE: the class-name token is an IdentifierToken child of the class
      // node; it IS a leaf in the reference's per-method trees only when the
      // method tree includes it — method subtrees never do. Skip storing it.
      if (is_punct("<")) skip_balanced("<", ">");
      if (is_punct(":")) {  // base list
        lx_.advance();
        parse_type();
        while (is_punct(",")) { lx_.advance(); parse_type(); }
      }
      while (is_ident() && lx_.cur().text == "where") {  // constraints
        while (!is_punct("{") && !at_end()) lx_.advance();
      }
      expect("{");
      while (!is_punct("}") && !at_end()) {
        CsNode* m = parse_member();
        if (m) ast_.add(cd, m);
      }
      expect("}");
      return cd;
    }
    if (is_kw("enum")) {
      lx_.advance();
      CsNode* ed = ast_.mk("EnumDeclaration");
      take();
      if (is_punct(":")) { lx_.advance(); parse_type(); }
      skip_balanced("{", "}");
      return ed;
    }
    return nullptr;
  }

  CsNode* parse_member() {
    skip_attributes_modifiers();
    if (is_punct(";")) { lx_.advance(); return nullptr; }
    if (is_kw("class") || is_kw("struct") || is_kw("interface") || is_kw("enum"))
      return parse_type_declaration();
    // constructor: Ident '('
    if (is_ident() && lx_.peek().kind == Tok::Punct && lx_.peek().text == "(") {
      CsNode* ctor = ast_.mk("ConstructorDeclaration");
      take();
      ast_.add(ctor, parse_parameter_list());
      if (is_punct(":")) {  // : base(...) / this(...)
        lx_.advance();
        take();
        skip_balanced("(", ")");
      }
      if (is_punct("{")) ast_.add(ctor, parse_block());
      else if (is_punct(";")) lx_.advance();
      return ctor;
    }
    // method / field / property: Type Name ...
    CsNode* type = parse_type();
    if (!is_ident() && !at_end()) {
      // operator overloads etc — skip the member robustly
      while (!is_punct(";") && !is_punct("{") && !at_end()) lx_.advance();
      if (is_punct("{")) skip_balanced("{", "}");
      else if (is_punct(";")) lx_.advance();
      return nullptr;
    }
    std::string name = take();
    if (is_punct("<")) skip_balanced("<", ">");   // generic method
    if (is_punct("(")) {
      CsNode* md = ast_.mk("MethodDeclaration");
      md->method_name = name;
      ast_.add(md, type);
      ast_.tok(md, name, /*method_name=*/true);
      ast_.add(md, parse_parameter_list());
      while (is_ident() && lx_.cur().text == "where")
        while (!is_punct("{") && !is_punct(";") && !at_end()) lx_.advance();
      if (is_punct("{")) {
        ast_.add(md, parse_block());
      } else if (is_punct("=>")) {   // expression-bodied: ArrowExpressionClause
        lx_.advance();
        CsNode* arrow = ast_.mk("ArrowExpressionClause");
        ast_.add(arrow, parse_expression());
        expect(";");
        ast_.add(md, arrow);
      } else if (is_punct(";")) {
        lx_.advance();
      }
      return md;
    }
    if (is_punct("{")) {  // property
      CsNode* pd = ast_.mk("PropertyDeclaration");
      ast_.add(pd, type);
      ast_.tok(pd, name);
      skip_balanced("{", "}");   // accessors (bodies skipped in v1)
      if (is_punct("=")) {       // initializer
        lx_.advance();
        CsNode* ev = ast_.mk("EqualsValueClause");
        ast_.add(ev, parse_expression());
        ast_.add(pd, ev);
        expect(";");
      }
      return pd;
    }
    // field: VariableDeclaration
    CsNode* fd = ast_.mk("FieldDeclaration");
    CsNode* vd = ast_.mk("VariableDeclaration");
    ast_.add(vd, type);
    ast_.add(vd, parse_variable_declarator(name));
    while (is_punct(",")) {
      lx_.advance();
      ast_.add(vd, parse_variable_declarator(take()));
    }
    ast_.add(fd, vd);
    if (is_punct(";")) lx_.advance();
    return fd;
  }

  CsNode* parse_parameter_list() {
    CsNode* pl = ast_.mk("ParameterList");
    expect("(");
    while (!is_punct(")") && !at_end()) {
      skip_attributes_modifiers();
      if (is_kw("ref") || is_kw("out") || is_kw("in") || is_kw("params"))
        lx_.advance();
      CsNode* p = ast_.mk("Parameter");
      CsNode* t = parse_type();
      ast_.add(p, t);
      if (is_ident()) ast_.tok(p, take());
      if (is_punct("=")) {
        lx_.advance();
        CsNode* ev = ast_.mk("EqualsValueClause");
        ast_.add(ev, parse_expression());
        ast_.add(p, ev);
      }
      ast_.add(pl, p);
      if (is_punct(",")) lx_.advance();
    }
    expect(")");
    return pl;
  }

  CsNode* parse_variable_declarator(std::string name) {
    CsNode* vd = ast_.mk("VariableDeclarator");
    ast_.tok(vd, name);
    if (is_punct("=")) {
      lx_.advance();
      CsNode* ev = ast_.mk("EqualsValueClause");
      ast_.add(ev, parse_expression());
      ast_.add(vd, ev);
    }
    return vd;
  }

  // ---- types ----
  bool looks_like_predefined() {
    return lx_.cur().kind == Tok::Keyword && kPredefined.count(lx_.cur().text);
  }

  CsNode* parse_type() {
    CsNode* t = parse_non_array_type();
    while (is_punct("[")) {
      // array type with rank specifier
      CsNode* at = ast_.mk("ArrayType");
      ast_.add(at, t);
      CsNode* rank = ast_.mk("ArrayRankSpecifier");
      lx_.advance();
      while (!is_punct("]") && !at_end()) {
        if (!is_punct(",")) ast_.add(rank, parse_expression());
        else lx_.advance();
      }
      expect("]");
      ast_.add(at, rank);
      t = at;
    }
    while (is_punct("?")) {  // nullable
      CsNode* nt = ast_.mk("NullableType");
      ast_.add(nt, t);
      lx_.advance();
      t = nt;
    }
    return t;
  }

  CsNode* parse_non_array_type() {
    if (looks_like_predefined()) {
      CsNode* pt = ast_.mk("PredefinedType");
      // Roslyn: the keyword token's parent is PredefinedType → it IS a leaf
      ast_.tok(pt, take());
      return pt;
    }
    if (is_kw("var") || (is_ident() && lx_.cur().text == "var")) {
      CsNode* in = ast_.mk("IdentifierName");
      // Leaf.IsLeafToken excludes `var` only in local-declaration position;
      // we conservatively never emit var as a leaf token
      take();
      return in;
    }
    CsNode* t = parse_simple_name();
    while (is_punct(".") ) {
      lx_.advance();
      CsNode* qn = ast_.mk("QualifiedName");
      ast_.add(qn, t);
      ast_.add(qn, parse_simple_name());
      t = qn;
    }
    return t;
  }

  CsNode* parse_simple_name() {
    std::string name = is_ident() ? take() : lx_.cur().text;
    if (!is_punct("<")) {
      CsNode* in = ast_.mk("IdentifierName");
      ast_.tok(in, name);
      return in;
    }
    CsNode* gn = ast_.mk("GenericName");
    ast_.tok(gn, name);
    CsNode* tal = ast_.mk("TypeArgumentList");
    lx_.advance();
    if (!is_punct(">")) {
      ast_.add(tal, parse_type());
      while (is_punct(",")) { lx_.advance(); ast_.add(tal, parse_type()); }
    }
    if (is_punct(">")) lx_.advance();
    else if (is_punct(">>") || is_punct(">>=")) lx_.split_gt();
    ast_.add(gn, tal);
    return gn;
  }

  // ---- statements ----

  CsNode* parse_block() {
    CsNode* b = ast_.mk("Block");
    expect("{");
    while (!is_punct("}") && !at_end()) ast_.add(b, parse_statement());
    expect("}");
    return b;
  }

  CsNode* parse_statement() {
    if (is_punct("{")) return parse_block();
    if (is_punct(";")) { lx_.advance(); return ast_.mk("EmptyStatement"); }
    if (is_kw("if")) {
      CsNode* s = ast_.mk("IfStatement");
      lx_.advance();
      expect("(");
      ast_.add(s, parse_expression());
      expect(")");
      ast_.add(s, parse_statement());
      if (is_kw("else")) {
        lx_.advance();
        CsNode* ec = ast_.mk("ElseClause");
        ast_.add(ec, parse_statement());
        ast_.add(s, ec);
      }
      return s;
    }
    if (is_kw("while")) {
      CsNode* s = ast_.mk("WhileStatement");
      lx_.advance();
      expect("(");
      ast_.add(s, parse_expression());
      expect(")");
      ast_.add(s, parse_statement());
      return s;
    }
    if (is_kw("do")) {
      CsNode* s = ast_.mk("DoStatement");
      lx_.advance();
      ast_.add(s, parse_statement());
      if (is_kw("while")) lx_.advance();
      expect("(");
      ast_.add(s, parse_expression());
      expect(")");
      expect(";");
      return s;
    }
    if (is_kw("for")) return parse_for();
    if (is_kw("foreach")) {
      CsNode* s = ast_.mk("ForEachStatement");
      lx_.advance();
      expect("(");
      ast_.add(s, parse_type());
      if (is_ident()) ast_.tok(s, take());
      if (is_kw("in")) lx_.advance();
      ast_.add(s, parse_expression());
      expect(")");
      ast_.add(s, parse_statement());
      return s;
    }
    if (is_kw("return")) {
      CsNode* s = ast_.mk("ReturnStatement");
      lx_.advance();
      if (!is_punct(";")) ast_.add(s, parse_expression());
      expect(";");
      return s;
    }
    if (is_kw("throw")) {
      CsNode* s = ast_.mk("ThrowStatement");
      lx_.advance();
      if (!is_punct(";")) ast_.add(s, parse_expression());
      expect(";");
      return s;
    }
    if (is_kw("break")) {
      lx_.advance();
      expect(";");
      return ast_.mk("BreakStatement");
    }
    if (is_kw("continue")) {
      lx_.advance();
      expect(";");
      return ast_.mk("ContinueStatement");
    }
    if (is_kw("try")) {
      CsNode* s = ast_.mk("TryStatement");
      lx_.advance();
      ast_.add(s, parse_block());
      while (is_kw("catch")) {
        lx_.advance();
        CsNode* cc = ast_.mk("CatchClause");
        if (is_punct("(")) {
          lx_.advance();
          CsNode* cd = ast_.mk("CatchDeclaration");
          ast_.add(cd, parse_type());
          if (is_ident()) ast_.tok(cd, take());
          expect(")");
          ast_.add(cc, cd);
        }
        ast_.add(cc, parse_block());
        ast_.add(s, cc);
      }
      if (is_kw("finally")) {
        lx_.advance();
        CsNode* fc = ast_.mk("FinallyClause");
        ast_.add(fc, parse_block());
        ast_.add(s, fc);
      }
      return s;
    }
    if (is_kw("switch")) {
      CsNode* s = ast_.mk("SwitchStatement");
      lx_.advance();
      expect("(");
      ast_.add(s, parse_expression());
      expect(")");
      expect("{");
      while (!is_punct("}") && !at_end()) {
        CsNode* sec = ast_.mk("SwitchSection");
        while (is_kw("case") || is_kw("default")) {
          if (is_kw("case")) {
            lx_.advance();
            CsNode* lab = ast_.mk("CaseSwitchLabel");
            ast_.add(lab, parse_expression());
            ast_.add(sec, lab);
          } else {
            lx_.advance();
            ast_.add(sec, ast_.mk("DefaultSwitchLabel"));
          }
          expect(":");
        }
        while (!is_kw("case") && !is_kw("default") && !is_punct("}") && !at_end())
          ast_.add(sec, parse_statement());
        ast_.add(s, sec);
      }
      expect("}");
      return s;
    }
    if (is_kw("using")) {  // using statement
      CsNode* s = ast_.mk("UsingStatement");
      lx_.advance();
      expect("(");
      if (starts_local_decl()) ast_.add(s, parse_variable_declaration());
      else ast_.add(s, parse_expression());
      expect(")");
      ast_.add(s, parse_statement());
      return s;
    }
    if (is_kw("lock")) {
      CsNode* s = ast_.mk("LockStatement");
      lx_.advance();
      expect("(");
      ast_.add(s, parse_expression());
      expect(")");
      ast_.add(s, parse_statement());
      return s;
    }
    if (starts_local_decl()) {
      CsNode* s = ast_.mk("LocalDeclarationStatement");
      ast_.add(s, parse_variable_declaration());
      expect(";");
      return s;
    }
    CsNode* s = ast_.mk("ExpressionStatement");
    ast_.add(s, parse_expression());
    expect(";");
    return s;
  }

  CsNode* parse_variable_declaration() {
    CsNode* vd = ast_.mk("VariableDeclaration");
    if (is_kw("const")) lx_.advance();
    ast_.add(vd, parse_type());
    ast_.add(vd, parse_variable_declarator(take()));
    while (is_punct(",")) {
      lx_.advance();
      ast_.add(vd, parse_variable_declarator(take()));
    }
    return vd;
  }

  CsNode* parse_for() {
    lx_.advance();
    expect("(");
    CsNode* f = ast_.mk("ForStatement");
    if (!is_punct(";")) {
      if (starts_local_decl()) ast_.add(f, parse_variable_declaration());
      else {
        ast_.add(f, parse_expression());
        while (is_punct(",")) { lx_.advance(); ast_.add(f, parse_expression()); }
      }
    }
    expect(";");
    if (!is_punct(";")) ast_.add(f, parse_expression());
    expect(";");
    if (!is_punct(")")) {
      ast_.add(f, parse_expression());
      while (is_punct(",")) { lx_.advance(); ast_.add(f, parse_expression()); }
    }
    expect(")");
    ast_.add(f, parse_statement());
    return f;
  }

  bool starts_local_decl() {
    if (is_kw("const")) return true;
    if (is_kw("var") || (is_ident() && lx_.cur().text == "var")) {
      return lx_.peek().kind == Tok::Ident;
    }
    if (looks_like_predefined()) return true;
    if (!is_ident()) return false;
    // probe: Name(.Name|<...>)*([])* Ident (=|;|,)
    CsLexer probe = lx_;
    auto at = [&](const char* p) {
      return probe.cur().kind == Tok::Punct && probe.cur().text == p;
    };
    if (probe.cur().kind != Tok::Ident) return false;
    probe.advance();
    for (;;) {
      if (at(".")) {
        probe.advance();
        if (probe.cur().kind != Tok::Ident) return false;
        probe.advance();
        continue;
      }
      break;
    }
    if (at("<")) {
      int depth = 0, guard = 0;
      do {
        if (at("<")) ++depth;
        else if (at(">")) --depth;
        else if (at(">>")) depth -= 2;
        else if (probe.cur().kind == Tok::End || at(";") || at("(")) return false;
        probe.advance();
        if (guard++ > 60) return false;
      } while (depth > 0);
    }
    while (at("[")) {
      probe.advance();
      if (!at("]")) return false;
      probe.advance();
    }
    if (at("?")) probe.advance();
    return probe.cur().kind == Tok::Ident;
  }

  // ---- expressions ----

  CsNode* parse_expression() { return parse_assignment(); }

  CsNode* parse_assignment() {
    CsNode* lhs = parse_ternary();
    static const std::pair<const char*, const char*> ops[] = {
        {"=", "SimpleAssignmentExpression"},
        {"+=", "AddAssignmentExpression"},
        {"-=", "SubtractAssignmentExpression"},
        {"*=", "MultiplyAssignmentExpression"},
        {"/=", "DivideAssignmentExpression"},
        {"%=", "ModuloAssignmentExpression"},
        {"&=", "AndAssignmentExpression"},
        {"|=", "OrAssignmentExpression"},
        {"^=", "ExclusiveOrAssignmentExpression"},
        {"<<=", "LeftShiftAssignmentExpression"},
        {">>=", "RightShiftAssignmentExpression"}};
    for (auto& [sym, kind] : ops) {
      if (is_punct(sym)) {
        lx_.advance();
        CsNode* a = ast_.mk(kind);
        ast_.add(a, lhs);
        ast_.add(a, parse_assignment());
        return a;
      }
    }
    return lhs;
  }

  CsNode* parse_ternary() {
    CsNode* c = parse_binary(0);
    if (is_punct("?")) {
      lx_.advance();
      CsNode* t = ast_.mk("ConditionalExpression");
      ast_.add(t, c);
      ast_.add(t, parse_expression());
      expect(":");
      ast_.add(t, parse_expression());
      return t;
    }
    return c;
  }

  struct BinOp {
    const char* sym;
    const char* kind;
    int prec;
  };
  static const BinOp* find_binop(const Token& t) {
    static const BinOp ops[] = {
        {"??", "CoalesceExpression", 1},
        {"||", "LogicalOrExpression", 2},
        {"&&", "LogicalAndExpression", 3},
        {"|", "BitwiseOrExpression", 4},
        {"^", "ExclusiveOrExpression", 5},
        {"&", "BitwiseAndExpression", 6},
        {"==", "EqualsExpression", 7},
        {"!=", "NotEqualsExpression", 7},
        {"<", "LessThanExpression", 8},
        {">", "GreaterThanExpression", 8},
        {"<=", "LessThanOrEqualExpression", 8},
        {">=", "GreaterThanOrEqualExpression", 8},
        {"<<", "LeftShiftExpression", 9},
        {">>", "RightShiftExpression", 9},
        {"+", "AddExpression", 10},
        {"-", "SubtractExpression", 10},
        {"*", "MultiplyExpression", 11},
        {"/", "DivideExpression", 11},
        {"%", "ModuloExpression", 11}};
    if (t.kind != Tok::Punct) return nullptr;
    for (const auto& op : ops)
      if (t.text == op.sym) return &op;
    return nullptr;
  }

  CsNode* parse_binary(int min_prec) {
    CsNode* lhs = parse_is_as(min_prec);
    for (;;) {
      const BinOp* op = find_binop(lx_.cur());
      if (!op || op->prec < min_prec) return lhs;
      lx_.advance();
      CsNode* rhs = parse_binary(op->prec + 1);
      CsNode* b = ast_.mk(op->kind);
      ast_.add(b, lhs);
      ast_.add(b, rhs);
      lhs = b;
    }
  }

  CsNode* parse_is_as(int min_prec) {
    CsNode* e = parse_unary();
    for (;;) {
      if (is_kw("is") && min_prec <= 8) {
        lx_.advance();
        CsNode* io = ast_.mk("IsExpression");
        ast_.add(io, e);
        ast_.add(io, parse_type());
        e = io;
      } else if (is_kw("as") && min_prec <= 8) {
        lx_.advance();
        CsNode* ao = ast_.mk("AsExpression");
        ast_.add(ao, e);
        ast_.add(ao, parse_type());
        e = ao;
      } else {
        return e;
      }
    }
  }

  CsNode* parse_unary() {
    static const std::pair<const char*, const char*> pre[] = {
        {"+", "UnaryPlusExpression"},     {"-", "UnaryMinusExpression"},
        {"!", "LogicalNotExpression"},    {"~", "BitwiseNotExpression"},
        {"++", "PreIncrementExpression"}, {"--", "PreDecrementExpression"}};
    for (auto& [sym, kind] : pre) {
      if (is_punct(sym)) {
        lx_.advance();
        CsNode* u = ast_.mk(kind);
        ast_.add(u, parse_unary());
        return u;
      }
    }
    if (is_punct("(") && cast_ahead()) {
      lx_.advance();
      CsNode* c = ast_.mk("CastExpression");
      ast_.add(c, parse_type());
      expect(")");
      ast_.add(c, parse_unary());
      return c;
    }
    return parse_postfix();
  }

  bool cast_ahead() {
    CsLexer probe = lx_;
    auto at = [&](const char* p) {
      return probe.cur().kind == Tok::Punct && probe.cur().text == p;
    };
    probe.advance();
    bool prim = probe.cur().kind == Tok::Keyword &&
                kPredefined.count(probe.cur().text);
    if (prim) {
      probe.advance();
      while (at("[")) {
        probe.advance();
        if (!at("]")) return false;
        probe.advance();
      }
      return at(")");
    }
    if (probe.cur().kind != Tok::Ident) return false;
    probe.advance();
    for (;;) {
      if (at(".")) {
        probe.advance();
        if (probe.cur().kind != Tok::Ident) return false;
        probe.advance();
        continue;
      }
      break;
    }
    while (at("[")) {
      probe.advance();
      if (!at("]")) return false;
      probe.advance();
    }
    if (!at(")")) return false;
    probe.advance();
    const Token& nx = probe.cur();
    return nx.kind == Tok::Ident || nx.kind == Tok::NumLit ||
           nx.kind == Tok::StrLit || nx.kind == Tok::CharLit ||
           (nx.kind == Tok::Keyword &&
            (nx.text == "this" || nx.text == "new" || nx.text == "null"));
  }

  CsNode* parse_argument_list(const char* kind, const char* open,
                              const char* close) {
    CsNode* al = ast_.mk(kind);
    expect(open);
    while (!is_punct(close) && !at_end()) {
      if (is_kw("ref") || is_kw("out") || is_kw("in")) lx_.advance();
      CsNode* arg = ast_.mk("Argument");
      ast_.add(arg, parse_expression());
      ast_.add(al, arg);
      if (is_punct(",")) lx_.advance();
    }
    expect(close);
    return al;
  }

  CsNode* parse_postfix() {
    CsNode* e = parse_primary();
    for (;;) {
      if (is_punct(".") || is_punct("?.")) {
        lx_.advance();
        CsNode* ma = ast_.mk("SimpleMemberAccessExpression");
        ast_.add(ma, e);
        ast_.add(ma, parse_simple_name());
        e = ma;
        continue;
      }
      if (is_punct("(")) {
        CsNode* inv = ast_.mk("InvocationExpression");
        ast_.add(inv, e);
        ast_.add(inv, parse_argument_list("ArgumentList", "(", ")"));
        e = inv;
        continue;
      }
      if (is_punct("[")) {
        CsNode* ea = ast_.mk("ElementAccessExpression");
        ast_.add(ea, e);
        ast_.add(ea, parse_argument_list("BracketedArgumentList", "[", "]"));
        e = ea;
        continue;
      }
      if (is_punct("++")) {
        lx_.advance();
        CsNode* u = ast_.mk("PostIncrementExpression");
        ast_.add(u, e);
        e = u;
        continue;
      }
      if (is_punct("--")) {
        lx_.advance();
        CsNode* u = ast_.mk("PostDecrementExpression");
        ast_.add(u, e);
        e = u;
        continue;
      }
      return e;
    }
  }

  CsNode* parse_primary() {
    const Token& t = lx_.cur();
    if (t.kind == Tok::NumLit) {
      CsNode* n = ast_.mk("NumericLiteralExpression");
      ast_.tok(n, take());
      return n;
    }
    if (t.kind == Tok::StrLit) {
      CsNode* n = ast_.mk("StringLiteralExpression");
      ast_.tok(n, take());
      return n;
    }
    if (t.kind == Tok::CharLit) {
      CsNode* n = ast_.mk("CharacterLiteralExpression");
      ast_.tok(n, take());
      return n;
    }
    if (is_kw("true")) { lx_.advance(); return ast_.mk("TrueLiteralExpression"); }
    if (is_kw("false")) { lx_.advance(); return ast_.mk("FalseLiteralExpression"); }
    if (is_kw("null")) { lx_.advance(); return ast_.mk("NullLiteralExpression"); }
    if (is_kw("this")) { lx_.advance(); return ast_.mk("ThisExpression"); }
    if (is_kw("base")) { lx_.advance(); return ast_.mk("BaseExpression"); }
    if (is_kw("typeof")) {
      lx_.advance();
      CsNode* te = ast_.mk("TypeOfExpression");
      expect("(");
      ast_.add(te, parse_type());
      expect(")");
      return te;
    }
    if (is_kw("new")) {
      lx_.advance();
      // array creation?
      CsNode* ty = parse_non_array_type();
      if (is_punct("[")) {
        CsNode* ac = ast_.mk("ArrayCreationExpression");
        CsNode* at = ast_.mk("ArrayType");
        ast_.add(at, ty);
        CsNode* rank = ast_.mk("ArrayRankSpecifier");
        lx_.advance();
        while (!is_punct("]") && !at_end()) {
          if (!is_punct(",")) ast_.add(rank, parse_expression());
          else lx_.advance();
        }
        expect("]");
        ast_.add(at, rank);
        ast_.add(ac, at);
        if (is_punct("{")) {
          CsNode* init = ast_.mk("ArrayInitializerExpression");
          lx_.advance();
          while (!is_punct("}") && !at_end()) {
            ast_.add(init, parse_expression());
            if (is_punct(",")) lx_.advance();
          }
          expect("}");
          ast_.add(ac, init);
        }
        return ac;
      }
      CsNode* oc = ast_.mk("ObjectCreationExpression");
      ast_.add(oc, ty);
      if (is_punct("(")) ast_.add(oc, parse_argument_list("ArgumentList", "(", ")"));
      if (is_punct("{")) {  // object initializer
        CsNode* init = ast_.mk("ObjectInitializerExpression");
        lx_.advance();
        while (!is_punct("}") && !at_end()) {
          ast_.add(init, parse_expression());
          if (is_punct(",")) lx_.advance();
        }
        expect("}");
        ast_.add(oc, init);
      }
      return oc;
    }
    if (is_punct("(")) {
      lx_.advance();
      CsNode* pe = ast_.mk("ParenthesizedExpression");
      ast_.add(pe, parse_expression());
      expect(")");
      return pe;
    }
    if (is_ident()) {
      // lambda: ident =>
      if (lx_.peek().kind == Tok::Punct && lx_.peek().text == "=>") {
        CsNode* le = ast_.mk("SimpleLambdaExpression");
        CsNode* p = ast_.mk("Parameter");
        ast_.tok(p, take());
        ast_.add(le, p);
        lx_.advance();  // =>
        if (is_punct("{")) ast_.add(le, parse_block());
        else ast_.add(le, parse_expression());
        return le;
      }
      return parse_simple_name();
    }
    if (looks_like_predefined()) {
      // predefined type in expression position: int.Parse(...) etc
      CsNode* pt = ast_.mk("PredefinedType");
      ast_.tok(pt, take());
      return pt;
    }
    throw CsParseError("unexpected token '" + lx_.cur().text + "'");
  }

  CsLexer lx_;
  CsAst& ast_;
};

// ---------------------------------------------------------------------------
// Naming / hashing utilities (Utilities.cs semantics)
// ---------------------------------------------------------------------------

static const std::set<std::string> kNumKeep = {"0", "1", "2", "3",
                                               "4", "5", "10"};

static std::string cs_normalize(const std::string& in) {
  std::string s;
  for (char c : in) s += (char)tolower((unsigned char)c);
  // remove literal "\\n" sequences (the reference's Replace("\\\\n", ""))
  std::string t;
  for (size_t i = 0; i < s.size();) {
    if (i + 1 < s.size() && s[i] == '\\' && s[i + 1] == 'n') { i += 2; continue; }
    t += s[i++];
  }
  // drop whitespace and non-ASCII
  std::string u;
  for (char c : t) {
    if (isspace((unsigned char)c)) continue;
    if ((unsigned char)c > 0x7E) continue;
    u += c;
  }
  std::string alpha;
  for (char c : u)
    if (isalpha((unsigned char)c)) alpha += c;
  if (!alpha.empty()) return alpha;
  bool all_digits = !u.empty();
  for (char c : u)
    if (!isdigit((unsigned char)c)) all_digits = false;
  if (all_digits) return kNumKeep.count(u) ? u : "NUM";
  return "";
}

static std::vector<std::string> cs_subtokens(const std::string& in) {
  std::string s = in;
  size_t b = s.find_first_not_of(" \t\r\n");
  size_t e = s.find_last_not_of(" \t\r\n");
  if (b == std::string::npos) return {};
  s = s.substr(b, e - b + 1);
  std::vector<std::string> parts;
  std::string cur;
  auto flush = [&]() {
    if (!cur.empty()) {
      std::string n = cs_normalize(cur);
      if (!n.empty()) parts.push_back(n);
      cur.clear();
    }
  };
  for (size_t i = 0; i < s.size(); ++i) {
    char c = s[i];
    if (c == '_' || isdigit((unsigned char)c) || isspace((unsigned char)c)) {
      flush();
      continue;
    }
    if (i > 0) {
      char p = s[i - 1];
      bool camel = islower((unsigned char)p) && isupper((unsigned char)c);
      bool acr = i + 1 < s.size() && isupper((unsigned char)p) &&
                 isupper((unsigned char)c) && islower((unsigned char)s[i + 1]);
      if (camel || acr) flush();
    }
    cur += c;
  }
  flush();
  return parts;
}

static std::string split_name_unless_empty(const std::string& original) {
  if (original == "METHOD_NAME") return original;
  auto subs = cs_subtokens(original);
  std::string name;
  for (size_t i = 0; i < subs.size(); ++i)
    name += (i ? "|" : "") + subs[i];
  if (name.empty()) name = cs_normalize(original);
  bool all_space = !name.empty();
  for (char c : name)
    if (!isspace((unsigned char)c)) all_space = false;
  if (all_space) name = "SPACE";
  if (name.empty()) name = "BLANK";
  return name;
}

// deterministic .NET-Framework-style 32-bit string hash (see header note)
static int32_t dotnet_hash(const std::string& s) {
  uint32_t hash1 = (5381u << 16) + 5381u;
  uint32_t hash2 = hash1;
  for (size_t i = 0; i < s.size(); i += 2) {
    hash1 = ((hash1 << 5) + hash1) ^ (uint8_t)s[i];
    if (i + 1 < s.size()) hash2 = ((hash2 << 5) + hash2) ^ (uint8_t)s[i + 1];
  }
  return (int32_t)(hash1 + hash2 * 1566083941u);
}

// ---------------------------------------------------------------------------
// Extraction (Extractor.cs / Variable.cs / PathFinder.cs semantics)
// ---------------------------------------------------------------------------

struct CsOptions {
  int max_length = 9;
  int max_width = 2;
  int max_contexts = 30000;
  bool no_hash = false;
  int threads = 1;
  std::string path = "./data/";
  std::string ofile;
};

static void assign_depths(CsNode* n, int d) {
  n->depth = d;
  for (CsNode* c : n->kids) assign_depths(c, d + 1);
}

static void collect_methods(CsNode* n, std::vector<CsNode*>& out) {
  if (n->kind == "MethodDeclaration") out.push_back(n);
  for (CsNode* c : n->kids) collect_methods(c, out);
}

static void collect_leaves(CsNode* n, std::vector<CsLeaf*>& out) {
  // Roslyn walker order: child nodes' leaves first, then own tokens
  for (CsNode* c : n->kids) collect_leaves(c, out);
  for (CsLeaf* l : n->tokens) out.push_back(l);
}

static int child_index(CsNode* parent, CsNode* child) {
  for (size_t i = 0; i < parent->kids.size(); ++i)
    if (parent->kids[i] == child) return (int)i;
  return -1;
}

static const std::set<std::string> kCsAddChildId = {
    "SimpleAssignmentExpression", "ElementAccessExpression",
    "SimpleMemberAccessExpression", "InvocationExpression",
    "BracketedArgumentList", "ArgumentList"};

struct CsPath {
  CsLeaf* left;
  std::vector<CsNode*> left_side;
  CsNode* ancestor;
  std::vector<CsNode*> right_side;
  CsLeaf* right;
};

static bool find_path(CsLeaf* l, CsLeaf* r, int max_len, int max_width,
                      CsPath& out) {
  CsNode* ln = l->parent;
  CsNode* rn = r->parent;
  CsNode* a = ln;
  CsNode* b = rn;
  while (a != b) {
    if (a->depth >= b->depth) a = a->parent;
    else b = b->parent;
    if (!a || !b) return false;
  }
  CsNode* lca = a;
  if (ln->depth + rn->depth - 2 * lca->depth + 2 > max_len) return false;
  std::vector<CsNode*> left_side, right_side;
  for (CsNode* c = ln; c != lca; c = c->parent) left_side.push_back(c);
  for (CsNode* c = rn; c != lca; c = c->parent) right_side.push_back(c);
  std::reverse(right_side.begin(), right_side.end());
  if (!left_side.empty() && !right_side.empty()) {
    int il = child_index(lca, left_side.back());
    int ir = child_index(lca, right_side.front());
    if (std::abs(il - ir) >= max_width) return false;
  }
  out = CsPath{l, left_side, lca, right_side, r};
  return true;
}

static std::string path_nodes_to_string(const CsPath& p) {
  std::string out;
  auto add_node = [&](CsNode* n) {
    out += n->kind;
    if (n->parent && kCsAddChildId.count(n->parent->kind)) {
      int idx = child_index(n->parent, n);
      out += std::to_string(std::min(idx, 3));
    }
  };
  for (CsNode* n : p.left_side) {
    add_node(n);
    out += "^";
  }
  out += p.ancestor->kind;
  for (CsNode* n : p.right_side) {
    out += "_";
    add_node(n);
  }
  return out;
}

static std::vector<std::string> extract_cs(const std::string& code,
                                           const CsOptions& opt) {
  CsAst ast;
  std::vector<Comment> comments;
  CsNode* root = nullptr;
  try {
    CsParser parser(code, ast, &comments);
    root = parser.parse_compilation_unit();
  } catch (const CsParseError&) {
    return {};
  }
  assign_depths(root, 0);

  // file-scope comment batches (the reference appends these to EVERY method)
  std::vector<std::string> comment_ctxs;
  for (const Comment& c : comments) {
    std::string txt = c.text;
    auto strip = [&](const std::string& s) {
      size_t b = s.find_first_not_of(" /*{}");
      size_t e = s.find_last_not_of(" /*{}");
      return b == std::string::npos ? std::string() : s.substr(b, e - b + 1);
    };
    std::string body = strip(txt);
    std::string norm = split_name_unless_empty(body);
    std::vector<std::string> parts;
    std::string cur;
    for (char ch : norm) {
      if (ch == '|') { parts.push_back(cur); cur.clear(); }
      else cur += ch;
    }
    parts.push_back(cur);
    for (size_t i = 0; i * 5 < parts.size(); ++i) {
      std::string batch;
      for (size_t j = i * 5; j < std::min(parts.size(), (i + 1) * 5); ++j)
        batch += (j > i * 5 ? "|" : "") + parts[j];
      comment_ctxs.push_back(batch + ",COMMENT," + batch);
    }
  }

  std::vector<CsNode*> methods;
  collect_methods(root, methods);
  std::vector<std::string> results;
  std::mt19937 rng(123457);  // deterministic (reference: time-seeded Random)

  for (CsNode* md : methods) {
    std::vector<CsLeaf*> leaves;
    collect_leaves(md, leaves);

    // variables: group leaves by name (METHOD_NAME for the decl identifier),
    // in first-appearance order
    std::vector<std::pair<std::string, std::vector<CsLeaf*>>> variables;
    std::map<std::string, size_t> name_to_var;
    for (CsLeaf* l : leaves) {
      std::string name = l->is_method_name ? "METHOD_NAME" : l->text;
      auto it = name_to_var.find(name);
      if (it == name_to_var.end()) {
        name_to_var[name] = variables.size();
        variables.push_back({name, {l}});
      } else {
        variables[it->second].second.push_back(l);
      }
    }

    // pairs: Choose2 ++ self-pairs, reservoir-sampled to max_contexts
    std::vector<std::pair<size_t, size_t>> pairs;
    long seen = 0;
    auto offer = [&](size_t i, size_t j) {
      ++seen;
      if ((int)pairs.size() < opt.max_contexts) {
        pairs.push_back({i, j});
      } else {
        long pos = (long)(rng() % seen);
        if (pos < (long)pairs.size()) pairs[pos] = {i, j};
      }
    };
    for (size_t i = 0; i < variables.size(); ++i)
      for (size_t j = i + 1; j < variables.size(); ++j) offer(i, j);
    for (size_t i = 0; i < variables.size(); ++i) offer(i, i);

    std::vector<std::string> ctxs;
    for (auto& [vi, vj] : pairs) {
      for (CsLeaf* rhs : variables[vj].second)
        for (CsLeaf* lhs : variables[vi].second) {
          if (lhs == rhs) continue;
          CsPath path;
          if (!find_path(lhs, rhs, opt.max_length, opt.max_width, path))
            continue;
          std::string ps = path_nodes_to_string(path);
          std::string hashed =
              opt.no_hash ? ps : std::to_string(dotnet_hash(ps));
          ctxs.push_back(split_name_unless_empty(variables[vi].first) + "," +
                         hashed + "," +
                         split_name_unless_empty(variables[vj].first));
        }
    }
    for (const std::string& c : comment_ctxs) ctxs.push_back(c);

    auto subs = cs_subtokens(md->method_name);
    std::string label;
    for (size_t i = 0; i < subs.size(); ++i) label += (i ? "|" : "") + subs[i];
    std::string line = label;
    for (const std::string& c : ctxs) line += " " + c;
    results.push_back(line);
  }
  return results;
}

// ---------------------------------------------------------------------------
// CLI (Program.cs semantics; -o append or stdout)
// ---------------------------------------------------------------------------

static std::string read_file_cs(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  std::ostringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

static void walk_cs(const std::string& dir, std::vector<std::string>& out) {
  DIR* d = opendir(dir.c_str());
  if (!d) return;
  struct dirent* ent;
  while ((ent = readdir(d)) != nullptr) {
    std::string name = ent->d_name;
    if (name == "." || name == "..") continue;
    std::string full = dir + "/" + name;
    struct stat st;
    if (stat(full.c_str(), &st) != 0) continue;
    if (S_ISDIR(st.st_mode)) walk_cs(full, out);
    else if (S_ISREG(st.st_mode) && name.size() > 3 &&
             name.substr(name.size() - 3) == ".cs")
      out.push_back(full);
  }
  closedir(d);
}

int main(int argc, char** argv) {
  CsOptions opt;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() -> std::string { return (i + 1 < argc) ? argv[++i] : ""; };
    if (a == "--path" || a == "-p") opt.path = next();
    else if (a == "--max_length" || a == "-l") opt.max_length = atoi(next().c_str());
    else if (a == "--max_width") opt.max_width = atoi(next().c_str());
    else if (a == "--max_contexts") opt.max_contexts = atoi(next().c_str());
    else if (a == "--no_hash" || a == "-h") opt.no_hash = true;
    else if (a == "--threads" || a == "-t") opt.threads = atoi(next().c_str());
    else if (a == "--ofile_name" || a == "-o") opt.ofile = next();
    else {
      std::cerr << "unknown option: " << a << "\n";
      return 2;
    }
  }

  std::vector<std::string> files;
  struct stat st;
  if (stat(opt.path.c_str(), &st) == 0 && S_ISDIR(st.st_mode))
    walk_cs(opt.path, files);
  else
    files.push_back(opt.path);
  std::sort(files.begin(), files.end());

  std::vector<std::vector<std::string>> results(files.size());
  std::atomic<size_t> next_idx(0);
  int nt = std::max(1, std::min<int>(opt.threads ? opt.threads : 1,
                                     (int)std::thread::hardware_concurrency()));
  auto work = [&]() {
    size_t i;
    while ((i = next_idx.fetch_add(1)) < files.size()) {
      try {
        results[i] = extract_cs(read_file_cs(files[i]), opt);
      } catch (...) {
        results[i].clear();
      }
    }
  };
  std::vector<std::thread> pool;
  for (int t = 0; t < nt; ++t) pool.emplace_back(work);
  for (auto& th : pool) th.join();

  std::ostream* out = &std::cout;
  std::ofstream fout;
  if (!opt.ofile.empty()) {
    fout.open(opt.ofile, std::ios::app);
    out = &fout;
  }
  for (const auto& rs : results)
    for (const std::string& r : rs)
      if (!r.empty()) (*out) << r << "\n";
  return 0;
}
