// c2v-extract — C++ AST path-context extractor for Java sources.
//
// Reimplements the reference JavaExtractor (Java + javaparser-core
// 3.0.0-alpha.4) output contract: one line per method,
//   `method|sub|tokens src,hash(path),tgt src,hash(path),tgt ...`
// where a path is the AST walk between two leaves. AST node-type names,
// children ordering, leaf rules, naming/normalization, and the path grammar
// follow the reference exactly — see NOTES.md for the grounding (children
// order was recovered from the reference jar's constructor bytecode).
//
// Build: g++ -O2 -std=c++17 -pthread -o c2v-extract c2v_extract.cpp

#include <algorithm>
#include <atomic>
#include <cstdint>
#include <cstring>
#include <deque>
#include <fstream>
#include <functional>
#include <iostream>
#include <memory>
#include <mutex>
#include <set>
#include <sstream>
#include <string>
#include <thread>
#include <vector>

#include <dirent.h>
#include <sys/stat.h>

// ---------------------------------------------------------------------------
// Lexer
// ---------------------------------------------------------------------------

enum class Tok {
  End, Ident, Keyword, IntLit, LongLit, FloatLit, DoubleLit, CharLit,
  StringLit, Punct
};

struct Token {
  Tok kind = Tok::End;
  std::string text;
};

static const std::set<std::string> kKeywords = {
    "abstract", "assert", "boolean", "break", "byte", "case", "catch", "char",
    "class", "const", "continue", "default", "do", "double", "else", "enum",
    "extends", "final", "finally", "float", "for", "goto", "if", "implements",
    "import", "instanceof", "int", "interface", "long", "native", "new",
    "package", "private", "protected", "public", "return", "short", "static",
    "strictfp", "super", "switch", "synchronized", "this", "throw", "throws",
    "transient", "try", "void", "volatile", "while", "true", "false", "null"};

struct ParseError : std::runtime_error {
  explicit ParseError(const std::string& m) : std::runtime_error(m) {}
};

class Lexer {
 public:
  explicit Lexer(const std::string& src) : s_(src) { advance(); }

  const Token& cur() const { return cur_; }
  const Token& peek() {
    if (!has_peek_) {
      peek_ = lex();
      has_peek_ = true;
    }
    return peek_;
  }

  // consume one '>' from a '>>'/'>>>' token (generic-closing ambiguity)
  void split_gt() {
    if (cur_.kind == Tok::Punct && cur_.text.size() > 1 &&
        cur_.text[0] == '>' )
      cur_.text.erase(cur_.text.begin());
    else
      advance();
  }

  void advance() {
    if (has_peek_) {
      cur_ = peek_;
      has_peek_ = false;
    } else {
      cur_ = lex();
    }
  }

  // position save/restore for error recovery (the reference member makes
  // the default copy-assignment unusable; copy-construct + restore_from
  // covers the rewind use case)
  void restore_from(const Lexer& other) {
    cur_ = other.cur_;
    peek_ = other.peek_;
    has_peek_ = other.has_peek_;
    i_ = other.i_;
  }

 private:
  Token lex() {
    skip_ws_comments();
    Token t;
    if (i_ >= s_.size()) return t;
    char c = s_[i_];
    // bytes >= 0x80 are UTF-8 lead/continuation bytes: Java allows unicode
    // identifiers (javaparser accepts them), so fold whole multi-byte
    // sequences into the identifier token instead of failing the member
    if (isalpha((unsigned char)c) || c == '_' || c == '$' ||
        (unsigned char)c >= 0x80) {
      size_t j = i_;
      while (j < s_.size() && (isalnum((unsigned char)s_[j]) || s_[j] == '_' ||
                               s_[j] == '$' || (unsigned char)s_[j] >= 0x80))
        ++j;
      t.text = s_.substr(i_, j - i_);
      t.kind = kKeywords.count(t.text) ? Tok::Keyword : Tok::Ident;
      i_ = j;
      return t;
    }
    if (isdigit((unsigned char)c) ||
        (c == '.' && i_ + 1 < s_.size() && isdigit((unsigned char)s_[i_ + 1]))) {
      return lex_number();
    }
    if (c == '\'') return lex_char();
    if (c == '"') return lex_string();
    // punctuation / operators: longest-match
    static const char* ops[] = {
        ">>>=", "<<=", ">>=", ">>>", "...", "->", "::", "++", "--", "&&", "||",
        "==", "!=", "<=", ">=", "+=", "-=", "*=", "/=", "%=", "&=", "|=", "^=",
        "<<", ">>"};
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

  Token lex_number() {
    size_t j = i_;
    bool is_fp = false;
    if (s_[j] == '0' && j + 1 < s_.size() && (s_[j + 1] == 'x' || s_[j + 1] == 'X')) {
      j += 2;
      while (j < s_.size() && (isxdigit((unsigned char)s_[j]) || s_[j] == '_')) ++j;
    } else if (s_[j] == '0' && j + 1 < s_.size() && (s_[j + 1] == 'b' || s_[j + 1] == 'B')) {
      j += 2;
      while (j < s_.size() && (s_[j] == '0' || s_[j] == '1' || s_[j] == '_')) ++j;
    } else {
      while (j < s_.size() && (isdigit((unsigned char)s_[j]) || s_[j] == '_')) ++j;
      if (j < s_.size() && s_[j] == '.') {
        is_fp = true;
        ++j;
        while (j < s_.size() && (isdigit((unsigned char)s_[j]) || s_[j] == '_')) ++j;
      }
      if (j < s_.size() && (s_[j] == 'e' || s_[j] == 'E')) {
        is_fp = true;
        ++j;
        if (j < s_.size() && (s_[j] == '+' || s_[j] == '-')) ++j;
        while (j < s_.size() && isdigit((unsigned char)s_[j])) ++j;
      }
    }
    Token t;
    t.text = s_.substr(i_, j - i_);
    if (j < s_.size() && (s_[j] == 'l' || s_[j] == 'L')) {
      t.kind = Tok::LongLit;
      t.text += s_[j];
      ++j;
    } else if (j < s_.size() && (s_[j] == 'f' || s_[j] == 'F')) {
      t.kind = Tok::FloatLit;
      ++j;
    } else if (j < s_.size() && (s_[j] == 'd' || s_[j] == 'D')) {
      t.kind = Tok::DoubleLit;
      ++j;
    } else {
      t.kind = is_fp ? Tok::DoubleLit : Tok::IntLit;
    }
    i_ = j;
    return t;
  }

  Token lex_char() {
    size_t j = i_ + 1;
    std::string v;
    while (j < s_.size() && s_[j] != '\'') {
      if (s_[j] == '\\' && j + 1 < s_.size()) {
        v += s_[j];
        v += s_[j + 1];
        j += 2;
      } else {
        v += s_[j];
        ++j;
      }
    }
    ++j;  // closing '
    Token t;
    t.kind = Tok::CharLit;
    t.text = v;
    i_ = j;
    return t;
  }

  Token lex_string() {
    size_t j = i_ + 1;
    std::string v;
    while (j < s_.size() && s_[j] != '"') {
      if (s_[j] == '\\' && j + 1 < s_.size()) {
        v += s_[j];
        v += s_[j + 1];
        j += 2;
      } else {
        v += s_[j];
        ++j;
      }
    }
    ++j;
    Token t;
    t.kind = Tok::StringLit;
    t.text = v;
    i_ = j;
    return t;
  }

  void skip_ws_comments() {
    for (;;) {
      while (i_ < s_.size() && isspace((unsigned char)s_[i_])) ++i_;
      if (i_ + 1 < s_.size() && s_[i_] == '/' && s_[i_ + 1] == '/') {
        while (i_ < s_.size() && s_[i_] != '\n') ++i_;
        continue;
      }
      if (i_ + 1 < s_.size() && s_[i_] == '/' && s_[i_ + 1] == '*') {
        i_ += 2;
        while (i_ + 1 < s_.size() && !(s_[i_] == '*' && s_[i_ + 1] == '/')) ++i_;
        i_ = std::min(i_ + 2, s_.size());
        continue;
      }
      break;
    }
  }

  const std::string& s_;
  size_t i_ = 0;
  Token cur_, peek_;
  bool has_peek_ = false;
};

// ---------------------------------------------------------------------------
// AST
// ---------------------------------------------------------------------------

struct Node {
  std::string type;       // JavaParser simple class name (RawType)
  std::string op;         // operator enum name for Binary/Unary/Assign
  std::string text;       // toString() equivalent for leaves
  std::vector<Node*> kids;
  Node* parent = nullptr;
  int childId = 0;
  bool isStmt = false;
  bool isNullLit = false;
  bool isIntLit = false;
  bool isBoxed = false;        // boxed ClassOrInterfaceType
  bool genericParent = false;  // COIT with type arguments
  std::string coitName;        // COIT simple name (for boxed/generic rules)
  std::string unboxed;         // unboxed type name for boxed COITs
  // method-only metadata
  std::string methodName;
  long bodyLines = 0;
  // computed during leaf collection
  std::string propType, propName;
};

class Ast {
 public:
  Node* mk(const char* type) {
    nodes_.emplace_back();
    Node* n = &nodes_.back();
    n->type = type;
    return n;
  }
  void add(Node* parent, Node* child) {
    if (!child) return;
    child->parent = parent;
    parent->kids.push_back(child);
  }
  std::deque<Node> nodes_;
};

static const std::set<std::string> kBoxed = {
    "Boolean", "Byte", "Character", "Double", "Float", "Integer", "Long",
    "Short"};
static const char* unbox(const std::string& n) {
  if (n == "Boolean") return "boolean";
  if (n == "Byte") return "byte";
  if (n == "Character") return "char";
  if (n == "Double") return "double";
  if (n == "Float") return "float";
  if (n == "Integer") return "int";
  if (n == "Long") return "long";
  if (n == "Short") return "short";
  return "";
}

// ---------------------------------------------------------------------------
// Parser (recursive descent over the Java subset the corpus uses)
// ---------------------------------------------------------------------------

class Parser {
 public:
  Parser(const std::string& src, Ast& ast) : lx_(src), ast_(ast) {}

  // members dropped by the per-member error recovery; the 3-stage retry
  // treats a parse with skips as soft-failed (a later wrapping stage may
  // recover MORE of the snippet)
  int skipped_members() const { return skipped_members_; }

  Node* parse_compilation_unit() {
    Node* cu = ast_.mk("CompilationUnit");
    if (is_kw("package")) {
      lx_.advance();
      Node* pd = ast_.mk("PackageDeclaration");
      ast_.add(pd, parse_qualified_name_expr());
      expect(";");
      ast_.add(cu, pd);
    }
    while (is_kw("import")) {
      lx_.advance();
      Node* imp = ast_.mk("SingleTypeImportDeclaration");
      // consume the dotted name (not modeled in detail; above methods anyway)
      while (!is_punct(";") && !at_end()) lx_.advance();
      expect(";");
      ast_.add(cu, imp);
    }
    while (!at_end()) {
      Node* td = parse_type_declaration();
      if (td) ast_.add(cu, td);
      else break;
    }
    return cu;
  }

 private:
  Node* parse_qualified_name_expr() {
    Node* n = ast_.mk("NameExpr");
    n->text = take();
    while (is_punct(".") ) {
      lx_.advance();
      Node* q = ast_.mk("QualifiedNameExpr");
      ast_.add(q, n);
      q->text = take();
      n = q;
    }
    return n;
  }

  // ---- token helpers ----
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
    if (!is_punct(p)) throw ParseError("expected " + std::string(p) + " got " + lx_.cur().text);
    lx_.advance();
  }
  void skip_modifiers_annotations() {
    static const std::set<std::string> mods = {
        "public", "private", "protected", "static", "final", "abstract",
        "native", "synchronized", "transient", "volatile", "strictfp",
        "default"};
    for (;;) {
      if (lx_.cur().kind == Tok::Keyword && mods.count(lx_.cur().text)) {
        lx_.advance();
        continue;
      }
      if (is_punct("@")) {  // annotation (not modeled as a node)
        lx_.advance();
        take();  // name
        while (is_punct(".")) { lx_.advance(); take(); }
        if (is_punct("(")) skip_balanced("(", ")");
        continue;
      }
      break;
    }
  }
  void skip_balanced(const char* open, const char* close) {
    int depth = 0;
    if (is_punct(open)) { ++depth; lx_.advance(); }
    while (depth > 0 && !at_end()) {
      if (is_punct(open)) ++depth;
      else if (is_punct(close)) --depth;
      lx_.advance();
    }
  }

  // ---- declarations ----

  Node* parse_type_declaration() {
    skip_modifiers_annotations();
    if (is_kw("class") || is_kw("interface")) return parse_class(false);
    if (is_kw("enum")) return parse_enum();
    if (at_end()) return nullptr;
    throw ParseError("unsupported type declaration at " + lx_.cur().text);
  }

  Node* parse_class(bool anonymous) {
    Node* cd = ast_.mk("ClassOrInterfaceDeclaration");
    if (!anonymous) {
      lx_.advance();  // class/interface
      // TypeDeclaration super-ctor: name first, then members; then typeParams,
      // extends, implements (NOTES.md). We add name first to match.
      Node* name = ast_.mk("NameExpr");
      name->text = take();
      ast_.add(cd, name);
      std::vector<Node*> tps, exts, impls;
      if (is_punct("<")) {  // type parameters
        lx_.advance();
        int depth = 1;
        while (depth > 0 && !at_end()) {
          if (is_punct("<")) ++depth;
          else if (is_punct(">")) --depth;
          else if (is_punct(">>")) depth -= 2;
          lx_.advance();
        }
      }
      if (is_kw("extends")) {
        lx_.advance();
        exts.push_back(parse_type());
        while (is_punct(",")) { lx_.advance(); exts.push_back(parse_type()); }
      }
      if (is_kw("implements")) {
        lx_.advance();
        impls.push_back(parse_type());
        while (is_punct(",")) { lx_.advance(); impls.push_back(parse_type()); }
      }
      std::vector<Node*> members = parse_class_body();
      for (Node* m : members) ast_.add(cd, m);
      for (Node* e : exts) ast_.add(cd, e);
      for (Node* i2 : impls) ast_.add(cd, i2);
    } else {
      std::vector<Node*> members = parse_class_body();
      for (Node* m : members) ast_.add(cd, m);
    }
    return cd;
  }

  Node* parse_enum() {
    lx_.advance();  // enum
    Node* ed = ast_.mk("EnumDeclaration");
    Node* name = ast_.mk("NameExpr");
    name->text = take();
    ast_.add(ed, name);
    if (is_kw("implements")) {
      lx_.advance();
      parse_type();
      while (is_punct(",")) { lx_.advance(); parse_type(); }
    }
    // enum bodies are walked for methods only (constants skipped coarsely)
    expect("{");
    while (!is_punct("}") && !at_end()) {
      if (is_punct(";")) { lx_.advance(); break; }
      take();  // constant name
      if (is_punct("(")) skip_balanced("(", ")");
      if (is_punct("{")) skip_balanced("{", "}");
      if (is_punct(",")) lx_.advance();
    }
    while (!is_punct("}") && !at_end()) {
      Node* m = parse_member();
      if (m) ast_.add(ed, m);
    }
    expect("}");
    return ed;
  }

  std::vector<Node*> parse_class_body() {
    std::vector<Node*> members;
    expect("{");
    while (!is_punct("}") && !at_end()) {
      // Per-member error recovery: a construct this grammar doesn't know
      // costs only its own member, not the whole file (JavaParser fails
      // whole-file too, but ITS grammar has no gaps — ours might). On a
      // parse error, rewind and skip one balanced member.
      Lexer save = lx_;
      try {
        Node* m = parse_member();
        if (m) members.push_back(m);
      } catch (const ParseError&) {
        lx_.restore_from(save);
        skip_one_member();
        ++skipped_members_;
      }
    }
    expect("}");
    return members;
  }

  // Skip one member: advance to the first ';' at brace depth 0 or past one
  // balanced '{...}' block (whichever comes first).
  void skip_one_member() {
    int depth = 0;
    bool moved = false;
    while (!at_end()) {
      const Token& t = lx_.cur();
      if (t.kind == Tok::Punct) {
        if (t.text == "{") {
          ++depth;
        } else if (t.text == "}") {
          if (depth == 0) return;  // enclosing class brace: stop before it
          --depth;
          lx_.advance();
          if (depth == 0) return;  // finished the member's block
          moved = true;
          continue;
        } else if (t.text == ";" && depth == 0) {
          lx_.advance();
          return;
        }
      }
      lx_.advance();
      moved = true;
    }
    (void)moved;
  }

  Node* parse_member() {
    skip_modifiers_annotations();
    if (is_punct(";")) { lx_.advance(); return nullptr; }
    if (is_kw("class") || is_kw("interface")) return parse_class(false);
    if (is_kw("enum")) return parse_enum();
    if (is_punct("{")) {  // initializer block
      Node* init = ast_.mk("InitializerDeclaration");
      ast_.add(init, parse_block());
      return init;
    }
    if (is_punct("<")) {  // method type parameters
      lx_.advance();
      int depth = 1;
      while (depth > 0 && !at_end()) {
        if (is_punct("<")) ++depth;
        else if (is_punct(">")) --depth;
        else if (is_punct(">>")) depth -= 2;
        lx_.advance();
      }
    }
    // constructor? ident followed by '('
    if (is_ident() && lx_.peek().kind == Tok::Punct && lx_.peek().text == "(") {
      Node* ctor = ast_.mk("ConstructorDeclaration");
      Node* name = ast_.mk("NameExpr");
      name->text = take();
      ast_.add(ctor, name);
      auto params = parse_parameters();
      for (Node* p : params) ast_.add(ctor, p);
      parse_throws(ctor);
      ast_.add(ctor, parse_block());
      return ctor;
    }
    // method or field: type name ...
    Node* type = parse_type();
    if (!is_ident() && !at_end())
      throw ParseError("expected member name, got " + lx_.cur().text);
    std::string name = take();
    if (is_punct("(")) {
      // MethodDeclaration children: elementType, name, params, brackets,
      // throws, body (typeParameters were skipped above)
      Node* md = ast_.mk("MethodDeclaration");
      md->methodName = name;
      ast_.add(md, type);
      Node* nm = ast_.mk("NameExpr");
      nm->text = name;
      ast_.add(md, nm);
      auto params = parse_parameters();
      for (Node* p : params) ast_.add(md, p);
      while (is_punct("[")) {  // brackets after parameter list
        lx_.advance();
        expect("]");
        ast_.add(md, ast_.mk("ArrayBracketPair"));
      }
      parse_throws(md);
      if (is_punct(";")) {
        lx_.advance();  // abstract/interface method: no body
      } else {
        ast_.add(md, parse_block());
      }
      return md;
    }
    // field declaration: elementType, variables..., brackets
    Node* fd = ast_.mk("FieldDeclaration");
    ast_.add(fd, type);
    ast_.add(fd, parse_variable_declarator(name));
    while (is_punct(",")) {
      lx_.advance();
      ast_.add(fd, parse_variable_declarator(take()));
    }
    expect(";");
    return fd;
  }

  void parse_throws(Node* owner) {
    if (is_kw("throws")) {
      lx_.advance();
      ast_.add(owner, parse_type());
      while (is_punct(",")) {
        lx_.advance();
        ast_.add(owner, parse_type());
      }
    }
  }

  std::vector<Node*> parse_parameters() {
    std::vector<Node*> out;
    expect("(");
    while (!is_punct(")") && !at_end()) {
      skip_modifiers_annotations();
      Node* p = ast_.mk("Parameter");
      Node* t = parse_type();
      bool varargs = false;
      if (is_punct("...")) { lx_.advance(); varargs = true; }
      (void)varargs;
      std::string nm = take();
      Node* id = ast_.mk("VariableDeclaratorId");
      id->text = nm;
      while (is_punct("[")) {  // brackets after id → part of the id toString
        lx_.advance();
        expect("]");
        id->text += "[]";
      }
      // Parameter children order: id BEFORE elementType (NOTES.md)
      ast_.add(p, id);
      ast_.add(p, t);
      out.push_back(p);
      if (is_punct(",")) lx_.advance();
    }
    expect(")");
    return out;
  }

  Node* parse_variable_declarator(std::string name) {
    Node* vd = ast_.mk("VariableDeclarator");
    Node* id = ast_.mk("VariableDeclaratorId");
    id->text = name;
    while (is_punct("[")) {
      lx_.advance();
      expect("]");
      id->text += "[]";
    }
    ast_.add(vd, id);
    if (is_punct("=")) {
      lx_.advance();
      ast_.add(vd, parse_variable_init());
    }
    return vd;
  }

  Node* parse_variable_init() {
    if (is_punct("{")) return parse_array_initializer();
    return parse_expression();
  }

  Node* parse_array_initializer() {
    Node* ai = ast_.mk("ArrayInitializerExpr");
    expect("{");
    while (!is_punct("}") && !at_end()) {
      ast_.add(ai, parse_variable_init());
      if (is_punct(",")) lx_.advance();
    }
    expect("}");
    return ai;
  }

  // ---- types ----

  bool looks_like_type() {
    const Token& t = lx_.cur();
    if (t.kind == Tok::Keyword)
      return t.text == "boolean" || t.text == "byte" || t.text == "char" ||
             t.text == "short" || t.text == "int" || t.text == "long" ||
             t.text == "float" || t.text == "double" || t.text == "void";
    return t.kind == Tok::Ident;
  }

  Node* parse_type() {
    Node* base = parse_non_array_type();
    while (is_punct("[")) {
      lx_.advance();
      expect("]");
      Node* at = ast_.mk("ArrayType");
      ast_.add(at, base);
      base = at;
    }
    return base;
  }

  Node* parse_non_array_type() {
    static const std::set<std::string> prim = {"boolean", "byte", "char",
                                              "short", "int", "long", "float",
                                              "double"};
    if (lx_.cur().kind == Tok::Keyword && prim.count(lx_.cur().text)) {
      Node* p = ast_.mk("PrimitiveType");
      p->text = take();
      return p;
    }
    if (is_kw("void")) {
      Node* v = ast_.mk("VoidType");
      v->text = take();
      return v;
    }
    // ClassOrInterfaceType with optional scope chain and generics
    Node* t = parse_coit();
    while (is_punct(".") && lx_.peek().kind == Tok::Ident) {
      lx_.advance();
      Node* outer = t;
      t = parse_coit();
      // scope is added as the FIRST child of the new COIT
      t->kids.insert(t->kids.begin(), outer);
      outer->parent = t;
    }
    return t;
  }

  Node* parse_coit() {
    Node* t = ast_.mk("ClassOrInterfaceType");
    t->coitName = take();
    t->text = t->coitName;
    t->isBoxed = kBoxed.count(t->coitName) > 0;
    if (t->isBoxed) t->unboxed = unbox(t->coitName);
    if (is_punct("<") && type_args_follow()) {
      lx_.advance();
      if (is_punct(">")) {  // diamond
        lx_.advance();
      } else {
        ast_.add(t, parse_type_argument());
        while (is_punct(",")) {
          lx_.advance();
          ast_.add(t, parse_type_argument());
        }
        close_generic();
        t->genericParent = !t->kids.empty();
      }
    }
    return t;
  }

  // after '<' in a type position we may actually be at a comparison in
  // expression context; callers in type context always want type args.
  bool type_args_follow() { return true; }

  void close_generic() {
    if (is_punct(">")) { lx_.advance(); return; }
    if (is_punct(">>") || is_punct(">>>")) { lx_.split_gt(); return; }
    throw ParseError("expected > in type arguments, got " + lx_.cur().text);
  }

  Node* parse_type_argument() {
    if (is_punct("?")) {
      lx_.advance();
      Node* w = ast_.mk("WildcardType");
      w->text = "?";
      if (is_kw("extends")) {
        lx_.advance();
        ast_.add(w, parse_type());
      } else if (is_kw("super")) {
        lx_.advance();
        ast_.add(w, parse_type());
      }
      return w;
    }
    return parse_type();
  }

  // ---- statements ----

  Node* parse_block() {
    Node* b = ast_.mk("BlockStmt");
    b->isStmt = true;
    expect("{");
    while (!is_punct("}") && !at_end()) ast_.add(b, parse_statement());
    expect("}");
    return b;
  }

  Node* mk_stmt(const char* t) {
    Node* n = ast_.mk(t);
    n->isStmt = true;
    return n;
  }

  Node* parse_statement() {
    if (is_punct("{")) return parse_block();
    if (is_punct(";")) { lx_.advance(); return mk_stmt("EmptyStmt"); }
    if (is_kw("if")) return parse_if();
    if (is_kw("while")) return parse_while();
    if (is_kw("do")) return parse_do();
    if (is_kw("for")) return parse_for();
    if (is_kw("return")) {
      Node* r = mk_stmt("ReturnStmt");
      lx_.advance();
      if (!is_punct(";")) ast_.add(r, parse_expression());
      expect(";");
      return r;
    }
    if (is_kw("throw")) {
      Node* t = mk_stmt("ThrowStmt");
      lx_.advance();
      ast_.add(t, parse_expression());
      expect(";");
      return t;
    }
    if (is_kw("break")) {
      Node* b = mk_stmt("BreakStmt");
      lx_.advance();
      if (is_ident()) b->text = take();
      expect(";");
      return b;
    }
    if (is_kw("continue")) {
      Node* c = mk_stmt("ContinueStmt");
      lx_.advance();
      if (is_ident()) c->text = take();
      expect(";");
      return c;
    }
    if (is_kw("try")) return parse_try();
    if (is_kw("switch")) return parse_switch();
    if (is_kw("synchronized")) {
      Node* s = mk_stmt("SynchronizedStmt");
      lx_.advance();
      expect("(");
      ast_.add(s, parse_expression());
      expect(")");
      ast_.add(s, parse_block());
      return s;
    }
    if (is_kw("assert")) {
      Node* a = mk_stmt("AssertStmt");
      lx_.advance();
      ast_.add(a, parse_expression());
      if (is_punct(":")) {
        lx_.advance();
        ast_.add(a, parse_expression());
      }
      expect(";");
      return a;
    }
    if (is_kw("class")) {  // local class
      Node* lc = mk_stmt("TypeDeclarationStmt");
      ast_.add(lc, parse_class(false));
      return lc;
    }
    // labeled statement: ident ':'
    if (is_ident() && lx_.peek().kind == Tok::Punct && lx_.peek().text == ":") {
      Node* l = mk_stmt("LabeledStmt");
      l->text = take();
      lx_.advance();  // ':'
      ast_.add(l, parse_statement());
      return l;
    }
    // local variable declaration or expression statement
    if (starts_local_var_decl()) {
      Node* es = mk_stmt("ExpressionStmt");
      ast_.add(es, parse_var_decl_expr());
      expect(";");
      return es;
    }
    Node* es = mk_stmt("ExpressionStmt");
    ast_.add(es, parse_expression());
    expect(";");
    return es;
  }

  Node* parse_var_decl_expr() {
    skip_modifiers_annotations();
    Node* vde = ast_.mk("VariableDeclarationExpr");
    ast_.add(vde, parse_type());
    ast_.add(vde, parse_variable_declarator(take()));
    while (is_punct(",")) {
      lx_.advance();
      ast_.add(vde, parse_variable_declarator(take()));
    }
    return vde;
  }

  // Heuristic lookahead: "Type ident (=|;|,|[...]ident)" starts a local decl.
  bool starts_local_var_decl() {
    if (is_kw("final")) return true;
    if (!looks_like_type()) return false;
    // primitive type always a decl
    if (lx_.cur().kind == Tok::Keyword) return true;
    // Save-and-restore by re-lexing is expensive; use a bounded textual scan.
    return scan_decl_ahead();
  }

  bool scan_decl_ahead() {
    // Clone the lexer state is not supported; instead do a conservative
    // grammar probe on a copy of the remaining token stream.
    Lexer probe = lx_;  // Lexer is copyable (indexes into the same string)
    // type: Name(.Name)*(<...>)?([])*  then identifier
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
      int depth = 0;
      do {
        if (at("<")) depth += 1;
        else if (at(">")) depth -= 1;
        else if (at(">>")) depth -= 2;
        else if (at(">>>")) depth -= 3;
        else if (probe.cur().kind == Tok::End) return false;
        else if (at(";") || at("(") || at("{")) return false;
        probe.advance();
      } while (depth > 0);
    }
    while (at("[")) {
      probe.advance();
      if (!at("]")) return false;
      probe.advance();
    }
    return probe.cur().kind == Tok::Ident;
  }

  Node* parse_if() {
    Node* s = mk_stmt("IfStmt");
    lx_.advance();
    expect("(");
    ast_.add(s, parse_expression());
    expect(")");
    ast_.add(s, parse_statement());
    if (is_kw("else")) {
      lx_.advance();
      ast_.add(s, parse_statement());
    }
    return s;
  }

  Node* parse_while() {
    Node* s = mk_stmt("WhileStmt");
    lx_.advance();
    expect("(");
    ast_.add(s, parse_expression());
    expect(")");
    ast_.add(s, parse_statement());
    return s;
  }

  Node* parse_do() {
    Node* s = mk_stmt("DoStmt");
    lx_.advance();
    ast_.add(s, parse_statement());
    if (!is_kw("while")) throw ParseError("expected while after do");
    lx_.advance();
    expect("(");
    ast_.add(s, parse_expression());
    expect(")");
    expect(";");
    return s;
  }

  Node* parse_for() {
    lx_.advance();
    expect("(");
    // foreach? probe: [final] Type ident ':'
    {
      Lexer probe = lx_;
      auto at = [&](const char* p) {
        return probe.cur().kind == Tok::Punct && probe.cur().text == p;
      };
      int guard = 0;
      bool colon = false;
      int depth = 0;
      while (probe.cur().kind != Tok::End && guard++ < 80) {
        if (at("(")) ++depth;
        if (at(")")) { if (depth == 0) break; --depth; }
        if (at(";") && depth == 0) break;
        if (at(":") && depth == 0) { colon = true; break; }
        probe.advance();
      }
      if (colon) {
        Node* fe = mk_stmt("ForeachStmt");
        ast_.add(fe, parse_var_decl_expr());
        expect(":");
        ast_.add(fe, parse_expression());
        expect(")");
        ast_.add(fe, parse_statement());
        return fe;
      }
    }
    // classic for — children order: compare FIRST, then init, update, body
    Node* f = mk_stmt("ForStmt");
    std::vector<Node*> init;
    if (!is_punct(";")) {
      if (starts_local_var_decl()) {
        init.push_back(parse_var_decl_expr());
      } else {
        init.push_back(parse_expression());
        while (is_punct(",")) {
          lx_.advance();
          init.push_back(parse_expression());
        }
      }
    }
    expect(";");
    Node* compare = nullptr;
    if (!is_punct(";")) compare = parse_expression();
    expect(";");
    std::vector<Node*> update;
    if (!is_punct(")")) {
      update.push_back(parse_expression());
      while (is_punct(",")) {
        lx_.advance();
        update.push_back(parse_expression());
      }
    }
    expect(")");
    Node* body = parse_statement();
    ast_.add(f, compare);
    for (Node* n : init) ast_.add(f, n);
    for (Node* n : update) ast_.add(f, n);
    ast_.add(f, body);
    return f;
  }

  Node* parse_try() {
    Node* t = mk_stmt("TryStmt");
    lx_.advance();
    if (is_punct("(")) {  // try-with-resources
      lx_.advance();
      ast_.add(t, parse_var_decl_expr());
      while (is_punct(";")) {
        lx_.advance();
        if (is_punct(")")) break;
        ast_.add(t, parse_var_decl_expr());
      }
      expect(")");
    }
    ast_.add(t, parse_block());
    while (is_kw("catch")) {
      lx_.advance();
      Node* cc = ast_.mk("CatchClause");
      expect("(");
      skip_modifiers_annotations();
      Node* p = ast_.mk("Parameter");
      Node* ty = parse_type();
      while (is_punct("|")) {  // union type
        lx_.advance();
        Node* u = ast_.mk("UnionType");
        ast_.add(u, ty);
        ast_.add(u, parse_type());
        ty = u;
      }
      Node* id = ast_.mk("VariableDeclaratorId");
      id->text = take();
      ast_.add(p, id);
      ast_.add(p, ty);
      expect(")");
      ast_.add(cc, p);
      ast_.add(cc, parse_block());
      ast_.add(t, cc);
    }
    if (is_kw("finally")) {
      lx_.advance();
      ast_.add(t, parse_block());
    }
    return t;
  }

  Node* parse_switch() {
    Node* s = mk_stmt("SwitchStmt");
    lx_.advance();
    expect("(");
    ast_.add(s, parse_expression());
    expect(")");
    expect("{");
    while (!is_punct("}") && !at_end()) {
      Node* e = mk_stmt("SwitchEntryStmt");
      if (is_kw("case")) {
        lx_.advance();
        ast_.add(e, parse_expression());
        expect(":");
      } else if (is_kw("default")) {
        lx_.advance();
        expect(":");
      } else {
        throw ParseError("expected case/default, got " + lx_.cur().text);
      }
      while (!is_kw("case") && !is_kw("default") && !is_punct("}") && !at_end())
        ast_.add(e, parse_statement());
      ast_.add(s, e);
    }
    expect("}");
    return s;
  }

  // ---- expressions ----

  // Any expression position may hold a lambda (JavaParser accepts
  // `return x -> e;`, initializers, ternary arms — not just call args)
  Node* parse_expression() { return parse_lambda_or_expr(); }

  Node* parse_assignment() {
    Node* lhs = parse_ternary();
    static const std::pair<const char*, const char*> ops[] = {
        {"=", "assign"}, {"+=", "plus"}, {"-=", "minus"}, {"*=", "star"},
        {"/=", "slash"}, {"&=", "and"}, {"|=", "or"}, {"^=", "xor"},
        {"%=", "rem"}, {"<<=", "lShift"}, {">>=", "rSignedShift"},
        {">>>=", "rUnsignedShift"}};
    for (auto& [sym, name] : ops) {
      if (is_punct(sym)) {
        lx_.advance();
        Node* a = ast_.mk("AssignExpr");
        a->op = name;
        ast_.add(a, lhs);
        ast_.add(a, parse_expression());  // RHS may be a lambda
        return a;
      }
    }
    return lhs;
  }

  Node* parse_ternary() {
    Node* c = parse_binary(0);
    if (is_punct("?")) {
      lx_.advance();
      Node* t = ast_.mk("ConditionalExpr");
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
    const char* name;
    int prec;
  };
  static const BinOp* find_binop(const Token& t, int generic_guard) {
    static const BinOp ops[] = {
        {"||", "or", 1},        {"&&", "and", 2},      {"|", "binOr", 3},
        {"^", "xor", 4},        {"&", "binAnd", 5},    {"==", "equals", 6},
        {"!=", "notEquals", 6}, {"<", "less", 7},      {">", "greater", 7},
        {"<=", "lessEquals", 7},{">=", "greaterEquals", 7},
        {"<<", "lShift", 8},    {">>", "rSignedShift", 8},
        {">>>", "rUnsignedShift", 8},
        {"+", "plus", 9},       {"-", "minus", 9},     {"*", "times", 10},
        {"/", "divide", 10},    {"%", "remainder", 10}};
    (void)generic_guard;
    if (t.kind != Tok::Punct) return nullptr;
    for (const auto& op : ops)
      if (t.text == op.sym) return &op;
    return nullptr;
  }

  Node* parse_binary(int min_prec) {
    Node* lhs = parse_instanceof();
    for (;;) {
      const BinOp* op = find_binop(lx_.cur(), 0);
      if (!op || op->prec < min_prec) return lhs;
      lx_.advance();
      Node* rhs = parse_binary(op->prec + 1);
      Node* b = ast_.mk("BinaryExpr");
      b->op = op->name;
      ast_.add(b, lhs);
      ast_.add(b, rhs);
      lhs = b;
    }
  }

  Node* parse_instanceof() {
    Node* e = parse_unary();
    while (is_kw("instanceof")) {
      lx_.advance();
      Node* io = ast_.mk("InstanceOfExpr");
      ast_.add(io, e);
      ast_.add(io, parse_type());
      e = io;
    }
    return e;
  }

  Node* parse_unary() {
    static const std::pair<const char*, const char*> pre[] = {
        {"+", "positive"}, {"-", "negative"}, {"++", "preIncrement"},
        {"--", "preDecrement"}, {"!", "not"}, {"~", "inverse"}};
    for (auto& [sym, name] : pre) {
      if (is_punct(sym)) {
        lx_.advance();
        Node* u = ast_.mk("UnaryExpr");
        u->op = name;
        ast_.add(u, parse_unary());
        return u;
      }
    }
    // cast: '(' Type ')' unary  — probe
    if (is_punct("(") && cast_ahead()) {
      lx_.advance();
      Node* c = ast_.mk("CastExpr");
      ast_.add(c, parse_type());
      expect(")");
      ast_.add(c, parse_unary());
      return c;
    }
    return parse_postfix();
  }

  bool cast_ahead() {
    Lexer probe = lx_;
    auto at = [&](const char* p) {
      return probe.cur().kind == Tok::Punct && probe.cur().text == p;
    };
    probe.advance();  // '('
    bool prim = probe.cur().kind == Tok::Keyword &&
                (probe.cur().text == "boolean" || probe.cur().text == "byte" ||
                 probe.cur().text == "char" || probe.cur().text == "short" ||
                 probe.cur().text == "int" || probe.cur().text == "long" ||
                 probe.cur().text == "float" || probe.cur().text == "double");
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
    if (at("<")) {
      int depth = 0, guard = 0;
      do {
        if (at("<")) depth += 1;
        else if (at(">")) depth -= 1;
        else if (at(">>")) depth -= 2;
        else if (at(">>>")) depth -= 3;
        else if (probe.cur().kind == Tok::End || at("(") || at(";")) return false;
        probe.advance();
        if (guard++ > 60) return false;
      } while (depth > 0);
    }
    while (at("[")) {
      probe.advance();
      if (!at("]")) return false;
      probe.advance();
    }
    if (!at(")")) return false;
    probe.advance();
    // after a cast: unary expression starts
    const Token& nx = probe.cur();
    if (nx.kind == Tok::Ident || nx.kind == Tok::IntLit ||
        nx.kind == Tok::LongLit || nx.kind == Tok::FloatLit ||
        nx.kind == Tok::DoubleLit || nx.kind == Tok::CharLit ||
        nx.kind == Tok::StringLit)
      return true;
    if (nx.kind == Tok::Keyword &&
        (nx.text == "new" || nx.text == "this" || nx.text == "super" ||
         nx.text == "true" || nx.text == "false" || nx.text == "null"))
      return true;
    if (nx.kind == Tok::Punct && (nx.text == "(" || nx.text == "!" ||
                                  nx.text == "~"))
      return true;
    return false;
  }

  Node* parse_postfix() {
    Node* e = parse_primary();
    for (;;) {
      if (is_punct(".")) {
        lx_.advance();
        if (is_kw("new")) {  // qualified new — treat as ObjectCreation w/scope
          lx_.advance();
          e = parse_object_creation(e);
          continue;
        }
        if (is_kw("class")) {  // shouldn't happen after '.', but guard
          lx_.advance();
          Node* ce = ast_.mk("ClassExpr");
          ast_.add(ce, e);
          e = ce;
          continue;
        }
        if (is_kw("this")) {
          lx_.advance();
          Node* te = ast_.mk("ThisExpr");
          ast_.add(te, e);
          e = te;
          continue;
        }
        if (is_punct("<")) {  // explicit type args on call — skip
          skip_balanced("<", ">");
        }
        std::string nm = take();
        if (is_punct("(")) {
          Node* mc = ast_.mk("MethodCallExpr");
          ast_.add(mc, e);
          Node* n = ast_.mk("NameExpr");
          n->text = nm;
          ast_.add(mc, n);
          parse_args(mc);
          e = mc;
        } else {
          Node* fa = ast_.mk("FieldAccessExpr");
          ast_.add(fa, e);
          Node* f = ast_.mk("NameExpr");
          f->text = nm;
          ast_.add(fa, f);
          e = fa;
        }
        continue;
      }
      if (is_punct("[")) {
        lx_.advance();
        Node* aa = ast_.mk("ArrayAccessExpr");
        ast_.add(aa, e);
        ast_.add(aa, parse_expression());
        expect("]");
        e = aa;
        continue;
      }
      if (is_punct("++")) {
        lx_.advance();
        Node* u = ast_.mk("UnaryExpr");
        u->op = "posIncrement";
        ast_.add(u, e);
        e = u;
        continue;
      }
      if (is_punct("--")) {
        lx_.advance();
        Node* u = ast_.mk("UnaryExpr");
        u->op = "posDecrement";
        ast_.add(u, e);
        e = u;
        continue;
      }
      if (is_punct("::")) {  // method reference
        lx_.advance();
        Node* mr = ast_.mk("MethodReferenceExpr");
        ast_.add(mr, e);
        mr->text = is_kw("new") ? take() : take();
        e = mr;
        continue;
      }
      return e;
    }
  }

  void parse_args(Node* call) {
    expect("(");
    while (!is_punct(")") && !at_end()) {
      ast_.add(call, parse_expression());
      if (is_punct(",")) lx_.advance();
    }
    expect(")");
  }

  Node* parse_lambda_or_expr() {
    // lambda probe: ident '->' | '(' [params] ')' '->'
    if (is_ident() && lx_.peek().kind == Tok::Punct && lx_.peek().text == "->") {
      Node* le = ast_.mk("LambdaExpr");
      Node* p = ast_.mk("Parameter");
      Node* id = ast_.mk("VariableDeclaratorId");
      id->text = take();
      ast_.add(p, id);
      ast_.add(le, p);
      lx_.advance();  // ->
      ast_.add(le, parse_lambda_body());
      return le;
    }
    if (is_punct("(")) {
      Lexer probe = lx_;
      int depth = 0, guard = 0;
      do {
        if (probe.cur().kind == Tok::Punct && probe.cur().text == "(") ++depth;
        else if (probe.cur().kind == Tok::Punct && probe.cur().text == ")") --depth;
        probe.advance();
        if (guard++ > 200 || probe.cur().kind == Tok::End) break;
      } while (depth > 0);
      if (probe.cur().kind == Tok::Punct && probe.cur().text == "->") {
        Node* le = ast_.mk("LambdaExpr");
        lx_.advance();  // (
        while (!is_punct(")") && !at_end()) {
          skip_modifiers_annotations();
          Node* p = ast_.mk("Parameter");
          // typed or untyped lambda param
          if (looks_like_type() &&
              (lx_.peek().kind == Tok::Ident)) {
            Node* t = parse_type();
            Node* id = ast_.mk("VariableDeclaratorId");
            id->text = take();
            ast_.add(p, id);
            ast_.add(p, t);
          } else {
            Node* id = ast_.mk("VariableDeclaratorId");
            id->text = take();
            ast_.add(p, id);
          }
          ast_.add(le, p);
          if (is_punct(",")) lx_.advance();
        }
        expect(")");
        expect("->");
        ast_.add(le, parse_lambda_body());
        return le;
      }
    }
    return parse_assignment();
  }

  Node* parse_lambda_body() {
    if (is_punct("{")) return parse_block();
    Node* es = mk_stmt("ExpressionStmt");
    ast_.add(es, parse_expression());
    return es;
  }

  Node* parse_object_creation(Node* scope) {
    Node* oc = ast_.mk("ObjectCreationExpr");
    if (scope) ast_.add(oc, scope);
    // array creation? new Type[...] or new Type[] {...}
    Node* t = parse_non_array_type();
    if (is_punct("[")) {
      Node* ac = ast_.mk("ArrayCreationExpr");
      // levels FIRST, then type, then initializer (NOTES.md)
      std::vector<Node*> levels;
      bool any_dim = false;
      while (is_punct("[")) {
        lx_.advance();
        if (!is_punct("]")) {
          levels.push_back(parse_expression());
          any_dim = true;
        }
        expect("]");
      }
      (void)any_dim;
      for (Node* l : levels) ast_.add(ac, l);
      ast_.add(ac, t);
      if (is_punct("{")) ast_.add(ac, parse_array_initializer());
      return ac;
    }
    ast_.add(oc, t);
    parse_args(oc);
    if (is_punct("{")) {  // anonymous class body
      Node* anon = parse_class(true);
      for (Node* m : anon->kids) ast_.add(oc, m);
    }
    return oc;
  }

  Node* parse_primary() {
    const Token& t = lx_.cur();
    switch (t.kind) {
      case Tok::IntLit: {
        Node* n = ast_.mk("IntegerLiteralExpr");
        n->isIntLit = true;
        n->text = take();
        return n;
      }
      case Tok::LongLit: {
        Node* n = ast_.mk("LongLiteralExpr");
        n->text = take();
        return n;
      }
      case Tok::FloatLit:
      case Tok::DoubleLit: {
        Node* n = ast_.mk("DoubleLiteralExpr");
        n->text = take();
        return n;
      }
      case Tok::CharLit: {
        Node* n = ast_.mk("CharLiteralExpr");
        n->text = "'" + take() + "'";
        return n;
      }
      case Tok::StringLit: {
        Node* n = ast_.mk("StringLiteralExpr");
        n->text = "\"" + take() + "\"";
        return n;
      }
      default:
        break;
    }
    if (is_kw("true") || is_kw("false")) {
      Node* n = ast_.mk("BooleanLiteralExpr");
      n->text = take();
      return n;
    }
    if (is_kw("null")) {
      lx_.advance();
      Node* n = ast_.mk("NullLiteralExpr");
      n->isNullLit = true;
      n->text = "null";
      return n;
    }
    if (is_kw("this")) {
      lx_.advance();
      Node* n = ast_.mk("ThisExpr");
      n->text = "this";
      return n;
    }
    if (is_kw("super")) {
      lx_.advance();
      Node* n = ast_.mk("SuperExpr");
      n->text = "super";
      return n;
    }
    if (is_kw("new")) {
      lx_.advance();
      return parse_object_creation(nullptr);
    }
    if (is_punct("(")) {
      lx_.advance();
      Node* en = ast_.mk("EnclosedExpr");
      ast_.add(en, parse_expression());
      expect(")");
      return en;
    }
    if (is_ident()) {
      // possibly Type.class or plain name / method call
      std::string nm = take();
      if (is_punct("(")) {
        Node* mc = ast_.mk("MethodCallExpr");
        Node* n = ast_.mk("NameExpr");
        n->text = nm;
        ast_.add(mc, n);
        parse_args(mc);
        return mc;
      }
      if (is_punct(".") && lx_.peek().kind == Tok::Keyword &&
          lx_.peek().text == "class") {
        lx_.advance();
        lx_.advance();
        Node* ce = ast_.mk("ClassExpr");
        Node* ty = ast_.mk("ClassOrInterfaceType");
        ty->coitName = nm;
        ty->text = nm;
        ast_.add(ce, ty);
        return ce;
      }
      Node* n = ast_.mk("NameExpr");
      n->text = nm;
      return n;
    }
    if (lx_.cur().kind == Tok::Keyword &&
        (lx_.cur().text == "boolean" || lx_.cur().text == "byte" ||
         lx_.cur().text == "char" || lx_.cur().text == "short" ||
         lx_.cur().text == "int" || lx_.cur().text == "long" ||
         lx_.cur().text == "float" || lx_.cur().text == "double" ||
         lx_.cur().text == "void")) {
      // primitive.class
      Node* ty = parse_type();
      if (is_punct(".") ) {
        lx_.advance();
        if (is_kw("class")) lx_.advance();
      }
      Node* ce = ast_.mk("ClassExpr");
      ast_.add(ce, ty);
      return ce;
    }
    throw ParseError("unexpected token '" + lx_.cur().text + "'");
  }

  Lexer lx_;
  int skipped_members_ = 0;
  Ast& ast_;
};

// ---------------------------------------------------------------------------
// Property computation + path generation (FeatureExtractor semantics)
// ---------------------------------------------------------------------------

static std::string normalize_name(const std::string& original_in,
                                  const std::string& def) {
  std::string s;
  s.reserve(original_in.size());
  for (char c : original_in)
    s += (char)tolower((unsigned char)c);
  // remove literal backslash-n sequences
  std::string t;
  for (size_t i = 0; i < s.size();) {
    if (s[i] == '\\' && i + 1 < s.size() && s[i + 1] == 'n') {
      i += 2;
      continue;
    }
    t += s[i++];
  }
  // remove "//s+" occurrences (the reference regex literally matches this)
  std::string u;
  for (size_t i = 0; i < t.size();) {
    if (t[i] == '/' && i + 1 < t.size() && t[i + 1] == '/' && i + 2 < t.size() &&
        t[i + 2] == 's') {
      size_t j = i + 2;
      while (j < t.size() && t[j] == 's') ++j;
      i = j;
      continue;
    }
    u += t[i++];
  }
  // remove quotes/apostrophes/commas and non-printables
  std::string v;
  for (char c : u) {
    if (c == '"' || c == '\'' || c == ',') continue;
    if ((unsigned char)c < 0x20 || (unsigned char)c > 0x7E) continue;
    v += c;
  }
  std::string stripped;
  for (char c : v)
    if (isalpha((unsigned char)c)) stripped += c;
  if (!stripped.empty()) return stripped;
  std::string careful;
  for (char c : v) careful += (c == ' ') ? '_' : c;
  if (careful.empty()) return def;
  return careful;
}

static std::vector<std::string> split_subtokens(const std::string& in) {
  // reference regex: (?<=[a-z])(?=[A-Z]) | _ | [0-9] | (?<=[A-Z])(?=[A-Z][a-z]) | \s+
  std::string s = in;
  // trim
  size_t b = s.find_first_not_of(" \t\r\n");
  size_t e = s.find_last_not_of(" \t\r\n");
  if (b == std::string::npos) return {};
  s = s.substr(b, e - b + 1);
  std::vector<std::string> parts;
  std::string cur;
  auto flush = [&]() {
    if (!cur.empty()) {
      std::string n = normalize_name(cur, "");
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
      bool acronym = i + 1 < s.size() && isupper((unsigned char)p) &&
                     isupper((unsigned char)c) &&
                     islower((unsigned char)s[i + 1]);
      if (camel || acronym) flush();
    }
    cur += c;
  }
  flush();
  return parts;
}

struct ExtractorOptions {
  int max_path_length = 8;
  int max_path_width = 2;
  int min_code_len = 1;
  int max_code_len = 10000;
  int max_child_id = INT32_MAX;
  bool no_hash = false;
  int num_threads = 32;
};

static const std::set<std::string> kParentTypesAddChildId = {
    "AssignExpr", "ArrayAccessExpr", "FieldAccessExpr", "MethodCallExpr"};

static int32_t java_hash(const std::string& s) {
  int32_t h = 0;
  for (unsigned char c : s) h = (int32_t)((uint32_t)h * 31u + c);
  return h;
}

struct LeafInfo {
  Node* node;
};

// Compute Property Type/Name for a node (Property.java:23-76 semantics).
static void compute_property(Node* n) {
  std::string type = n->type;
  if (n->type == "ClassOrInterfaceType" && n->isBoxed) type = "PrimitiveType";
  if (!n->op.empty()) type += ":" + n->op;
  bool is_leaf = n->kids.empty();
  if (n->genericParent && is_leaf) type = "GenericClass";

  std::string name = normalize_name(n->text, "BLANK");
  if (name.size() > 50) {
    name = name.substr(0, 50);
  } else if (n->type == "ClassOrInterfaceType" && n->isBoxed) {
    name = n->unboxed;
  }
  // METHOD_NAME masking
  if (n->type == "NameExpr" && n->parent &&
      n->parent->type == "MethodDeclaration") {
    name = "METHOD_NAME";
  }
  n->propType = type;
  n->propName = name;
}

static void collect_subtree(Node* n, std::vector<Node*>& leaves) {
  // preorder, assigns childIds and Properties (LeavesCollectorVisitor)
  if (n->parent) {
    int cid = 0;
    for (Node* c : n->parent->kids) {
      if (c == n) break;
      ++cid;
    }
    n->childId = cid;
  } else {
    n->childId = 0;
  }
  compute_property(n);
  if (n->kids.empty() && !n->isStmt) {
    const std::string& ts = n->text;
    if (!ts.empty() && (ts != "null" || n->isNullLit)) leaves.push_back(n);
  }
  for (Node* c : n->kids) collect_subtree(c, leaves);
}

static bool in_subtree_props(Node* n) { return !n->propType.empty(); }

static std::string generate_path(Node* source, Node* target,
                                 const ExtractorOptions& opt) {
  std::vector<Node*> ss, ts;
  for (Node* c = source; c; c = c->parent) ss.push_back(c);
  for (Node* c = target; c; c = c->parent) ts.push_back(c);
  int si = (int)ss.size() - 1, ti = (int)ts.size() - 1;
  int common = 0;
  while (si >= 0 && ti >= 0 && ss[si] == ts[ti]) {
    ++common;
    --si;
    --ti;
  }
  int path_length = (int)ss.size() + (int)ts.size() - 2 * common;
  if (path_length > opt.max_path_length) return "";
  if (si >= 0 && ti >= 0) {
    int width = ts[ti]->childId - ss[si]->childId;
    if (width > opt.max_path_width) return "";
  }
  auto sat = [&](int cid) {
    return std::to_string(std::min(cid, opt.max_child_id));
  };
  std::string out;
  for (int i = 0; i < (int)ss.size() - common; ++i) {
    Node* cur = ss[i];
    std::string child_id;
    std::string parent_raw =
        (cur->parent && in_subtree_props(cur->parent)) ? cur->parent->type : "";
    if (i == 0 || kParentTypesAddChildId.count(parent_raw))
      child_id = sat(cur->childId);
    out += "(" + cur->propType + child_id + ")^";
  }
  Node* common_node = ss[(int)ss.size() - common];
  std::string common_child_id;
  std::string common_parent_raw =
      (common_node->parent && in_subtree_props(common_node->parent))
          ? common_node->parent->type
          : "";
  if (kParentTypesAddChildId.count(common_parent_raw))
    common_child_id = sat(common_node->childId);
  out += "(" + common_node->propType + common_child_id + ")";
  for (int i = (int)ts.size() - common - 1; i >= 0; --i) {
    Node* cur = ts[i];
    std::string child_id;
    if (i == 0 || kParentTypesAddChildId.count(cur->type))
      child_id = sat(cur->childId);
    out += "_(" + cur->propType + child_id + ")";
  }
  return out;
}

static void find_methods(Node* n, std::vector<Node*>& methods) {
  if (n->type == "MethodDeclaration") {
    bool has_body = false;
    for (Node* c : n->kids)
      if (c->type == "BlockStmt") has_body = true;
    if (has_body) methods.push_back(n);
  }
  for (Node* c : n->kids) find_methods(c, methods);
}

static std::string extract_file_content(const std::string& code,
                                        const ExtractorOptions& opt) {
  Ast ast;
  Node* cu = nullptr;
  // 3-stage parse retry (FeatureExtractor.java:51-75). A stage that parsed
  // but DROPPED members via the per-member recovery is only accepted if no
  // later stage does better (most methods wins; clean parses win outright).
  std::vector<std::string> attempts = {
      code,
      "public class Test {SomeUnknownReturnType f() {" + code +
          "return noSuchReturnValue; }}",
      "public class Test {" + code + "}"};
  Ast best_ast;
  Node* best_cu = nullptr;
  size_t best_methods = 0;
  for (const std::string& attempt : attempts) {
    Ast fresh;
    try {
      Parser p(attempt, fresh);
      Node* parsed = p.parse_compilation_unit();
      if (p.skipped_members() == 0) {
        ast.nodes_.swap(fresh.nodes_);
        cu = parsed;
        break;
      }
      std::vector<Node*> ms;
      find_methods(parsed, ms);
      // keep the FIRST stage that recovered any methods: later stages wrap
      // the snippet in synthetic members that would leak into the output
      if (best_cu == nullptr && !ms.empty()) {
        best_ast.nodes_.swap(fresh.nodes_);
        best_cu = parsed;
        best_methods = ms.size();
      }
    } catch (const ParseError&) {
      continue;
    }
  }
  if (!cu && best_cu) {
    ast.nodes_.swap(best_ast.nodes_);
    cu = best_cu;
  }
  if (!cu) return "";

  std::vector<Node*> methods;
  find_methods(cu, methods);
  std::vector<std::string> lines;
  for (Node* md : methods) {
    // method target label
    std::string norm = normalize_name(md->methodName, "BLANK");
    auto parts = split_subtokens(md->methodName);
    std::string label = norm;
    if (!parts.empty()) {
      label = parts[0];
      for (size_t i = 1; i < parts.size(); ++i) label += "|" + parts[i];
    }
    std::vector<Node*> leaves;
    collect_subtree(md, leaves);
    // method length filter: count leaves-derived statements is wrong; use
    // a simple statement-line proxy: number of statement nodes
    long stmt_count = 0;
    std::function<void(Node*)> cnt = [&](Node* x) {
      if (x->isStmt && x->type != "BlockStmt") ++stmt_count;
      for (Node* c : x->kids) cnt(c);
    };
    cnt(md);
    if (stmt_count < opt.min_code_len || stmt_count > opt.max_code_len)
      continue;

    std::string line = label;
    bool any = false;
    for (size_t i = 0; i < leaves.size(); ++i) {
      for (size_t j = i + 1; j < leaves.size(); ++j) {
        std::string path = generate_path(leaves[i], leaves[j], opt);
        if (path.empty()) continue;
        const std::string hashed =
            opt.no_hash ? path : std::to_string(java_hash(path));
        line += " " + leaves[i]->propName + "," + hashed + "," +
                leaves[j]->propName;
        any = true;
      }
    }
    if (any) lines.push_back(line);
  }
  std::string out;
  for (size_t i = 0; i < lines.size(); ++i) {
    out += lines[i];
    if (i + 1 < lines.size()) out += "\n";
  }
  return out;
}

// ---------------------------------------------------------------------------
// CLI driver
// ---------------------------------------------------------------------------

static std::string read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  std::ostringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

static void walk_dir(const std::string& dir, std::vector<std::string>& out) {
  DIR* d = opendir(dir.c_str());
  if (!d) return;
  struct dirent* ent;
  while ((ent = readdir(d)) != nullptr) {
    std::string name = ent->d_name;
    if (name == "." || name == "..") continue;
    std::string full = dir + "/" + name;
    struct stat st;
    if (stat(full.c_str(), &st) != 0) continue;
    if (S_ISDIR(st.st_mode)) {
      walk_dir(full, out);
    } else if (S_ISREG(st.st_mode)) {
      std::string lower = name;
      std::transform(lower.begin(), lower.end(), lower.begin(), ::tolower);
      if (lower.size() > 5 && lower.substr(lower.size() - 5) == ".java")
        out.push_back(full);
    }
  }
  closedir(d);
}

int main(int argc, char** argv) {
  ExtractorOptions opt;
  std::string file, dir;
  for (int i = 1; i < argc; ++i) {
    std::string a = argv[i];
    auto next = [&]() -> std::string {
      return (i + 1 < argc) ? argv[++i] : "";
    };
    if (a == "--file") file = next();
    else if (a == "--dir") dir = next();
    else if (a == "--max_path_length") opt.max_path_length = atoi(next().c_str());
    else if (a == "--max_path_width") opt.max_path_width = atoi(next().c_str());
    else if (a == "--min_code_len") opt.min_code_len = atoi(next().c_str());
    else if (a == "--max_code_len") opt.max_code_len = atoi(next().c_str());
    else if (a == "--max_child_id") opt.max_child_id = atoi(next().c_str());
    else if (a == "--no_hash") opt.no_hash = true;
    else if (a == "--num_threads") opt.num_threads = atoi(next().c_str());
    else if (a == "--pretty_print") { /* accepted for compat */ }
    else {
      std::cerr << "unknown option: " << a << "\n";
      return 2;
    }
  }

  if (!file.empty()) {
    std::string out = extract_file_content(read_file(file), opt);
    if (!out.empty()) std::cout << out << "\n";
    return 0;
  }
  if (!dir.empty()) {
    std::vector<std::string> files;
    walk_dir(dir, files);
    std::sort(files.begin(), files.end());
    std::vector<std::string> results(files.size());
    std::atomic<size_t> next_idx(0);
    int nt = std::max(1, std::min<int>(opt.num_threads,
                                       (int)std::thread::hardware_concurrency()));
    auto work = [&]() {
      size_t i;
      while ((i = next_idx.fetch_add(1)) < files.size()) {
        try {
          results[i] = extract_file_content(read_file(files[i]), opt);
        } catch (...) {
          results[i].clear();
        }
      }
    };
    std::vector<std::thread> pool;
    for (int t = 0; t < nt; ++t) pool.emplace_back(work);
    for (auto& th : pool) th.join();
    for (const std::string& r : results)
      if (!r.empty()) std::cout << r << "\n";
    return 0;
  }
  std::cerr << "usage: c2v-extract --file F | --dir D --max_path_length N "
               "--max_path_width N [--no_hash] [--num_threads N]\n";
  return 2;
}
