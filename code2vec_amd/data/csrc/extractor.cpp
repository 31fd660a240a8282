// Offline Java -> AST path-context extractor (native C++).
//
// Re-implements the capability of the reference's Scala/JavaParser
// preprocessing notebook (create_path_contexts.ipynb cells 4-12; format
// spec docs/PREPROCESSING.md): per method, build a normalized AST,
// anonymize parameter/local names to @var_N and the method's own name to
// @method_0, replace literals with @*_literal tokens, collect terminal
// leaves, and emit every ordered leaf pair whose connecting AST path is
// short (<= max_length nodes) and narrow (sibling spread at the LCA
// <= max_width) as a (start-terminal, path, end-terminal) triple.
// Output files are byte-compatible with the reference corpus format
// (corpus.txt / terminal_idxs.txt / path_idxs.txt / actual_methods.txt /
// params.txt) and round-trip through data/reader.py.
//
// This is a clean-room extractor with its own pragmatic Java parser
// (lexer + recursive-descent statements + precedence-climbing
// expressions).  It handles the bulk of ordinary Java method bodies;
// methods it cannot parse are skipped and counted, mirroring the
// notebook's warn-and-continue behavior (cell 11).  AST node names are
// JavaParser-like but not guaranteed identical; they only feed the path
// vocabulary, which is corpus-defined.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <algorithm>
#include <cctype>
#include <cstdint>
#include <fstream>
#include <map>
#include <set>
#include <sstream>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

namespace {

// ---------------------------------------------------------------------------
// Lexer
enum class Tk { Ident, Kw, Int, Float, Str, Char, Op, Punct, End };

struct Token {
  Tk kind;
  std::string text;
};

const std::set<std::string> kKeywords = {
    "abstract", "assert", "boolean", "break", "byte", "case", "catch",
    "char", "class", "const", "continue", "default", "do", "double",
    "else", "enum", "extends", "final", "finally", "float", "for", "goto",
    "if", "implements", "import", "instanceof", "int", "interface", "long",
    "native", "new", "package", "private", "protected", "public", "return",
    "short", "static", "strictfp", "super", "switch", "synchronized",
    "this", "throw", "throws", "transient", "try", "void", "volatile",
    "while", "true", "false", "null", "var", "record", "yield"};

struct ParseError : std::runtime_error {
  explicit ParseError(const std::string& m) : std::runtime_error(m) {}
};

std::vector<Token> lex(const std::string& src) {
  std::vector<Token> out;
  size_t i = 0, n = src.size();
  while (i < n) {
    char c = src[i];
    if (isspace((unsigned char)c)) { ++i; continue; }
    if (c == '/' && i + 1 < n && src[i + 1] == '/') {
      while (i < n && src[i] != '\n') ++i;
      continue;
    }
    if (c == '/' && i + 1 < n && src[i + 1] == '*') {
      i += 2;
      while (i + 1 < n && !(src[i] == '*' && src[i + 1] == '/')) ++i;
      i = std::min(n, i + 2);
      continue;
    }
    if (c == '"') {
      // text blocks (""") collapse to one string literal
      if (i + 2 < n && src[i + 1] == '"' && src[i + 2] == '"') {
        i += 3;
        while (i + 2 < n && !(src[i] == '"' && src[i + 1] == '"' &&
                              src[i + 2] == '"'))
          ++i;
        i = std::min(n, i + 3);
        out.push_back({Tk::Str, "\"\""});
        continue;
      }
      ++i;
      while (i < n && src[i] != '"') {
        if (src[i] == '\\') ++i;
        ++i;
      }
      ++i;
      out.push_back({Tk::Str, "\"\""});
      continue;
    }
    if (c == '\'') {
      ++i;
      while (i < n && src[i] != '\'') {
        if (src[i] == '\\') ++i;
        ++i;
      }
      ++i;
      out.push_back({Tk::Char, "''"});
      continue;
    }
    if (isdigit((unsigned char)c) ||
        (c == '.' && i + 1 < n && isdigit((unsigned char)src[i + 1]))) {
      size_t j = i;
      bool flt = false;
      if (src[j] == '0' && j + 1 < n && (src[j + 1] == 'x' || src[j + 1] == 'X')) {
        j += 2;
        while (j < n && (isxdigit((unsigned char)src[j]) || src[j] == '_')) ++j;
      } else {
        while (j < n && (isdigit((unsigned char)src[j]) || src[j] == '_')) ++j;
        if (j < n && src[j] == '.') {
          flt = true;
          ++j;
          while (j < n && (isdigit((unsigned char)src[j]) || src[j] == '_')) ++j;
        }
        if (j < n && (src[j] == 'e' || src[j] == 'E')) {
          flt = true;
          ++j;
          if (j < n && (src[j] == '+' || src[j] == '-')) ++j;
          while (j < n && isdigit((unsigned char)src[j])) ++j;
        }
      }
      if (j < n && (src[j] == 'f' || src[j] == 'F' || src[j] == 'd' ||
                    src[j] == 'D'))
        flt = true, ++j;
      if (j < n && (src[j] == 'l' || src[j] == 'L')) ++j;
      out.push_back({flt ? Tk::Float : Tk::Int, src.substr(i, j - i)});
      i = j;
      continue;
    }
    if (isalpha((unsigned char)c) || c == '_' || c == '$') {
      size_t j = i;
      while (j < n && (isalnum((unsigned char)src[j]) || src[j] == '_' ||
                       src[j] == '$'))
        ++j;
      std::string t = src.substr(i, j - i);
      out.push_back({kKeywords.count(t) ? Tk::Kw : Tk::Ident, t});
      i = j;
      continue;
    }
    // operators (longest-match over the common multi-char set)
    static const char* ops3[] = {">>>=", nullptr};
    static const char* ops2[] = {"<<=", ">>=", ">>>", "==", "!=", "<=",
                                 ">=", "&&", "||", "++", "--", "+=", "-=",
                                 "*=", "/=", "%=", "&=", "|=", "^=", "<<",
                                 ">>", "->", "::", nullptr};
    bool matched = false;
    for (int k = 0; ops3[k]; ++k) {
      size_t len = strlen(ops3[k]);
      if (src.compare(i, len, ops3[k]) == 0) {
        out.push_back({Tk::Op, ops3[k]});
        i += len;
        matched = true;
        break;
      }
    }
    if (matched) continue;
    for (int k = 0; ops2[k]; ++k) {
      size_t len = strlen(ops2[k]);
      if (src.compare(i, len, ops2[k]) == 0) {
        out.push_back({Tk::Op, ops2[k]});
        i += len;
        matched = true;
        break;
      }
    }
    if (matched) continue;
    if (strchr("+-*/%=<>!&|^~?", c)) {
      out.push_back({Tk::Op, std::string(1, c)});
      ++i;
      continue;
    }
    if (strchr("(){}[];,.:@", c)) {
      out.push_back({Tk::Punct, std::string(1, c)});
      ++i;
      continue;
    }
    ++i;  // unknown byte: skip
  }
  out.push_back({Tk::End, ""});
  return out;
}

// ---------------------------------------------------------------------------
// AST arena
struct Node {
  std::string name;      // node type, e.g. "IfStmt", "BinaryExpr:plus"
  std::string term;      // terminal value ("" for internal nodes)
  int parent = -1;
  int child_idx = 0;
  std::vector<int> kids;
};

struct Ast {
  std::vector<Node> nodes;
  int add(const std::string& name, int parent, const std::string& term = "") {
    int id = (int)nodes.size();
    nodes.push_back({name, term, parent, 0, {}});
    if (parent >= 0) {
      nodes[parent].kids.push_back(id);
      nodes[id].child_idx = (int)nodes[parent].kids.size() - 1;
    }
    return id;
  }
};

const char* op_name(const std::string& op) {
  static const std::unordered_map<std::string, const char*> m = {
      {"+", "plus"}, {"-", "minus"}, {"*", "times"}, {"/", "divide"},
      {"%", "remainder"}, {"==", "equals"}, {"!=", "notEquals"},
      {"<", "less"}, {">", "greater"}, {"<=", "lessEquals"},
      {">=", "greaterEquals"}, {"&&", "and"}, {"||", "or"},
      {"&", "binAnd"}, {"|", "binOr"}, {"^", "xor"}, {"<<", "lShift"},
      {">>", "rSignedShift"}, {">>>", "rUnsignedShift"}, {"!", "not"},
      {"~", "complement"}, {"++", "increment"}, {"--", "decrement"},
      {"=", "assign"}, {"+=", "plus"}, {"-=", "minus"}, {"*=", "times"},
      {"/=", "divide"}, {"%=", "remainder"}, {"&=", "binAnd"},
      {"|=", "binOr"}, {"^=", "xor"}, {"<<=", "lShift"},
      {">>=", "rSignedShift"}, {">>>=", "rUnsignedShift"}};
  auto it = m.find(op);
  return it == m.end() ? "op" : it->second;
}

// ---------------------------------------------------------------------------
// Method-body parser: tokens -> normalized AST
struct MethodParser {
  const std::vector<Token>& t;
  size_t p = 0;
  Ast& ast;
  // anonymization state (reference notebook cell 5/6 VarEnv)
  std::unordered_map<std::string, std::string>& var_alias;  // orig -> @var_N
  const std::set<std::string>& class_methods;
  std::unordered_map<std::string, std::string>& method_alias;
  const std::string& own_name;
  int guard = 0;

  MethodParser(const std::vector<Token>& toks, Ast& a,
               std::unordered_map<std::string, std::string>& va,
               const std::set<std::string>& cm,
               std::unordered_map<std::string, std::string>& ma,
               const std::string& own)
      : t(toks), ast(a), var_alias(va), class_methods(cm), method_alias(ma),
        own_name(own) {}

  const Token& cur() const { return t[std::min(p, t.size() - 1)]; }
  const Token& peek(int k = 1) const {
    return t[std::min(p + k, t.size() - 1)];
  }
  bool is(Tk k, const char* s = nullptr) const {
    return cur().kind == k && (!s || cur().text == s);
  }
  void expect(Tk k, const char* s) {
    if (!is(k, s)) throw ParseError("expected " + std::string(s ? s : "?") +
                                    " got '" + cur().text + "'");
    ++p;
  }
  bool accept(Tk k, const char* s) {
    if (is(k, s)) { ++p; return true; }
    return false;
  }
  void bump_guard() {
    if (++guard > 2000000) throw ParseError("guard");
  }

  std::string var_of(const std::string& name) {
    auto it = var_alias.find(name);
    if (it != var_alias.end()) return it->second;
    return name;  // field / class / unknown identifier: keep
  }
  void declare_var(const std::string& name) {
    if (!var_alias.count(name))
      var_alias[name] = "@var_" + std::to_string(var_alias.size());
  }
  std::string method_of(const std::string& name) {
    if (name == own_name) return "@method_0";
    if (class_methods.count(name)) {
      auto it = method_alias.find(name);
      if (it != method_alias.end()) return it->second;
      std::string a = "@method_" + std::to_string(method_alias.size() + 1);
      method_alias[name] = a;
      return a;
    }
    return name;
  }

  // ---- types (consumed, not represented except as Type terminals)
  bool looks_like_type() const {
    if (cur().kind == Tk::Kw)
      return cur().text == "int" || cur().text == "long" ||
             cur().text == "short" || cur().text == "byte" ||
             cur().text == "char" || cur().text == "float" ||
             cur().text == "double" || cur().text == "boolean" ||
             cur().text == "void" || cur().text == "final" ||
             cur().text == "var";
    return cur().kind == Tk::Ident;
  }
  void skip_generics() {
    if (!is(Tk::Op, "<")) return;
    int depth = 0;
    size_t q = p;
    while (q < t.size()) {
      const auto& tk = t[q];
      if (tk.kind == Tk::Op && tk.text == "<") ++depth;
      else if (tk.kind == Tk::Op && (tk.text == ">" || tk.text == ">>" ||
                                     tk.text == ">>>")) {
        depth -= (int)tk.text.size();
        if (depth <= 0) { p = q + 1; return; }
      } else if (tk.kind == Tk::Punct &&
                 (tk.text == ";" || tk.text == "{" || tk.text == ")")) {
        return;  // not generics after all
      } else if (tk.kind == Tk::End) return;
      ++q;
    }
  }
  // consume a type reference; returns its printable name
  std::string parse_type() {
    std::string name = cur().text;
    if (!looks_like_type()) throw ParseError("type?");
    ++p;
    while (accept(Tk::Punct, ".")) {
      name = cur().text;
      ++p;
    }
    skip_generics();
    while (is(Tk::Punct, "[") && peek().kind == Tk::Punct &&
           peek().text == "]") {
      p += 2;
      name += "[]";
    }
    return name;
  }

  // ---- expressions (precedence climbing)
  int prec(const std::string& o) {
    if (o == "=" || o == "+=" || o == "-=" || o == "*=" || o == "/=" ||
        o == "%=" || o == "&=" || o == "|=" || o == "^=" || o == "<<=" ||
        o == ">>=" || o == ">>>=")
      return 1;
    if (o == "?") return 2;
    if (o == "||") return 3;
    if (o == "&&") return 4;
    if (o == "|") return 5;
    if (o == "^") return 6;
    if (o == "&") return 7;
    if (o == "==" || o == "!=") return 8;
    if (o == "<" || o == ">" || o == "<=" || o == ">=") return 9;
    if (o == "<<" || o == ">>" || o == ">>>") return 10;
    if (o == "+" || o == "-") return 11;
    if (o == "*" || o == "/" || o == "%") return 12;
    return -1;
  }

  int parse_expr(int parent, int min_prec = 1) {
    bump_guard();
    int lhs = parse_unary(parent);
    for (;;) {
      bump_guard();
      if (cur().kind == Tk::Kw && cur().text == "instanceof") {
        ++p;
        int node = reparent("InstanceOfExpr", parent, lhs);
        std::string ty = parse_type();
        if (cur().kind == Tk::Ident) {  // pattern variable
          declare_var(cur().text);
          ast.add("NameExpr", node, var_of(cur().text));
          ++p;
        } else {
          ast.add("ClassExpr", node, ty);
        }
        lhs = node;
        continue;
      }
      if (cur().kind != Tk::Op) break;
      std::string o = cur().text;
      int pr = prec(o);
      if (pr < min_prec) break;
      if (o == "?") {  // ternary
        ++p;
        int node = reparent("ConditionalExpr", parent, lhs);
        parse_expr(node, 1);
        expect(Tk::Punct, ":");
        parse_expr(node, 1);
        lhs = node;
        continue;
      }
      bool assign = pr == 1;
      ++p;
      std::string label = assign
          ? std::string("AssignExpr:") + op_name(o)
          : std::string("BinaryExpr:") + op_name(o);
      int node = reparent(label, parent, lhs);
      parse_expr(node, assign ? pr : pr + 1);  // assignment right-assoc
      lhs = node;
    }
    return lhs;
  }

  // wrap an already-built subtree under a fresh node (for infix forms)
  int reparent(const std::string& name, int parent, int child) {
    int node = ast.add(name, parent);
    // move child under node: child was last added under parent
    auto& pk = ast.nodes[parent].kids;
    auto it = std::find(pk.begin(), pk.end(), child);
    if (it != pk.end()) pk.erase(it);
    ast.nodes[child].parent = node;
    ast.nodes[node].kids.push_back(child);
    // fix child indexes
    for (size_t i = 0; i < pk.size(); ++i) ast.nodes[pk[i]].child_idx = (int)i;
    ast.nodes[child].child_idx = 0;
    return node;
  }

  int parse_unary(int parent) {
    bump_guard();
    if (cur().kind == Tk::Op &&
        (cur().text == "!" || cur().text == "-" || cur().text == "+" ||
         cur().text == "~" || cur().text == "++" || cur().text == "--")) {
      std::string o = cur().text;
      ++p;
      int node = ast.add(std::string("UnaryExpr:") + op_name(o), parent);
      parse_unary(node);
      return node;
    }
    // cast: '(' type ')' unary — only for primitive casts (unambiguous)
    if (is(Tk::Punct, "(") && peek().kind == Tk::Kw &&
        (peek().text == "int" || peek().text == "long" ||
         peek().text == "short" || peek().text == "byte" ||
         peek().text == "char" || peek().text == "float" ||
         peek().text == "double" || peek().text == "boolean")) {
      ++p;
      int node = ast.add("CastExpr", parent);
      ast.add("PrimitiveType", node, cur().text);
      ++p;
      expect(Tk::Punct, ")");
      parse_unary(node);
      return node;
    }
    return parse_postfix(parent);
  }

  int parse_postfix(int parent) {
    int e = parse_primary(parent);
    for (;;) {
      bump_guard();
      if (is(Tk::Punct, ".")) {
        // method call or field access
        if (peek().kind == Tk::Ident && peek(2).kind == Tk::Punct &&
            peek(2).text == "(") {
          ++p;  // '.'
          std::string m = cur().text;
          ++p;
          int node = reparent("MethodCallExpr", parent, e);
          ast.add("SimpleName", node, method_of(m));
          ++p;  // '('
          parse_args(node);
          e = node;
          continue;
        }
        if (peek().kind == Tk::Ident || (peek().kind == Tk::Kw &&
                                         (peek().text == "class" ||
                                          peek().text == "this"))) {
          ++p;
          int node = reparent("FieldAccessExpr", parent, e);
          ast.add("SimpleName", node, cur().text);
          ++p;
          e = node;
          continue;
        }
        if (peek().kind == Tk::Op && peek().text == "<") {
          ++p;  // explicit generic method call: consume and retry
          skip_generics();
          continue;
        }
        throw ParseError("postfix .");
      }
      if (is(Tk::Punct, "[")) {
        ++p;
        int node = reparent("ArrayAccessExpr", parent, e);
        parse_expr(node);
        expect(Tk::Punct, "]");
        e = node;
        continue;
      }
      if (cur().kind == Tk::Op &&
          (cur().text == "++" || cur().text == "--")) {
        int node = reparent(
            std::string("UnaryExpr:post") + op_name(cur().text), parent, e);
        ++p;
        e = node;
        continue;
      }
      if (cur().kind == Tk::Op && cur().text == "::") {
        ++p;
        int node = reparent("MethodReferenceExpr", parent, e);
        ast.add("SimpleName", node, cur().text);
        ++p;
        e = node;
        continue;
      }
      break;
    }
    return e;
  }

  void parse_args(int call_node) {
    if (accept(Tk::Punct, ")")) return;
    for (;;) {
      parse_expr(call_node);
      if (accept(Tk::Punct, ",")) continue;
      expect(Tk::Punct, ")");
      break;
    }
  }

  bool lambda_ahead() const {
    // '(' [idents] ')' '->'  or  ident '->'
    if (cur().kind == Tk::Ident && peek().kind == Tk::Op &&
        peek().text == "->")
      return true;
    if (!is(Tk::Punct, "(")) return false;
    size_t q = p + 1;
    int depth = 1;
    while (q < t.size() && depth > 0) {
      if (t[q].kind == Tk::Punct && t[q].text == "(") ++depth;
      else if (t[q].kind == Tk::Punct && t[q].text == ")") --depth;
      else if (t[q].kind == Tk::Punct && (t[q].text == ";" || t[q].text == "{"))
        return false;
      ++q;
    }
    return q < t.size() && t[q].kind == Tk::Op && t[q].text == "->";
  }

  int parse_lambda(int parent) {
    int node = ast.add("LambdaExpr", parent);
    if (cur().kind == Tk::Ident) {
      declare_var(cur().text);
      ast.add("Parameter", node, var_of(cur().text));
      ++p;
    } else {
      expect(Tk::Punct, "(");
      while (!accept(Tk::Punct, ")")) {
        bump_guard();
        if (cur().kind == Tk::Ident || cur().kind == Tk::Kw) {
          // possibly typed param: type name  |  name
          if (peek().kind == Tk::Ident) {
            parse_type();
          }
          if (cur().kind == Tk::Ident) {
            declare_var(cur().text);
            ast.add("Parameter", node, var_of(cur().text));
            ++p;
          }
        }
        accept(Tk::Punct, ",");
      }
    }
    expect(Tk::Op, "->");
    if (is(Tk::Punct, "{"))
      parse_block(node);
    else
      parse_expr(node);
    return node;
  }

  int parse_primary(int parent) {
    bump_guard();
    if (lambda_ahead()) return parse_lambda(parent);
    if (accept(Tk::Punct, "(")) {
      int node = ast.add("EnclosedExpr", parent);
      parse_expr(node);
      expect(Tk::Punct, ")");
      return node;
    }
    const Token& c = cur();
    switch (c.kind) {
      case Tk::Int: ++p; return ast.add("IntegerLiteralExpr", parent, "@int_literal");
      case Tk::Float: ++p; return ast.add("DoubleLiteralExpr", parent, "@double_literal");
      case Tk::Str: ++p; return ast.add("StringLiteralExpr", parent, "@string_literal");
      case Tk::Char: ++p; return ast.add("CharLiteralExpr", parent, "@char_literal");
      default: break;
    }
    if (c.kind == Tk::Kw) {
      if (c.text == "true" || c.text == "false") {
        ++p;
        return ast.add("BooleanLiteralExpr", parent, c.text);
      }
      if (c.text == "null") {
        ++p;
        return ast.add("NullLiteralExpr", parent, "null");
      }
      if (c.text == "this") {
        ++p;
        return ast.add("ThisExpr", parent, "this");
      }
      if (c.text == "super") {
        ++p;
        if (is(Tk::Punct, "(")) {  // super(...) call
          int node = ast.add("ExplicitConstructorInvocationStmt", parent);
          ++p;
          parse_args(node);
          return node;
        }
        return ast.add("SuperExpr", parent, "super");
      }
      if (c.text == "new") {
        ++p;
        std::string ty = parse_type();
        if (is(Tk::Punct, "[")) {  // array creation
          int node = ast.add("ArrayCreationExpr", parent);
          ast.add("ClassOrInterfaceType", node, ty);
          while (accept(Tk::Punct, "[")) {
            if (!is(Tk::Punct, "]")) parse_expr(node);
            expect(Tk::Punct, "]");
          }
          if (is(Tk::Punct, "{")) parse_array_init(node);
          return node;
        }
        int node = ast.add("ObjectCreationExpr", parent);
        ast.add("ClassOrInterfaceType", node, ty);
        if (accept(Tk::Punct, "(")) parse_args(node);
        if (is(Tk::Punct, "{")) {  // anonymous class body: skip tokens
          skip_braced();
        }
        return node;
      }
      if (c.text == "int" || c.text == "long" || c.text == "double" ||
          c.text == "float" || c.text == "boolean" || c.text == "char" ||
          c.text == "byte" || c.text == "short" || c.text == "void") {
        // e.g. int.class / int[].class
        std::string ty = parse_type();
        return ast.add("ClassExpr", parent, ty);
      }
      if (c.text == "switch") return parse_switch(parent, /*expr=*/true);
    }
    if (c.kind == Tk::Ident) {
      // method call on implicit this / plain name
      if (peek().kind == Tk::Punct && peek().text == "(") {
        std::string m = c.text;
        p += 2;
        int node = ast.add("MethodCallExpr", parent);
        ast.add("SimpleName", node, method_of(m));
        parse_args(node);
        return node;
      }
      ++p;
      return ast.add("NameExpr", parent, var_of(c.text));
    }
    throw ParseError("primary '" + c.text + "'");
  }

  void parse_array_init(int parent) {
    expect(Tk::Punct, "{");
    int node = ast.add("ArrayInitializerExpr", parent);
    if (accept(Tk::Punct, "}")) return;
    for (;;) {
      bump_guard();
      if (is(Tk::Punct, "{"))
        parse_array_init(node);
      else
        parse_expr(node);
      if (accept(Tk::Punct, ",")) {
        if (accept(Tk::Punct, "}")) return;
        continue;
      }
      expect(Tk::Punct, "}");
      return;
    }
  }

  void skip_braced() {
    expect(Tk::Punct, "{");
    int depth = 1;
    while (depth > 0 && cur().kind != Tk::End) {
      if (is(Tk::Punct, "{")) ++depth;
      if (is(Tk::Punct, "}")) --depth;
      ++p;
    }
  }

  // ---- statements
  int parse_block(int parent) {
    expect(Tk::Punct, "{");
    int node = ast.add("BlockStmt", parent);
    while (!accept(Tk::Punct, "}")) {
      bump_guard();
      if (cur().kind == Tk::End) throw ParseError("eof in block");
      parse_stmt(node);
    }
    return node;
  }

  bool var_decl_ahead() const {
    // [final] type name (= | , | ;) — heuristic lookahead
    size_t q = p;
    if (t[q].kind == Tk::Kw && t[q].text == "final") ++q;
    if (!(t[q].kind == Tk::Ident ||
          (t[q].kind == Tk::Kw &&
           (t[q].text == "int" || t[q].text == "long" || t[q].text == "short" ||
            t[q].text == "byte" || t[q].text == "char" || t[q].text == "float" ||
            t[q].text == "double" || t[q].text == "boolean" ||
            t[q].text == "var"))))
      return false;
    ++q;
    // qualified type / generics / arrays
    int angle = 0;
    while (q < t.size()) {
      const auto& tk = t[q];
      if (tk.kind == Tk::Punct && tk.text == "." && angle == 0) { q += 2; continue; }
      if (tk.kind == Tk::Op && tk.text == "<") { ++angle; ++q; continue; }
      if (angle > 0) {
        if (tk.kind == Tk::Op && (tk.text == ">" || tk.text == ">>" ||
                                  tk.text == ">>>")) {
          angle -= (int)tk.text.size();
          if (angle < 0) return false;
        } else if (tk.kind == Tk::Punct && (tk.text == ";" || tk.text == "{"))
          return false;
        ++q;
        continue;
      }
      if (tk.kind == Tk::Punct && tk.text == "[" && q + 1 < t.size() &&
          t[q + 1].text == "]") { q += 2; continue; }
      break;
    }
    if (q >= t.size() || t[q].kind != Tk::Ident) return false;
    ++q;
    while (q + 1 < t.size() && t[q].kind == Tk::Punct && t[q].text == "[" &&
           t[q + 1].text == "]")
      q += 2;
    if (q >= t.size()) return false;
    return (t[q].kind == Tk::Op && t[q].text == "=") ||
           (t[q].kind == Tk::Punct && (t[q].text == ";" || t[q].text == ","));
  }

  void parse_var_decl(int parent, bool stmt_semi) {
    int node = ast.add("VariableDeclarationExpr", parent);
    accept(Tk::Kw, "final");
    std::string ty = parse_type();
    for (;;) {
      bump_guard();
      int d = ast.add("VariableDeclarator", node);
      ast.add("ClassOrInterfaceType", d, ty);
      if (cur().kind != Tk::Ident) throw ParseError("var name");
      declare_var(cur().text);
      ast.add("SimpleName", d, var_of(cur().text));
      ++p;
      while (is(Tk::Punct, "[") && peek().text == "]") p += 2;
      if (accept(Tk::Op, "=")) {
        if (is(Tk::Punct, "{"))
          parse_array_init(d);
        else
          parse_expr(d);
      }
      if (accept(Tk::Punct, ",")) continue;
      break;
    }
    if (stmt_semi) expect(Tk::Punct, ";");
  }

  int parse_switch(int parent, bool expr) {
    expect(Tk::Kw, "switch");
    int node = ast.add(expr ? "SwitchExpr" : "SwitchStmt", parent);
    expect(Tk::Punct, "(");
    parse_expr(node);
    expect(Tk::Punct, ")");
    expect(Tk::Punct, "{");
    while (!accept(Tk::Punct, "}")) {
      bump_guard();
      if (cur().kind == Tk::End) throw ParseError("eof in switch");
      int entry = ast.add("SwitchEntry", node);
      if (accept(Tk::Kw, "default")) {
      } else {
        expect(Tk::Kw, "case");
        parse_expr(entry);
        while (accept(Tk::Punct, ",")) parse_expr(entry);
      }
      if (accept(Tk::Op, "->")) {
        if (is(Tk::Punct, "{"))
          parse_block(entry);
        else {
          parse_stmt(entry);
          continue;
        }
      } else {
        expect(Tk::Punct, ":");
        while (!is(Tk::Kw, "case") && !is(Tk::Kw, "default") &&
               !is(Tk::Punct, "}")) {
          bump_guard();
          parse_stmt(entry);
        }
      }
    }
    return node;
  }

  void parse_stmt(int parent) {
    bump_guard();
    if (is(Tk::Punct, "{")) { parse_block(parent); return; }
    if (accept(Tk::Punct, ";")) { ast.add("EmptyStmt", parent); return; }
    if (is(Tk::Kw, "if")) {
      ++p;
      int node = ast.add("IfStmt", parent);
      expect(Tk::Punct, "(");
      parse_expr(node);
      expect(Tk::Punct, ")");
      parse_stmt(node);
      if (accept(Tk::Kw, "else")) parse_stmt(node);
      return;
    }
    if (is(Tk::Kw, "while")) {
      ++p;
      int node = ast.add("WhileStmt", parent);
      expect(Tk::Punct, "(");
      parse_expr(node);
      expect(Tk::Punct, ")");
      parse_stmt(node);
      return;
    }
    if (is(Tk::Kw, "do")) {
      ++p;
      int node = ast.add("DoStmt", parent);
      parse_stmt(node);
      expect(Tk::Kw, "while");
      expect(Tk::Punct, "(");
      parse_expr(node);
      expect(Tk::Punct, ")");
      expect(Tk::Punct, ";");
      return;
    }
    if (is(Tk::Kw, "for")) {
      ++p;
      expect(Tk::Punct, "(");
      // foreach?  [final] type name ':' expr
      size_t save = p;
      bool foreach = false;
      {
        size_t q = p;
        int depth = 0;
        while (q < t.size() && !(depth == 0 && t[q].kind == Tk::Punct &&
                                 (t[q].text == ";" || t[q].text == ")"))) {
          if (t[q].kind == Tk::Punct && t[q].text == "(") ++depth;
          if (t[q].kind == Tk::Punct && t[q].text == ")") --depth;
          if (depth == 0 && t[q].kind == Tk::Punct && t[q].text == ":") {
            foreach = true;
            break;
          }
          ++q;
        }
      }
      if (foreach) {
        int node = ast.add("ForEachStmt", parent);
        accept(Tk::Kw, "final");
        std::string ty = parse_type();
        int d = ast.add("VariableDeclarator", node);
        ast.add("ClassOrInterfaceType", d, ty);
        declare_var(cur().text);
        ast.add("SimpleName", d, var_of(cur().text));
        ++p;
        expect(Tk::Punct, ":");
        parse_expr(node);
        expect(Tk::Punct, ")");
        parse_stmt(node);
        return;
      }
      p = save;
      int node = ast.add("ForStmt", parent);
      if (!accept(Tk::Punct, ";")) {
        if (var_decl_ahead())
          parse_var_decl(node, false);
        else {
          parse_expr(node);
          while (accept(Tk::Punct, ",")) parse_expr(node);
        }
        expect(Tk::Punct, ";");
      }
      if (!accept(Tk::Punct, ";")) {
        parse_expr(node);
        expect(Tk::Punct, ";");
      }
      if (!is(Tk::Punct, ")")) {
        parse_expr(node);
        while (accept(Tk::Punct, ",")) parse_expr(node);
      }
      expect(Tk::Punct, ")");
      parse_stmt(node);
      return;
    }
    if (is(Tk::Kw, "return")) {
      ++p;
      int node = ast.add("ReturnStmt", parent);
      if (!is(Tk::Punct, ";")) parse_expr(node);
      expect(Tk::Punct, ";");
      return;
    }
    if (is(Tk::Kw, "throw")) {
      ++p;
      int node = ast.add("ThrowStmt", parent);
      parse_expr(node);
      expect(Tk::Punct, ";");
      return;
    }
    if (is(Tk::Kw, "break") || is(Tk::Kw, "continue")) {
      std::string kind = cur().text == "break" ? "BreakStmt" : "ContinueStmt";
      ++p;
      int node = ast.add(kind, parent);
      if (cur().kind == Tk::Ident) {
        ast.add("SimpleName", node, cur().text);
        ++p;
      }
      expect(Tk::Punct, ";");
      return;
    }
    if (is(Tk::Kw, "try")) {
      ++p;
      int node = ast.add("TryStmt", parent);
      if (accept(Tk::Punct, "(")) {  // try-with-resources
        for (;;) {
          bump_guard();
          if (var_decl_ahead())
            parse_var_decl(node, false);
          else
            parse_expr(node);
          if (accept(Tk::Punct, ";")) {
            if (is(Tk::Punct, ")")) { ++p; break; }
            continue;
          }
          expect(Tk::Punct, ")");
          break;
        }
      }
      parse_block(node);
      while (is(Tk::Kw, "catch")) {
        ++p;
        int cl = ast.add("CatchClause", node);
        expect(Tk::Punct, "(");
        accept(Tk::Kw, "final");
        parse_type();
        while (accept(Tk::Op, "|")) parse_type();
        declare_var(cur().text);
        ast.add("Parameter", cl, var_of(cur().text));
        ++p;
        expect(Tk::Punct, ")");
        parse_block(cl);
      }
      if (accept(Tk::Kw, "finally")) parse_block(node);
      return;
    }
    if (is(Tk::Kw, "synchronized")) {
      ++p;
      int node = ast.add("SynchronizedStmt", parent);
      expect(Tk::Punct, "(");
      parse_expr(node);
      expect(Tk::Punct, ")");
      parse_block(node);
      return;
    }
    if (is(Tk::Kw, "assert")) {
      ++p;
      int node = ast.add("AssertStmt", parent);
      parse_expr(node);
      if (accept(Tk::Punct, ":")) parse_expr(node);
      expect(Tk::Punct, ";");
      return;
    }
    if (is(Tk::Kw, "switch")) {
      parse_switch(parent, /*expr=*/false);
      return;
    }
    if (is(Tk::Kw, "yield")) {
      ++p;
      int node = ast.add("YieldStmt", parent);
      parse_expr(node);
      expect(Tk::Punct, ";");
      return;
    }
    // labeled statement: ident ':'
    if (cur().kind == Tk::Ident && peek().kind == Tk::Punct &&
        peek().text == ":") {
      int node = ast.add("LabeledStmt", parent);
      ast.add("SimpleName", node, "@label_0");
      p += 2;
      parse_stmt(node);
      return;
    }
    if (var_decl_ahead()) {
      int node = ast.add("ExpressionStmt", parent);
      parse_var_decl(node, true);
      return;
    }
    int node = ast.add("ExpressionStmt", parent);
    parse_expr(node);
    expect(Tk::Punct, ";");
  }
};

// ---------------------------------------------------------------------------
// Method discovery in a compilation unit
struct MethodDecl {
  std::string name;
  std::vector<std::string> params;
  size_t body_lo = 0, body_hi = 0;  // token range incl. braces (0,0 = none)
  bool is_abstract = false;
};

std::vector<MethodDecl> find_methods(const std::vector<Token>& toks) {
  std::vector<MethodDecl> out;
  const size_t n = toks.size();
  for (size_t i = 0; i + 1 < n; ++i) {
    if (toks[i].kind != Tk::Ident) continue;
    if (toks[i + 1].kind != Tk::Punct || toks[i + 1].text != "(") continue;
    // must be preceded by a plausible type or modifier (not '.', 'new', etc.)
    if (i == 0) continue;
    const Token& prev = toks[i - 1];
    bool prev_ok =
        (prev.kind == Tk::Ident) ||
        (prev.kind == Tk::Punct && prev.text == "]") ||
        (prev.kind == Tk::Op && (prev.text == ">" || prev.text == ">>")) ||
        (prev.kind == Tk::Kw &&
         (prev.text == "void" || prev.text == "int" || prev.text == "long" ||
          prev.text == "short" || prev.text == "byte" || prev.text == "char" ||
          prev.text == "float" || prev.text == "double" ||
          prev.text == "boolean" || prev.text == "public" ||
          prev.text == "private" || prev.text == "protected" ||
          prev.text == "static" || prev.text == "final" ||
          prev.text == "synchronized" || prev.text == "abstract" ||
          prev.text == "native" || prev.text == "default"));
    if (!prev_ok) continue;
    // find matching ')'
    size_t q = i + 2;
    int depth = 1;
    while (q < n && depth > 0) {
      if (toks[q].kind == Tk::Punct && toks[q].text == "(") ++depth;
      else if (toks[q].kind == Tk::Punct && toks[q].text == ")") --depth;
      ++q;
    }
    if (depth != 0) continue;
    // after params: optional "throws A, B", then '{' (decl) or ';' (abstract)
    size_t r = q;
    if (r < n && toks[r].kind == Tk::Kw && toks[r].text == "throws") {
      ++r;
      while (r < n && !(toks[r].kind == Tk::Punct &&
                        (toks[r].text == "{" || toks[r].text == ";")))
        ++r;
    }
    if (r >= n || toks[r].kind != Tk::Punct) continue;
    bool has_body = toks[r].text == "{";
    if (!has_body && toks[r].text != ";") continue;
    // modifier scan backwards for 'abstract'
    bool is_abs = !has_body;
    MethodDecl md;
    md.name = toks[i].text;
    md.is_abstract = is_abs;
    // params: idents preceding ',' or ')' at depth 1 (skip defaults etc.)
    {
      size_t a = i + 2;
      int d = 1;
      int angle = 0;
      std::string last_ident;
      while (a < q) {
        const Token& tk = toks[a];
        if (tk.kind == Tk::Punct && tk.text == "(") ++d;
        else if (tk.kind == Tk::Punct && tk.text == ")") --d;
        else if (tk.kind == Tk::Op && tk.text == "<") ++angle;
        else if (tk.kind == Tk::Op && (tk.text == ">" || tk.text == ">>" ||
                                       tk.text == ">>>"))
          angle = std::max(0, angle - (int)tk.text.size());
        else if (d == 1 && angle == 0) {
          if (tk.kind == Tk::Ident) last_ident = tk.text;
          if (tk.kind == Tk::Punct && tk.text == "," && !last_ident.empty()) {
            md.params.push_back(last_ident);
            last_ident.clear();
          }
        }
        ++a;
      }
      if (!last_ident.empty()) md.params.push_back(last_ident);
    }
    if (has_body) {
      size_t b = r;
      int bd = 0;
      size_t e = b;
      while (e < n) {
        if (toks[e].kind == Tk::Punct && toks[e].text == "{") ++bd;
        else if (toks[e].kind == Tk::Punct && toks[e].text == "}") {
          --bd;
          if (bd == 0) { ++e; break; }
        }
        ++e;
      }
      md.body_lo = b;
      md.body_hi = e;
    }
    out.push_back(std::move(md));
    i = q - 1;
  }
  return out;
}

// reference cell 4: trivial getter/setter + Object methods + abstract
bool is_ignorable(const MethodDecl& md, const std::vector<Token>& toks) {
  static const std::set<std::string> objm = {"clone", "equals", "finalize",
                                             "hashCode", "toString"};
  if (md.is_abstract || md.body_lo == md.body_hi) return true;
  if (objm.count(md.name)) return true;
  auto body_stmts = [&]() {  // tokens strictly inside the outer braces
    return std::make_pair(md.body_lo + 1, md.body_hi - 1);
  };
  auto count_semis = [&]() {
    int s = 0;
    for (size_t q = md.body_lo + 1; q + 1 < md.body_hi; ++q)
      if (toks[q].kind == Tk::Punct && toks[q].text == ";") ++s;
    return s;
  };
  if (md.name.rfind("set", 0) == 0 && md.params.size() == 1) {
    // single statement that is an assignment
    auto [lo, hi] = body_stmts();
    bool has_assign = false;
    for (size_t q = lo; q < hi; ++q)
      if (toks[q].kind == Tk::Op && toks[q].text == "=") has_assign = true;
    if (count_semis() == 1 && has_assign) return true;
  }
  if ((md.name.rfind("get", 0) == 0 || md.name.rfind("is", 0) == 0) &&
      md.params.empty()) {
    auto [lo, hi] = body_stmts();
    if (count_semis() == 1 && lo < hi && toks[lo].kind == Tk::Kw &&
        toks[lo].text == "return")
      return true;
  }
  return false;
}

// ---------------------------------------------------------------------------
// Path-context extraction from a built AST
struct Extraction {
  std::vector<std::pair<std::string, std::string>> aliases;  // orig, @var_N
  std::vector<std::tuple<std::string, std::string, std::string>> contexts;
  int n_terminals = 0;
};

void collect_terminals(const Ast& ast, int node, std::vector<int>& out) {
  const Node& nd = ast.nodes[node];
  if (!nd.term.empty()) out.push_back(node);
  for (int k : nd.kids) collect_terminals(ast, k, out);
}

std::string lower(std::string s) {
  for (auto& c : s) c = (char)tolower((unsigned char)c);
  return s;
}

Extraction extract_paths(const Ast& ast, int root, int max_length,
                         int max_width) {
  Extraction ex;
  std::vector<int> terms;
  collect_terminals(ast, root, terms);
  ex.n_terminals = (int)terms.size();
  // per-terminal root chains
  std::vector<std::vector<int>> chain(terms.size());
  for (size_t i = 0; i < terms.size(); ++i) {
    int x = terms[i];
    while (x != -1) {
      chain[i].push_back(x);
      x = ast.nodes[x].parent;
    }
  }
  for (size_t i = 0; i < terms.size(); ++i) {
    for (size_t j = i + 1; j < terms.size(); ++j) {
      // LCA via suffix match of root chains
      const auto& a = chain[i];
      const auto& b = chain[j];
      int ai = (int)a.size() - 1, bi = (int)b.size() - 1;
      while (ai > 0 && bi > 0 && a[ai - 1] == b[bi - 1]) { --ai; --bi; }
      // up-path a[0..ai], LCA = a[ai], down-path b[bi-1..0]
      const int up_len = ai;        // nodes strictly below LCA on start side
      const int down_len = bi;
      if (up_len + down_len + 1 > max_length) continue;
      if (up_len > 0 && down_len > 0) {
        const int ca = ast.nodes[a[ai - 1]].child_idx;
        const int cb = ast.nodes[b[bi - 1]].child_idx;
        if (std::abs(ca - cb) > max_width) continue;
      }
      std::string path;
      for (int k = 0; k < up_len; ++k) {
        path += ast.nodes[a[k]].name;
        path += "\xE2\x86\x91";  // ↑
      }
      path += ast.nodes[a[ai]].name;
      for (int k = down_len - 1; k >= 0; --k) {
        path += "\xE2\x86\x93";  // ↓
        path += ast.nodes[b[k]].name;
      }
      ex.contexts.emplace_back(lower(ast.nodes[terms[i]].term), path,
                               lower(ast.nodes[terms[j]].term));
    }
  }
  return ex;
}

// Parse one method into contexts.  Throws ParseError on failure.
Extraction extract_method(const std::vector<Token>& toks,
                          const MethodDecl& md,
                          const std::set<std::string>& class_methods,
                          int max_length, int max_width) {
  Ast ast;
  int root = ast.add("MethodDeclaration", -1);
  ast.add("SimpleName", root, "@method_0");
  std::unordered_map<std::string, std::string> var_alias;
  std::unordered_map<std::string, std::string> method_alias;
  for (const auto& pn : md.params) {
    if (!var_alias.count(pn))
      var_alias[pn] = "@var_" + std::to_string(var_alias.size());
    int pnode = ast.add("Parameter", root);
    ast.add("SimpleName", pnode, var_alias[pn]);
  }
  std::vector<Token> body(toks.begin() + md.body_lo,
                          toks.begin() + md.body_hi);
  body.push_back({Tk::End, ""});
  MethodParser mp(body, ast, var_alias, class_methods, method_alias, md.name);
  mp.parse_block(root);
  if (mp.cur().kind != Tk::End) throw ParseError("trailing tokens");
  Extraction ex = extract_paths(ast, root, max_length, max_width);
  std::vector<std::pair<std::string, std::string>> al(var_alias.begin(),
                                                      var_alias.end());
  std::sort(al.begin(), al.end(), [](const auto& x, const auto& y) {
    return x.second < y.second;
  });
  ex.aliases = std::move(al);
  return ex;
}

// ---------------------------------------------------------------------------
// createDataset equivalent (notebook cells 11-12)
struct Vocab {
  std::unordered_map<std::string, int> stoi;
  std::vector<std::string> itos{"<PAD/>"};
  int get(const std::string& s) {
    auto it = stoi.find(s);
    if (it != stoi.end()) return it->second;
    int id = (int)itos.size();
    stoi.emplace(s, id);
    itos.push_back(s);
    return id;
  }
};

std::string read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  if (!f) throw std::runtime_error("cannot open " + path);
  std::stringstream ss;
  ss << f.rdbuf();
  return ss.str();
}

py::dict extract_to_dataset(const std::string& methods_file,
                            const std::string& src_root,
                            const std::string& out_dir, int max_length,
                            int max_width) {
  std::ifstream mf(methods_file);
  if (!mf) throw std::runtime_error("cannot open " + methods_file);
  std::ofstream corpus(out_dir + "/corpus.txt");
  std::ofstream actual(out_dir + "/actual_methods.txt");
  Vocab terms, paths;
  std::string line;
  int id = 0, written = 0, skipped_parse = 0, skipped_ignorable = 0,
      missing = 0;
  std::string cached_file;
  std::vector<Token> toks;
  std::vector<MethodDecl> decls;
  std::set<std::string> class_methods;
  while (std::getline(mf, line)) {
    if (line.empty()) continue;
    auto tab = line.find('\t');
    if (tab == std::string::npos) continue;
    std::string file = line.substr(0, tab);
    std::string mname = line.substr(tab + 1);
    std::string full = src_root.empty() ? file : src_root + "/" + file;
    if (file != cached_file) {
      cached_file = file;
      decls.clear();
      class_methods.clear();
      try {
        toks = lex(read_file(full));
        decls = find_methods(toks);
        for (const auto& d : decls) class_methods.insert(d.name);
      } catch (const std::exception&) {
        toks.clear();
      }
    }
    const MethodDecl* found = nullptr;
    for (const auto& d : decls)
      if (d.name == mname) { found = &d; break; }
    ++id;
    if (!found) { ++missing; continue; }
    if (is_ignorable(*found, toks)) { ++skipped_ignorable; continue; }
    Extraction ex;
    try {
      ex = extract_method(toks, *found, class_methods, max_length,
                          max_width);
    } catch (const std::exception&) {
      ++skipped_parse;
      continue;
    }
    if (ex.contexts.empty()) continue;
    corpus << "#" << id << "\n";
    corpus << "label:" << mname << "\n";
    corpus << "class:" << file << "\n";
    corpus << "paths:\n";
    for (const auto& [s, pth, e] : ex.contexts)
      corpus << terms.get(s) << "\t" << paths.get(pth) << "\t"
             << terms.get(e) << "\n";
    if (!ex.aliases.empty()) {
      corpus << "vars:\n";
      for (const auto& [orig, alias] : ex.aliases)
        corpus << orig << "\t" << alias << "\n";
    }
    corpus << "\n";
    actual << file << "\t" << mname << "\n";
    ++written;
  }
  {
    std::ofstream tf(out_dir + "/terminal_idxs.txt");
    for (size_t i = 0; i < terms.itos.size(); ++i)
      tf << i << "\t" << terms.itos[i] << "\n";
    std::ofstream pf(out_dir + "/path_idxs.txt");
    for (size_t i = 0; i < paths.itos.size(); ++i)
      pf << i << "\t" << paths.itos[i] << "\n";
    std::ofstream sf(out_dir + "/params.txt");
    sf << "MAX_PATH_LENGTH: " << max_length << "\n";
    sf << "MAX_PATH_WIDTH: " << max_width << "\n";
    sf << "EXTRACTOR: code2vec_amd native C++\n";
    sf << "terminal vocab size: " << terms.itos.size() - 1 << "\n";
    sf << "path vocab size: " << paths.itos.size() - 1 << "\n";
    sf << "method count: " << written << "\n";
  }
  py::dict r;
  r["methods_written"] = written;
  r["skipped_unparseable"] = skipped_parse;
  r["skipped_ignorable"] = skipped_ignorable;
  r["missing"] = missing;
  r["terminal_vocab"] = (int)terms.itos.size() - 1;
  r["path_vocab"] = (int)paths.itos.size() - 1;
  return r;
}

// Single-source helper for tests: returns per-method extraction results.
py::list extract_source(const std::string& source, int max_length,
                        int max_width) {
  auto toks = lex(source);
  auto decls = find_methods(toks);
  std::set<std::string> class_methods;
  for (const auto& d : decls) class_methods.insert(d.name);
  py::list out;
  for (const auto& d : decls) {
    py::dict rec;
    rec["name"] = d.name;
    rec["ignorable"] = is_ignorable(d, toks);
    if (is_ignorable(d, toks)) {
      out.append(rec);
      continue;
    }
    try {
      Extraction ex =
          extract_method(toks, d, class_methods, max_length, max_width);
      py::list ctx;
      for (const auto& [s, p2, e] : ex.contexts)
        ctx.append(py::make_tuple(s, p2, e));
      rec["contexts"] = ctx;
      py::dict al;
      for (const auto& [orig, alias] : ex.aliases) al[py::str(orig)] = alias;
      rec["aliases"] = al;
      rec["n_terminals"] = ex.n_terminals;
    } catch (const std::exception& e) {
      rec["error"] = std::string(e.what());
    }
    out.append(rec);
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.doc() = "Native Java -> AST path-context extractor";
  m.def("extract_to_dataset", &extract_to_dataset, py::arg("methods_file"),
        py::arg("src_root"), py::arg("out_dir"), py::arg("max_length") = 8,
        py::arg("max_width") = 3);
  m.def("extract_source", &extract_source, py::arg("source"),
        py::arg("max_length") = 8, py::arg("max_width") = 3);
}
