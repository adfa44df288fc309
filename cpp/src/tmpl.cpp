#include "cpilot/tmpl.hpp"

#include <cctype>
#include <cstdio>
#include <cstring>
#include <map>
#include <memory>
#include <regex>
#include <stdexcept>
#include <vector>

extern char** environ;

namespace cpilot {
namespace {

// ---------- values ----------

struct Value;
using ValueList = std::vector<Value>;

struct Value {
  enum class Kind { Nil, Str, Int, Float, Bool, List };
  Kind kind = Kind::Nil;
  std::string s;
  int64_t i = 0;
  double f = 0;
  bool b = false;
  std::shared_ptr<ValueList> list;

  static Value nil() { return Value{}; }
  static Value str(std::string v) {
    Value r;
    r.kind = Kind::Str;
    r.s = std::move(v);
    return r;
  }
  static Value integer(int64_t v) {
    Value r;
    r.kind = Kind::Int;
    r.i = v;
    return r;
  }
  static Value number(double v) {
    Value r;
    r.kind = Kind::Float;
    r.f = v;
    return r;
  }
  static Value boolean(bool v) {
    Value r;
    r.kind = Kind::Bool;
    r.b = v;
    return r;
  }
  static Value mklist(ValueList v) {
    Value r;
    r.kind = Kind::List;
    r.list = std::make_shared<ValueList>(std::move(v));
    return r;
  }

  bool truthy() const {
    switch (kind) {
      case Kind::Nil: return false;
      case Kind::Str: return !s.empty();
      case Kind::Int: return i != 0;
      case Kind::Float: return f != 0;
      case Kind::Bool: return b;
      case Kind::List: return list && !list->empty();
    }
    return false;
  }

  std::string print() const {
    switch (kind) {
      case Kind::Nil: return "";
      case Kind::Str: return s;
      case Kind::Int: return std::to_string(i);
      case Kind::Float: {
        char buf[40];
        snprintf(buf, sizeof(buf), "%g", f);
        return buf;
      }
      case Kind::Bool: return b ? "true" : "false";
      case Kind::List: {
        std::string out = "[";
        bool first = true;
        for (auto& e : *list) {
          if (!first) out += " ";
          first = false;
          out += e.print();
        }
        return out + "]";
      }
    }
    return "";
  }

  int64_t toInt() const {
    switch (kind) {
      case Kind::Int: return i;
      case Kind::Float: return (int64_t)f;
      case Kind::Str: {
        char* end = nullptr;
        long v = strtol(s.c_str(), &end, 10);
        if (!end || *end != '\0')
          throw std::runtime_error("cannot convert '" + s + "' to integer");
        return v;
      }
      default:
        throw std::runtime_error("cannot convert value to integer");
    }
  }
};

// ---------- expression AST ----------

struct Expr;
using ExprPtr = std::shared_ptr<Expr>;

struct Expr {
  enum class Kind { Field, Var, StrLit, NumLit, Call, Pipeline, Dot } kind;
  std::vector<std::string> path;  // Field: .A.B  / Var: name
  std::string strval;
  Value numval;
  std::string fname;            // Call
  std::vector<ExprPtr> args;    // Call
  std::vector<ExprPtr> stages;  // Pipeline
};

// ---------- template AST ----------

struct Node;
using NodePtr = std::shared_ptr<Node>;

struct Node {
  enum class Kind { Text, Action, If, Range, Assign, With } kind;
  std::string text;               // Text
  ExprPtr expr;                   // Action/If/Range/With/Assign rhs
  std::string rangeVar;           // Range: $i (index var) / Assign: name
  std::string rangeVar2;          // Range: $v (value var, two-var form)
  std::vector<NodePtr> body;      // If/Range/With
  std::vector<NodePtr> elseBody;  // If/With
};

// ---------- lexer for actions ----------

struct ActionTok {
  enum class Kind { Ident, Field, Var, Str, Num, LParen, RParen, Pipe, Assign,
                    Comma, End } kind;
  std::string text;
  Value num;
};

class ActionLexer {
 public:
  explicit ActionLexer(std::string s) : s_(std::move(s)) {}

  ActionTok next() {
    skipWs();
    if (pos_ >= s_.size()) return {ActionTok::Kind::End, "", {}};
    char c = s_[pos_];
    if (c == '(') { pos_++; return {ActionTok::Kind::LParen, "(", {}}; }
    if (c == ')') { pos_++; return {ActionTok::Kind::RParen, ")", {}}; }
    if (c == '|') { pos_++; return {ActionTok::Kind::Pipe, "|", {}}; }
    if (c == ',') { pos_++; return {ActionTok::Kind::Comma, ",", {}}; }
    if (c == ':' && pos_ + 1 < s_.size() && s_[pos_ + 1] == '=') {
      pos_ += 2;
      return {ActionTok::Kind::Assign, ":=", {}};
    }
    if (c == '=') {
      pos_++;
      return {ActionTok::Kind::Assign, "=", {}};
    }
    if (c == '"' || c == '\'' || c == '`') return lexString(c);
    if (c == '.') {
      pos_++;
      std::string path;
      while (pos_ < s_.size() &&
             (isalnum((unsigned char)s_[pos_]) || s_[pos_] == '_' ||
              s_[pos_] == '.')) {
        path += s_[pos_++];
      }
      return {ActionTok::Kind::Field, path, {}};
    }
    if (c == '$') {
      pos_++;
      std::string name;
      while (pos_ < s_.size() &&
             (isalnum((unsigned char)s_[pos_]) || s_[pos_] == '_'))
        name += s_[pos_++];
      return {ActionTok::Kind::Var, name, {}};
    }
    if (isdigit((unsigned char)c) || c == '-' || c == '+') {
      size_t start = pos_;
      pos_++;
      bool isFloat = false;
      while (pos_ < s_.size() &&
             (isdigit((unsigned char)s_[pos_]) || s_[pos_] == '.' ||
              s_[pos_] == 'e' || s_[pos_] == 'E')) {
        if (s_[pos_] == '.' || s_[pos_] == 'e' || s_[pos_] == 'E')
          isFloat = true;
        pos_++;
      }
      std::string num = s_.substr(start, pos_ - start);
      ActionTok t{ActionTok::Kind::Num, num, {}};
      t.num = isFloat ? Value::number(strtod(num.c_str(), nullptr))
                      : Value::integer(strtoll(num.c_str(), nullptr, 10));
      return t;
    }
    if (isalpha((unsigned char)c) || c == '_') {
      std::string name;
      while (pos_ < s_.size() &&
             (isalnum((unsigned char)s_[pos_]) || s_[pos_] == '_'))
        name += s_[pos_++];
      return {ActionTok::Kind::Ident, name, {}};
    }
    throw std::runtime_error(std::string("template: unexpected character '") +
                             c + "' in action");
  }

 private:
  std::string s_;
  size_t pos_ = 0;

  void skipWs() {
    while (pos_ < s_.size() && isspace((unsigned char)s_[pos_])) pos_++;
  }

  ActionTok lexString(char quote) {
    pos_++;
    std::string out;
    while (pos_ < s_.size() && s_[pos_] != quote) {
      char c = s_[pos_++];
      if (c == '\\' && quote != '`' && pos_ < s_.size()) {
        char e = s_[pos_++];
        switch (e) {
          case 'n': out += '\n'; break;
          case 't': out += '\t'; break;
          case 'r': out += '\r'; break;
          default: out += e;
        }
      } else {
        out += c;
      }
    }
    if (pos_ >= s_.size()) throw std::runtime_error("template: unterminated string");
    pos_++;
    return {ActionTok::Kind::Str, out, {}};
  }
};

// ---------- action parser ----------

std::vector<std::string> splitPath(const std::string& p) {
  std::vector<std::string> out;
  std::string cur;
  for (char c : p) {
    if (c == '.') {
      if (!cur.empty()) out.push_back(cur);
      cur.clear();
    } else {
      cur += c;
    }
  }
  if (!cur.empty()) out.push_back(cur);
  return out;
}

class ActionParser {
 public:
  explicit ActionParser(std::string s) : lex_(std::move(s)) { advance(); }

  // parse full pipeline (for action body)
  ExprPtr parsePipeline() {
    auto first = parseCommand();
    if (tok_.kind != ActionTok::Kind::Pipe) return first;
    auto pipe = std::make_shared<Expr>();
    pipe->kind = Expr::Kind::Pipeline;
    pipe->stages.push_back(first);
    while (tok_.kind == ActionTok::Kind::Pipe) {
      advance();
      pipe->stages.push_back(parseCommand());
    }
    return pipe;
  }

  bool atEnd() const { return tok_.kind == ActionTok::Kind::End; }

  // parse "range $i := ..." / "range $i, $v := ..." header; returns
  // (var1, var2) — empty strings when absent
  std::pair<std::string, std::string> parseRangeVars() {
    std::pair<std::string, std::string> out;
    if (tok_.kind != ActionTok::Kind::Var) return out;
    out.first = tok_.text;
    advance();
    if (tok_.kind == ActionTok::Kind::Comma) {
      advance();
      if (tok_.kind != ActionTok::Kind::Var)
        throw std::runtime_error("template: expected variable after ','");
      out.second = tok_.text;
      advance();
    }
    if (tok_.kind != ActionTok::Kind::Assign)
      throw std::runtime_error("template: expected := after range variable");
    advance();
    return out;
  }

  // "$x := pipeline" as a plain action; returns (name, expr) or ("",null)
  std::pair<std::string, ExprPtr> tryParseAssignment() {
    if (tok_.kind != ActionTok::Kind::Var) return {"", nullptr};
    std::string name = tok_.text;
    ActionLexer save = lex_;
    ActionTok savedTok = tok_;
    advance();
    if (tok_.kind == ActionTok::Kind::Assign) {
      advance();
      return {name, parsePipeline()};
    }
    lex_ = save;
    tok_ = savedTok;
    return {"", nullptr};
  }

 private:
  ActionLexer lex_;
  ActionTok tok_;

  void advance() { tok_ = lex_.next(); }

  // command: ident arg*  |  operand
  ExprPtr parseCommand() {
    if (tok_.kind == ActionTok::Kind::Ident) {
      auto call = std::make_shared<Expr>();
      call->kind = Expr::Kind::Call;
      call->fname = tok_.text;
      advance();
      while (tok_.kind != ActionTok::Kind::End &&
             tok_.kind != ActionTok::Kind::Pipe &&
             tok_.kind != ActionTok::Kind::RParen) {
        call->args.push_back(parseOperand());
      }
      return call;
    }
    return parseOperand();
  }

  ExprPtr parseOperand() {
    auto e = std::make_shared<Expr>();
    switch (tok_.kind) {
      case ActionTok::Kind::Field:
        if (tok_.text.empty()) {
          e->kind = Expr::Kind::Dot;
        } else {
          e->kind = Expr::Kind::Field;
          e->path = splitPath(tok_.text);
        }
        advance();
        return e;
      case ActionTok::Kind::Var:
        e->kind = Expr::Kind::Var;
        e->path = {tok_.text};
        advance();
        return e;
      case ActionTok::Kind::Str:
        e->kind = Expr::Kind::StrLit;
        e->strval = tok_.text;
        advance();
        return e;
      case ActionTok::Kind::Num:
        e->kind = Expr::Kind::NumLit;
        e->numval = tok_.num;
        advance();
        return e;
      case ActionTok::Kind::LParen: {
        advance();
        auto inner = parsePipeline();
        if (tok_.kind != ActionTok::Kind::RParen)
          throw std::runtime_error("template: expected ')'");
        advance();
        return inner;
      }
      default:
        throw std::runtime_error("template: unexpected token in action");
    }
  }
};

// ---------- template parser ----------

struct RawAction {
  std::string body;  // inner text of {{ ... }} with trim markers removed
  bool trimLeft = false, trimRight = false;
};

class TemplateParser {
 public:
  explicit TemplateParser(const std::string& text) : text_(text) {}

  std::vector<NodePtr> parse() {
    auto nodes = parseNodes(nullptr);
    return nodes;
  }

 private:
  const std::string& text_;
  size_t pos_ = 0;
  bool pendingTrim_ = false;  // previous action had a right-trim marker
  int depth_ = 0;

  // terminator: if non-null, stop at {{end}} / {{else}}; sets *terminator
  std::vector<NodePtr> parseNodes(std::string* terminator) {
    if (++depth_ > 200)
      throw std::runtime_error("template: block nesting too deep");
    std::vector<NodePtr> nodes;
    struct DepthGuard {
      int& d;
      ~DepthGuard() { d--; }
    } guard{depth_};
    while (pos_ < text_.size()) {
      size_t open = text_.find("{{", pos_);
      std::string raw = text_.substr(
          pos_, (open == std::string::npos ? text_.size() : open) - pos_);
      if (open == std::string::npos) {
        emitText(nodes, raw, false);
        pos_ = text_.size();
        break;
      }
      size_t close = text_.find("}}", open + 2);
      if (close == std::string::npos)
        throw std::runtime_error("template: unclosed action");
      std::string body = text_.substr(open + 2, close - open - 2);
      bool trimL = false, trimR = false;
      if (!body.empty() && body.front() == '-' &&
          (body.size() == 1 || isspace((unsigned char)body[1]))) {
        trimL = true;
        body = body.substr(1);
      }
      if (!body.empty() && body.back() == '-' &&
          (body.size() == 1 || isspace((unsigned char)body[body.size() - 2]))) {
        trimR = true;
        body.pop_back();
      }
      emitText(nodes, raw, trimL);
      pos_ = close + 2;
      pendingTrim_ = trimR;

      std::string trimmed = trim(body);
      // template comments: {{/* ... */}}
      if (trimmed.rfind("/*", 0) == 0 &&
          trimmed.size() >= 4 &&
          trimmed.compare(trimmed.size() - 2, 2, "*/") == 0) {
        continue;
      }
      std::string keyword = firstWord(trimmed);
      if (keyword == "end" || keyword == "else") {
        if (!terminator)
          throw std::runtime_error("template: unexpected {{" + keyword + "}}");
        *terminator = trimmed;  // full body: "else if ..." keeps its tail
        return nodes;
      }
      if (keyword == "if" || keyword == "with") {
        nodes.push_back(parseIfChain(trimmed));
        continue;
      }
      if (keyword == "range") {
        auto node = std::make_shared<Node>();
        node->kind = Node::Kind::Range;
        ActionParser ap(trimmed.substr(5));
        auto vars = ap.parseRangeVars();
        node->rangeVar = vars.first;
        node->rangeVar2 = vars.second;
        node->expr = ap.parsePipeline();
        std::string term;
        node->body = parseNodes(&term);
        if (term != "end")
          throw std::runtime_error("template: expected {{end}} to close range");
        nodes.push_back(node);
        continue;
      }
      // assignment action: {{ $x := expr }}
      {
        ActionParser ap(trimmed);
        auto assign = ap.tryParseAssignment();
        if (!assign.first.empty()) {
          auto node = std::make_shared<Node>();
          node->kind = Node::Kind::Assign;
          node->rangeVar = assign.first;
          node->expr = assign.second;
          nodes.push_back(node);
          continue;
        }
      }
      // plain action
      auto node = std::make_shared<Node>();
      node->kind = Node::Kind::Action;
      ActionParser ap(trimmed);
      node->expr = ap.parsePipeline();
      nodes.push_back(node);
    }
    if (terminator && pos_ >= text_.size() && terminator->empty())
      throw std::runtime_error("template: unexpected EOF, expected {{end}}");
    return nodes;
  }

  // parse "{{if expr}}" or "{{with expr}}" including else / else-if
  // chains, consuming exactly one {{end}}
  NodePtr parseIfChain(const std::string& header) {
    bool isWith = firstWord(header) == "with";
    auto node = std::make_shared<Node>();
    node->kind = isWith ? Node::Kind::With : Node::Kind::If;
    ActionParser ap(header.substr(isWith ? 4 : 2));
    node->expr = ap.parsePipeline();
    std::string term;
    node->body = parseNodes(&term);
    if (firstWord(term) == "else") {
      std::string rest = trim(term.substr(4));
      if (!rest.empty()) {
        // {{else if ...}}: nested chain shares our {{end}}
        if (firstWord(rest) != "if")
          throw std::runtime_error("template: expected 'if' after 'else'");
        node->elseBody.push_back(parseIfChain(rest));
      } else {
        std::string term2;
        node->elseBody = parseNodes(&term2);
        if (term2 != "end")
          throw std::runtime_error("template: expected {{end}}");
      }
    } else if (term != "end") {
      throw std::runtime_error("template: expected {{end}}");
    }
    return node;
  }

  void emitText(std::vector<NodePtr>& nodes, std::string raw, bool trimRightOfText) {
    if (pendingTrim_) {
      size_t i = 0;
      while (i < raw.size() && isspace((unsigned char)raw[i])) i++;
      raw = raw.substr(i);
      pendingTrim_ = false;
    }
    if (trimRightOfText) {
      size_t i = raw.size();
      while (i > 0 && isspace((unsigned char)raw[i - 1])) i--;
      raw = raw.substr(0, i);
    }
    if (raw.empty()) return;
    auto node = std::make_shared<Node>();
    node->kind = Node::Kind::Text;
    node->text = std::move(raw);
    nodes.push_back(node);
  }

  static std::string trim(const std::string& s) {
    size_t a = 0, b = s.size();
    while (a < b && isspace((unsigned char)s[a])) a++;
    while (b > a && isspace((unsigned char)s[b - 1])) b--;
    return s.substr(a, b - a);
  }

  static std::string firstWord(const std::string& s) {
    size_t i = 0;
    while (i < s.size() && !isspace((unsigned char)s[i])) i++;
    return s.substr(0, i);
  }
};

// ---------- functions (template.go:19-140) ----------

Value fnDefault(const std::vector<Value>& args) {
  if (args.size() != 2)
    throw std::runtime_error("template: wrong number of args for default");
  const Value& defVal = args[0];
  const Value& tmplVal = args[1];
  if (tmplVal.kind == Value::Kind::Str && !tmplVal.s.empty())
    return tmplVal;
  if (defVal.kind == Value::Kind::Str) return defVal;
  return Value::str(defVal.print());
}

Value fnEnv(const std::vector<Value>& args) {
  if (args.size() != 1)
    throw std::runtime_error("template: wrong number of args for env");
  const char* v = getenv(args[0].print().c_str());
  return Value::str(v ? v : "");
}

Value fnSplit(const std::vector<Value>& args) {
  if (args.size() != 2)
    throw std::runtime_error("template: wrong number of args for split");
  std::string sep = args[0].print();
  std::string s = args[1].print();
  // TrimSpace first (template.go:19-25)
  size_t a = 0, b = s.size();
  while (a < b && isspace((unsigned char)s[a])) a++;
  while (b > a && isspace((unsigned char)s[b - 1])) b--;
  s = s.substr(a, b - a);
  ValueList out;
  if (s.empty()) return Value::mklist(out);
  if (sep.empty()) {
    for (char c : s) out.push_back(Value::str(std::string(1, c)));
    return Value::mklist(out);
  }
  size_t pos = 0;
  while (true) {
    size_t next = s.find(sep, pos);
    if (next == std::string::npos) {
      out.push_back(Value::str(s.substr(pos)));
      break;
    }
    out.push_back(Value::str(s.substr(pos, next - pos)));
    pos = next + sep.size();
  }
  return Value::mklist(out);
}

Value fnJoin(const std::vector<Value>& args) {
  if (args.size() != 2)
    throw std::runtime_error("template: wrong number of args for join");
  std::string sep = args[0].print();
  if (args[1].kind != Value::Kind::List)
    throw std::runtime_error("template: join expects a list");
  std::string out;
  bool first = true;
  for (auto& e : *args[1].list) {
    if (!first) out += sep;
    first = false;
    out += e.print();
  }
  return Value::str(out);
}

Value fnReplaceAll(const std::vector<Value>& args) {
  if (args.size() != 3)
    throw std::runtime_error("template: wrong number of args for replaceAll");
  std::string from = args[0].print(), to = args[1].print(),
              s = args[2].print();
  if (from.empty()) return Value::str(s);
  std::string out;
  size_t pos = 0;
  while (true) {
    size_t next = s.find(from, pos);
    if (next == std::string::npos) {
      out += s.substr(pos);
      break;
    }
    out += s.substr(pos, next - pos);
    out += to;
    pos = next + from.size();
  }
  return Value::str(out);
}

Value fnRegexReplaceAll(const std::vector<Value>& args) {
  if (args.size() != 3)
    throw std::runtime_error(
        "template: wrong number of args for regexReplaceAll");
  std::regex re(args[0].print(), std::regex::ECMAScript);
  return Value::str(std::regex_replace(args[2].print(), re, args[1].print()));
}

Value fnLoop(const std::vector<Value>& args) {
  int64_t start = 0, stop = 0;
  if (args.size() == 1) {
    stop = args[0].toInt();
  } else if (args.size() == 2) {
    start = args[0].toInt();
    stop = args[1].toInt();
  } else {
    throw std::runtime_error(
        "loop: wrong number of arguments, expected 1 or 2, but got " +
        std::to_string(args.size()));
  }
  ValueList out;
  if (stop < start) {
    for (int64_t i = start; i > stop; i--) out.push_back(Value::integer(i));
  } else {
    for (int64_t i = start; i < stop; i++) out.push_back(Value::integer(i));
  }
  return Value::mklist(out);
}

// ---------- Go text/template builtins (always available in Go) ----------

bool valueEq(const Value& a, const Value& b) {
  bool aNum = a.kind == Value::Kind::Int || a.kind == Value::Kind::Float;
  bool bNum = b.kind == Value::Kind::Int || b.kind == Value::Kind::Float;
  if (aNum && bNum) {
    double av = a.kind == Value::Kind::Int ? (double)a.i : a.f;
    double bv = b.kind == Value::Kind::Int ? (double)b.i : b.f;
    return av == bv;
  }
  if (a.kind == Value::Kind::Bool || b.kind == Value::Kind::Bool)
    return a.truthy() == b.truthy();
  return a.print() == b.print();
}

int valueCmp(const Value& a, const Value& b) {
  bool aNum = a.kind == Value::Kind::Int || a.kind == Value::Kind::Float;
  bool bNum = b.kind == Value::Kind::Int || b.kind == Value::Kind::Float;
  if (aNum && bNum) {
    double av = a.kind == Value::Kind::Int ? (double)a.i : a.f;
    double bv = b.kind == Value::Kind::Int ? (double)b.i : b.f;
    return av < bv ? -1 : (av > bv ? 1 : 0);
  }
  return a.print().compare(b.print()) < 0
             ? -1
             : (a.print() == b.print() ? 0 : 1);
}

Value fnLen(const std::vector<Value>& args) {
  if (args.size() != 1)
    throw std::runtime_error("template: wrong number of args for len");
  const Value& v = args[0];
  if (v.kind == Value::Kind::List) return Value::integer((int64_t)v.list->size());
  if (v.kind == Value::Kind::Str) return Value::integer((int64_t)v.s.size());
  throw std::runtime_error("template: len of unsupported type");
}

Value fnIndex(const std::vector<Value>& args) {
  if (args.size() != 2)
    throw std::runtime_error("template: wrong number of args for index");
  int64_t i = args[1].toInt();
  if (args[0].kind == Value::Kind::List) {
    if (i < 0 || (size_t)i >= args[0].list->size())
      throw std::runtime_error("template: index out of range");
    return (*args[0].list)[i];
  }
  if (args[0].kind == Value::Kind::Str) {
    if (i < 0 || (size_t)i >= args[0].s.size())
      throw std::runtime_error("template: index out of range");
    return Value::integer((unsigned char)args[0].s[i]);
  }
  throw std::runtime_error("template: index of unsupported type");
}

Value fnPrint(const std::vector<Value>& args, bool newline) {
  // fmt.Sprint semantics: spaces between operands when neither is a string
  std::string out;
  for (size_t i = 0; i < args.size(); i++) {
    if (i > 0 && args[i - 1].kind != Value::Kind::Str &&
        args[i].kind != Value::Kind::Str && !newline)
      out += " ";
    else if (i > 0 && newline)
      out += " ";  // Sprintln: always spaces
    out += args[i].print();
  }
  if (newline) out += "\n";
  return Value::str(out);
}

Value fnPrintf(const std::vector<Value>& args) {
  if (args.empty())
    throw std::runtime_error("template: printf needs a format string");
  const std::string& fmt = args[0].kind == Value::Kind::Str
                               ? args[0].s
                               : throw std::runtime_error(
                                     "template: printf format must be string");
  std::string out;
  size_t argi = 1;
  for (size_t i = 0; i < fmt.size(); i++) {
    if (fmt[i] != '%') {
      out += fmt[i];
      continue;
    }
    if (i + 1 >= fmt.size()) break;
    char spec = fmt[++i];
    if (spec == '%') {
      out += '%';
      continue;
    }
    if (argi >= args.size())
      throw std::runtime_error("template: printf: not enough args");
    const Value& a = args[argi++];
    switch (spec) {
      case 's':
      case 'v':
        out += a.print();
        break;
      case 'd':
        out += std::to_string(a.toInt());
        break;
      case 'f': {
        char buf[40];
        snprintf(buf, sizeof(buf), "%f",
                 a.kind == Value::Kind::Float ? a.f : (double)a.toInt());
        out += buf;
        break;
      }
      default:
        throw std::runtime_error(std::string("template: printf: unsupported verb %") + spec);
    }
  }
  return Value::str(out);
}

// ---------- evaluator ----------

struct Scope {
  std::map<std::string, Value> vars;
  Value dot;  // current "." (Nil means "the env map")
  bool dotIsEnv = true;
};

class Evaluator {
 public:
  Evaluator() {
    for (char** e = environ; *e; e++) {
      const char* eq = strchr(*e, '=');
      if (!eq) continue;
      env_[std::string(*e, eq - *e)] = std::string(eq + 1);
    }
  }

  std::string exec(const std::vector<NodePtr>& nodes) {
    Scope scope;
    std::string out;
    execNodes(nodes, scope, out);
    return out;
  }

 private:
  std::map<std::string, std::string> env_;

  void execNodes(const std::vector<NodePtr>& nodes, Scope& scope,
                 std::string& out) {
    for (auto& n : nodes) {
      switch (n->kind) {
        case Node::Kind::Text:
          out += n->text;
          break;
        case Node::Kind::Action:
          out += eval(n->expr, scope).print();
          break;
        case Node::Kind::If: {
          Value cond = eval(n->expr, scope);
          if (cond.truthy())
            execNodes(n->body, scope, out);
          else
            execNodes(n->elseBody, scope, out);
          break;
        }
        case Node::Kind::Range: {
          Value coll = eval(n->expr, scope);
          if (coll.kind != Value::Kind::List)
            throw std::runtime_error("template: range over non-list value");
          int64_t idx = 0;
          for (auto& item : *coll.list) {
            Scope inner = scope;
            inner.dot = item;
            inner.dotIsEnv = false;
            if (!n->rangeVar2.empty()) {
              // two-var form: $i = index, $v = value
              inner.vars[n->rangeVar] = Value::integer(idx);
              inner.vars[n->rangeVar2] = item;
            } else if (!n->rangeVar.empty()) {
              inner.vars[n->rangeVar] = item;
            }
            execNodes(n->body, inner, out);
            idx++;
          }
          break;
        }
        case Node::Kind::Assign:
          scope.vars[n->rangeVar] = eval(n->expr, scope);
          break;
        case Node::Kind::With: {
          Value v = eval(n->expr, scope);
          if (v.truthy()) {
            Scope inner = scope;
            inner.dot = v;
            inner.dotIsEnv = false;
            execNodes(n->body, inner, out);
          } else {
            execNodes(n->elseBody, scope, out);
          }
          break;
        }
      }
    }
  }

  Value eval(const ExprPtr& e, Scope& scope) {
    switch (e->kind) {
      case Expr::Kind::StrLit:
        return Value::str(e->strval);
      case Expr::Kind::NumLit:
        return e->numval;
      case Expr::Kind::Dot:
        return scope.dot;
      case Expr::Kind::Field: {
        if (!scope.dotIsEnv) {
          // field access on a non-map value: missingkey=zero -> empty
          return Value::str("");
        }
        // .A -> env lookup; missing key = zero value ("")
        auto it = env_.find(e->path[0]);
        std::string v = (it == env_.end()) ? "" : it->second;
        if (e->path.size() > 1) return Value::str("");  // .A.B on a string
        return Value::str(v);
      }
      case Expr::Kind::Var: {
        auto it = scope.vars.find(e->path[0]);
        if (it == scope.vars.end())
          throw std::runtime_error("template: undefined variable $" +
                                   e->path[0]);
        return it->second;
      }
      case Expr::Kind::Call: {
        std::vector<Value> args;
        for (auto& a : e->args) args.push_back(eval(a, scope));
        return callFn(e->fname, args);
      }
      case Expr::Kind::Pipeline: {
        Value v = eval(e->stages[0], scope);
        for (size_t i = 1; i < e->stages.size(); i++) {
          auto& stage = e->stages[i];
          if (stage->kind != Expr::Kind::Call)
            throw std::runtime_error("template: pipeline stage must be a call");
          std::vector<Value> args;
          for (auto& a : stage->args) args.push_back(eval(a, scope));
          args.push_back(v);  // piped value becomes the last argument
          v = callFn(stage->fname, args);
        }
        return v;
      }
    }
    return Value::nil();
  }

  Value callFn(const std::string& name, const std::vector<Value>& args) {
    if (name == "default") return fnDefault(args);
    if (name == "env") return fnEnv(args);
    if (name == "split") return fnSplit(args);
    if (name == "join") return fnJoin(args);
    if (name == "replaceAll") return fnReplaceAll(args);
    if (name == "regexReplaceAll") return fnRegexReplaceAll(args);
    if (name == "loop") return fnLoop(args);
    if (name == "printf") return fnPrintf(args);
    // Go text/template builtins (template.go's FuncMap only ADDS funcs;
    // the language builtins are always available to reference configs)
    if (name == "eq") {
      if (args.size() < 2)
        throw std::runtime_error("template: wrong number of args for eq");
      for (size_t i = 1; i < args.size(); i++)
        if (valueEq(args[0], args[i])) return Value::boolean(true);
      return Value::boolean(false);
    }
    if (name == "ne") {
      if (args.size() != 2)
        throw std::runtime_error("template: wrong number of args for ne");
      return Value::boolean(!valueEq(args[0], args[1]));
    }
    if (name == "lt" || name == "le" || name == "gt" || name == "ge") {
      if (args.size() != 2)
        throw std::runtime_error("template: wrong number of args for " + name);
      int c = valueCmp(args[0], args[1]);
      bool r = (name == "lt")   ? c < 0
               : (name == "le") ? c <= 0
               : (name == "gt") ? c > 0
                                : c >= 0;
      return Value::boolean(r);
    }
    if (name == "and") {
      if (args.empty())
        throw std::runtime_error("template: wrong number of args for and");
      for (auto& a : args)
        if (!a.truthy()) return a;
      return args.back();
    }
    if (name == "or") {
      if (args.empty())
        throw std::runtime_error("template: wrong number of args for or");
      for (auto& a : args)
        if (a.truthy()) return a;
      return args.back();
    }
    if (name == "not") {
      if (args.size() != 1)
        throw std::runtime_error("template: wrong number of args for not");
      return Value::boolean(!args[0].truthy());
    }
    if (name == "len") return fnLen(args);
    if (name == "index") return fnIndex(args);
    if (name == "print") return fnPrint(args, false);
    if (name == "println") return fnPrint(args, true);
    throw std::runtime_error("template: function \"" + name + "\" not defined");
  }
};

}  // namespace

std::string renderTemplate(const std::string& text) {
  TemplateParser parser(text);
  auto nodes = parser.parse();
  Evaluator ev;
  return ev.exec(nodes);
}

}  // namespace cpilot
