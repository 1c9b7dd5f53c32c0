// xmlio.cpp — gates.xsd XML writer + strict hand-rolled reader.
//
// The reference uses libxml2 (state.c); this implementation carries no
// external dependency: a minimal well-formed-XML parser adequate for the
// gates.xsd vocabulary (elements, attributes, comments, XML declaration,
// the five predefined entities), followed by the same semantic validation
// the reference performs on load (state.c:260-411): gate inputs must
// reference earlier gates, IN gates must form a prefix of at most 8, arity
// must match the type, LUT `function` is one hex byte, outputs are unique
// bits 0-7, and every truth table is recomputed from scratch.

#include "sbg/xmlio.hpp"

#include <cctype>
#include <cstdio>
#include <cstring>
#include <memory>
#include <vector>

namespace sbg {

namespace {

struct XmlNode {
  std::string name;
  std::vector<std::pair<std::string, std::string>> attrs;
  std::vector<std::unique_ptr<XmlNode>> children;

  const char* attr(const char* key) const {
    for (const auto& kv : attrs) {
      if (kv.first == key) return kv.second.c_str();
    }
    return nullptr;
  }
};

class XmlParser {
 public:
  explicit XmlParser(const std::string& text) : s_(text) {}

  std::unique_ptr<XmlNode> parse(std::string* err) {
    skip_misc();
    auto root = parse_element(err);
    if (root == nullptr) return nullptr;
    skip_misc();
    if (pos_ != s_.size()) {
      set_err(err, "trailing content after root element");
      return nullptr;
    }
    return root;
  }

 private:
  const std::string& s_;
  size_t pos_ = 0;

  void set_err(std::string* err, const char* msg) {
    if (err != nullptr) {
      char buf[128];
      std::snprintf(buf, sizeof(buf), "XML parse error at offset %zu: %s", pos_, msg);
      *err = buf;
    }
  }

  bool starts_with(const char* p) const {
    return s_.compare(pos_, std::strlen(p), p) == 0;
  }

  void skip_ws() {
    while (pos_ < s_.size() && std::isspace(static_cast<unsigned char>(s_[pos_]))) pos_++;
  }

  // Skips whitespace, comments, the XML declaration and processing
  // instructions / DOCTYPE (ignored, not validated).
  void skip_misc() {
    for (;;) {
      skip_ws();
      if (starts_with("<!--")) {
        size_t end = s_.find("-->", pos_ + 4);
        pos_ = end == std::string::npos ? s_.size() : end + 3;
      } else if (starts_with("<?")) {
        size_t end = s_.find("?>", pos_ + 2);
        pos_ = end == std::string::npos ? s_.size() : end + 2;
      } else if (starts_with("<!")) {
        size_t end = s_.find('>', pos_ + 2);
        pos_ = end == std::string::npos ? s_.size() : end + 1;
      } else {
        return;
      }
    }
  }

  static bool name_char(char c) {
    return std::isalnum(static_cast<unsigned char>(c)) || c == '_' || c == '-' ||
           c == '.' || c == ':';
  }

  std::string parse_name() {
    size_t start = pos_;
    while (pos_ < s_.size() && name_char(s_[pos_])) pos_++;
    return s_.substr(start, pos_ - start);
  }

  bool parse_entity(std::string* out, std::string* err) {
    // pos_ is at '&'.
    size_t semi = s_.find(';', pos_);
    if (semi == std::string::npos || semi - pos_ > 8) {
      set_err(err, "bad entity");
      return false;
    }
    std::string ent = s_.substr(pos_ + 1, semi - pos_ - 1);
    pos_ = semi + 1;
    if (ent == "lt") *out += '<';
    else if (ent == "gt") *out += '>';
    else if (ent == "amp") *out += '&';
    else if (ent == "quot") *out += '"';
    else if (ent == "apos") *out += '\'';
    else if (!ent.empty() && ent[0] == '#') {
      long v = std::strtol(ent.c_str() + (ent[1] == 'x' ? 2 : 1), nullptr,
                           ent[1] == 'x' ? 16 : 10);
      *out += static_cast<char>(v);
    } else {
      set_err(err, "unknown entity");
      return false;
    }
    return true;
  }

  std::unique_ptr<XmlNode> parse_element(std::string* err) {
    if (pos_ >= s_.size() || s_[pos_] != '<') {
      set_err(err, "expected element");
      return nullptr;
    }
    pos_++;  // '<'
    auto node = std::make_unique<XmlNode>();
    node->name = parse_name();
    if (node->name.empty()) {
      set_err(err, "empty element name");
      return nullptr;
    }
    // Attributes.
    for (;;) {
      skip_ws();
      if (pos_ >= s_.size()) { set_err(err, "unexpected end in tag"); return nullptr; }
      if (s_[pos_] == '/' || s_[pos_] == '>') break;
      std::string key = parse_name();
      if (key.empty()) { set_err(err, "bad attribute name"); return nullptr; }
      skip_ws();
      if (pos_ >= s_.size() || s_[pos_] != '=') { set_err(err, "expected '='"); return nullptr; }
      pos_++;
      skip_ws();
      if (pos_ >= s_.size() || (s_[pos_] != '"' && s_[pos_] != '\'')) {
        set_err(err, "expected quoted attribute value");
        return nullptr;
      }
      char quote = s_[pos_++];
      std::string val;
      while (pos_ < s_.size() && s_[pos_] != quote) {
        if (s_[pos_] == '&') {
          if (!parse_entity(&val, err)) return nullptr;
        } else {
          val += s_[pos_++];
        }
      }
      if (pos_ >= s_.size()) { set_err(err, "unterminated attribute"); return nullptr; }
      pos_++;  // closing quote
      node->attrs.emplace_back(std::move(key), std::move(val));
    }
    if (s_[pos_] == '/') {
      pos_++;
      if (pos_ >= s_.size() || s_[pos_] != '>') { set_err(err, "expected '>'"); return nullptr; }
      pos_++;
      return node;
    }
    pos_++;  // '>'
    // Children: elements and ignorable text/comments until </name>.
    for (;;) {
      // Text content (ignored — the schema has none).
      while (pos_ < s_.size() && s_[pos_] != '<') pos_++;
      if (pos_ >= s_.size()) { set_err(err, "missing end tag"); return nullptr; }
      if (starts_with("<!--")) {
        size_t end = s_.find("-->", pos_ + 4);
        pos_ = end == std::string::npos ? s_.size() : end + 3;
        continue;
      }
      if (starts_with("</")) {
        pos_ += 2;
        std::string closing = parse_name();
        skip_ws();
        if (closing != node->name || pos_ >= s_.size() || s_[pos_] != '>') {
          set_err(err, "mismatched end tag");
          return nullptr;
        }
        pos_++;
        return node;
      }
      auto child = parse_element(err);
      if (child == nullptr) return nullptr;
      node->children.push_back(std::move(child));
    }
  }
};

int gate_type_from_name(const char* name) {
  for (int t = 0; t <= LUT; t++) {
    if (std::strcmp(name, gate_name[t]) == 0) return t;
  }
  return -1;
}

bool parse_uint(const char* s, long* out) {
  if (s == nullptr || *s == '\0') return false;
  char* end = nullptr;
  long v = std::strtol(s, &end, 10);
  if (*end != '\0' || v < 0) return false;
  *out = v;
  return true;
}

}  // namespace

std::string state_to_xml(const state& st) {
  // Text format parity with the reference writer (state.c:127-166).
  std::string out;
  out.reserve(4096);
  char buf[96];
  out += "<?xml version=\"1.0\" encoding=\"UTF-8\" ?>\n";
  out += "<gates>\n";
  for (int i = 0; i < 8; i++) {
    if (st.outputs[i] != NO_GATE) {
      std::snprintf(buf, sizeof(buf), "  <output bit=\"%d\" gate=\"%d\" />\n", i,
                    st.outputs[i]);
      out += buf;
    }
  }
  for (int i = 0; i < st.num_gates; i++) {
    const gate& g = st.gates[i];
    if (g.type == IN) {
      out += "  <gate type=\"IN\" />\n";
      continue;
    }
    if (g.type == LUT) {
      std::snprintf(buf, sizeof(buf), "  <gate type=\"LUT\" function=\"%02x\">\n",
                    g.function);
    } else {
      std::snprintf(buf, sizeof(buf), "  <gate type=\"%s\">\n", gate_name[g.type]);
    }
    out += buf;
    const gatenum ins[3] = {g.in1, g.in2, g.in3};
    for (gatenum in : ins) {
      if (in != NO_GATE) {
        std::snprintf(buf, sizeof(buf), "    <input gate=\"%d\" />\n", in);
        out += buf;
      }
    }
    out += "  </gate>\n";
  }
  out += "</gates>\n";
  return out;
}

std::string save_state(const state& st, const std::string& dir) {
  std::string path = state_file_name(st);
  if (!dir.empty()) path = dir + "/" + path;
  // Atomic checkpoint: write a temp file and rename into place, so a
  // process killed mid-write (budget windows SIGKILL at arbitrary
  // points) can never leave a truncated state file for --resume-dir to
  // trip over. Found by the interrupt/resume soak (tools/resume_soak.py).
  std::string tmp = path + ".tmp";
  FILE* fp = std::fopen(tmp.c_str(), "w");
  if (fp == nullptr) {
    std::fprintf(stderr, "sboxgates: error opening %s for writing\n", tmp.c_str());
    return "";
  }
  std::string xml = state_to_xml(st);
  bool ok = std::fwrite(xml.data(), 1, xml.size(), fp) == xml.size();
  ok = std::fclose(fp) == 0 && ok;
  if (ok) ok = std::rename(tmp.c_str(), path.c_str()) == 0;
  if (!ok) {
    std::remove(tmp.c_str());
    std::fprintf(stderr, "sboxgates: error writing %s\n", path.c_str());
    return "";
  }
  return path;
}

bool state_from_xml(const std::string& xml, state* out, std::string* err) {
  XmlParser parser(xml);
  auto root = parser.parse(err);
  if (root == nullptr) return false;
  if (root->name != "gates") {
    if (err != nullptr) *err = "root element is not <gates>";
    return false;
  }

  auto fail = [err](const char* msg) {
    if (err != nullptr) *err = msg;
    return false;
  };

  state st;
  std::memset(&st, 0, sizeof(state));
  st.max_gates = MAX_GATES;
  st.max_sat_metric = INT_MAX;
  for (int i = 0; i < 8; i++) st.outputs[i] = NO_GATE;

  // Pass 1: gates, in document order. Inputs must reference earlier gates;
  // IN gates must form a prefix of length <= 8 (parity: state.c:300-380).
  for (const auto& node : root->children) {
    if (node->name != "gate") continue;
    if (st.num_gates >= MAX_GATES) return fail("too many gates");

    const char* typestr = node->attr("type");
    if (typestr == nullptr) return fail("gate without type");
    int type = gate_type_from_name(typestr);
    if (type < 0) return fail("unknown gate type");

    long func = 0;
    const char* funcstr = node->attr("function");
    if (funcstr != nullptr) {
      char* end = nullptr;
      func = std::strtol(funcstr, &end, 16);
      // Full-token consumption: trailing garbage ("1g") is an error, not 1.
      if (end == funcstr || *end != '\0' || func <= 0 || func > 255) {
        return fail("bad LUT function");
      }
    }
    if (type != LUT && func != 0) return fail("function on non-LUT gate");
    if (type == LUT && funcstr == nullptr) return fail("LUT gate without function");

    int inp = 0;
    gatenum inputs[3] = {NO_GATE, NO_GATE, NO_GATE};
    for (const auto& child : node->children) {
      if (child->name != "input") continue;
      if (inp >= 3) return fail("too many gate inputs");
      long g = 0;
      if (!parse_uint(child->attr("gate"), &g)) return fail("bad input gate id");
      if (g >= st.num_gates) return fail("input references later gate");
      inputs[inp++] = static_cast<gatenum>(g);
    }

    ttable table;
    if (type <= TRUE_GATE) {
      if (inp != 2) return fail("2-input gate arity mismatch");
      table = gen_ttable_2(type, st.gates[inputs[0]].table, st.gates[inputs[1]].table);
    } else if (type == NOT) {
      if (inp != 1) return fail("NOT gate arity mismatch");
      table = ~st.gates[inputs[0]].table;
    } else if (type == IN) {
      if (inp != 0) return fail("IN gate with inputs");
      if (st.num_gates >= 8) return fail("more than 8 input gates");
      if (st.num_gates != 0 && st.gates[st.num_gates - 1].type != IN) {
        return fail("IN gates must precede all other gates");
      }
      table = generate_target(static_cast<u8>(st.num_gates), nullptr);
    } else {  // LUT
      if (inp != 3) return fail("LUT arity mismatch");
      table = gen_lut_ttable(static_cast<u8>(func), st.gates[inputs[0]].table,
                             st.gates[inputs[1]].table, st.gates[inputs[2]].table);
    }

    gate& g = st.gates[st.num_gates];
    g.table = table;
    g.type = type;
    g.in1 = inputs[0];
    g.in2 = inputs[1];
    g.in3 = inputs[2];
    g.function = static_cast<u8>(func);
    st.num_gates += 1;
  }

  // Pass 2: outputs (parity: state.c:383-411).
  for (const auto& node : root->children) {
    if (node->name != "output") continue;
    long bit = 0, g = 0;
    if (!parse_uint(node->attr("bit"), &bit) || bit >= 8) return fail("bad output bit");
    if (st.outputs[bit] != NO_GATE) return fail("duplicate output bit");
    if (!parse_uint(node->attr("gate"), &g) || g >= st.num_gates) {
      return fail("bad output gate");
    }
    st.outputs[bit] = static_cast<gatenum>(g);
  }

  // SAT metric: zero when any LUT is present (parity: state.c:399-407).
  st.sat_metric = 0;
  for (int i = 0; i < st.num_gates; i++) {
    if (st.gates[i].type == LUT) {
      st.sat_metric = 0;
      break;
    }
    st.sat_metric += sat_metric_of(st.gates[i].type);
  }

  *out = st;
  return true;
}

bool load_state(const std::string& path, state* out, std::string* err) {
  FILE* fp = std::fopen(path.c_str(), "rb");
  if (fp == nullptr) {
    if (err != nullptr) *err = "cannot open " + path;
    return false;
  }
  std::string xml;
  char buf[65536];
  size_t n;
  while ((n = std::fread(buf, 1, sizeof(buf), fp)) > 0) xml.append(buf, n);
  std::fclose(fp);
  return state_from_xml(xml, out, err);
}

}  // namespace sbg
