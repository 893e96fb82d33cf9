// Minimal YAML subset parser for the sched-pipeline input schemas.
//
// The reference links yaml-cpp (reference src-native/CMakeLists.txt); this
// environment ships no yaml-cpp, so we parse the subset the PipeEdge schemas
// actually use (README_Scheduler.md:44-264): block mappings, block sequences
// ("- item" / "- key: value"), flow sequences "[a, b, ...]" (possibly
// spanning lines), and plain scalars (strings / ints / floats). Anchors,
// aliases, multi-docs, and block scalars are intentionally unsupported.
#pragma once

#include <algorithm>
#include <cctype>
#include <cstdlib>
#include <fstream>
#include <map>
#include <memory>
#include <sstream>
#include <stdexcept>
#include <string>
#include <vector>

namespace minyaml {

struct Node;
using NodePtr = std::shared_ptr<Node>;

struct Node {
    enum Kind { Scalar, Map, Seq } kind = Scalar;
    std::string scalar;
    std::vector<std::pair<std::string, NodePtr>> map;  // ordered
    std::vector<NodePtr> seq;

    const NodePtr& at(const std::string& key) const {
        for (const auto& kv : map)
            if (kv.first == key) return kv.second;
        throw std::runtime_error("minyaml: missing key: " + key);
    }
    bool has(const std::string& key) const {
        for (const auto& kv : map)
            if (kv.first == key) return true;
        return false;
    }
    double as_double() const { return std::strtod(scalar.c_str(), nullptr); }
    long as_long() const { return std::strtol(scalar.c_str(), nullptr, 10); }
    std::vector<double> as_double_list() const {
        std::vector<double> out;
        out.reserve(seq.size());
        for (const auto& n : seq) out.push_back(n->as_double());
        return out;
    }
    std::vector<long> as_long_list() const {
        std::vector<long> out;
        out.reserve(seq.size());
        for (const auto& n : seq) out.push_back(n->as_long());
        return out;
    }
};

namespace detail {

struct Line {
    int indent;
    std::string text;  // content without leading spaces / trailing comment
};

inline std::string strip(const std::string& s) {
    size_t a = s.find_first_not_of(" \t\r\n");
    if (a == std::string::npos) return "";
    size_t b = s.find_last_not_of(" \t\r\n");
    return s.substr(a, b - a + 1);
}

inline std::string unquote(const std::string& s) {
    if (s.size() >= 2 && ((s.front() == '"' && s.back() == '"') ||
                          (s.front() == '\'' && s.back() == '\'')))
        return s.substr(1, s.size() - 2);
    return s;
}

inline std::vector<Line> read_lines(std::istream& in) {
    std::vector<Line> lines;
    std::string raw;
    while (std::getline(in, raw)) {
        // strip comments (naive: '#' outside quotes)
        std::string s;
        bool in_sq = false, in_dq = false;
        for (char c : raw) {
            if (c == '\'' && !in_dq) in_sq = !in_sq;
            else if (c == '"' && !in_sq) in_dq = !in_dq;
            if (c == '#' && !in_sq && !in_dq) break;
            s += c;
        }
        std::string content = strip(s);
        if (content.empty() || content == "---") continue;
        int indent = 0;
        while (indent < (int)s.size() && s[indent] == ' ') ++indent;
        lines.push_back({indent, content});
    }
    return lines;
}

// split "key: value" at the first ':' that is followed by space/EOL and not
// inside brackets/quotes
inline bool split_kv(const std::string& s, std::string& key,
                     std::string& val) {
    int depth = 0;
    bool in_sq = false, in_dq = false;
    for (size_t i = 0; i < s.size(); ++i) {
        char c = s[i];
        if (c == '\'' && !in_dq) in_sq = !in_sq;
        else if (c == '"' && !in_sq) in_dq = !in_dq;
        else if (!in_sq && !in_dq) {
            if (c == '[' || c == '{') ++depth;
            else if (c == ']' || c == '}') --depth;
            else if (c == ':' && depth == 0 &&
                     (i + 1 == s.size() || s[i + 1] == ' ')) {
                key = unquote(strip(s.substr(0, i)));
                val = strip(s.substr(i + 1));
                return true;
            }
        }
    }
    return false;
}

inline NodePtr parse_flow_seq(const std::string& s) {
    auto node = std::make_shared<Node>();
    node->kind = Node::Seq;
    // s starts with '[' and ends with ']'
    std::string body = s.substr(1, s.size() - 2);
    std::string cur;
    int depth = 0;
    for (char c : body) {
        if (c == '[' || c == '{') ++depth;
        if (c == ']' || c == '}') --depth;
        if (c == ',' && depth == 0) {
            std::string item = strip(cur);
            if (!item.empty()) {
                auto child = std::make_shared<Node>();
                child->scalar = unquote(item);
                node->seq.push_back(child);
            }
            cur.clear();
        } else {
            cur += c;
        }
    }
    std::string item = strip(cur);
    if (!item.empty()) {
        auto child = std::make_shared<Node>();
        child->scalar = unquote(item);
        node->seq.push_back(child);
    }
    return node;
}

class Parser {
  public:
    explicit Parser(std::vector<Line> lines) : lines_(std::move(lines)) {}

    NodePtr parse() {
        pos_ = 0;
        if (lines_.empty()) return std::make_shared<Node>();
        return parse_block(lines_[0].indent);
    }

  private:
    std::vector<Line> lines_;
    size_t pos_ = 0;

    // join continuation lines of a flow sequence "[1, 2," + "3, 4]"
    std::string collect_flow(std::string val, int) {
        while (val.find('[') != std::string::npos &&
               std::count(val.begin(), val.end(), '[') >
                   std::count(val.begin(), val.end(), ']')) {
            if (pos_ >= lines_.size())
                throw std::runtime_error("minyaml: unterminated flow seq");
            val += " " + lines_[pos_++].text;
        }
        return val;
    }

    NodePtr value_node(const std::string& raw, int indent) {
        std::string val = raw;
        if (!val.empty() && val[0] == '[') val = collect_flow(val, indent);
        auto node = std::make_shared<Node>();
        if (!val.empty() && val[0] == '[') {
            return parse_flow_seq(val);
        }
        node->scalar = unquote(val);
        return node;
    }

    NodePtr parse_block(int indent) {
        auto node = std::make_shared<Node>();
        bool first = true;
        while (pos_ < lines_.size()) {
            const Line& ln = lines_[pos_];
            if (ln.indent < indent) break;
            if (ln.indent > indent) {
                throw std::runtime_error("minyaml: unexpected indent: " +
                                         ln.text);
            }
            if (ln.text.rfind("- ", 0) == 0 || ln.text == "-") {
                if (first) {
                    node->kind = Node::Seq;
                    first = false;
                } else if (node->kind != Node::Seq) {
                    break;
                }
                node->seq.push_back(parse_seq_item(indent));
            } else {
                std::string key, val;
                if (!split_kv(ln.text, key, val))
                    throw std::runtime_error("minyaml: bad line: " + ln.text);
                if (first) {
                    node->kind = Node::Map;
                    first = false;
                } else if (node->kind != Node::Map) {
                    break;
                }
                ++pos_;
                // a block sequence may sit at the SAME indent as its key
                const bool seq_same_indent =
                    pos_ < lines_.size() && lines_[pos_].indent == indent &&
                    (lines_[pos_].text.rfind("- ", 0) == 0 ||
                     lines_[pos_].text == "-");
                if (!val.empty()) {
                    node->map.emplace_back(key, value_node(val, indent));
                } else if (pos_ < lines_.size() &&
                           (lines_[pos_].indent > indent || seq_same_indent)) {
                    node->map.emplace_back(key,
                                           parse_block(lines_[pos_].indent));
                } else {
                    auto empty = std::make_shared<Node>();
                    empty->kind = Node::Map;
                    node->map.emplace_back(key, empty);
                }
            }
        }
        return node;
    }

    NodePtr parse_seq_item(int indent) {
        Line ln = lines_[pos_];
        std::string rest = ln.text == "-" ? "" : strip(ln.text.substr(2));
        ++pos_;
        if (rest.empty()) {
            if (pos_ < lines_.size() && lines_[pos_].indent > indent)
                return parse_block(lines_[pos_].indent);
            return std::make_shared<Node>();
        }
        std::string key, val;
        if (split_kv(rest, key, val)) {
            // "- key: value" starts an inline map; continuation keys are
            // indented deeper than the dash
            auto node = std::make_shared<Node>();
            node->kind = Node::Map;
            auto next_is_seq = [&]() {
                return pos_ < lines_.size() &&
                       lines_[pos_].indent == indent + 2 &&
                       (lines_[pos_].text.rfind("- ", 0) == 0 ||
                        lines_[pos_].text == "-");
            };
            if (!val.empty()) {
                node->map.emplace_back(key, value_node(val, indent));
            } else if (pos_ < lines_.size() &&
                       (lines_[pos_].indent > indent + 2 || next_is_seq())) {
                node->map.emplace_back(key, parse_block(lines_[pos_].indent));
            }
            while (pos_ < lines_.size() && lines_[pos_].indent == indent + 2) {
                std::string k2, v2;
                if (!split_kv(lines_[pos_].text, k2, v2)) break;
                ++pos_;
                if (!v2.empty()) {
                    node->map.emplace_back(k2, value_node(v2, indent + 2));
                } else if (pos_ < lines_.size() &&
                           (lines_[pos_].indent > indent + 2 ||
                            next_is_seq())) {
                    node->map.emplace_back(k2,
                                           parse_block(lines_[pos_].indent));
                }
            }
            return node;
        }
        return value_node(rest, indent);
    }
};

}  // namespace detail

inline NodePtr parse_stream(std::istream& in) {
    detail::Parser p(detail::read_lines(in));
    return p.parse();
}

inline NodePtr parse_file(const std::string& path) {
    std::ifstream f(path);
    if (!f) throw std::runtime_error("minyaml: cannot open " + path);
    return parse_stream(f);
}

}  // namespace minyaml
