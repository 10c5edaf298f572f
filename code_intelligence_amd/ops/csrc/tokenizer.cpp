// Native tokenizer core for bulk data prep (reference scale: 16.7M issue
// docs tokenized on a 72-core box for an hour, 02_fastai_DataBunch:213).
//
// Token-exact C++ port of text/tokenizer.py's hot path for ASCII input:
//   replace_rep (char x4+ -> "xxrep N c"), replace_wrep (word x4+ ->
//   "xxwrep N w"), word split (\w+ | single non-space), replace_all_caps,
//   deal_caps. String-level rules that cannot change the token stream
//   (spec_add_spaces, rm_useless_spaces) are no-ops at token level and
//   skipped. fix_html/markdown stay in Python (string-level, cheap).
// The Python wrapper routes only pure-ASCII strings here (exact parity);
// anything else falls back to the Python rules.
#include <torch/extension.h>

#include <cctype>
#include <string>
#include <vector>

namespace ci {

static inline bool is_word(char c) {
  return (c >= 'a' && c <= 'z') || (c >= 'A' && c <= 'Z') ||
         (c >= '0' && c <= '9') || c == '_';
}

static inline bool is_space(unsigned char c) { return std::isspace(c) != 0; }

// pass 1: character repeats (regex (\S)(\1{3,}) -> " xxrep N c ")
static std::string pass_rep(const std::string& s) {
  std::string out;
  out.reserve(s.size() + 16);
  size_t i = 0;
  while (i < s.size()) {
    char c = s[i];
    if (!is_space((unsigned char)c)) {
      size_t j = i + 1;
      while (j < s.size() && s[j] == c) ++j;
      size_t n = j - i;
      if (n >= 4) {
        out += " xxrep ";
        out += std::to_string(n);
        out += ' ';
        out += c;
        out += ' ';
        i = j;
        continue;
      }
    }
    out += c;
    ++i;
  }
  return out;
}

// pass 2: word repeats ((?:\s|^)(\w+)((?:\s+\1){3,})(\s|$))
static std::string pass_wrep(const std::string& s) {
  // split into (whitespace, word-or-symbol chunk) runs, then scan runs of
  // identical \w+ chunks separated only by whitespace.
  struct Tok { std::string text; bool word; };
  std::vector<Tok> toks;     // non-space chunks (maximal \S+ runs)
  std::vector<std::string> gaps;  // gaps[i] = whitespace before toks[i]
  std::string cur, gap;
  size_t i = 0;
  while (i <= s.size()) {
    if (i == s.size() || is_space((unsigned char)s[i])) {
      if (!cur.empty()) {
        bool w = true;
        for (char c : cur) w = w && is_word(c);
        gaps.push_back(gap);
        toks.push_back({cur, w});
        cur.clear();
        gap.clear();
      }
      if (i < s.size()) gap += s[i];
      ++i;
    } else {
      cur += s[i];
      ++i;
    }
  }
  std::string out;
  out.reserve(s.size() + 16);
  size_t k = 0;
  while (k < toks.size()) {
    if (toks[k].word) {
      size_t j = k + 1;
      while (j < toks.size() && toks[j].text == toks[k].text) ++j;
      size_t n = j - k;
      if (n >= 4) {
        out += gaps[k].empty() && k == 0 ? "" : " ";
        out += "xxwrep ";
        out += std::to_string(n);
        out += ' ';
        out += toks[k].text;
        out += ' ';
        k = j;
        continue;
      }
    }
    out += gaps[k];
    out += toks[k].text;
    ++k;
  }
  out += gap;  // trailing whitespace (empty or last gap content)
  return out;
}

// split \w+ | single non-space, then caps rules
static void split_and_caps(const std::string& s, std::vector<std::string>& out) {
  size_t i = 0;
  std::string tok;
  auto emit = [&out](const std::string& t) {
    // replace_all_caps: len>1, all upper alpha -> xxup + lower
    if (t.size() > 1) {
      bool all_upper = true, all_alpha = true;
      for (char c : t) {
        all_alpha = all_alpha && std::isalpha((unsigned char)c);
        all_upper = all_upper && (!std::isalpha((unsigned char)c) ||
                                  std::isupper((unsigned char)c));
      }
      if (all_alpha && all_upper) {
        out.push_back("xxup");
        std::string l = t;
        for (char& c : l) c = std::tolower((unsigned char)c);
        out.push_back(l);
        return;
      }
      // deal_caps: Xxxx (first upper, rest lower, all alpha) -> xxmaj + lower
      bool first_upper = std::isupper((unsigned char)t[0]);
      bool rest_lower = true;
      for (size_t k = 1; k < t.size(); ++k)
        rest_lower = rest_lower && std::islower((unsigned char)t[k]);
      if (all_alpha && first_upper && rest_lower) {
        out.push_back("xxmaj");
        std::string l = t;
        l[0] = std::tolower((unsigned char)l[0]);
        out.push_back(l);
        return;
      }
    }
    out.push_back(t);
  };
  while (i < s.size()) {
    unsigned char c = s[i];
    if (is_word((char)c)) {
      tok.clear();
      while (i < s.size() && is_word(s[i])) tok += s[i++];
      emit(tok);
    } else if (!is_space(c)) {
      emit(std::string(1, (char)c));
      ++i;
    } else {
      ++i;
    }
  }
}

std::vector<std::string> tokenize_core(const std::string& text) {
  std::vector<std::string> out;
  split_and_caps(pass_wrep(pass_rep(text)), out);
  return out;
}

std::vector<std::vector<std::string>> tokenize_core_batch(
    const std::vector<std::string>& texts) {
  std::vector<std::vector<std::string>> out(texts.size());
  {
    pybind11::gil_scoped_release nogil;
    for (size_t i = 0; i < texts.size(); ++i)
      split_and_caps(pass_wrep(pass_rep(texts[i])), out[i]);
  }
  return out;
}

}  // namespace ci
