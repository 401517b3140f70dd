// Inline WPM encoder shared by the tokenizer binding and the native
// batcher (see wpm_tokenizer.cpp for semantics/doc).
#pragma once

#include <torch/extension.h>

#include <algorithm>
#include <atomic>
#include <cstdint>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

namespace lingvo_amd {

constexpr const char* kWordMark = "\xe2\x96\x81";  // '▁'

class WpmEncoder {
 public:
  WpmEncoder(std::vector<std::string> pieces, int64_t unk_id)
      : unk_id_(unk_id) {
    pieces_ = std::move(pieces);
    max_len_ = 1;
    for (size_t i = 0; i < pieces_.size(); ++i) {
      vocab_.emplace(pieces_[i], static_cast<int64_t>(i));
      max_len_ = std::max(max_len_, pieces_[i].size());
    }
  }

  size_t VocabSize() const { return pieces_.size(); }

  // Greedy longest-match over one marked word (already '▁'-prefixed).
  void EncodePiece(const std::string& s, std::vector<int64_t>* out) const {
    size_t i = 0;
    const size_t n = s.size();
    while (i < n) {
      size_t j = std::min(n, i + max_len_);
      bool hit = false;
      for (; j > i; --j) {
        auto it = vocab_.find(s.substr(i, j - i));
        if (it != vocab_.end()) {
          out->push_back(it->second);
          hit = true;
          break;
        }
      }
      if (!hit) {
        out->push_back(unk_id_);
        // advance one UTF-8 codepoint: skip continuation bytes (10xxxxxx)
        j = i + 1;
        while (j < n && (static_cast<uint8_t>(s[j]) & 0xC0) == 0x80) ++j;
      }
      i = j;
    }
  }

  std::vector<int64_t> Encode(const std::string& text) const {
    std::vector<int64_t> out;
    size_t i = 0;
    const size_t n = text.size();
    while (i < n) {
      while (i < n && IsSpace(text[i])) ++i;
      size_t w = i;
      while (w < n && !IsSpace(text[w])) ++w;
      if (w > i) EncodePiece(kWordMark + text.substr(i, w - i), &out);
      i = w;
    }
    return out;
  }

  std::vector<std::vector<int64_t>> EncodeBatch(
      const std::vector<std::string>& lines, int num_threads) const {
    std::vector<std::vector<int64_t>> out(lines.size());
    pybind11::gil_scoped_release release;
    if (num_threads <= 1 || lines.size() < 2) {
      for (size_t i = 0; i < lines.size(); ++i) out[i] = Encode(lines[i]);
      return out;
    }
    const int nt = std::min<int>(num_threads, lines.size());
    std::vector<std::thread> threads;
    std::atomic<size_t> next{0};
    for (int t = 0; t < nt; ++t) {
      threads.emplace_back([&] {
        for (size_t i = next.fetch_add(1); i < lines.size();
             i = next.fetch_add(1)) {
          out[i] = Encode(lines[i]);
        }
      });
    }
    for (auto& th : threads) th.join();
    return out;
  }

  std::string Decode(const std::vector<int64_t>& ids) const {
    std::string s;
    for (int64_t id : ids) {
      if (id >= 0 && static_cast<size_t>(id) < pieces_.size()) {
        s += pieces_[id];
      } else {
        s += '?';
      }
    }
    // '▁' -> ' ', then strip
    std::string t;
    size_t i = 0;
    while (i < s.size()) {
      if (s.compare(i, 3, kWordMark) == 0) {
        t += ' ';
        i += 3;
      } else {
        t += s[i++];
      }
    }
    size_t b = t.find_first_not_of(' ');
    size_t e = t.find_last_not_of(' ');
    return b == std::string::npos ? std::string() : t.substr(b, e - b + 1);
  }

 private:
  static bool IsSpace(char c) {
    return c == ' ' || c == '\t' || c == '\n' || c == '\r' || c == '\f' ||
           c == '\v';
  }

  std::vector<std::string> pieces_;
  std::unordered_map<std::string, int64_t> vocab_;
  size_t max_len_;
  int64_t unk_id_;
};

}  // namespace lingvo_amd
