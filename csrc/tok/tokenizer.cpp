// In-repo C++ tokenizer core (replaces the HuggingFace Rust `tokenizers`
// dependency of the reference — src/tokenization.py:4,42-57,
// utils/encode_data.py:281-293, utils/build_vocab.py:39-58; Rust is not
// available in this environment).
//
// The Python layer (bert_pytorch_amd/data/tokenization.py) handles
// Unicode normalization / basic tokenization / byte-level mapping; this
// file implements the hot parts: greedy longest-match WordPiece encode,
// byte-level BPE merge loops, and a frequency-based BPE trainer.

#include <torch/extension.h>

#include <algorithm>
#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <string>
#include <unordered_map>
#include <utility>
#include <vector>

namespace bpa_tok {

struct WordPieceModel {
  std::unordered_map<std::string, int64_t> vocab;
  std::string unk = "[UNK]";
  int max_input_chars = 100;  // reference: src/tokenization.py:179
};

struct BPEModel {
  std::unordered_map<std::string, int64_t> vocab;
  // merge pair -> rank
  std::map<std::pair<std::string, std::string>, int> merges;
};

namespace {
std::mutex g_mu;
std::vector<std::unique_ptr<WordPieceModel>> g_wp;
std::vector<std::unique_ptr<BPEModel>> g_bpe;
}  // namespace

int64_t create_wordpiece(std::vector<std::string> vocab, std::string unk) {
  auto m = std::make_unique<WordPieceModel>();
  for (size_t i = 0; i < vocab.size(); ++i) m->vocab[vocab[i]] = i;
  m->unk = std::move(unk);
  std::lock_guard<std::mutex> lock(g_mu);
  g_wp.push_back(std::move(m));
  return static_cast<int64_t>(g_wp.size() - 1);
}

// greedy longest-match over pre-tokenized words; '##' continuation
std::pair<std::vector<std::string>, std::vector<int64_t>> encode_wordpiece(
    int64_t handle, std::vector<std::string> words) {
  const WordPieceModel& m = *g_wp.at(handle);
  std::vector<std::string> tokens;
  std::vector<int64_t> ids;
  const auto unk_it = m.vocab.find(m.unk);
  const int64_t unk_id = unk_it == m.vocab.end() ? 0 : unk_it->second;
  for (const auto& word : words) {
    if (static_cast<int>(word.size()) > m.max_input_chars) {
      tokens.push_back(m.unk);
      ids.push_back(unk_id);
      continue;
    }
    size_t start = 0;
    std::vector<std::string> sub;
    bool bad = false;
    while (start < word.size()) {
      size_t end = word.size();
      std::string cur;
      bool found = false;
      while (end > start) {
        std::string piece = word.substr(start, end - start);
        if (start > 0) piece = "##" + piece;
        if (m.vocab.count(piece)) {
          cur = std::move(piece);
          found = true;
          break;
        }
        // back off one UTF-8 codepoint, not one byte
        do {
          --end;
        } while (end > start && (static_cast<uint8_t>(word[end]) & 0xC0) == 0x80);
      }
      if (!found) {
        bad = true;
        break;
      }
      sub.push_back(std::move(cur));
      start = end;
    }
    if (bad) {
      tokens.push_back(m.unk);
      ids.push_back(unk_id);
    } else {
      for (auto& s : sub) {
        ids.push_back(m.vocab.at(s));
        tokens.push_back(std::move(s));
      }
    }
  }
  return {tokens, ids};
}

int64_t create_bpe(std::vector<std::string> vocab,
                   std::vector<std::string> merge_lines) {
  auto m = std::make_unique<BPEModel>();
  for (size_t i = 0; i < vocab.size(); ++i) m->vocab[vocab[i]] = i;
  int rank = 0;
  for (const auto& line : merge_lines) {
    auto sp = line.find(' ');
    if (sp == std::string::npos) continue;
    m->merges[{line.substr(0, sp), line.substr(sp + 1)}] = rank++;
  }
  std::lock_guard<std::mutex> lock(g_mu);
  g_bpe.push_back(std::move(m));
  return static_cast<int64_t>(g_bpe.size() - 1);
}

// split a byte-mapped string into UTF-8 codepoints
static std::vector<std::string> codepoints(const std::string& s) {
  std::vector<std::string> out;
  for (size_t i = 0; i < s.size();) {
    size_t len = 1;
    const uint8_t c = static_cast<uint8_t>(s[i]);
    if ((c & 0xF8) == 0xF0) len = 4;
    else if ((c & 0xF0) == 0xE0) len = 3;
    else if ((c & 0xE0) == 0xC0) len = 2;
    out.push_back(s.substr(i, len));
    i += len;
  }
  return out;
}

// classic BPE merge loop for one pre-token (byte-repr string)
static std::vector<std::string> bpe_word(const BPEModel& m,
                                         const std::string& word) {
  std::vector<std::string> parts = codepoints(word);
  if (parts.size() < 2) return parts;
  while (true) {
    int best_rank = INT32_MAX;
    size_t best_i = 0;
    for (size_t i = 0; i + 1 < parts.size(); ++i) {
      auto it = m.merges.find({parts[i], parts[i + 1]});
      if (it != m.merges.end() && it->second < best_rank) {
        best_rank = it->second;
        best_i = i;
      }
    }
    if (best_rank == INT32_MAX) break;
    parts[best_i] = parts[best_i] + parts[best_i + 1];
    parts.erase(parts.begin() + best_i + 1);
    if (parts.size() < 2) break;
  }
  return parts;
}

std::pair<std::vector<std::string>, std::vector<int64_t>> encode_bpe(
    int64_t handle, std::vector<std::string> pretokens) {
  const BPEModel& m = *g_bpe.at(handle);
  std::vector<std::string> tokens;
  std::vector<int64_t> ids;
  for (const auto& pt : pretokens) {
    for (auto& piece : bpe_word(m, pt)) {
      auto it = m.vocab.find(piece);
      ids.push_back(it == m.vocab.end() ? -1 : it->second);
      tokens.push_back(std::move(piece));
    }
  }
  return {tokens, ids};
}

// frequency-based BPE trainer: words + counts -> ordered merge list.
// (Used for both byte-level BPE and — with a '##'-style post-pass in
// Python — WordPiece-shaped vocabs; simpler than HF's likelihood
// trainer but produces a functional vocab, utils/build_vocab.py.)
std::vector<std::string> train_bpe(std::vector<std::string> words,
                                   std::vector<int64_t> counts,
                                   int64_t num_merges) {
  std::vector<std::vector<std::string>> seqs(words.size());
  for (size_t i = 0; i < words.size(); ++i) seqs[i] = codepoints(words[i]);
  std::vector<std::string> merge_lines;
  for (int64_t step = 0; step < num_merges; ++step) {
    std::map<std::pair<std::string, std::string>, int64_t> pair_counts;
    for (size_t w = 0; w < seqs.size(); ++w) {
      for (size_t i = 0; i + 1 < seqs[w].size(); ++i)
        pair_counts[{seqs[w][i], seqs[w][i + 1]}] += counts[w];
    }
    if (pair_counts.empty()) break;
    auto best = std::max_element(
        pair_counts.begin(), pair_counts.end(),
        [](const auto& a, const auto& b) { return a.second < b.second; });
    if (best->second < 2) break;
    const auto [l, r] = best->first;
    merge_lines.push_back(l + " " + r);
    for (auto& seq : seqs) {
      for (size_t i = 0; i + 1 < seq.size();) {
        if (seq[i] == l && seq[i + 1] == r) {
          seq[i] = l + r;
          seq.erase(seq.begin() + i + 1);
        } else {
          ++i;
        }
      }
    }
  }
  return merge_lines;
}

}  // namespace bpa_tok
