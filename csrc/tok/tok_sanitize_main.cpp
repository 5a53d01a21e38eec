// Sanitizer driver for the C++ tokenizer core (SURVEY.md §5
// "sanitizers"): exercises WordPiece encode, byte-BPE encode and the
// BPE trainer under ASan+UBSan with adversarial inputs — empty
// strings, over-long words, invalid UTF-8 bytes, degenerate vocabs,
// merge rules referencing unknown symbols. Built and run on CPU by
// scripts/sanitize_check.sh (no GPU, no torch linkage: the tokenizer
// core uses only the C++ standard library).

#include <cstdint>
#include <cstdio>
#include <string>
#include <utility>
#include <vector>

namespace bpa_tok {
int64_t create_wordpiece(std::vector<std::string> vocab, std::string unk);
std::pair<std::vector<std::string>, std::vector<int64_t>> encode_wordpiece(
    int64_t handle, std::vector<std::string> words);
int64_t create_bpe(std::vector<std::string> vocab,
                   std::vector<std::string> merge_lines);
std::pair<std::vector<std::string>, std::vector<int64_t>> encode_bpe(
    int64_t handle, std::vector<std::string> pretokens);
std::vector<std::string> train_bpe(std::vector<std::string> words,
                                   std::vector<int64_t> counts,
                                   int64_t num_merges);
}  // namespace bpa_tok

int main() {
  using namespace bpa_tok;

  // --- WordPiece ---
  std::vector<std::string> vocab = {"[PAD]", "[UNK]", "[CLS]", "[SEP]",
                                    "the",   "th",    "##e",   "##ere",
                                    "a",     "##b",   "##c"};
  const int64_t wp = create_wordpiece(vocab, "[UNK]");
  std::vector<std::string> words = {
      "the",
      "there",
      "abc",
      "",                              // empty word
      std::string(500, 'x'),           // > max_input_chars
      "\xff\xfe\x80",                  // invalid UTF-8 bytes
      std::string("nul\0byte", 8),     // embedded NUL
      "\xe4\xbd\xa0\xe5\xa5\xbd",      // CJK multibyte
  };
  auto r1 = encode_wordpiece(wp, words);
  if (r1.first.size() != r1.second.size()) {
    std::fprintf(stderr, "wordpiece size mismatch\n");
    return 1;
  }

  // --- byte-BPE encode ---
  std::vector<std::string> bvocab;
  for (int i = 0; i < 256; ++i) bvocab.push_back(std::string(1, char(i)));
  bvocab.push_back("ab");
  bvocab.push_back("abc");
  std::vector<std::string> merges = {"a b", "ab c", "zz zz",
                                     "missing tokens", "", "a"};
  const int64_t bp = create_bpe(bvocab, merges);
  auto r2 = encode_bpe(
      bp, {"abcabc", "", std::string(1, '\0'), std::string(300, 'a'),
           "\xf0\x9f\x98\x80"});
  if (r2.first.size() != r2.second.size()) {
    std::fprintf(stderr, "bpe size mismatch\n");
    return 1;
  }

  // --- BPE trainer ---
  auto learned = train_bpe({"low", "lower", "lowest", "newest", "wide", ""},
                           {5, 2, 7, 9, 3, 1}, 32);
  // over-ask merges on a tiny corpus: must terminate without OOB
  auto learned2 = train_bpe({"aa"}, {1}, 1000);
  std::printf("SANITIZE OK wp=%zu bpe=%zu merges=%zu/%zu\n", r1.first.size(),
              r2.first.size(), learned.size(), learned2.size());
  return 0;
}
