// Native unigram-tokenizer core: trie + Viterbi encode over
// sentencepiece-style pieces, batch API with the GIL released. The Python
// layer (mi355/tokenize/) owns the model file parsing and the T5Tokenizer
// surface; this core is the hot loop (BatchMapper tokenization of 4096-row
// batches — reference Model_finetuning_and_batch_inference.ipynb:cell-27).
#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cmath>
#include <cstdint>
#include <limits>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

namespace {

struct TrieNode {
  std::unordered_map<uint8_t, int32_t> next;  // byte -> node index
  int32_t piece_id = -1;
  float score = 0.f;
};

struct UnigramCore {
  std::vector<TrieNode> trie{1};
  int32_t unk_id = 2;
  float unk_score = -20.f;
  int max_len = 1;

  void add_piece(const std::string& bytes, int32_t id, float score) {
    int node = 0;
    for (unsigned char c : bytes) {
      auto it = trie[node].next.find(c);
      if (it == trie[node].next.end()) {
        trie[node].next.emplace(c, (int32_t)trie.size());
        node = (int)trie.size();
        trie.emplace_back();
      } else {
        node = it->second;
      }
    }
    trie[node].piece_id = id;
    trie[node].score = score;
    max_len = std::max(max_len, (int)bytes.size());
  }

  // Viterbi over the normalized byte string (UTF-8; unk fallback advances
  // one full UTF-8 codepoint, matching the Python reference per-char step).
  std::vector<int32_t> encode(const std::string& text) const {
    const int n = (int)text.size();
    constexpr float NEG = -std::numeric_limits<float>::infinity();
    std::vector<float> best(n + 1, NEG);
    std::vector<int32_t> prev(n + 1, -1), pid(n + 1, -1);
    best[0] = 0.f;
    for (int i = 0; i < n; ++i) {
      if (best[i] == NEG) continue;
      // walk the trie from position i
      int node = 0;
      for (int j = i; j < n; ++j) {
        auto it = trie[node].next.find((uint8_t)text[j]);
        if (it == trie[node].next.end()) break;
        node = it->second;
        if (trie[node].piece_id >= 0) {
          float sc = best[i] + trie[node].score;
          if (sc > best[j + 1]) {
            best[j + 1] = sc;
            prev[j + 1] = i;
            pid[j + 1] = trie[node].piece_id;
          }
        }
      }
      // unk fallback: one UTF-8 codepoint
      int adv = 1;
      unsigned char c = text[i];
      if ((c & 0xE0) == 0xC0) adv = 2;
      else if ((c & 0xF0) == 0xE0) adv = 3;
      else if ((c & 0xF8) == 0xF0) adv = 4;
      int j = std::min(n, i + adv);
      float sc = best[i] + unk_score;
      if (sc > best[j]) {
        best[j] = sc;
        prev[j] = i;
        pid[j] = unk_id;
      }
    }
    std::vector<int32_t> out;
    for (int j = n; j > 0; j = prev[j]) out.push_back(pid[j]);
    std::reverse(out.begin(), out.end());
    return out;
  }
};

}  // namespace

PYBIND11_MODULE(_tokenize, m) {
  py::class_<UnigramCore>(m, "UnigramCore")
      .def(py::init<>())
      .def_readwrite("unk_id", &UnigramCore::unk_id)
      .def_readwrite("unk_score", &UnigramCore::unk_score)
      .def("add_piece", &UnigramCore::add_piece)
      .def("encode",
           [](const UnigramCore& c, const std::string& s) {
             std::vector<int32_t> r;
             {
               py::gil_scoped_release rel;
               r = c.encode(s);
             }
             return r;
           })
      .def("encode_batch", [](const UnigramCore& c, std::vector<std::string> texts) {
        std::vector<std::vector<int32_t>> out(texts.size());
        {
          py::gil_scoped_release rel;
          for (size_t i = 0; i < texts.size(); ++i) out[i] = c.encode(texts[i]);
        }
        return out;
      });
}
