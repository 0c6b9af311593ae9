// Byte-level BPE token counter for usage estimation when a backend
// omits the `usage` field — the native analog of the reference's
// tiktoken cl100k_base fallback (llmlb/src/token/mod.rs:217-223).
//
// Consumes the standard tiktoken file format (base64(token_bytes) SP
// rank per line), so deployments can point LLMLB_TIKTOKEN_FILE at a
// real cl100k_base.tiktoken for exact parity; the in-tree default
// table (gateway/data/llmlb.tiktoken) is a 24k-vocab byte-level BPE
// trained offline on English+code text (gateway/tools/train_bpe.py).
#pragma once

#include <string>
#include <string_view>
#include <unordered_map>
#include <vector>

namespace gw {

class BpeTokenizer {
 public:
  // loads a tiktoken-format file; false on error
  bool load(const std::string& path);
  bool loaded() const { return !ranks_.empty(); }

  // GPT-2-style pre-split + greedy lowest-rank BPE merge; returns the
  // token count (the estimation use case needs no ids)
  long long count(std::string_view text) const;

  // the process-wide default table: LLMLB_TIKTOKEN_FILE, else
  // gateway/data/llmlb.tiktoken next to the executable; nullptr when
  // neither loads (callers fall back to a chars/4 heuristic)
  static const BpeTokenizer* default_instance();

  // chars/4 heuristic used when no table is available
  static long long heuristic_count(std::string_view text) {
    return (long long)(text.size() + 3) / 4;
  }

 private:
  long long count_piece(std::string_view piece) const;
  std::unordered_map<std::string, int> ranks_;
};

}  // namespace gw
