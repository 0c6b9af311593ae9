// Anthropic-native /v1/messages surface: Anthropic<->OpenAI request and
// response conversion, and the stateful OpenAI-SSE -> Anthropic-SSE
// stream re-encoder (reference behavior: api/anthropic.rs:70-1533).
#pragma once

#include <string>
#include <string_view>

#include "json.h"

namespace gw {

class Gateway;
class ClientConn;
struct ProxyOp;

// entry: POST /v1/messages
void handle_anthropic_messages(Gateway* gw, ClientConn* c);

// Anthropic request JSON -> OpenAI chat/completions request JSON.
// Returns false + err on malformed input.
bool anthropic_request_to_openai(const Json& in, Json* out,
                                 std::string* err);

// OpenAI chat response (raw JSON) -> Anthropic message response.
bool anthropic_convert_response(std::string_view openai_json,
                                const std::string& public_model,
                                std::string* out);

// streaming re-encoder: fed decoded OpenAI SSE payload, emits Anthropic
// SSE events with chunked framing into the client out buffer
class AnthropicStreamEncoder {
 public:
  explicit AnthropicStreamEncoder(std::string model)
      : model_(std::move(model)) {}
  // feed decoded SSE payload bytes; appends chunked output to `out`
  void feed(std::string_view payload, std::string& out);
  void finish(std::string& out);

 private:
  void event(std::string_view name, const Json& data, std::string& out);
  void data_line(std::string_view json_body, std::string& out);
  void handle_event(std::string_view body, std::string& out);
  void ensure_message_start(std::string& out);
  void close_block(std::string& out);

  std::string model_;
  std::string carry_;
  std::string msg_id_;
  bool started_ = false;
  bool finished_ = false;
  int block_index_ = -1;
  enum class Block { kNone, kText, kTool } block_ = Block::kNone;
  std::string finish_reason_;
  long long input_tokens_ = 0, output_tokens_ = 0;
  long long text_deltas_ = 0;
};

// helpers used by proxy.cpp
void anthropic_reencode_feed(ProxyOp* op, ClientConn* c,
                             std::string_view payload);
void anthropic_reencode_finish(ProxyOp* op, ClientConn* c);

}  // namespace gw
