#include "anthropic.h"

#include "gateway.h"

namespace gw {

// declared in proxy.cpp
void start_proxy_direct(Gateway* gw, ClientConn* c, EndpointPtr ep,
                        ApiKind kind, const std::string& upstream_path,
                        std::string_view content_type);

// ---------------------------------------------------------------------
// Request conversion (anthropic.rs:1048-1298)
// ---------------------------------------------------------------------
static Json content_blocks_to_openai_text(const Json& content) {
  // string content passes through; block arrays concatenate text blocks
  if (content.is_str()) return content;
  std::string text;
  for (auto& blk : content.as_arr()) {
    if (blk["type"].as_str() == "text") {
      if (!text.empty()) text += "\n";
      text += blk["text"].as_str();
    }
  }
  return Json(text);
}

bool anthropic_request_to_openai(const Json& in, Json* out,
                                 std::string* err) {
  if (!in.is_obj()) {
    *err = "body must be a JSON object";
    return false;
  }
  const std::string& model = in["model"].as_str();
  if (model.empty()) {
    *err = "model is required";
    return false;
  }
  if (!in.find("max_tokens") || !in["max_tokens"].is_num()) {
    *err = "max_tokens is required";
    return false;
  }
  JsonArray messages;
  // system: string or content-block array -> leading system message
  const Json* system = in.find("system");
  if (system && !system->is_null()) {
    JsonObject m;
    m.emplace_back("role", Json("system"));
    m.emplace_back("content", content_blocks_to_openai_text(*system));
    messages.emplace_back(Json(std::move(m)));
  }
  for (auto& msg : in["messages"].as_arr()) {
    const std::string& role = msg["role"].as_str();
    const Json& content = msg["content"];
    if (content.is_str()) {
      JsonObject m;
      m.emplace_back("role", Json(role));
      m.emplace_back("content", content);
      messages.emplace_back(Json(std::move(m)));
      continue;
    }
    // block array: split into text / tool_use / tool_result pieces
    std::string text;
    JsonArray tool_calls;
    std::vector<std::pair<std::string, std::string>> tool_results;
    for (auto& blk : content.as_arr()) {
      const std::string& t = blk["type"].as_str();
      if (t == "text") {
        if (!text.empty()) text += "\n";
        text += blk["text"].as_str();
      } else if (t == "tool_use") {
        JsonObject fn;
        fn.emplace_back("name", blk["name"]);
        fn.emplace_back("arguments", Json(blk["input"].dump()));
        JsonObject tc;
        tc.emplace_back("id", blk["id"]);
        tc.emplace_back("type", Json("function"));
        tc.emplace_back("function", Json(std::move(fn)));
        tool_calls.emplace_back(Json(std::move(tc)));
      } else if (t == "tool_result") {
        std::string rc = blk["content"].is_str()
                             ? blk["content"].as_str()
                             : content_blocks_to_openai_text(blk["content"])
                                   .as_str();
        tool_results.emplace_back(blk["tool_use_id"].as_str(), rc);
      }
    }
    if (!tool_results.empty()) {
      for (auto& [id, rc] : tool_results) {
        JsonObject m;
        m.emplace_back("role", Json("tool"));
        m.emplace_back("tool_call_id", Json(id));
        m.emplace_back("content", Json(rc));
        messages.emplace_back(Json(std::move(m)));
      }
    }
    if (!text.empty() || !tool_calls.empty() || tool_results.empty()) {
      JsonObject m;
      m.emplace_back("role", Json(role));
      m.emplace_back("content", Json(text));
      if (!tool_calls.empty())
        m.emplace_back("tool_calls", Json(std::move(tool_calls)));
      messages.emplace_back(Json(std::move(m)));
    }
  }
  JsonObject o;
  o.emplace_back("model", Json(model));
  o.emplace_back("messages", Json(std::move(messages)));
  o.emplace_back("max_tokens", in["max_tokens"]);
  if (in.find("temperature")) o.emplace_back("temperature",
                                             in["temperature"]);
  if (in.find("top_p")) o.emplace_back("top_p", in["top_p"]);
  if (in.find("top_k")) o.emplace_back("top_k", in["top_k"]);
  if (in.find("stream")) o.emplace_back("stream", in["stream"]);
  const Json* stops = in.find("stop_sequences");
  if (stops && stops->is_arr()) o.emplace_back("stop", *stops);
  // tools (anthropic.rs:1218-1298)
  const Json* tools = in.find("tools");
  if (tools && tools->is_arr() && !tools->as_arr().empty()) {
    JsonArray otools;
    for (auto& t : tools->as_arr()) {
      JsonObject fn;
      fn.emplace_back("name", t["name"]);
      fn.emplace_back("description", t["description"]);
      fn.emplace_back("parameters", t["input_schema"]);
      JsonObject ot;
      ot.emplace_back("type", Json("function"));
      ot.emplace_back("function", Json(std::move(fn)));
      otools.emplace_back(Json(std::move(ot)));
    }
    o.emplace_back("tools", Json(std::move(otools)));
    const Json* tc = in.find("tool_choice");
    if (tc && tc->is_obj()) {
      const std::string& ty = (*tc)["type"].as_str();
      if (ty == "auto") o.emplace_back("tool_choice", Json("auto"));
      else if (ty == "any") o.emplace_back("tool_choice", Json("required"));
      else if (ty == "tool") {
        JsonObject fn;
        fn.emplace_back("name", (*tc)["name"]);
        JsonObject choice;
        choice.emplace_back("type", Json("function"));
        choice.emplace_back("function", Json(std::move(fn)));
        o.emplace_back("tool_choice", Json(std::move(choice)));
      }
    }
  }
  // streaming chat requests get usage injected like the OpenAI path
  if (in["stream"].as_bool())
    o.emplace_back("stream_options", [] {
      JsonObject so;
      so.emplace_back("include_usage", Json(true));
      return Json(std::move(so));
    }());
  *out = Json(std::move(o));
  return true;
}

// ---------------------------------------------------------------------
// Non-streaming response conversion (anthropic.rs:1435-1533)
// ---------------------------------------------------------------------
bool anthropic_convert_response(std::string_view openai_json,
                                const std::string& public_model,
                                std::string* out) {
  Json resp;
  if (!Json::parse(openai_json, resp)) return false;
  const Json& choice0 = resp["choices"].as_arr().empty()
                            ? Json()
                            : resp["choices"].as_arr()[0];
  const Json& msg = choice0["message"];
  JsonArray content;
  const std::string& text = msg["content"].as_str();
  if (!text.empty()) {
    JsonObject blk;
    blk.emplace_back("type", Json("text"));
    blk.emplace_back("text", Json(text));
    content.emplace_back(Json(std::move(blk)));
  }
  const Json* tcs = msg.find("tool_calls");
  if (tcs && tcs->is_arr()) {
    for (auto& tc : tcs->as_arr()) {
      JsonObject blk;
      blk.emplace_back("type", Json("tool_use"));
      blk.emplace_back("id", tc["id"]);
      blk.emplace_back("name", tc["function"]["name"]);
      Json input;
      if (!Json::parse(tc["function"]["arguments"].as_str(), input))
        input = Json(JsonObject{});
      blk.emplace_back("input", std::move(input));
      content.emplace_back(Json(std::move(blk)));
    }
  }
  const std::string& fin = choice0["finish_reason"].as_str();
  std::string stop_reason = fin == "length" ? "max_tokens"
                            : fin == "tool_calls" ? "tool_use"
                                                  : "end_turn";
  JsonObject usage;
  usage.emplace_back("input_tokens",
                     Json(resp["usage"]["prompt_tokens"].as_int()));
  usage.emplace_back("output_tokens",
                     Json(resp["usage"]["completion_tokens"].as_int()));
  JsonObject o;
  std::string id = resp["id"].as_str();
  o.emplace_back("id", Json(id.empty() ? "msg_" + uuid4() : "msg_" + id));
  o.emplace_back("type", Json("message"));
  o.emplace_back("role", Json("assistant"));
  o.emplace_back("model", Json(public_model));
  o.emplace_back("content", Json(std::move(content)));
  o.emplace_back("stop_reason", Json(stop_reason));
  o.emplace_back("stop_sequence", Json(nullptr));
  o.emplace_back("usage", Json(std::move(usage)));
  *out = Json(std::move(o)).dump();
  return true;
}

// ---------------------------------------------------------------------
// Streaming re-encoder (anthropic.rs:728-1046)
// ---------------------------------------------------------------------
void AnthropicStreamEncoder::data_line(std::string_view json_body,
                                       std::string& out) {
  // one SSE event with chunked framing
  char lenbuf[16];
  size_t total = json_body.size();
  (void)total;
  std::string payload;
  payload.append(json_body);
  int n = snprintf(lenbuf, sizeof lenbuf, "%zx\r\n", payload.size());
  out.append(lenbuf, n);
  out.append(payload);
  out.append("\r\n");
}

void AnthropicStreamEncoder::event(std::string_view name, const Json& data,
                                   std::string& out) {
  std::string body = "event: ";
  body.append(name);
  body += "\ndata: ";
  data.dump(body);
  body += "\n\n";
  data_line(body, out);
}

void AnthropicStreamEncoder::ensure_message_start(std::string& out) {
  if (started_) return;
  started_ = true;
  if (msg_id_.empty()) msg_id_ = "msg_" + uuid4();
  JsonObject usage;
  usage.emplace_back("input_tokens", Json(input_tokens_));
  usage.emplace_back("output_tokens", Json(0));
  JsonObject msg;
  msg.emplace_back("id", Json(msg_id_));
  msg.emplace_back("type", Json("message"));
  msg.emplace_back("role", Json("assistant"));
  msg.emplace_back("model", Json(model_));
  msg.emplace_back("content", Json(JsonArray{}));
  msg.emplace_back("stop_reason", Json(nullptr));
  msg.emplace_back("usage", Json(std::move(usage)));
  JsonObject ev;
  ev.emplace_back("type", Json("message_start"));
  ev.emplace_back("message", Json(std::move(msg)));
  event("message_start", Json(std::move(ev)), out);
}

void AnthropicStreamEncoder::close_block(std::string& out) {
  if (block_ == Block::kNone) return;
  JsonObject ev;
  ev.emplace_back("type", Json("content_block_stop"));
  ev.emplace_back("index", Json((long long)block_index_));
  event("content_block_stop", Json(std::move(ev)), out);
  block_ = Block::kNone;
}

void AnthropicStreamEncoder::feed(std::string_view payload,
                                  std::string& out) {
  size_t start = 0;
  while (true) {
    size_t nl = payload.find('\n', start);
    if (nl == std::string_view::npos) {
      carry_.append(payload.substr(start));
      return;
    }
    std::string_view line;
    std::string tmp;
    if (!carry_.empty()) {
      carry_.append(payload.substr(start, nl - start));
      tmp.swap(carry_);
      line = tmp;
    } else {
      line = payload.substr(start, nl - start);
    }
    while (!line.empty() && line.back() == '\r') line.remove_suffix(1);
    if (starts_with(line, "data:")) {
      std::string_view body = trim(line.substr(5));
      if (body == "[DONE]") {
        // terminal events emitted in finish()
      } else if (!body.empty()) {
        handle_event(body, out);
      }
    }
    start = nl + 1;
  }
}

void AnthropicStreamEncoder::handle_event(std::string_view body,
                                          std::string& out) {
  Json chunk;
  if (!Json::parse(body, chunk)) return;
  if (msg_id_.empty()) {
    const std::string& id = chunk["id"].as_str();
    if (!id.empty()) msg_id_ = "msg_" + id;
  }
  const Json* usage = chunk.find("usage");
  if (usage && usage->is_obj()) {
    input_tokens_ = (*usage)["prompt_tokens"].as_int(input_tokens_);
    output_tokens_ = (*usage)["completion_tokens"].as_int(output_tokens_);
  }
  const JsonArray& choices = chunk["choices"].as_arr();
  if (choices.empty()) return;
  const Json& choice = choices[0];
  const Json& delta = choice["delta"];
  const std::string& fin = choice["finish_reason"].as_str();
  ensure_message_start(out);

  const std::string& text = delta["content"].as_str();
  if (!text.empty()) {
    if (block_ != Block::kText) {
      close_block(out);
      ++block_index_;
      block_ = Block::kText;
      JsonObject cb;
      cb.emplace_back("type", Json("text"));
      cb.emplace_back("text", Json(""));
      JsonObject ev;
      ev.emplace_back("type", Json("content_block_start"));
      ev.emplace_back("index", Json((long long)block_index_));
      ev.emplace_back("content_block", Json(std::move(cb)));
      event("content_block_start", Json(std::move(ev)), out);
    }
    ++text_deltas_;
    JsonObject d;
    d.emplace_back("type", Json("text_delta"));
    d.emplace_back("text", Json(text));
    JsonObject ev;
    ev.emplace_back("type", Json("content_block_delta"));
    ev.emplace_back("index", Json((long long)block_index_));
    ev.emplace_back("delta", Json(std::move(d)));
    event("content_block_delta", Json(std::move(ev)), out);
  }
  const Json* tcs = delta.find("tool_calls");
  if (tcs && tcs->is_arr()) {
    for (auto& tc : tcs->as_arr()) {
      const Json& fn = tc["function"];
      const std::string& name = fn["name"].as_str();
      if (!name.empty()) {
        close_block(out);
        ++block_index_;
        block_ = Block::kTool;
        JsonObject cb;
        cb.emplace_back("type", Json("tool_use"));
        std::string id = tc["id"].as_str();
        cb.emplace_back("id", Json(id.empty() ? "toolu_" + uuid4() : id));
        cb.emplace_back("name", Json(name));
        cb.emplace_back("input", Json(JsonObject{}));
        JsonObject ev;
        ev.emplace_back("type", Json("content_block_start"));
        ev.emplace_back("index", Json((long long)block_index_));
        ev.emplace_back("content_block", Json(std::move(cb)));
        event("content_block_start", Json(std::move(ev)), out);
      }
      const std::string& args = fn["arguments"].as_str();
      if (!args.empty() && block_ == Block::kTool) {
        JsonObject d;
        d.emplace_back("type", Json("input_json_delta"));
        d.emplace_back("partial_json", Json(args));
        JsonObject ev;
        ev.emplace_back("type", Json("content_block_delta"));
        ev.emplace_back("index", Json((long long)block_index_));
        ev.emplace_back("delta", Json(std::move(d)));
        event("content_block_delta", Json(std::move(ev)), out);
      }
    }
  }
  if (!fin.empty())
    finish_reason_ = fin;
}

void AnthropicStreamEncoder::finish(std::string& out) {
  if (finished_) return;
  finished_ = true;
  ensure_message_start(out);
  close_block(out);
  std::string stop_reason = finish_reason_ == "length" ? "max_tokens"
                            : finish_reason_ == "tool_calls"
                                ? "tool_use"
                                : "end_turn";
  {
    JsonObject d;
    d.emplace_back("stop_reason", Json(stop_reason));
    d.emplace_back("stop_sequence", Json(nullptr));
    JsonObject usage;
    usage.emplace_back("output_tokens",
                       Json(output_tokens_ ? output_tokens_
                                           : text_deltas_));
    JsonObject ev;
    ev.emplace_back("type", Json("message_delta"));
    ev.emplace_back("delta", Json(std::move(d)));
    ev.emplace_back("usage", Json(std::move(usage)));
    event("message_delta", Json(std::move(ev)), out);
  }
  {
    JsonObject ev;
    ev.emplace_back("type", Json("message_stop"));
    event("message_stop", Json(std::move(ev)), out);
  }
  out.append("0\r\n\r\n");  // terminal chunk
}

// ---------------------------------------------------------------------
// /v1/messages entry (anthropic.rs:70-727)
// ---------------------------------------------------------------------
void handle_anthropic_messages(Gateway* gw, ClientConn* c) {
  const HttpRequest& r = c->req;
  std::string_view raw =
      c->in.view().substr(r.header_len, (size_t)r.content_length);
  if (r.header("anthropic-version").empty()) {
    c->send_simple(400, "application/json",
                   "{\"type\":\"error\",\"error\":{\"type\":"
                   "\"invalid_request_error\",\"message\":"
                   "\"anthropic-version header is required\"}}");
    return;
  }
  std::string model = raw_string_field(raw, "model");
  if (starts_with(model, "anthropic:")) {
    // native cloud passthrough
    gw->start_cloud_proxy(c, ApiKind::kMessages, "anthropic",
                          model.substr(10), "/v1/messages", true);
    return;
  }
  Json in;
  std::string perr;
  if (!Json::parse(raw, in, &perr)) {
    c->send_simple(400, "application/json",
                   "{\"type\":\"error\",\"error\":{\"type\":"
                   "\"invalid_request_error\",\"message\":\"invalid "
                   "JSON\"}}");
    return;
  }
  Json converted;
  std::string err;
  if (!anthropic_request_to_openai(in, &converted, &err)) {
    std::string body = "{\"type\":\"error\",\"error\":{\"type\":"
                       "\"invalid_request_error\",\"message\":";
    Json::escape(err, body);
    body += "}}";
    c->send_simple(400, "application/json", body);
    return;
  }
  // select a chat endpoint for the model
  auto table = gw->registry.table();
  SelectResult sel = select_endpoint(*table, model, ApiKind::kMessages,
                                     gw->cfg.admission, wall_ms(), kCapChat);
  if (sel.err_status != 0) {
    std::string body = "{\"type\":\"error\",\"error\":{\"type\":"
                       "\"not_found_error\",\"message\":\"model not "
                       "available\"}}";
    c->send_simple(sel.err_status, "application/json", body);
    return;
  }
  if (sel.upstream_model != model)
    converted.set("model", Json(sel.upstream_model));
  std::string out_body = converted.dump();
  bool want_stream = in["stream"].as_bool();

  UpstreamConn* uc = c->w->acquire_upstream(sel.ep, sel.ep->id,
                                            sel.ep->host, sel.ep->port);
  if (!uc) {
    gw->metrics.upstream_errors.fetch_add(1);
    c->send_simple(502, "application/json",
                   "{\"type\":\"error\",\"error\":{\"type\":"
                   "\"api_error\",\"message\":\"endpoint unreachable\"}}");
    return;
  }
  auto op = std::make_unique<ProxyOp>();
  op->client = c;
  op->upstream = uc;
  op->start_us = now_us();
  op->public_model = model;
  op->upstream_model = sel.upstream_model;
  op->endpoint_id = sel.ep->id;
  op->kind = ApiKind::kMessages;
  op->client_ip = c->peer_ip;
  op->anthropic_reencode = true;
  if (want_stream)
    op->reenc = std::make_unique<AnthropicStreamEncoder>(model);
  op->lease = RequestLease(sel.ep, &gw->registry, sel.canonical,
                           ApiKind::kMessages);
  c->op = std::move(op);
  uc->op = c->op.get();

  char head[512];
  int n = snprintf(head, sizeof head,
                   "POST %s/v1/chat/completions HTTP/1.1\r\nHost: %s:%d\r\n"
                   "Content-Type: application/json\r\nContent-Length: "
                   "%zu\r\nAccept: */*\r\n",
                   sel.ep->path_prefix.c_str(), sel.ep->host.c_str(),
                   sel.ep->port, out_body.size());
  uc->out.append(head, n);
  if (!sel.ep->api_key.empty()) {
    uc->out.append("Authorization: Bearer ");
    uc->out.append(sel.ep->api_key);
    uc->out.append("\r\n");
  }
  uc->out.append("Connection: keep-alive\r\n\r\n");
  uc->out.append(out_body);
  gw->metrics.active_requests.fetch_add(1);
  // reuse proxy.cpp's timeout arming via a local lambda-equivalent
  extern void arm_timeout_public(Gateway* gw, ClientConn* c, int secs);
  arm_timeout_public(gw, c, sel.ep->inference_timeout_s);
  if (uc->connected) uc->send_buffered();
}

void anthropic_reencode_feed(ProxyOp* op, ClientConn* c,
                             std::string_view payload) {
  if (!op->reenc) return;
  std::string outstr;
  op->reenc->feed(payload, outstr);
  c->out.append(outstr);
  op->tap.feed(payload);  // token accounting still on the OpenAI stream
}

void anthropic_reencode_finish(ProxyOp* op, ClientConn* c) {
  if (!op->reenc) return;
  std::string outstr;
  op->reenc->finish(outstr);
  c->out.append(outstr);
  c->flush_out();
}

}  // namespace gw
