// Byte-level JSON grammar FSM for constrained decoding (CPU, C++).
//
// the reference repairs invalid LLM JSON after the fact
// (pkg/utils/json.go:16-190 CleanJSON/ExtractField); this FSM makes invalid
// JSON impossible instead: each decode step it computes an allowed-token
// bitmask over the vocabulary which the GPU sampling kernel
// (sampling.hip: oa_masked_argmax) fuses into the logits argmax.
//
// Modes:
//   0 = generic JSON value (top level must be an object)
//   1 = ToolPrompt schema (ref pkg/tools/tool.go:29-38): fixed key sequence
//       {"question": s, "thought": s, "action": {"name": s, "input": s},
//        "observation": s, "final_answer": s} — string values only
//   2 = tool_calls wire schema: {"tool_calls": [{"name": s, "arguments": o}]}
//
// Schema modes are template-driven: LIT (forced bytes) | STRVAL (JSON string
// content) | JSONVAL (embedded generic JSON value). The mask for each token
// is computed by simulating the token's bytes on a copy of the machine state.
//
// Exposed as a C API, built into libopsagent_grammar.so (plain g++, no deps),
// wrapped by opsagent_amd/engine/grammar.py.

#include <cstdint>
#include <cstring>
#include <map>
#include <mutex>
#include <string>
#include <unordered_map>
#include <utility>
#include <vector>

namespace {

enum State : uint8_t {
    S_VALUE,        // expect start of a JSON value
    S_STR,          // inside string content
    S_STR_ESC,      // after backslash
    S_STR_U0, S_STR_U1, S_STR_U2, S_STR_U3,  // \uXXXX hex digits
    S_NUM_MINUS,    // after '-'
    S_NUM_ZERO,     // after leading 0
    S_NUM_INT,      // integer digits
    S_NUM_DOT,      // after '.'
    S_NUM_FRAC,     // fraction digits
    S_NUM_E,        // after e/E
    S_NUM_ESIGN,    // after exponent sign
    S_NUM_EXP,      // exponent digits
    S_LIT,          // inside true/false/null
    S_AFTER_VALUE,  // value complete; delimiters next
    S_OBJ_FIRST,    // after '{': '"' or '}'
    S_OBJ_KEY,      // after ',' in object: '"'
    S_OBJ_COLON,    // after key string: ':'
    S_DONE,         // machine complete
};

enum Ctx : uint8_t { CTX_OBJ, CTX_ARR };

enum TplKind : uint8_t { T_LIT, T_STRVAL, T_JSONVAL, T_NAMES };

struct TplItem {
    TplKind kind;
    std::string lit;
};

constexpr int MAX_DEPTH = 64;

struct MachineState {
    uint8_t state = S_VALUE;
    uint8_t stack[MAX_DEPTH];
    int depth = 0;
    // string sub-state
    bool in_key = false;
    // literal progress
    const char* lit = nullptr;
    uint8_t lit_pos = 0;
    // template progress (schema modes); -1 = generic json top-level
    int tpl_idx = 0;
    int tpl_lit_pos = 0;
    // T_NAMES (constrained tool-name alternation): bitmask of still-matching
    // options and the byte position inside them
    uint32_t alt_alive = 0;
    int alt_pos = 0;
    bool in_jsonval = false;  // inside an embedded JSONVAL
    bool arr_fresh = false;   // directly after '[' (allows the empty array ']')
};

inline bool is_ws(uint8_t c) { return c == ' ' || c == '\t' || c == '\n' || c == '\r'; }
inline bool is_digit(uint8_t c) { return c >= '0' && c <= '9'; }
inline bool is_hex(uint8_t c) {
    return is_digit(c) || (c >= 'a' && c <= 'f') || (c >= 'A' && c <= 'F');
}
// raw string-content byte (RFC 8259: anything except '"', '\', control)
inline bool is_str_byte(uint8_t c) { return c >= 0x20 && c != '"' && c != '\\'; }

class Grammar {
public:
    Grammar(int mode) : mode_(mode) {
        // NOTE: STRVAL consumes its closing '"', so following literals start
        // after the quote.
        if (mode == 1) {
            tpl_ = {
                {T_LIT, "{\"question\": \""}, {T_STRVAL, ""},
                {T_LIT, ", \"thought\": \""}, {T_STRVAL, ""},
                {T_LIT, ", \"action\": {\"name\": \""}, {T_STRVAL, ""},
                {T_LIT, ", \"input\": \""}, {T_STRVAL, ""},
                {T_LIT, "}, \"observation\": \""}, {T_STRVAL, ""},
                {T_LIT, ", \"final_answer\": \""}, {T_STRVAL, ""},
                {T_LIT, "}"},
            };
        } else if (mode == 2) {
            tpl_ = {
                {T_LIT, "{\"tool_calls\": [{\"name\": \""}, {T_STRVAL, ""},
                {T_LIT, ", \"arguments\": "}, {T_JSONVAL, ""},
                {T_LIT, "}]}"},
            };
        }
        reset();
    }

    // names non-empty: the template's name field becomes a T_NAMES item that
    // only accepts one of the declared tool names (tool_choice / declared
    // tools constrain the call STRUCTURALLY, not just the JSON shape)
    void set_names(std::vector<std::string> names) {
        names_ = std::move(names);
        if (!names_.empty()) {
            // the STRVAL immediately after a literal ending in `"name": "`
            // is the tool-name field (holds for both schema modes)
            static const std::string marker = "\"name\": \"";
            for (size_t i = 0; i + 1 < tpl_.size(); ++i) {
                const std::string& lit = tpl_[i].lit;
                if (tpl_[i].kind == T_LIT && lit.size() >= marker.size() &&
                    lit.compare(lit.size() - marker.size(), marker.size(),
                                marker) == 0 &&
                    tpl_[i + 1].kind == T_STRVAL) {
                    tpl_[i + 1].kind = T_NAMES;
                    break;
                }
            }
        }
        reset();
    }

    void reset() {
        st_ = MachineState{};
        if (mode_ == 0) {
            st_.state = S_VALUE;
            st_.tpl_idx = -1;
        } else {
            st_.state = S_DONE;  // placeholder; template drives
            st_.tpl_idx = 0;
            st_.tpl_lit_pos = 0;
            enter_tpl_item(st_);
        }
    }

    // advance the machine by one byte; returns false if byte not allowed
    static bool step(MachineState& s, uint8_t c, const Grammar& g) {
        if (s.tpl_idx >= 0 && !s.in_jsonval) {
            // template-driven
            if (s.tpl_idx >= (int)g.tpl_.size()) return false;  // complete: no bytes
            const TplItem& it = g.tpl_[s.tpl_idx];
            if (it.kind == T_LIT) {
                if (c != (uint8_t)it.lit[s.tpl_lit_pos]) return false;
                if (++s.tpl_lit_pos == (int)it.lit.size()) advance_tpl(s, g);
                return true;
            }
            if (it.kind == T_NAMES) {
                if (c == '"') {
                    for (int i = 0; i < (int)g.names_.size(); ++i)
                        if ((s.alt_alive >> i) & 1u &&
                            (int)g.names_[i].size() == s.alt_pos) {
                            advance_tpl(s, g);
                            return true;
                        }
                    return false;
                }
                uint32_t next = 0;
                for (int i = 0; i < (int)g.names_.size(); ++i)
                    if ((s.alt_alive >> i) & 1u &&
                        s.alt_pos < (int)g.names_[i].size() &&
                        (uint8_t)g.names_[i][s.alt_pos] == c)
                        next |= 1u << i;
                if (!next) return false;
                s.alt_alive = next;
                s.alt_pos++;
                return true;
            }
            if (it.kind == T_STRVAL) {
                switch (s.state) {
                    case S_STR:
                        if (c == '"') { advance_tpl(s, g); return true; }
                        if (c == '\\') { s.state = S_STR_ESC; return true; }
                        return is_str_byte(c);
                    case S_STR_ESC:
                        if (c == 'u') { s.state = S_STR_U0; return true; }
                        if (strchr("\"\\/bfnrt", c)) { s.state = S_STR; return true; }
                        return false;
                    case S_STR_U0: case S_STR_U1: case S_STR_U2:
                        if (!is_hex(c)) return false;
                        s.state = (State)(s.state + 1);
                        return true;
                    case S_STR_U3:
                        if (!is_hex(c)) return false;
                        s.state = S_STR;
                        return true;
                    default:
                        return false;
                }
            }
            // T_JSONVAL handled via in_jsonval flag set at entry
            return false;
        }
        return json_step(s, c, g);
    }

    static void enter_tpl_item(MachineState& s) {
        // set sub-state for the current template item (caller set tpl_idx)
        s.tpl_lit_pos = 0;
        s.state = S_STR;       // for STRVAL
        s.in_jsonval = false;
        s.alt_alive = 0xffffffffu;  // T_NAMES: all options alive
        s.alt_pos = 0;
    }

    static void advance_tpl(MachineState& s, const Grammar& g) {
        s.tpl_idx++;
        enter_tpl_item(s);
        if (s.tpl_idx < (int)g.tpl_.size() && g.tpl_[s.tpl_idx].kind == T_JSONVAL) {
            s.in_jsonval = true;
            s.state = S_VALUE;
            s.depth = 0;
        }
    }

    // generic JSON machine; also used for embedded JSONVAL
    static bool json_step(MachineState& s, uint8_t c, const Grammar& g) {
        switch (s.state) {
            case S_VALUE:
                if (is_ws(c)) return true;
                if (c == '{') {
                    if (s.depth >= MAX_DEPTH) return false;
                    s.stack[s.depth++] = CTX_OBJ;
                    s.state = S_OBJ_FIRST;
                    return true;
                }
                if (c == '[') {
                    if (s.depth >= MAX_DEPTH) return false;
                    // top-level of mode 0 must be an object
                    if (s.tpl_idx < 0 && s.depth == 0 && g.mode_ == 0) return false;
                    s.stack[s.depth++] = CTX_ARR;
                    s.state = S_VALUE;
                    s.arr_fresh = true;  // ']' allowed only for the empty array
                    return true;
                }
                if (s.depth == 0 && s.tpl_idx < 0 && g.mode_ == 0) return false;  // non-object top
                if (c == '"') { s.state = S_STR; s.in_key = false; return true; }
                if (c == '-') { s.state = S_NUM_MINUS; return true; }
                if (c == '0') { s.state = S_NUM_ZERO; return true; }
                if (is_digit(c)) { s.state = S_NUM_INT; return true; }
                if (c == 't') { s.state = S_LIT; s.lit = "true"; s.lit_pos = 1; return true; }
                if (c == 'f') { s.state = S_LIT; s.lit = "false"; s.lit_pos = 1; return true; }
                if (c == 'n') { s.state = S_LIT; s.lit = "null"; s.lit_pos = 1; return true; }
                // empty array: ']' only directly after '[' (no trailing commas)
                if (c == ']' && s.arr_fresh && s.depth > 0 &&
                    s.stack[s.depth - 1] == CTX_ARR) {
                    s.depth--;
                    return value_done(s, g);
                }
                return false;
            case S_STR:
                if (c == '"') {
                    if (s.in_key) { s.state = S_OBJ_COLON; return true; }
                    return value_done(s, g);
                }
                if (c == '\\') { s.state = S_STR_ESC; return true; }
                return is_str_byte(c);
            case S_STR_ESC:
                if (c == 'u') { s.state = S_STR_U0; return true; }
                if (strchr("\"\\/bfnrt", c)) { s.state = S_STR; return true; }
                return false;
            case S_STR_U0: case S_STR_U1: case S_STR_U2:
                if (!is_hex(c)) return false;
                s.state = (State)(s.state + 1);
                return true;
            case S_STR_U3:
                if (!is_hex(c)) return false;
                s.state = S_STR;
                return true;
            case S_NUM_MINUS:
                if (c == '0') { s.state = S_NUM_ZERO; return true; }
                if (is_digit(c)) { s.state = S_NUM_INT; return true; }
                return false;
            case S_NUM_ZERO:
                if (c == '.') { s.state = S_NUM_DOT; return true; }
                if (c == 'e' || c == 'E') { s.state = S_NUM_E; return true; }
                return end_number_then(s, c, g);
            case S_NUM_INT:
                if (is_digit(c)) return true;
                if (c == '.') { s.state = S_NUM_DOT; return true; }
                if (c == 'e' || c == 'E') { s.state = S_NUM_E; return true; }
                return end_number_then(s, c, g);
            case S_NUM_DOT:
                if (is_digit(c)) { s.state = S_NUM_FRAC; return true; }
                return false;
            case S_NUM_FRAC:
                if (is_digit(c)) return true;
                if (c == 'e' || c == 'E') { s.state = S_NUM_E; return true; }
                return end_number_then(s, c, g);
            case S_NUM_E:
                if (c == '+' || c == '-') { s.state = S_NUM_ESIGN; return true; }
                if (is_digit(c)) { s.state = S_NUM_EXP; return true; }
                return false;
            case S_NUM_ESIGN:
                if (is_digit(c)) { s.state = S_NUM_EXP; return true; }
                return false;
            case S_NUM_EXP:
                if (is_digit(c)) return true;
                return end_number_then(s, c, g);
            case S_LIT:
                if (c != (uint8_t)s.lit[s.lit_pos]) return false;
                if (s.lit[++s.lit_pos] == '\0') return value_done(s, g);
                return true;
            case S_AFTER_VALUE: {
                if (is_ws(c)) return true;
                if (s.depth == 0) return false;  // only ws after a complete doc
                uint8_t top = s.stack[s.depth - 1];
                if (top == CTX_OBJ) {
                    if (c == ',') { s.state = S_OBJ_KEY; return true; }
                    if (c == '}') { s.depth--; return value_done(s, g); }
                    return false;
                }
                if (c == ',') { s.state = S_VALUE; s.arr_fresh = false; return true; }
                if (c == ']') { s.depth--; return value_done(s, g); }
                return false;
            }
            case S_OBJ_FIRST:
                if (is_ws(c)) return true;
                if (c == '}') { s.depth--; return value_done(s, g); }
                if (c == '"') { s.state = S_STR; s.in_key = true; return true; }
                return false;
            case S_OBJ_KEY:
                if (is_ws(c)) return true;
                if (c == '"') { s.state = S_STR; s.in_key = true; return true; }
                return false;
            case S_OBJ_COLON:
                if (is_ws(c)) return true;
                if (c == ':') { s.state = S_VALUE; return true; }
                return false;
            case S_DONE:
                return false;
        }
        return false;
    }

    static bool value_done(MachineState& s, const Grammar& g) {
        if (s.depth == 0) {
            if (s.in_jsonval) {
                // embedded JSONVAL complete -> leave json mode, advance template
                s.in_jsonval = false;
                s.tpl_idx++;
                enter_tpl_item(s);
                return true;
            }
            s.state = S_AFTER_VALUE;  // complete doc: only ws / EOS next
            return true;
        }
        s.state = S_AFTER_VALUE;
        return true;
    }

    static bool end_number_then(MachineState& s, uint8_t c, const Grammar& g) {
        // the number is complete; re-dispatch c as a delimiter
        if (!value_done(s, g)) return false;
        return step(s, c, g);
    }

    bool is_complete(const MachineState& s) const {
        if (s.tpl_idx >= 0 && !s.in_jsonval) return s.tpl_idx >= (int)tpl_.size();
        return s.state == S_AFTER_VALUE && s.depth == 0 && !s.in_jsonval;
    }

    bool accept_byte(uint8_t c) { return step(st_, c, *this); }

    MachineState st_;
    int mode_;
    std::vector<TplItem> tpl_;
    std::vector<std::string> names_;
};

struct Vocab {
    std::vector<std::pair<const uint8_t*, int>> tokens;  // ptr,len per id
    std::vector<uint8_t> token_store;
    std::vector<int> realizable;  // token ids with len > 0 (mask loop skip)
    int vocab;
    int eos_id;
};

// Allowed-token mask cache. The mask is a pure function of (vocab, grammar
// template, machine state); realistic vocabularies (32k-128k BPE tokens)
// make the per-step simulate-every-token scan ~milliseconds on the host, but
// generation revisits a small set of state classes (string bodies, template
// positions), so masks are computed once per class and memcpy'd after.
struct MaskCache {
    std::mutex mu;
    std::unordered_map<uint64_t, std::vector<uint32_t>> m;
};

// MachineState signature. Safe to hash the `lit` pointer: it only ever
// points at the static "true"/"false"/"null" literals (process-stable).
static uint64_t state_sig(const MachineState& s) {
    uint64_t h = 1469598103934665603ull;
    auto mix = [&](uint64_t v) { h ^= v; h *= 1099511628211ull; };
    mix(s.state);
    mix((uint64_t)s.depth);
    for (int i = 0; i < s.depth; ++i) mix(s.stack[i]);
    mix(s.in_key);
    mix((uint64_t)(uintptr_t)s.lit);
    mix(s.lit_pos);
    mix((uint64_t)(int64_t)s.tpl_idx);
    mix((uint64_t)s.tpl_lit_pos);
    mix(s.alt_alive);
    mix((uint64_t)s.alt_pos);
    mix(s.in_jsonval);
    mix(s.arr_fresh);
    return h;
}

// grammar identity for cache sharing across requests: mode + tool names
static uint64_t grammar_sig(int mode, const std::vector<std::string>& names) {
    uint64_t h = 1469598103934665603ull;
    auto mixb = [&](uint8_t v) { h ^= v; h *= 1099511628211ull; };
    mixb((uint8_t)mode);
    for (const auto& n : names) {
        mixb(0xff);
        for (char c : n) mixb((uint8_t)c);
    }
    return h;
}

static std::mutex g_cache_mu;
static std::map<std::pair<const Vocab*, uint64_t>, MaskCache*> g_caches;

static MaskCache* get_mask_cache(const Vocab* vb, uint64_t gsig) {
    std::lock_guard<std::mutex> lk(g_cache_mu);
    auto key = std::make_pair(vb, gsig);
    auto it = g_caches.find(key);
    if (it != g_caches.end()) return it->second;
    MaskCache* c = new MaskCache();
    g_caches[key] = c;
    return c;
}

struct Ctx2 {
    Grammar g;
    const Vocab* vb;  // shared, not owned
    MaskCache* mc;    // shared per (vocab, grammar identity), not owned
    Ctx2(int mode, const Vocab* v)
        : g(mode), vb(v), mc(get_mask_cache(v, grammar_sig(mode, {}))) {}
};

}  // namespace

extern "C" {

// Shared token table — built ONCE per tokenizer (the per-request grammar
// machines reference it), so request setup is O(1) not O(vocab).
void* oa_vocab_create(const int32_t* token_lens, const uint8_t* token_bytes_concat,
                      int vocab, int eos_id) {
    Vocab* vb = new Vocab();
    vb->vocab = vocab;
    vb->eos_id = eos_id;
    int64_t total = 0;
    for (int i = 0; i < vocab; ++i) total += token_lens[i];
    vb->token_store.assign(token_bytes_concat, token_bytes_concat + total);
    vb->tokens.resize(vocab);
    int64_t off = 0;
    for (int i = 0; i < vocab; ++i) {
        vb->tokens[i] = {vb->token_store.data() + off, token_lens[i]};
        off += token_lens[i];
        if (token_lens[i] > 0) vb->realizable.push_back(i);
    }
    return vb;
}

void oa_vocab_destroy(void* v) { delete (Vocab*)v; }

void* oa_grammar_create(int mode, void* vocab_handle) {
    return new Ctx2(mode, (const Vocab*)vocab_handle);
}

// Constrained tool names: the template's name field only accepts one of
// the declared names (<= 32; byte-concatenated with per-name lengths).
void* oa_grammar_create_names(int mode, void* vocab_handle,
                              const uint8_t* names_concat,
                              const int32_t* name_lens, int n_names) {
    Ctx2* c = new Ctx2(mode, (const Vocab*)vocab_handle);
    if (n_names > 0 && n_names <= 32) {
        std::vector<std::string> names;
        int64_t off = 0;
        for (int i = 0; i < n_names; ++i) {
            names.emplace_back((const char*)names_concat + off, name_lens[i]);
            off += name_lens[i];
        }
        c->mc = get_mask_cache(c->vb, grammar_sig(mode, names));
        c->g.set_names(std::move(names));
    }
    return c;
}

void oa_grammar_destroy(void* h) { delete (Ctx2*)h; }

void oa_grammar_reset(void* h) { ((Ctx2*)h)->g.reset(); }

int oa_grammar_is_complete(void* h) {
    Ctx2* c = (Ctx2*)h;
    return c->g.is_complete(c->g.st_) ? 1 : 0;
}

// advance by a sampled token; returns 0 ok, -1 token not allowed
int oa_grammar_accept_token(void* h, int token) {
    Ctx2* c = (Ctx2*)h;
    if (token == c->vb->eos_id) return c->g.is_complete(c->g.st_) ? 0 : -1;
    if (token < 0 || token >= c->vb->vocab) return -1;
    auto [ptr, len] = c->vb->tokens[token];
    if (len == 0) return -1;
    MachineState backup = c->g.st_;
    for (int i = 0; i < len; ++i) {
        if (!Grammar::step(c->g.st_, ptr[i], c->g)) {
            c->g.st_ = backup;
            return -1;
        }
    }
    return 0;
}

// Shortest completion: bytes that drive the machine from its current state to
// acceptance (used when the token budget runs out, so constrained output is
// ALWAYS valid JSON — never a truncated document). Returns length, or -1 if
// out of space / impossible. Does not mutate the live state.
int oa_grammar_completion(void* h, uint8_t* out, int max_len) {
    Ctx2* c = (Ctx2*)h;
    MachineState s = c->g.st_;
    const Grammar& g = c->g;
    int n = 0;
    auto emit = [&](uint8_t b) -> bool {
        if (n >= max_len) return false;
        if (!Grammar::step(s, b, g)) return false;
        out[n++] = b;
        return true;
    };
    int guard = max_len;
    while (!g.is_complete(s) && guard-- > 0) {
        uint8_t b;
        // template literal in progress?
        if (s.tpl_idx >= 0 && !s.in_jsonval && s.tpl_idx < (int)g.tpl_.size() &&
            g.tpl_[s.tpl_idx].kind == T_LIT) {
            b = (uint8_t)g.tpl_[s.tpl_idx].lit[s.tpl_lit_pos];
        } else if (s.tpl_idx >= 0 && !s.in_jsonval &&
                   s.tpl_idx < (int)g.tpl_.size() &&
                   g.tpl_[s.tpl_idx].kind == T_NAMES) {
            b = '"';
            for (int i = 0; i < (int)g.names_.size(); ++i)
                if ((s.alt_alive >> i) & 1u) {
                    b = (s.alt_pos < (int)g.names_[i].size())
                            ? (uint8_t)g.names_[i][s.alt_pos]
                            : (uint8_t)'"';
                    break;
                }
        } else {
            switch (s.state) {
                case S_STR: b = '"'; break;
                case S_STR_ESC: b = 'n'; break;
                case S_STR_U0: case S_STR_U1: case S_STR_U2: case S_STR_U3:
                    b = '0'; break;
                case S_NUM_MINUS: case S_NUM_DOT: case S_NUM_E: case S_NUM_ESIGN:
                    b = '0'; break;
                case S_NUM_ZERO: case S_NUM_INT: case S_NUM_FRAC: case S_NUM_EXP:
                    // number is terminable: close the enclosing container
                    b = (s.depth > 0 && s.stack[s.depth - 1] == CTX_ARR) ? ']' : '}';
                    break;
                case S_LIT: b = (uint8_t)s.lit[s.lit_pos]; break;
                case S_VALUE:
                    // empty array close / top-level-object open / null
                    b = s.arr_fresh ? ']'
                        : (s.depth == 0 && s.tpl_idx < 0 && g.mode_ == 0) ? '{'
                                                                          : 'n';
                    break;
                case S_AFTER_VALUE:
                    b = (s.depth > 0 && s.stack[s.depth - 1] == CTX_ARR) ? ']' : '}';
                    break;
                case S_OBJ_FIRST: b = '}'; break;
                case S_OBJ_KEY: b = '"'; break;   // then S_STR(key) closes + colon
                case S_OBJ_COLON: b = ':'; break;
                default: return -1;
            }
        }
        if (!emit(b)) return -1;
    }
    return g.is_complete(s) ? n : -1;
}

// Forced run (jump-ahead decoding): while the grammar allows exactly ONE
// token (template literals, structural bytes), emit its bytes and advance —
// the masked argmax could only ever pick that token, so the engine appends
// it without a model forward. If fewer than min_tokens forced tokens are
// found the state is RESTORED and 0 returned: a short jump is not worth the
// KV catch-up pass (~2 decode-step-equivalents of eager prefill).
// Returns bytes written (== tokens for the byte-level vocab).
int oa_grammar_forced_run(void* h, uint8_t* out, int max_bytes, int min_tokens) {
    Ctx2* c = (Ctx2*)h;
    Grammar save = c->g;
    int n = 0, ntok = 0;
    while (n < max_bytes && !c->g.is_complete(c->g.st_)) {
        int cand = -1, count = 0;
        for (int t : c->vb->realizable) {
            auto [ptr, len] = c->vb->tokens[t];
            MachineState s = c->g.st_;
            bool ok = true;
            for (int i = 0; i < len && ok; ++i) ok = Grammar::step(s, ptr[i], c->g);
            if (ok && ++count > 1) break;
            if (ok) cand = t;
        }
        if (count != 1) break;
        auto [ptr, len] = c->vb->tokens[cand];
        if (n + len > max_bytes) break;
        for (int i = 0; i < len; ++i) {
            Grammar::step(c->g.st_, ptr[i], c->g);
            out[n++] = ptr[i];
        }
        ++ntok;
    }
    if (ntok < min_tokens) {
        c->g = save;
        return 0;
    }
    return n;
}

// Forced BYTES peek (BPE jump-ahead): walk the byte-level machine while the
// allowed next-byte set is a singleton, WITHOUT touching the live state.
// Unlike oa_grammar_forced_run (which needs a singleton allowed-TOKEN set —
// never true for BPE vocabs, where many tokens share a forced prefix), this
// identifies the unique byte continuation; the engine tokenizes it and
// accepts whole in-run tokens via oa_grammar_accept_token.
int oa_grammar_forced_bytes(void* h, uint8_t* out, int max_bytes) {
    Ctx2* c = (Ctx2*)h;
    MachineState s = c->g.st_;
    int n = 0;
    while (n < max_bytes && !c->g.is_complete(s)) {
        int cand = -1, count = 0;
        for (int b = 0; b < 256; ++b) {
            MachineState t = s;
            if (Grammar::step(t, (uint8_t)b, c->g)) {
                if (++count > 1) break;
                cand = b;
            }
        }
        if (count != 1) break;
        Grammar::step(s, (uint8_t)cand, c->g);
        out[n++] = (uint8_t)cand;
    }
    return n;
}

// fill the allowed-token bitmask for an arbitrary machine state
static void fill_mask_state(Ctx2* c, const MachineState& st, uint32_t* mask_words) {
    const int words = (c->vb->vocab + 31) / 32;
    const uint64_t sig = state_sig(st);
    {
        std::lock_guard<std::mutex> lk(c->mc->mu);
        auto it = c->mc->m.find(sig);
        if (it != c->mc->m.end()) {
            memcpy(mask_words, it->second.data(), words * 4);
            return;
        }
    }
    memset(mask_words, 0, words * 4);
    if (c->g.is_complete(st)) {
        const int t = c->vb->eos_id;
        mask_words[t >> 5] |= 1u << (t & 31);
    }
    for (int t : c->vb->realizable) {
        auto [ptr, len] = c->vb->tokens[t];
        MachineState s = st;
        bool ok = true;
        for (int i = 0; i < len && ok; ++i) ok = Grammar::step(s, ptr[i], c->g);
        if (ok) mask_words[t >> 5] |= 1u << (t & 31);
    }
    std::lock_guard<std::mutex> lk(c->mc->mu);
    if (c->mc->m.size() < 4096)
        c->mc->m.emplace(sig, std::vector<uint32_t>(mask_words, mask_words + words));
}

// fill the allowed-token bitmask (vocab bits, 32 per word, little-endian bit order)
void oa_grammar_fill_mask(void* h, uint32_t* mask_words) {
    Ctx2* c = (Ctx2*)h;
    fill_mask_state(c, c->g.st_, mask_words);
}

// length of the grammar-legal prefix of a token sequence, SIMULATED from the
// current state (the live state is untouched) — used to pre-filter
// speculative proposals
int oa_grammar_check_tokens(void* h, const int32_t* toks, int n) {
    Ctx2* c = (Ctx2*)h;
    MachineState s = c->g.st_;
    for (int i = 0; i < n; ++i) {
        const int t = toks[i];
        if (t < 0 || t >= c->vb->vocab) return i;
        auto [ptr, len] = c->vb->tokens[t];
        if (len == 0) {
            // specials: only EOS, and only at completion
            if (t != c->vb->eos_id || !c->g.is_complete(s)) return i;
            continue;
        }
        for (int j = 0; j < len; ++j)
            if (!Grammar::step(s, ptr[j], c->g)) return i;
    }
    return n;
}

// masks along a speculative proposal path (grammar-aware verify): writes
// mask i (i = 0..m) = the allowed set at the state reached after accepting
// toks[0..i-1]; stops at the first illegal token. Returns the number of
// masks written (= legal prefix length + 1). The live state is untouched.
int oa_grammar_masks_along(void* h, const int32_t* toks, int n,
                           uint32_t* masks_out) {
    Ctx2* c = (Ctx2*)h;
    const int words = (c->vb->vocab + 31) / 32;
    MachineState s = c->g.st_;
    int written = 0;
    for (int i = 0; i <= n; ++i) {
        fill_mask_state(c, s, masks_out + (size_t)i * words);
        ++written;
        if (i == n) break;
        const int t = toks[i];
        if (t < 0 || t >= c->vb->vocab) break;
        auto [ptr, len] = c->vb->tokens[t];
        bool ok = len > 0;
        for (int j = 0; j < len && ok; ++j) ok = Grammar::step(s, ptr[j], c->g);
        if (!ok) break;
    }
    return written;
}

}  // extern "C"
