// Native tokenizer (reference parity: the C++ BPE/SPM tokenizer inside
// llama.cpp that Ollama uses — SURVEY.md §2.3 row "Tokenizer").
// Byte-level BPE with merge ranks + <0xNN> byte-fallback vocabularies.
#include <cstdint>
#include <map>
#include <string>
#include <unordered_map>
#include <vector>

namespace cla {

namespace {

// GPT-2 byte <-> unicode table (standard public construction).
const std::vector<std::string>& byte_to_unicode() {
    static std::vector<std::string> table = [] {
        std::vector<int> bs;
        for (int b = int('!'); b <= int('~'); b++) bs.push_back(b);
        for (int b = 0xA1; b <= 0xAC; b++) bs.push_back(b);
        for (int b = 0xAE; b <= 0xFF; b++) bs.push_back(b);
        std::vector<int> cs = bs;
        int n = 0;
        for (int b = 0; b < 256; b++) {
            bool found = false;
            for (int x : bs)
                if (x == b) { found = true; break; }
            if (!found) {
                bs.push_back(b);
                cs.push_back(256 + n++);
            }
        }
        std::vector<std::string> t(256);
        for (size_t i = 0; i < bs.size(); i++) {
            // encode cs[i] as UTF-8
            int cp = cs[i];
            std::string u;
            if (cp < 0x80) {
                u += (char)cp;
            } else if (cp < 0x800) {
                u += (char)(0xC0 | (cp >> 6));
                u += (char)(0x80 | (cp & 0x3F));
            } else {
                u += (char)(0xE0 | (cp >> 12));
                u += (char)(0x80 | ((cp >> 6) & 0x3F));
                u += (char)(0x80 | (cp & 0x3F));
            }
            t[bs[i]] = u;
        }
        return t;
    }();
    return table;
}

}  // namespace

class Tokenizer {
public:
    Tokenizer(std::vector<std::string> tokens,
              const std::vector<std::string>& merges,
              int bos_id, int eos_id)
        : tokens_(std::move(tokens)), bos_(bos_id), eos_(eos_id) {
        for (size_t i = 0; i < tokens_.size(); i++)
            vocab_[tokens_[i]] = (int)i;
        for (size_t i = 0; i < merges.size(); i++) {
            auto sp = merges[i].find(' ');
            if (sp == std::string::npos) continue;
            ranks_[merges[i].substr(0, sp) + "\x01" + merges[i].substr(sp + 1)]
                = (int)i;
        }
        for (size_t i = 0; i < tokens_.size(); i++) {
            const std::string& t = tokens_[i];
            if (t.size() == 6 && t.rfind("<0x", 0) == 0 && t.back() == '>') {
                int v = std::stoi(t.substr(3, 2), nullptr, 16);
                byte_tok_[v] = (int)i;
            }
        }
    }

    std::vector<int32_t> encode(const std::string& text, bool add_bos) const {
        std::vector<int32_t> out;
        if (add_bos && bos_ >= 0) out.push_back(bos_);
        if (!ranks_.empty()) {
            // byte-level BPE over the unicode-mapped text
            const auto& b2u = byte_to_unicode();
            std::vector<std::string> parts;
            for (unsigned char c : text) parts.push_back(b2u[c]);
            while (parts.size() > 1) {
                int best = -1, best_rank = INT32_MAX;
                for (size_t i = 0; i + 1 < parts.size(); i++) {
                    auto it = ranks_.find(parts[i] + "\x01" + parts[i + 1]);
                    if (it != ranks_.end() && it->second < best_rank) {
                        best = (int)i;
                        best_rank = it->second;
                    }
                }
                if (best < 0) break;
                parts[best] += parts[best + 1];
                parts.erase(parts.begin() + best + 1);
            }
            for (auto& p : parts) {
                auto it = vocab_.find(p);
                if (it != vocab_.end()) {
                    out.push_back(it->second);
                } else {
                    for (char ch : p) {
                        auto i2 = vocab_.find(std::string(1, ch));
                        if (i2 != vocab_.end()) out.push_back(i2->second);
                    }
                }
            }
        } else if (!byte_tok_.empty()) {
            for (unsigned char c : text) {
                auto it = byte_tok_.find(c);
                if (it != byte_tok_.end()) out.push_back(it->second);
            }
        } else {
            for (unsigned char c : text) {
                auto it = vocab_.find(std::string(1, (char)c));
                if (it != vocab_.end()) out.push_back(it->second);
            }
        }
        return out;
    }

    std::string decode(const std::vector<int32_t>& ids) const {
        // reverse byte-unicode map
        static std::unordered_map<std::string, int> u2b = [] {
            std::unordered_map<std::string, int> m;
            const auto& t = byte_to_unicode();
            for (int b = 0; b < 256; b++) m[t[b]] = b;
            return m;
        }();
        std::string out;
        std::unordered_map<int, int> rev_byte;
        for (auto& kv : byte_tok_) rev_byte[kv.second] = kv.first;
        for (int32_t id : ids) {
            if (id == bos_ || id == eos_) continue;
            if (id < 0 || id >= (int)tokens_.size()) continue;
            auto rb = rev_byte.find(id);
            if (rb != rev_byte.end()) {
                out += (char)rb->second;
                continue;
            }
            const std::string& tok = tokens_[id];
            if (!ranks_.empty()) {
                // walk UTF-8 chars; map back through the byte table
                size_t i = 0;
                while (i < tok.size()) {
                    size_t len = 1;
                    unsigned char c = tok[i];
                    if ((c & 0xE0) == 0xC0) len = 2;
                    else if ((c & 0xF0) == 0xE0) len = 3;
                    else if ((c & 0xF8) == 0xF0) len = 4;
                    std::string ch = tok.substr(i, len);
                    auto it = u2b.find(ch);
                    if (it != u2b.end()) out += (char)it->second;
                    else out += ch;
                    i += len;
                }
            } else {
                std::string t = tok;
                size_t pos;
                while ((pos = t.find("\xE2\x96\x81")) != std::string::npos)
                    t.replace(pos, 3, " ");  // SPM space marker
                out += t;
            }
        }
        return out;
    }

    int bos() const { return bos_; }
    int eos() const { return eos_; }
    size_t size() const { return tokens_.size(); }

private:
    std::vector<std::string> tokens_;
    std::unordered_map<std::string, int> vocab_;
    std::unordered_map<std::string, int> ranks_;
    std::map<int, int> byte_tok_;
    int bos_, eos_;
};

}  // namespace cla

namespace cla {
Tokenizer* tokenizer_new(std::vector<std::string> tokens,
                         const std::vector<std::string>& merges,
                         int bos_id, int eos_id) {
    return new Tokenizer(std::move(tokens), merges, bos_id, eos_id);
}
void tokenizer_free(Tokenizer* t) { delete t; }
std::vector<int32_t> tokenizer_encode(const Tokenizer* t,
                                      const std::string& s, bool add_bos) {
    return t->encode(s, add_bos);
}
std::string tokenizer_decode(const Tokenizer* t,
                             const std::vector<int32_t>& ids) {
    return t->decode(ids);
}
int tokenizer_bos(const Tokenizer* t) { return t->bos(); }
int tokenizer_eos(const Tokenizer* t) { return t->eos(); }
size_t tokenizer_size(const Tokenizer* t) { return t->size(); }
}  // namespace cla
