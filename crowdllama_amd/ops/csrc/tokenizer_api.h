#pragma once
// Native tokenizer interface (implementation in tokenizer.cpp).
#include <cstdint>
#include <string>
#include <vector>

namespace cla {
class Tokenizer;
Tokenizer* tokenizer_new(std::vector<std::string> tokens,
                         const std::vector<std::string>& merges,
                         int bos_id, int eos_id);
void tokenizer_free(Tokenizer*);
std::vector<int32_t> tokenizer_encode(const Tokenizer*, const std::string&,
                                      bool add_bos);
std::string tokenizer_decode(const Tokenizer*, const std::vector<int32_t>&);
int tokenizer_bos(const Tokenizer*);
int tokenizer_eos(const Tokenizer*);
size_t tokenizer_size(const Tokenizer*);
}  // namespace cla
