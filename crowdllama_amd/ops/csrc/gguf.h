// Minimal C++ GGUF v3 reader: mmap the file, parse metadata + tensor table.
// (MI355X-native replacement for the GGUF loading the reference delegates to
// Ollama/llama.cpp — SURVEY.md §2.3 row "GGUF loader".)
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <variant>
#include <vector>

namespace cla {

struct GGUFTensor {
    std::string name;
    std::vector<int64_t> shape;  // logical, outer-first (numpy order)
    int32_t ggml_type = 0;
    uint64_t offset = 0;         // from data-section start
    uint64_t nbytes = 0;
    const uint8_t* data = nullptr;  // pointer into the mmap
};

using GGUFValue = std::variant<int64_t, double, bool, std::string,
                               std::vector<int64_t>, std::vector<double>,
                               std::vector<std::string>>;

class GGUFFile {
public:
    explicit GGUFFile(const std::string& path);
    ~GGUFFile();
    GGUFFile(const GGUFFile&) = delete;
    GGUFFile& operator=(const GGUFFile&) = delete;

    const GGUFTensor& tensor(const std::string& name) const;
    bool has_tensor(const std::string& name) const;
    const std::map<std::string, GGUFTensor>& tensors() const { return tensors_; }

    int64_t meta_int(const std::string& key, int64_t fallback) const;
    double meta_float(const std::string& key, double fallback) const;
    std::string meta_str(const std::string& key, const std::string& fallback) const;
    bool has_meta(const std::string& key) const;
    const std::map<std::string, GGUFValue>& metadata() const { return meta_; }

private:
    void parse();
    int fd_ = -1;
    size_t size_ = 0;
    const uint8_t* base_ = nullptr;
    std::map<std::string, GGUFTensor> tensors_;
    std::map<std::string, GGUFValue> meta_;
};

int64_t ggml_row_bytes(int32_t ggml_type, int64_t n_cols);

}  // namespace cla
