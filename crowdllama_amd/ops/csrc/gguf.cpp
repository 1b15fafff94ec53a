#include "gguf.h"

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstring>
#include <stdexcept>

namespace cla {

namespace {

constexpr uint32_t kMagic = 0x46554747;  // "GGUF"

enum VType : uint32_t {
    V_U8 = 0, V_I8 = 1, V_U16 = 2, V_I16 = 3, V_U32 = 4, V_I32 = 5,
    V_F32 = 6, V_BOOL = 7, V_STR = 8, V_ARR = 9, V_U64 = 10, V_I64 = 11,
    V_F64 = 12,
};

struct Cursor {
    const uint8_t* p;
    const uint8_t* end;

    template <typename T>
    T read() {
        if (p + sizeof(T) > end) throw std::runtime_error("truncated GGUF");
        T v;
        std::memcpy(&v, p, sizeof(T));
        p += sizeof(T);
        return v;
    }
    std::string read_str() {
        uint64_t n = read<uint64_t>();
        if (p + n > end) throw std::runtime_error("truncated GGUF string");
        std::string s(reinterpret_cast<const char*>(p), n);
        p += n;
        return s;
    }
};

GGUFValue read_value(Cursor& c, uint32_t vt);

GGUFValue read_scalar(Cursor& c, uint32_t vt) {
    switch (vt) {
        case V_U8: return (int64_t)c.read<uint8_t>();
        case V_I8: return (int64_t)c.read<int8_t>();
        case V_U16: return (int64_t)c.read<uint16_t>();
        case V_I16: return (int64_t)c.read<int16_t>();
        case V_U32: return (int64_t)c.read<uint32_t>();
        case V_I32: return (int64_t)c.read<int32_t>();
        case V_U64: return (int64_t)c.read<uint64_t>();
        case V_I64: return (int64_t)c.read<int64_t>();
        case V_F32: return (double)c.read<float>();
        case V_F64: return c.read<double>();
        case V_BOOL: return c.read<uint8_t>() != 0;
        case V_STR: return c.read_str();
        default: throw std::runtime_error("bad GGUF value type");
    }
}

GGUFValue read_value(Cursor& c, uint32_t vt) {
    if (vt != V_ARR) return read_scalar(c, vt);
    uint32_t et = c.read<uint32_t>();
    uint64_t n = c.read<uint64_t>();
    if (et == V_STR) {
        std::vector<std::string> out;
        out.reserve(n);
        for (uint64_t i = 0; i < n; i++) out.push_back(c.read_str());
        return out;
    }
    if (et == V_F32 || et == V_F64) {
        std::vector<double> out;
        out.reserve(n);
        for (uint64_t i = 0; i < n; i++)
            out.push_back(std::get<double>(read_scalar(c, et)));
        return out;
    }
    std::vector<int64_t> out;
    out.reserve(n);
    for (uint64_t i = 0; i < n; i++)
        out.push_back(std::get<int64_t>(read_scalar(c, et)));
    return out;
}

}  // namespace

int64_t ggml_row_bytes(int32_t t, int64_t k) {
    switch (t) {
        case 0: return k * 4;             // F32
        case 1: return k * 2;             // F16
        case 30: return k * 2;            // BF16
        case 8: return k / 32 * 34;       // Q8_0
        case 12: return k / 256 * 144;    // Q4_K
        case 14: return k / 256 * 210;    // Q6_K
        default:
            throw std::runtime_error("unsupported ggml type " + std::to_string(t));
    }
}

GGUFFile::GGUFFile(const std::string& path) {
    fd_ = ::open(path.c_str(), O_RDONLY);
    if (fd_ < 0) throw std::runtime_error("cannot open " + path);
    struct stat st;
    if (fstat(fd_, &st) != 0) throw std::runtime_error("fstat failed: " + path);
    size_ = st.st_size;
    void* m = mmap(nullptr, size_, PROT_READ, MAP_PRIVATE, fd_, 0);
    if (m == MAP_FAILED) throw std::runtime_error("mmap failed: " + path);
    base_ = static_cast<const uint8_t*>(m);
    madvise(const_cast<uint8_t*>(base_), size_, MADV_SEQUENTIAL);
    parse();
}

GGUFFile::~GGUFFile() {
    if (base_) munmap(const_cast<uint8_t*>(base_), size_);
    if (fd_ >= 0) ::close(fd_);
}

void GGUFFile::parse() {
    Cursor c{base_, base_ + size_};
    if (c.read<uint32_t>() != kMagic) throw std::runtime_error("not a GGUF file");
    uint32_t version = c.read<uint32_t>();
    if (version != 2 && version != 3)
        throw std::runtime_error("unsupported GGUF version");
    uint64_t n_tensors = c.read<uint64_t>();
    uint64_t n_kv = c.read<uint64_t>();
    for (uint64_t i = 0; i < n_kv; i++) {
        std::string key = c.read_str();
        uint32_t vt = c.read<uint32_t>();
        meta_[key] = read_value(c, vt);
    }
    int64_t align = meta_int("general.alignment", 32);
    struct RawInfo { std::string name; std::vector<int64_t> shape; int32_t t; uint64_t off; };
    std::vector<RawInfo> infos;
    for (uint64_t i = 0; i < n_tensors; i++) {
        RawInfo ri;
        ri.name = c.read_str();
        uint32_t nd = c.read<uint32_t>();
        std::vector<int64_t> ne(nd);
        for (uint32_t d = 0; d < nd; d++) ne[d] = c.read<uint64_t>();
        ri.shape.assign(ne.rbegin(), ne.rend());  // GGUF stores innermost first
        ri.t = c.read<uint32_t>();
        ri.off = c.read<uint64_t>();
        infos.push_back(std::move(ri));
    }
    size_t data_start = (size_t)((c.p - base_ + align - 1) / align * align);
    for (auto& ri : infos) {
        GGUFTensor t;
        t.name = ri.name;
        t.shape = ri.shape;
        t.ggml_type = ri.t;
        t.offset = ri.off;
        int64_t rows = 1;
        for (size_t d = 0; d + 1 < ri.shape.size(); d++) rows *= ri.shape[d];
        t.nbytes = rows * ggml_row_bytes(ri.t, ri.shape.back());
        if (data_start + t.offset + t.nbytes > size_)
            throw std::runtime_error("tensor " + t.name + " out of file bounds");
        t.data = base_ + data_start + t.offset;
        tensors_[t.name] = std::move(t);
    }
}

const GGUFTensor& GGUFFile::tensor(const std::string& name) const {
    auto it = tensors_.find(name);
    if (it == tensors_.end())
        throw std::runtime_error("missing tensor: " + name);
    return it->second;
}

bool GGUFFile::has_tensor(const std::string& name) const {
    return tensors_.count(name) != 0;
}

bool GGUFFile::has_meta(const std::string& key) const { return meta_.count(key) != 0; }

int64_t GGUFFile::meta_int(const std::string& key, int64_t fallback) const {
    auto it = meta_.find(key);
    if (it == meta_.end()) return fallback;
    if (auto* v = std::get_if<int64_t>(&it->second)) return *v;
    if (auto* v = std::get_if<double>(&it->second)) return (int64_t)*v;
    return fallback;
}

double GGUFFile::meta_float(const std::string& key, double fallback) const {
    auto it = meta_.find(key);
    if (it == meta_.end()) return fallback;
    if (auto* v = std::get_if<double>(&it->second)) return *v;
    if (auto* v = std::get_if<int64_t>(&it->second)) return (double)*v;
    return fallback;
}

std::string GGUFFile::meta_str(const std::string& key, const std::string& fallback) const {
    auto it = meta_.find(key);
    if (it == meta_.end()) return fallback;
    if (auto* v = std::get_if<std::string>(&it->second)) return *v;
    return fallback;
}

}  // namespace cla
