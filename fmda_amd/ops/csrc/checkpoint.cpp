// Native reader/writer for the PyTorch zip-format state_dict
// (`model_params.pt`, reference training notebook cell 39 / predict.py:104).
//
// This is a from-scratch implementation of the two formats involved —
// the STORE-only zip container and the pickle stream torch emits for an
// OrderedDict[str, Tensor] — NOT a call into torch serialization. It
// round-trips byte-compatibly with torch.save/torch.load (verified in
// tests/test_checkpoint_cpp.py against both directions):
//   archive/data.pkl   pickle protocol 2: OrderedDict of
//                      torch._utils._rebuild_tensor_v2(storage, offset,
//                      shape, stride, False, OrderedDict())
//   archive/data/<N>   raw little-endian storage bytes
//   archive/version    "3"
// Reader implements a restricted pickle VM (exactly the opcodes the torch
// pickler emits for fp32/bf16/fp64/int64 CPU state_dicts); writer emits a
// canonical stream the (weights_only) torch unpickler accepts.
#include <torch/extension.h>

#include <cstdint>
#include <cstring>
#include <fstream>
#include <functional>
#include <map>
#include <sstream>
#include <string>
#include <vector>

namespace fmda_ckpt {

// --------------------------- CRC32 (zip) ----------------------------------
static uint32_t crc_table[256];
static bool crc_init_done = false;
static void crc_init() {
    if (crc_init_done) return;
    for (uint32_t i = 0; i < 256; ++i) {
        uint32_t c = i;
        for (int k = 0; k < 8; ++k)
            c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
        crc_table[i] = c;
    }
    crc_init_done = true;
}
static uint32_t crc32(const uint8_t* p, size_t n) {
    crc_init();
    uint32_t c = 0xFFFFFFFFu;
    for (size_t i = 0; i < n; ++i)
        c = crc_table[(c ^ p[i]) & 0xFF] ^ (c >> 8);
    return c ^ 0xFFFFFFFFu;
}

// --------------------------- zip writer (STORE) ----------------------------
struct ZipEntry {
    std::string name;
    uint32_t crc, size, offset;
};

class ZipWriter {
  public:
    explicit ZipWriter(const std::string& path) : f_(path, std::ios::binary) {
        TORCH_CHECK(f_.good(), "cannot open ", path, " for writing");
    }
    void add(const std::string& name, const void* data, size_t n) {
        ZipEntry e{name, crc32((const uint8_t*)data, n), (uint32_t)n,
                   (uint32_t)f_.tellp()};
        wr32(0x04034b50); wr16(20); wr16(0); wr16(0);  // local header
        wr16(0); wr16(0);                               // time/date
        wr32(e.crc); wr32(e.size); wr32(e.size);
        wr16((uint16_t)name.size()); wr16(0);
        f_.write(name.data(), name.size());
        f_.write((const char*)data, n);
        entries_.push_back(e);
    }
    void finish() {
        uint32_t cd_off = (uint32_t)f_.tellp();
        for (const auto& e : entries_) {
            wr32(0x02014b50); wr16(20); wr16(20); wr16(0); wr16(0);
            wr16(0); wr16(0);
            wr32(e.crc); wr32(e.size); wr32(e.size);
            wr16((uint16_t)e.name.size()); wr16(0); wr16(0);
            wr16(0); wr16(0); wr32(0);
            wr32(e.offset);
            f_.write(e.name.data(), e.name.size());
        }
        uint32_t cd_size = (uint32_t)f_.tellp() - cd_off;
        wr32(0x06054b50); wr16(0); wr16(0);
        wr16((uint16_t)entries_.size()); wr16((uint16_t)entries_.size());
        wr32(cd_size); wr32(cd_off); wr16(0);
        f_.close();
    }

  private:
    void wr16(uint16_t v) { f_.write((const char*)&v, 2); }
    void wr32(uint32_t v) { f_.write((const char*)&v, 4); }
    std::ofstream f_;
    std::vector<ZipEntry> entries_;
};

// --------------------------- zip reader ------------------------------------
class ZipReader {
  public:
    explicit ZipReader(const std::string& path) {
        std::ifstream f(path, std::ios::binary | std::ios::ate);
        TORCH_CHECK(f.good(), "cannot open ", path);
        size_t n = (size_t)f.tellg();
        buf_.resize(n);
        f.seekg(0);
        f.read((char*)buf_.data(), n);
        // find end-of-central-directory
        TORCH_CHECK(n >= 22, "not a zip: ", path);
        size_t i = n - 22;
        while (true) {
            if (rd32(i) == 0x06054b50) break;
            TORCH_CHECK(i > 0 && n - i < 22 + 65536, "zip EOCD not found");
            --i;
        }
        uint16_t count = rd16(i + 10);
        size_t cd = rd32(i + 16);
        for (uint16_t k = 0; k < count; ++k) {
            TORCH_CHECK(rd32(cd) == 0x02014b50, "bad central header");
            uint16_t method = rd16(cd + 10);
            uint32_t size = rd32(cd + 24);
            uint16_t nlen = rd16(cd + 28), xlen = rd16(cd + 30),
                     clen = rd16(cd + 32);
            uint32_t lho = rd32(cd + 42);
            std::string name((const char*)&buf_[cd + 46], nlen);
            TORCH_CHECK(method == 0, "zip entry ", name,
                        " is compressed; only STORE supported");
            // local header: skip its (possibly different) name/extra
            uint16_t lnlen = rd16(lho + 26), lxlen = rd16(lho + 28);
            files_[name] = {lho + 30 + (size_t)lnlen + lxlen, size};
            cd += 46 + nlen + xlen + clen;
        }
    }
    bool has(const std::string& suffix) const { return find(suffix) != nullptr; }
    std::pair<const uint8_t*, size_t> get(const std::string& suffix) const {
        const auto* e = find(suffix);
        TORCH_CHECK(e, "zip entry *", suffix, " not found");
        return {buf_.data() + e->first, e->second};
    }

  private:
    const std::pair<size_t, uint32_t>* find(const std::string& sfx) const {
        for (const auto& kv : files_) {
            const std::string& nm = kv.first;
            if (nm.size() >= sfx.size() &&
                nm.compare(nm.size() - sfx.size(), sfx.size(), sfx) == 0)
                return &kv.second;
        }
        return nullptr;
    }
    uint16_t rd16(size_t o) const { uint16_t v; memcpy(&v, &buf_[o], 2); return v; }
    uint32_t rd32(size_t o) const { uint32_t v; memcpy(&v, &buf_[o], 4); return v; }
    std::vector<uint8_t> buf_;
    std::map<std::string, std::pair<size_t, uint32_t>> files_;
};

// --------------------------- pickle writer ---------------------------------
// Canonical protocol-2 stream for OrderedDict[str, Tensor], storages by
// persistent id ('storage', torch.<T>Storage, '<n>', 'cpu', numel).
class PickleWriter {
  public:
    std::string out;
    void proto() { put('\x80'); put(2); }
    void global_(const char* mod, const char* name) {
        put('c'); raw(mod); put('\n'); raw(name); put('\n'); memo();
    }
    void empty_tuple() { put(')'); }
    // caller always pushes MARK first; 't' pops to the mark for any arity
    void tuple_from_mark() { put('t'); memo(); }
    void mark() { put('('); }
    void reduce() { put('R'); memo(); }
    void binint(int64_t v) {
        TORCH_CHECK(v >= 0 && v <= 0x7FFFFFFF, "int range");
        if (v <= 0xFF) { put('K'); put((char)v); }
        else if (v <= 0xFFFF) { put('M'); put((char)(v & 0xFF)); put((char)(v >> 8)); }
        else { put('J'); le32((uint32_t)v); }
    }
    void unicode(const std::string& s) {
        put('X'); le32((uint32_t)s.size()); out += s; memo();
    }
    void newfalse() { put('\x89'); }
    void empty_dict() { put('}'); memo(); }
    void setitems_mark() { put('('); }
    void setitems() { put('u'); }
    void binpersid() { put('Q'); }
    void stop() { put('.'); }
    void memo() {  // BINPUT/LONG_BINPUT with running index
        if (memo_ < 256) { put('q'); put((char)memo_); }
        else { put('r'); le32((uint32_t)memo_); }
        ++memo_;
    }

  private:
    void put(char c) { out.push_back(c); }
    void raw(const char* s) { out += s; }
    void le32(uint32_t v) { out.append((const char*)&v, 4); }
    int memo_ = 0;
};

struct DType {
    const char* storage;     // torch pickle storage class name
    at::ScalarType st;
    size_t esize;
};
static const DType kDTypes[] = {
    {"FloatStorage", at::kFloat, 4},
    {"DoubleStorage", at::kDouble, 8},
    {"LongStorage", at::kLong, 8},
    {"IntStorage", at::kInt, 4},
    {"HalfStorage", at::kHalf, 2},
    {"BFloat16Storage", at::kBFloat16, 2},
    {"BoolStorage", at::kBool, 1},
};
static const DType& dtype_of(at::ScalarType st) {
    for (const auto& d : kDTypes)
        if (d.st == st) return d;
    TORCH_CHECK(false, "unsupported dtype for native checkpoint");
}
static const DType& dtype_by_name(const std::string& nm) {
    for (const auto& d : kDTypes)
        if (nm == d.storage) return d;
    TORCH_CHECK(false, "unsupported storage class ", nm);
}

void save_state_dict(const std::string& path,
                     const std::vector<std::string>& keys,
                     const std::vector<torch::Tensor>& tensors) {
    TORCH_CHECK(keys.size() == tensors.size());
    PickleWriter p;
    p.proto();
    p.global_("collections", "OrderedDict");
    p.empty_tuple();
    p.reduce();
    p.setitems_mark();
    std::vector<torch::Tensor> contig;
    for (size_t i = 0; i < keys.size(); ++i) {
        torch::Tensor t = tensors[i].detach().cpu().contiguous();
        contig.push_back(t);
        const DType& dt = dtype_of(t.scalar_type());
        p.unicode(keys[i]);
        p.global_("torch._utils", "_rebuild_tensor_v2");
        p.mark();
        {   // persistent id tuple ('storage', Storage, key, 'cpu', numel)
            p.mark();
            p.unicode("storage");
            p.global_("torch", dt.storage);
            p.unicode(std::to_string(i));
            p.unicode("cpu");
            p.binint(t.numel());
            p.tuple_from_mark();
            p.binpersid();
        }
        p.binint(0);  // storage offset
        p.mark();
        for (auto s : t.sizes()) p.binint(s);
        p.tuple_from_mark();
        p.mark();
        for (auto s : t.strides()) p.binint(s);
        p.tuple_from_mark();
        p.newfalse();             // requires_grad
        p.empty_dict();           // backward_hooks (OrderedDict-compatible)
        p.tuple_from_mark();
        p.reduce();
    }
    p.setitems();
    p.stop();

    ZipWriter z(path);
    z.add("archive/data.pkl", p.out.data(), p.out.size());
    for (size_t i = 0; i < contig.size(); ++i) {
        const auto& t = contig[i];
        z.add("archive/data/" + std::to_string(i), t.data_ptr(),
              t.numel() * t.element_size());
    }
    z.add("archive/version", "3\n", 2);
    z.finish();
}

// --------------------------- pickle reader ---------------------------------
struct PVal {  // tagged value for the mini VM
    enum Kind { NONE, INT, STR, BOOL, TUPLE, DICT_MARKER, GLOBAL, TENSOR,
                PERSID, MARKOBJ, ODICT } kind = NONE;
    int64_t i = 0;
    std::string s;
    std::vector<PVal> tup;
    torch::Tensor ten;
};

// Mini pickle VM. `data_of` resolves a storage key to raw bytes; on
// return, `end_pos` (when non-null) receives the position just past the
// STOP opcode so several back-to-back pickles can be parsed (legacy
// container). `keys_out` collects a top-level list of strings if the
// pickle's result is a list (the legacy storage-key list).
static std::vector<std::pair<std::string, torch::Tensor>> run_pickle(
        const uint8_t* pkl, size_t pkl_n,
        const std::function<std::pair<const uint8_t*, size_t>(
            const std::string&)>& data_of,
        size_t* end_pos = nullptr,
        std::vector<std::string>* keys_out = nullptr) {
    std::vector<PVal> stack;
    std::vector<size_t> marks;
    std::vector<PVal> memo;
    std::vector<std::pair<std::string, torch::Tensor>> result;

    size_t i = 0;
    auto need = [&](size_t k) { TORCH_CHECK(i + k <= pkl_n, "pickle truncated"); };
    auto rd8 = [&]() { need(1); return pkl[i++]; };
    auto rd32v = [&]() { need(4); uint32_t v; memcpy(&v, pkl + i, 4); i += 4; return v; };

    while (true) {
        uint8_t op = rd8();
        switch (op) {
            case 0x80: rd8(); break;                       // PROTO
            case '(': { PVal m; m.kind = PVal::MARKOBJ;
                        marks.push_back(stack.size()); stack.push_back(m); break; }
            case ')': { PVal t; t.kind = PVal::TUPLE; stack.push_back(t); break; }
            case '}': { PVal d; d.kind = PVal::ODICT; stack.push_back(d); break; }
            case ']': { PVal l; l.kind = PVal::TUPLE; stack.push_back(l); break; }
            case 'c': {                                    // GLOBAL
                std::string mod, name;
                while (true) { char c = (char)rd8(); if (c == '\n') break; mod += c; }
                while (true) { char c = (char)rd8(); if (c == '\n') break; name += c; }
                PVal g; g.kind = PVal::GLOBAL; g.s = mod + "." + name;
                stack.push_back(g); break;
            }
            case 'K': { PVal v; v.kind = PVal::INT; v.i = rd8(); stack.push_back(v); break; }
            case 'M': { PVal v; v.kind = PVal::INT; v.i = rd8(); v.i |= (int64_t)rd8() << 8;
                        stack.push_back(v); break; }
            case 'J': { PVal v; v.kind = PVal::INT; v.i = (int32_t)rd32v();
                        stack.push_back(v); break; }
            case 0x8a: {                                   // LONG1
                uint8_t n = rd8(); int64_t v = 0;
                for (int k = 0; k < n; ++k) {                  // >8 bytes:
                    uint8_t byt = rd8();                        // keep low 64
                    if (k < 8) v |= (int64_t)byt << (8 * k);
                }
                if (n && n <= 8 && (pkl[i - 1] & 0x80))
                    v -= (int64_t)1 << (8 * (int)n);
                PVal pv; pv.kind = PVal::INT; pv.i = v; stack.push_back(pv); break;
            }
            case 'N': { PVal v; stack.push_back(v); break; }   // NONE
            case 'a': {                                        // APPEND
                PVal v = stack.back(); stack.pop_back();
                stack.back().tup.push_back(v); break;
            }
            case 'e': {                                        // APPENDS
                size_t m = marks.back(); marks.pop_back();
                std::vector<PVal> items(stack.begin() + m + 1, stack.end());
                stack.resize(m);
                for (auto& it : items) stack.back().tup.push_back(it);
                break;
            }
            case 0x88 + 0x100: break;  // unreachable
            case 'X': { uint32_t n = rd32v(); need(n);
                        PVal v; v.kind = PVal::STR;
                        v.s.assign((const char*)pkl + i, n); i += n;
                        stack.push_back(v); break; }
            case 'U': { uint8_t n = rd8(); need(n);
                        PVal v; v.kind = PVal::STR;
                        v.s.assign((const char*)pkl + i, n); i += n;
                        stack.push_back(v); break; }
            case 0x8c: { uint8_t n = rd8(); need(n);       // SHORT_BINUNICODE
                        PVal v; v.kind = PVal::STR;
                        v.s.assign((const char*)pkl + i, n); i += n;
                        stack.push_back(v); break; }
            case 0x88: { PVal v; v.kind = PVal::BOOL; v.i = 1; stack.push_back(v); break; }
            case 0x89: { PVal v; v.kind = PVal::BOOL; v.i = 0; stack.push_back(v); break; }
            case 'q': { uint8_t n = rd8(); if (memo.size() <= n) memo.resize(n + 1);
                        memo[n] = stack.back(); break; }
            case 'r': { uint32_t n = rd32v(); if (memo.size() <= n) memo.resize(n + 1);
                        memo[n] = stack.back(); break; }
            case 0x94: { memo.push_back(stack.back()); break; }   // MEMOIZE
            case 'h': { uint8_t n = rd8(); stack.push_back(memo.at(n)); break; }
            case 'j': { uint32_t n = rd32v(); stack.push_back(memo.at(n)); break; }
            case 0x85: case 0x86: case 0x87: {             // TUPLE1..3
                int n = op - 0x84;
                PVal t; t.kind = PVal::TUPLE;
                t.tup.assign(stack.end() - n, stack.end());
                stack.resize(stack.size() - n);
                stack.push_back(t); break;
            }
            case 't': {                                    // TUPLE (mark)
                size_t m = marks.back(); marks.pop_back();
                PVal t; t.kind = PVal::TUPLE;
                t.tup.assign(stack.begin() + m + 1, stack.end());
                stack.resize(m);
                stack.push_back(t); break;
            }
            case 'Q': {                                    // BINPERSID
                PVal pid = stack.back(); stack.pop_back();
                TORCH_CHECK(pid.kind == PVal::TUPLE && pid.tup.size() >= 5 &&
                            pid.tup[0].s == "storage", "unexpected persistent id");
                PVal v; v.kind = PVal::PERSID;
                v.s = pid.tup[2].s;                         // data key
                v.tup.push_back(pid.tup[1]);                // storage global
                v.i = pid.tup[4].i;                         // numel
                stack.push_back(v); break;
            }
            case 'R': {                                    // REDUCE
                PVal args = stack.back(); stack.pop_back();
                PVal fn = stack.back(); stack.pop_back();
                if (fn.kind == PVal::GLOBAL &&
                    fn.s == "torch._utils._rebuild_tensor_v2") {
                    const auto& a = args.tup;
                    const PVal& st = a[0];
                    TORCH_CHECK(st.kind == PVal::PERSID, "bad storage arg");
                    std::string cls = st.tup[0].s;          // torch.XStorage
                    const DType& dt = dtype_by_name(cls.substr(cls.find('.') + 1));
                    auto [ptr, nbytes] = data_of(st.s);
                    int64_t offset = a[1].i;
                    std::vector<int64_t> sizes, strides;
                    for (const auto& v : a[2].tup) sizes.push_back(v.i);
                    for (const auto& v : a[3].tup) strides.push_back(v.i);
                    // nbytes == 0: structural pass over a legacy container
                    // (layout not known yet) — build a zero storage of the
                    // persistent id's element count instead.
                    const int64_t sn =
                        nbytes ? (int64_t)(nbytes / dt.esize) : st.i;
                    torch::Tensor storage = torch::zeros(
                        {sn}, torch::TensorOptions().dtype(dt.st));
                    if (nbytes) memcpy(storage.data_ptr(), ptr, nbytes);
                    PVal out; out.kind = PVal::TENSOR;
                    out.ten = storage.as_strided(sizes, strides, offset)
                                  .contiguous();
                    stack.push_back(out);
                } else {
                    // e.g. collections.OrderedDict(()) -> empty dict
                    PVal d; d.kind = PVal::ODICT; stack.push_back(d);
                }
                break;
            }
            case 's': {                                    // SETITEM
                PVal v = stack.back(); stack.pop_back();
                PVal k = stack.back(); stack.pop_back();
                TORCH_CHECK(stack.back().kind == PVal::ODICT, "SETITEM not on dict");
                if (v.kind == PVal::TENSOR) result.emplace_back(k.s, v.ten);
                break;
            }
            case 'u': {                                    // SETITEMS
                size_t m = marks.back(); marks.pop_back();
                for (size_t k = m + 1; k + 1 < stack.size() + 0; k += 2) {
                    const PVal& key = stack[k];
                    const PVal& val = stack[k + 1];
                    if (val.kind == PVal::TENSOR)
                        result.emplace_back(key.s, val.ten);
                }
                stack.resize(m);
                TORCH_CHECK(!stack.empty() && stack.back().kind == PVal::ODICT,
                            "SETITEMS not on dict");
                break;
            }
            case 'b': {                                    // BUILD
                // attaches state (e.g. OrderedDict._metadata) to the
                // object below; the state never holds tensors we need.
                stack.pop_back();
                break;
            }
            case '.':
                if (end_pos) *end_pos = i;
                if (keys_out && !stack.empty())
                    for (const auto& it : stack.back().tup)
                        keys_out->push_back(it.s);
                return result;
            case '2': break;                               // DUP (unused)
            case 'G': { need(8); i += 8;                   // BINFLOAT
                        PVal v; v.kind = PVal::INT; stack.push_back(v); break; }
            case 0x95: { need(8); i += 8; break; }         // FRAME
            default:
                {
                    char hx[8];
                    snprintf(hx, sizeof hx, "%02x", (int)op);
                    TORCH_CHECK(false, "unsupported pickle opcode 0x", hx);
                }
        }
    }
}

std::vector<std::pair<std::string, torch::Tensor>> load_state_dict(
        const std::string& path) {
    std::ifstream f(path, std::ios::binary | std::ios::ate);
    TORCH_CHECK(f.good(), "cannot open ", path);
    size_t n = (size_t)f.tellg();
    std::vector<uint8_t> buf(n);
    f.seekg(0);
    f.read((char*)buf.data(), n);

    if (n >= 4 && buf[0] == 'P' && buf[1] == 'K') {
        ZipReader z(path);
        auto [pkl, pkl_n] = z.get("data.pkl");
        return run_pickle(pkl, pkl_n, [&](const std::string& key) {
            return z.get("data/" + key);
        });
    }

    // Legacy (pre-zip) torch container, as shipped in the reference
    // model_params.pt: magic pickle, protocol pickle, sys_info pickle,
    // the object pickle, the storage-key list pickle, then per key an
    // int64 numel followed by the raw storage bytes.
    TORCH_CHECK(n >= 2 && buf[0] == 0x80, "not a torch checkpoint: ", path);
    auto none_data = [](const std::string&)
        -> std::pair<const uint8_t*, size_t> {
        TORCH_CHECK(false, "storage requested during structural pass");
    };
    size_t pos = 0, next = 0;
    run_pickle(buf.data() + pos, n - pos, none_data, &next);  // magic
    pos += next;
    run_pickle(buf.data() + pos, n - pos, none_data, &next);  // protocol
    pos += next;
    run_pickle(buf.data() + pos, n - pos, none_data, &next);  // sys_info
    pos += next;
    const size_t obj_pos = pos;
    {   // structural pass over the object pickle just to find its end;
        // storages resolve to empty (the VM builds zero tensors then)
        auto zero_data = [&](const std::string&)
            -> std::pair<const uint8_t*, size_t> {
            return {nullptr, 0};
        };
        run_pickle(buf.data() + pos, n - pos, zero_data, &next);
        pos += next;
    }
    std::vector<std::string> keys;
    run_pickle(buf.data() + pos, n - pos, none_data, &next, &keys);
    pos += next;

    // Data section: per key (in key-list order) an int64 element count,
    // then numel * esize raw bytes. The element size is uniform per file
    // in practice; find the E in {4, 8, 2, 1} whose walk lands exactly on
    // EOF.
    const size_t data_pos = pos;
    std::map<std::string, std::pair<size_t, int64_t>> slots;  // pos, numel
    size_t E_found = 0;
    for (size_t E : {4, 8, 2, 1}) {
        size_t p2 = data_pos;
        std::map<std::string, std::pair<size_t, int64_t>> trial;
        bool ok = true;
        for (const auto& k : keys) {
            if (p2 + 8 > n) { ok = false; break; }
            int64_t numel;
            memcpy(&numel, buf.data() + p2, 8);
            p2 += 8;
            if (numel < 0 || p2 + (size_t)numel * E > n) { ok = false; break; }
            trial[k] = {p2, numel};
            p2 += (size_t)numel * E;
        }
        if (ok && p2 == n) { slots = trial; E_found = E; break; }
    }
    TORCH_CHECK(E_found, "legacy storage layout not understood");
    auto data_of2 = [&](const std::string& key)
        -> std::pair<const uint8_t*, size_t> {
        auto it = slots.find(key);
        TORCH_CHECK(it != slots.end(), "unknown storage key ", key);
        return {buf.data() + it->second.first,
                (size_t)it->second.second * E_found};
    };
    return run_pickle(buf.data() + obj_pos, n - obj_pos, data_of2);
}

}  // namespace fmda_ckpt
