// bifrost_amd: bfMap — the user-function JIT engine,
// re-built on hipRTC for gfx950.  Behaviour contract: reference
// src/map.cpp:110-605 semantics:
//   * elementwise surface: named arrays + scalars, numpy-style trailing
//     broadcast, .real/.imag access, extra_code;
//   * explicit axis-indexed form `c(i,j) = a(j,i)` with shape/axis_names:
//     per-arg accessor structs with embedded constant shapes/strides,
//     tail-aligned index broadcasting and Python-style single-wrap
//     negative indices (reference ArrayIndexer.cuh:66-82), the implicit
//     index vector `_` (IndexVec arithmetic, reference IndexArray.cuh)
//     and `.shape()`;
//   * packed sub-byte dtypes (ci4 etc.) are not supported in map.
//
// Codegen strategy: computation shape and every argument's (broadcast-
// aligned) shape/strides are embedded as compile-time constants, so index
// math folds to shifts/mads; kernels are cached in-process by source hash
// (BF_MAP_KERNEL_CACHE_SIZE entries, LRU-ish).

#include <bifrost/map.h>

#include <hip/hip_runtime.h>
#include <hip/hiprtc.h>

#include <cctype>
#include <cstdio>
#include <sys/stat.h>
#include <cstring>
#include <list>
#include <memory>
#include <mutex>
#include <sstream>
#include <string>
#include <unordered_map>
#include <vector>

#include "dtype.hpp"
#include "hipctx.hpp"
#include "status.hpp"

namespace {

using bfamd::StatusError;

const char* kPrelude = R"(
typedef signed char      i8;
typedef short            i16;
typedef int              i32;
typedef long long        i64;
typedef unsigned char    u8;
typedef unsigned short   u16;
typedef unsigned int     u32;
typedef unsigned long long u64;
typedef float            f32;
typedef double           f64;

template<typename T>
struct Complex {
    T real, imag;
    __device__ Complex() : real(0), imag(0) {}
    __device__ Complex(T r, T i = 0) : real(r), imag(i) {}
    template<typename U>
    __device__ Complex(const Complex<U>& o)
        : real((T)o.real), imag((T)o.imag) {}
    template<typename U>
    __device__ Complex& operator=(const Complex<U>& o) {
        real = (T)o.real; imag = (T)o.imag; return *this;
    }
    __device__ Complex& operator=(T v) { real = v; imag = 0; return *this; }
    __device__ Complex operator+(const Complex& o) const {
        return Complex(real + o.real, imag + o.imag);
    }
    __device__ Complex operator-(const Complex& o) const {
        return Complex(real - o.real, imag - o.imag);
    }
    __device__ Complex operator*(const Complex& o) const {
        return Complex(real * o.real - imag * o.imag,
                       real * o.imag + imag * o.real);
    }
    __device__ Complex operator*(T s) const {
        return Complex(real * s, imag * s);
    }
    __device__ Complex& operator+=(const Complex& o) {
        real += o.real; imag += o.imag; return *this;
    }
    __device__ Complex& operator-=(const Complex& o) {
        real -= o.real; imag -= o.imag; return *this;
    }
    __device__ Complex& operator*=(const Complex& o) {
        T r = real * o.real - imag * o.imag;
        imag = real * o.imag + imag * o.real;
        real = r; return *this;
    }
    __device__ T mag2() const { return real * real + imag * imag; }
    __device__ Complex conj() const { return Complex(real, -imag); }
    __device__ Complex& assign(T r, T i) {
        real = r; imag = i; return *this;
    }
};
template<typename T>
__device__ Complex<T> operator*(T s, const Complex<T>& c) {
    return Complex<T>(s * c.real, s * c.imag);
}
template<typename T>
__device__ Complex<T> operator*(int s, const Complex<T>& c) {
    return Complex<T>((T)s * c.real, (T)s * c.imag);
}
template<typename T>
__device__ Complex<T> conj(const Complex<T>& c) {
    return Complex<T>(c.real, -c.imag);
}
// Packed 4+4-bit complex: ONE byte per element (re in the HIGH nibble,
// the ci4 convention of src/Complex.hpp:149-168), so it is byte-addressed
// like every other dtype.  Participates in map expressions as a value
// convertible to/from Complex<T>; direct .real/.imag member access is not
// available on the packed form (use cf32_t(a).real etc.).
struct ci4_t {
    signed char b;
    __device__ signed char real_get() const { return (signed char)(b >> 4); }
    __device__ signed char imag_get() const {
        return (signed char)((signed char)(b << 4) >> 4);
    }
    template<typename U>
    __device__ operator Complex<U>() const {
        return Complex<U>((U)real_get(), (U)imag_get());
    }
    template<typename U>
    __device__ ci4_t& operator=(const Complex<U>& o) {
        int re = (int)o.real, im = (int)o.imag;
        b = (signed char)(((re & 0xF) << 4) | (im & 0xF));
        return *this;
    }
    __device__ ci4_t& assign(int r, int i) {
        b = (signed char)(((r & 0xF) << 4) | (i & 0xF));
        return *this;
    }
    __device__ Complex<signed char> conj() const {
        return Complex<signed char>(real_get(), (signed char)-imag_get());
    }
    __device__ int mag2() const {
        int r = real_get(), i = imag_get();
        return r * r + i * i;
    }
};
#define CI4_BINOP(op)                                                    \
template<typename U>                                                     \
__device__ Complex<U> operator op(const ci4_t& a, const Complex<U>& b) { \
    return Complex<U>((U)a.real_get(), (U)a.imag_get()) op b;            \
}                                                                        \
template<typename U>                                                     \
__device__ Complex<U> operator op(const Complex<U>& a, const ci4_t& b) { \
    return a op Complex<U>((U)b.real_get(), (U)b.imag_get());            \
}
CI4_BINOP(+)
CI4_BINOP(-)
CI4_BINOP(*)
__device__ inline Complex<float> operator*(const ci4_t& a, float s) {
    return Complex<float>(a.real_get() * s, a.imag_get() * s);
}
__device__ inline Complex<float> operator*(float s, const ci4_t& a) {
    return a * s;
}
__device__ inline Complex<signed char> conj(const ci4_t& a) {
    return a.conj();
}
typedef Complex<signed char> ci8_t;
typedef Complex<short>       ci16_t;
typedef Complex<int>         ci32_t;
typedef Complex<long long>   ci64_t;
typedef Complex<float>       cf32_t;
typedef Complex<double>      cf64_t;

template<typename T, int N> struct Vec {
    T v[N];
    typedef T value_type;
    __device__ Vec() {}
    template<typename... As>
    __device__ Vec(As... as) : v{T(as)...} {}
    __device__ T& operator[](int i) { return v[i]; }
    __device__ const T& operator[](int i) const { return v[i]; }
};

template<int N> struct IndexVec {
    long v[N];
    __device__ long operator[](int i) const { return v[i]; }
};
#define IV_OP(op)                                                        \
template<int N>                                                          \
__device__ IndexVec<N> operator op(const IndexVec<N>& a,                 \
                                   const IndexVec<N>& b) {               \
    IndexVec<N> r;                                                       \
    for (int i = 0; i < N; ++i) r.v[i] = a.v[i] op b.v[i];               \
    return r;                                                            \
}                                                                        \
template<int N>                                                          \
__device__ IndexVec<N> operator op(const IndexVec<N>& a, long b) {       \
    IndexVec<N> r;                                                       \
    for (int i = 0; i < N; ++i) r.v[i] = a.v[i] op b;                    \
    return r;                                                            \
}
IV_OP(+)
IV_OP(-)
IV_OP(*)
IV_OP(/)
IV_OP(%)

template<typename S>
__device__ inline void iv_put(long* b, int& n, S x) { b[n++] = (long)x; }
template<int K>
__device__ inline void iv_put(long* b, int& n, const IndexVec<K>& x) {
    for (int i = 0; i < K; ++i) b[n++] = x.v[i];
}
__device__ inline void iv_puts(long*, int&) {}
template<typename T0, typename... Ts>
__device__ inline void iv_puts(long* b, int& n, T0 a, Ts... rest) {
    iv_put(b, n, a);
    iv_puts(b, n, rest...);
}
)";

std::string dtype_ctype(BFdtype dt) {
    bool cplx = dt & BF_DTYPE_COMPLEX_BIT;
    int nbit = dt & BF_DTYPE_NBIT_BITS;
    int type = dt & BF_DTYPE_TYPE_BITS;
    int veclen = ((dt & BF_DTYPE_VECTOR_BITS) >> BF_DTYPE_VECTOR_BIT0) + 1;
    std::ostringstream os;
    if (cplx) {
        if (type == BF_DTYPE_FLOAT_TYPE) os << "cf" << nbit << "_t";
        else os << "ci" << nbit << "_t";
    } else {
        if (type == BF_DTYPE_FLOAT_TYPE) os << "f" << nbit;
        else if (type == BF_DTYPE_UINT_TYPE) os << "u" << nbit;
        else os << "i" << nbit;
    }
    if (veclen > 1) return "Vec<" + os.str() + "," +
                           std::to_string(veclen) + ">";
    return os.str();
}

bool dtype_supported(BFdtype dt) {
    int nbit = dt & BF_DTYPE_NBIT_BITS;
    bool cplx = dt & BF_DTYPE_COMPLEX_BIT;
    // ci4 packs 4+4 bits in ONE byte per element — byte-addressable, so
    // map supports it (as the reference does via Complex<FourBit>).
    // Plain i4/u4 and narrower (2+ elements per byte) stay unsupported.
    if (nbit == 4 && cplx) {
        return (dt & BF_DTYPE_TYPE_BITS) == BF_DTYPE_INT_TYPE;
    }
    if (nbit < 8) return false;  // true sub-byte not supported in map
    int type = dt & BF_DTYPE_TYPE_BITS;
    if (type != BF_DTYPE_INT_TYPE && type != BF_DTYPE_UINT_TYPE &&
        type != BF_DTYPE_FLOAT_TYPE)
        return false;
    if (type == BF_DTYPE_FLOAT_TYPE && nbit == 16) return false;  // no f16
    return true;
}

// Entries are reference-counted: a thread launching a kernel holds a
// shared_ptr copied out under the lock, so eviction (or bfMapClearCache)
// from another thread cannot hipModuleUnload a module with a launch in
// flight — the unload happens in the destructor when the last user drops.
struct CacheEntry {
    hipModule_t module = nullptr;
    hipFunction_t func = nullptr;
    CacheEntry() = default;
    CacheEntry(const CacheEntry&) = delete;
    CacheEntry& operator=(const CacheEntry&) = delete;
    ~CacheEntry() {
        if (module) (void)hipModuleUnload(module);
    }
};
using CacheRef = std::shared_ptr<CacheEntry>;

std::mutex g_cache_mutex;
std::unordered_map<std::string, CacheRef> g_cache;
std::list<std::string> g_cache_order;

void cache_evict_locked() {
    while (g_cache.size() > BF_MAP_KERNEL_CACHE_SIZE && !g_cache_order.empty()) {
        auto key = g_cache_order.front();
        g_cache_order.pop_front();
        g_cache.erase(key);  // module unloads when the last ref drops
    }
}

// ---- on-disk code cache ----------------------------------------------------
// Like the reference's ~/.bifrost map kernel disk cache: compiled code
// objects keyed by an FNV-1a hash of the generated source, so cold
// processes skip hipRTC entirely.  Disable with BIFROST_NO_DISK_CACHE=1.

std::string disk_cache_dir() {
    static std::string dir = [] {
        if (getenv("BIFROST_NO_DISK_CACHE")) return std::string();
        const char* home = getenv("HOME");
        if (!home || !*home) return std::string();
        std::string d = std::string(home) + "/.bifrost_amd";
        mkdir(d.c_str(), 0755);
        d += "/mapcache_gfx950";
        mkdir(d.c_str(), 0755);
        return d;
    }();
    return dir;
}

std::string src_hash_hex(const std::string& src) {
    unsigned long long h = 1469598103934665603ULL;  // FNV-1a 64
    for (unsigned char c : src) {
        h ^= c;
        h *= 1099511628211ULL;
    }
    char buf[20];
    std::snprintf(buf, sizeof(buf), "%016llx", h);
    return buf;
}

bool disk_cache_load(const std::string& src, std::vector<char>* code) {
    std::string dir = disk_cache_dir();
    if (dir.empty()) return false;
    std::string path = dir + "/" + src_hash_hex(src) + ".hsaco";
    FILE* f = std::fopen(path.c_str(), "rb");
    if (!f) return false;
    std::fseek(f, 0, SEEK_END);
    long n = std::ftell(f);
    std::fseek(f, 0, SEEK_SET);
    if (n <= 0) { std::fclose(f); return false; }
    code->resize((size_t)n);
    bool ok = std::fread(code->data(), 1, (size_t)n, f) == (size_t)n;
    std::fclose(f);
    return ok;
}

void disk_cache_store(const std::string& src,
                      const std::vector<char>& code) {
    std::string dir = disk_cache_dir();
    if (dir.empty()) return;
    std::string path = dir + "/" + src_hash_hex(src) + ".hsaco";
    std::string tmp = path + ".tmp";
    FILE* f = std::fopen(tmp.c_str(), "wb");
    if (!f) return;
    bool ok = std::fwrite(code.data(), 1, code.size(), f) == code.size();
    std::fclose(f);
    if (ok) std::rename(tmp.c_str(), path.c_str());
    else std::remove(tmp.c_str());
}

BFstatus load_module(const std::vector<char>& code, const std::string& src,
                     CacheRef* out) {
    auto e = std::make_shared<CacheEntry>();
    if (hipModuleLoadData(&e->module, code.data()) != hipSuccess) {
        e->module = nullptr;
        return BF_STATUS_DEVICE_ERROR;
    }
    if (hipModuleGetFunction(&e->func, e->module, "bfmap_kernel") !=
        hipSuccess) {
        return BF_STATUS_DEVICE_ERROR;  // dtor unloads
    }
    std::lock_guard<std::mutex> lk(g_cache_mutex);
    g_cache[src] = e;
    g_cache_order.push_back(src);
    cache_evict_locked();
    *out = e;
    return BF_STATUS_SUCCESS;
}

BFstatus compile_and_cache(const std::string& src, CacheRef* out) {
    {
        std::lock_guard<std::mutex> lk(g_cache_mutex);
        auto it = g_cache.find(src);
        if (it != g_cache.end()) {
            *out = it->second;
            return BF_STATUS_SUCCESS;
        }
    }
    {
        std::vector<char> code;
        if (disk_cache_load(src, &code) &&
            load_module(code, src, out) == BF_STATUS_SUCCESS)
            return BF_STATUS_SUCCESS;
    }
    hiprtcProgram prog;
    if (hiprtcCreateProgram(&prog, src.c_str(), "bfmap.hip", 0, nullptr,
                            nullptr) != HIPRTC_SUCCESS)
        return BF_STATUS_INTERNAL_ERROR;
    const char* opts[] = {"--offload-arch=gfx950", "-O3", "-std=c++17"};
    hiprtcResult cres = hiprtcCompileProgram(prog, 3, opts);
    if (cres != HIPRTC_SUCCESS) {
        if (bfamd::debug_enabled()) {
            size_t lsz = 0;
            hiprtcGetProgramLogSize(prog, &lsz);
            std::string log(lsz, '\0');
            hiprtcGetProgramLog(prog, &log[0]);
            std::fprintf(stderr, "[bifrost_amd] bfMap compile log:\n%s\n",
                         log.c_str());
        }
        hiprtcDestroyProgram(&prog);
        return BF_STATUS_INVALID_ARGUMENT;
    }
    size_t code_size = 0;
    hiprtcGetCodeSize(prog, &code_size);
    std::vector<char> code(code_size);
    hiprtcGetCode(prog, code.data());
    hiprtcDestroyProgram(&prog);

    disk_cache_store(src, code);
    return load_module(code, src, out);
}

}  // namespace

extern "C" BFstatus bfMap(int ndim, long const* shape,
                          char const* const* axis_names, int narg,
                          BFarray const* const* args,
                          char const* const* arg_names,
                          char const* func_name, char const* func,
                          char const* extra_code, int const* block_shape,
                          int const* block_axes) {
    using namespace bfamd;
    BF_ASSERT(func && args && arg_names, BF_STATUS_INVALID_POINTER);
    BF_ASSERT(narg >= 1 && narg <= 16, BF_STATUS_INVALID_ARGUMENT);
    (void)func_name;
    (void)block_shape;
    (void)block_axes;

    // Which args are used in the explicit indexed form `name(...)` /
    // `name.shape()`?  Those get accessor structs; plain uses get a bound
    // reference (the elementwise path, constant-folded offsets).
    bool indexed[16] = {false};
    bool any_indexed = false;
    // A match only counts with a non-identifier character on the LEFT (an
    // arg named 'a' must not match inside "data(..)"), mirroring the bare
    // '_' word-boundary check below.
    auto find_word = [&](const char* hay, const std::string& pat) {
        for (const char* p = std::strstr(hay, pat.c_str()); p;
             p = std::strstr(p + 1, pat.c_str())) {
            if (p == hay ||
                (!isalnum((unsigned char)p[-1]) && p[-1] != '_'))
                return true;
        }
        return false;
    };
    for (int a = 0; a < narg; ++a) {
        std::string pat1 = std::string(arg_names[a]) + "(";
        std::string pat2 = std::string(arg_names[a]) + ".shape";
        std::string pat3 = std::string(arg_names[a]) + " (";
        if (find_word(func, pat1) || find_word(func, pat2) ||
            find_word(func, pat3)) {
            indexed[a] = true;
            any_indexed = true;
        }
    }
    // bare `_` (the implicit index vector)?
    bool uses_underscore = false;
    for (const char* p = std::strchr(func, '_'); p;
         p = std::strchr(p + 1, '_')) {
        bool lok = (p == func) ||
                   (!isalnum((unsigned char)p[-1]) && p[-1] != '_');
        bool rok = !isalnum((unsigned char)p[1]) && p[1] != '_';
        if (lok && rok) { uses_underscore = true; break; }
    }

    // Computation shape: explicit, or broadcast of the args.
    long cshape[BF_MAX_DIMS];
    int cndim = 0;
    if (shape && ndim > 0) {
        cndim = ndim;
        for (int d = 0; d < ndim; ++d) cshape[d] = shape[d];
    } else {
        for (int a = 0; a < narg; ++a)
            cndim = std::max(cndim, args[a]->ndim);
        for (int d = 0; d < cndim; ++d) cshape[d] = 1;
        for (int a = 0; a < narg; ++a) {
            int off = cndim - args[a]->ndim;
            for (int d = 0; d < args[a]->ndim; ++d) {
                long ext = args[a]->shape[d];
                if (ext != 1) {
                    BF_ASSERT(cshape[off + d] == 1 || cshape[off + d] == ext,
                              BF_STATUS_INVALID_SHAPE);
                    cshape[off + d] = ext;
                }
            }
        }
        if (cndim == 0) { cndim = 1; cshape[0] = 1; }
    }
    long n = 1;
    for (int d = 0; d < cndim; ++d) n *= cshape[d];

    // All arrays must be device-accessible (GPU-only op, as the reference).
    for (int a = 0; a < narg; ++a) {
        BF_ASSERT(space_device_accessible(args[a]->space),
                  BF_STATUS_UNSUPPORTED_SPACE);
        BF_ASSERT(dtype_supported(args[a]->dtype), BF_STATUS_UNSUPPORTED_DTYPE);
    }

    // ---- generate source ----
    std::ostringstream os;
    os << kPrelude;
    if (extra_code) os << extra_code << "\n";
    // accessor structs for indexed args: shapes/strides as literals;
    // tail-aligned index broadcast with single-wrap negative indices
    // (reference ArrayIndexer.cuh:66-82)
    for (int a = 0; a < narg; ++a) {
        if (!indexed[a]) continue;
        const BFarray* arr = args[a];
        std::string ct = dtype_ctype(arr->dtype);
        int andim = arr->ndim;
        os << "struct Acc_" << arg_names[a] << " {\n";
        os << "  char* p;\n";
        os << "  __device__ static long shp(int d) { return ";
        for (int d = 0; d < andim; ++d)
            os << "d==" << d << " ? " << arr->shape[d] << "L : ";
        os << "1L; }\n";
        os << "  __device__ static long str(int d) { return ";
        for (int d = 0; d < andim; ++d)
            os << "d==" << d << " ? " << arr->strides[d] << "L : ";
        os << "0L; }\n";
        os << "  __device__ static long shape(int d) { return shp(d); }\n";
        os << "  __device__ IndexVec<" << andim << "> shape() const {\n";
        os << "    IndexVec<" << andim << "> r = {{";
        for (int d = 0; d < andim; ++d)
            os << (d ? ", " : "") << arr->shape[d] << "L";
        os << "}}; return r; }\n";
        os << "  __device__ " << ct << "& ref(const long* ind, int nind) "
              "const {\n";
        os << "    long off = 0;\n";
        os << "    int lead = nind - " << andim << "; "
              "if (lead < 0) lead = 0;\n";
        os << "    int nd = nind < " << andim << " ? nind : " << andim
           << ";\n";
        os << "    for (int d = 0; d < nd; ++d) {\n";
        os << "      long i2 = ind[d + lead];\n";
        os << "      i2 += (i2 < 0) * shp(d);\n";
        os << "      off += (shp(d) != 1) * i2 * str(d);\n";
        os << "    }\n";
        os << "    return *(" << ct << "*)(p + off);\n";
        os << "  }\n";
        os << "  template<typename... Is>\n";
        os << "  __device__ " << ct << "& operator()(Is... is) const {\n";
        os << "    long buf[16]; int n2 = 0; iv_puts(buf, n2, is...);\n";
        os << "    return ref(buf, n2);\n";
        os << "  }\n";
        os << "};\n";
    }
    // <name>_type typedefs (the reference exposes each arg's element
    // type under this name, e.g. Complex<b_type>)
    for (int a = 0; a < narg; ++a) {
        os << "typedef " << dtype_ctype(args[a]->dtype) << " "
           << arg_names[a] << "_type;\n";
    }
    os << "extern \"C\" __global__ void bfmap_kernel(";
    for (int a = 0; a < narg; ++a) {
        os << (a ? ", " : "") << "char* __restrict__ _bf_p" << a;
    }
    os << ") {\n";
    os << "  const long _bf_N = " << n << "L;\n";
    os << "  for (long _bf_i = (long)blockIdx.x * blockDim.x + threadIdx.x;\n"
          "       _bf_i < _bf_N; _bf_i += (long)gridDim.x * blockDim.x) {\n";
    // decompose i (row-major over cshape) and compute per-arg offsets
    os << "    long _bf_rem = _bf_i;\n";
    for (int d = cndim - 1; d >= 0; --d) {
        os << "    long _bf_idx" << d << " = _bf_rem % " << cshape[d]
           << "L; _bf_rem /= " << cshape[d] << "L;\n";
    }
    // named axes (explicit-shape indexed form)
    if (axis_names) {
        for (int d = 0; d < cndim; ++d) {
            if (axis_names[d] && axis_names[d][0])
                os << "    long " << axis_names[d] << " = _bf_idx" << d
                   << ";\n";
        }
    }
    if (uses_underscore || any_indexed) {
        os << "    IndexVec<" << cndim << "> _ = {{";
        for (int d = 0; d < cndim; ++d) os << (d ? ", " : "") << "_bf_idx" << d;
        os << "}};\n    (void)_;\n";
    }
    for (int a = 0; a < narg; ++a) {
        const BFarray* arr = args[a];
        if (indexed[a]) {
            os << "    Acc_" << arg_names[a] << " " << arg_names[a]
               << "{_bf_p" << a << "};\n";
            continue;
        }
        int off = cndim - arr->ndim;
        os << "    long _bf_off" << a << " = 0";
        for (int d = 0; d < arr->ndim; ++d) {
            long ext = arr->shape[d];
            long strd = arr->strides[d];
            if (ext != 1 && strd != 0 && off + d >= 0) {
                os << " + _bf_idx" << (off + d) << " * " << strd << "L";
            }
        }
        os << ";\n";
        std::string ct = dtype_ctype(arr->dtype);
        os << "    " << ct << "& " << arg_names[a] << " = *(" << ct
           << "*)(_bf_p" << a << " + _bf_off" << a << ");\n";
    }
    os << "    " << func << ";\n";
    os << "  }\n}\n";

    CacheRef entry;
    BF_CHECK(compile_and_cache(os.str(), &entry));

    // ---- launch ----
    void* ptrs[16];
    void* kargs[16];
    for (int a = 0; a < narg; ++a) {
        ptrs[a] = args[a]->data;
        kargs[a] = &ptrs[a];
    }
    unsigned blocks = (unsigned)std::min<long>((n + 255) / 256, 32768L);
    if (blocks == 0) blocks = 1;
    hipError_t err = hipModuleLaunchKernel(
        entry->func, blocks, 1, 1, 256, 1, 1, 0, bfamd::thread_stream(),
        kargs, nullptr);
    BF_CHECK_HIP(err);
    return BF_STATUS_SUCCESS;
}

extern "C" BFstatus bfMapClearCache() {
    std::lock_guard<std::mutex> lk(g_cache_mutex);
    g_cache.clear();  // each module unloads when its last user drops
    g_cache_order.clear();
    return BF_STATUS_SUCCESS;
}
