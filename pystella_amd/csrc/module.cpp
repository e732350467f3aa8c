// pystella_amd._C — native runtime for MI355X (gfx950).
//
// Exposes:
//  * AOT hand-written CDNA4 kernels (derivs.hip — stencil family);
//  * an hiprtc JIT for the expression-specialized kernel templates
//    (fused RK stages / elementwise maps / reductions / histograms):
//    Python splices user physics expressions into hand-written HIP
//    templates and compiles them here, with an on-disk code cache.
//
// The module is deliberately torch-free: Python passes raw device
// pointers (tensor.data_ptr()) and the current HIP stream
// (torch.cuda.current_stream().cuda_stream), so the extension builds
// with plain hipcc in seconds and has no ABI coupling to torch.

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <hip/hip_runtime.h>
#include <hip/hiprtc.h>

#include <cstring>
#include <fstream>
#include <sstream>
#include <stdexcept>
#include <string>
#include <unordered_map>
#include <vector>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                   \
    do {                                                                  \
        hipError_t _e = (expr);                                           \
        if (_e != hipSuccess)                                             \
            throw std::runtime_error(std::string("HIP error: ") +         \
                                     hipGetErrorString(_e) + " at " +     \
                                     __FILE__ + ":" +                     \
                                     std::to_string(__LINE__));           \
    } while (0)

// ---------------------------------------------------------------------------
// AOT kernels (derivs.hip); dtype: 0 = fp64, 1 = fp32
extern "C" int pystella_gradlap(const void *, void *, void *, void *,
                                void *, long long, int, int, int, int, int,
                                double, double, double, int, void *);
extern "C" int pystella_pd(const void *, void *, int, int, int, int, int,
                           int, int, double, int, void *);

static void check_knl(int err, const char *what)
{
    if (err == 2)
        throw std::runtime_error(std::string(what) +
                                 ": unsupported dtype (fp64/fp32 only)");
    if (err != 0)
        throw std::runtime_error(std::string(what) + " launch failed: " +
                                 hipGetErrorString((hipError_t)err));
}

static void gradlap(uintptr_t f, uintptr_t lap, uintptr_t pdx, uintptr_t pdy,
                    uintptr_t pdz, int64_t g_fstride, int h, int nx, int ny,
                    int nz, int nf, double dx, double dy, double dz,
                    int dtype, uintptr_t stream)
{
    check_knl(pystella_gradlap((const void *)f, (void *)lap,
                               (void *)pdx, (void *)pdy, (void *)pdz,
                               (long long)g_fstride, h, nx, ny, nz, nf,
                               dx, dy, dz, dtype, (void *)stream),
              "gradlap");
}

static void pd(uintptr_t f, uintptr_t out, int h, int axis, int accum,
               int nx, int ny, int nz, int nf, double d, int dtype,
               uintptr_t stream)
{
    check_knl(pystella_pd((const void *)f, (void *)out, h, axis, accum,
                          nx, ny, nz, nf, d, dtype, (void *)stream),
              "pd");
}

// ---------------------------------------------------------------------------
// hiprtc JIT with in-memory + on-disk cache

struct JitKernel {
    hipModule_t module = nullptr;
    hipFunction_t fn = nullptr;
};

static std::unordered_map<std::string, JitKernel> g_kernels;
static std::string g_cache_dir;

static std::string arch_name()
{
    hipDeviceProp_t props;
    int dev = 0;
    HIP_CHECK(hipGetDevice(&dev));
    HIP_CHECK(hipGetDeviceProperties(&props, dev));
    std::string arch = props.gcnArchName;
    auto colon = arch.find(':');
    if (colon != std::string::npos) arch = arch.substr(0, colon);
    return arch;
}

static std::string fnv1a(const std::string &s)
{
    uint64_t h = 1469598103934665603ull;
    for (unsigned char c : s) {
        h ^= c;
        h *= 1099511628211ull;
    }
    char buf[32];
    snprintf(buf, sizeof buf, "%016llx", (unsigned long long)h);
    return buf;
}

static std::vector<char> compile_to_code(const std::string &src,
                                         const std::string &name)
{
    hiprtcProgram prog;
    if (hiprtcCreateProgram(&prog, src.c_str(), (name + ".hip").c_str(), 0,
                            nullptr, nullptr) != HIPRTC_SUCCESS)
        throw std::runtime_error("hiprtcCreateProgram failed");

    std::string arch_opt = "--offload-arch=" + arch_name();
    std::vector<const char *> opts = {arch_opt.c_str(), "-O3",
                                      "-std=c++17", "-ffp-contract=fast"};
    hiprtcResult res = hiprtcCompileProgram(prog, (int)opts.size(),
                                            opts.data());
    size_t log_size = 0;
    hiprtcGetProgramLogSize(prog, &log_size);
    std::string log(log_size, '\0');
    if (log_size > 1) hiprtcGetProgramLog(prog, log.data());
    if (res != HIPRTC_SUCCESS) {
        hiprtcDestroyProgram(&prog);
        throw std::runtime_error("hiprtc compile of " + name +
                                 " failed:\n" + log + "\nsource:\n" + src);
    }
    size_t code_size = 0;
    hiprtcGetCodeSize(prog, &code_size);
    std::vector<char> code(code_size);
    hiprtcGetCode(prog, code.data());
    hiprtcDestroyProgram(&prog);
    return code;
}

static void set_cache_dir(const std::string &d) { g_cache_dir = d; }

// Compile (or load from cache) and register under `key`; returns key.
static std::string jit_compile(const std::string &src,
                               const std::string &name)
{
    const std::string key = name + "_" + fnv1a(src + arch_name());
    if (g_kernels.count(key)) return key;

    std::vector<char> code;
    std::string cache_file;
    if (!g_cache_dir.empty()) {
        cache_file = g_cache_dir + "/" + key + ".hsaco";
        std::ifstream in(cache_file, std::ios::binary);
        if (in) {
            code.assign(std::istreambuf_iterator<char>(in),
                        std::istreambuf_iterator<char>());
        }
    }
    if (code.empty()) {
        code = compile_to_code(src, name);
        if (!cache_file.empty()) {
            std::string tmp = cache_file + ".tmp." +
                              std::to_string((uintptr_t)&code);
            std::ofstream out(tmp, std::ios::binary);
            out.write(code.data(), (std::streamsize)code.size());
            out.close();
            std::rename(tmp.c_str(), cache_file.c_str());
        }
    }

    JitKernel k;
    HIP_CHECK(hipModuleLoadData(&k.module, code.data()));
    HIP_CHECK(hipModuleGetFunction(&k.fn, k.module, name.c_str()));
    g_kernels[key] = k;
    return key;
}

// Launch a JIT kernel.  Args are passed as three vectors in declaration
// order: device pointers first, then int32 scalars, then doubles — the
// Python side generates kernel signatures following this convention.
static void jit_launch(const std::string &key, int gx, int gy, int gz,
                       int bx, int by, int bz, int shmem, uintptr_t stream,
                       const std::vector<uintptr_t> &ptrs,
                       const std::vector<int64_t> &ints,
                       const std::vector<double> &doubles)
{
    auto it = g_kernels.find(key);
    if (it == g_kernels.end())
        throw std::runtime_error("unknown JIT kernel " + key);

    struct Slot { alignas(16) unsigned char data[16]; };
    std::vector<Slot> storage(ptrs.size() + ints.size() + doubles.size());
    std::vector<void *> params;
    size_t n = 0;
    for (uintptr_t p : ptrs) {
        std::memcpy(storage[n].data, &p, sizeof p);
        params.push_back(storage[n].data);
        n++;
    }
    for (int64_t v : ints) {
        int iv = (int)v;
        std::memcpy(storage[n].data, &iv, sizeof iv);
        params.push_back(storage[n].data);
        n++;
    }
    for (double d : doubles) {
        std::memcpy(storage[n].data, &d, sizeof d);
        params.push_back(storage[n].data);
        n++;
    }

    HIP_CHECK(hipModuleLaunchKernel(it->second.fn, gx, gy, gz, bx, by, bz,
                                    shmem, (hipStream_t)stream,
                                    params.data(), nullptr));
}

static std::string get_arch() { return arch_name(); }

static int device_count()
{
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
}

extern "C" void launch_tt_project_mfma(
    const double *hij, double *out, const double *kx, const double *ky,
    const double *kz, int ny, int nz, long vol, hipStream_t stream);

static void tt_project(uintptr_t hij, uintptr_t out, uintptr_t kx,
                       uintptr_t ky, uintptr_t kz, int ny, int nz,
                       long vol, uintptr_t stream)
{
    launch_tt_project_mfma(
        (const double *)hij, (double *)out, (const double *)kx,
        (const double *)ky, (const double *)kz, ny, nz, vol,
        (hipStream_t)stream);
    HIP_CHECK(hipGetLastError());
}

PYBIND11_MODULE(_C, m)
{
    m.doc() = "pystella_amd native runtime (gfx950)";
    m.def("gradlap", &gradlap, "fused gradient/Laplacian stencil");
    m.def("tt_project", &tt_project,
          "transverse-traceless projection via f64 MFMA");
    m.def("pd", &pd, "single-axis first derivative");
    m.def("jit_compile", &jit_compile, "compile HIP source via hiprtc");
    m.def("jit_launch", &jit_launch, "launch a JIT kernel");
    m.def("set_cache_dir", &set_cache_dir);
    m.def("arch", &get_arch);
    m.def("device_count", &device_count);
}
