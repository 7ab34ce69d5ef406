// _healthprobe — gfx950 GPU self-test kernel.
//
// The plugin's GPU self-test for nodes where a GPU is
// host-driver-resident (SR-IOV PFs): an HBM write/readback at bandwidth
// plus an MFMA-unit exercise confirms the silicon answers.  Invoked by
// operators (`make test-gpu`) and by the driver's smoke(); the daemon's
// in-band health signals stay event-driven (vfio nodes + AMD-SMI).
// The reference has no GPU compute at all (SURVEY.md §2 — its only
// native code is the NVML binding); this is the MI355X-native extra
// that makes "is this GPU healthy" answerable beyond "the /dev node
// exists".
//
// CDNA4 notes (per /opt/skills/guides/MI355X_MICROARCH.md): 64-wide
// wavefronts, 256 threads/block, grid-stride loops sized ≫256
// workgroups to cover all 8 XCDs; dwordx4 vector traffic for HBM3E.

#include <pybind11/pybind11.h>

#include <hip/hip_runtime.h>

#include <cstdint>
#include <stdexcept>
#include <string>

namespace py = pybind11;

#define HIP_CHECK(expr)                                                   \
  do {                                                                    \
    hipError_t _e = (expr);                                               \
    if (_e != hipSuccess)                                                 \
      throw std::runtime_error(std::string(#expr) + ": " +                \
                               hipGetErrorString(_e));                    \
  } while (0)

namespace {

// Pattern fill: each lane writes a value derived from its index so the
// verify pass can detect addressing faults, not just stuck bits.
// Non-temporal dwordx4 stores: the buffer is write-once-read-once, so
// bypassing L2 keeps the stream at HBM3E rate instead of thrashing the
// per-XCD L2.  Contiguous per-wavefront accesses (i = global thread id,
// stride = grid) coalesce into full 4 KiB bursts.
__global__ void fill_kernel(uint4 *__restrict__ out, size_t n_vec,
                            uint32_t seed) {
  size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  for (; i < n_vec; i += stride) {
    uint32_t base = seed ^ (uint32_t)(i * 2654435761u);
    uint4 v = make_uint4(base, base + 1, base + 2, base + 3);
    __builtin_nontemporal_store(v.x, &out[i].x);
    __builtin_nontemporal_store(v.y, &out[i].y);
    __builtin_nontemporal_store(v.z, &out[i].z);
    __builtin_nontemporal_store(v.w, &out[i].w);
  }
}

// Verify + bandwidth read pass: mismatches flip the error counter.
__global__ void verify_kernel(const uint4 *__restrict__ in, size_t n_vec,
                              uint32_t seed,
                              unsigned long long *__restrict__ errors) {
  size_t i = blockIdx.x * (size_t)blockDim.x + threadIdx.x;
  size_t stride = (size_t)gridDim.x * blockDim.x;
  unsigned long long local = 0;
  for (; i < n_vec; i += stride) {
    uint32_t base = seed ^ (uint32_t)(i * 2654435761u);
    uint4 v;
    v.x = __builtin_nontemporal_load(&in[i].x);
    v.y = __builtin_nontemporal_load(&in[i].y);
    v.z = __builtin_nontemporal_load(&in[i].z);
    v.w = __builtin_nontemporal_load(&in[i].w);
    local += (v.x != base) + (v.y != base + 1) + (v.z != base + 2) +
             (v.w != base + 3);
  }
  if (local) atomicAdd(errors, local);
}

// LDS exercise: every workgroup fills its 64 KB LDS slab with an
// index-derived pattern, cross-reads it lane-swizzled (hits all 32
// banks), and counts mismatches.  Launched with ≫256 workgroups so
// every CU's LDS gets touched.
__global__ void lds_kernel(unsigned long long *__restrict__ errors,
                           uint32_t seed) {
  __shared__ uint32_t slab[16384];  // 64 KB
  unsigned t = threadIdx.x;
  for (unsigned i = t; i < 16384; i += blockDim.x)
    slab[i] = seed ^ (i * 2246822519u) ^ blockIdx.x;
  __syncthreads();
  unsigned long long local = 0;
  for (unsigned i = t; i < 16384; i += blockDim.x) {
    unsigned j = (i * 33) & 16383;  // stride-33 → bank-sweeping reads
    if (slab[j] != (seed ^ (j * 2246822519u) ^ blockIdx.x)) ++local;
  }
  if (local) atomicAdd(errors, local);
}

// Device-scope atomics: every thread of every workgroup (spread across
// all 8 XCDs) increments one counter; the total must be exact — a
// quick cross-XCD coherency check.
__global__ void atomic_kernel(unsigned long long *__restrict__ counter) {
  atomicAdd(counter, 1ull);
}

// Minimal matrix-core exercise: one MFMA per wavefront, result checked
// on host.  Confirms the XCD compute path beyond plain VALU/HBM.
__global__ void mfma_kernel(float *__restrict__ out) {
#if defined(__gfx950__) || defined(__gfx942__) || defined(__gfx90a__)
  using f32x4 = __attribute__((__vector_size__(4 * sizeof(float)))) float;
  f32x4 acc = {0.f, 0.f, 0.f, 0.f};
  // C += A*B with A=B=1 on a 16x16x4 tile: every output element = 4.
  acc = __builtin_amdgcn_mfma_f32_16x16x4f32(1.0f, 1.0f, acc, 0, 0, 0);
  int lane = threadIdx.x & 63;
  if (blockIdx.x == 0 && threadIdx.x < 64) {
    out[lane * 4 + 0] = acc[0];
    out[lane * 4 + 1] = acc[1];
    out[lane * 4 + 2] = acc[2];
    out[lane * 4 + 3] = acc[3];
  }
#else
  if (blockIdx.x == 0 && threadIdx.x == 0) out[0] = -1.0f;
#endif
}

py::dict probe(int device, size_t mib, int blocks_per_cu) {
  py::dict result;
  int ndev = 0;
  HIP_CHECK(hipGetDeviceCount(&ndev));
  result["device_count"] = ndev;
  if (device >= ndev)
    throw std::runtime_error("device index out of range");
  HIP_CHECK(hipSetDevice(device));

  hipDeviceProp_t prop;
  HIP_CHECK(hipGetDeviceProperties(&prop, device));
  result["name"] = std::string(prop.name);
  result["gcn_arch"] = std::string(prop.gcnArchName);
  result["vram_gib"] =
      static_cast<double>(prop.totalGlobalMem) / (1 << 30);
  result["multi_processor_count"] = prop.multiProcessorCount;

  size_t bytes = mib << 20;
  size_t n_vec = bytes / sizeof(uint4);
  uint4 *buf = nullptr;
  HIP_CHECK(hipMalloc(&buf, bytes));
  unsigned long long *errors = nullptr;
  HIP_CHECK(hipMalloc(&errors, sizeof(unsigned long long)));
  HIP_CHECK(hipMemset(errors, 0, sizeof(unsigned long long)));

  // ≫256 workgroups so all 8 XCDs are covered.
  int blocks = prop.multiProcessorCount * blocks_per_cu;
  dim3 grid(blocks), block(256);
  uint32_t seed = 0xA31DBEEF;

  hipEvent_t t0, t1, t2;
  HIP_CHECK(hipEventCreate(&t0));
  HIP_CHECK(hipEventCreate(&t1));
  HIP_CHECK(hipEventCreate(&t2));

  // Kernel shape settled by an on-hardware sweep
  // (profiles/probe_sweep_r09/r10.log): the plain grid-stride dwordx4
  // stream is at the bandwidth envelope; per-thread unrolls either
  // break write coalescing (32 B lane stride) or stay within noise.
  auto launch_fill = [&] {
    hipLaunchKernelGGL(fill_kernel, grid, block, 0, 0, buf, n_vec,
                       seed);
  };
  auto launch_verify = [&] {
    hipLaunchKernelGGL(verify_kernel, grid, block, 0, 0, buf, n_vec,
                       seed, errors);
  };

  // warmup
  launch_fill();
  HIP_CHECK(hipDeviceSynchronize());

  HIP_CHECK(hipEventRecord(t0));
  launch_fill();
  HIP_CHECK(hipEventRecord(t1));
  launch_verify();
  HIP_CHECK(hipEventRecord(t2));
  HIP_CHECK(hipEventSynchronize(t2));

  float fill_ms = 0, verify_ms = 0;
  HIP_CHECK(hipEventElapsedTime(&fill_ms, t0, t1));
  HIP_CHECK(hipEventElapsedTime(&verify_ms, t1, t2));
  unsigned long long h_errors = 0;
  HIP_CHECK(hipMemcpy(&h_errors, errors, sizeof h_errors,
                      hipMemcpyDeviceToHost));

  // LDS banks on every CU
  HIP_CHECK(hipMemset(errors, 0, sizeof(unsigned long long)));
  hipLaunchKernelGGL(lds_kernel, grid, block, 0, 0, errors, seed);
  HIP_CHECK(hipDeviceSynchronize());
  unsigned long long lds_errors = 0;
  HIP_CHECK(hipMemcpy(&lds_errors, errors, sizeof lds_errors,
                      hipMemcpyDeviceToHost));

  // device-scope atomic coherency across XCDs
  unsigned long long *counter = nullptr;
  HIP_CHECK(hipMalloc(&counter, sizeof(unsigned long long)));
  HIP_CHECK(hipMemset(counter, 0, sizeof(unsigned long long)));
  hipLaunchKernelGGL(atomic_kernel, grid, block, 0, 0, counter);
  HIP_CHECK(hipDeviceSynchronize());
  unsigned long long h_counter = 0;
  HIP_CHECK(hipMemcpy(&h_counter, counter, sizeof h_counter,
                      hipMemcpyDeviceToHost));
  HIP_CHECK(hipFree(counter));
  bool atomics_ok =
      h_counter == (unsigned long long)blocks * block.x;

  float *mfma_out = nullptr;
  HIP_CHECK(hipMalloc(&mfma_out, 256 * sizeof(float)));
  HIP_CHECK(hipMemset(mfma_out, 0, 256 * sizeof(float)));
  hipLaunchKernelGGL(mfma_kernel, dim3(1), dim3(64), 0, 0, mfma_out);
  HIP_CHECK(hipDeviceSynchronize());
  float h_mfma[256];
  HIP_CHECK(hipMemcpy(h_mfma, mfma_out, sizeof h_mfma,
                      hipMemcpyDeviceToHost));
  bool mfma_ok = true;
  for (int i = 0; i < 256; ++i)
    if (h_mfma[i] != 4.0f) mfma_ok = false;

  HIP_CHECK(hipFree(mfma_out));
  HIP_CHECK(hipFree(errors));
  HIP_CHECK(hipFree(buf));
  (void)hipEventDestroy(t0);
  (void)hipEventDestroy(t1);
  (void)hipEventDestroy(t2);

  double gib = static_cast<double>(bytes) / (1 << 30);
  result["bytes"] = bytes;
  result["write_gbps"] = gib / (fill_ms / 1e3);
  result["read_gbps"] = gib / (verify_ms / 1e3);
  result["pattern_errors"] = h_errors;
  result["lds_errors"] = lds_errors;
  result["atomics_ok"] = atomics_ok;
  result["mfma_ok"] = mfma_ok;
  result["ok"] = (h_errors == 0) && (lds_errors == 0) && atomics_ok &&
                 mfma_ok;
  return result;
}

}  // namespace

PYBIND11_MODULE(_healthprobe, m) {
  m.doc() = "gfx950 GPU self-test: HBM pattern + bandwidth + MFMA";
  m.def("probe", &probe, py::arg("device") = 0, py::arg("mib") = 1024,
        py::arg("blocks_per_cu") = 8);
  m.def("device_count", [] {
    int n = 0;
    if (hipGetDeviceCount(&n) != hipSuccess) return 0;
    return n;
  });
}
