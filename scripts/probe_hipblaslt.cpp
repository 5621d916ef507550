// hipBLASLt algo search for the decode GEMM shapes.
//
// torch's F.linear goes through hipBLASLt with torch's own heuristic
// pick; this probe asks hipBLASLt for its full heuristic candidate list
// per shape (y[M,N] = x[M,K] @ W[N,K]^T, bf16 in / bf16 out, f32
// accumulate — identical to the serving path) and times each, printing
// the best few.  If the best beats torch's pick (timed separately in
// python on the same box), a fixed algo index is worth wiring into a
// custom linear op.
//
// Build: hipcc --offload-arch=gfx950 -O2 scripts/probe_hipblaslt.cpp \
//          -o /tmp/probe_hipblaslt -lhipblaslt
#include <hip/hip_runtime.h>
#include <hip/hip_bf16.h>
#include <hipblaslt/hipblaslt-ext.hpp>

#include <algorithm>
#include <cstdio>
#include <vector>

#define CK(x)                                                      \
  do {                                                             \
    auto e_ = (x);                                                 \
    if (e_ != hipSuccess) {                                        \
      printf("HIP error %d at line %d\n", int(e_), __LINE__);      \
      return 1;                                                    \
    }                                                              \
  } while (0)
#define CB(x)                                                      \
  do {                                                             \
    auto e_ = (x);                                                 \
    if (e_ != HIPBLAS_STATUS_SUCCESS) {                            \
      printf("hipblaslt error %d at line %d\n", int(e_), __LINE__);\
      continue;                                                    \
    }                                                              \
  } while (0)

struct Shape {
  const char* name;
  int M, N, K;
};

int main() {
  // decode shapes at batch 64 (Llama-3-8B) + lm_head
  Shape shapes[] = {
      {"qkv   ", 64, 6144, 4096},    {"o     ", 64, 4096, 4096},
      {"gateup", 64, 28672, 4096},   {"down  ", 64, 4096, 14336},
      {"lmhead", 64, 128256, 4096},
  };
  hipblasLtHandle_t handle;
  if (hipblasLtCreate(&handle) != HIPBLAS_STATUS_SUCCESS) {
    printf("hipblasLtCreate failed\n");
    return 1;
  }
  hipStream_t stream;
  CK(hipStreamCreate(&stream));
  void* ws;
  const size_t ws_size = size_t(128) << 20;
  CK(hipMalloc(&ws, ws_size));

  for (const Shape& s : shapes) {
    // allocations (bf16); values irrelevant for timing
    __hip_bfloat16 *x, *w, *y;
    CK(hipMalloc(&x, size_t(s.M) * s.K * 2));
    CK(hipMalloc(&w, size_t(s.N) * s.K * 2));
    CK(hipMalloc(&y, size_t(s.M) * s.N * 2));
    CK(hipMemset(x, 0x3c, size_t(s.M) * s.K * 2));
    CK(hipMemset(w, 0x3c, size_t(s.N) * s.K * 2));

    // C_col[N,M] = W_col[K,N]^T * x_col[K,M]
    hipblaslt_ext::Gemm gemm(handle, HIPBLAS_OP_T, HIPBLAS_OP_N,
                             HIP_R_16BF, HIP_R_16BF, HIP_R_16BF, HIP_R_16BF,
                             HIPBLAS_COMPUTE_32F);
    hipblaslt_ext::GemmEpilogue ep;   // default: no epilogue
    hipblaslt_ext::GemmInputs in;
    float alpha = 1.f, beta = 0.f;
    in.setA(w);
    in.setB(x);
    in.setC(y);
    in.setD(y);
    in.setAlpha(&alpha);
    in.setBeta(&beta);
    if (gemm.setProblem(s.N, s.M, s.K, 1, ep, in) != HIPBLAS_STATUS_SUCCESS) {
      printf("%s setProblem failed\n", s.name);
      continue;
    }
    hipblaslt_ext::GemmPreference pref;
    pref.setMaxWorkspaceBytes(ws_size);
    std::vector<hipblasLtMatmulHeuristicResult_t> algos;
    if (gemm.algoGetHeuristic(128, pref, algos) != HIPBLAS_STATUS_SUCCESS ||
        algos.empty()) {
      printf("%s no heuristic algos\n", s.name);
      continue;
    }

    struct Res { double us; int idx; };
    std::vector<Res> results;
    hipEvent_t ev0, ev1;
    CK(hipEventCreate(&ev0));
    CK(hipEventCreate(&ev1));
    for (size_t ai = 0; ai < algos.size(); ++ai) {
      size_t need = 0;
      if (gemm.isAlgoSupported(algos[ai].algo, need) !=
              HIPBLAS_STATUS_SUCCESS ||
          need > ws_size)
        continue;
      CB(gemm.initialize(algos[ai].algo, ws));
      // warmup
      for (int it = 0; it < 10; ++it) CB(gemm.run(stream));
      CK(hipStreamSynchronize(stream));
      CK(hipEventRecord(ev0, stream));
      const int iters = 100;
      for (int it = 0; it < iters; ++it) CB(gemm.run(stream));
      CK(hipEventRecord(ev1, stream));
      CK(hipEventSynchronize(ev1));
      float ms = 0.f;
      CK(hipEventElapsedTime(&ms, ev0, ev1));
      results.push_back({double(ms) * 1e3 / iters,
                         int(hipblaslt_ext::getIndexFromAlgo(
                             const_cast<hipblasLtMatmulAlgo_t&>(algos[ai].algo)))});
    }
    std::sort(results.begin(), results.end(),
              [](const Res& a, const Res& b) { return a.us < b.us; });
    const double gb = double(s.N) * s.K * 2 / 1e9;   // weight bytes
    printf("%s M=%d N=%d K=%d: %zu algos timed\n", s.name, s.M, s.N, s.K,
           results.size());
    for (size_t i = 0; i < results.size() && i < 4; ++i)
      printf("   #%zu  algo_index=%-6d %8.2f us  (%5.2f TB/s weight-stream)\n",
             i, results[i].idx, results[i].us, gb / results[i].us * 1e3);
    CK(hipEventDestroy(ev0));
    CK(hipEventDestroy(ev1));
    CK(hipFree(x));
    CK(hipFree(w));
    CK(hipFree(y));
  }
  CK(hipFree(ws));
  hipblasLtDestroy(handle);
  printf("PROBE DONE\n");
  return 0;
}
