// Probe which hipBLASLt epilogue configs have heuristic algos.
#include <hip/hip_runtime.h>
#include <hipblaslt/hipblaslt.h>
#include <cstdio>
#include <vector>

int main() {
  hipblasLtHandle_t h;
  hipblasLtCreate(&h);
  struct Cfg { const char* name; hipblasLtEpilogue_t epi; bool aux; };
  std::vector<Cfg> cfgs = {
    {"BIAS", HIPBLASLT_EPILOGUE_BIAS, false},
    {"GELU_BIAS", HIPBLASLT_EPILOGUE_GELU_BIAS, false},
    {"GELU_AUX_BIAS", HIPBLASLT_EPILOGUE_GELU_AUX_BIAS, true},
    {"DGELU", HIPBLASLT_EPILOGUE_DGELU, true},
    {"DGELU_BGRAD", HIPBLASLT_EPILOGUE_DGELU_BGRAD, true},
    {"BGRADB", HIPBLASLT_EPILOGUE_BGRADB, false},
    {"BGRADA", HIPBLASLT_EPILOGUE_BGRADA, false},
  };
  struct Shape { int64_t m, n, k; hipblasOperation_t oa, ob; };
  std::vector<Shape> shapes = {
    {1024, 256, 256, HIPBLAS_OP_T, HIPBLAS_OP_N},
    {6400, 16384, 1600, HIPBLAS_OP_T, HIPBLAS_OP_N},
    {6400, 16384, 1600, HIPBLAS_OP_N, HIPBLAS_OP_N},
    {6400, 1600, 16384, HIPBLAS_OP_N, HIPBLAS_OP_T},
  };
  void* dummy;
  hipMalloc(&dummy, 1 << 20);
  for (auto& sh : shapes) {
    for (auto& c : cfgs) {
      for (int auxty = 0; auxty < (c.aux ? 2 : 1); ++auxty) {
        hipblasLtMatmulDesc_t d;
        hipblasLtMatmulDescCreate(&d, HIPBLAS_COMPUTE_32F, HIP_R_32F);
        hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_TRANSA,
                                        &sh.oa, sizeof(sh.oa));
        hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_TRANSB,
                                        &sh.ob, sizeof(sh.ob));
        hipblasLtMatmulDescSetAttribute(d, HIPBLASLT_MATMUL_DESC_EPILOGUE,
                                        &c.epi, sizeof(c.epi));
        hipblasLtMatmulDescSetAttribute(
            d, HIPBLASLT_MATMUL_DESC_BIAS_POINTER, &dummy, sizeof(dummy));
        if (c.aux) {
          hipblasLtMatmulDescSetAttribute(
              d, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_POINTER, &dummy,
              sizeof(dummy));
          int64_t ld = sh.m;
          hipblasLtMatmulDescSetAttribute(
              d, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_LD, &ld, sizeof(ld));
          if (auxty == 1) {
            int32_t t = HIP_R_16BF;
            hipblasLtMatmulDescSetAttribute(
                d, HIPBLASLT_MATMUL_DESC_EPILOGUE_AUX_DATA_TYPE, &t,
                sizeof(t));
          }
        }
        int64_t ar = (sh.oa == HIPBLAS_OP_N) ? sh.m : sh.k;
        int64_t ac = (sh.oa == HIPBLAS_OP_N) ? sh.k : sh.m;
        int64_t br = (sh.ob == HIPBLAS_OP_N) ? sh.k : sh.n;
        int64_t bc = (sh.ob == HIPBLAS_OP_N) ? sh.n : sh.k;
        hipblasLtMatrixLayout_t la, lb, lc;
        hipblasLtMatrixLayoutCreate(&la, HIP_R_16BF, ar, ac, ar);
        hipblasLtMatrixLayoutCreate(&lb, HIP_R_16BF, br, bc, br);
        hipblasLtMatrixLayoutCreate(&lc, HIP_R_16BF, sh.m, sh.n, sh.m);
        hipblasLtMatmulPreference_t pref;
        hipblasLtMatmulPreferenceCreate(&pref);
        size_t ws = 64u << 20;
        hipblasLtMatmulPreferenceSetAttribute(
            pref, HIPBLASLT_MATMUL_PREF_MAX_WORKSPACE_BYTES, &ws,
            sizeof(ws));
        hipblasLtMatmulHeuristicResult_t res[8];
        int found = 0;
        hipblasStatus_t st = hipblasLtMatmulAlgoGetHeuristic(
            h, d, la, lb, lc, lc, pref, 8, res, &found);
        printf("m=%-5ld n=%-5ld k=%-5ld %c%c %-14s auxty=%d st=%d found=%d\n",
               (long)sh.m, (long)sh.n, (long)sh.k,
               sh.oa == HIPBLAS_OP_N ? 'N' : 'T',
               sh.ob == HIPBLAS_OP_N ? 'N' : 'T', c.name, auxty,
               (int)st, found);
        hipblasLtMatmulPreferenceDestroy(pref);
        hipblasLtMatrixLayoutDestroy(la);
        hipblasLtMatrixLayoutDestroy(lb);
        hipblasLtMatrixLayoutDestroy(lc);
        hipblasLtMatmulDescDestroy(d);
      }
    }
  }
  printf("probe done\n");
  return 0;
}
