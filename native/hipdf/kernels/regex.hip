// Device regex matcher: backtracking VM over bytecode compiled by
// spark_rapids_amd/ops/regex_compiler.py (reference analogue: cudf's device
// regex engine behind RLike — SURVEY.md §2.4 regex transpiler row).
//
// Find semantics (java Matcher.find): try a match at every start offset.
// One thread per row; explicit backtrack stack; step/stack overflow bumps a
// global counter and the host re-runs the column on CPU (exactness over
// silent divergence).
#include "hipdf_common.h"

enum RxOp : int {
  RX_CHAR = 0,
  RX_ANY = 1,
  RX_CLASS = 2,
  RX_MATCH = 3,
  RX_JMP = 4,
  RX_SPLIT = 5,
  RX_BOL = 6,
  RX_EOL = 7,
};

#define RX_STACK 64
#define RX_MAX_STEPS 200000

__device__ __forceinline__ bool class_has(const uint8_t* bitmap, uint8_t b) {
  return (bitmap[b >> 3] >> (b & 7)) & 1;
}

__global__ void k_regex_match(const int32_t* __restrict__ prog, int nops,
                              const uint8_t* __restrict__ classes,
                              const int32_t* __restrict__ offsets,
                              const uint8_t* __restrict__ bytes,
                              uint8_t* __restrict__ out,
                              int32_t* __restrict__ overflow, int64_t n) {
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    int32_t begin = offsets[row];
    int32_t end = offsets[row + 1];
    int32_t len = end - begin;
    const uint8_t* s = bytes + begin;
    uint64_t stack[RX_STACK];
    bool matched = false;
    bool blown = false;
    int64_t steps = 0;
    for (int32_t start = 0; start <= len && !matched && !blown; ++start) {
      int sp_depth = 0;
      int32_t pc = 0, sp = start;
      while (true) {
        if (++steps > RX_MAX_STEPS) {
          blown = true;
          break;
        }
        bool ok;
        int op = prog[3 * pc];
        int a0 = prog[3 * pc + 1];
        int a1 = prog[3 * pc + 2];
        switch (op) {
          case RX_MATCH:
            matched = true;
            break;
          case RX_JMP:
            pc = a0;
            continue;
          case RX_SPLIT:
            if (sp_depth >= RX_STACK) {
              blown = true;
              break;
            }
            stack[sp_depth++] = ((uint64_t)(uint32_t)a1 << 32) |
                                (uint32_t)sp;
            pc = a0;
            continue;
          case RX_BOL:
            ok = sp == 0;
            if (ok) {
              ++pc;
              continue;
            }
            break;
          case RX_EOL:
            ok = sp == len;
            if (ok) {
              ++pc;
              continue;
            }
            break;
          case RX_CHAR:
            ok = sp < len && s[sp] == (uint8_t)a0;
            if (ok) {
              ++sp;
              ++pc;
              continue;
            }
            break;
          case RX_ANY:
            ok = sp < len && s[sp] != (uint8_t)'\n';
            if (ok) {
              ++sp;
              ++pc;
              continue;
            }
            break;
          case RX_CLASS: {
            bool in = sp < len && class_has(classes + 32 * a0, s[sp]);
            if (a1) in = sp < len && !in;
            if (in) {
              ++sp;
              ++pc;
              continue;
            }
            break;
          }
          default:
            break;
        }
        if (matched || blown) break;
        // fail: backtrack
        if (sp_depth == 0) break;
        uint64_t top = stack[--sp_depth];
        pc = (int32_t)(top >> 32);
        sp = (int32_t)(uint32_t)top;
      }
    }
    if (blown) {
      atomicAdd(overflow, 1);
      out[row] = 0;
    } else {
      out[row] = matched ? 1 : 0;
    }
  }
}

extern "C" {

void hipdf_regex_match(const void* prog, int nops, const void* classes,
                       const void* offsets, const void* bytes, void* out,
                       void* overflow, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_regex_match, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)prog, nops,
                     (const uint8_t*)classes, (const int32_t*)offsets,
                     (const uint8_t*)bytes, (uint8_t*)out, (int32_t*)overflow,
                     n);
}

}  // extern "C"
