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
  RX_SAVE = 8,  // arg0 = slot: capture-group position
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
          case RX_SAVE:
            ++pc;  // match-only kernel ignores captures
            continue;
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

// ---- capture-group engine (regexp_extract / regexp_replace) --------------
// Backtracking with save slots: SAVE writes go through an undo log so a
// backtrack restores the capture state of the resumed alternative.
#define RX_SLOTS 20   // whole match + 9 groups
#define RX_UNDO 96

// try to match at `start`; on success returns end position and fills
// saves[] (saves[0]=start, saves[1]=end, 2g/2g+1 = group g). Returns -1 on
// no match, -2 on resource overflow.
__device__ int32_t rx_try(const int32_t* __restrict__ prog,
                          const uint8_t* __restrict__ classes,
                          const uint8_t* __restrict__ s, int32_t len,
                          int32_t start, int32_t* __restrict__ saves,
                          int64_t* __restrict__ steps) {
  uint64_t stack[RX_STACK];
  uint8_t sdepth[RX_STACK];
  int64_t undo[RX_UNDO];  // (slot << 32) | (uint32)oldval
  int nundo = 0;
  int sp_depth = 0;
  for (int k = 0; k < RX_SLOTS; ++k) saves[k] = -1;
  int32_t pc = 0, sp = start;
  while (true) {
    if (++*steps > RX_MAX_STEPS) return -2;
    int op = prog[3 * pc];
    int a0 = prog[3 * pc + 1];
    int a1 = prog[3 * pc + 2];
    bool fail = false;
    switch (op) {
      case RX_MATCH:
        saves[0] = start;
        saves[1] = sp;
        return sp;
      case RX_JMP:
        pc = a0;
        continue;
      case RX_SPLIT:
        if (sp_depth >= RX_STACK) return -2;
        sdepth[sp_depth] = (uint8_t)nundo;
        stack[sp_depth++] = ((uint64_t)(uint32_t)a1 << 32) | (uint32_t)sp;
        pc = a0;
        continue;
      case RX_SAVE:
        if (a0 < RX_SLOTS) {
          if (nundo >= RX_UNDO) return -2;
          undo[nundo++] = ((int64_t)a0 << 32) | (uint32_t)saves[a0];
          saves[a0] = sp;
        }
        ++pc;
        continue;
      case RX_BOL:
        fail = sp != 0;
        break;
      case RX_EOL:
        fail = sp != len;
        break;
      case RX_CHAR:
        fail = !(sp < len && s[sp] == (uint8_t)a0);
        if (!fail) ++sp;
        break;
      case RX_ANY:
        fail = !(sp < len && s[sp] != (uint8_t)'\n');
        if (!fail) ++sp;
        break;
      case RX_CLASS: {
        bool in = sp < len && class_has(classes + 32 * a0, s[sp]);
        if (a1) in = sp < len && !in;
        fail = !in;
        if (!fail) ++sp;
        break;
      }
      default:
        fail = true;
        break;
    }
    if (!fail) {
      ++pc;
      continue;
    }
    if (sp_depth == 0) return -1;
    uint64_t top = stack[--sp_depth];
    int back = sdepth[sp_depth];
    while (nundo > back) {
      int64_t u = undo[--nundo];
      saves[(int)(u >> 32)] = (int32_t)(uint32_t)u;
    }
    pc = (int32_t)(top >> 32);
    sp = (int32_t)(uint32_t)top;
  }
}

// leftmost match at or after `from`: -1 none, -2 overflow
__device__ __forceinline__ int32_t rx_find_from(
    const int32_t* prog, const uint8_t* classes, const uint8_t* s,
    int32_t len, int32_t from, int32_t* saves, int64_t* steps) {
  for (int32_t st = from; st <= len; ++st) {
    int32_t e = rx_try(prog, classes, s, len, st, saves, steps);
    if (e != -1) return e;  // match or overflow
  }
  return -1;
}

// regexp_extract: absolute (start, len) of the requested group per row;
// empty when no match or the group did not participate (Spark returns "")
__global__ void k_regex_extract(const int32_t* __restrict__ prog, int nops,
                                const uint8_t* __restrict__ classes,
                                const int32_t* __restrict__ offsets,
                                const uint8_t* __restrict__ bytes, int group,
                                int32_t* __restrict__ out_start,
                                int64_t* __restrict__ out_len,
                                int32_t* __restrict__ overflow, int64_t n) {
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    int32_t begin = offsets[row];
    int32_t len = offsets[row + 1] - begin;
    const uint8_t* s = bytes + begin;
    int32_t saves[RX_SLOTS];
    int64_t steps = 0;
    int32_t e = rx_find_from(prog, classes, s, len, 0, saves, &steps);
    out_start[row] = begin;
    out_len[row] = 0;
    if (e == -2) {
      atomicAdd(overflow, 1);
    } else if (e >= 0) {
      int32_t gs = saves[2 * group], ge = saves[2 * group + 1];
      if (gs >= 0 && ge >= gs) {
        out_start[row] = begin + gs;
        out_len[row] = ge - gs;
      }
    }
  }
}

// replacement template op: kind 0 = literal (a0=offset into lit, a1=len),
// kind 1 = group a0
// mode 0: compute output length per row; mode 1: write bytes at out_off
__global__ void k_regex_replace(const int32_t* __restrict__ prog, int nops,
                                const uint8_t* __restrict__ classes,
                                const int32_t* __restrict__ offsets,
                                const uint8_t* __restrict__ bytes,
                                const int32_t* __restrict__ repl_ops,
                                int nrepl,
                                const uint8_t* __restrict__ lit,
                                const int64_t* __restrict__ out_off,
                                int64_t* __restrict__ out_len,
                                uint8_t* __restrict__ out_bytes, int mode,
                                int32_t* __restrict__ overflow, int64_t n) {
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    int32_t begin = offsets[row];
    int32_t len = offsets[row + 1] - begin;
    const uint8_t* s = bytes + begin;
    uint8_t* dst = mode ? out_bytes + out_off[row] : nullptr;
    int64_t w = 0;
    int32_t saves[RX_SLOTS];
    int64_t steps = 0;
    int32_t pos = 0;
    bool blown = false;
    while (pos <= len) {
      int32_t e = rx_find_from(prog, classes, s, len, pos, saves, &steps);
      if (e == -2) {
        blown = true;
        break;
      }
      if (e < 0) break;
      int32_t ms = saves[0], me = saves[1];
      for (int32_t k = pos; k < ms; ++k, ++w)
        if (mode) dst[w] = s[k];
      for (int r = 0; r < nrepl; ++r) {
        int kind = repl_ops[3 * r];
        int a0 = repl_ops[3 * r + 1];
        int a1 = repl_ops[3 * r + 2];
        if (kind == 0) {
          for (int k = 0; k < a1; ++k, ++w)
            if (mode) dst[w] = lit[a0 + k];
        } else {
          int32_t gs = saves[2 * a0], ge = saves[2 * a0 + 1];
          if (gs >= 0)
            for (int32_t k = gs; k < ge; ++k, ++w)
              if (mode) dst[w] = s[k];
        }
      }
      if (me > ms) {
        pos = me;
      } else {  // empty match: copy the next char and advance (java)
        if (ms < len) {
          if (mode) dst[w] = s[ms];
          ++w;
        }
        pos = ms + 1;
      }
    }
    if (blown) {
      atomicAdd(overflow, 1);
      out_len[row] = 0;
      continue;
    }
    for (int32_t k = pos; k <= len - 1; ++k, ++w)
      if (mode) dst[w] = s[k];
    if (!mode) out_len[row] = w;
  }
}

// regexp_extract_all: every non-overlapping match's group span.
// mode 0: per-row match count; mode 1: (start, len) spans at part_off.
__global__ void k_regex_extract_all(const int32_t* __restrict__ prog,
                                    int nops,
                                    const uint8_t* __restrict__ classes,
                                    const int32_t* __restrict__ offsets,
                                    const uint8_t* __restrict__ bytes,
                                    int group,
                                    const int64_t* __restrict__ part_off,
                                    int64_t* __restrict__ counts,
                                    int32_t* __restrict__ out_ss,
                                    int64_t* __restrict__ out_sl, int mode,
                                    int32_t* __restrict__ overflow,
                                    int64_t n) {
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < n;
       row += (int64_t)gridDim.x * blockDim.x) {
    int32_t begin = offsets[row];
    int32_t len = offsets[row + 1] - begin;
    const uint8_t* s = bytes + begin;
    int32_t saves[RX_SLOTS];
    int64_t steps = 0;
    int32_t pos = 0;
    int64_t cnt = 0;
    int64_t w = mode ? part_off[row] : 0;
    bool blown = false;
    while (pos <= len) {
      int32_t e = rx_find_from(prog, classes, s, len, pos, saves, &steps);
      if (e == -2) {
        blown = true;
        break;
      }
      if (e < 0) break;
      int32_t ms = saves[0], me = saves[1];
      if (mode) {
        int32_t gs = saves[2 * group], ge = saves[2 * group + 1];
        if (gs >= 0 && ge >= gs) {
          out_ss[w] = begin + gs;
          out_sl[w] = ge - gs;
        } else {
          out_ss[w] = begin;
          out_sl[w] = 0;
        }
        ++w;
      }
      ++cnt;
      pos = me > ms ? me : ms + 1;
    }
    if (blown) atomicAdd(overflow, 1);
    if (!mode) counts[row] = blown ? 0 : cnt;
  }
}

extern "C" {


void hipdf_regex_extract(const void* prog, int nops, const void* classes,
                         const void* offsets, const void* bytes, int group,
                         void* out_start, void* out_len, void* overflow,
                         int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_regex_extract, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)prog, nops,
                     (const uint8_t*)classes, (const int32_t*)offsets,
                     (const uint8_t*)bytes, group, (int32_t*)out_start,
                     (int64_t*)out_len, (int32_t*)overflow, n);
}

void hipdf_regex_extract_all(const void* prog, int nops,
                             const void* classes, const void* offsets,
                             const void* bytes, int group,
                             const void* part_off, void* counts,
                             void* out_ss, void* out_sl, int mode,
                             void* overflow, int64_t n, hipStream_t stream) {
  hipLaunchKernelGGL(k_regex_extract_all, flat_grid(n), dim3(HIPDF_BLOCK),
                     0, stream, (const int32_t*)prog, nops,
                     (const uint8_t*)classes, (const int32_t*)offsets,
                     (const uint8_t*)bytes, group,
                     (const int64_t*)part_off, (int64_t*)counts,
                     (int32_t*)out_ss, (int64_t*)out_sl, mode,
                     (int32_t*)overflow, n);
}

void hipdf_regex_replace(const void* prog, int nops, const void* classes,
                         const void* offsets, const void* bytes,
                         const void* repl_ops, int nrepl, const void* lit,
                         const void* out_off, void* out_len, void* out_bytes,
                         int mode, void* overflow, int64_t n,
                         hipStream_t stream) {
  hipLaunchKernelGGL(k_regex_replace, flat_grid(n), dim3(HIPDF_BLOCK), 0,
                     stream, (const int32_t*)prog, nops,
                     (const uint8_t*)classes, (const int32_t*)offsets,
                     (const uint8_t*)bytes, (const int32_t*)repl_ops, nrepl,
                     (const uint8_t*)lit, (const int64_t*)out_off,
                     (int64_t*)out_len, (uint8_t*)out_bytes, mode,
                     (int32_t*)overflow, n);
}

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
