// Fused expression interpreter — gfx950 (MI355X, CDNA4).
//
// Role parity (reference, capability mapping — no code ported): the
// tokio-less equivalent of datafusion's fused physical-expr evaluation
// inside ProjectExec/FilterExec (/root/reference/native-engine/
// datafusion-ext-plans/src/project_exec.rs, filter_exec.rs): one kernel
// evaluates a whole scalar expression tree per row, instead of one
// at::native launch (plus an intermediate HBM round-trip) per operator.
//
// Model: a typed stack machine. The host compiles an expression tree to
// postfix ExInstr ops (slot types resolved at compile time, so the
// kernel has no runtime type dispatch beyond column loads). Each thread
// evaluates one row per grid-stride step. The opcode switch is
// wave-uniform (every lane runs the same instruction sequence), so there
// is no divergence; per-row state is an LDS stack (dynamic size = the
// program's max depth, passed at launch) plus a null bitmask register.
//
// SQL three-valued logic: every slot carries a null bit; arithmetic and
// comparisons propagate nulls, AND/OR are Kleene, IFELSE treats a null
// condition as false, DIV/MOD by zero yields null (non-ANSI Spark).

#include <hip/hip_runtime.h>
#include <stdint.h>

#define AU_EXPORT extern "C" __attribute__((visibility("default")))

enum AuDType : int32_t {
  AU_BOOL = 0, AU_INT8 = 1, AU_INT16 = 2, AU_INT32 = 3, AU_INT64 = 4,
  AU_FLOAT32 = 5, AU_FLOAT64 = 6, AU_DATE32 = 7, AU_STRING = 8,
  AU_DECIMAL64 = 9, AU_LIST = 10, AU_TIMESTAMP = 11,
};

struct AuColDesc {
  const void* data;
  const int64_t* offsets;
  const uint8_t* validity;
  int32_t dtype;
  int32_t scale;
};

struct ExInstr {    // 16 bytes; imm carries int64 or double (bit pattern)
  int32_t op;
  int32_t a;
  int64_t imm;
};

enum ExOp : int32_t {
  EX_PUSH_COL = 0, EX_PUSH_LIT_I = 1, EX_PUSH_LIT_D = 2, EX_PUSH_NULL = 3,
  EX_I2D = 4, EX_D2I_TRUNC = 5, EX_D2I_ROUND = 6, EX_TRUNC_I = 7, EX_NEZ = 8,
  EX_ADDI = 9, EX_SUBI = 10, EX_MULI = 11,
  EX_ADDD = 12, EX_SUBD = 13, EX_MULD = 14,
  EX_DIVD = 15, EX_MODI = 16, EX_DIVI = 17,
  EX_LTI = 18, EX_LEI = 19, EX_GTI = 20, EX_GEI = 21, EX_EQI = 22, EX_NEI = 23,
  EX_LTD = 24, EX_LED = 25, EX_GTD = 26, EX_GED = 27, EX_EQD = 28, EX_NED = 29,
  EX_ANDB = 30, EX_ORB = 31, EX_NOTB = 32,
  EX_ISNULL = 33, EX_ISNOTNULL = 34,
  EX_IFELSE = 35, EX_COALESCE2 = 36,
  EX_OUT = 37,
  EX_PICK = 38,  // push a copy of slot sp-1-a
  EX_NIP = 39,   // drop slot sp-2 (keep top)
};

#define EX_BLOCK 256
#define EX_MAX_INSTR 192
#define EX_MAX_COLS 16
#define EX_MAX_OUTS 8

// column/output descriptors ride the kernel-argument block (copied by the
// runtime at launch — no explicit H2D staging, no extra buffers to manage)
struct ExParams {
  AuColDesc cols[EX_MAX_COLS];
  AuColDesc outs[EX_MAX_OUTS];
};

union ExVal { int64_t i; double d; };

__global__ void __launch_bounds__(EX_BLOCK) k_expr_exec(
    const ExInstr* __restrict__ prog, int n_instr,
    const ExParams p,
    int64_t n) {
  const AuColDesc* __restrict__ cols = p.cols;
  const AuColDesc* __restrict__ outs = p.outs;
  extern __shared__ int64_t stk[];  // [max_depth][EX_BLOCK]
  __shared__ ExInstr sprog[EX_MAX_INSTR];
  for (int i = threadIdx.x; i < n_instr; i += blockDim.x) sprog[i] = prog[i];
  __syncthreads();
  const int tid = threadIdx.x;
#define STK(k) stk[(int64_t)(k) * EX_BLOCK + tid]
  const int64_t stride = (int64_t)gridDim.x * blockDim.x;
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + tid; row < n;
       row += stride) {
    int sp = 0;                 // next free slot
    uint32_t nulls = 0;         // bit k set => slot k is null
    for (int i = 0; i < n_instr; i++) {
      const ExInstr ins = sprog[i];
      switch (ins.op) {
        case EX_PUSH_COL: {
          const AuColDesc c = cols[ins.a];
          ExVal v; v.i = 0;
          switch (c.dtype) {
            case AU_BOOL:   v.i = ((const uint8_t*)c.data)[row] != 0; break;
            case AU_INT8:   v.i = ((const int8_t*)c.data)[row]; break;
            case AU_INT16:  v.i = ((const int16_t*)c.data)[row]; break;
            case AU_INT32:
            case AU_DATE32: v.i = ((const int32_t*)c.data)[row]; break;
            case AU_INT64:
            case AU_DECIMAL64:
            case AU_TIMESTAMP: v.i = ((const int64_t*)c.data)[row]; break;
            case AU_FLOAT32: v.d = ((const float*)c.data)[row]; break;
            case AU_FLOAT64: v.d = ((const double*)c.data)[row]; break;
            default: break;
          }
          if (c.validity && !c.validity[row]) nulls |= 1u << sp;
          else nulls &= ~(1u << sp);
          STK(sp) = v.i; sp++;
          break;
        }
        case EX_PUSH_LIT_I: case EX_PUSH_LIT_D:
          nulls &= ~(1u << sp); STK(sp) = ins.imm; sp++; break;
        case EX_PUSH_NULL:
          nulls |= 1u << sp; STK(sp) = 0; sp++; break;
        case EX_I2D: {
          ExVal v; v.d = (double)STK(sp - 1); STK(sp - 1) = v.i; break;
        }
        case EX_D2I_TRUNC: {
          ExVal v; v.i = STK(sp - 1); STK(sp - 1) = (int64_t)v.d; break;
        }
        case EX_D2I_ROUND: {
          ExVal v; v.i = STK(sp - 1);
          STK(sp - 1) = (int64_t)nearbyint(v.d); break;
        }
        case EX_TRUNC_I: {
          int sh = 64 - (int)ins.imm;
          STK(sp - 1) = (STK(sp - 1) << sh) >> sh; break;
        }
        case EX_NEZ: STK(sp - 1) = STK(sp - 1) != 0; break;
        case EX_ADDI: case EX_SUBI: case EX_MULI:
        case EX_LTI: case EX_LEI: case EX_GTI: case EX_GEI:
        case EX_EQI: case EX_NEI: case EX_DIVI: case EX_MODI: {
          int64_t b = STK(sp - 1), a = STK(sp - 2);
          int64_t r = 0;
          bool nl = (nulls >> (sp - 1) & 1u) | (nulls >> (sp - 2) & 1u);
          switch (ins.op) {
            case EX_ADDI: r = a + b; break;
            case EX_SUBI: r = a - b; break;
            case EX_MULI: r = a * b; break;
            case EX_LTI: r = a < b; break;
            case EX_LEI: r = a <= b; break;
            case EX_GTI: r = a > b; break;
            case EX_GEI: r = a >= b; break;
            case EX_EQI: r = a == b; break;
            case EX_NEI: r = a != b; break;
            case EX_DIVI: if (b == 0) nl = true; else r = a / b; break;
            case EX_MODI: if (b == 0) nl = true; else r = a % b; break;
          }
          sp--; STK(sp - 1) = r;
          if (nl) nulls |= 1u << (sp - 1); else nulls &= ~(1u << (sp - 1));
          break;
        }
        case EX_ADDD: case EX_SUBD: case EX_MULD: case EX_DIVD:
        case EX_LTD: case EX_LED: case EX_GTD: case EX_GED:
        case EX_EQD: case EX_NED: {
          ExVal va, vb; vb.i = STK(sp - 1); va.i = STK(sp - 2);
          double a = va.d, b = vb.d;
          ExVal r; r.i = 0;
          bool nl = (nulls >> (sp - 1) & 1u) | (nulls >> (sp - 2) & 1u);
          switch (ins.op) {
            case EX_ADDD: r.d = a + b; break;
            case EX_SUBD: r.d = a - b; break;
            case EX_MULD: r.d = a * b; break;
            case EX_DIVD: if (b == 0.0) nl = true; else r.d = a / b; break;
            case EX_LTD: r.i = a < b; break;
            case EX_LED: r.i = a <= b; break;
            case EX_GTD: r.i = a > b; break;
            case EX_GED: r.i = a >= b; break;
            case EX_EQD: r.i = a == b; break;
            case EX_NED: r.i = a != b; break;
          }
          sp--; STK(sp - 1) = r.i;
          if (nl) nulls |= 1u << (sp - 1); else nulls &= ~(1u << (sp - 1));
          break;
        }
        case EX_ANDB: case EX_ORB: {
          // Kleene: FALSE dominates null for AND, TRUE for OR
          bool bv = STK(sp - 1) != 0, av = STK(sp - 2) != 0;
          bool bn = nulls >> (sp - 1) & 1u, an = nulls >> (sp - 2) & 1u;
          bool r, nl;
          if (ins.op == EX_ANDB) {
            nl = !((!an && !bn) || (!an && !av) || (!bn && !bv));
            r = (av || an) && (bv || bn);
          } else {
            nl = !((!an && !bn) || (!an && av) || (!bn && bv));
            r = (av && !an) || (bv && !bn);
          }
          sp--; STK(sp - 1) = r;
          if (nl) nulls |= 1u << (sp - 1); else nulls &= ~(1u << (sp - 1));
          break;
        }
        case EX_NOTB: STK(sp - 1) = STK(sp - 1) == 0; break;
        case EX_ISNULL:
          STK(sp - 1) = nulls >> (sp - 1) & 1u;
          nulls &= ~(1u << (sp - 1)); break;
        case EX_ISNOTNULL:
          STK(sp - 1) = !(nulls >> (sp - 1) & 1u);
          nulls &= ~(1u << (sp - 1)); break;
        case EX_IFELSE: {
          // stack: cond, then, else (top) — null cond selects else
          bool cn = nulls >> (sp - 3) & 1u;
          bool cv = STK(sp - 3) != 0 && !cn;
          int src = cv ? sp - 2 : sp - 1;
          int64_t v = STK(src);
          bool nl = nulls >> src & 1u;
          sp -= 2; STK(sp - 1) = v;
          if (nl) nulls |= 1u << (sp - 1); else nulls &= ~(1u << (sp - 1));
          break;
        }
        case EX_COALESCE2: {
          bool an = nulls >> (sp - 2) & 1u;
          int src = an ? sp - 1 : sp - 2;
          int64_t v = STK(src);
          bool nl = nulls >> src & 1u;
          sp--; STK(sp - 1) = v;
          if (nl) nulls |= 1u << (sp - 1); else nulls &= ~(1u << (sp - 1));
          break;
        }
        case EX_OUT: {
          const AuColDesc o = outs[ins.a];
          ExVal v; v.i = STK(sp - 1);
          bool nl = nulls >> (sp - 1) & 1u;
          sp--;
          switch (o.dtype) {
            case AU_BOOL:   ((uint8_t*)o.data)[row] = v.i != 0; break;
            case AU_INT8:   ((int8_t*)o.data)[row] = (int8_t)v.i; break;
            case AU_INT16:  ((int16_t*)o.data)[row] = (int16_t)v.i; break;
            case AU_INT32:
            case AU_DATE32: ((int32_t*)o.data)[row] = (int32_t)v.i; break;
            case AU_INT64:
            case AU_DECIMAL64:
            case AU_TIMESTAMP: ((int64_t*)o.data)[row] = v.i; break;
            case AU_FLOAT32: ((float*)o.data)[row] = (float)v.d; break;
            case AU_FLOAT64: ((double*)o.data)[row] = v.d; break;
            default: break;
          }
          if (o.validity) ((uint8_t*)o.validity)[row] = !nl;
          break;
        }
        case EX_PICK: {
          int src = sp - 1 - ins.a;
          int64_t v = STK(src);
          bool nl = nulls >> src & 1u;
          STK(sp) = v;
          if (nl) nulls |= 1u << sp; else nulls &= ~(1u << sp);
          sp++;
          break;
        }
        case EX_NIP: {
          int64_t v = STK(sp - 1);
          bool nl = nulls >> (sp - 1) & 1u;
          sp--; STK(sp - 1) = v;
          if (nl) nulls |= 1u << (sp - 1); else nulls &= ~(1u << (sp - 1));
          break;
        }
        default: break;
      }
    }
  }
#undef STK
}

AU_EXPORT int au_expr_exec(const void* prog_dev, int n_instr,
                           const void* host_cols, int n_cols,
                           const void* host_outs, int n_outs,
                           int max_depth, int64_t n, void* stream) {
  // host_cols/host_outs are HOST arrays of AuColDesc (whose data/validity
  // members are device pointers); they are copied into the by-value
  // kernel-arg block here, so the launch needs no descriptor upload
  if (n <= 0 || n_instr <= 0) return 0;
  if (n_instr > EX_MAX_INSTR || n_cols > EX_MAX_COLS || n_outs > EX_MAX_OUTS)
    return 1001;
  ExParams p{};
  const AuColDesc* hc = (const AuColDesc*)host_cols;
  const AuColDesc* ho = (const AuColDesc*)host_outs;
  for (int i = 0; i < n_cols; i++) p.cols[i] = hc[i];
  for (int i = 0; i < n_outs; i++) p.outs[i] = ho[i];
  int64_t g = (n + EX_BLOCK - 1) / EX_BLOCK;
  if (g > 2048) g = 2048;  // 256 CU x 8 XCD-filling grid-stride
  size_t lds = (size_t)max_depth * EX_BLOCK * sizeof(int64_t);
  hipLaunchKernelGGL(k_expr_exec, dim3((uint32_t)g), dim3(EX_BLOCK), lds,
                     (hipStream_t)stream,
                     (const ExInstr*)prog_dev, n_instr, p, n);
  return (int)hipGetLastError();
}
