// dist.hpp — distributed-coordination interface for the multi-GPU search.
//
// The reference's MPI SPMD layer (sboxgates.c:618-642, lut.c:116-487,
// 665-740: broadcast + Isend/Irecv early exit + cancel/barrier dance) is
// replaced by a deliberately tiny, deadlock-free contract: a byte
// broadcast, a min-allreduce and a winner broadcast, called in a fixed
// chunked cadence computed identically on every rank. Implementations:
//   * LocalCtx — single process (world 1), all no-ops.
//   * Python (sboxgates_amd.parallel) — torch.distributed over RCCL/xGMI
//     (or gloo on CPU), bound through pybind11 callbacks.
// Payloads are tiny (<= ~33 KB state blob, 4 B keys, 20 B winners), so the
// layer is latency-bound: single-shot collectives, no rings.
#pragma once

#include <cstddef>

#include "sbg/common.hpp"
#include "sbg/state.hpp"
#include "sbg/ttable.hpp"

namespace sbg {

class DistCtx {
 public:
  virtual ~DistCtx() = default;
  virtual int rank() const { return 0; }
  virtual int world() const { return 1; }
  // Broadcast bytes from rank `root` to all ranks (in-place).
  virtual void bcast(void* data, size_t n, int root) { (void)data; (void)n; (void)root; }
  // Global minimum of a per-rank int.
  virtual int allreduce_min(int v) { return v; }
  virtual void barrier() {}
};

// Single-process context.
class LocalCtx : public DistCtx {};

// Work descriptor broadcast from rank 0 to workers before each distributed
// LUT search round (the RCCL analog of the reference's mpi_work,
// sboxgates.h:68-76). POD: broadcast as raw bytes.
struct WorkMsg {
  i32 kind;          // 0 = quit, 1 = distributed 5/7-LUT search
  i32 verbosity;
  u64 seed;          // per-round randomization base, shared by all ranks
  ttable target;
  ttable mask;
  i8 inbits[8];
  state st;
};

constexpr i32 WORK_QUIT = 0;
constexpr i32 WORK_LUT_SEARCH = 1;

// The WorkMsg is broadcast as raw bytes between ranks (same-arch SPMD);
// lock its ABI so a header change cannot silently desynchronize ranks
// built from different trees.
static_assert(sizeof(WorkMsg) == 32160, "WorkMsg wire format changed");

}  // namespace sbg
