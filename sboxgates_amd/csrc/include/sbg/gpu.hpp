// gpu.hpp — single-GPU scan engine: CDNA4 (gfx950) HIP kernels for the
// 3/5/7-LUT candidate scans. Implementation: csrc/hip/kernels.hip.
//
// One GpuEngine owns one device (HIP_VISIBLE_DEVICES / torchrun LOCAL_RANK
// selects which): multi-GPU runs use one process per GPU with RCCL-backed
// coordination at the Engine/DistCtx layer, never inside this class.
#pragma once

#include <memory>
#include <string>

#include "sbg/scan.hpp"

namespace sbg {

class GpuEngine {
 public:
  // Returns nullptr when no HIP device is visible (err describes why).
  static std::unique_ptr<GpuEngine> create(int device, std::string* err);
  ~GpuEngine();

  // Scans combinations [begin, end) of C(rq.n, k), k in {3,4,5,7}.
  // Blocking; early-exits via an in-kernel abort flag unless rq.count_all.
  ScanResult scan(int k, const ScanRequest& rq, i64 begin, i64 end);

  // True when k=4 scans go through the persistent scan-service kernel
  // (~5-8 us per call) instead of one-shot launches (~30 us API floor).
  // Lazily creates the per-device service on first query.
  bool scan4_service_active();

  int device() const;
  std::string device_name() const;

  struct Impl;

 private:
  explicit GpuEngine(Impl* impl) : impl_(impl) {}
  Impl* impl_;
};

// True if the process can see at least one HIP device.
bool gpu_available();
int gpu_count();

}  // namespace sbg
