// gpu_stub.cpp — GpuEngine stub for sanitizer builds of the host engine
// (selftest links host sources with g++ + ASan/UBSan/TSan; the HIP
// runtime is out of scope there). The stub reports "no GPU": every scan
// takes the CPU path.

#include "sbg/gpu.hpp"

namespace sbg {

bool gpu_available() { return false; }
int gpu_count() { return 0; }

std::unique_ptr<GpuEngine> GpuEngine::create(int, std::string* err) {
  if (err != nullptr) *err = "sanitizer build: GPU engine stubbed out";
  return nullptr;
}

GpuEngine::~GpuEngine() = default;
int GpuEngine::device() const { return -1; }
std::string GpuEngine::device_name() const { return "stub"; }
ScanResult GpuEngine::scan(int, const ScanRequest&, i64, i64) { return {}; }
bool GpuEngine::scan4_service_active() { return false; }

}  // namespace sbg
