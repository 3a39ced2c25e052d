// Linked ONLY by the CPU-only CMake configuration (a host compiler that
// is not hipcc): the GPU engine factory honors its documented contract
// ("returns nullptr when no HIP device is available") so the PumiTally
// facade and the engine_api example fall back to the CPU engine.  The
// hipcc build replaces this TU with the real factory in
// csrc/hip/engine_gpu.hip.
#include "engine.h"

namespace pumitally {

std::unique_ptr<Engine> make_gpu_engine(Mesh, int64_t, int, int, int) {
  return nullptr;
}

} // namespace pumitally
