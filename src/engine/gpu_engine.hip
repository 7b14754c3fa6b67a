// GpuEngine — MI355X data plane. Placeholder for M0 (fails loudly); the real
// engine lands with the M1 milestone (HBM arena, pinned staging, HIP kernels).
#include <stdexcept>
#include "engine.h"

namespace edgehip {
std::unique_ptr<Engine> make_gpu_engine(const Conf&) {
    throw std::runtime_error("GpuEngine: not built in this milestone");
}
}  // namespace edgehip
