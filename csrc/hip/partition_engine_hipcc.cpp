// CMake-friendly wrapper: hipcc compiles .cpp TUs as HIP, but CMake's
// language dispatch mangles bare .hip files; this TU carries the engine.
#include "partition_engine.hip"
