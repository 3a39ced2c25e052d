// CMake-friendly wrapper: hipcc compiles .cpp TUs as HIP, but CMake's
// language dispatch mangles bare .hip files; this TU carries the RCCL comm.
#include "comm_rccl.hip"
