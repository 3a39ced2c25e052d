// Small fp64 vector geometry used by both the CPU engine and the HIP kernels.
//
// Everything here is header-only and marked host+device so the exact same
// arithmetic runs on the CPU oracle and on the MI355X walk kernels
// (differential tests compare them bitwise on paths, 1e-12 on flux).
#pragma once

#include <cmath>
#include <cstdint>

#if defined(__HIPCC__)
#include <hip/hip_runtime.h>
#define PT_HD __host__ __device__ __forceinline__
#else
#define PT_HD inline
#endif

namespace pumitally {

struct Vec3 {
  double x, y, z;
};

PT_HD Vec3 operator+(Vec3 a, Vec3 b) { return {a.x + b.x, a.y + b.y, a.z + b.z}; }
PT_HD Vec3 operator-(Vec3 a, Vec3 b) { return {a.x - b.x, a.y - b.y, a.z - b.z}; }
PT_HD Vec3 operator*(double s, Vec3 a) { return {s * a.x, s * a.y, s * a.z}; }
PT_HD double dot(Vec3 a, Vec3 b) { return a.x * b.x + a.y * b.y + a.z * b.z; }
PT_HD Vec3 cross(Vec3 a, Vec3 b) {
  return {a.y * b.z - a.z * b.y, a.z * b.x - a.x * b.z, a.x * b.y - a.y * b.x};
}
PT_HD double norm(Vec3 a) { return sqrt(dot(a, a)); }

// One face plane of a tet, stored inward-positive: a point p is on the inside
// of the face iff dot(n,p) - c >= 0.  n is unit length, so the value is a
// signed distance.  Layout matches double4 (32 B) for coalesced GPU loads.
struct Plane {
  double nx, ny, nz, c;
};

PT_HD double plane_eval(const Plane &pl, Vec3 p) {
  return pl.nx * p.x + pl.ny * p.y + pl.nz * p.z - pl.c;
}

// fp32 plane (float4 layout, 16 B) for the traversal fast path: exit-face
// CANDIDATE decisions in fp32, the chosen face's crossing parameter and all
// tallies in fp64 (walk.h).
struct Plane32 {
  float nx, ny, nz, c;
};

struct Vec3f {
  float x, y, z;
};

PT_HD float plane_eval32(const Plane32 &pl, Vec3f p) {
  return pl.nx * p.x + pl.ny * p.y + pl.nz * p.z - pl.c;
}

// Signed volume of tet (a,b,c,d): positive when d is on the positive side of
// triangle (a,b,c) oriented by the right-hand rule.
PT_HD double signed_volume(Vec3 a, Vec3 b, Vec3 c, Vec3 d) {
  return dot(b - a, cross(c - a, d - a)) / 6.0;
}

} // namespace pumitally
