// Host (CPU) shim for compiling the CDNA4 kernels as single-lane C++.
//
// The kernels' control flow is wave-uniform by construction: lanes only
// diverge inside the byte-parallel primitives (skip_ws, string_end,
// put_escaped, wave_copy, ...), every one of which degrades to a correct
// serial loop at WAVE=1 with these shims.  Built by ops/build.py with
//   -DGGRMCP_HOST_SIM -DWAVE=1
// into _hostsim.so, which the CPU test suite uses as a bit-exact oracle of
// the device kernels (tests/test_hostsim.py) and which makes kernel logic
// debuggable with ASAN/gdb on GPU-less machines.
#pragma once

#include <math.h>
#include <stdint.h>

#define __global__
#define __device__
#define __host__
#define __forceinline__ inline
#define __constant__ const
#define __shared__ static
#define __launch_bounds__(x)

struct SimDim3 {
  unsigned x = 0, y = 0, z = 0;
};

// set by the simulator driver loop (one "wave" at a time)
extern SimDim3 threadIdx, blockIdx, blockDim, gridDim;

// single-lane wave intrinsics
static inline uint32_t __shfl(uint32_t v, int, int) { return v; }
static inline uint64_t __shfl(uint64_t v, int, int) { return v; }
static inline int __shfl(int v, int, int) { return v; }
static inline uint32_t __shfl_up(uint32_t, int, int) { return 0; }
static inline uint64_t __ballot(bool p) { return p ? 1ull : 0ull; }
static inline int __ffsll(long long m) { return __builtin_ffsll(m); }
static inline bool __all(int p) { return p != 0; }

static inline void __builtin_amdgcn_wave_barrier() {}
