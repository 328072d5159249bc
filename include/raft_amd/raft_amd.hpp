// raft_amd — public C++ API of the native MI355X (gfx950) kernel library.
//
// Umbrella header: the host-callable entry points implemented in csrc/ and
// compiled into libraft_amd / the Python extension. Plain C++ functions over
// device pointers + a hipStream_t, usable from any C++ host code linked
// against build/ext objects or the CMake `raft_amd::raft_amd` target — no
// Python, no torch required (bindings.cpp binds them 1:1 for Python).
//
// Reference parity: the raft_runtime precompiled-API idea
// (cpp/include/raft_runtime in rapidsai/raft) — non-templated entry points a
// consumer can call without a device compiler.
#pragma once

#include "cluster.hpp"
#include "core/mdspan.hpp"
#include "core/resources.hpp"
#include "core.hpp"
#include "distance.hpp"
#include "linalg.hpp"
#include "matrix.hpp"
#include "neighbors.hpp"
#include "random.hpp"
#include "reductions.hpp"
#include "sparse.hpp"
#include "mdspan_api.hpp"
