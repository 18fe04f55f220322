// NativeLane: the GIL-free engram fast lane between the bobraccel DAG
// core (_core.so) and the HIP kernel library (_hipops.so).
//
// Role (VERDICT r1 #2 / SURVEY §2.6): the reference runs engram bodies in
// separate pods; round 1 ran them as Python callables on worker threads —
// correct, but the Python body (hydrate, context, dispatch, offload) is
// GIL-serialized and dominates sub-millisecond stories.  This lane lets
// the core dispatch BUILT-IN engrams (embed, allgather-join, ...) straight
// into C++/HIP: the launch enqueues kernels on a pool stream and returns a
// ticket; the core's loop thread polls tickets (hipEventQuery behind the
// function pointer) and completes steps without ever taking the GIL.
//
// The struct crosses the .so boundary as a raw pointer inside a Python
// capsule; JValue is the shared header-only payload type (both modules
// compile against the same libstdc++, Itanium ABI).
#pragma once

#include <string>

#include "jvalue.h"

namespace bobraccel {

struct NativeLane {
  int version = 1;
  void* self = nullptr;
  // Launch engram work of `kind` with the step's resolved input and the
  // engram's config.  Returns a ticket (>0), or <=0 when this input is
  // not nativable (the caller falls back to the host-language launcher).
  long (*launch)(void* self, int kind, const JValue* cfg, const JValue* input,
                 int device) = nullptr;
  // Poll a ticket: 0 = pending; 1 = done (out filled; ticket consumed);
  // -1 = failed (err filled; ticket consumed; exit class terminal).
  int (*poll)(void* self, long ticket, JValue* out, std::string* err) = nullptr;
  // Drop a lane-held payload (a "$storageRef" key minted by this lane).
  void (*free_key)(void* self, const char* key) = nullptr;
};

}  // namespace bobraccel
