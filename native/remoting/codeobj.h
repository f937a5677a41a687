// Code-object introspection for the remoting worker: kernarg layouts from
// amdgcn ELF metadata (NT_AMDGPU_METADATA msgpack, read via libamd_comgr).
// Needed to re-pack hipLaunchKernel's void** args into the flat kernarg
// buffer the worker launches with (hipModuleLaunchKernel +
// HIP_LAUNCH_PARAM_BUFFER_POINTER).
#pragma once

#include <cstdint>
#include <map>
#include <string>
#include <vector>

namespace tfrpc {

struct KernArg {
  uint32_t size;
  uint32_t offset;
  bool hidden;  // hidden_* value kinds (runtime-filled; not packed by client)
};

struct KernelSig {
  std::string name;
  uint32_t kernarg_segment_size = 0;  // full segment incl. hidden args
  uint32_t explicit_bytes = 0;        // bytes the client must pack
  std::vector<KernArg> args;          // explicit args only, ordered
};

// Parse every kernel signature out of `image` (raw amdgcn ELF executable or
// a clang offload bundle — compressed CCOB handled via comgr unbundling).
// Returns false (with err filled) when nothing could be parsed.
bool parse_kernel_signatures(const void* image, size_t size,
                             std::map<std::string, KernelSig>* out,
                             std::string* err);

}  // namespace tfrpc
