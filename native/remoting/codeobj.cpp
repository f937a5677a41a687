// comgr-backed kernel-signature extraction (see codeobj.h).
#include "codeobj.h"

#include <dlfcn.h>
#include <string.h>

#include <cstdio>

namespace tfrpc {
namespace {

// ---- minimal comgr ABI (amd_comgr.h mirrored; dlopen'd) -----------------
typedef int cstatus;  // amd_comgr_status_t, 0 == success
struct cdata {
  uint64_t handle;
};
struct cdataset {
  uint64_t handle;
};
struct caction {
  uint64_t handle;
};
struct cmeta {
  uint64_t handle;
};

enum : int {
  KIND_RELOCATABLE = 0x5,
  KIND_EXECUTABLE = 0x8,
  KIND_OBJ_BUNDLE = 0x14,
  ACTION_UNBUNDLE = 0xF,
  META_STRING = 0x1,
  META_MAP = 0x2,
  META_LIST = 0x3,
};

struct Api {
  cstatus (*create_data)(int kind, cdata*);
  cstatus (*release_data)(cdata);
  cstatus (*set_data)(cdata, size_t, const char*);
  cstatus (*set_data_name)(cdata, const char*);
  cstatus (*get_data)(cdata, size_t*, char*);
  cstatus (*get_data_kind)(cdata, int*);
  cstatus (*create_data_set)(cdataset*);
  cstatus (*destroy_data_set)(cdataset);
  cstatus (*data_set_add)(cdataset, cdata);
  cstatus (*action_data_count)(cdataset, int kind, size_t*);
  cstatus (*action_data_get_data)(cdataset, int kind, size_t index, cdata*);
  cstatus (*create_action_info)(caction*);
  cstatus (*destroy_action_info)(caction);
  cstatus (*set_bundle_entry_ids)(caction, const char**, size_t);
  cstatus (*do_action)(int action, caction, cdataset, cdataset);
  cstatus (*get_data_metadata)(cdata, cmeta*);
  cstatus (*destroy_metadata)(cmeta);
  cstatus (*metadata_lookup)(cmeta, const char*, cmeta*);
  cstatus (*get_metadata_kind)(cmeta, int*);
  cstatus (*get_metadata_string)(cmeta, size_t*, char*);
  cstatus (*get_metadata_list_size)(cmeta, size_t*);
  cstatus (*index_list_metadata)(cmeta, size_t, cmeta*);
  bool ok = false;
};

Api& api() {
  static Api a = [] {
    Api x{};
    void* h = dlopen("libamd_comgr.so", RTLD_LAZY);
    if (!h) h = dlopen("libamd_comgr.so.3", RTLD_LAZY);
    if (!h) h = dlopen("libamd_comgr.so.2", RTLD_LAZY);
    if (!h) h = dlopen("/opt/rocm/lib/libamd_comgr.so", RTLD_LAZY);
    if (!h) return x;
#define R(f, sym)                                                   \
  x.f = reinterpret_cast<decltype(x.f)>(dlsym(h, sym));             \
  if (!x.f) return x;
    R(create_data, "amd_comgr_create_data")
    R(release_data, "amd_comgr_release_data")
    R(set_data, "amd_comgr_set_data")
    R(set_data_name, "amd_comgr_set_data_name")
    R(get_data, "amd_comgr_get_data")
    R(get_data_kind, "amd_comgr_get_data_kind")
    R(create_data_set, "amd_comgr_create_data_set")
    R(destroy_data_set, "amd_comgr_destroy_data_set")
    R(data_set_add, "amd_comgr_data_set_add")
    R(action_data_count, "amd_comgr_action_data_count")
    R(action_data_get_data, "amd_comgr_action_data_get_data")
    R(create_action_info, "amd_comgr_create_action_info")
    R(destroy_action_info, "amd_comgr_destroy_action_info")
    R(set_bundle_entry_ids, "amd_comgr_action_info_set_bundle_entry_ids")
    R(do_action, "amd_comgr_do_action")
    R(get_data_metadata, "amd_comgr_get_data_metadata")
    R(destroy_metadata, "amd_comgr_destroy_metadata")
    R(metadata_lookup, "amd_comgr_metadata_lookup")
    R(get_metadata_kind, "amd_comgr_get_metadata_kind")
    R(get_metadata_string, "amd_comgr_get_metadata_string")
    R(get_metadata_list_size, "amd_comgr_get_metadata_list_size")
    R(index_list_metadata, "amd_comgr_index_list_metadata")
#undef R
    x.ok = true;
    return x;
  }();
  return a;
}

std::string meta_string(cmeta m) {
  auto& a = api();
  size_t sz = 0;
  if (a.get_metadata_string(m, &sz, nullptr) != 0 || sz == 0) return "";
  std::string s(sz, '\0');
  a.get_metadata_string(m, &sz, s.data());
  if (!s.empty() && s.back() == '\0') s.pop_back();
  return s;
}

std::string lookup_string(cmeta map, const char* key) {
  auto& a = api();
  cmeta v{};
  if (a.metadata_lookup(map, key, &v) != 0) return "";
  std::string s = meta_string(v);
  a.destroy_metadata(v);
  return s;
}

uint64_t lookup_u64(cmeta map, const char* key, uint64_t dflt) {
  std::string s = lookup_string(map, key);
  if (s.empty()) return dflt;
  return strtoull(s.c_str(), nullptr, 10);
}

// Extract kernels from ONE executable data object's metadata.
bool kernels_from_exec(cdata exec, std::map<std::string, KernelSig>* out,
                       std::string* err) {
  auto& a = api();
  cmeta root{};
  if (a.get_data_metadata(exec, &root) != 0) {
    *err = "get_data_metadata failed";
    return false;
  }
  bool any = false;
  cmeta kernels{};
  if (a.metadata_lookup(root, "amdhsa.kernels", &kernels) == 0) {
    size_t n = 0;
    a.get_metadata_list_size(kernels, &n);
    for (size_t i = 0; i < n; ++i) {
      cmeta k{};
      if (a.index_list_metadata(kernels, i, &k) != 0) continue;
      KernelSig sig;
      sig.name = lookup_string(k, ".name");
      sig.kernarg_segment_size =
          (uint32_t)lookup_u64(k, ".kernarg_segment_size", 0);
      cmeta args{};
      if (a.metadata_lookup(k, ".args", &args) == 0) {
        size_t na = 0;
        a.get_metadata_list_size(args, &na);
        for (size_t j = 0; j < na; ++j) {
          cmeta arg{};
          if (a.index_list_metadata(args, j, &arg) != 0) continue;
          KernArg ka;
          ka.size = (uint32_t)lookup_u64(arg, ".size", 0);
          ka.offset = (uint32_t)lookup_u64(arg, ".offset", 0);
          std::string vk = lookup_string(arg, ".value_kind");
          ka.hidden = vk.rfind("hidden", 0) == 0;
          if (!ka.hidden) {
            sig.args.push_back(ka);
            uint32_t end = ka.offset + ka.size;
            if (end > sig.explicit_bytes) sig.explicit_bytes = end;
          }
          a.destroy_metadata(arg);
        }
        a.destroy_metadata(args);
      }
      if (!sig.name.empty()) {
        (*out)[sig.name] = std::move(sig);
        any = true;
      }
      a.destroy_metadata(k);
    }
    a.destroy_metadata(kernels);
  }
  a.destroy_metadata(root);
  if (!any) *err = "no amdhsa.kernels metadata";
  return any;
}

bool try_exec_blob(const void* image, size_t size,
                   std::map<std::string, KernelSig>* out, std::string* err) {
  auto& a = api();
  cdata d{};
  if (a.create_data(KIND_EXECUTABLE, &d) != 0) {
    *err = "create_data failed";
    return false;
  }
  bool ok = false;
  if (a.set_data(d, size, (const char*)image) == 0)
    ok = kernels_from_exec(d, out, err);
  a.release_data(d);
  return ok;
}

bool try_unbundle(const void* image, size_t size,
                  std::map<std::string, KernelSig>* out, std::string* err) {
  auto& a = api();
  cdata in{};
  if (a.create_data(KIND_OBJ_BUNDLE, &in) != 0) return false;
  a.set_data_name(in, "fatbin.bundle");
  if (a.set_data(in, size, (const char*)image) != 0) {
    a.release_data(in);
    *err = "set_data(bundle) failed";
    return false;
  }
  cdataset inset{}, outset{};
  a.create_data_set(&inset);
  a.create_data_set(&outset);
  a.data_set_add(inset, in);
  caction act{};
  a.create_action_info(&act);
  static const char* ids[] = {
      "hipv4-amdgcn-amd-amdhsa--gfx950",
      "hip-amdgcn-amd-amdhsa--gfx950",
  };
  bool ok = false;
  for (const char* id : ids) {
    const char* one[] = {id};
    if (a.set_bundle_entry_ids(act, one, 1) != 0) continue;
    if (a.do_action(ACTION_UNBUNDLE, act, inset, outset) != 0) continue;
    size_t n = 0;
    a.action_data_count(outset, KIND_EXECUTABLE, &n);
    for (size_t i = 0; i < n && !ok; ++i) {
      cdata e{};
      if (a.action_data_get_data(outset, KIND_EXECUTABLE, i, &e) != 0)
        continue;
      size_t esz = 0;
      a.get_data(e, &esz, nullptr);
      if (esz > 64) ok = kernels_from_exec(e, out, err);
      a.release_data(e);
    }
    if (ok) break;
  }
  a.destroy_action_info(act);
  a.destroy_data_set(inset);
  a.destroy_data_set(outset);
  a.release_data(in);
  if (!ok && err->empty()) *err = "unbundle produced no gfx950 executable";
  return ok;
}

// Uncompressed clang offload bundle: extract gfx950 slice directly.
bool try_plain_bundle(const void* image, size_t size,
                      std::map<std::string, KernelSig>* out,
                      std::string* err) {
  static const char MAGIC[] = "__CLANG_OFFLOAD_BUNDLE__";
  const uint8_t* p = (const uint8_t*)image;
  if (size < 32 || memcmp(p, MAGIC, 24) != 0) return false;
  uint64_t n;
  memcpy(&n, p + 24, 8);
  size_t off = 32;
  for (uint64_t i = 0; i < n && off + 24 <= size; ++i) {
    uint64_t eoff, esz, tlen;
    memcpy(&eoff, p + off, 8);
    memcpy(&esz, p + off + 8, 8);
    memcpy(&tlen, p + off + 16, 8);
    off += 24;
    if (off + tlen > size) break;
    std::string triple((const char*)p + off, tlen);
    off += tlen;
    if (triple.find("gfx950") != std::string::npos && eoff + esz <= size &&
        esz > 64) {
      return try_exec_blob(p + eoff, esz, out, err);
    }
  }
  return false;
}

}  // namespace

bool parse_kernel_signatures(const void* image, size_t size,
                             std::map<std::string, KernelSig>* out,
                             std::string* err) {
  err->clear();
  if (!api().ok) {
    *err = "libamd_comgr unavailable";
    return false;
  }
  const uint8_t* p = (const uint8_t*)image;
  if (size >= 4 && memcmp(p, "\x7f" "ELF", 4) == 0)
    return try_exec_blob(image, size, out, err);
  if (try_plain_bundle(image, size, out, err)) return true;
  return try_unbundle(image, size, out, err);
}

}  // namespace tfrpc
