// tf_codeobj_dump <file> — prints parsed kernel signatures as JSON
// (tests/test_codeobj.py drives it against a hipcc --genco artifact).
#include <stdio.h>
#include <stdlib.h>

#include "codeobj.h"

int main(int argc, char** argv) {
  if (argc < 2) return 2;
  FILE* f = fopen(argv[1], "rb");
  if (!f) return 2;
  fseek(f, 0, SEEK_END);
  long sz = ftell(f);
  fseek(f, 0, SEEK_SET);
  char* buf = (char*)malloc(sz);
  if (fread(buf, 1, sz, f) != (size_t)sz) return 2;
  fclose(f);
  std::map<std::string, tfrpc::KernelSig> sigs;
  std::string err;
  if (!tfrpc::parse_kernel_signatures(buf, sz, &sigs, &err)) {
    printf("{\"error\": \"%s\"}\n", err.c_str());
    return 1;
  }
  printf("{");
  bool first = true;
  for (auto& [name, s] : sigs) {
    printf("%s\"%s\": {\"kernarg_size\": %u, \"explicit\": %u, \"args\": [",
           first ? "" : ", ", name.c_str(), s.kernarg_segment_size,
           s.explicit_bytes);
    for (size_t i = 0; i < s.args.size(); ++i)
      printf("%s[%u, %u]", i ? ", " : "", s.args[i].size, s.args[i].offset);
    printf("]}");
    first = false;
  }
  printf("}\n");
  return 0;
}
