/* SHIM: logging/print macros used by the reference headers compiled in
 * oracle/_ref — all no-ops here (standalone, no oblog). OURS. */
#ifndef OBX_REF_SHIM_PRINT_UTILS_H_
#define OBX_REF_SHIM_PRINT_UTILS_H_
#define K(x)
#define KP(x)
#define K_(x)
#define KP_(x)
#define STORAGE_LOG(level, ...) do {} while (0)
#define TO_STRING_KV(...)                                   \
  int64_t to_string(char *, const int64_t) const { return 0; }
#define VIRTUAL_TO_STRING_KV(...)                           \
  virtual int64_t to_string(char *, const int64_t) const { return 0; }
#define KPC(x)
#define INHERIT_TO_STRING_KV(...)                           \
  virtual int64_t to_string(char *, const int64_t) const override { return 0; }
#define KPHEX(...)
#endif
