/* SHIM (ours): error codes used by the codec headers */
#ifndef OBX_REF_SHIM_ERRNO_H_
#define OBX_REF_SHIM_ERRNO_H_
#include <limits.h>
#include <stdint.h>
#include <stddef.h>
namespace oceanbase { namespace common {
constexpr int OB_SUCCESS = 0;
constexpr int OB_ERROR = -4000;
constexpr int OB_INVALID_ARGUMENT = -4002;
constexpr int OB_BUF_NOT_ENOUGH = -4009;
constexpr int OB_NOT_SUPPORTED = -4007;
constexpr int OB_ERR_UNEXPECTED = -4016;
constexpr int OB_ALLOCATE_MEMORY_FAILED = -4013;
constexpr int OB_DESERIALIZE_ERROR = -4034;
constexpr int OB_NOT_INIT = -4006;
constexpr int OB_INDEX_OUT_OF_RANGE = -4008;
/* minimal allocator interface for the codec headers (they only keep a
 * pointer; the wrapper never exercises allocator paths) */
class ObIAllocator {
 public:
  virtual ~ObIAllocator() {}
  virtual void *alloc(int64_t) { return nullptr; }
  virtual void free(void *) {}
};
} }
#endif
