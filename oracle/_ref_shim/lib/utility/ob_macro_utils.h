/* SHIM (ours) */
#ifndef OBX_REF_SHIM_MACRO_UTILS_H_
#define OBX_REF_SHIM_MACRO_UTILS_H_
#include "share/ob_define.h"
#include "lib/ob_errno.h"
namespace oceanbase { namespace common {} }
using namespace oceanbase::common;
#define OB_FAIL(x) __builtin_expect((ret = (x)) != ::oceanbase::common::OB_SUCCESS, 0)
#define OB_SUCC(x) __builtin_expect((ret = (x)) == ::oceanbase::common::OB_SUCCESS, 1)
#define OB_ISNULL(x) __builtin_expect((x) == nullptr, 0)
#define OB_NOT_NULL(x) __builtin_expect((x) != nullptr, 1)
#define LST_DO_CODE(...)
#define MEMCPY(dst, src, n) memcpy((dst), (src), (n))
#define MEMSET(p, v, n) memset((p), (v), (n))
#define MEMMOVE(dst, src, n) memmove((dst), (src), (n))
#define FAILEDx(x) \
  __builtin_expect((ret == ::oceanbase::common::OB_SUCCESS) && \
                       ((ret = (x)) != ::oceanbase::common::OB_SUCCESS), 0)
#endif
