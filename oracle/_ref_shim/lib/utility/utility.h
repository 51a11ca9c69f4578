/* SHIM (ours): minimal surface for the codec sources */
#ifndef OBX_REF_SHIM_UTILITY_H_
#define OBX_REF_SHIM_UTILITY_H_
#include "share/ob_define.h"
#include "lib/ob_errno.h"
#include "lib/oblog/ob_log_module.h"
#include "lib/utility/ob_print_utils.h"
#include <stdint.h>
/* oblib's global fixed-width typedefs (lib/ob_define.h) */
typedef uint8_t uint8;
typedef uint16_t uint16;
typedef uint32_t uint32;
typedef uint64_t uint64;
typedef int8_t int8;
typedef int16_t int16;
typedef int32_t int32;
using namespace oceanbase::common;
#endif
