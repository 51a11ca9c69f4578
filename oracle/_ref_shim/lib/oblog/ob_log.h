/* SHIM (ours) */
#ifndef OBX_REF_SHIM_LOG_H_
#define OBX_REF_SHIM_LOG_H_
#include "lib/oblog/ob_log_module.h"
#endif
