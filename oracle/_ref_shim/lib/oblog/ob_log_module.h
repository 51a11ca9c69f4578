/* SHIM (ours): no-op logging */
#ifndef OBX_REF_SHIM_LOG_MODULE_H_
#define OBX_REF_SHIM_LOG_MODULE_H_
#define LIB_LOG(...) do {} while (0)
#define COMMON_LOG(...) do {} while (0)
#define SHARE_LOG(...) do {} while (0)
#define _LIB_LOG(...) do {} while (0)
#endif
