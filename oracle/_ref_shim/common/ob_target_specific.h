/* SHIM (ours): the codec headers only ask is_arch_supported(AVX2); the
 * standalone pin always takes the scalar path (the
 * CPU_ARCH_INDEPENDANT_SCALAR layout our restatement follows). */
#ifndef OBX_REF_SHIM_TARGET_SPECIFIC_H_
#define OBX_REF_SHIM_TARGET_SPECIFIC_H_
namespace oceanbase {
enum class ObTargetArch { Default = 0, SSE42 = 1, AVX = 2, AVX2 = 3, AVX512 = 4 };
namespace common {
inline bool is_arch_supported(ObTargetArch) { return false; }
}
}
#endif
