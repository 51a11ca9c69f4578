/* SHIM header for the oracle/_ref standalone compile of reference leaf
 * codecs (oracle/ref_bitstream.cpp). The reference's real ob_define.h
 * drags in the whole oblib; ob_bit_stream.h only needs the macros and
 * error codes below. This file is OURS (no reference code copied). */
#ifndef OBX_REF_SHIM_OB_DEFINE_H_
#define OBX_REF_SHIM_OB_DEFINE_H_
#include <stdint.h>
#include <stddef.h>
#include <algorithm>
#include <cstring>

#define OB_INLINE inline __attribute__((always_inline))
#define OB_NOINLINE __attribute__((noinline))
#define OB_LIKELY(x) __builtin_expect(!!(x), 1)
#define OB_UNLIKELY(x) __builtin_expect(!!(x), 0)
#define UNUSED(x) ((void)(x))
/* the reference's own definition: co_var.h:26
 * "#define RLOCAL_INLINE(TYPE, VAR) thread_local TYPE VAR
#define DISALLOW_COPY_AND_ASSIGN(T) \
  T(const T &) = delete;            \
  T &operator=(const T &) = delete" */
#define RLOCAL_INLINE(TYPE, VAR) thread_local TYPE VAR
#define DISALLOW_COPY_AND_ASSIGN(T) \
  T(const T &) = delete;            \
  T &operator=(const T &) = delete

#include "lib/ob_errno.h"
#endif
