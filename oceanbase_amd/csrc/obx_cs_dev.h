/*
 * obx_cs_dev.h — host<->device descriptors for the CS (cs_encoding)
 * load-time transform (obx_cs_load.cpp plans, obx_cs_kernels.hip decodes).
 *
 * A "stream task" is one unit of device work. Phase 1 tasks read the raw
 * CS bytes (src buffer); phase 2 tasks read phase-1 outputs from the
 * arena (decoded int64 arrays in the scratch tail). The host launches the
 * two phases as two kernel dispatches.
 */
#ifndef OBX_CS_DEV_H_
#define OBX_CS_DEV_H_

#if defined(__HIPRTC__)
/* not used under hipRTC */
#else
#include <stdint.h>
#endif

enum {
  /* phase 1 (read src) */
  OBX_CSOP_INT64 = 0,      /* int stream -> int64 array in arena */
  OBX_CSOP_COPY = 1,       /* raw bytes -> arena */
  OBX_CSOP_EXT_BITMAP = 2, /* MSB-first null bitmap -> LSB ext bits */
  /* phase 2 (read arena) */
  OBX_CSOP_REFS = 3,        /* int64 array -> packed 1/2-byte refs */
  OBX_CSOP_CONSTREF = 4,    /* const-ref array -> per-row packed refs */
  OBX_CSOP_I64_TO_BYTES = 5,/* int64 array -> fixed-width LE cells */
  OBX_CSOP_EXT_REPLACE = 6, /* ext bits from value == replace */
};

typedef struct cs_dev_stream {
  uint64_t src_byte;  /* phase 1: offset in the CS src buffer;
                         phase 2: offset in the arena */
  uint64_t out_byte;  /* output offset in the arena */
  uint32_t src_len;   /* byte length of the input (bounds/copy size) */
  uint32_t count;     /* values / rows */
  int64_t base;       /* INT64: stream base; EXT_REPLACE: null_replaced */
  uint32_t row0;      /* CONSTREF: first global row of this chunk */
  uint8_t op;
  uint8_t enc_type;   /* OBX_CS_ENC_* for INT64 */
  uint8_t wb;         /* element width bytes (INT64) / ref width (REFS) */
  uint8_t pad;
} cs_dev_stream;

#endif /* OBX_CS_DEV_H_ */
