/* SHIM (ours): the codec headers reference the compressor-type enum */
#ifndef OBX_REF_SHIM_COMPRESS_UTIL_H_
#define OBX_REF_SHIM_COMPRESS_UTIL_H_
namespace oceanbase { namespace common {
enum ObCompressorType {
  INVALID_COMPRESSOR = 0,
  NONE_COMPRESSOR = 1,
  LZ4_COMPRESSOR = 2,
  SNAPPY_COMPRESSOR = 3,
  ZLIB_COMPRESSOR = 4,
  ZSTD_COMPRESSOR = 5,
};
} }
#endif
