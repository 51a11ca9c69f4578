/*
 * ref_bitstream.cpp — oracle/_ref: the REFERENCE's own ObBitStream
 * compiled standalone (cross-binary parity pin).
 *
 * This translation unit #includes the reference header IN PLACE
 * (/root/reference/src/storage/blocksstable/encoding/ob_bit_stream.h)
 * against the tiny shim headers in oracle/_ref_shim (no reference
 * sources are copied into the repo; the built .so lands in oracle/_ref/,
 * which is git-ignored but ships to the GPU box). The exported wrappers
 * let tests compare our bit-stream restatement against the reference's
 * actual set/get code, and generate committed golden vectors
 * (tests/golden/ref_bitstream.json).
 */
#include "share/ob_define.h"          /* shim */
#include "lib/utility/ob_print_utils.h" /* shim */

#include "storage/blocksstable/encoding/ob_bit_stream.h"

using oceanbase::blocksstable::ObBitStream;

extern "C" {

/* pack `value` at bit `offset` (cnt bits) into buf (the caller zeroes) */
int ref_bs_set(unsigned char *buf, int64_t buf_len, int64_t offset,
               int64_t cnt, int64_t value) {
  ObBitStream bs(buf, buf_len);
  return bs.set(offset, cnt, value);
}

int ref_bs_get(const unsigned char *buf, int64_t buf_len, int64_t offset,
               int64_t cnt, int64_t *value) {
  ObBitStream bs(const_cast<unsigned char *>(buf), buf_len);
  int64_t v = 0;
  int ret = bs.get(offset, cnt, v);
  *value = v;
  return ret;
}

/* the templated fast unpack paths (ObBitStreamUnpackType) */
int ref_bs_get_unpack(const unsigned char *buf, int64_t offset, int64_t cnt,
                      int64_t bs_len_bits, int which, int64_t *value) {
  int64_t v = 0;
  int ret;
  switch (which) {
    case 0:
      ret = ObBitStream::get<ObBitStream::PACKED_LEN_LESS_THAN_10>(
          buf, offset, cnt, bs_len_bits, v);
      break;
    case 1:
      ret = ObBitStream::get<ObBitStream::PACKED_LEN_LESS_THAN_26>(
          buf, offset, cnt, bs_len_bits, v);
      break;
    default:
      ret = ObBitStream::get<ObBitStream::DEFAULT>(buf, offset, cnt,
                                                   bs_len_bits, v);
      break;
  }
  *value = v;
  return ret;
}

}  /* extern "C" */
