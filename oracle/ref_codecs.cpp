/*
 * ref_codecs.cpp — oracle/_ref: the REFERENCE's integer-stream codec
 * family (deps/oblib/src/lib/codec) compiled standalone against the
 * oracle/_ref_shim headers — the cross-binary pin for the CS
 * (cs_encoding) stream layer.
 *
 * Headers are included from /root/reference IN PLACE (nothing copied).
 * Exports one encode/decode pair dispatching on the
 * ObIntegerStream::EncodingType codes our container stores
 * (oracle/obx_cs.h), always in the CPU_ARCH_INDEPENDANT_SCALAR layout
 * (the codecs' default since 4.3.3 — ob_codecs.h:82).
 */
#include "share/ob_define.h"
#include "lib/utility/ob_print_utils.h"

#include "lib/codec/ob_delta_zigzag_rle.h"
#include "lib/codec/ob_double_delta_zigzag_rle.h"
#include "lib/codec/ob_delta_zigzag_pfor.h"
#include "lib/codec/ob_double_delta_zigzag_pfor.h"
#include "lib/codec/ob_simd_fixed_pfor.h"
#include "lib/codec/ob_xor_fixed_pfor.h"

using namespace oceanbase::common;

static ObCodec *codec_for(int enc_type) {
  static thread_local ObDeltaZigzagRle dzr;
  static thread_local ObDoubleDeltaZigzagRle ddzr;
  static thread_local ObDeltaZigzagPFor dzp;
  static thread_local ObDoubleDeltaZigzagPFor ddzp;
  static thread_local ObSIMDFixedPFor fpfor;
  static thread_local ObXorFixedPfor xpfor;
  switch (enc_type) { /* ObIntegerStream::EncodingType values */
    case 2: return &ddzr;
    case 3: return &ddzp;
    case 4: return &dzr;
    case 5: return &dzp;
    case 6: return &fpfor;
    case 8: return &xpfor;
    default: return nullptr;
  }
}

extern "C" {

/* encode `count` fixed-width (wb bytes) uints; returns bytes written or
 * a negative reference error code */
int64_t ref_codec_encode(int enc_type, const unsigned char *in,
                         uint64_t count, int wb, unsigned char *out,
                         uint64_t out_cap) {
  ObCodec *c = codec_for(enc_type);
  if (!c) return -1;
  c->set_uint_bytes((uint8_t)wb);
  c->set_pfor_packing_type(ObCodec::CPU_ARCH_INDEPENDANT_SCALAR);
  uint64_t out_pos = 0;
  int ret = c->encode((const char *)in, count * (uint64_t)wb, (char *)out,
                      out_cap, out_pos);
  return ret == OB_SUCCESS ? (int64_t)out_pos : (int64_t)ret;
}

int64_t ref_codec_decode(int enc_type, const unsigned char *in,
                         uint64_t in_len, uint64_t count, int wb,
                         unsigned char *out, uint64_t out_cap) {
  ObCodec *c = codec_for(enc_type);
  if (!c) return -1;
  c->set_uint_bytes((uint8_t)wb);
  c->set_pfor_packing_type(ObCodec::CPU_ARCH_INDEPENDANT_SCALAR);
  uint64_t in_pos = 0, out_pos = 0;
  int ret = c->decode((const char *)in, in_len, in_pos, count, (char *)out,
                      out_cap, out_pos);
  return ret == OB_SUCCESS ? (int64_t)in_pos : (int64_t)ret;
}

}  /* extern "C" */
