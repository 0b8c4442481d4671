// GPU protobuf decode: serialized message per row -> typed columns.
//
// Reference parity: Protobuf.java:25-50 (4-pass design: scan/count ->
// prefix-sum offsets -> extract -> build), protobuf_kernels.cu
// (extract_varint_kernel / extract_fixed_kernel / enum validate),
// ProtobufSchemaDescriptor.java (host-side flattened field tables).
//
// v1 scope: top-level scalar fields (varint int64/int32/bool/enum with
// optional zigzag, fixed64 double, fixed32 float, length-delimited
// string/bytes); last-one-wins per proto3; unknown fields skipped; nested
// messages are skipped as length-delimited blobs (extractable as BYTES).
#include "srj_common.hpp"

namespace srj {

enum PbKind : int32_t {
  PB_VARINT_I64 = 0,
  PB_VARINT_I32 = 1,
  PB_VARINT_BOOL = 2,
  PB_VARINT_SINT64 = 3,  // zigzag
  PB_FIXED64_DOUBLE = 4,
  PB_FIXED32_FLOAT = 5,
  PB_BYTES = 6,          // string/bytes/submessage blob
  PB_REP_I64 = 7,        // repeated int64 varint (packed or unpacked) -> LIST
};

constexpr int PB_MAX_REPEATED = 8;

struct PbField {
  int32_t field_number;
  int32_t kind;
  int32_t rep_slot;  // REP_*: index into the per-row repeated cursors
  int32_t _pad;
  void* data;        // output fixed-width data (null for BYTES in phase 0)
  uint8_t* valid;    // output validity
  int32_t* lens;     // BYTES/REP phase 0: per-row length/count
  const int32_t* offsets;  // BYTES/REP phase 1: column offsets
  char* chars;       // BYTES phase 1
};

__device__ inline bool pb_varint(const uint8_t* p, int64_t len, int64_t* pos,
                                 uint64_t* out) {
  uint64_t v = 0;
  int shift = 0;
  while (*pos < len && shift < 64) {
    uint8_t b = p[(*pos)++];
    v |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) {
      *out = v;
      return true;
    }
    shift += 7;
  }
  return false;
}

template <bool WRITE_BYTES>
__global__ void pb_decode_kernel(ColDesc in, int64_t nrows,
                                 PbField* __restrict__ fields, int32_t nfields,
                                 uint8_t* __restrict__ row_ok) {
  int64_t stride = (int64_t)gridDim.x * blockDim.x;
  int64_t npad = (nrows + WAVE - 1) & ~(int64_t)(WAVE - 1);
  for (int64_t row = (int64_t)blockIdx.x * blockDim.x + threadIdx.x; row < npad;
       row += stride) {
    bool in_range = row < nrows;
    bool valid = in_range && is_valid(in.valid, row);
    // per-field presence this row (up to 64 fields via bitmask)
    uint64_t present = 0;
    int32_t rcnt[PB_MAX_REPEATED];
#pragma unroll
    for (int k = 0; k < PB_MAX_REPEATED; ++k) rcnt[k] = 0;
    bool ok = valid;
    if (valid) {
      StrView s = get_string(in, row);
      const uint8_t* p = reinterpret_cast<const uint8_t*>(s.ptr);
      int64_t pos = 0, len = s.len;
      while (pos < len && ok) {
        uint64_t tag;
        if (!pb_varint(p, len, &pos, &tag)) { ok = false; break; }
        int32_t fnum = (int32_t)(tag >> 3);
        int wire = (int)(tag & 7);
        // find schema field (nfields is small; linear scan)
        int fi = -1;
        for (int k = 0; k < nfields; ++k)
          if (fields[k].field_number == fnum) { fi = k; break; }
        switch (wire) {
          case 0: {  // varint
            uint64_t v;
            if (!pb_varint(p, len, &pos, &v)) { ok = false; break; }
            if (fi >= 0) {
              PbField& f = fields[fi];
              present |= 1ull << fi;
              switch (f.kind) {
                case PB_REP_I64: {
                  int32_t c0 = rcnt[f.rep_slot]++;
                  if (WRITE_BYTES)
                    reinterpret_cast<int64_t*>(f.data)[f.offsets[row] + c0] =
                        (int64_t)v;
                  break;
                }
                case PB_VARINT_I64:
                  reinterpret_cast<int64_t*>(f.data)[row] = (int64_t)v;
                  break;
                case PB_VARINT_I32:
                  reinterpret_cast<int32_t*>(f.data)[row] = (int32_t)v;
                  break;
                case PB_VARINT_BOOL:
                  reinterpret_cast<int8_t*>(f.data)[row] = v != 0;
                  break;
                case PB_VARINT_SINT64:
                  reinterpret_cast<int64_t*>(f.data)[row] =
                      (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
                  break;
                default:
                  present &= ~(1ull << fi);  // wire/kind mismatch: skip
              }
            }
            break;
          }
          case 1: {  // fixed64
            if (pos + 8 > len) { ok = false; break; }
            if (fi >= 0 && fields[fi].kind == PB_FIXED64_DOUBLE) {
              double d;
              __builtin_memcpy(&d, p + pos, 8);
              reinterpret_cast<double*>(fields[fi].data)[row] = d;
              present |= 1ull << fi;
            }
            pos += 8;
            break;
          }
          case 5: {  // fixed32
            if (pos + 4 > len) { ok = false; break; }
            if (fi >= 0 && fields[fi].kind == PB_FIXED32_FLOAT) {
              float d;
              __builtin_memcpy(&d, p + pos, 4);
              reinterpret_cast<float*>(fields[fi].data)[row] = d;
              present |= 1ull << fi;
            }
            pos += 4;
            break;
          }
          case 2: {  // length-delimited
            uint64_t blen;
            if (!pb_varint(p, len, &pos, &blen) || pos + (int64_t)blen > len) {
              ok = false;
              break;
            }
            if (fi >= 0 && fields[fi].kind == PB_BYTES) {
              PbField& f = fields[fi];
              present |= 1ull << fi;
              if (WRITE_BYTES) {
                int32_t o = f.offsets[row];
                for (uint64_t k = 0; k < blen; ++k)
                  f.chars[o + k] = (char)p[pos + k];
              } else {
                f.lens[row] = (int32_t)blen;
              }
            } else if (fi >= 0 && fields[fi].kind == PB_REP_I64) {
              // packed repeated varints
              PbField& f = fields[fi];
              present |= 1ull << fi;
              int64_t bpos = pos, bend = pos + (int64_t)blen;
              while (bpos < bend) {
                uint64_t v;
                if (!pb_varint(p, bend, &bpos, &v)) { ok = false; break; }
                int32_t c0 = rcnt[f.rep_slot]++;
                if (WRITE_BYTES)
                  reinterpret_cast<int64_t*>(f.data)[f.offsets[row] + c0] =
                      (int64_t)v;
              }
            }
            pos += blen;
            break;
          }
          default:
            ok = false;  // wire types 3/4 (groups) unsupported
        }
      }
    }
    if (in_range && row_ok) row_ok[row] = ok;
    // validity: field present AND row parsed ok
    for (int k = 0; k < nfields; ++k) {
      bool fv = ok && ((present >> k) & 1);
      ballot_write_validity(fields[k].valid, row, fv);
      if (in_range && !fv && fields[k].kind == PB_BYTES && !WRITE_BYTES)
        fields[k].lens[row] = 0;
      if (in_range && fields[k].kind == PB_REP_I64 && !WRITE_BYTES)
        fields[k].lens[row] = fv ? rcnt[fields[k].rep_slot] : 0;
    }
  }
}

}  // namespace srj

using namespace srj;

extern "C" {

void srj_pb_decode(const void* in, int64_t nrows, void* fields, int32_t nfields,
                   uint8_t* row_ok, int32_t write_bytes, hipStream_t stream) {
  ColDesc c = *reinterpret_cast<const ColDesc*>(in);
  if (write_bytes)
    pb_decode_kernel<true><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, reinterpret_cast<PbField*>(fields), nfields, row_ok);
  else
    pb_decode_kernel<false><<<grid_1d(nrows), DEFAULT_BLOCK, 0, stream>>>(
        c, nrows, reinterpret_cast<PbField*>(fields), nfields, row_ok);
}

}  // extern "C"
