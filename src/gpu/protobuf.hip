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
  // repeated fields -> LIST columns (packed or unpacked encodings;
  // reference protobuf_kernels.cuh:150-361 batched variants)
  PB_REP_I64 = 7,
  PB_REP_I32 = 8,
  PB_REP_BOOL = 9,
  PB_REP_SINT64 = 10,
  PB_REP_DOUBLE = 11,
  PB_REP_FLOAT = 12,
  PB_REP_BYTES = 13,     // repeated string/bytes -> LIST<STRING>
};

constexpr int PB_MAX_REPEATED = 8;

__device__ inline bool pb_is_rep(int32_t k) {
  return k >= PB_REP_I64 && k <= PB_REP_BYTES;
}

struct PbField {
  int32_t field_number;
  int32_t kind;
  int32_t rep_slot;  // REP_*: index into the per-row repeated cursors
  int32_t _pad;
  void* data;        // output fixed-width data (null for BYTES in phase 0)
  uint8_t* valid;    // output validity
  int32_t* lens;     // BYTES/REP phase 0: per-row length/count
  const int32_t* offsets;  // BYTES/REP phase 1: column/list offsets
  char* chars;       // BYTES/REP_BYTES phase 1: char output
  int32_t* lens2;    // REP_BYTES phase 0: per-row char total
  int32_t* elem_offsets;   // REP_BYTES phase 1: per-element char offsets
  const int32_t* char_base;  // REP_BYTES phase 1: per-row char base
};

// append one repeated numeric element (phase-1 store by kind)
__device__ inline void pb_store_rep(const PbField& f, int64_t at,
                                    uint64_t v, int32_t kind) {
  switch (kind) {
    case PB_REP_I64:
      reinterpret_cast<int64_t*>(f.data)[at] = (int64_t)v;
      break;
    case PB_REP_I32:
      reinterpret_cast<int32_t*>(f.data)[at] = (int32_t)v;
      break;
    case PB_REP_BOOL:
      reinterpret_cast<int8_t*>(f.data)[at] = v != 0;
      break;
    case PB_REP_SINT64:
      reinterpret_cast<int64_t*>(f.data)[at] =
          (int64_t)(v >> 1) ^ -(int64_t)(v & 1);
      break;
  }
}

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
    int32_t rcnt[PB_MAX_REPEATED];   // element counts per repeated slot
    int32_t ccnt[PB_MAX_REPEATED];   // char counts (REP_BYTES)
#pragma unroll
    for (int k = 0; k < PB_MAX_REPEATED; ++k) { rcnt[k] = 0; ccnt[k] = 0; }
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
                case PB_REP_I64:
                case PB_REP_I32:
                case PB_REP_BOOL:
                case PB_REP_SINT64: {
                  int32_t c0 = rcnt[f.rep_slot]++;
                  if (WRITE_BYTES)
                    pb_store_rep(f, (int64_t)f.offsets[row] + c0, v, f.kind);
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
            } else if (fi >= 0 && fields[fi].kind == PB_REP_DOUBLE) {
              PbField& f = fields[fi];
              present |= 1ull << fi;
              int32_t c0 = rcnt[f.rep_slot]++;
              if (WRITE_BYTES) {
                double d;
                __builtin_memcpy(&d, p + pos, 8);
                reinterpret_cast<double*>(f.data)[f.offsets[row] + c0] = d;
              }
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
            } else if (fi >= 0 && fields[fi].kind == PB_REP_FLOAT) {
              PbField& f = fields[fi];
              present |= 1ull << fi;
              int32_t c0 = rcnt[f.rep_slot]++;
              if (WRITE_BYTES) {
                float d;
                __builtin_memcpy(&d, p + pos, 4);
                reinterpret_cast<float*>(f.data)[f.offsets[row] + c0] = d;
              }
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
            } else if (fi >= 0 && fields[fi].kind >= PB_REP_I64 &&
                       fields[fi].kind <= PB_REP_SINT64) {
              // packed repeated varints
              PbField& f = fields[fi];
              present |= 1ull << fi;
              int64_t bpos = pos, bend = pos + (int64_t)blen;
              while (bpos < bend) {
                uint64_t v;
                if (!pb_varint(p, bend, &bpos, &v)) { ok = false; break; }
                int32_t c0 = rcnt[f.rep_slot]++;
                if (WRITE_BYTES)
                  pb_store_rep(f, (int64_t)f.offsets[row] + c0, v, f.kind);
              }
            } else if (fi >= 0 && fields[fi].kind == PB_REP_DOUBLE) {
              // packed fixed64 elements
              PbField& f = fields[fi];
              present |= 1ull << fi;
              for (uint64_t b = 0; b + 8 <= blen; b += 8) {
                int32_t c0 = rcnt[f.rep_slot]++;
                if (WRITE_BYTES) {
                  double d;
                  __builtin_memcpy(&d, p + pos + b, 8);
                  reinterpret_cast<double*>(f.data)[f.offsets[row] + c0] = d;
                }
              }
            } else if (fi >= 0 && fields[fi].kind == PB_REP_FLOAT) {
              // packed fixed32 elements
              PbField& f = fields[fi];
              present |= 1ull << fi;
              for (uint64_t b = 0; b + 4 <= blen; b += 4) {
                int32_t c0 = rcnt[f.rep_slot]++;
                if (WRITE_BYTES) {
                  float d;
                  __builtin_memcpy(&d, p + pos + b, 4);
                  reinterpret_cast<float*>(f.data)[f.offsets[row] + c0] = d;
                }
              }
            } else if (fi >= 0 && fields[fi].kind == PB_REP_BYTES) {
              // one element per length-delimited occurrence
              PbField& f = fields[fi];
              present |= 1ull << fi;
              int32_t c0 = rcnt[f.rep_slot]++;
              if (WRITE_BYTES) {
                int32_t off = f.char_base[row] + ccnt[f.rep_slot];
                for (uint64_t k = 0; k < blen; ++k)
                  f.chars[off + k] = (char)p[pos + k];
                f.elem_offsets[f.offsets[row] + c0] = off;
              }
              ccnt[f.rep_slot] += (int32_t)blen;
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
      if (in_range && pb_is_rep(fields[k].kind) && !WRITE_BYTES) {
        fields[k].lens[row] = fv ? rcnt[fields[k].rep_slot] : 0;
        if (fields[k].kind == PB_REP_BYTES)
          fields[k].lens2[row] = fv ? ccnt[fields[k].rep_slot] : 0;
      }
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
