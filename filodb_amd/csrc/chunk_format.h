/* Shared constants/structs for the frozen BinaryVector chunk format (host + device).
 *
 * The byte layouts reproduce the reference's off-heap frozen vectors exactly
 * (DESIGN.md §2):
 *   WireFormat word  = (subType << 8) | majorType, 16 bits at +4
 *                      (core/.../format/WireFormat.scala:51-53)
 *   length word at +0 counts the bytes AFTER it (BinaryVector.scala:115)
 *   primitive header: +4 u16 wf; +6 u8 nbits|signed<<7; +7 u8 bitShift; +8 data
 *                      (BinaryVector.scala:511-533)
 *   drop/reset bit   = bit 15 of the u16 at +6 (PrimitiveVector.DropMask,
 *                      BinaryVector.scala:519)
 */
#ifndef FDB_CHUNK_FORMAT_H
#define FDB_CHUNK_FORMAT_H

#include <stdint.h>

/* WireFormat major types (core/.../format/WireFormat.scala:8-17) */
#define FDB_VECTORTYPE_BINSIMPLE  0x06
#define FDB_VECTORTYPE_DELTA2     0x08
#define FDB_VECTORTYPE_HISTOGRAM  0x09
/* sub types (WireFormat.scala:25-38) */
#define FDB_SUBTYPE_PRIMITIVE        0x00
#define FDB_SUBTYPE_PRIMITIVE_NOMASK 0x05
#define FDB_SUBTYPE_REPEATED         0x06
#define FDB_SUBTYPE_INT              0x07
#define FDB_SUBTYPE_INT_NOMASK       0x08
#define FDB_SUBTYPE_H_SECTDELTA      0x12

/* Sect-delta histogram vector (HistogramVector.scala:237-254,491-545):
 *  +0  i32 length word (bytes after it)
 *  +4  u16 wireformat 0x1209
 *  +6  u16 numHistograms
 *  +8  u8  format code (0x03 = geometric + NibblePacked delta longs)
 *  +9  u16 bucketDefNumBytes
 *  +11 bucket def: u16 numBuckets, f64 firstBucket, f64 multiplier (geometric)
 *  +11+def: sections; each section (Section.scala:17-24):
 *     +0 u16 bytes after 4-byte header; +2 u8 numElements; +3 u8 type(0|1=drop)
 *     elements: u16 len + NibblePack stream */
#define FDB_WF_HIST_SECTDELTA 0x1209
#define FDB_HIST_FMT_GEOMETRIC_DELTA 0x03
#define FDB_HIST_OFF_NUMHIST   6
#define FDB_HIST_OFF_FMT       8
#define FDB_HIST_OFF_DEFSIZE   9
#define FDB_HIST_OFF_DEF       11
#define FDB_HIST_MAX_PER_SECTION 16   /* AppendableSectDeltaHistVector :501 */

#define FDB_WF(major, sub) ((uint16_t)((((sub) & 0xff) << 8) | ((major) & 0xff)))

/* Composite wireformat words actually produced on this path */
#define FDB_WF_DDV        FDB_WF(FDB_VECTORTYPE_DELTA2, FDB_SUBTYPE_INT_NOMASK)    /* 0x0808 */
#define FDB_WF_DDV_CONST  FDB_WF(FDB_VECTORTYPE_DELTA2, FDB_SUBTYPE_REPEATED)      /* 0x0608 */
#define FDB_WF_PRIM64     FDB_WF(FDB_VECTORTYPE_BINSIMPLE, FDB_SUBTYPE_PRIMITIVE_NOMASK) /* 0x0506 */
#define FDB_WF_INT_NOMASK FDB_WF(FDB_VECTORTYPE_BINSIMPLE, FDB_SUBTYPE_INT_NOMASK) /* 0x0806 */

#define FDB_NBITS_MASK 0x7f
#define FDB_SIGN_MASK  0x80
#define FDB_DROP_MASK  0x8000

/* DDV header offsets (DeltaDeltaVector.scala:138-146) */
#define FDB_DDV_OFF_INIT   8
#define FDB_DDV_OFF_SLOPE  16
#define FDB_DDV_OFF_INNER  20
/* const DDV (DeltaDeltaVector.scala:89-106) */
#define FDB_DDVC_OFF_NELEM 8
#define FDB_DDVC_OFF_INIT  12
#define FDB_DDVC_OFF_SLOPE 20
#define FDB_DDVC_BYTES     24
/* primitive vector (BinaryVector.scala:511-517) */
#define FDB_PRIM_OFF_DATA  8

/* approx-const acceptance band for timestamp DDV (DeltaDeltaVector.scala:46-47) */
#define FDB_DDV_MAX_APPROX_DELTA  250
/* default chunk row cap (conf/timeseries-filodb-server: block max-chunk-size) */
#define FDB_DEFAULT_MAX_ROWS 400

/* Chunk directory entry: the cached ChunkSetInfoReader fields
 * (core/.../store/ChunkSetInfoReader.scala:53-66; record layout
 *  ChunkSetInfo.scala:133-154). Offsets index the dataset's contiguous blob. */
typedef struct {
  uint64_t ts_off;
  uint64_t val_off;
  int64_t  start_time;
  int64_t  end_time;
  int32_t  num_rows;
  int32_t  _pad;
  /* histogram companion double columns (otel max/min schema,
   * AggrOverTimeFunctions.scala:612-813); 0 = column absent. The first
   * vector in the blob is always a timestamp vector, so offset 0 is never a
   * companion. */
  uint64_t max_off;
  uint64_t min_off;
} fdb_dir_entry_t;

#endif /* FDB_CHUNK_FORMAT_H */
