/* filodb_amd — C-ABI for the MI355X-native FiloDB chunk-scan + range-vector engine.
 *
 * This is the drop-in boundary. Each entry point names the
 * reference interface it replaces. The shape follows the reference's own native-crossing
 * precedent: SimdNativeMethods.simdSumDouble(dataAddr, start, end, ignoreNaN) — raw
 * pointers in, numbers out, no exceptions across the boundary
 * (core/src/main/scala/filodb.memory/format/vectors/SimdNativeMethods.scala:75,
 *  core/src/rust/filodb_core/src/simd_vectors.rs:174-213).
 *
 * Ownership: the caller owns every buffer it passes; the engine owns its scratch and
 * device memory. Errors: negative int return + thread-local message via fdb_last_error().
 */
#ifndef FILODB_AMD_H
#define FILODB_AMD_H

#include <stdint.h>
#include <stddef.h>

#ifdef __cplusplus
extern "C" {
#endif

/* ---- error handling ------------------------------------------------------ */
/* Returns the thread-local message for the last failing fdb_* call. */
const char* fdb_last_error(void);

#define FDB_OK            0
#define FDB_ERR          -1   /* generic; see fdb_last_error() */
#define FDB_ERR_NOGPU    -2   /* compute entry point called with no HIP device */
#define FDB_ERR_BADARG   -3

/* ---- chunk store (host): builds the reference's frozen chunk format ------ */
/* Replaces the ingest/encode path TimeSeriesPartition.switchBuffers(encode=true) →
 * optimizeAll (core/.../memstore/TimeSeriesPartition.scala:130; encoding decisions:
 * DeltaDeltaVector.scala:63-85, DoubleVector.scala:86-96,457-476,
 * LongBinaryVector.scala:333-341). Host-side C++; produces bit-identical frozen
 * BinaryVector bytes plus the chunk directory (ChunkSetInfoReader cached fields,
 * core/.../store/ChunkSetInfoReader.scala:53-66). */
typedef struct fdb_store fdb_store_t;

/* series value temporality / column kind */
#define FDB_COL_GAUGE    0   /* plain double column                        */
#define FDB_COL_COUNTER  1   /* drop-detecting counter column (DoubleCounterAppender) */
#define FDB_COL_HIST     2   /* sect-delta histogram column (AppendableSectDeltaHistVector,
                                core/.../vectors/HistogramVector.scala:491-545) */

fdb_store_t* fdb_store_create(int64_t expected_series);
void         fdb_store_destroy(fdb_store_t* s);

/* Adds a series; returns its dense series id (>=0). group_id is the precomputed
 * cross-series aggregation key (the label-subset key of AggregateMapReduce,
 * query/.../exec/AggrOverRangeVectors.scala:150-159, resolved by the host). */
int32_t fdb_store_add_series(fdb_store_t* s, int32_t group_id, int32_t col_kind);

/* Appends samples to a series' write buffer. NaN values allowed (Prometheus stale
 * markers). Timestamps must be nondecreasing. */
int32_t fdb_series_append(fdb_store_t* s, int32_t series_id,
                          const int64_t* ts, const double* vals, int32_t n);

/* Appends histogram samples to a FDB_COL_HIST series. bucket_values is
 * row-major [n × num_buckets] cumulative-LE counts (each bucket an increasing
 * counter). The bucket scheme is geometric: top(i) = first * mult^i
 * (GeometricBuckets, core/.../vectors/Histogram.scala:609-626); it must be
 * identical across a series. Encoded per the reference's sect-delta format:
 * sections of <=16 histograms, first element NibblePack-delta packed raw,
 * later elements packed as bucket-delta diffs vs the section base, counter
 * drops forcing a TypeDrop section (HistogramVector.scala:491-545,
 * NibblePack.scala:304-350, Section.scala). */
int32_t fdb_series_append_hist(fdb_store_t* s, int32_t series_id,
                               const int64_t* ts, const uint64_t* bucket_values,
                               int32_t n, int32_t num_buckets,
                               double bucket_first, double bucket_mult);

/* Forces a chunk boundary: encodes the write buffer into frozen vectors
 * (= switchBuffers(encode=true)). Called automatically when a buffer reaches
 * max_rows (default 400 = filodb-defaults.conf:835 sourced chunk cap). */
int32_t fdb_series_cut_chunk(fdb_store_t* s, int32_t series_id);
int32_t fdb_store_set_max_rows(fdb_store_t* s, int32_t max_rows);
/* Encodes every series' outstanding buffer. Call before upload/inspect. */
int32_t fdb_store_seal(fdb_store_t* s);

/* -- introspection (tests, oracle, upload) -- */
int32_t  fdb_store_num_series(const fdb_store_t* s);
int32_t  fdb_series_num_chunks(const fdb_store_t* s, int32_t series_id);
/* Chunk directory record — the cached ChunkSetInfoReader fields. */
typedef struct {
  const uint8_t* ts_vec;    /* frozen timestamp BinaryVector bytes  */
  const uint8_t* val_vec;   /* frozen value BinaryVector bytes      */
  int32_t  num_rows;
  int64_t  start_time;      /* first timestamp in chunk  */
  int64_t  end_time;        /* last timestamp in chunk   */
  int32_t  ts_vec_len;      /* total bytes incl. 4B length word     */
  int32_t  val_vec_len;
} fdb_chunk_info_t;
int32_t fdb_chunk_get(const fdb_store_t* s, int32_t series_id, int32_t chunk_idx,
                      fdb_chunk_info_t* out);

/* Flattened whole-store view over a sealed store: one contiguous blob of frozen
 * vector bytes plus the SoA chunk directory. This is exactly what
 * fdb_dataset_upload copies to HBM, and what the test-side oracle reads.
 * Pointers remain owned by the store and valid until it is mutated/destroyed.
 * dir entries are `fdb_dir_entry_t` as defined in filodb_amd/csrc/chunk_format.h
 * ({u64 ts_off, u64 val_off, i64 start_time, i64 end_time, i32 num_rows, i32 pad}). */
typedef struct {
  const uint8_t* blob;
  int64_t        blob_len;
  const void*    dir;            /* fdb_dir_entry_t[num_chunks]  */
  int64_t        num_chunks;
  const int32_t* series_first;   /* first chunk index per series */
  const int32_t* series_nchunks;
  const int32_t* group_ids;      /* per series                   */
  int32_t        num_series;
  int32_t        _pad;
} fdb_view_t;
int32_t fdb_store_view(const fdb_store_t* s, fdb_view_t* out);

/* Synthetic workload generator (host, parallel): reproduces the data shapes of
 * the reference's TestTimeseriesProducer gauge/counter series
 * (gateway/src/main/scala/filodb/timeseries/TestTimeseriesProducer.scala:75-199)
 * at BASELINE.json config scale. kind: FDB_COL_COUNTER → cumulative Poisson(lam)
 * counts with reset probability reset_p per sample; FDB_COL_GAUGE → non-integral
 * random walk. Timestamps jittered ±jitter_ms around the step grid. Series s
 * gets group id s % n_groups. Deterministic in seed. */
int32_t fdb_synth_generate(fdb_store_t* s, int32_t kind, int32_t n_series,
                           int32_t n_samples, int64_t start_ts, int32_t step_ms,
                           int32_t jitter_ms, double lam, double reset_p,
                           int32_t n_groups, uint64_t seed);

/* ---- standalone codec entry points (host; tests/golden-vector parity) ---- */
/* NibblePack.pack8 / unpack8 (core/.../format/NibblePack.scala:108-183,395-447). */
int32_t fdb_nibblepack_pack8(const int64_t in[8], uint8_t* out, int32_t outcap);
int32_t fdb_nibblepack_unpack8(const uint8_t* in, int32_t inlen, int64_t out[8],
                               int32_t* consumed);
/* NibblePack.packDelta / packDoubles (NibblePack.scala:37-98). Return bytes written. */
int32_t fdb_nibblepack_pack_delta(const int64_t* in, int32_t n, uint8_t* out, int32_t outcap);
int32_t fdb_nibblepack_pack_doubles(const double* in, int32_t n, uint8_t* out, int32_t outcap);

/* ---- query definition ---------------------------------------------------- */
/* Range functions: the dispatch table of RangeFunction.generatorFor
 * (query/.../rangefn/RangeFunction.scala:294-410). */
#define FDB_FN_RATE             0   /* ChunkedRateFunction       */
#define FDB_FN_INCREASE         1   /* ChunkedIncreaseFunction   */
#define FDB_FN_DELTA            2   /* ChunkedDeltaFunction      */
#define FDB_FN_SUM_OVER_TIME    3
#define FDB_FN_COUNT_OVER_TIME  4
#define FDB_FN_AVG_OVER_TIME    5
#define FDB_FN_MIN_OVER_TIME    6
#define FDB_FN_MAX_OVER_TIME    7
#define FDB_FN_STDDEV_OVER_TIME 8
#define FDB_FN_STDVAR_OVER_TIME 9
#define FDB_FN_CHANGES         10
#define FDB_FN_HIST_RATE       11   /* HistRateFunction (RateFunctions.scala:330-400):
                                       per-bucket counter-corrected extrapolated rate */
#define FDB_FN_LAST            12   /* LastSampleChunkedFunctionD (RangeFunction.scala:595-694):
                                       raw value of the last sample <= wEnd within the window
                                       (NaN stale markers propagate); the reference uses
                                       window = stale-sample-after + 1 = 300001ms for raw
                                       queries (PeriodicSamplesMapper.scala:79-81) */
#define FDB_FN_PRESENT         13   /* PresentOverTimeChunkedFunctionD
                                       (RangeFunction.scala:725-745): 1 for a non-NaN last
                                       sample; NaN stale markers step back one row */
#define FDB_FN_TIMESTAMP       14   /* TimestampChunkedFunction (RangeFunction.scala:705-723):
                                       last sample's timestamp <= wEnd, in seconds */
#define FDB_FN_QUANTILE_OT     16   /* QuantileOverTimeChunkedFunctionD
                                       (AggrOverTimeFunctions.scala:1272-1299);
                                       q in fdb_query_t.param */
#define FDB_FN_MAD_OT          17   /* MedianAbsoluteDeviationOverTime...
                                       (AggrOverTimeFunctions.scala:1302-1330) */
#define FDB_FN_PREDICT_LINEAR  18   /* PredictLinearChunkedFunctionD
                                       (AggrOverTimeFunctions.scala:1507-1554);
                                       t offset seconds in param */
#define FDB_FN_RATE_OVER_DELTA 19   /* RateOverDeltaChunkedFunctionD
                                       (RateFunctions.scala:424-445) */
#define FDB_FN_HOLT_WINTERS    20   /* HoltWintersChunkedFunctionD
                                       (AggrOverTimeFunctions.scala:1379-1453):
                                       sf in param, tf in param2. SINGLE-CHUNK
                                       series only (fast-eligible datasets):
                                       across chunk boundaries the reference
                                       feeds one decoded-past-endRow value —
                                       undefined memory — into the recurrence,
                                       so exact multi-chunk parity does not
                                       exist; the engine rejects those loudly */
#define FDB_FN_ZSCORE          15   /* ZScoreChunkedFunctionD
                                       (AggrOverTimeFunctions.scala:1592-1603):
                                       (lastSample - mean) / stddev over the window */

/* Cross-series aggregation: RowAggregator implementations
 * (query/.../exec/aggregator/RowAggregator.scala:28-150). */
#define FDB_AGG_NONE   0   /* emit the full [series × windows] grid  */
#define FDB_AGG_SUM    1   /* SumRowAggregator.scala:12-34           */
#define FDB_AGG_COUNT  2
#define FDB_AGG_MIN    3
#define FDB_AGG_MAX    4
#define FDB_AGG_AVG    5   /* AvgRowAggregator.scala:8-41 (sum,count partials) */
#define FDB_AGG_TOPK   6   /* TopBottomKRowAggregator.scala:29-100: k largest
                              non-NaN series values per (group, window).
                              q.param = k (<=16). out: [G × W × k] values sorted
                              descending (NaN-padded); out_counts reinterpreted
                              as [G × W × k] doubles holding the series ids
                              (-1 padding). */
#define FDB_AGG_BOTTOMK 7  /* same, k smallest; values sorted ascending */
#define FDB_AGG_STDDEV  8  /* StddevRowAggregator.scala:39-58 (merge is
                              algebraically sum/sumsq/count; partials stack
                              raw sums+sumsq in a 2x out grid, see below) */
#define FDB_AGG_STDVAR  9  /* StdvarRowAggregator (same, without the sqrt) */
#define FDB_AGG_QUANTILE 11 /* QuantileRowAggregator.scala:21-76: per-cell
                               t-digest (compression 100, the published
                               merging algorithm restated in
                               filodb_amd/csrc/tdigest_impl.h — the
                               com.tdunning dep is absent from the reference
                               tree); present = quantile(q.param).
                               No partial (multi-shard) mode. */
#define FDB_AGG_COUNT_VALUES 12 /* CountValuesRowAggregator.scala:26-100 —
                               use fdb_query_exec_count_values */
#define FDB_AGG_GROUP  10  /* GroupRowAggregator: 1 where any non-NaN row
                              contributed, NaN otherwise */

typedef struct {
  int64_t start;     /* first window end timestamp (ms)                      */
  int64_t step;      /* ms; numWindows = (end-start)/step + 1                */
  int64_t end;       /* last window end timestamp                            */
  int64_t window;    /* lookback length (ms); wStart = wEnd - window
                        (inclusive-range=true, filodb-defaults.conf:590)     */
  int32_t func_id;   /* FDB_FN_*  (PeriodicSamplesMapper.functionId)         */
  int32_t agg_id;    /* FDB_AGG_* (AggregateMapReduce)                       */
  int32_t num_groups;/* required when agg_id != NONE                         */
  int32_t _pad;
  double  param;     /* function parameter (quantile q for the histogram
                        present step); 0 when unused                          */
  double  param2;    /* second parameter (holt_winters trend factor); 0 when
                        unused                                               */
} fdb_query_t;

static inline int32_t fdb_num_windows(const fdb_query_t* q) {
  return (int32_t)((q->end - q->start) / q->step) + 1;
}

/* ---- GPU engine ---------------------------------------------------------- */
typedef struct fdb_engine  fdb_engine_t;
typedef struct fdb_dataset fdb_dataset_t;

/* Creates an engine bound to HIP device `device`. Fails (FDB_ERR_NOGPU) when no
 * HIP device is present — there is deliberately no CPU fallback. */
fdb_engine_t* fdb_engine_create(int32_t device);
void          fdb_engine_destroy(fdb_engine_t* e);
int32_t       fdb_engine_synchronize(fdb_engine_t* e);

/* Uploads a sealed store into one contiguous HBM blob + SoA chunk directory
 * (DESIGN.md §2). Returns NULL on error. */
fdb_dataset_t* fdb_dataset_upload(fdb_engine_t* e, const fdb_store_t* s);
void           fdb_dataset_destroy(fdb_dataset_t* d);
int64_t        fdb_dataset_bytes(const fdb_dataset_t* d);   /* chunk payload bytes in HBM */
int64_t        fdb_dataset_samples(const fdb_dataset_t* d); /* total rows across chunks   */

/* Executes one (shard, query): the batched equivalent of folding
 * PeriodicSamplesMapper (+ AggregateMapReduce when agg_id != NONE) over every
 * RawDataRangeVector of the shard (ExecPlan.scala:404-419).
 *
 * agg_id == FDB_AGG_NONE: out must hold series×windows doubles; out_counts ignored.
 * agg_id != FDB_AGG_NONE: out must hold num_groups×windows doubles; for FDB_AGG_AVG
 *   (and cross-GPU merges) out_counts (num_groups×windows doubles, may be NULL
 *   otherwise) receives the count partials — the reduction schema of
 *   AvgRowAggregator.scala:8-41.
 * Passing out_counts selects PARTIAL mode: out holds raw sums (0 where empty;
 *   MIN/MAX keep NaN) for the caller to merge across shards and present.
 *   For FDB_AGG_STDDEV/STDVAR partials, out must hold 2×num_groups×windows
 *   doubles — raw sums then raw sum-of-squares — merged by addition
 *   (algebraically StddevRowAggregator.scala:36-52's reduction schema).
 * out/out_counts may be HOST pointers (out_on_device=0) or DEVICE pointers
 * (out_on_device=1, e.g. torch tensor data_ptr for the RCCL all-reduce).
 * Synchronous: returns after the result is materialized. */
int32_t fdb_query_exec(fdb_engine_t* e, const fdb_dataset_t* d, const fdb_query_t* q,
                       double* out, double* out_counts, int32_t out_on_device);

/* count_values cross-series aggregation (CountValuesRowAggregator.scala:
 * 26-100): per (group, window) cell the distinct non-NaN values with their
 * frequencies, sorted ascending by value. out_vals/out_cnts are
 * [num_groups × windows × k_cap] host buffers, out_n [num_groups × windows].
 * A cell exceeding k_cap distinct values errors (the reference throws at its
 * 1000-value limit; k_cap <= 1000). q->agg_id is ignored. */
int32_t fdb_query_exec_count_values(fdb_engine_t* e, const fdb_dataset_t* d,
                                    const fdb_query_t* q, int32_t k_cap,
                                    double* out_vals, double* out_cnts,
                                    int32_t* out_n);

/* Histogram rows with otel max/min companion double columns
 * (SumAndMaxOverTimeFuncHD / CumulativeHistRateAndMinMaxFunction inputs,
 *  AggrOverTimeFunctions.scala:612-813). */
int32_t fdb_series_append_hist_mm(fdb_store_t* s, int32_t sid,
                                  const int64_t* ts,
                                  const uint64_t* bucket_values,
                                  const double* maxs, const double* mins,
                                  int32_t n, int32_t num_buckets,
                                  double bucket_first, double bucket_mult);

/* Histogram query with companion max/min outputs [num_groups × windows]
 * (merged across series with maxIgnoreNaN/minIgnoreNaN per
 *  HistMaxMinSumAggregator). func_id FDB_FN_HIST_RATE = counter-corrected
 * rate + min/max; FDB_FN_SUM_OVER_TIME = SumOverTime of the histograms
 * (SumAndMaxOverTimeFuncHD shape). out_max/out_min may be NULL. */
/* ------------------------------------------------------------------ */
/* BinaryRecord v2 ingestion containers (layouts restated
 * from core/.../binaryrecord2/RecordContainer.scala:15-27,
 * RecordSchema.scala:60-75, RecordBuilder.scala:109-125,372-404,461-478).
 * The builder produces on-wire containers of the gauge/counter ingestion
 * schema {timestamp, value, metric, tags}; fdb_store_ingest_brv2 consumes
 * them into a chunk store, keying series by the binary partition-key
 * region exactly as the reference compares part keys. */
typedef struct fdb_brv2_builder fdb_brv2_builder_t;
typedef struct fdb_brv2_index fdb_brv2_index_t;

fdb_brv2_builder_t* fdb_brv2_builder_create(int64_t ts_header);
void fdb_brv2_builder_destroy(fdb_brv2_builder_t* b);
/* tag_kv = [k0, v0, k1, v1, ...] (ntags pairs) */
int32_t fdb_brv2_add_record(fdb_brv2_builder_t* b, int64_t ts, double val,
                            const char* metric, const char* const* tag_kv,
                            int32_t ntags, int32_t schema_id);
int32_t fdb_brv2_finish(fdb_brv2_builder_t* b, uint8_t* out, int32_t cap);
int32_t fdb_brv2_read(const uint8_t* bytes, int32_t len, int32_t idx,
                      int64_t* ts, double* val, uint8_t* pk_out,
                      int32_t pk_cap, int32_t* pk_len, int32_t* schema_id,
                      int32_t* part_hash);
fdb_brv2_index_t* fdb_brv2_index_create(void);
void fdb_brv2_index_destroy(fdb_brv2_index_t* ix);
int32_t fdb_store_ingest_brv2(fdb_store_t* s, fdb_brv2_index_t* ix,
                              const uint8_t* bytes, int32_t len,
                              int32_t col_kind, int32_t* out_new_series);

/* Cassandra chunk-table persistence (paging side).
 * Row shape restated from cassandra/.../columnstore/TimeSeriesChunksTable.scala:35-103:
 * (partition blob, chunkid bigint, info = first 28 B of the ChunkSetInfo record
 * {chunkID i64, numRows i32, ingestionTime i64, endTime i64}
 * (core/.../store/ChunkSetInfo.scala:133-154, toBytes :250-254), chunks =
 * frozen per-column vector blobs in schema order). chunkid packing from
 * core/.../store/package.scala:112-123 (startTimeShift = 22, :16). Rows are
 * framed into a flat byte stream (no Cassandra here):
 * u32 pk_len + pk | i64 chunkid | u32 info_len + info | u16 nchunks |
 * (u32 len + bytes)*. fdb_store_restore feeds the frozen bytes back
 * UNCHANGED via fdb_store_add_encoded_chunk (a bit-faithful round trip). */
int64_t fdb_chunkid(int64_t start_time, int64_t ingestion_time);
int64_t fdb_chunkid_start_time(int64_t chunkid);
int32_t fdb_store_persist(const fdb_store_t* s, int32_t series_id,
                          const uint8_t* partkey, int32_t pk_len,
                          int64_t ingestion_time,
                          uint8_t* out, int32_t cap, int32_t* out_len);
int32_t fdb_store_restore(fdb_store_t* s, fdb_brv2_index_t* ix,
                          const uint8_t* bytes, int32_t len,
                          int32_t col_kind, int32_t* out_rows);
int32_t fdb_store_add_encoded_chunk(fdb_store_t* s, int32_t series_id,
                                    const uint8_t* ts_bytes, int32_t ts_len,
                                    const uint8_t* val_bytes, int32_t val_len,
                                    int32_t num_rows,
                                    int64_t start_time, int64_t end_time);

/* GPU ingest-side chunk encoder: encodes num_chunks scalar
 * chunks (ts rows + double rows, delimited by row_offs[c]..row_offs[c+1])
 * into frozen vectors BYTE-IDENTICAL to the host encoder / the reference's
 * formats (DeltaDeltaVector.scala:63-135, IntBinaryVector.scala:52-177,
 * DoubleVector.scala:86-96,457-476). One wavefront per chunk. Outputs land in
 * `out` at the returned per-chunk offsets/lengths; histogram columns stay
 * host-side. */
/* Downsampled scalar series: pre-aggregated sums with per-row sample counts
 * (the downsample schema's avg path). fdb_series_append_sc stores the count
 * column alongside the sum column; fdb_query_exec_avg_sc computes
 * avg(window) = SumOverTime(sum col) / SumOverTime(count col) per
 * AvgWithSumAndCountOverTimeFuncD (AggrOverTimeFunctions.scala:820-860),
 * emitting the [series x windows] grid. */
int32_t fdb_series_append_sc(fdb_store_t* s, int32_t series_id,
                             const int64_t* ts, const double* sums,
                             const double* counts, int32_t n);
int32_t fdb_query_exec_avg_sc(fdb_engine_t* e, const fdb_dataset_t* d,
                              const fdb_query_t* q, double* out,
                              int32_t out_on_device);

int32_t fdb_gpu_encode_chunks(fdb_engine_t* e,
                              const int64_t* ts, const double* vals,
                              const int64_t* row_offs,
                              int32_t num_chunks, int32_t col_kind,
                              uint8_t* out, int64_t out_cap,
                              int64_t* out_ts_off, int64_t* out_val_off,
                              int32_t* out_ts_len, int32_t* out_val_len);

int32_t fdb_query_exec_hist_mm(fdb_engine_t* e, const fdb_dataset_t* d,
                               const fdb_query_t* q, int32_t nb,
                               double* out_bucket_sums, double* out_counts,
                               double* out_max, double* out_min,
                               double* out_quantile, int32_t out_on_device);

/* Histogram pipeline for BASELINE config #4:
 * histogram_quantile(param, sum(rate(hist[window])) by group)
 * — HistRateFunction per series (per-bucket extrapolated rate with counter
 * correction), HistSumRowAggregator across series per group
 * (aggregator/HistSumRowAggregator.scala:20-29), Histogram.quantile present
 * step (core/.../vectors/Histogram.scala:63-108).
 * out_bucket_sums: [num_groups × windows × num_buckets] per-bucket rate sums
 *                  (the cross-shard/RCCL merge payload); may be NULL.
 * out_counts:      [num_groups × windows] contributing-series counts; may be NULL.
 * out_quantile:    [num_groups × windows] quantile(param) of the summed rate
 *                  histogram (NaN where no contributions); may be NULL.
 * Round-1 engine limit: one chunk per series per query span. */
int32_t fdb_query_exec_hist(fdb_engine_t* e, const fdb_dataset_t* d,
                            const fdb_query_t* q, int32_t num_buckets,
                            double* out_bucket_sums, double* out_counts,
                            double* out_quantile, int32_t out_on_device);

/* Timing variant for bench.py: runs the same launch `iters` times and returns the
 * average per-iteration kernel milliseconds measured with HIP events on the
 * engine's stream. Results land in out like fdb_query_exec. */
int32_t fdb_query_bench(fdb_engine_t* e, const fdb_dataset_t* d, const fdb_query_t* q,
                        double* out, double* out_counts, int32_t out_on_device,
                        int32_t warmup, int32_t iters, double* avg_kernel_ms);

#ifdef __cplusplus
} /* extern "C" */
#endif
#endif /* FILODB_AMD_H */
