/*
 * agg_oracle.c — CPU oracle for the m3aggregator rollup loop.
 * TEST INFRASTRUCTURE ONLY — see m3tsz_oracle.h header comment.
 *
 * Restates, with reference file:line cites:
 *   src/aggregator/aggregation/counter.go:31-131 (Counter)
 *   src/aggregator/aggregation/gauge.go:31-165 (Gauge incl. NaN rules :87-98)
 *   src/aggregator/aggregation/timer.go:31-153 (Timer)
 *   src/aggregator/aggregation/common.go:29-36 (stdev)
 *   src/aggregator/aggregation/quantile/cm/{stream.go,heap.go,list.go}
 *     (CKMS biased quantiles; defaults options.go:30-32: eps=1e-3,
 *      capacity=32, insertAndCompressEvery=1024)
 *   window semantics: aggregator/aggregator/generic_elem.go:219-235 (AddValue
 *     truncates timestamp to the resolution window), list.go:541-543 (flush
 *     timestamp = window end).
 */
#include "m3tsz_oracle.h"
#include <math.h>
#include <stdlib.h>
#include <string.h>

/* metrics/aggregation/type.go:31-56 */
enum {
    AGG_UNKNOWN = 0, AGG_LAST, AGG_MIN, AGG_MAX, AGG_MEAN, AGG_MEDIAN,
    AGG_COUNT, AGG_SUM, AGG_SUMSQ, AGG_STDEV,
    AGG_P10, AGG_P20, AGG_P30, AGG_P40, AGG_P50, AGG_P60, AGG_P70, AGG_P80,
    AGG_P90, AGG_P95, AGG_P99, AGG_P999, AGG_P9999, AGG_P25, AGG_P75,
};

/* common.go:29-36 */
static double m3_stdev(int64_t count, double sum_sq, double sum) {
    int64_t div = count * (count - 1);
    if (div == 0) return 0.0;
    double num = (double)count * sum_sq - sum * sum;
    return sqrt(num / (double)div);
}

/* ====================== CKMS stream (quantile/cm) ====================== */

#define MIN_SAMPLES_TO_COMPRESS 3 /* stream.go:27 */

typedef struct sample {
    double value;
    int64_t num_ranks; /* sample.go */
    int64_t delta;
    struct sample *prev, *next;
} cm_sample;

typedef struct cm_chunk {
    struct cm_chunk* next;
    cm_sample items[1024];
} cm_chunk;

typedef struct {
    cm_chunk* chunks;       /* stable arena: chunks never move */
    int chunk_used;         /* used slots in head chunk */
    cm_sample** free_list;  /* released samples */
    int64_t free_len, free_cap;
    cm_sample *head, *tail;
    int64_t list_len;

    double* buf_less; int64_t less_len, less_cap;  /* minHeap */
    double* buf_more; int64_t more_len, more_cap;

    const double* quantiles; int nq;
    double* computed;
    int64_t* thr_rank; int64_t* thr_thresh;

    cm_sample *insert_cursor, *compress_cursor;
    int64_t num_values;
    int64_t compress_min_rank;
    int insert_and_compress_counter;
    int insert_and_compress_every;
    double eps;
    int flushed;
} cm_stream;

/* --- minHeap (heap.go:27-56): push keeps heap; SortDesc heapsorts asc=desc? ---
 * heap.go SortDesc: repeated pop-min to end => array ends sorted DESCENDING.
 * Behavior-equivalent restatement: we only need Push + SortDesc + swap. */
static void heap_push(double** h, int64_t* len, int64_t* cap, double value) {
    if (*len == *cap) {
        *cap = *cap ? *cap * 2 : 64;
        *h = (double*)realloc(*h, (size_t)(*cap) * sizeof(double));
    }
    double* heap = *h;
    int64_t i = (*len)++;
    heap[i] = value;
    while (i > 0) {
        int64_t parent = (i - 1) / 2;
        if (heap[parent] <= heap[i]) break;
        double t = heap[parent]; heap[parent] = heap[i]; heap[i] = t;
        i = parent;
    }
}
static void heap_sort_desc(double* heap, int64_t len) {
    for (int64_t n = len - 1; n > 0; n--) {
        double t = heap[0]; heap[0] = heap[n]; heap[n] = t;
        int64_t i = 0;
        for (;;) {
            int64_t smallest = i, left = 2 * i + 1, right = left + 1;
            if (left < n && heap[left] < heap[smallest]) smallest = left;
            if (right < n && heap[right] < heap[smallest]) smallest = right;
            if (smallest == i) break;
            t = heap[i]; heap[i] = heap[smallest]; heap[smallest] = t;
            i = smallest;
        }
    }
}

/* --- sampleList (list.go) over a stable chunked arena --- */
static cm_sample* list_acquire(cm_stream* s) {
    if (s->free_len > 0) return s->free_list[--s->free_len];
    if (!s->chunks || s->chunk_used == 1024) {
        cm_chunk* c = (cm_chunk*)malloc(sizeof(cm_chunk));
        c->next = s->chunks;
        s->chunks = c;
        s->chunk_used = 0;
    }
    cm_sample* smp = &s->chunks->items[s->chunk_used++];
    smp->prev = smp->next = NULL;
    return smp;
}
static void list_push_back(cm_stream* s, cm_sample* sample) {
    sample->prev = s->tail;
    sample->next = NULL;
    if (!s->head) s->head = sample;
    else s->tail->next = sample;
    s->tail = sample;
    s->list_len++;
}
static void list_insert_before(cm_stream* s, cm_sample* sample, cm_sample* mark) {
    cm_sample* prev = NULL;
    if (!mark->prev) s->head = sample;
    else { prev = mark->prev; prev->next = sample; }
    mark->prev = sample;
    sample->next = mark;
    sample->prev = prev;
    s->list_len++;
}
static void list_remove(cm_stream* s, cm_sample* sample) {
    cm_sample* prev = sample->prev;
    cm_sample* next = sample->next;
    sample->prev = sample->next = NULL;
    if (!prev) s->head = next; else prev->next = next;
    if (!next) s->tail = prev; else next->prev = prev;
    /* release back to free list */
    if (s->free_len == s->free_cap) {
        s->free_cap = s->free_cap ? s->free_cap * 2 : 64;
        s->free_list = (cm_sample**)realloc(s->free_list, (size_t)s->free_cap * sizeof(cm_sample*));
    }
    s->free_list[s->free_len++] = sample;
    s->list_len--;
}

static void stream_init(cm_stream* s, const double* quantiles, int nq,
                        double eps, int insert_and_compress_every) {
    memset(s, 0, sizeof(*s));
    s->quantiles = quantiles;
    s->nq = nq;
    s->eps = eps;
    s->insert_and_compress_every = insert_and_compress_every;
    s->computed = (double*)calloc((size_t)(nq > 0 ? nq : 1), sizeof(double));
    s->thr_rank = (int64_t*)calloc((size_t)(nq > 0 ? nq : 1), sizeof(int64_t));
    s->thr_thresh = (int64_t*)calloc((size_t)(nq > 0 ? nq : 1), sizeof(int64_t));
}
static void stream_free(cm_stream* s) {
    cm_chunk* c = s->chunks;
    while (c) { cm_chunk* n = c->next; free(c); c = n; }
    free(s->free_list); free(s->buf_less); free(s->buf_more);
    free(s->computed); free(s->thr_rank); free(s->thr_thresh);
    memset(s, 0, sizeof(*s));
}

static void stream_reset_insert_cursor(cm_stream* s) { /* stream.go:425-429 */
    double* tb = s->buf_less; int64_t tl = s->less_len, tc = s->less_cap;
    s->buf_less = s->buf_more; s->less_len = s->more_len; s->less_cap = s->more_cap;
    s->buf_more = tb; s->more_len = tl; s->more_cap = tc;
    s->insert_cursor = s->head;
}

/* stream.go:280-331 insert */
static void stream_insert(cm_stream* s) {
    cm_sample* comp_cur = s->compress_cursor;
    double comp_value = nan("");
    if (comp_cur) comp_value = comp_cur->value;

    heap_sort_desc(s->buf_more, s->more_len);
    double* vals = s->buf_more;
    int64_t idx = s->more_len - 1;

    while (s->insert_cursor && idx < s->more_len) {
        cm_sample* curr = s->insert_cursor;
        double insert_point_value = curr->value;
        while (idx >= 0 && vals[idx] <= insert_point_value) {
            double val = vals[idx];
            idx--;
            cm_sample* sample = list_acquire(s);
            sample->value = val;
            sample->num_ranks = 1;
            sample->delta = curr->num_ranks + curr->delta - 1;
            list_insert_before(s, sample, curr);
            if (comp_value >= val) s->compress_min_rank++;
            s->num_values++;
        }
        s->insert_cursor = s->insert_cursor->next;
    }

    if (!s->insert_cursor && idx < s->more_len) {
        while (idx >= 0 && vals[idx] >= s->tail->value) {
            double val = vals[idx];
            idx--;
            cm_sample* sample = list_acquire(s);
            sample->value = val;
            sample->num_ranks = 1;
            sample->delta = 0;
            list_push_back(s, sample);
            s->num_values++;
        }
    }

    s->more_len = 0;
    stream_reset_insert_cursor(s);
}

/* stream.go:333-401 compress */
static void stream_compress(cm_stream* s) {
    if (s->list_len < MIN_SAMPLES_TO_COMPRESS) return;

    if (!s->compress_cursor) {
        s->compress_cursor = s->tail->prev;
        s->compress_min_rank = s->num_values - 1 - s->compress_cursor->num_ranks;
        s->compress_cursor = s->compress_cursor->prev;
    }

    int64_t num_vals = s->num_values;
    double eps = 2.0 * s->eps;

    while (s->compress_cursor && s->compress_cursor != s->head) {
        cm_sample* curr = s->compress_cursor;
        cm_sample* next = curr->next;
        cm_sample* prev = curr->prev;
        int64_t max_rank = s->compress_min_rank + curr->num_ranks + curr->delta;

        int64_t threshold = INT64_MAX;
        for (int i = 0; i < s->nq; i++) {
            int64_t quantile_min;
            /* Go: maxRank >= int64(quantiles[i]*float64(numVals)) */
            if (max_rank >= (int64_t)(s->quantiles[i] * (double)num_vals)) {
                quantile_min = (int64_t)(eps * (double)max_rank / s->quantiles[i]);
            } else {
                quantile_min = (int64_t)(eps * (double)(num_vals - max_rank) / (1.0 - s->quantiles[i]));
            }
            if (quantile_min < threshold) threshold = quantile_min;
        }

        s->compress_min_rank -= curr->num_ranks;
        int64_t test_val = curr->num_ranks + next->num_ranks + next->delta;

        if (test_val <= threshold) {
            if (s->insert_cursor == curr) s->insert_cursor = next;
            next->num_ranks += curr->num_ranks;
            list_remove(s, curr);
        }
        s->compress_cursor = prev;
    }

    if (s->compress_cursor == s->head) s->compress_cursor = NULL;
}

/* stream.go:362-377 threshold(rank) */
static int64_t stream_threshold(cm_stream* s, int64_t rank) {
    int64_t min_val = INT64_MAX;
    int64_t num_vals = s->num_values;
    double eps = 2.0 * s->eps;
    for (int i = 0; i < s->nq; i++) {
        int64_t quantile_min;
        if (rank >= (int64_t)(s->quantiles[i] * (double)num_vals)) {
            quantile_min = (int64_t)(eps * (double)rank / s->quantiles[i]);
        } else {
            quantile_min = (int64_t)(eps * (double)(num_vals - rank) / (1.0 - s->quantiles[i]));
        }
        if (quantile_min < min_val) min_val = quantile_min;
    }
    return min_val;
}

/* stream.go:210-229 quantilesFromBuf */
static void stream_quantiles_from_buf(cm_stream* s) {
    double buf[MIN_SAMPLES_TO_COMPRESS + 1];
    int n = 0;
    for (cm_sample* curr = s->head; curr; curr = curr->next) buf[n++] = curr->value;
    for (int i = 0; i < s->nq; i++) {
        int idx = (int)(s->quantiles[i] * (double)n);
        if (idx >= n) idx = n - 1;
        s->computed[i] = buf[idx];
    }
}

/* stream.go:231-277 calcQuantiles */
static void stream_calc_quantiles(cm_stream* s) {
    if (s->nq == 0 || s->num_values == 0) return;
    if (s->num_values <= MIN_SAMPLES_TO_COMPRESS) { stream_quantiles_from_buf(s); return; }

    int64_t min_rank = 0, max_rank = 0;
    int idx = 0;
    cm_sample* curr = s->head;
    cm_sample* prev = s->head;

    for (int i = 0; i < s->nq; i++) {
        int64_t rank = (int64_t)ceil(s->quantiles[i] * (double)s->num_values);
        s->thr_rank[i] = rank;
        s->thr_thresh[i] = (int64_t)ceil((double)stream_threshold(s, rank) / 2.0);
    }

    while (curr && idx < s->nq) {
        max_rank = min_rank + curr->num_ranks + curr->delta;
        int64_t rank = s->thr_rank[idx], threshold = s->thr_thresh[idx];
        if (max_rank > rank + threshold || min_rank > rank) {
            s->computed[idx] = prev->value;
            idx++;
        }
        min_rank += curr->num_ranks;
        prev = curr;
        curr = curr->next;
    }

    for (int i = idx; i < s->nq; i++) {
        int64_t rank = s->thr_rank[i], threshold = s->thr_thresh[i];
        if (max_rank >= rank + threshold || min_rank > rank) {
            s->computed[i] = prev->value;
        }
    }
}

/* stream.go:77-116 AddBatch */
static void stream_add_batch(cm_stream* s, const double* values, int64_t n) {
    s->flushed = 0;
    if (n == 0) return;
    int64_t start = 0;
    if (s->list_len == 0) {
        cm_sample* sample = list_acquire(s);
        sample->value = values[0];
        sample->num_ranks = 1;
        sample->delta = 0;
        list_push_back(s, sample);
        s->insert_cursor = s->head;
        s->num_values++;
        start = 1;
    }
    double insert_point_value = s->insert_cursor->value;
    int insert_counter = s->insert_and_compress_counter;
    for (int64_t i = start; i < n; i++) {
        double value = values[i];
        if (value < insert_point_value) heap_push(&s->buf_less, &s->less_len, &s->less_cap, value);
        else heap_push(&s->buf_more, &s->more_len, &s->more_cap, value);
        if (insert_counter == s->insert_and_compress_every) {
            stream_insert(s);
            stream_compress(s);
            insert_counter = 0;
        }
        insert_counter++;
    }
    s->insert_and_compress_counter = insert_counter;
}

/* stream.go:123-137 Flush */
static void stream_flush(cm_stream* s) {
    if (s->flushed) return;
    while (s->less_len > 0 || s->more_len > 0) {
        if (s->more_len == 0) stream_reset_insert_cursor(s);
        stream_insert(s);
        stream_compress(s);
    }
    stream_calc_quantiles(s);
    s->flushed = 1;
}

/* stream.go:150-171 Quantile (call after Flush) */
static double stream_quantile(cm_stream* s, double q) {
    if (q < 0.0 || q > 1.0) return nan("");
    if (s->list_len == 0) return 0.0;
    if (q == 0.0) return s->head->value;
    if (q == 1.0) return s->tail->value;
    for (int i = 0; i < s->nq; i++) {
        if (s->quantiles[i] >= q) return s->computed[i];
    }
    return nan("");
}

/* ==================== exported CKMS surface (tests) ==================== */

/* Feed n values through a CKMS stream and return the post-flush sample
 * list length (test sizing aid for the GPU engine's sample-list cap). */
int64_t oracle_ckms_list_len(const double* values, int64_t n,
                             const double* quantiles, int nq,
                             double eps, int insert_and_compress_every) {
    cm_stream s;
    stream_init(&s, quantiles, nq, eps, insert_and_compress_every);
    stream_add_batch(&s, values, n);
    stream_flush(&s);
    int64_t len = s.list_len;
    stream_free(&s);
    return len;
}


/* Feed n values through a CKMS stream (AddBatch), Flush, and evaluate the
 * q[] quantiles (which must equal the stream's registered quantiles).
 * Returns 0. */
int oracle_ckms_quantiles(const double* values, int64_t n,
                          const double* quantiles, int nq,
                          double eps, int insert_and_compress_every,
                          double* out, double* out_min, double* out_max) {
    cm_stream s;
    stream_init(&s, quantiles, nq, eps, insert_and_compress_every);
    stream_add_batch(&s, values, n);
    stream_flush(&s);
    for (int i = 0; i < nq; i++) out[i] = stream_quantile(&s, quantiles[i]);
    if (out_min) *out_min = stream_quantile(&s, 0.0);
    if (out_max) *out_max = stream_quantile(&s, 1.0);
    stream_free(&s);
    return 0;
}

/* ========================= windowed rollup ========================= */

/* Metric types for the rollup entry point. */
enum { METRIC_COUNTER = 0, METRIC_GAUGE = 1, METRIC_TIMER = 2 };

/* Output slot layout per (series, bucket): one double per requested agg
 * type, in the order given by agg_types[]. Quantile agg types use the m3
 * default CKMS options (eps=1e-3, every=1024) via a Timer stream
 * (timer.go:44-48, options.go:30-32).
 *
 * Window semantics (generic_elem.go:219-221, list.go:541-543):
 *   bucket index = ts // window_ns  (timestamps are >0, Go Truncate)
 *   out_window_ts[b] = window_start + window_ns  (window END)
 *
 * Values are assigned to buckets relative to base_ns:
 *   bucket = (ts - base_ns) / window_ns, 0 <= bucket < nbuckets,
 * where base_ns must be window-aligned. Values outside are an error (-1).
 *
 * Counter input values are int64 via Go conversion int64(value) of the f64
 * input (entry.go passes counter values as int64; our batch carries f64
 * storage of exact ints).
 */
static double agg_quantile_of(int t);

int oracle_rollup_series_opts(
    const int64_t* ts_ns, const double* vals, int64_t n,
    int metric_type, int64_t base_ns, int64_t window_ns, int64_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* out /* nbuckets x naggs */, int64_t* out_window_ts /* nbuckets, or NULL */,
    double eps, int every) {
    /* collect per-bucket state */
    typedef struct {
        int64_t count;
        int64_t isum, imin, imax, isumsq;     /* counter */
        double fsum, fsumsq, fmin, fmax, last; /* gauge/timer */
        int64_t last_at;
        int any;
    } bstate;
    bstate* st = (bstate*)calloc((size_t)nbuckets, sizeof(bstate));
    for (int64_t b = 0; b < nbuckets; b++) {
        st[b].imin = INT64_MAX; st[b].imax = INT64_MIN; /* counter.go:44-47 */
        st[b].fmin = nan(""); st[b].fmax = nan("");      /* gauge.go:56-59 */
        st[b].last = 0; st[b].last_at = 0;
    }

    /* quantile streams only where needed */
    int need_stream = 0;
    double qlist[32]; int nql = 0;
    for (int i = 0; i < naggs; i++) {
        double q = agg_quantile_of(agg_types[i]);
        if (q >= 0) { need_stream = 1; if (nql < 32) qlist[nql++] = q; }
    }
    /* m3 registers the stream with the sorted unique quantile list */
    if (nql > 1) {
        for (int i = 1; i < nql; i++) {
            double k = qlist[i]; int j = i - 1;
            while (j >= 0 && qlist[j] > k) { qlist[j + 1] = qlist[j]; j--; }
            qlist[j + 1] = k;
        }
        int m = 0;
        for (int i = 0; i < nql; i++) if (m == 0 || qlist[m - 1] != qlist[i]) qlist[m++] = qlist[i];
        nql = m;
    }

    cm_stream* streams = NULL;
    if (need_stream && metric_type == METRIC_TIMER) {
        streams = (cm_stream*)malloc((size_t)nbuckets * sizeof(cm_stream));
        for (int64_t b = 0; b < nbuckets; b++)
            stream_init(&streams[b], qlist, nql, eps, every);
    }

    int err = 0;
    for (int64_t i = 0; i < n && !err; i++) {
        int64_t b = (ts_ns[i] - base_ns) / window_ns;
        if (ts_ns[i] < base_ns || b >= nbuckets) { err = -1; break; }
        bstate* bs = &st[b];
        double v = vals[i];
        int64_t t = ts_ns[i];
        switch (metric_type) {
        case METRIC_COUNTER: { /* counter.go:52-78 */
            int64_t iv = (int64_t)v; /* in-range by contract */
            bs->isum += iv;
            bs->count++;
            if (bs->imax < iv) bs->imax = iv;
            if (bs->imin > iv) bs->imin = iv;
            bs->isumsq += iv * iv;
            bs->any = 1;
            break;
        }
        case METRIC_GAUGE: { /* gauge.go:73-103 updateTotals */
            if (bs->last_at == 0 || t > bs->last_at) { bs->last_at = t; bs->last = v; }
            bs->count++;
            if (v != v) { bs->any = 1; break; } /* NaN: count only */
            bs->fsum += v;
            if (bs->fmax != bs->fmax || bs->fmax < v) bs->fmax = v;
            if (bs->fmin != bs->fmin || bs->fmin > v) bs->fmin = v;
            bs->fsumsq += v * v;
            bs->any = 1;
            break;
        }
        case METRIC_TIMER: { /* timer.go:56-75 AddBatch */
            bs->count++;
            bs->fsum += v;
            bs->fsumsq += v * v;
            if (streams) stream_add_batch(&streams[b], &v, 1);
            bs->any = 1;
            break;
        }
        }
    }

    for (int64_t b = 0; b < nbuckets && !err; b++) {
        if (out_window_ts) out_window_ts[b] = base_ns + (b + 1) * window_ns; /* list.go:541-543 */
        bstate* bs = &st[b];
        if (streams && bs->any) stream_flush(&streams[b]);
        for (int i = 0; i < naggs; i++) {
            int t = agg_types[i];
            double r = 0;
            double q = agg_quantile_of(t);
            if (metric_type == METRIC_COUNTER) { /* counter.go:112-131 ValueOf */
                switch (t) {
                case AGG_MIN: r = (double)bs->imin; break;
                case AGG_MAX: r = (double)bs->imax; break;
                case AGG_MEAN: r = bs->count ? (double)bs->isum / (double)bs->count : 0; break;
                case AGG_COUNT: r = (double)bs->count; break;
                case AGG_SUM: r = (double)bs->isum; break;
                case AGG_SUMSQ: r = (double)bs->isumsq; break;
                case AGG_STDEV: r = m3_stdev(bs->count, (double)bs->isumsq, (double)bs->isum); break;
                default: r = 0; break;
                }
            } else if (metric_type == METRIC_GAUGE) { /* gauge.go:144-165 ValueOf */
                switch (t) {
                case AGG_LAST: r = bs->last; break;
                case AGG_MIN: r = bs->fmin; break;
                case AGG_MAX: r = bs->fmax; break;
                case AGG_MEAN: r = bs->count ? bs->fsum / (double)bs->count : 0.0; break;
                case AGG_COUNT: r = (double)bs->count; break;
                case AGG_SUM: r = bs->fsum; break;
                case AGG_SUMSQ: r = bs->fsumsq; break;
                case AGG_STDEV: r = m3_stdev(bs->count, bs->fsumsq, bs->fsum); break;
                default: r = 0; break;
                }
            } else { /* timer.go:131-153 ValueOf */
                if (q >= 0) {
                    r = streams ? stream_quantile(&streams[b], q) : 0.0;
                } else {
                    switch (t) {
                    case AGG_MIN: r = streams ? stream_quantile(&streams[b], 0.0) : 0.0; break;
                    case AGG_MAX: r = streams ? stream_quantile(&streams[b], 1.0) : 0.0; break;
                    case AGG_MEAN: r = bs->count ? bs->fsum / (double)bs->count : 0.0; break;
                    case AGG_COUNT: r = (double)bs->count; break;
                    case AGG_SUM: r = bs->fsum; break;
                    case AGG_SUMSQ: r = bs->fsumsq; break;
                    case AGG_STDEV: r = m3_stdev(bs->count, bs->fsumsq, bs->fsum); break;
                    default: r = 0; break;
                    }
                }
            }
            out[b * naggs + i] = r;
        }
    }

    if (streams) {
        for (int64_t b = 0; b < nbuckets; b++) stream_free(&streams[b]);
        free(streams);
    }
    free(st);
    return err;
}

/* type.go Quantile(): maps P* types to q values, Median=0.5 */
static double agg_quantile_of(int t) {
    switch (t) {
    case AGG_MEDIAN: return 0.5;
    case AGG_P10: return 0.1;
    case AGG_P20: return 0.2;
    case AGG_P25: return 0.25;
    case AGG_P30: return 0.3;
    case AGG_P40: return 0.4;
    case AGG_P50: return 0.5;
    case AGG_P60: return 0.6;
    case AGG_P70: return 0.7;
    case AGG_P75: return 0.75;
    case AGG_P80: return 0.8;
    case AGG_P90: return 0.9;
    case AGG_P95: return 0.95;
    case AGG_P99: return 0.99;
    case AGG_P999: return 0.999;
    case AGG_P9999: return 0.9999;
    default: return -1.0;
    }
}

/* Batch rollup over SoA decoded series (one thread per series chunk). */
int oracle_rollup_series(
    const int64_t* ts_ns, const double* vals, int64_t n,
    int metric_type, int64_t base_ns, int64_t window_ns, int64_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* out, int64_t* out_window_ts) {
    /* m3 CKMS production defaults (cm options.go:30-32) */
    return oracle_rollup_series_opts(ts_ns, vals, n, metric_type, base_ns,
                                     window_ns, nbuckets, agg_types, naggs,
                                     out, out_window_ts, 1e-3, 1024);
}

int oracle_rollup_batch_opts(
    const int64_t* ts_ns, const double* vals, const uint32_t* counts,
    int64_t nseries, int64_t stride,
    int metric_type, int64_t window_ns, int64_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* out /* nseries x nbuckets x naggs */,
    int64_t* out_window_ts /* nseries x nbuckets, or NULL */,
    int nthreads, double eps, int every) {
    int err = 0;
    (void)nthreads;
#pragma omp parallel for schedule(dynamic, 16) num_threads(nthreads)
    for (int64_t i = 0; i < nseries; i++) {
        if (err) continue;
        int64_t base = (ts_ns[i * stride] / window_ns) * window_ns;
        int r = oracle_rollup_series_opts(
            ts_ns + i * stride, vals + i * stride, (int64_t)counts[i],
            metric_type, base, window_ns, nbuckets, agg_types, naggs,
            out + i * nbuckets * naggs,
            out_window_ts ? out_window_ts + i * nbuckets : NULL, eps, every);
        if (r) {
#pragma omp critical
            err = r;
        }
    }
    return err;
}

int oracle_rollup_batch(
    const int64_t* ts_ns, const double* vals, const uint32_t* counts,
    int64_t nseries, int64_t stride,
    int metric_type, int64_t window_ns, int64_t nbuckets,
    const int32_t* agg_types, int naggs,
    double* out, int64_t* out_window_ts, int nthreads) {
    return oracle_rollup_batch_opts(ts_ns, vals, counts, nseries, stride,
                                    metric_type, window_ns, nbuckets,
                                    agg_types, naggs, out, out_window_ts,
                                    nthreads, 1e-3, 1024);
}

/* ==================== multi-replica deduplicating merge ====================
 * Restates the reference's MultiReaderIterator over ONE slice of R replica
 * iterators (dbnode/encoding/multi_reader_iterator.go:62-155 with the
 * iterators collection, dbnode/encoding/iterators.go:56-237), operating on
 * already-decoded (ts, val) arrays:
 *  - `values` keeps push order; an exhausted iterator is removed by
 *    swapping the TAIL into its slot (iterators.go:181-196);
 *  - the `earliest` set is rebuilt by scanning `values` in order
 *    (iterators.go:211-215 tryAddEarliest);
 *  - equal timestamps resolve via IterateLastPushed (the default,
 *    iterators_types.go:41-56): current() returns earliest[len-1]
 *    (iterators.go:60-90);
 *  - after advancing, an equal earliest time is deduped by advancing again
 *    (multi_reader_iterator.go:139-158), a smaller one is
 *    errOutOfOrderIterator (iterators.go:229-236).
 */

int oracle_merge_series(
    const int64_t* const* ts_rows, const double* const* val_rows,
    const uint32_t* counts, int nreplicas,
    int64_t* out_ts, double* out_vals, int64_t cap, int64_t* out_n) {
    int values[64]; /* replica ids in `values` order */
    int64_t cursor[64];
    int nvals = 0;
    if (nreplicas > 64) return -2;
    for (int r = 0; r < nreplicas; r++) {
        if (counts[r] > 0) {
            values[nvals] = r;
            cursor[r] = 0;
            nvals++;
        }
    }
    int64_t n = 0;
    int64_t prev_at = 0;
    int first = 1;
    while (nvals > 0) {
        /* rebuild earliest: scan values in order */
        int64_t earliest_at = INT64_MAX;
        int winner = -1; /* last in scan order among minima */
        for (int i = 0; i < nvals; i++) {
            int r = values[i];
            int64_t t = ts_rows[r][cursor[r]];
            if (t < earliest_at) {
                earliest_at = t;
                winner = r;
            } else if (t == earliest_at) {
                winner = r; /* IterateLastPushed: later scan position wins */
            }
        }
        if (!first) {
            if (earliest_at < prev_at) { *out_n = n; return -1; } /* errOutOfOrderIterator */
            if (earliest_at == prev_at) {
                /* dedupe: advance the earliest set again without emitting */
            }
        }
        if (first || earliest_at != prev_at) {
            if (n >= cap) { *out_n = n; return -3; }
            out_ts[n] = earliest_at;
            out_vals[n] = val_rows[winner][cursor[winner]];
            n++;
            prev_at = earliest_at;
            first = 0;
        }
        /* moveToValidNext: advance every iterator at earliest_at; remove
         * exhausted ones by swap-with-tail */
        for (int i = 0; i < nvals; i++) {
            int r = values[i];
            if (ts_rows[r][cursor[r]] == earliest_at) {
                cursor[r]++;
                if (cursor[r] >= (int64_t)counts[r]) {
                    values[i] = values[nvals - 1];
                    nvals--;
                    i--; /* re-examine the swapped-in entry */
                }
            }
        }
    }
    *out_n = n;
    return 0;
}

/* Batch form over SoA rows: replica-major layout ts[(r*nseries + i)*stride]. */
int oracle_merge_batch(
    const int64_t* ts, const double* vals, const uint32_t* counts,
    int nreplicas, int64_t nseries, int64_t stride,
    int64_t* out_ts, double* out_vals, uint32_t* out_counts,
    int64_t out_stride, int32_t* out_errs, int nthreads) {
    (void)nthreads;
#pragma omp parallel for schedule(dynamic, 64) num_threads(nthreads)
    for (int64_t i = 0; i < nseries; i++) {
        const int64_t* tr[64];
        const double* vr[64];
        uint32_t cr[64];
        for (int r = 0; r < nreplicas && r < 64; r++) {
            tr[r] = ts + ((int64_t)r * nseries + i) * stride;
            vr[r] = vals + ((int64_t)r * nseries + i) * stride;
            cr[r] = counts[(int64_t)r * nseries + i];
        }
        int64_t n = 0;
        int rc = oracle_merge_series(tr, vr, cr, nreplicas,
                                     out_ts + i * out_stride,
                                     out_vals + i * out_stride,
                                     out_stride, &n);
        out_counts[i] = (uint32_t)n;
        out_errs[i] = rc;
    }
    return 0;
}
