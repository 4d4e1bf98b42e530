/* fsdr_fg.cpp — minimal native flowgraph driver for 1-in/1-out chains.
 *
 * Mirrors the reference runtime semantics for the hot path (cited):
 *  - Flowgraph::add / Flowgraph::stream wiring — src/runtime/flowgraph.rs:
 *    227-241, 364-423 (typed BufferWriter -> BufferReader pairs).
 *  - The block actor loop — src/runtime/wrapped_kernel.rs:106-229: call
 *    work() while progress is possible; a block finishes when its upstream
 *    finished and work did not return InsufficientOutput
 *    (src/blocks/fir.rs:89-91); NullSource never finishes, Head bounds the
 *    stream (src/blocks/head.rs:23-57).
 *  - Stream buffers keep a min_items-1 history prefix for the reader, the
 *    Slab reserved-prefix mechanism (src/runtime/buffer/slab.rs:369-399),
 *    realized here as a device-resident linear buffer with tail compaction.
 *  - Scheduling: single-threaded round-robin over blocks = the semantics of
 *    the reference's smol1 scheduler configuration (scheduler/smol.rs:62-106
 *    with one worker), which is perf/fir's default (perf/fir/fir.rs:77-80).
 *
 * Data stays resident in HBM between blocks; only sources/sinks touch host
 * memory. All data-plane work (kernels, D2D compaction, source fills) is
 * ENQUEUED asynchronously on one HIP stream — the host actor loop's
 * bookkeeping is pure status math, so it runs ahead of the device and the
 * only synchronization points are capture sinks and run() completion.
 * This is the harness the fg-level tests and INTEGRATION.md refer to; the
 * fused fsdr_chain_* path is the production pipeline.
 */
#include <hip/hip_runtime.h>

#include <cstring>
#include <vector>

#include "../../include/futuresdr_hip.h"

extern "C" size_t fsdr_filter_item_sizes(const fsdr_filter*, size_t*);

namespace {

struct Edge {
    void* dev = nullptr;       /* device ring */
    size_t cap = 0;            /* items */
    size_t item_bytes = 8;
    size_t r = 0, w = 0;       /* cursors (r <= w <= cap) */
    size_t headroom = 0;       /* >= reader min_items (capacity slack) */
    int dst_block = -1;
    bool writer_finished = false;

    size_t readable() const { return w - r; }
    size_t writable() const { return cap - w; }
};

enum BKind { B_NULL_SRC, B_VEC_SRC, B_HEAD, B_FILTER, B_NULL_SINK,
             B_VEC_SINK };

struct Block {
    BKind kind;
    fsdr_filter* filter = nullptr;       /* B_FILTER (borrowed) */
    std::vector<char> vec;               /* vector source data / sink copy */
    size_t vec_pos = 0;
    unsigned long long head_n = 0;       /* B_HEAD remaining */
    unsigned long long n_received = 0;   /* null/vector sink */
    int in_edge = -1, out_edge = -1;
    bool finished = false;
    size_t item_in = 8, item_out = 8;
};

}  // namespace

struct fsdr_fg {
    std::vector<Block> blocks;
    std::vector<Edge> edges;
    size_t default_cap = 1 << 18; /* items per stream buffer */
    bool dead = false;
    /* all data-plane work is ENQUEUED on this stream; the actor loop's
     * bookkeeping is host-side status math (no readbacks), so the
     * round-robin runs ahead of the GPU and the only syncs are capture
     * sinks and run() end — the async analogue of the reference's
     * per-block tasks overlapping on one queue. */
    hipStream_t stream = nullptr;
};

static bool fg_have_gpu() {
    int n = 0;
    return hipGetDeviceCount(&n) == hipSuccess && n > 0;
}

extern "C" fsdr_fg* fsdr_fg_create(void) {
    if (!fg_have_gpu()) return nullptr; /* product path: no CPU fallback */
    fsdr_fg* fg = new fsdr_fg();
    if (hipStreamCreate(&fg->stream) != hipSuccess) {
        delete fg;
        return nullptr;
    }
    return fg;
}

extern "C" void fsdr_fg_destroy(fsdr_fg* fg) {
    if (!fg) return;
    if (fg->stream) {
        (void)hipStreamSynchronize(fg->stream);
        (void)hipStreamDestroy(fg->stream);
    }
    for (auto& e : fg->edges)
        if (e.dev) (void)hipFree(e.dev);
    delete fg;
}

static int add_block(fsdr_fg* fg, Block b) {
    fg->blocks.push_back(std::move(b));
    return (int)fg->blocks.size() - 1;
}

extern "C" int fsdr_fg_add_null_source_cf32(fsdr_fg* fg) {
    Block b; b.kind = B_NULL_SRC;
    return add_block(fg, std::move(b));
}

extern "C" int fsdr_fg_add_vector_source_cf32(fsdr_fg* fg,
                                              const fsdr_cf32* data,
                                              size_t n) {
    Block b; b.kind = B_VEC_SRC;
    b.vec.assign((const char*)data, (const char*)data + n * sizeof(fsdr_cf32));
    return add_block(fg, std::move(b));
}

extern "C" int fsdr_fg_add_head(fsdr_fg* fg, unsigned long long n) {
    Block b; b.kind = B_HEAD; b.head_n = n;
    return add_block(fg, std::move(b));
}

extern "C" int fsdr_fg_add_filter(fsdr_fg* fg, fsdr_filter* f) {
    Block b; b.kind = B_FILTER; b.filter = f;
    /* mag2 narrows cf32 -> f32; everything else on this path is 8B items */
    size_t out_b = 8;
    b.item_in = fsdr_filter_item_sizes(f, &out_b);
    b.item_out = out_b;
    return add_block(fg, std::move(b));
}

extern "C" int fsdr_fg_add_null_sink(fsdr_fg* fg) {
    Block b; b.kind = B_NULL_SINK;
    return add_block(fg, std::move(b));
}

extern "C" int fsdr_fg_add_vector_sink(fsdr_fg* fg) {
    Block b; b.kind = B_VEC_SINK;
    return add_block(fg, std::move(b));
}

/* connect!(fg, a > b): flowgraph.rs:364-423 — allocates the shared stream
 * buffer and sets reserved history = reader min_items - 1 (slab.rs:113,
 * blocks/fir.rs:48-49 set_min_items(filter.length())). */
extern "C" int fsdr_fg_stream(fsdr_fg* fg, int src, int dst) {
    if (!fg || src < 0 || dst < 0 || src >= (int)fg->blocks.size() ||
        dst >= (int)fg->blocks.size())
        return FSDR_ERR_INVALID;
    Block& bs = fg->blocks[src];
    Block& bd = fg->blocks[dst];
    Edge e;
    e.item_bytes = bs.item_out;
    size_t min_items =
        bd.kind == B_FILTER ? fsdr_filter_length(bd.filter) : 1;
    e.headroom = min_items;
    e.cap = fg->default_cap + e.headroom;
    e.dst_block = dst;
    if (hipMalloc(&e.dev, e.cap * e.item_bytes) != hipSuccess)
        return FSDR_ERR_HIP;
    fg->edges.push_back(e);
    int eid = (int)fg->edges.size() - 1;
    bs.out_edge = eid;
    bd.in_edge = eid;
    bd.item_in = bs.item_out;
    return FSDR_OK;
}

/* compaction = the slab reserved-prefix tail copy (slab.rs:369-399):
 * move [r - reserved, w) to the buffer start so the writer regains space
 * while the reader keeps its history. */
static int edge_compact(Edge& e, hipStream_t st) {
    size_t keep_from = e.r;
    size_t keep = e.w - keep_from;
    if (keep_from == 0) return FSDR_OK; /* nothing to gain */
    if (keep > 0) {
        /* same-buffer D2D copy; regions may overlap only if keep >
         * keep_from, which cannot happen since keep <= cap - keep_from
         * and we only compact when the writer is starved; use a bounce
         * via memcpyDtoD which requires non-overlap — guard it. */
        if (keep_from >= keep) {
            if (hipMemcpyAsync((char*)e.dev,
                               (char*)e.dev + keep_from * e.item_bytes,
                               keep * e.item_bytes,
                               hipMemcpyDeviceToDevice, st) != hipSuccess)
                return FSDR_ERR_HIP;
        } else {
            /* overlapping: chunked forward copy (same stream = ordered) */
            size_t done = 0;
            while (done < keep) {
                size_t c = keep_from < keep - done ? keep_from : keep - done;
                if (hipMemcpyAsync(
                        (char*)e.dev + done * e.item_bytes,
                        (char*)e.dev + (keep_from + done) * e.item_bytes,
                        c * e.item_bytes, hipMemcpyDeviceToDevice,
                        st) != hipSuccess)
                    return FSDR_ERR_HIP;
                done += c;
            }
        }
    }
    e.r -= keep_from;
    e.w -= keep_from;
    return FSDR_OK;
}

/* one work() call; returns items of progress (consumed+produced) */
static long long block_work(fsdr_fg* fg, Block& b) {
    Edge* ie = b.in_edge >= 0 ? &fg->edges[b.in_edge] : nullptr;
    Edge* oe = b.out_edge >= 0 ? &fg->edges[b.out_edge] : nullptr;
    if (b.finished) return 0;
    switch (b.kind) {
        case B_NULL_SRC: { /* null_source.rs:53-66: zero-fill all space */
            if (!oe) return 0;
            if (oe->writable() == 0) {
                if (edge_compact(*oe, fg->stream) != FSDR_OK) return -1;
            }
            size_t n = oe->writable();
            if (n == 0) return 0;
            if (hipMemsetAsync((char*)oe->dev + oe->w * oe->item_bytes, 0,
                               n * oe->item_bytes,
                               fg->stream) != hipSuccess)
                return -1;
            oe->w += n;
            return (long long)n;
        }
        case B_VEC_SRC: { /* vector_source.rs: emit once, then finish */
            if (!oe) return 0;
            size_t left = (b.vec.size() - b.vec_pos) / oe->item_bytes;
            if (left == 0) {
                b.finished = true;
                oe->writer_finished = true;
                return 0;
            }
            if (oe->writable() == 0 && edge_compact(*oe, fg->stream) != FSDR_OK)
                return -1;
            size_t n = oe->writable() < left ? oe->writable() : left;
            if (n == 0) return 0;
            if (hipMemcpyAsync((char*)oe->dev + oe->w * oe->item_bytes,
                               b.vec.data() + b.vec_pos,
                               n * oe->item_bytes, hipMemcpyHostToDevice,
                               fg->stream) != hipSuccess)
                return -1;
            oe->w += n;
            b.vec_pos += n * oe->item_bytes;
            return (long long)n;
        }
        case B_HEAD: { /* head.rs:23-57 */
            if (!ie || !oe) return 0;
            size_t avail = ie->readable();
            size_t n = avail;
            if ((unsigned long long)n > b.head_n) n = (size_t)b.head_n;
            if (oe->writable() < n) {
                if (edge_compact(*oe, fg->stream) != FSDR_OK) return -1;
                if (oe->writable() < n) n = oe->writable();
            }
            if (n > 0) {
                if (hipMemcpyAsync((char*)oe->dev + oe->w * oe->item_bytes,
                                   (char*)ie->dev + ie->r * ie->item_bytes,
                                   n * ie->item_bytes,
                                   hipMemcpyDeviceToDevice,
                                   fg->stream) != hipSuccess)
                    return -1;
                ie->r += n;
                oe->w += n;
                b.head_n -= n;
            }
            if (b.head_n == 0 ||
                (ie->writer_finished && ie->readable() == 0)) {
                b.finished = true;
                oe->writer_finished = true;
            }
            return (long long)n;
        }
        case B_FILTER: {
            if (!ie || !oe) return 0;
            /* slice = [r, w): the filter consumes `consumed` items and
             * leaves the taps-1 tail unconsumed (fir.rs:69-91); that tail
             * is the history the next call sees. */
            size_t n_in = ie->readable();
            if (oe->writable() < fg->default_cap / 2 &&
                edge_compact(*oe, fg->stream) != FSDR_OK)
                return -1;
            size_t n_out = oe->writable();
            fsdr_filter_result r;
            int rc = fsdr_filter_dev(
                b.filter, (char*)ie->dev + ie->r * ie->item_bytes, n_in,
                (char*)oe->dev + oe->w * oe->item_bytes, n_out, fg->stream,
                &r);
            if (rc != FSDR_OK) return -1;
            ie->r += r.consumed;
            oe->w += r.produced;
            if (ie->r > 0) (void)edge_compact(*ie, fg->stream);
            /* fir.rs:89-91 / wrapped_kernel finish propagation: upstream
             * finished and not output-limited -> done (the unconsumable
             * tail < min_items is dropped, like the reference) */
            if (ie->writer_finished && r.status != FSDR_INSUFFICIENT_OUTPUT &&
                r.consumed == 0 && r.produced == 0) {
                b.finished = true;
                oe->writer_finished = true;
            }
            return (long long)(r.consumed + r.produced);
        }
        case B_NULL_SINK: { /* null_sink.rs:63-71 */
            if (!ie) return 0;
            size_t n = ie->readable();
            b.n_received += n;
            ie->r += n;
            if (n > 0) (void)edge_compact(*ie, fg->stream);
            if (ie->writer_finished && ie->readable() == 0)
                b.finished = true;
            return (long long)n;
        }
        case B_VEC_SINK: { /* vector_sink.rs:20-39: capture everything */
            if (!ie) return 0;
            size_t n = ie->readable();
            if (n > 0) {
                size_t old = b.vec.size();
                b.vec.resize(old + n * ie->item_bytes);
                /* capture sink: drain enqueued work, then a blocking
                 * copy into the (reallocatable) host vector */
                if (hipStreamSynchronize(fg->stream) != hipSuccess)
                    return -1;
                if (hipMemcpy(b.vec.data() + old,
                              (char*)ie->dev + ie->r * ie->item_bytes,
                              n * ie->item_bytes,
                              hipMemcpyDeviceToHost) != hipSuccess)
                    return -1;
                ie->r += n;
                b.n_received += n;
                (void)edge_compact(*ie, fg->stream);
            }
            if (ie->writer_finished && ie->readable() == 0)
                b.finished = true;
            return (long long)n;
        }
    }
    return 0;
}

/* Runtime::run (runtime.rs:169-215): loop until every block with a finite
 * stream is finished or no block makes progress. */
extern "C" int fsdr_fg_run(fsdr_fg* fg) {
    if (!fg) return FSDR_ERR_INVALID;
    for (;;) {
        long long progress = 0;
        bool all_done = true;
        for (auto& b : fg->blocks) {
            long long p = block_work(fg, b);
            if (p < 0) return FSDR_ERR_HIP;
            progress += p;
            if (!b.finished) all_done = false;
        }
        for (int i = (int)fg->blocks.size() - 1; i >= 0; i--) {
            Block& b = fg->blocks[i];
            if (!b.finished && b.out_edge >= 0) {
                int dst = fg->edges[b.out_edge].dst_block;
                if (dst >= 0 && fg->blocks[dst].finished) b.finished = true;
            }
        }
        all_done = true;
        for (auto& b : fg->blocks)
            if (!b.finished) all_done = false;
        if (all_done) break;
        if (progress == 0) {
            /* deadlock = a NullSource-fed graph without Head, or a stuck
             * block; the reference would keep blocking — we error out */
            bool has_unbounded_src = false;
            for (auto& b : fg->blocks)
                if (b.kind == B_NULL_SRC && !b.finished)
                    has_unbounded_src = true;
            if (!has_unbounded_src) break;
            return FSDR_ERR_INVALID;
        }
    }
    (void)hipStreamSynchronize(fg->stream);
    return FSDR_OK;
}

extern "C" unsigned long long fsdr_fg_n_received(fsdr_fg* fg, int block) {
    if (!fg || block < 0 || block >= (int)fg->blocks.size()) return 0;
    return fg->blocks[block].n_received;
}

extern "C" size_t fsdr_fg_vector_sink_get(fsdr_fg* fg, int block, void* out,
                                          size_t cap_bytes) {
    if (!fg || block < 0 || block >= (int)fg->blocks.size()) return 0;
    Block& b = fg->blocks[block];
    size_t n = b.vec.size() < cap_bytes ? b.vec.size() : cap_bytes;
    if (out && n) memcpy(out, b.vec.data(), n);
    return b.vec.size();
}
