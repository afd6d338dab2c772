/*
 * staging.c — host-side heap-page → columnar SoA staging shim.
 *
 * The real provider's BeginCustomScan stages a relation once per
 * query/session: it walks heap pages (under the buffer lock, snapshot
 * visibility applied by heapgetpage — heapam.c:388,505), deforms each
 * visible tuple (heap_deform_tuple, access/common/heaptuple.c:936), and
 * lands attributes in per-column host arrays that otbx_memcpy_h2d ships to
 * the device column cache. This file restates the page
 * and tuple FORMAT walk of that staging step as plain host C; MVCC snapshot
 * checks stay on the server side of the boundary — here a line pointer
 * counts as visible iff LP_NORMAL, the post-heapgetpage state.
 *
 * On-disk layout restated from the reference headers (cited per struct):
 *  - PageHeaderData: 24-byte header {pd_lsn 8, pd_checksum 2, pd_flags 2,
 *    pd_lower 2, pd_upper 2, pd_special 2, pd_pagesize_version 2,
 *    pd_prune_xid 4}, then the ItemId array up to pd_lower
 *    (storage/bufpage.h:157-168).
 *  - ItemIdData: 4 bytes {lp_off:15, lp_flags:2, lp_len:15}; LP_NORMAL=1
 *    (storage/itemid.h:25,40).
 *  - HeapTupleHeaderData: {t_xmin 4, t_xmax 4, t_cid 4, t_ctid 6,
 *    t_infomask2 2, t_infomask 2, t_hoff 1}, then the optional NULL bitmap
 *    (present iff t_infomask & HEAP_HASNULL), padding to t_hoff, then the
 *    attribute data (access/htup_details.h:118-166). natts =
 *    t_infomask2 & HEAP_NATTS_MASK (0x07FF, htup_details.h:272).
 *  - Attribute placement: each fixed-width attribute is aligned to its
 *    typalign within the tuple data area (att_align_nominal,
 *    access/tupmacs.h:126); NULL attributes occupy no space.
 *
 * No reference code is copied: the layout above is the public on-disk
 * format, restated; this walker is an independent implementation.
 */
#include <stdint.h>
#include <string.h>

typedef int32_t otbx_status_i;
#define ST_OK 0
#define ST_ERR_INVALID 3 /* keep in sync with otbx_status (include/otbx.h) */

typedef struct {
    uint16_t attlen;   /* 1, 2, 4 or 8 (fixed-width pass-by-value types) */
    uint16_t attalign; /* alignment in bytes (typalign 'c'=1 's'=2 'i'=4 'd'=8) */
} otbx_attdesc;

#define PAGE_HEADER_BYTES 24
#define LP_NORMAL 1u
#define HEAP_HASNULL 0x0001u
#define HEAP_NATTS_MASK 0x07FFu

static inline uint16_t rd16(const uint8_t *p) { uint16_t v; memcpy(&v, p, 2); return v; }
static inline uint32_t rd32(const uint8_t *p) { uint32_t v; memcpy(&v, p, 4); return v; }

otbx_status_i otbx_stage_pages(const void *pages, int64_t npages,
                               size_t page_size, const otbx_attdesc *atts,
                               int32_t natts, void **out_cols,
                               uint8_t **out_nulls, int64_t cap_rows,
                               int64_t *nrows_out)
{
    if (!pages || !atts || !out_cols || !nrows_out || natts <= 0 ||
        page_size < PAGE_HEADER_BYTES || npages < 0)
        return ST_ERR_INVALID;
    for (int32_t a = 0; a < natts; a++) {
        uint16_t w = atts[a].attlen;
        if (w != 1 && w != 2 && w != 4 && w != 8) return ST_ERR_INVALID;
        uint16_t al = atts[a].attalign;
        if (al != 1 && al != 2 && al != 4 && al != 8) return ST_ERR_INVALID;
    }
    int64_t row = 0;
    for (int64_t pg = 0; pg < npages; pg++) {
        const uint8_t *page = (const uint8_t *)pages + (size_t)pg * page_size;
        uint16_t pd_lower = rd16(page + 12);
        uint16_t pd_upper = rd16(page + 14);
        if (pd_lower < PAGE_HEADER_BYTES || pd_lower > page_size ||
            pd_upper > page_size)
            return ST_ERR_INVALID; /* PageIsValid-style sanity */
        uint32_t nitems = (pd_lower - PAGE_HEADER_BYTES) / 4;
        for (uint32_t it = 0; it < nitems; it++) {
            uint32_t lp = rd32(page + PAGE_HEADER_BYTES + 4 * it);
            uint32_t lp_off = lp & 0x7FFFu;
            uint32_t lp_flags = (lp >> 15) & 0x3u;
            uint32_t lp_len = (lp >> 17) & 0x7FFFu;
            if (lp_flags != LP_NORMAL)
                continue; /* unused/dead/redirect: invisible post-heapgetpage */
            if (lp_off + lp_len > page_size || lp_len < 23 ||
                lp_off < pd_upper || lp_off < PAGE_HEADER_BYTES)
                return ST_ERR_INVALID; /* item must sit in the tuple area
                                        * [pd_upper, page_size) —
                                        * PageGetItemIdCareful analog */
            const uint8_t *tup = page + lp_off;
            uint16_t infomask2 = rd16(tup + 18);
            uint16_t infomask = rd16(tup + 20);
            uint8_t t_hoff = tup[22];
            uint32_t tup_natts = infomask2 & HEAP_NATTS_MASK;
            if (t_hoff > lp_len) return ST_ERR_INVALID;
            /* the NULL bitmap (when present) lives at [23, t_hoff): a
             * malformed t_hoff that doesn't cover (tup_natts+7)/8 bitmap
             * bytes would send the bits[] reads past the tuple — and past
             * the caller's buffer on the last page */
            if ((infomask & HEAP_HASNULL) &&
                (uint32_t)t_hoff < 23u + (tup_natts + 7u) / 8u)
                return ST_ERR_INVALID;
            const uint8_t *bits =
                (infomask & HEAP_HASNULL) ? tup + 23 : (const uint8_t *)0;
            if (row >= cap_rows) return ST_ERR_INVALID;
            /* deform: walk attributes in order, aligning as stored
             * (heap_deform_tuple's fixed-width fast path) */
            size_t off = t_hoff;
            for (int32_t a = 0; a < natts; a++) {
                int isnull =
                    a >= (int32_t)tup_natts /* added column: NULL
                                             * (heap_getattr semantics) */
                    || (bits && !(bits[a >> 3] & (1u << (a & 7))));
                if (out_nulls && out_nulls[a]) out_nulls[a][row] = (uint8_t)isnull;
                uint16_t w = atts[a].attlen;
                if (isnull) {
                    if (!out_nulls || !out_nulls[a])
                        return ST_ERR_INVALID; /* caller declared NOT NULL */
                    memset((uint8_t *)out_cols[a] + (size_t)row * w, 0, w);
                    continue; /* NULL occupies no space in the tuple */
                }
                size_t al = atts[a].attalign;
                off = (off + al - 1) & ~(al - 1);
                if (lp_off + off + w > (size_t)lp_off + lp_len)
                    return ST_ERR_INVALID;
                memcpy((uint8_t *)out_cols[a] + (size_t)row * w, tup + off, w);
                off += w;
            }
            row++;
        }
    }
    *nrows_out = row;
    return ST_OK;
}
