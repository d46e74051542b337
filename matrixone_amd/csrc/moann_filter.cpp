/* Host-side docfilter membership evaluation: turns MatrixOne's serialized
 * membership filters into the slot/key bitsets the HIP search kernels
 * consume (the cuVS precedent is eval_filter_bitmap_cpu,
 * cgo/cuvs/filter.hpp:855-960; the per-candidate predicate being replaced
 * is filtered_search_membership_cb, cgo/usearchex.c:27-49).
 *
 * Filter kinds and payload formats mirror the reference exactly
 * (tag bytes cgo/usearchex.h:30-33; MUST stay in sync with
 * pkg/common/docfilter):
 *   0 BLOOM    — bloomfilter_t flat bytes {magic "XXBF", k u32, nbits u64,
 *                seed u64, bitmap u64[]} (cgo/bloom.h:36-42); membership =
 *                k double-hash probes (h1 + i*h2) & (nbits-1) of
 *                XXH3_128bits_withSeed(key8, seed) (cgo/bloom.c:33-79,177).
 *   2 CROARING — CRoaring roaring64 PORTABLE serialization
 *                (cgo/croaring.c:99-112); parsed read-only here.
 *   3 CBITMAP  — dense little-endian [base u64][nbits u64][words u64[]]
 *                (cgo/cbitmap.c:159-201); bit i == value base+i.
 *   4 SORTED64 — [count u64][count sorted u64] binary search
 *                (cgo/sorted64.c:26-57).
 * Unknown kinds fail CLOSED (usearchex.c:42-48) and report an error.
 *
 * All implementations here are written fresh against the formats above and
 * validated in tests/test_docfilter.py against the reference's own
 * compiled filters (oracle/_ref) and golden payloads. */

#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <string>
#include <vector>

#include "../../include/moann.h"

namespace {

/* ---------------- XXH3 128-bit, 8-byte-input path ------------------------
 * Restatement of the published xxHash (BSD-2) XXH3_len_4to8_128b algorithm
 * with the default secret, specialized to len == 8 — the only length the
 * search path hashes (doc keys are uint64; bloom_calculate_hash forces all
 * fixed ints through the 8-byte route, cgo/bloom.c:40-56). Validated
 * bit-exactly against python-xxhash and the reference bloom.c build. */

static const uint8_t kXxhSecret16to32[16] = {
    0xde, 0xd4, 0x6d, 0xe9, 0x83, 0x90, 0x97, 0xdb,
    0x72, 0x40, 0xa4, 0xa4, 0xb7, 0xb3, 0x67, 0x1f,
};

static inline uint64_t rd64(const uint8_t* p) {
    uint64_t v;
    std::memcpy(&v, p, 8);
    return v;
}
static inline uint32_t rd32(const uint8_t* p) {
    uint32_t v;
    std::memcpy(&v, p, 4);
    return v;
}
static inline uint32_t swap32(uint32_t x) { return __builtin_bswap32(x); }

struct U128 {
    uint64_t lo, hi;
};

static inline U128 mult64to128(uint64_t a, uint64_t b) {
    __uint128_t p = (__uint128_t)a * b;
    return {(uint64_t)p, (uint64_t)(p >> 64)};
}
static inline uint64_t xorshift64(uint64_t v, int s) { return v ^ (v >> s); }
static inline uint64_t xxh3_avalanche(uint64_t h) {
    h ^= h >> 37;
    h *= 0x165667919E3779F9ULL;
    h ^= h >> 32;
    return h;
}

static U128 xxh3_128_len8_seed(uint64_t key_le, uint64_t seed) {
    const uint64_t PRIME64_1 = 0x9E3779B185EBCA87ULL;
    const size_t len = 8;
    uint8_t data[8];
    std::memcpy(data, &key_le, 8);
    seed ^= (uint64_t)swap32((uint32_t)seed) << 32;
    const uint32_t input_lo = rd32(data);
    const uint32_t input_hi = rd32(data + len - 4);
    const uint64_t input_64 = input_lo + ((uint64_t)input_hi << 32);
    const uint64_t bitflip =
        (rd64(kXxhSecret16to32) ^ rd64(kXxhSecret16to32 + 8)) + seed;
    const uint64_t keyed = input_64 ^ bitflip;
    U128 m128 = mult64to128(keyed, PRIME64_1 + (len << 2));
    m128.hi += m128.lo << 1;
    m128.lo ^= m128.hi >> 3;
    m128.lo = xorshift64(m128.lo, 35);
    m128.lo *= 0x9FB21C651E98DF25ULL;
    m128.lo = xorshift64(m128.lo, 28);
    m128.hi = xxh3_avalanche(m128.hi);
    return m128;
}

/* ---------------- bloom ------------------------------------------------- */

struct BloomView {
    uint32_t k = 0;
    uint64_t nbits = 0, seed = 0;
    const uint64_t* words = nullptr;
    uint64_t nwords = 0;
    bool ok = false;
};

static BloomView bloom_view(const uint8_t* p, uint64_t len) {
    BloomView v;
    if (len < 24 || std::memcmp(p, "XXBF", 4) != 0) return v;
    std::memcpy(&v.k, p + 4, 4);
    std::memcpy(&v.nbits, p + 8, 8);
    std::memcpy(&v.seed, p + 16, 8);
    v.words = (const uint64_t*)(p + 24);
    v.nwords = (len - 24) / 8;
    if ((v.nbits + 63) / 64 > v.nwords || v.k > 64) return v;
    v.ok = true;
    return v;
}

static bool bloom_test(const BloomView& v, uint64_t key) {
    if (v.nbits == 0) return false;
    const U128 h = xxh3_128_len8_seed(key, v.seed);
    for (uint32_t i = 0; i < v.k; ++i) {
        const uint64_t pos = (h.lo + (uint64_t)i * h.hi) & (v.nbits - 1);
        if (!((v.words[pos >> 6] >> (pos & 63)) & 1ull)) return false;
    }
    return true;
}

/* ---------------- cbitmap ------------------------------------------------ */

static bool cbitmap_test(const uint8_t* p, uint64_t len, uint64_t key) {
    if (len < 16) return false;
    uint64_t base, nbits;
    std::memcpy(&base, p, 8);
    std::memcpy(&nbits, p + 8, 8);
    if (key < base) return false;
    const uint64_t bit = key - base;
    if (bit >= nbits) return false;
    const uint64_t w = bit >> 6;
    if (16 + (w + 1) * 8 > len) return false;
    uint64_t word;
    std::memcpy(&word, p + 16 + w * 8, 8);
    return (word >> (bit & 63)) & 1ull;
}

/* ---------------- sorted64 ----------------------------------------------- */

static bool sorted64_test(const uint8_t* p, uint64_t len, uint64_t key) {
    if (len < 8) return false;
    uint64_t count;
    std::memcpy(&count, p, 8);
    if (count > (len - 8) / 8) count = (len - 8) / 8; /* defensive clamp */
    const uint8_t* vals = p + 8;
    uint64_t lo = 0, hi = count;
    while (lo < hi) {
        const uint64_t mid = lo + (hi - lo) / 2;
        uint64_t v;
        std::memcpy(&v, vals + mid * 8, 8);
        if (v < key)
            lo = mid + 1;
        else
            hi = mid;
    }
    if (lo >= count) return false;
    uint64_t v;
    std::memcpy(&v, vals + lo * 8, 8);
    return v == key;
}

/* ---------------- roaring64 portable view -------------------------------
 * Read-only parser for CRoaring's roaring64 portable format: u64 bucket
 * count, then per bucket a u32 high-32 key and a portable roaring32
 * bitmap (RoaringFormatSpec: cookie, descriptive headers, optional offset
 * table, then array/bitset/run containers). */

struct R32Container {
    uint16_t key;
    uint32_t cardinality; /* card-1 stored; here the real value */
    bool is_run;
    const uint8_t* data;  /* container payload start */
};

struct R32View {
    std::vector<R32Container> conts;
};

struct R64View {
    std::vector<uint32_t> high;
    std::vector<R32View> maps;
    bool ok = false;
};

static bool parse_r32(const uint8_t*& p, const uint8_t* end, R32View* out) {
    if (p + 4 > end) return false;
    const uint32_t cookie = rd32(p);
    p += 4;
    uint32_t size;
    std::vector<uint8_t> runflags;
    const bool has_run = (cookie & 0xFFFF) == 12347;
    if (has_run) {
        size = (cookie >> 16) + 1;
        const uint32_t fb = (size + 7) / 8;
        if (p + fb > end) return false;
        runflags.assign(p, p + fb);
        p += fb;
    } else if (cookie == 12346) {
        if (p + 4 > end) return false;
        size = rd32(p);
        p += 4;
    } else {
        return false;
    }
    if (p + (size_t)size * 4 > end) return false;
    out->conts.resize(size);
    for (uint32_t i = 0; i < size; ++i) {
        out->conts[i].key = (uint16_t)(p[0] | (p[1] << 8));
        out->conts[i].cardinality = (uint32_t)(p[2] | (p[3] << 8)) + 1;
        out->conts[i].is_run =
            has_run && ((runflags[i / 8] >> (i % 8)) & 1);
        p += 4;
    }
    /* offset table present when no-run cookie, or >= 4 containers */
    if (!has_run || size >= 4) {
        if (p + (size_t)size * 4 > end) return false;
        p += (size_t)size * 4;
    }
    for (uint32_t i = 0; i < size; ++i) {
        auto& c = out->conts[i];
        c.data = p;
        size_t bytes;
        if (c.is_run) {
            if (p + 2 > end) return false;
            const uint16_t nruns = (uint16_t)(p[0] | (p[1] << 8));
            bytes = 2 + (size_t)nruns * 4;
        } else if (c.cardinality > 4096) {
            bytes = 8192;
        } else {
            bytes = (size_t)c.cardinality * 2;
        }
        if (p + bytes > end) return false;
        p += bytes;
    }
    return true;
}

static bool r32_contains(const R32View& v, uint32_t low32) {
    const uint16_t key = (uint16_t)(low32 >> 16);
    const uint16_t val = (uint16_t)(low32 & 0xFFFF);
    size_t lo = 0, hi = v.conts.size();
    while (lo < hi) {
        const size_t mid = (lo + hi) / 2;
        if (v.conts[mid].key < key)
            lo = mid + 1;
        else
            hi = mid;
    }
    if (lo >= v.conts.size() || v.conts[lo].key != key) return false;
    const R32Container& c = v.conts[lo];
    const uint8_t* d = c.data;
    if (c.is_run) {
        const uint16_t nruns = (uint16_t)(d[0] | (d[1] << 8));
        d += 2;
        size_t a = 0, b = nruns;
        while (a < b) { /* runs sorted by start */
            const size_t mid = (a + b) / 2;
            const uint16_t start = (uint16_t)(d[mid * 4] | (d[mid * 4 + 1] << 8));
            const uint16_t rlen = (uint16_t)(d[mid * 4 + 2] | (d[mid * 4 + 3] << 8));
            if (val < start)
                b = mid;
            else if ((uint32_t)val > (uint32_t)start + rlen)
                a = mid + 1;
            else
                return true;
        }
        return false;
    }
    if (c.cardinality > 4096) { /* bitset */
        return (d[val >> 3] >> (val & 7)) & 1;
    }
    /* sorted u16 array */
    size_t a = 0, b = c.cardinality;
    while (a < b) {
        const size_t mid = (a + b) / 2;
        const uint16_t x = (uint16_t)(d[mid * 2] | (d[mid * 2 + 1] << 8));
        if (x < val)
            a = mid + 1;
        else
            b = mid;
    }
    return a < c.cardinality &&
           (uint16_t)(d[a * 2] | (d[a * 2 + 1] << 8)) == val;
}

static R64View parse_r64(const uint8_t* p, uint64_t len) {
    R64View v;
    const uint8_t* end = p + len;
    if (p + 8 > end) return v;
    uint64_t nbuckets;
    std::memcpy(&nbuckets, p, 8);
    p += 8;
    /* never reserve from an untrusted count: each bucket needs >= 8
     * payload bytes, so cap by what the buffer could possibly hold */
    const uint64_t cap = (uint64_t)(end - p) / 8 + 1;
    const uint64_t r = nbuckets < cap ? nbuckets : cap;
    v.high.reserve(r);
    v.maps.reserve(r);
    for (uint64_t i = 0; i < nbuckets; ++i) {
        if (p + 4 > end) return v;
        v.high.push_back(rd32(p));
        p += 4;
        v.maps.emplace_back();
        if (!parse_r32(p, end, &v.maps.back())) {
            v.high.pop_back();
            v.maps.pop_back();
            return v;
        }
    }
    v.ok = true;
    return v;
}

static bool r64_contains(const R64View& v, uint64_t key) {
    const uint32_t hi = (uint32_t)(key >> 32);
    size_t lo = 0, hh = v.high.size();
    while (lo < hh) {
        const size_t mid = (lo + hh) / 2;
        if (v.high[mid] < hi)
            lo = mid + 1;
        else
            hh = mid;
    }
    if (lo >= v.high.size() || v.high[lo] != hi) return false;
    return r32_contains(v.maps[lo], (uint32_t)key);
}

static void set_err(void* errmsg, const char* where, const char* what) {
    if (!errmsg) return;
    std::string m = std::string(where) + ": " + what;
    char* c = (char*)malloc(m.size() + 1);
    if (c) std::memcpy(c, m.c_str(), m.size() + 1);
    *(char**)errmsg = c;
}

}  // namespace

extern "C" int moann_docfilter_eval(int kind, const void* payload,
                                    uint64_t payload_len, const int64_t* keys,
                                    uint64_t nkeys, uint32_t* out_words,
                                    void* errmsg) {
    const uint64_t nw = (nkeys + 31) / 32;
    std::memset(out_words, 0, nw * 4);
    if (!payload) return 0; /* NULL filter passes nothing into the bitset
                               path; the NULL-passes-all convention is the
                               caller's (it just skips filtering) */
    const uint8_t* p = (const uint8_t*)payload;
    switch (kind) {
    case MOANN_FILTER_BLOOM: {
        const BloomView v = bloom_view(p, payload_len);
        if (!v.ok) {
            set_err(errmsg, "moann_docfilter_eval", "bad bloom payload");
            return -1;
        }
        for (uint64_t i = 0; i < nkeys; ++i)
            if (bloom_test(v, (uint64_t)keys[i]))
                out_words[i >> 5] |= 1u << (i & 31);
        return 0;
    }
    case MOANN_FILTER_CROARING: {
        const R64View v = parse_r64(p, payload_len);
        if (!v.ok) {
            set_err(errmsg, "moann_docfilter_eval", "bad roaring64 payload");
            return -1;
        }
        for (uint64_t i = 0; i < nkeys; ++i)
            if (r64_contains(v, (uint64_t)keys[i]))
                out_words[i >> 5] |= 1u << (i & 31);
        return 0;
    }
    case MOANN_FILTER_CBITMAP:
        for (uint64_t i = 0; i < nkeys; ++i)
            if (cbitmap_test(p, payload_len, (uint64_t)keys[i]))
                out_words[i >> 5] |= 1u << (i & 31);
        return 0;
    case MOANN_FILTER_SORTED64:
        for (uint64_t i = 0; i < nkeys; ++i)
            if (sorted64_test(p, payload_len, (uint64_t)keys[i]))
                out_words[i >> 5] |= 1u << (i & 31);
        return 0;
    default:
        /* fail closed, like usearchex.c:42-48 */
        set_err(errmsg, "moann_docfilter_eval", "unknown filter kind");
        return -1;
    }
}

/* ---- docfilter handles (the Go bridge's per-filter C handle shape,
 * cgo/usearchex.h:38-42 / pkg/vectorindex/usearchex/search.go:37-109) ----- */

namespace {
struct MoannDocFilter {
    int kind;
    std::vector<uint8_t> payload;
};
}  // namespace

extern "C" moann_docfilter_c moann_docfilter_parse(int kind,
                                                   const void* payload,
                                                   uint64_t payload_len,
                                                   void* errmsg) {
    try {
        if (!payload) {
            set_err(errmsg, "moann_docfilter_parse", "null payload");
            return nullptr;
        }
        /* validate by a trial evaluation over zero keys' parse step */
        const uint8_t* p = (const uint8_t*)payload;
        switch (kind) {
        case MOANN_FILTER_BLOOM:
            if (!bloom_view(p, payload_len).ok) {
                set_err(errmsg, "moann_docfilter_parse", "bad bloom payload");
                return nullptr;
            }
            break;
        case MOANN_FILTER_CROARING:
            if (!parse_r64(p, payload_len).ok) {
                set_err(errmsg, "moann_docfilter_parse",
                        "bad roaring64 payload");
                return nullptr;
            }
            break;
        case MOANN_FILTER_CBITMAP:
            if (payload_len < 16) {
                set_err(errmsg, "moann_docfilter_parse",
                        "bad cbitmap payload");
                return nullptr;
            }
            break;
        case MOANN_FILTER_SORTED64:
            if (payload_len < 8) {
                set_err(errmsg, "moann_docfilter_parse",
                        "bad sorted64 payload");
                return nullptr;
            }
            break;
        default:
            set_err(errmsg, "moann_docfilter_parse", "unknown filter kind");
            return nullptr;
        }
        auto f = new MoannDocFilter;
        f->kind = kind;
        f->payload.assign(p, p + payload_len);
        return (moann_docfilter_c)f;
    } catch (const std::exception& e) {
        set_err(errmsg, "moann_docfilter_parse", e.what());
        return nullptr;
    }
}

extern "C" int moann_docfilter_test(moann_docfilter_c fh, const int64_t* keys,
                                    uint64_t nkeys, uint32_t* out_words,
                                    void* errmsg) {
    if (!fh) {
        set_err(errmsg, "moann_docfilter_test", "null filter");
        return -1;
    }
    auto f = (MoannDocFilter*)fh;
    return moann_docfilter_eval(f->kind, f->payload.data(),
                                f->payload.size(), keys, nkeys, out_words,
                                errmsg);
}

extern "C" void moann_docfilter_free(moann_docfilter_c fh) {
    delete (MoannDocFilter*)fh;
}
