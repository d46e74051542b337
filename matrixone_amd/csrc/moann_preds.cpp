/* Columnar JSON-predicate filter evaluation — the host bitset builder the
 * reference's GPU path uses for SQL residual filters
 * (cgo/cuvs/filter.hpp:855-960 eval_filter_bitmap_cpu; predicate wire
 * format filter.hpp:414-423; column meta filter.hpp:597-640). Restated
 * fresh against those semantics:
 *   - preds_json: [{"col":N,"op":"<=","val":5.0}, {"col":-1,"op":"in",
 *     "vals":[...]}, {"col":2,"op":"between","lo":a,"hi":b},
 *     {"col":3,"op":"is_null"}] — conjunctive AND; empty list = all pass.
 *   - cols_json:  [{"name":"a","type":T}] with T in {0 int32, 1 int64,
 *     2 float32, 3 float64, 4 uint64 (varchar hash)}.
 *   - col == -1 is the virtual PK column (__mo_pk_host_id,
 *     pkg/sql/plan/filter_predicate.go): evaluated against the index's
 *     host ids; PKs are non-nullable (filter.hpp:861-875 of
 *     eval_pred_word's virtual-column block).
 *   - SQL three-valued logic: a NULL cell fails every value comparison;
 *     is_null / is_not_null consult only the null bitmap.
 *   - numeric literals keep int64, uint64 and double faces and are
 *     re-narrowed per column type exactly like pred_value_as
 *     (filter.hpp:683-688).
 * Output: LSB-first uint32 words, bit i = row i passes (raft bitset
 * layout, filter.hpp:40-45). */

#include <cstdint>
#include <cstdlib>
#include <cstring>
#include <stdexcept>
#include <string>
#include <vector>

#include "../../include/moann.h"

namespace {

enum class PT { EQ, NE, LT, LE, GT, GE, BETWEEN, IN, IS_NULL, IS_NOT_NULL };

struct PV {
    int64_t i64 = 0;
    uint64_t u64 = 0;
    double f64 = 0.0;
};

struct Pred {
    int64_t col = 0; /* -1 = virtual PK column */
    PT op = PT::EQ;
    PV val, lo, hi;
    std::vector<PV> in_vals;
};

struct Col {
    std::string name;
    int type = 0; /* 0 i32, 1 i64, 2 f32, 3 f64, 4 u64 */
};

/* ---------------- minimal JSON reader ------------------------------------ */

struct Rd {
    const std::string& s;
    size_t i = 0;
    void ws() {
        while (i < s.size() && (s[i] == ' ' || s[i] == '\t' || s[i] == '\n' ||
                                s[i] == '\r'))
            ++i;
    }
    bool eat(char c) {
        ws();
        if (i < s.size() && s[i] == c) {
            ++i;
            return true;
        }
        return false;
    }
    std::string str() {
        ws();
        if (i >= s.size() || s[i] != '"')
            throw std::runtime_error("preds: expected string");
        ++i;
        std::string out;
        while (i < s.size() && s[i] != '"') {
            if (s[i] == '\\' && i + 1 < s.size()) {
                out.push_back(s[i + 1]);
                i += 2;
            } else
                out.push_back(s[i++]);
        }
        if (i >= s.size()) throw std::runtime_error("preds: bad string");
        ++i;
        return out;
    }
    PV num() {
        ws();
        const size_t start = i;
        if (i < s.size() && (s[i] == '-' || s[i] == '+')) ++i;
        bool dig = false, flt = false, dot = false, ex = false;
        while (i < s.size()) {
            const char c = s[i];
            if (c >= '0' && c <= '9') {
                dig = true;
                ++i;
            } else if (c == '.' && !dot && !ex) {
                dot = flt = true;
                ++i;
            } else if ((c == 'e' || c == 'E') && !ex && dig) {
                ex = flt = true;
                ++i;
                if (i < s.size() && (s[i] == '-' || s[i] == '+')) ++i;
            } else
                break;
        }
        if (!dig) throw std::runtime_error("preds: expected number");
        const std::string tok = s.substr(start, i - start);
        PV v;
        if (flt) {
            v.f64 = std::stod(tok);
            v.i64 = (int64_t)v.f64;
            v.u64 = (uint64_t)v.f64;
        } else {
            v.i64 = std::stoll(tok);
            v.u64 = (uint64_t)v.i64;
            v.f64 = (double)v.i64;
        }
        return v;
    }
};

PT op_of(const std::string& s) {
    if (s == "=" || s == "==" || s == "eq") return PT::EQ;
    if (s == "!=" || s == "<>" || s == "ne") return PT::NE;
    if (s == "<" || s == "lt") return PT::LT;
    if (s == "<=" || s == "le") return PT::LE;
    if (s == ">" || s == "gt") return PT::GT;
    if (s == ">=" || s == "ge") return PT::GE;
    if (s == "between") return PT::BETWEEN;
    if (s == "in") return PT::IN;
    if (s == "is_null" || s == "isnull") return PT::IS_NULL;
    if (s == "is_not_null" || s == "isnotnull") return PT::IS_NOT_NULL;
    throw std::runtime_error("preds: unknown op '" + s + "'");
}

std::vector<Pred> parse_preds(const std::string& js) {
    std::vector<Pred> out;
    Rd r{js};
    r.ws();
    if (r.i >= js.size()) return out;
    if (!r.eat('[')) throw std::runtime_error("preds: expected '['");
    while (true) {
        r.ws();
        if (r.eat(']')) break;
        if (!r.eat('{')) throw std::runtime_error("preds: expected '{'");
        Pred p;
        bool has_col = false, has_op = false;
        while (true) {
            r.ws();
            if (r.eat('}')) break;
            const std::string key = r.str();
            if (!r.eat(':')) throw std::runtime_error("preds: expected ':'");
            if (key == "col") {
                p.col = r.num().i64;
                has_col = true;
            } else if (key == "op") {
                p.op = op_of(r.str());
                has_op = true;
            } else if (key == "val")
                p.val = r.num();
            else if (key == "lo")
                p.lo = r.num();
            else if (key == "hi")
                p.hi = r.num();
            else if (key == "vals") {
                if (!r.eat('[')) throw std::runtime_error("preds: vals '['");
                while (true) {
                    r.ws();
                    if (r.eat(']')) break;
                    p.in_vals.push_back(r.num());
                    r.eat(',');
                }
            } else
                throw std::runtime_error("preds: unknown key '" + key + "'");
            r.eat(',');
        }
        if (!has_col || !has_op)
            throw std::runtime_error("preds: predicate needs col and op");
        out.push_back(std::move(p));
        r.eat(',');
    }
    return out;
}

std::vector<Col> parse_cols(const std::string& js) {
    std::vector<Col> out;
    Rd r{js};
    r.ws();
    if (r.i >= js.size()) return out;
    if (!r.eat('[')) throw std::runtime_error("cols: expected '['");
    while (true) {
        r.ws();
        if (r.eat(']')) break;
        if (!r.eat('{')) throw std::runtime_error("cols: expected '{'");
        Col c;
        while (true) {
            r.ws();
            if (r.eat('}')) break;
            const std::string key = r.str();
            if (!r.eat(':')) throw std::runtime_error("cols: expected ':'");
            if (key == "name")
                c.name = r.str();
            else if (key == "type") {
                c.type = (int)r.num().i64;
                if (c.type < 0 || c.type > 4)
                    throw std::runtime_error("cols: type out of range");
            } else
                throw std::runtime_error("cols: unknown key '" + key + "'");
            r.eat(',');
        }
        out.push_back(std::move(c));
        r.eat(',');
    }
    return out;
}

/* ---------------- typed row evaluator ------------------------------------ */

template <typename T>
T as(const PV& v);
template <>
int32_t as<int32_t>(const PV& v) { return (int32_t)v.i64; }
template <>
int64_t as<int64_t>(const PV& v) { return v.i64; }
template <>
uint64_t as<uint64_t>(const PV& v) { return v.u64; }
template <>
float as<float>(const PV& v) { return (float)v.f64; }
template <>
double as<double>(const PV& v) { return v.f64; }

template <typename T>
bool eval_row(const T x, const Pred& p) {
    switch (p.op) {
    case PT::EQ: return x == as<T>(p.val);
    case PT::NE: return x != as<T>(p.val);
    case PT::LT: return x < as<T>(p.val);
    case PT::LE: return x <= as<T>(p.val);
    case PT::GT: return x > as<T>(p.val);
    case PT::GE: return x >= as<T>(p.val);
    case PT::BETWEEN: return x >= as<T>(p.lo) && x <= as<T>(p.hi);
    case PT::IN: {
        for (const auto& v : p.in_vals)
            if (x == as<T>(v)) return true;
        return false;
    }
    default: return false; /* null ops handled before dispatch */
    }
}

bool typed_eval(int type, const void* base, uint64_t row, const Pred& p) {
    switch (type) {
    case 0: return eval_row<int32_t>(((const int32_t*)base)[row], p);
    case 1: return eval_row<int64_t>(((const int64_t*)base)[row], p);
    case 2: return eval_row<float>(((const float*)base)[row], p);
    case 3: return eval_row<double>(((const double*)base)[row], p);
    default: return eval_row<uint64_t>(((const uint64_t*)base)[row], p);
    }
}

inline bool null_at(const uint32_t* nulls, uint64_t row) {
    return nulls && ((nulls[row >> 5] >> (row & 31)) & 1u);
}

void set_err(void* errmsg, const char* what) {
    if (!errmsg) return;
    std::string m = std::string("moann_preds_eval: ") + what;
    char* c = (char*)malloc(m.size() + 1);
    if (c) std::memcpy(c, m.c_str(), m.size() + 1);
    *(char**)errmsg = c;
}

}  // namespace

extern "C" int moann_preds_eval(const char* cols_json, const char* preds_json,
                                const void* const* col_data,
                                const uint32_t* const* col_nulls,
                                const int64_t* host_ids, uint64_t nrows,
                                uint32_t* out_words, void* errmsg) {
    const uint64_t nw = (nrows + 31) / 32;
    try {
        const std::vector<Col> cols = parse_cols(cols_json ? cols_json : "");
        const std::vector<Pred> preds =
            parse_preds(preds_json ? preds_json : "");
        /* empty predicate list: all rows pass (filter.hpp:873-879) */
        std::memset(out_words, 0xFF, nw * 4);
        if (nrows & 31)
            out_words[nw - 1] = (1u << (nrows & 31)) - 1u;
        if (preds.empty()) return 0;
        for (const auto& p : preds) {
            if (p.col != -1 &&
                (p.col < 0 || (uint64_t)p.col >= cols.size()))
                throw std::runtime_error("predicate column out of range");
            if (p.col != -1 && !col_data[p.col])
                throw std::runtime_error("predicate column data missing");
        }
        for (uint64_t row = 0; row < nrows; ++row) {
            bool pass = true;
            for (const auto& p : preds) {
                bool b;
                if (p.col == -1) {
                    /* virtual PK column: non-nullable; absent view =
                     * pass-through (the planner's residual filter stays
                     * authoritative — filter.hpp eval_pred_word) */
                    if (p.op == PT::IS_NULL) b = false;
                    else if (p.op == PT::IS_NOT_NULL) b = true;
                    else if (!host_ids) b = true;
                    else b = eval_row<int64_t>(host_ids[row], p);
                } else if (p.op == PT::IS_NULL) {
                    b = null_at(col_nulls ? col_nulls[p.col] : nullptr, row);
                } else if (p.op == PT::IS_NOT_NULL) {
                    b = !null_at(col_nulls ? col_nulls[p.col] : nullptr, row);
                } else if (null_at(col_nulls ? col_nulls[p.col] : nullptr,
                                   row)) {
                    b = false; /* NULL fails every value comparison */
                } else {
                    b = typed_eval(cols[p.col].type, col_data[p.col], row, p);
                }
                if (!b) {
                    pass = false;
                    break;
                }
            }
            if (!pass) out_words[row >> 5] &= ~(1u << (row & 31));
        }
        return 0;
    } catch (const std::exception& e) {
        std::memset(out_words, 0, nw * 4); /* fail closed */
        set_err(errmsg, e.what());
        return -1;
    }
}
