// thrift_compact.h — minimal Apache Thrift *compact protocol* reader, enough
// to parse Parquet file metadata and page headers (the footer format the
// reference's SST writer emits through parquet-rs 53.2.0,
// storage.rs:193-213; format pinned by the Apache Parquet spec — parity via
// pyarrow 25.0.0). Hand-written: no thrift codegen in image.
#pragma once
#include <cstdint>
#include <cstddef>
#include <stdexcept>
#include <string>

namespace hx {

struct ThriftReader {
    const uint8_t* p;
    const uint8_t* end;

    ThriftReader(const uint8_t* buf, size_t len) : p(buf), end(buf + len) {}

    [[noreturn]] void fail(const char* msg) const {
        throw std::runtime_error(std::string("thrift: ") + msg);
    }
    uint8_t byte() {
        if (p >= end) fail("eof");
        return *p++;
    }
    uint64_t varint() {
        uint64_t r = 0;
        int s = 0;
        for (;;) {
            uint8_t b = byte();
            r |= uint64_t(b & 0x7f) << s;
            if (!(b & 0x80)) return r;
            s += 7;
            if (s > 63) fail("varint too long");
        }
    }
    int64_t zigzag() {
        uint64_t u = varint();
        return int64_t(u >> 1) ^ -int64_t(u & 1);
    }
    // compact type codes
    enum {
        T_STOP = 0, T_TRUE = 1, T_FALSE = 2, T_BYTE = 3, T_I16 = 4,
        T_I32 = 5, T_I64 = 6, T_DOUBLE = 7, T_BINARY = 8, T_LIST = 9,
        T_SET = 10, T_MAP = 11, T_STRUCT = 12,
    };

    // Field header inside a struct. Returns false at STOP.
    bool field(int16_t& last_id, int16_t& id, uint8_t& type) {
        uint8_t b = byte();
        if (b == 0) return false;
        type = b & 0x0f;
        uint8_t delta = b >> 4;
        if (delta == 0) {
            id = int16_t(zigzag());
        } else {
            id = int16_t(last_id + delta);
        }
        last_id = id;
        return true;
    }

    void list_header(uint8_t& elem_type, uint32_t& size) {
        uint8_t b = byte();
        elem_type = b & 0x0f;
        size = b >> 4;
        if (size == 15) size = uint32_t(varint());
    }

    std::string binary() {
        uint64_t n = varint();
        if (p + n > end) fail("binary overruns");
        std::string s(reinterpret_cast<const char*>(p), n);
        p += n;
        return s;
    }
    void skip_binary() {
        uint64_t n = varint();
        if (p + n > end) fail("binary overruns");
        p += n;
    }

    void skip(uint8_t type) {
        switch (type) {
            case T_TRUE: case T_FALSE: return;       // value in type nibble
            case T_BYTE: byte(); return;
            case T_I16: case T_I32: case T_I64: zigzag(); return;
            case T_DOUBLE:
                if (p + 8 > end) fail("double overruns");
                p += 8;
                return;
            case T_BINARY: skip_binary(); return;
            case T_LIST: case T_SET: {
                uint8_t et; uint32_t n;
                list_header(et, n);
                for (uint32_t i = 0; i < n; i++) skip_elem(et);
                return;
            }
            case T_MAP: {
                uint8_t b = byte();           // size varint came first? compact map:
                // compact map: varint size, then (if size>0) 1 byte key/val types
                // NOTE: byte() above consumed first varint byte — reparse:
                p--;
                uint64_t n = varint();
                if (n > 0) {
                    uint8_t kv = byte();
                    uint8_t kt = kv >> 4, vt = kv & 0x0f;
                    for (uint64_t i = 0; i < n; i++) { skip_elem(kt); skip_elem(vt); }
                }
                (void)b;
                return;
            }
            case T_STRUCT: {
                int16_t last = 0, id; uint8_t t;
                while (field(last, id, t)) skip(t);
                return;
            }
            default: fail("unknown type in skip");
        }
    }
    void skip_elem(uint8_t type) {
        // list elements: bool is 1 byte (T_TRUE/T_FALSE code as full byte)
        if (type == T_TRUE || type == T_FALSE) { byte(); return; }
        skip(type);
    }
    size_t offset(const uint8_t* base) const { return size_t(p - base); }
};

}  // namespace hx
