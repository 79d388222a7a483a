// HPACK encode/decode helper for the raw-socket HTTP/2 test clients.
//
// The wire-conformance suites (tests/test_grpcgo_conformance.py,
// tests/test_fastserver_wirefuzz.py) drive the native DevicePlugin server
// with hand-built HTTP/2 frames.  Request-side HPACK is hand-rolled in
// Python (so exotic-but-legal encodings can be produced); this module only
// needs to DECODE the server's response header blocks — which nghttp2
// huffman-encodes — and optionally provide a reference encoder.  It wraps
// libnghttp2's stable nghttp2_hd_* API via the same dlopen pattern the
// fast server uses (nghttp2_abi.h).

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <dlfcn.h>

#include <stdexcept>
#include <string>
#include <utility>
#include <vector>

#include "nghttp2_abi.h"

namespace py = pybind11;

extern "C" {
typedef struct nghttp2_hd_deflater nghttp2_hd_deflater;
typedef struct nghttp2_hd_inflater nghttp2_hd_inflater;
}

enum {
    NGHTTP2_HD_INFLATE_NONE = 0,
    NGHTTP2_HD_INFLATE_FINAL = 0x01,
    NGHTTP2_HD_INFLATE_EMIT = 0x02,
};

struct HdApi {
    int (*deflate_new)(nghttp2_hd_deflater **, size_t);
    void (*deflate_del)(nghttp2_hd_deflater *);
    ssize_t (*deflate_hd)(nghttp2_hd_deflater *, uint8_t *, size_t,
                          const nghttp2_nv *, size_t);
    size_t (*deflate_bound)(nghttp2_hd_deflater *, const nghttp2_nv *, size_t);
    int (*inflate_new)(nghttp2_hd_inflater **);
    void (*inflate_del)(nghttp2_hd_inflater *);
    ssize_t (*inflate_hd2)(nghttp2_hd_inflater *, nghttp2_nv *, int *,
                           const uint8_t *, size_t, int);
    int (*inflate_end_headers)(nghttp2_hd_inflater *);

    static HdApi &get() {
        static HdApi inst;
        return inst;
    }

    HdApi() {
        void *h = dlopen("libnghttp2.so.14", RTLD_NOW | RTLD_GLOBAL);
        if (!h) h = dlopen("libnghttp2.so", RTLD_NOW | RTLD_GLOBAL);
        if (!h) throw std::runtime_error("libnghttp2 not found");
        auto sym = [&](const char *name) {
            void *p = dlsym(h, name);
            if (!p)
                throw std::runtime_error(std::string("missing symbol ") + name);
            return p;
        };
#define LOAD(field, name) field = reinterpret_cast<decltype(field)>(sym(name))
        LOAD(deflate_new, "nghttp2_hd_deflate_new");
        LOAD(deflate_del, "nghttp2_hd_deflate_del");
        LOAD(deflate_hd, "nghttp2_hd_deflate_hd");
        LOAD(deflate_bound, "nghttp2_hd_deflate_bound");
        LOAD(inflate_new, "nghttp2_hd_inflate_new");
        LOAD(inflate_del, "nghttp2_hd_inflate_del");
        LOAD(inflate_hd2, "nghttp2_hd_inflate_hd2");
        LOAD(inflate_end_headers, "nghttp2_hd_inflate_end_headers");
#undef LOAD
    }
};

class HpackDecoder {
  public:
    HpackDecoder() {
        if (HdApi::get().inflate_new(&inf_) != 0)
            throw std::runtime_error("nghttp2_hd_inflate_new failed");
    }
    ~HpackDecoder() { HdApi::get().inflate_del(inf_); }
    HpackDecoder(const HpackDecoder &) = delete;
    HpackDecoder &operator=(const HpackDecoder &) = delete;

    // Decode one complete header block.  The inflater's dynamic table
    // persists across calls, as it must across a connection's header
    // blocks (RFC 7541 §2.2).
    std::vector<std::pair<std::string, std::string>> decode(py::bytes block) {
        auto &api = HdApi::get();
        std::string buf = block;
        const uint8_t *in = reinterpret_cast<const uint8_t *>(buf.data());
        size_t inlen = buf.size();
        std::vector<std::pair<std::string, std::string>> out;
        for (;;) {
            nghttp2_nv nv;
            int flags = 0;
            ssize_t rv = api.inflate_hd2(inf_, &nv, &flags, in, inlen, 1);
            if (rv < 0)
                throw std::runtime_error("hpack decode error " +
                                         std::to_string(rv));
            in += rv;
            inlen -= rv;
            if (flags & NGHTTP2_HD_INFLATE_EMIT) {
                out.emplace_back(
                    std::string(reinterpret_cast<char *>(nv.name), nv.namelen),
                    std::string(reinterpret_cast<char *>(nv.value),
                                nv.valuelen));
            }
            if (flags & NGHTTP2_HD_INFLATE_FINAL) {
                api.inflate_end_headers(inf_);
                break;
            }
            if (inlen == 0 && !(flags & NGHTTP2_HD_INFLATE_EMIT)) {
                // input exhausted without FINAL: truncated block
                throw std::runtime_error("hpack block truncated");
            }
        }
        return out;
    }

  private:
    nghttp2_hd_inflater *inf_;
};

class HpackEncoder {
  public:
    explicit HpackEncoder(size_t table_size = 4096) {
        if (HdApi::get().deflate_new(&def_, table_size) != 0)
            throw std::runtime_error("nghttp2_hd_deflate_new failed");
    }
    ~HpackEncoder() { HdApi::get().deflate_del(def_); }
    HpackEncoder(const HpackEncoder &) = delete;
    HpackEncoder &operator=(const HpackEncoder &) = delete;

    py::bytes encode(
        const std::vector<std::pair<std::string, std::string>> &headers) {
        auto &api = HdApi::get();
        std::vector<nghttp2_nv> nva;
        nva.reserve(headers.size());
        for (auto &h : headers) {
            nghttp2_nv nv;
            nv.name = reinterpret_cast<uint8_t *>(const_cast<char *>(
                h.first.data()));
            nv.namelen = h.first.size();
            nv.value = reinterpret_cast<uint8_t *>(const_cast<char *>(
                h.second.data()));
            nv.valuelen = h.second.size();
            nv.flags = NGHTTP2_NV_FLAG_NONE;
            nva.push_back(nv);
        }
        size_t bound = api.deflate_bound(def_, nva.data(), nva.size());
        std::vector<uint8_t> buf(bound);
        ssize_t rv =
            api.deflate_hd(def_, buf.data(), buf.size(), nva.data(), nva.size());
        if (rv < 0)
            throw std::runtime_error("hpack encode error " +
                                     std::to_string(rv));
        return py::bytes(reinterpret_cast<char *>(buf.data()), rv);
    }

  private:
    nghttp2_hd_deflater *def_;
};

PYBIND11_MODULE(_h2tool, m) {
    m.doc() = "HPACK encode/decode via the system libnghttp2 (tests only)";
    py::class_<HpackDecoder>(m, "HpackDecoder")
        .def(py::init<>())
        .def("decode", &HpackDecoder::decode);
    py::class_<HpackEncoder>(m, "HpackEncoder")
        .def(py::init<size_t>(), py::arg("table_size") = 4096)
        .def("encode", &HpackEncoder::encode);
}
