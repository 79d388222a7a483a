// Minimal libnghttp2 ABI declarations for dlopen use.
//
// The build image ships libnghttp2.so.14 (v1.43) but no development
// headers, so the handful of entry points the fast server needs are
// declared here from the documented, long-stable public API (all exist
// since nghttp2 1.0; struct layouts below are part of the installed .so's
// ABI).  Resolved at runtime via dlopen/dlsym — no link-time dependency.

#pragma once

#include <dlfcn.h>
#include <stddef.h>
#include <stdint.h>
#include <sys/types.h>

#include <stdexcept>
#include <string>

extern "C" {

typedef struct nghttp2_session nghttp2_session;
typedef struct nghttp2_session_callbacks nghttp2_session_callbacks;
typedef struct nghttp2_option nghttp2_option;

typedef struct {
    uint8_t *name;
    uint8_t *value;
    size_t namelen;
    size_t valuelen;
    uint8_t flags;
} nghttp2_nv;

typedef struct {
    size_t length;
    int32_t stream_id;
    uint8_t type;
    uint8_t flags;
    uint8_t reserved;
} nghttp2_frame_hd;

// We only ever read the frame header, which is the first member of every
// frame struct in the nghttp2_frame union.
typedef struct {
    nghttp2_frame_hd hd;
} nghttp2_frame;

typedef struct {
    int32_t settings_id;
    uint32_t value;
} nghttp2_settings_entry;

typedef union {
    int fd;
    void *ptr;
} nghttp2_data_source;

typedef ssize_t (*nghttp2_data_source_read_callback)(
    nghttp2_session *session, int32_t stream_id, uint8_t *buf, size_t length,
    uint32_t *data_flags, nghttp2_data_source *source, void *user_data);

typedef struct {
    nghttp2_data_source source;
    nghttp2_data_source_read_callback read_callback;
} nghttp2_data_provider;

typedef int (*nghttp2_on_frame_recv_callback)(nghttp2_session *,
                                              const nghttp2_frame *, void *);
typedef int (*nghttp2_on_data_chunk_recv_callback)(nghttp2_session *, uint8_t,
                                                   int32_t, const uint8_t *,
                                                   size_t, void *);
typedef int (*nghttp2_on_header_callback)(nghttp2_session *,
                                          const nghttp2_frame *,
                                          const uint8_t *, size_t,
                                          const uint8_t *, size_t, uint8_t,
                                          void *);
typedef int (*nghttp2_on_begin_headers_callback)(nghttp2_session *,
                                                 const nghttp2_frame *, void *);
typedef int (*nghttp2_on_stream_close_callback)(nghttp2_session *, int32_t,
                                                uint32_t, void *);

}  // extern "C"

// error codes / flags (nghttp2.h, stable ABI constants)
enum {
    NGHTTP2_ERR_WOULDBLOCK = -504,
    NGHTTP2_ERR_EOF = -507,
    NGHTTP2_ERR_DEFERRED = -508,
    NGHTTP2_ERR_TEMPORAL_CALLBACK_FAILURE = -521,
    NGHTTP2_ERR_CALLBACK_FAILURE = -902,
};
enum {
    NGHTTP2_FLAG_NONE = 0,
    NGHTTP2_FLAG_END_STREAM = 0x01,
};
enum {
    NGHTTP2_DATA_FLAG_NONE = 0,
    NGHTTP2_DATA_FLAG_EOF = 0x01,
    NGHTTP2_DATA_FLAG_NO_END_STREAM = 0x02,
};
enum {
    NGHTTP2_FRAME_DATA = 0,
    NGHTTP2_FRAME_HEADERS = 1,
};
enum {
    NGHTTP2_NV_FLAG_NONE = 0,
};
enum {
    NGHTTP2_SETTINGS_MAX_CONCURRENT_STREAMS = 3,
};

// Resolved entry points.
struct NgHttp2 {
    int (*session_callbacks_new)(nghttp2_session_callbacks **);
    void (*session_callbacks_del)(nghttp2_session_callbacks *);
    void (*set_on_frame_recv)(nghttp2_session_callbacks *,
                              nghttp2_on_frame_recv_callback);
    void (*set_on_data_chunk_recv)(nghttp2_session_callbacks *,
                                   nghttp2_on_data_chunk_recv_callback);
    void (*set_on_header)(nghttp2_session_callbacks *,
                          nghttp2_on_header_callback);
    void (*set_on_begin_headers)(nghttp2_session_callbacks *,
                                 nghttp2_on_begin_headers_callback);
    void (*set_on_stream_close)(nghttp2_session_callbacks *,
                                nghttp2_on_stream_close_callback);
    int (*session_server_new)(nghttp2_session **,
                              const nghttp2_session_callbacks *, void *);
    void (*session_del)(nghttp2_session *);
    ssize_t (*session_mem_recv)(nghttp2_session *, const uint8_t *, size_t);
    ssize_t (*session_mem_send)(nghttp2_session *, const uint8_t **);
    int (*session_want_write)(nghttp2_session *);
    int (*submit_settings)(nghttp2_session *, uint8_t,
                           const nghttp2_settings_entry *, size_t);
    int (*submit_response)(nghttp2_session *, int32_t, const nghttp2_nv *,
                           size_t, const nghttp2_data_provider *);
    int (*submit_trailer)(nghttp2_session *, int32_t, const nghttp2_nv *,
                          size_t);
    int (*session_resume_data)(nghttp2_session *, int32_t);
    // client side (used by the native bench client)
    int (*session_client_new)(nghttp2_session **,
                              const nghttp2_session_callbacks *, void *);
    int32_t (*submit_request)(nghttp2_session *, const void *priority_spec,
                              const nghttp2_nv *, size_t,
                              const nghttp2_data_provider *, void *);

    static NgHttp2 &get() {
        static NgHttp2 inst;
        return inst;
    }

    NgHttp2() {
        void *h = dlopen("libnghttp2.so.14", RTLD_NOW | RTLD_GLOBAL);
        if (!h) h = dlopen("libnghttp2.so", RTLD_NOW | RTLD_GLOBAL);
        if (!h)
            throw std::runtime_error("libnghttp2.so.14 not found (fast server "
                                     "requires the system nghttp2 runtime)");
        auto sym = [&](const char *name) {
            void *p = dlsym(h, name);
            if (!p)
                throw std::runtime_error(std::string("missing nghttp2 symbol ") +
                                         name);
            return p;
        };
#define LOAD(field, name) field = reinterpret_cast<decltype(field)>(sym(name))
        LOAD(session_callbacks_new, "nghttp2_session_callbacks_new");
        LOAD(session_callbacks_del, "nghttp2_session_callbacks_del");
        LOAD(set_on_frame_recv,
             "nghttp2_session_callbacks_set_on_frame_recv_callback");
        LOAD(set_on_data_chunk_recv,
             "nghttp2_session_callbacks_set_on_data_chunk_recv_callback");
        LOAD(set_on_header, "nghttp2_session_callbacks_set_on_header_callback");
        LOAD(set_on_begin_headers,
             "nghttp2_session_callbacks_set_on_begin_headers_callback");
        LOAD(set_on_stream_close,
             "nghttp2_session_callbacks_set_on_stream_close_callback");
        LOAD(session_server_new, "nghttp2_session_server_new");
        LOAD(session_del, "nghttp2_session_del");
        LOAD(session_mem_recv, "nghttp2_session_mem_recv");
        LOAD(session_mem_send, "nghttp2_session_mem_send");
        LOAD(session_want_write, "nghttp2_session_want_write");
        LOAD(submit_settings, "nghttp2_submit_settings");
        LOAD(submit_response, "nghttp2_submit_response");
        LOAD(submit_trailer, "nghttp2_submit_trailer");
        LOAD(session_resume_data, "nghttp2_session_resume_data");
        LOAD(session_client_new, "nghttp2_session_client_new");
        LOAD(submit_request, "nghttp2_submit_request");
#undef LOAD
    }
};
