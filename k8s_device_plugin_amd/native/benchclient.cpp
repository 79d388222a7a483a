// Native gRPC bench client (nghttp2 over UDS, standalone binary).
//
// Measures the DevicePlugin server's Allocate round trip with a
// C-speed client — the Python grpc client in bench.py adds ~100+ µs of
// its own, so this is the latency a Go kubelet actually sees.  Also
// serves as a third independent HTTP/2 implementation exercising the
// server (after grpc C-core and the conformance tests).
//
// Build: g++ -O2 -std=c++17 benchclient.cpp -o benchclient -ldl
// Usage: ./benchclient <socket> <device-id> [iters]

#include <arpa/inet.h>
#include <errno.h>
#include <poll.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <algorithm>
#include <chrono>
#include <cstdio>
#include <cstdlib>
#include <string>
#include <vector>

#include "nghttp2_abi.h"

namespace {

struct Ctx {
    int fd = -1;
    nghttp2_session *session = nullptr;
    std::string req_body;     // gRPC-framed request message
    size_t req_off = 0;
    bool stream_done = false; // current request's stream closed
    std::string resp;
};

ssize_t body_read(nghttp2_session *, int32_t, uint8_t *buf, size_t length,
                  uint32_t *flags, nghttp2_data_source *src, void *) {
    auto *ctx = static_cast<Ctx *>(src->ptr);
    size_t n = std::min(length, ctx->req_body.size() - ctx->req_off);
    memcpy(buf, ctx->req_body.data() + ctx->req_off, n);
    ctx->req_off += n;
    if (ctx->req_off == ctx->req_body.size()) *flags |= NGHTTP2_DATA_FLAG_EOF;
    return (ssize_t)n;
}

int on_data(nghttp2_session *, uint8_t, int32_t, const uint8_t *d, size_t n,
            void *user) {
    static_cast<Ctx *>(user)->resp.append((const char *)d, n);
    return 0;
}

int on_close(nghttp2_session *, int32_t, uint32_t, void *user) {
    static_cast<Ctx *>(user)->stream_done = true;
    return 0;
}

void pump(Ctx &ctx) {
    auto &ng = NgHttp2::get();
    uint8_t buf[65536];
    while (!ctx.stream_done) {
        for (;;) {
            const uint8_t *data;
            ssize_t len = ng.session_mem_send(ctx.session, &data);
            if (len <= 0) break;
            ssize_t off = 0;
            while (off < len) {
                ssize_t n = ::write(ctx.fd, data + off, (size_t)(len - off));
                if (n < 0) { perror("write"); exit(1); }
                off += n;
            }
        }
        if (ctx.stream_done) break;
        pollfd p{ctx.fd, POLLIN, 0};
        if (::poll(&p, 1, 5000) <= 0) { fprintf(stderr, "timeout\n"); exit(1); }
        ssize_t n = ::read(ctx.fd, buf, sizeof(buf));
        if (n <= 0) { fprintf(stderr, "conn closed\n"); exit(1); }
        if (ng.session_mem_recv(ctx.session, buf, (size_t)n) < 0) {
            fprintf(stderr, "h2 error\n");
            exit(1);
        }
    }
}

void put_varint(std::string &out, uint64_t v) {
    while (v >= 0x80) { out.push_back((char)(v | 0x80)); v >>= 7; }
    out.push_back((char)v);
}

std::string allocate_request_bytes(const std::string &device_id) {
    // ContainerAllocateRequest{devices_ids: [id]}
    std::string car;
    car.push_back('\x0a');
    put_varint(car, device_id.size());
    car += device_id;
    // AllocateRequest{container_requests: [car]}
    std::string msg;
    msg.push_back('\x0a');
    put_varint(msg, car.size());
    msg += car;
    // gRPC frame
    std::string framed;
    framed.push_back('\0');
    uint32_t be = htonl((uint32_t)msg.size());
    framed.append((const char *)&be, 4);
    framed += msg;
    return framed;
}

std::string preferred_request_bytes(const std::vector<std::string> &ids,
                                    int size) {
    // ContainerPreferredAllocationRequest{available_deviceIDs: ids,
    //                                     allocation_size: size}
    std::string cr;
    for (auto &id : ids) {
        cr.push_back('\x0a');  // field 1, length-delimited
        put_varint(cr, id.size());
        cr += id;
    }
    cr.push_back('\x18');  // field 3 varint (allocation_size)
    put_varint(cr, (uint64_t)size);
    // PreferredAllocationRequest{container_requests: [cr]}
    std::string msg;
    msg.push_back('\x0a');
    put_varint(msg, cr.size());
    msg += cr;
    std::string framed;
    framed.push_back('\0');
    uint32_t be = htonl((uint32_t)msg.size());
    framed.append((const char *)&be, 4);
    framed += msg;
    return framed;
}

#define NV(n, v) \
    {(uint8_t *)(n), (uint8_t *)(v), sizeof(n) - 1, sizeof(v) - 1, 0}

}  // namespace

int main(int argc, char **argv) {
    if (argc < 3) {
        fprintf(stderr,
                "usage: %s <socket> <device-id> [iters] "
                "[pref_ids_csv] [pref_size]\n", argv[0]);
        return 2;
    }
    std::string sock_path = argv[1], device_id = argv[2];
    int iters = argc > 3 ? atoi(argv[3]) : 2000;
    std::vector<std::string> pref_ids;
    if (argc > 4) {
        std::string csv = argv[4];
        size_t pos = 0;
        while (pos <= csv.size()) {
            size_t c = csv.find(',', pos);
            if (c == std::string::npos) c = csv.size();
            if (c > pos) pref_ids.push_back(csv.substr(pos, c - pos));
            pos = c + 1;
        }
    }
    int pref_size = argc > 5 ? atoi(argv[5]) : 1;

    Ctx ctx;
    ctx.fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
    sockaddr_un addr{};
    addr.sun_family = AF_UNIX;
    strncpy(addr.sun_path, sock_path.c_str(), sizeof(addr.sun_path) - 1);
    if (::connect(ctx.fd, (sockaddr *)&addr, sizeof(addr)) != 0) {
        perror("connect");
        return 1;
    }

    auto &ng = NgHttp2::get();
    nghttp2_session_callbacks *cbs;
    ng.session_callbacks_new(&cbs);
    ng.set_on_data_chunk_recv(cbs, on_data);
    ng.set_on_stream_close(cbs, on_close);
    ng.session_client_new(&ctx.session, cbs, &ctx);
    ng.session_callbacks_del(cbs);
    ng.submit_settings(ctx.session, NGHTTP2_FLAG_NONE, nullptr, 0);

    std::string alloc_req = allocate_request_bytes(device_id);

    auto one_call = [&](const std::string &req, const char *path) {
        ctx.req_body = req;
        ctx.req_off = 0;
        ctx.stream_done = false;
        ctx.resp.clear();
        nghttp2_nv hdrs[] = {
            NV(":method", "POST"),
            NV(":scheme", "http"),
            {(uint8_t *)":path", (uint8_t *)path, 5, strlen(path), 0},
            NV(":authority", "localhost"),
            NV("content-type", "application/grpc"),
            NV("te", "trailers"),
        };
        nghttp2_data_provider prov;
        prov.source.ptr = &ctx;
        prov.read_callback = body_read;
        int32_t sid = ng.submit_request(ctx.session, nullptr, hdrs, 6,
                                        &prov, nullptr);
        if (sid < 0) { fprintf(stderr, "submit failed\n"); exit(1); }
        pump(ctx);
        if (ctx.resp.size() < 6) { fprintf(stderr, "short response\n"); exit(1); }
    };

    auto measure = [&](const std::string &req, const char *path,
                       std::vector<double> &lat_us) {
        for (int i = 0; i < 200; ++i) one_call(req, path);  // warmup
        for (int i = 0; i < iters; ++i) {
            auto t0 = std::chrono::steady_clock::now();
            one_call(req, path);
            auto t1 = std::chrono::steady_clock::now();
            lat_us.push_back(
                std::chrono::duration<double, std::micro>(t1 - t0).count());
        }
        std::sort(lat_us.begin(), lat_us.end());
    };

    std::vector<double> lat_us;
    lat_us.reserve(iters);
    measure(alloc_req, "/v1beta1.DevicePlugin/Allocate", lat_us);

    std::vector<double> pref_us;
    if (!pref_ids.empty()) {
        std::string pref_req = preferred_request_bytes(pref_ids, pref_size);
        pref_us.reserve(iters);
        measure(pref_req, "/v1beta1.DevicePlugin/GetPreferredAllocation",
                pref_us);
    }

    printf("{\"client\": \"native-nghttp2\", \"iters\": %d, "
           "\"allocate_p50_us\": %.1f, \"p90_us\": %.1f, \"p99_us\": %.1f",
           iters, lat_us[lat_us.size() / 2],
           lat_us[(size_t)(lat_us.size() * 0.9)],
           lat_us[(size_t)(lat_us.size() * 0.99)]);
    if (!pref_us.empty()) {
        printf(", \"preferred_p50_us\": %.1f, \"preferred_p99_us\": %.1f, "
               "\"preferred_size\": %d, \"preferred_pool\": %zu",
               pref_us[pref_us.size() / 2],
               pref_us[(size_t)(pref_us.size() * 0.99)], pref_size,
               pref_ids.size());
    }
    printf("}\n");
    ng.session_del(ctx.session);
    ::close(ctx.fd);
    return 0;
}
