// Native DevicePlugin v1beta1 gRPC server (C++ / nghttp2 over a unix socket).
//
// The reference's runtime is a compiled Go daemon; this is the MI355X
// build's native equivalent for the kubelet-facing hot path.  Python owns
// discovery/health/lifecycle and pushes PRE-SERIALIZED state (device list
// bytes, per-device Allocate fragments, allocator tables); this server
// owns the wire: HTTP/2 via the system libnghttp2 (dlopen, nghttp2_abi.h),
// gRPC framing, request parsing and the preferred-allocation search — no
// Python in the request path, so Allocate latency is the kernel's UDS
// round trip plus microseconds of C++.
//
// Protocol surface (identical to the Python grpc server, conformance-tested
// against the Python grpc client in tests/test_fastserver.py):
//   /v1beta1.DevicePlugin/GetDevicePluginOptions   unary, fixed bytes
//   /v1beta1.DevicePlugin/PreStartContainer        unary, empty
//   /v1beta1.DevicePlugin/Allocate                 unary, assembled reply
//   /v1beta1.DevicePlugin/GetPreferredAllocation   unary, native search
//   /v1beta1.DevicePlugin/ListAndWatch             server-streaming + pushes

#include <arpa/inet.h>
#include <errno.h>
#include <fcntl.h>
#include <poll.h>
#include <string.h>
#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <algorithm>
#include <atomic>
#include <chrono>
#include <cstdint>
#include <map>
#include <memory>
#include <mutex>
#include <set>
#include <string>
#include <thread>
#include <unordered_map>
#include <vector>

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include "nghttp2_abi.h"

namespace py = pybind11;

namespace {

// ---------------- protobuf wire helpers ----------------

void put_varint(std::string &out, uint64_t v) {
    while (v >= 0x80) {
        out.push_back((char)(v | 0x80));
        v >>= 7;
    }
    out.push_back((char)v);
}

bool get_varint(const uint8_t *&p, const uint8_t *end, uint64_t &v) {
    v = 0;
    int shift = 0;
    while (p < end && shift < 64) {
        uint8_t b = *p++;
        v |= (uint64_t)(b & 0x7F) << shift;
        if (!(b & 0x80)) return true;
        shift += 7;
    }
    return false;
}

void put_len_delim(std::string &out, int field, const std::string &bytes) {
    put_varint(out, (uint64_t)(field << 3) | 2);
    put_varint(out, bytes.size());
    out += bytes;
}

// Iterate fields of a serialized message; cb(field_no, wire_type, ptr, len_or_varint)
template <typename F>
bool for_each_field(const uint8_t *p, const uint8_t *end, F cb) {
    while (p < end) {
        uint64_t key;
        if (!get_varint(p, end, key)) return false;
        int field = (int)(key >> 3), wt = (int)(key & 7);
        if (wt == 0) {  // varint
            uint64_t v;
            if (!get_varint(p, end, v)) return false;
            cb(field, wt, (const uint8_t *)nullptr, v);
        } else if (wt == 2) {  // len-delim
            uint64_t len;
            if (!get_varint(p, end, len)) return false;
            if ((uint64_t)(end - p) < len) return false;
            cb(field, wt, p, len);
            p += len;
        } else if (wt == 5) {
            if (end - p < 4) return false;
            p += 4;
        } else if (wt == 1) {
            if (end - p < 8) return false;
            p += 8;
        } else {
            return false;
        }
    }
    return true;
}

// ---------------- gRPC framing ----------------

std::string grpc_frame(const std::string &msg) {
    std::string out;
    out.reserve(msg.size() + 5);
    out.push_back('\0');
    uint32_t n = htonl((uint32_t)msg.size());
    out.append((const char *)&n, 4);
    out += msg;
    return out;
}

// ---------------- native preferred-allocation search ----------------
// Mirrors allocator/besteffort.py exactly (same grouping, ordering, seeds,
// BFS with parent-set dedup); cross-validated in tests/test_fastserver.py.

struct AllocState {
    std::unordered_map<std::string, int> node_of_id;
    std::unordered_map<int, std::string> id_of_node;
    // (group sort key = parent id, member node ids ascending)
    std::vector<std::pair<std::string, std::vector<int>>> groups;
    std::unordered_map<uint64_t, int> weights;  // (min<<32|max) -> weight
    bool ready = false;

    // Uniform group-pair weight table: the reference's weight model
    // derives a pair's score from per-GPU facts (same devID, link type,
    // NUMA — device.go:136-158), so every partition pair of the same two
    // physical GPUs scores identically.  When that uniformity actually
    // holds in the loaded topology (verified below, not assumed), the
    // search can score candidates from per-group counts in O(G) per
    // extension instead of O(|subset|) — exact, same totals, same
    // tie-break order.  Any non-uniform pair (e.g. a partition missing a
    // link entry while its siblings have one) disables the fast path.
    bool uniform = false;
    std::vector<std::vector<long>> gw;       // [group][group] pair weight
    std::unordered_map<int, int> group_of_node;

    int weight(int a, int b) const {
        if (a > b) std::swap(a, b);
        auto it = weights.find(((uint64_t)(uint32_t)a << 32) | (uint32_t)b);
        return it == weights.end() ? 0 : it->second;
    }

    void build_uniform_table() {
        size_t G = groups.size();
        group_of_node.clear();
        for (size_t g = 0; g < G; ++g)
            for (int n : groups[g].second) group_of_node[n] = (int)g;
        gw.assign(G, std::vector<long>(G, 0));
        uniform = true;
        for (size_t a = 0; a < G && uniform; ++a) {
            for (size_t b = a; b < G && uniform; ++b) {
                bool first = true;
                long w0 = 0;
                for (int i : groups[a].second) {
                    for (int j : groups[b].second) {
                        if (a == b && i >= j) continue;
                        long w = weight(i, j);
                        if (first) { w0 = w; first = false; }
                        else if (w != w0) { uniform = false; break; }
                    }
                    if (!uniform) break;
                }
                gw[a][b] = gw[b][a] = w0;
            }
        }
    }
};

struct Subset {
    std::vector<int> ids;
    uint64_t parents = 0;  // bitset over filtered-group indices (<=64 groups)
    long weight = 0;
};

bool preferred_alloc(const AllocState &st,
                     const std::vector<std::string> &available,
                     const std::vector<std::string> &required, int size,
                     std::vector<std::string> &out, std::string &err) {
    if (!st.ready) { err = "allocator not initialized"; return false; }
    if (size <= 0) { err = "allocation size must be a positive integer"; return false; }
    if ((int)available.size() < size) { err = "available devices count less than allocation size"; return false; }
    if ((int)required.size() > size) { err = "must-include set larger than allocation size"; return false; }
    if (required.size() > available.size()) { err = "must-include set larger than available set"; return false; }
    if ((int)available.size() == size) { out = available; return true; }
    if ((int)required.size() == size) { out = required; return true; }
    std::set<std::string> avail_set(available.begin(), available.end());
    for (auto &r : required)
        if (!avail_set.count(r)) { err = "must-include devices not all available"; return false; }

    std::set<int> avail_nodes, req_nodes;
    std::vector<int> req_node_list;
    for (auto &a : available) {
        auto it = st.node_of_id.find(a);
        if (it != st.node_of_id.end()) avail_nodes.insert(it->second);
    }
    for (auto &r : required) {
        auto it = st.node_of_id.find(r);
        if (it != st.node_of_id.end()) {
            req_nodes.insert(it->second);
            req_node_list.push_back(it->second);
        }
    }

    // filtered groups, sorted by (len asc, parent key asc)
    std::vector<std::pair<std::string, std::vector<int>>> groups;
    for (auto &g : st.groups) {
        std::vector<int> ids;
        for (int n : g.second)
            if (avail_nodes.count(n) && !req_nodes.count(n)) ids.push_back(n);
        if (!ids.empty()) groups.emplace_back(g.first, std::move(ids));
    }
    // stable: Python's sort is stable, and equal (size, parent) keys must
    // tie-break identically for result parity
    std::stable_sort(groups.begin(), groups.end(),
                     [](auto &a, auto &b) {
                         if (a.second.size() != b.second.size())
                             return a.second.size() < b.second.size();
                         return a.first < b.first;
                     });
    if (groups.size() > 64) { err = "too many device groups"; return false; }

    int new_size = size - (int)req_node_list.size();

    const size_t n_orig = st.groups.size();
    if (st.uniform && n_orig <= 64) {
        // ---- closed-form fast path (exact under verified uniformity) ----
        //
        // BFS invariant: every group added to an incomplete state is
        // consumed FULLY (the per-group loop only stops early when the
        // request completes), so an intermediate state is fully described
        // by (parent set, group add order, total size, weight).  Adding m
        // nodes of group b to a state with per-group counts c[] costs
        //   m * sum_h c[h]*gw[b][h]  +  C(m,2)*gw[b][b]
        // — O(G) per GROUP instead of O(|subset|) per NODE, with no heap
        // per state.  Candidate id lists are reconstructed only for the
        // single winner, preserving the exact generation/insertion order
        // of the general path (seed group ascending, then groups in added
        // order, last group as a prefix, required ids appended last).
        struct FS {
            uint64_t parents = 0;
            long weight = 0;
            int size = 0;
            uint8_t norder = 0;
            uint8_t order[64];            // filtered group idx, add order
            long rowsum[64];              // sum_h c[h]*gw[b][h] per ORIG b
        };
        // map filtered idx -> original group idx
        std::vector<int> orig_of(groups.size());
        {
            std::unordered_map<std::string, int> orig_idx;
            for (size_t i = 0; i < st.groups.size(); ++i)
                orig_idx[st.groups[i].first] = (int)i;
            // parent key may repeat only if dev_ids collide — they don't
            for (size_t i = 0; i < groups.size(); ++i) {
                // find by first member node (parent keys can be "")
                orig_of[i] = st.group_of_node.at(groups[i].second[0]);
            }
        }
        auto add_group = [&](FS &s, size_t fidx, int m) {
            int b = orig_of[fidx];
            s.weight += (long)m * s.rowsum[b] +
                        (long)m * (m - 1) / 2 * st.gw[b][b];
            for (size_t h = 0; h < n_orig; ++h)
                s.rowsum[h] += (long)m * st.gw[b][h];
            s.parents |= 1ull << fidx;
            s.order[s.norder++] = (uint8_t)fidx;
            s.size += m;
        };
        auto add_required = [&](FS &s) {
            for (int rn : req_node_list) {
                int b = st.group_of_node.at(rn);
                s.weight += s.rowsum[b];
                for (size_t h = 0; h < n_orig; ++h)
                    s.rowsum[h] += st.gw[b][h];
            }
        };

        std::vector<FS> queue;
        queue.reserve(groups.size() <= 16 ? (1u << groups.size())
                                          : 4096);
        // parent-set dedup: direct bitmap when 2^G fits, else a set
        std::vector<bool> seen_bm;
        std::set<uint64_t> seen_set;
        const bool use_bm = groups.size() <= 20;
        if (use_bm) seen_bm.assign(1u << groups.size(), false);
        auto seen_test_set = [&](uint64_t p) {
            if (use_bm) {
                if (seen_bm[p]) return true;
                seen_bm[p] = true;
                return false;
            }
            return !seen_set.insert(p).second;
        };
        // track the best final on the fly (first strict minimum, same
        // tie-break as collecting then scanning) — avoids storing finals
        FS best{};
        bool have_best = false;
        auto emit_final = [&](FS s) {
            add_required(s);
            if (!have_best || s.weight < best.weight) {
                best = s;
                have_best = true;
            }
        };
        for (size_t idx = 0; idx < groups.size(); ++idx) {
            FS s{};
            int take = std::min((int)groups[idx].second.size(), new_size);
            add_group(s, idx, take);
            if (s.size == new_size) emit_final(s);
            else { seen_test_set(s.parents); queue.push_back(s); }
        }
        for (size_t qi = 0; qi < queue.size(); ++qi) {
            FS cur = queue[qi];
            if (__builtin_popcountll(cur.parents) == (int)groups.size())
                continue;
            for (size_t idx = 0; idx < groups.size(); ++idx) {
                if (cur.parents & (1ull << idx)) continue;
                FS s = cur;
                int take = std::min((int)groups[idx].second.size(),
                                    new_size - s.size);
                add_group(s, idx, take);
                if (s.size == new_size) emit_final(s);
                else if (!seen_test_set(s.parents)) {
                    queue.push_back(s);
                }
            }
        }
        if (!have_best) { err = "no candidate subset found"; return false; }
        // reconstruct the winner's id list in generation order
        int remaining = new_size;
        for (int oi = 0; oi < best.norder && remaining > 0; ++oi) {
            const auto &g = groups[best.order[oi]].second;
            int take = std::min((int)g.size(), remaining);
            for (int i = 0; i < take; ++i) {
                auto it = st.id_of_node.find(g[i]);
                if (it != st.id_of_node.end()) out.push_back(it->second);
            }
            remaining -= take;
        }
        for (int rn : req_node_list) {
            auto it = st.id_of_node.find(rn);
            if (it != st.id_of_node.end()) out.push_back(it->second);
        }
        return true;
    }

    // ---- general path (non-uniform weights): per-node extension ----
    auto extend = [&](Subset s, int nid, int parent_idx) {
        for (int other : s.ids) s.weight += st.weight(other, nid);
        s.ids.push_back(nid);
        if (parent_idx >= 0) s.parents |= (1ull << parent_idx);
        return s;
    };
    auto finish = [&](Subset s) {
        for (int rn : req_node_list) s = extend(std::move(s), rn, -1);
        return s;
    };

    std::vector<Subset> final_sets, queue;
    std::set<uint64_t> seen;
    for (size_t idx = 0; idx < groups.size(); ++idx) {
        Subset s;
        s.ids.push_back(groups[idx].second[0]);
        s.parents = 1ull << idx;
        if (new_size == 1) { final_sets.push_back(finish(s)); continue; }
        bool fulfilled = false;
        for (size_t i = 1; i < groups[idx].second.size(); ++i) {
            s = extend(std::move(s), groups[idx].second[i], (int)idx);
            if ((int)s.ids.size() == new_size) { fulfilled = true; break; }
        }
        if (fulfilled) final_sets.push_back(finish(s));
        else { seen.insert(s.parents); queue.push_back(std::move(s)); }
    }
    for (size_t qi = 0; qi < queue.size(); ++qi) {
        Subset cur = queue[qi];
        if (__builtin_popcountll(cur.parents) == (int)groups.size()) continue;
        for (size_t idx = 0; idx < groups.size(); ++idx) {
            if (cur.parents & (1ull << idx)) continue;
            Subset s = cur;
            s.parents |= 1ull << idx;
            bool done = false;
            for (int nid : groups[idx].second) {
                s = extend(std::move(s), nid, (int)idx);
                if ((int)s.ids.size() == new_size) {
                    final_sets.push_back(finish(s));
                    done = true;
                    break;
                }
            }
            if (!done && !seen.count(s.parents)) {
                seen.insert(s.parents);
                queue.push_back(std::move(s));
            }
        }
    }
    if (final_sets.empty()) { err = "no candidate subset found"; return false; }
    const Subset *best = &final_sets[0];
    for (auto &s : final_sets)
        if (s.weight < best->weight) best = &s;
    for (int nid : best->ids) {
        auto it = st.id_of_node.find(nid);
        if (it != st.id_of_node.end() && avail_set.count(it->second))
            out.push_back(it->second);
    }
    return true;
}

// ---------------- server ----------------

struct Conn;

struct Stream {
    std::string path;
    std::string req_body;
    // outgoing byte queue for the data provider
    std::string out;
    size_t out_off = 0;
    bool is_listwatch = false;
    bool trailer_sent = false;
    std::string grpc_status = "0";
    std::string grpc_message;
    Conn *conn = nullptr;
};

class Server;

struct Conn {
    int fd = -1;
    nghttp2_session *session = nullptr;
    std::map<int32_t, std::unique_ptr<Stream>> streams;
    std::string wbuf;  // pending bytes the socket couldn't take yet
    Server *srv = nullptr;
    bool dead = false;
};

class Server {
  public:
    explicit Server(std::string socket_path) : path_(std::move(socket_path)) {}
    ~Server() { stop(); }

    // ---- state pushed from Python (all pre-serialized protobuf) ----
    void set_options_response(py::bytes b) {
        std::lock_guard<std::mutex> g(mu_);
        options_ = std::string(b);
    }
    void set_kfd_spec(py::bytes b) {
        std::lock_guard<std::mutex> g(mu_);
        kfd_spec_ = std::string(b);
    }
    void set_device_specs(const std::map<std::string, py::bytes> &specs) {
        std::lock_guard<std::mutex> g(mu_);
        dev_specs_.clear();
        for (auto &kv : specs) dev_specs_[kv.first] = std::string(kv.second);
    }
    void set_list_response(py::bytes b) {
        std::lock_guard<std::mutex> g(mu_);
        list_bytes_ = std::string(b);
    }
    void push_list_update(py::bytes b) {
        {
            std::lock_guard<std::mutex> g(mu_);
            list_bytes_ = std::string(b);
            pending_push_ = true;
        }
        wake();
    }
    void set_prestart_paths(const std::map<std::string, std::string> &paths) {
        std::lock_guard<std::mutex> g(mu_);
        prestart_paths_.clear();
        for (auto &kv : paths) prestart_paths_[kv.first] = kv.second;
    }

    py::dict stats() {
        py::dict d;
        d["allocate_total"] = (uint64_t)n_allocate_.load();
        d["preferred_allocation_total"] = (uint64_t)n_preferred_.load();
        d["list_and_watch_streams_total"] = (uint64_t)n_listwatch_.load();
        d["options_total"] = (uint64_t)n_options_.load();
        d["prestart_total"] = (uint64_t)n_prestart_.load();
        d["unknown_method_total"] = (uint64_t)n_unknown_.load();
        d["list_pushes_total"] = (uint64_t)n_pushes_.load();
        d["connections_total"] = (uint64_t)n_conns_.load();
        // handler-time sums (ns): with the *_total counters these give
        // Prometheus-style average handler latency via rate()/rate()
        d["allocate_handler_ns_total"] = (uint64_t)allocate_ns_.load();
        d["preferred_handler_ns_total"] = (uint64_t)preferred_ns_.load();
        return d;
    }

    void set_allocator_state(
        const std::vector<std::pair<std::string, std::vector<int>>> &groups,
        const std::map<std::string, int> &node_of_id,
        const std::vector<std::tuple<int, int, int>> &weights) {
        std::lock_guard<std::mutex> g(mu_);
        alloc_.groups = groups;
        alloc_.node_of_id.clear();
        alloc_.id_of_node.clear();
        for (auto &kv : node_of_id) {
            alloc_.node_of_id[kv.first] = kv.second;
            alloc_.id_of_node[kv.second] = kv.first;
        }
        alloc_.weights.clear();
        for (auto &t : weights) {
            int a = std::get<0>(t), b = std::get<1>(t);
            if (a > b) std::swap(a, b);
            alloc_.weights[((uint64_t)(uint32_t)a << 32) | (uint32_t)b] =
                std::get<2>(t);
        }
        // weights may legitimately be empty: on a no-links topology
        // (1-kfd-visible box) the policy degrades to uniform weights and
        // every pair lookup scores 0 — the search is still well-defined
        alloc_.ready = !alloc_.groups.empty();
        alloc_.build_uniform_table();
    }

    void start() {
        if (running_.exchange(true)) return;
        ::unlink(path_.c_str());
        listen_fd_ = ::socket(AF_UNIX, SOCK_STREAM | SOCK_NONBLOCK, 0);
        if (listen_fd_ < 0) throw std::runtime_error("socket() failed");
        sockaddr_un addr{};
        addr.sun_family = AF_UNIX;
        if (path_.size() >= sizeof(addr.sun_path))
            throw std::runtime_error("socket path too long");
        strncpy(addr.sun_path, path_.c_str(), sizeof(addr.sun_path) - 1);
        if (::bind(listen_fd_, (sockaddr *)&addr, sizeof(addr)) != 0)
            throw std::runtime_error("bind() failed: " +
                                     std::string(strerror(errno)));
        if (::listen(listen_fd_, 16) != 0)
            throw std::runtime_error("listen() failed");
        if (::pipe2(wake_pipe_, O_NONBLOCK) != 0)
            throw std::runtime_error("pipe2() failed");
        loop_ = std::thread([this] { run_loop(); });
    }

    void stop() {
        if (!running_.exchange(false)) return;
        wake();
        if (loop_.joinable()) loop_.join();
        for (auto &c : conns_) destroy_conn(c.get(), false);
        conns_.clear();
        if (listen_fd_ >= 0) ::close(listen_fd_);
        ::close(wake_pipe_[0]);
        ::close(wake_pipe_[1]);
        listen_fd_ = -1;
        ::unlink(path_.c_str());
    }

  private:
    // ---------------- nghttp2 callbacks ----------------

    static int on_begin_headers(nghttp2_session *, const nghttp2_frame *frame,
                                void *user) {
        auto *conn = static_cast<Conn *>(user);
        if (frame->hd.type == NGHTTP2_FRAME_HEADERS) {
            auto st = std::make_unique<Stream>();
            st->conn = conn;
            conn->streams[frame->hd.stream_id] = std::move(st);
        }
        return 0;
    }

    static int on_header(nghttp2_session *, const nghttp2_frame *frame,
                         const uint8_t *name, size_t namelen,
                         const uint8_t *value, size_t valuelen, uint8_t,
                         void *user) {
        auto *conn = static_cast<Conn *>(user);
        auto it = conn->streams.find(frame->hd.stream_id);
        if (it == conn->streams.end()) return 0;
        if (namelen == 5 && memcmp(name, ":path", 5) == 0)
            it->second->path.assign((const char *)value, valuelen);
        return 0;
    }

    static int on_data_chunk(nghttp2_session *, uint8_t, int32_t stream_id,
                             const uint8_t *data, size_t len, void *user) {
        auto *conn = static_cast<Conn *>(user);
        auto it = conn->streams.find(stream_id);
        if (it != conn->streams.end())
            it->second->req_body.append((const char *)data, len);
        return 0;
    }

    static int on_frame_recv(nghttp2_session *, const nghttp2_frame *frame,
                             void *user) {
        auto *conn = static_cast<Conn *>(user);
        if ((frame->hd.type == NGHTTP2_FRAME_HEADERS ||
             frame->hd.type == NGHTTP2_FRAME_DATA) &&
            (frame->hd.flags & NGHTTP2_FLAG_END_STREAM)) {
            auto it = conn->streams.find(frame->hd.stream_id);
            if (it != conn->streams.end())
                conn->srv->dispatch(conn, frame->hd.stream_id,
                                    it->second.get());
        }
        return 0;
    }

    static int on_stream_close(nghttp2_session *, int32_t stream_id, uint32_t,
                               void *user) {
        auto *conn = static_cast<Conn *>(user);
        conn->streams.erase(stream_id);
        return 0;
    }

    static ssize_t data_read(nghttp2_session *, int32_t stream_id, uint8_t *buf,
                             size_t length, uint32_t *data_flags,
                             nghttp2_data_source *source, void *user) {
        auto *conn = static_cast<Conn *>(user);
        auto *st = static_cast<Stream *>(source->ptr);
        auto &ng = NgHttp2::get();
        size_t avail;
        {
            std::lock_guard<std::mutex> g(conn->srv->mu_);
            avail = st->out.size() - st->out_off;
            if (avail == 0 && st->is_listwatch) return NGHTTP2_ERR_DEFERRED;
            size_t n = std::min(avail, length);
            memcpy(buf, st->out.data() + st->out_off, n);
            st->out_off += n;
            avail -= n;
            if (st->is_listwatch) {
                if (st->out_off == st->out.size()) {
                    st->out.clear();
                    st->out_off = 0;
                }
                // server-streaming: never EOF until shutdown
                return (ssize_t)n;
            }
            if (avail == 0) {
                *data_flags |= NGHTTP2_DATA_FLAG_EOF | NGHTTP2_DATA_FLAG_NO_END_STREAM;
                if (!st->trailer_sent) {
                    st->trailer_sent = true;
                    nghttp2_nv tr[2];
                    static const char kStatus[] = "grpc-status";
                    static const char kMsg[] = "grpc-message";
                    tr[0] = {(uint8_t *)kStatus, (uint8_t *)st->grpc_status.data(),
                             sizeof(kStatus) - 1, st->grpc_status.size(),
                             NGHTTP2_NV_FLAG_NONE};
                    size_t ntr = 1;
                    if (!st->grpc_message.empty()) {
                        tr[1] = {(uint8_t *)kMsg,
                                 (uint8_t *)st->grpc_message.data(),
                                 sizeof(kMsg) - 1, st->grpc_message.size(),
                                 NGHTTP2_NV_FLAG_NONE};
                        ntr = 2;
                    }
                    ng.submit_trailer(conn->session, stream_id, tr, ntr);
                }
            }
            return (ssize_t)n;
        }
    }

    // ---------------- request dispatch ----------------

    void dispatch(Conn *conn, int32_t stream_id, Stream *st) {
        // Validate the gRPC length-prefixed message framing before looking
        // at the path.  A compressed-flag byte other than 0 means the
        // client negotiated a message codec we do not implement
        // (grpc-go only sets it after a grpc-encoding handshake) →
        // UNIMPLEMENTED(12), matching grpc-go's own unsupported-codec
        // status.  A body shorter than the declared length (truncated
        // frame) or a stray partial prefix is a malformed request →
        // INTERNAL(13).  Treating either as an empty request would turn
        // garbage into a bogus success (e.g. an Allocate response with
        // only /dev/kfd).
        std::string msg;
        if (!st->req_body.empty()) {
            if (st->req_body.size() < 5) {
                st->grpc_status = "13";  // INTERNAL: truncated frame prefix
                st->grpc_message = "malformed grpc frame: short prefix";
                submit_unary(conn, stream_id, st, "", false);
                return;
            }
            uint8_t compressed = (uint8_t)st->req_body[0];
            uint32_t len;
            memcpy(&len, st->req_body.data() + 1, 4);
            len = ntohl(len);
            if (compressed != 0) {
                st->grpc_status = "12";  // UNIMPLEMENTED: compression
                st->grpc_message = "grpc message compression not supported";
                submit_unary(conn, stream_id, st, "", false);
                return;
            }
            if (st->req_body.size() < (size_t)5 + len) {
                st->grpc_status = "13";  // INTERNAL: truncated body
                st->grpc_message = "malformed grpc frame: truncated body";
                submit_unary(conn, stream_id, st, "", false);
                return;
            }
            msg = st->req_body.substr(5, len);
        }

        const std::string &p = st->path;
        std::string resp;
        if (p == "/v1beta1.DevicePlugin/GetDevicePluginOptions") {
            n_options_.fetch_add(1, std::memory_order_relaxed);
            std::lock_guard<std::mutex> g(mu_);
            resp = options_;
        } else if (p == "/v1beta1.DevicePlugin/PreStartContainer") {
            n_prestart_.fetch_add(1, std::memory_order_relaxed);
            std::string bad;
            if (!handle_prestart(msg, bad)) {
                st->grpc_status = "9";  // FAILED_PRECONDITION
                st->grpc_message =
                    "device " + bad + " failed the pre-start probe";
                submit_unary(conn, stream_id, st, "", false);
                return;
            }
            resp = "";
        } else if (p == "/v1beta1.DevicePlugin/Allocate") {
            n_allocate_.fetch_add(1, std::memory_order_relaxed);
            auto t0 = std::chrono::steady_clock::now();
            resp = handle_allocate(msg);
            allocate_ns_.fetch_add(
                (uint64_t)std::chrono::duration_cast<std::chrono::nanoseconds>(
                    std::chrono::steady_clock::now() - t0).count(),
                std::memory_order_relaxed);
        } else if (p == "/v1beta1.DevicePlugin/GetPreferredAllocation") {
            n_preferred_.fetch_add(1, std::memory_order_relaxed);
            auto t0 = std::chrono::steady_clock::now();
            std::string err;
            bool ok = handle_preferred(msg, resp, err);
            preferred_ns_.fetch_add(
                (uint64_t)std::chrono::duration_cast<std::chrono::nanoseconds>(
                    std::chrono::steady_clock::now() - t0).count(),
                std::memory_order_relaxed);
            if (!ok) {
                st->grpc_status = "3";  // INVALID_ARGUMENT
                st->grpc_message = err;
                resp.clear();
                submit_unary(conn, stream_id, st, "", /*with_body=*/false);
                return;
            }
        } else if (p == "/v1beta1.DevicePlugin/ListAndWatch") {
            n_listwatch_.fetch_add(1, std::memory_order_relaxed);
            st->is_listwatch = true;
            {
                std::lock_guard<std::mutex> g(mu_);
                st->out = grpc_frame(list_bytes_);
                st->out_off = 0;
            }
            submit_stream_response(conn, stream_id, st);
            return;
        } else {
            n_unknown_.fetch_add(1, std::memory_order_relaxed);
            st->grpc_status = "12";  // UNIMPLEMENTED
            submit_unary(conn, stream_id, st, "", false);
            return;
        }
        submit_unary(conn, stream_id, st, resp, true);
    }

    void submit_headers_common(Conn *conn, int32_t stream_id, Stream *st) {
        auto &ng = NgHttp2::get();
        static const char kS[] = ":status", k200[] = "200";
        static const char kCT[] = "content-type", kGrpc[] = "application/grpc";
        nghttp2_nv hdrs[2] = {
            {(uint8_t *)kS, (uint8_t *)k200, sizeof(kS) - 1, sizeof(k200) - 1,
             NGHTTP2_NV_FLAG_NONE},
            {(uint8_t *)kCT, (uint8_t *)kGrpc, sizeof(kCT) - 1,
             sizeof(kGrpc) - 1, NGHTTP2_NV_FLAG_NONE},
        };
        nghttp2_data_provider prov;
        prov.source.ptr = st;
        prov.read_callback = data_read;
        ng.submit_response(conn->session, stream_id, hdrs, 2, &prov);
    }

    void submit_unary(Conn *conn, int32_t stream_id, Stream *st,
                      const std::string &msg, bool with_body) {
        {
            std::lock_guard<std::mutex> g(mu_);
            st->out = with_body ? grpc_frame(msg) : std::string();
            st->out_off = 0;
        }
        submit_headers_common(conn, stream_id, st);
    }

    void submit_stream_response(Conn *conn, int32_t stream_id, Stream *st) {
        submit_headers_common(conn, stream_id, st);
    }

    // true = every requested device opens; on failure `bad` names it
    bool handle_prestart(const std::string &msg, std::string &bad) {
        std::unordered_map<std::string, std::string> paths;
        {
            std::lock_guard<std::mutex> g(mu_);
            if (prestart_paths_.empty()) return true;  // probe disabled
            paths = prestart_paths_;
        }
        bool ok = true;
        const uint8_t *p = (const uint8_t *)msg.data();
        for_each_field(p, p + msg.size(), [&](int field, int wt,
                                              const uint8_t *data,
                                              uint64_t len) {
            if (!ok || field != 1 || wt != 2) return;
            std::string id((const char *)data, len);
            auto it = paths.find(id);
            if (it == paths.end()) return;
            int fd = ::open(it->second.c_str(), O_RDWR | O_CLOEXEC);
            if (fd < 0) {
                ok = false;
                bad = id;
            } else {
                ::close(fd);
            }
        });
        return ok;
    }

    std::string handle_allocate(const std::string &msg) {
        std::lock_guard<std::mutex> g(mu_);
        std::string resp;
        const uint8_t *p = (const uint8_t *)msg.data();
        for_each_field(p, p + msg.size(), [&](int field, int wt,
                                              const uint8_t *data,
                                              uint64_t len) {
            if (field != 1 || wt != 2) return;
            // one ContainerAllocateRequest
            std::string car = kfd_spec_;
            for_each_field(data, data + len, [&](int f2, int wt2,
                                                 const uint8_t *d2,
                                                 uint64_t l2) {
                if (f2 != 1 || wt2 != 2) return;
                std::string id((const char *)d2, l2);
                auto it = dev_specs_.find(id);
                if (it != dev_specs_.end()) car += it->second;
            });
            put_len_delim(resp, 1, car);
        });
        return resp;
    }

    bool handle_preferred(const std::string &msg, std::string &resp,
                          std::string &err) {
        std::lock_guard<std::mutex> g(mu_);
        bool ok = true;
        const uint8_t *p = (const uint8_t *)msg.data();
        for_each_field(p, p + msg.size(), [&](int field, int wt,
                                              const uint8_t *data,
                                              uint64_t len) {
            if (!ok || field != 1 || wt != 2) return;
            std::vector<std::string> available, required;
            int size = 0;
            for_each_field(data, data + len, [&](int f2, int wt2,
                                                 const uint8_t *d2,
                                                 uint64_t l2) {
                if (f2 == 1 && wt2 == 2)
                    available.emplace_back((const char *)d2, l2);
                else if (f2 == 2 && wt2 == 2)
                    required.emplace_back((const char *)d2, l2);
                else if (f2 == 3 && wt2 == 0)
                    size = (int)l2;
            });
            std::vector<std::string> chosen;
            if (!preferred_alloc(alloc_, available, required, size, chosen,
                                 err)) {
                ok = false;
                return;
            }
            std::string car;
            for (auto &id : chosen) put_len_delim(car, 1, id);
            put_len_delim(resp, 1, car);
        });
        return ok;
    }

    // ---------------- event loop ----------------

    void wake() {
        char c = 1;
        ssize_t rc = ::write(wake_pipe_[1], &c, 1);
        (void)rc;
    }

    // Connection cap: the kubelet needs 1-2 connections; a runaway local
    // client must not exhaust the daemon's fds (the socket dir is
    // root-only, so this is belt-and-braces, not a security boundary).
    static constexpr size_t kMaxConns = 256;

    void accept_conn() {
        for (;;) {
            int fd = ::accept4(listen_fd_, nullptr, nullptr, SOCK_NONBLOCK);
            if (fd < 0) return;
            if (conns_.size() >= kMaxConns) {
                ::close(fd);  // refuse; peer sees ECONNRESET and retries
                continue;
            }
            auto &ng = NgHttp2::get();
            nghttp2_session_callbacks *cbs;
            ng.session_callbacks_new(&cbs);
            ng.set_on_begin_headers(cbs, on_begin_headers);
            ng.set_on_header(cbs, on_header);
            ng.set_on_data_chunk_recv(cbs, on_data_chunk);
            ng.set_on_frame_recv(cbs, on_frame_recv);
            ng.set_on_stream_close(cbs, on_stream_close);
            auto conn = std::make_unique<Conn>();
            conn->fd = fd;
            conn->srv = this;
            ng.session_server_new(&conn->session, cbs, conn.get());
            ng.session_callbacks_del(cbs);
            nghttp2_settings_entry st[1] = {
                {NGHTTP2_SETTINGS_MAX_CONCURRENT_STREAMS, 128}};
            ng.submit_settings(conn->session, NGHTTP2_FLAG_NONE, st, 1);
            n_conns_.fetch_add(1, std::memory_order_relaxed);
            conns_.push_back(std::move(conn));
        }
    }

    void destroy_conn(Conn *c, bool erase) {
        auto &ng = NgHttp2::get();
        if (c->session) ng.session_del(c->session);
        if (c->fd >= 0) ::close(c->fd);
        c->session = nullptr;
        c->fd = -1;
        c->dead = true;
        (void)erase;
    }

    // returns false when the connection died
    bool flush_conn(Conn *c) {
        auto &ng = NgHttp2::get();
        // drain pending wbuf first
        while (!c->wbuf.empty()) {
            ssize_t n = ::write(c->fd, c->wbuf.data(), c->wbuf.size());
            if (n < 0) {
                if (errno == EAGAIN || errno == EWOULDBLOCK) return true;
                return false;
            }
            c->wbuf.erase(0, (size_t)n);
        }
        for (;;) {
            const uint8_t *data = nullptr;
            ssize_t len = ng.session_mem_send(c->session, &data);
            if (len < 0) return false;
            if (len == 0) break;
            ssize_t off = 0;
            while (off < len) {
                ssize_t n = ::write(c->fd, data + off, (size_t)(len - off));
                if (n < 0) {
                    if (errno == EAGAIN || errno == EWOULDBLOCK) {
                        c->wbuf.assign((const char *)data + off,
                                       (size_t)(len - off));
                        return true;
                    }
                    return false;
                }
                off += n;
            }
        }
        return true;
    }

    void run_loop() {
        auto &ng = NgHttp2::get();
        std::vector<pollfd> pfds;
        uint8_t buf[65536];
        while (running_.load()) {
            pfds.clear();
            pfds.push_back({listen_fd_, POLLIN, 0});
            pfds.push_back({wake_pipe_[0], POLLIN, 0});
            for (auto &c : conns_) {
                short ev = POLLIN;
                if (!c->wbuf.empty() || ng.session_want_write(c->session))
                    ev |= POLLOUT;
                pfds.push_back({c->fd, ev, 0});
            }
            int rc = ::poll(pfds.data(), pfds.size(), 500);
            if (rc < 0 && errno != EINTR) break;
            if (!running_.load()) break;

            if (pfds[1].revents & POLLIN) {
                char tmp[64];
                while (::read(wake_pipe_[0], tmp, sizeof(tmp)) > 0) {
                }
                bool do_push;
                std::string bytes;
                {
                    std::lock_guard<std::mutex> g(mu_);
                    do_push = pending_push_;
                    pending_push_ = false;
                    bytes = list_bytes_;
                }
                if (do_push) {
                    n_pushes_.fetch_add(1, std::memory_order_relaxed);
                    std::string framed = grpc_frame(bytes);
                    for (auto &c : conns_) {
                        for (auto &skv : c->streams) {
                            Stream *st = skv.second.get();
                            if (!st->is_listwatch) continue;
                            {
                                std::lock_guard<std::mutex> g(mu_);
                                st->out += framed;
                            }
                            ng.session_resume_data(c->session, skv.first);
                        }
                    }
                }
            }

            if (pfds[0].revents & POLLIN) accept_conn();

            size_t pi = 2;
            for (auto &c : conns_) {
                short rev = pi < pfds.size() ? pfds[pi].revents : 0;
                ++pi;
                if (c->dead) continue;
                if (rev & (POLLERR | POLLHUP)) {
                    destroy_conn(c.get(), true);
                    continue;
                }
                if (rev & POLLIN) {
                    for (;;) {
                        ssize_t n = ::read(c->fd, buf, sizeof(buf));
                        if (n > 0) {
                            if (ng.session_mem_recv(c->session, buf,
                                                    (size_t)n) < 0) {
                                destroy_conn(c.get(), true);
                                break;
                            }
                        } else if (n == 0) {
                            destroy_conn(c.get(), true);
                            break;
                        } else {
                            if (errno != EAGAIN && errno != EWOULDBLOCK)
                                destroy_conn(c.get(), true);
                            break;
                        }
                    }
                }
                if (!c->dead && !flush_conn(c.get()))
                    destroy_conn(c.get(), true);
            }
            conns_.erase(std::remove_if(conns_.begin(), conns_.end(),
                                        [](auto &c) { return c->dead; }),
                         conns_.end());
        }
    }

    std::string path_;
    int listen_fd_ = -1;
    int wake_pipe_[2] = {-1, -1};
    std::thread loop_;
    std::atomic<bool> running_{false};
    std::vector<std::unique_ptr<Conn>> conns_;

    std::atomic<uint64_t> allocate_ns_{0}, preferred_ns_{0};
    std::atomic<uint64_t> n_allocate_{0}, n_preferred_{0}, n_listwatch_{0},
        n_options_{0}, n_prestart_{0}, n_unknown_{0}, n_pushes_{0},
        n_conns_{0};

    std::mutex mu_;
    std::string options_, kfd_spec_, list_bytes_;
    std::unordered_map<std::string, std::string> dev_specs_;
    std::unordered_map<std::string, std::string> prestart_paths_;
    AllocState alloc_;
    bool pending_push_ = false;

    friend struct Conn;
};

}  // namespace

PYBIND11_MODULE(_fastserver, m) {
    m.doc() = "native DevicePlugin v1beta1 gRPC server (nghttp2 over UDS)";
    py::class_<Server>(m, "Server")
        .def(py::init<std::string>())
        .def("set_options_response", &Server::set_options_response)
        .def("set_kfd_spec", &Server::set_kfd_spec)
        .def("set_device_specs", &Server::set_device_specs)
        .def("set_list_response", &Server::set_list_response)
        .def("push_list_update", &Server::push_list_update)
        .def("set_allocator_state", &Server::set_allocator_state)
        .def("stats", &Server::stats)
        .def("set_prestart_paths", &Server::set_prestart_paths)
        .def("start", &Server::start,
             py::call_guard<py::gil_scoped_release>())
        .def("stop", &Server::stop, py::call_guard<py::gil_scoped_release>());
}
