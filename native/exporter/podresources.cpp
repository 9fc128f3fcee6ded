#include "podresources.h"

#include <sys/socket.h>
#include <sys/un.h>
#include <unistd.h>

#include <cstring>
#include <map>

namespace mi355x {

// ---------------------------------------------------------------------------
// protobuf wire-format walker (proto3, kubelet podresources v1)
//
// message ListPodResourcesResponse { repeated PodResources pod_resources=1; }
// message PodResources { string name=1; string namespace=2;
//                        repeated ContainerResources containers=3; }
// message ContainerResources { string name=1;
//                              repeated ContainerDevices devices=2; ... }
// message ContainerDevices { string resource_name=1;
//                            repeated string device_ids=2; ... }
// Unknown fields are skipped, so newer kubelets parse fine.
// ---------------------------------------------------------------------------

namespace {

struct Cursor {
    const uint8_t* p;
    const uint8_t* end;
    bool ok = true;

    uint64_t varint()
    {
        uint64_t v = 0;
        int shift = 0;
        while (p < end && shift < 64) {
            uint8_t b = *p++;
            v |= (uint64_t)(b & 0x7f) << shift;
            if (!(b & 0x80)) return v;
            shift += 7;
        }
        ok = false;
        return 0;
    }

    bool field(uint32_t* num, uint32_t* wire)
    {
        if (p >= end) return false;
        uint64_t tag = varint();
        if (!ok) return false;
        *num = (uint32_t)(tag >> 3);
        *wire = (uint32_t)(tag & 7);
        return true;
    }

    // returns (ptr,len) for a length-delimited field. The length is an
    // unvalidated varint from the wire: compare LENGTHS, never `p + l`
    // (a huge l would overflow the pointer arithmetic — UB — before the
    // bound check could reject it).
    bool bytes(const uint8_t** data, size_t* len)
    {
        uint64_t l = varint();
        if (!ok || l > (uint64_t)(end - p)) {
            ok = false;
            return false;
        }
        *data = p;
        *len = (size_t)l;
        p += l;
        return true;
    }

    void skip(uint32_t wire)
    {
        switch (wire) {
            case 0: varint(); break;
            case 1:
                if (8 > (size_t)(end - p)) { ok = false; return; }
                p += 8;
                break;
            case 2: {
                const uint8_t* d;
                size_t l;
                bytes(&d, &l);
                break;
            }
            case 5:
                if (4 > (size_t)(end - p)) { ok = false; return; }
                p += 4;
                break;
            default: ok = false;
        }
    }
};

std::string to_str(const uint8_t* d, size_t l) { return std::string((const char*)d, l); }

void parse_container_devices(const uint8_t* d, size_t l, DeviceAllocation* alloc)
{
    Cursor c{d, d + l};
    uint32_t num, wire;
    while (c.ok && c.field(&num, &wire)) {
        if (num == 1 && wire == 2) {
            const uint8_t* s;
            size_t sl;
            if (c.bytes(&s, &sl)) alloc->resource_name = to_str(s, sl);
        } else if (num == 2 && wire == 2) {
            const uint8_t* s;
            size_t sl;
            if (c.bytes(&s, &sl)) alloc->device_ids.push_back(to_str(s, sl));
        } else {
            c.skip(wire);
        }
    }
}

void parse_container(const uint8_t* d, size_t l, const std::string& pod,
                     const std::string& ns, std::vector<DeviceAllocation>* out)
{
    Cursor c{d, d + l};
    uint32_t num, wire;
    std::string cname;
    std::vector<std::pair<const uint8_t*, size_t>> devs;
    while (c.ok && c.field(&num, &wire)) {
        if (num == 1 && wire == 2) {
            const uint8_t* s;
            size_t sl;
            if (c.bytes(&s, &sl)) cname = to_str(s, sl);
        } else if (num == 2 && wire == 2) {
            const uint8_t* s;
            size_t sl;
            if (c.bytes(&s, &sl)) devs.push_back({s, sl});
        } else {
            c.skip(wire);
        }
    }
    for (auto& [dp, dl] : devs) {
        DeviceAllocation a;
        a.pod = pod;
        a.ns = ns;
        a.container = cname;
        parse_container_devices(dp, dl, &a);
        if (!a.device_ids.empty()) out->push_back(std::move(a));
    }
}

void parse_pod(const uint8_t* d, size_t l, std::vector<DeviceAllocation>* out)
{
    Cursor c{d, d + l};
    uint32_t num, wire;
    std::string pod, ns;
    std::vector<std::pair<const uint8_t*, size_t>> containers;
    while (c.ok && c.field(&num, &wire)) {
        if (num == 1 && wire == 2) {
            const uint8_t* s;
            size_t sl;
            if (c.bytes(&s, &sl)) pod = to_str(s, sl);
        } else if (num == 2 && wire == 2) {
            const uint8_t* s;
            size_t sl;
            if (c.bytes(&s, &sl)) ns = to_str(s, sl);
        } else if (num == 3 && wire == 2) {
            const uint8_t* s;
            size_t sl;
            if (c.bytes(&s, &sl)) containers.push_back({s, sl});
        } else {
            c.skip(wire);
        }
    }
    for (auto& [cp, cl] : containers) parse_container(cp, cl, pod, ns, out);
}

} // namespace

namespace wire {

bool parse_list_response(const uint8_t* data, size_t len,
                         std::vector<DeviceAllocation>* out, std::string* err)
{
    Cursor c{data, data + len};
    uint32_t num, wirev;
    while (c.ok && c.field(&num, &wirev)) {
        if (num == 1 && wirev == 2) {
            const uint8_t* s;
            size_t sl;
            if (c.bytes(&s, &sl)) parse_pod(s, sl, out);
        } else {
            c.skip(wirev);
        }
    }
    if (!c.ok) {
        if (err) *err = "malformed ListPodResourcesResponse";
        return false;
    }
    return true;
}

// HPACK (RFC 7541) block: static-table indexed fields where possible,
// literal-without-indexing otherwise, no huffman.
std::vector<uint8_t> build_request_headers(const std::string& authority,
                                           const std::string& path)
{
    std::vector<uint8_t> h;
    auto lit_str = [&](const std::string& s) {
        // 7-bit prefix length, huffman bit clear; values here are < 127
        h.push_back((uint8_t)s.size());
        h.insert(h.end(), s.begin(), s.end());
    };
    auto lit_idx_name = [&](int idx, const std::string& val) {
        // literal without indexing, 4-bit prefixed name index
        if (idx < 15) {
            h.push_back((uint8_t)idx);
        } else {
            h.push_back(0x0f);
            h.push_back((uint8_t)(idx - 15)); // fits one byte for idx<142
        }
        lit_str(val);
    };
    auto lit_new_name = [&](const std::string& name, const std::string& val) {
        h.push_back(0x00);
        lit_str(name);
        lit_str(val);
    };
    h.push_back(0x83);                   // :method POST   (static idx 3)
    h.push_back(0x86);                   // :scheme http   (static idx 6)
    lit_idx_name(4, path);               // :path          (static idx 4 name)
    lit_idx_name(1, authority);          // :authority     (static idx 1 name)
    lit_idx_name(31, "application/grpc");// content-type   (static idx 31 name)
    lit_new_name("te", "trailers");
    return h;
}

} // namespace wire

// ---------------------------------------------------------------------------
// HTTP/2 transport: one unary gRPC call on a unix socket
// ---------------------------------------------------------------------------

namespace {

struct Conn {
    int fd = -1;
    ~Conn()
    {
        if (fd >= 0) ::close(fd);
    }

    bool send_all(const uint8_t* d, size_t l)
    {
        size_t off = 0;
        while (off < l) {
            ssize_t n = ::send(fd, d + off, l - off, MSG_NOSIGNAL);
            if (n <= 0) return false;
            off += (size_t)n;
        }
        return true;
    }
    bool send_all(const std::vector<uint8_t>& v) { return send_all(v.data(), v.size()); }

    bool recv_exact(uint8_t* d, size_t l)
    {
        size_t off = 0;
        while (off < l) {
            ssize_t n = ::recv(fd, d + off, l - off, 0);
            if (n <= 0) return false;
            off += (size_t)n;
        }
        return true;
    }
};

void put_frame_header(std::vector<uint8_t>* out, uint32_t len, uint8_t type,
                      uint8_t flags, uint32_t stream)
{
    out->push_back((len >> 16) & 0xff);
    out->push_back((len >> 8) & 0xff);
    out->push_back(len & 0xff);
    out->push_back(type);
    out->push_back(flags);
    out->push_back((stream >> 24) & 0x7f);
    out->push_back((stream >> 16) & 0xff);
    out->push_back((stream >> 8) & 0xff);
    out->push_back(stream & 0xff);
}

} // namespace

bool list_pod_resources(const std::string& socket_path,
                        std::vector<DeviceAllocation>* out, std::string* err)
{
    Conn conn;
    conn.fd = ::socket(AF_UNIX, SOCK_STREAM, 0);
    if (conn.fd < 0) {
        if (err) *err = "socket() failed";
        return false;
    }
    timeval tv{5, 0};
    setsockopt(conn.fd, SOL_SOCKET, SO_RCVTIMEO, &tv, sizeof(tv));
    setsockopt(conn.fd, SOL_SOCKET, SO_SNDTIMEO, &tv, sizeof(tv));
    sockaddr_un addr;
    std::memset(&addr, 0, sizeof(addr));
    addr.sun_family = AF_UNIX;
    if (socket_path.size() >= sizeof(addr.sun_path)) {
        if (err) *err = "socket path too long";
        return false;
    }
    std::strcpy(addr.sun_path, socket_path.c_str());
    if (::connect(conn.fd, (sockaddr*)&addr, sizeof(addr)) != 0) {
        if (err) *err = "connect(" + socket_path + ") failed";
        return false;
    }

    // --- send: preface, SETTINGS, WINDOW_UPDATE, HEADERS, DATA ---
    std::vector<uint8_t> msg;
    const char* preface = "PRI * HTTP/2.0\r\n\r\nSM\r\n\r\n";
    msg.insert(msg.end(), preface, preface + 24);
    put_frame_header(&msg, 0, 0x4, 0, 0); // empty SETTINGS
    // connection window: +2^30 so any response streams without stalls
    put_frame_header(&msg, 4, 0x8, 0, 0);
    msg.insert(msg.end(), {0x40, 0x00, 0x00, 0x00});

    auto hdrs = wire::build_request_headers("localhost", "/v1.PodResourcesLister/List");
    put_frame_header(&msg, (uint32_t)hdrs.size(), 0x1, 0x4 /*END_HEADERS*/, 1);
    msg.insert(msg.end(), hdrs.begin(), hdrs.end());

    // DATA: gRPC frame of the empty ListPodResourcesRequest
    put_frame_header(&msg, 5, 0x0, 0x1 /*END_STREAM*/, 1);
    msg.insert(msg.end(), {0, 0, 0, 0, 0});
    // stream window bump too (some servers respect only stream windows)
    put_frame_header(&msg, 4, 0x8, 0, 1);
    msg.insert(msg.end(), {0x40, 0x00, 0x00, 0x00});

    if (!conn.send_all(msg)) {
        if (err) *err = "send failed";
        return false;
    }

    // --- receive frames until stream 1 ends ---
    std::vector<uint8_t> grpc_payload;
    bool headers_seen = false;
    bool stream_done = false;
    int frames = 0;
    while (!stream_done && ++frames < 4096) {
        uint8_t fh[9];
        if (!conn.recv_exact(fh, 9)) {
            if (err) *err = "recv frame header failed / timeout";
            return false;
        }
        uint32_t len = (fh[0] << 16) | (fh[1] << 8) | fh[2];
        uint8_t type = fh[3], flags = fh[4];
        uint32_t stream = ((fh[5] & 0x7f) << 24) | (fh[6] << 16) | (fh[7] << 8) | fh[8];
        std::vector<uint8_t> payload(len);
        if (len && !conn.recv_exact(payload.data(), len)) {
            if (err) *err = "recv frame payload failed";
            return false;
        }
        switch (type) {
            case 0x0: // DATA
                if (stream == 1) {
                    grpc_payload.insert(grpc_payload.end(), payload.begin(),
                                        payload.end());
                    if (flags & 0x1) stream_done = true;
                }
                break;
            case 0x1: // HEADERS (response headers, then trailers)
                if (stream == 1) {
                    if (headers_seen && (flags & 0x1)) stream_done = true;
                    if (flags & 0x1) stream_done = true;
                    headers_seen = true;
                }
                break;
            case 0x3: // RST_STREAM
                if (err) *err = "stream reset by server";
                return false;
            case 0x4: // SETTINGS
                if (!(flags & 0x1)) {
                    std::vector<uint8_t> ack;
                    put_frame_header(&ack, 0, 0x4, 0x1, 0);
                    conn.send_all(ack);
                }
                break;
            case 0x6: { // PING
                if (!(flags & 0x1)) {
                    std::vector<uint8_t> pong;
                    put_frame_header(&pong, len, 0x6, 0x1, 0);
                    pong.insert(pong.end(), payload.begin(), payload.end());
                    conn.send_all(pong);
                }
                break;
            }
            case 0x7: // GOAWAY
                if (grpc_payload.empty()) {
                    if (err) *err = "server sent GOAWAY before response";
                    return false;
                }
                stream_done = true;
                break;
            default: break; // WINDOW_UPDATE, PUSH_PROMISE(none), etc.
        }
    }

    if (grpc_payload.size() < 5) {
        if (err) *err = "no gRPC response message (empty DATA)";
        return false;
    }
    // gRPC framing: 1-byte compression flag + 4-byte length
    if (grpc_payload[0] != 0) {
        if (err) *err = "compressed gRPC response unsupported";
        return false;
    }
    uint32_t mlen = (grpc_payload[1] << 24) | (grpc_payload[2] << 16) |
                    (grpc_payload[3] << 8) | grpc_payload[4];
    if (5 + mlen > grpc_payload.size()) {
        if (err) *err = "truncated gRPC message";
        return false;
    }
    return wire::parse_list_response(grpc_payload.data() + 5, mlen, out, err);
}

AttributionMap build_attribution(const std::vector<DeviceAllocation>& allocs,
                                 const std::vector<GpuInfo>& gpus,
                                 const std::string& id_type)
{
    AttributionMap out;
    auto ends_with = [](const std::string& s, const std::string& suf) {
        return s.size() >= suf.size() &&
               s.compare(s.size() - suf.size(), suf.size(), suf) == 0;
    };
    for (const auto& a : allocs) {
        if (!ends_with(a.resource_name, "/gpu")) continue;
        for (const auto& id : a.device_ids) {
            // find the GPU this device id names
            for (const auto& g : gpus) {
                std::string key = attribution_key(g, id_type);
                bool match = (id == key) || (id == g.uuid) || (id == g.drm_render) ||
                             (id == g.pci_bdf) || (id == std::to_string(g.index));
                if (!match && !g.drm_render.empty() &&
                    id.find(g.drm_render) != std::string::npos)
                    match = true;
                if (!match && !g.uuid.empty() && id.find(g.uuid) != std::string::npos)
                    match = true;
                if (match) {
                    out[key] = PodAttribution{a.pod, a.ns, a.container};
                    break;
                }
            }
        }
    }
    return out;
}

} // namespace mi355x
