// Native codec core for the waldb replication stream.
//
// The reference ships a native coordination-client library
// (deps/zookeeper/libzookeeper_mt.a); this build's native piece sits on
// its data plane instead: WAL record framing (u32 len | u32 crc32 |
// payload, big-endian) encode/validate/scan.  These run per replicated
// chunk on every standby and over every segment on crash recovery —
// the tightest CPU loops in the system.
//
// Exposed (module manatee_amd.native._codec):
//   crc32(data) -> int
//   encode_frame(payload) -> bytes
//   parse_frames(data) -> list[(frame_len, payload)]   (raises ValueError)
//   scan_records(data, want_offsets=False)
//       -> (valid_bytes, count, offsets)  — validate a segment prefix

#include <pybind11/pybind11.h>
#include <pybind11/stl.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <vector>

namespace py = pybind11;

namespace {

// CRC-32 (IEEE 802.3, reflected 0xEDB88320) — identical to zlib.crc32.
// Slice-by-8 for ~8 bytes/iteration.
struct Crc32Tables {
    uint32_t t[8][256];
    Crc32Tables() {
        for (uint32_t i = 0; i < 256; i++) {
            uint32_t c = i;
            for (int k = 0; k < 8; k++)
                c = (c & 1) ? 0xEDB88320u ^ (c >> 1) : c >> 1;
            t[0][i] = c;
        }
        for (uint32_t i = 0; i < 256; i++)
            for (int j = 1; j < 8; j++)
                t[j][i] = (t[j - 1][i] >> 8) ^ t[0][t[j - 1][i] & 0xFF];
    }
};

const Crc32Tables kCrc;

uint32_t crc32_update(uint32_t crc, const uint8_t *p, size_t n) {
    crc = ~crc;
    while (n >= 8) {
        crc ^= (uint32_t)p[0] | ((uint32_t)p[1] << 8) |
               ((uint32_t)p[2] << 16) | ((uint32_t)p[3] << 24);
        uint32_t hi = (uint32_t)p[4] | ((uint32_t)p[5] << 8) |
                      ((uint32_t)p[6] << 16) | ((uint32_t)p[7] << 24);
        crc = kCrc.t[7][crc & 0xFF] ^ kCrc.t[6][(crc >> 8) & 0xFF] ^
              kCrc.t[5][(crc >> 16) & 0xFF] ^ kCrc.t[4][crc >> 24] ^
              kCrc.t[3][hi & 0xFF] ^ kCrc.t[2][(hi >> 8) & 0xFF] ^
              kCrc.t[1][(hi >> 16) & 0xFF] ^ kCrc.t[0][hi >> 24];
        p += 8;
        n -= 8;
    }
    while (n--)
        crc = kCrc.t[0][(crc ^ *p++) & 0xFF] ^ (crc >> 8);
    return ~crc;
}

inline uint32_t rd_u32be(const uint8_t *p) {
    return ((uint32_t)p[0] << 24) | ((uint32_t)p[1] << 16) |
           ((uint32_t)p[2] << 8) | (uint32_t)p[3];
}

inline void wr_u32be(uint8_t *p, uint32_t v) {
    p[0] = (uint8_t)(v >> 24);
    p[1] = (uint8_t)(v >> 16);
    p[2] = (uint8_t)(v >> 8);
    p[3] = (uint8_t)v;
}

constexpr size_t kHdr = 8;
constexpr uint32_t kMaxRecord = 64u * 1024 * 1024;

}  // namespace

static uint64_t py_crc32(py::buffer data) {
    py::buffer_info info = data.request();
    return crc32_update(0, static_cast<const uint8_t *>(info.ptr),
                        (size_t)info.size);
}

static py::bytes py_encode_frame(py::buffer payload) {
    py::buffer_info info = payload.request();
    const uint8_t *p = static_cast<const uint8_t *>(info.ptr);
    size_t n = (size_t)info.size;
    if (n > kMaxRecord)
        throw std::invalid_argument("record too large");
    std::vector<uint8_t> out(kHdr + n);
    wr_u32be(out.data(), (uint32_t)n);
    wr_u32be(out.data() + 4, crc32_update(0, p, n));
    std::memcpy(out.data() + kHdr, p, n);
    return py::bytes(reinterpret_cast<const char *>(out.data()), out.size());
}

// parse_frames: whole-chunk validation (replicated chunks are always
// record-aligned).  Mirrors wal.parse_frames exactly.
static py::list py_parse_frames(py::buffer data) {
    py::buffer_info info = data.request();
    const uint8_t *p = static_cast<const uint8_t *>(info.ptr);
    size_t size = (size_t)info.size;
    py::list out;
    size_t pos = 0;
    while (pos + kHdr <= size) {
        uint32_t length = rd_u32be(p + pos);
        uint32_t crc = rd_u32be(p + pos + 4);
        if (length > kMaxRecord || pos + kHdr + length > size)
            throw py::value_error("corrupt replicated WAL frame");
        const uint8_t *payload = p + pos + kHdr;
        if (crc32_update(0, payload, length) != crc)
            throw py::value_error("corrupt replicated WAL frame");
        out.append(py::make_tuple(
            (uint64_t)(kHdr + length),
            py::bytes(reinterpret_cast<const char *>(payload), length)));
        pos += kHdr + length;
    }
    if (pos != size)
        throw py::value_error("non-record-aligned replicated chunk");
    return out;
}

// scan_records: validate a segment prefix (crash recovery).  Returns
// (valid_bytes, record_count, offsets) where offsets is a list of
// (payload_offset, payload_length) pairs when want_offsets, else empty.
// Stops cleanly at the first torn/corrupt frame — the caller truncates.
static py::tuple py_scan_records(py::buffer data, bool want_offsets) {
    py::buffer_info info = data.request();
    const uint8_t *p = static_cast<const uint8_t *>(info.ptr);
    size_t size = (size_t)info.size;
    py::list offsets;
    size_t pos = 0;
    uint64_t count = 0;
    while (pos + kHdr <= size) {
        uint32_t length = rd_u32be(p + pos);
        uint32_t crc = rd_u32be(p + pos + 4);
        if (length > kMaxRecord || pos + kHdr + length > size)
            break;
        if (crc32_update(0, p + pos + kHdr, length) != crc)
            break;
        if (want_offsets)
            offsets.append(py::make_tuple((uint64_t)(pos + kHdr),
                                          (uint64_t)length));
        pos += kHdr + length;
        count++;
    }
    return py::make_tuple((uint64_t)pos, count, offsets);
}

PYBIND11_MODULE(_codec, m) {
    m.doc() = "native WAL frame codec (crc32, framing, segment scan)";
    m.def("crc32", &py_crc32, py::arg("data"));
    m.def("encode_frame", &py_encode_frame, py::arg("payload"));
    m.def("parse_frames", &py_parse_frames, py::arg("data"));
    m.def("scan_records", &py_scan_records, py::arg("data"),
          py::arg("want_offsets") = false);
}
