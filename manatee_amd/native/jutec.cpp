// Native jute (ZooKeeper wire protocol) primitive codec.
//
// The reference ships a prebuilt native ZooKeeper client library
// (deps/zookeeper/libzookeeper_mt.a); this is the
// from-scratch equivalent of that native layer for this build: the
// big-endian primitive writer/reader every jute record is built from
// (ref lib/zookeeperMgr.js uses the pure-JS client on top of the same
// wire format).  manatee_amd/coord/jute.py defines the record layer on
// top and selects this extension when present (MANATEE_PURE_PY=1 forces
// the pure-Python fallback; tests/test_native_jute.py fuzzes byte
// parity between the two).

#include <pybind11/pybind11.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>

namespace py = pybind11;

namespace {

inline void put_be32(std::string &out, int32_t v) {
    uint32_t u = static_cast<uint32_t>(v);
    char b[4] = {static_cast<char>(u >> 24), static_cast<char>(u >> 16),
                 static_cast<char>(u >> 8), static_cast<char>(u)};
    out.append(b, 4);
}

inline void put_be64(std::string &out, int64_t v) {
    uint64_t u = static_cast<uint64_t>(v);
    char b[8] = {static_cast<char>(u >> 56), static_cast<char>(u >> 48),
                 static_cast<char>(u >> 40), static_cast<char>(u >> 32),
                 static_cast<char>(u >> 24), static_cast<char>(u >> 16),
                 static_cast<char>(u >> 8),  static_cast<char>(u)};
    out.append(b, 8);
}

class Writer {
  public:
    Writer() { buf_.reserve(128); }

    Writer &int32(int64_t v) {
        if (v < INT32_MIN || v > INT32_MAX)
            throw py::value_error("int32 out of range");
        put_be32(buf_, static_cast<int32_t>(v));
        return *this;
    }

    Writer &int64(int64_t v) {
        put_be64(buf_, v);
        return *this;
    }

    Writer &boolean(bool v) {
        buf_.push_back(v ? '\x01' : '\x00');
        return *this;
    }

    Writer &buffer(py::object v) {
        if (v.is_none()) {
            put_be32(buf_, -1);
            return *this;
        }
        py::bytes b = py::reinterpret_borrow<py::bytes>(v);
        char *data;
        Py_ssize_t len;
        if (PyBytes_AsStringAndSize(b.ptr(), &data, &len) != 0)
            throw py::error_already_set();
        put_be32(buf_, static_cast<int32_t>(len));
        buf_.append(data, static_cast<size_t>(len));
        return *this;
    }

    Writer &ustring(py::object v) {
        if (v.is_none()) {
            put_be32(buf_, -1);
            return *this;
        }
        std::string s = py::cast<std::string>(v);  // utf-8 encodes py::str
        put_be32(buf_, static_cast<int32_t>(s.size()));
        buf_.append(s);
        return *this;
    }

    Writer &raw(py::bytes v) {
        char *data;
        Py_ssize_t len;
        if (PyBytes_AsStringAndSize(v.ptr(), &data, &len) != 0)
            throw py::error_already_set();
        buf_.append(data, static_cast<size_t>(len));
        return *this;
    }

    py::bytes tobytes() const { return py::bytes(buf_); }

    py::bytes framed() const {
        std::string out;
        out.reserve(buf_.size() + 4);
        put_be32(out, static_cast<int32_t>(buf_.size()));
        out.append(buf_);
        return py::bytes(out);
    }

  private:
    std::string buf_;
};

class Reader {
  public:
    explicit Reader(py::bytes buf) : pos_(0) {
        char *data;
        Py_ssize_t len;
        if (PyBytes_AsStringAndSize(buf.ptr(), &data, &len) != 0)
            throw py::error_already_set();
        buf_.assign(data, static_cast<size_t>(len));
    }

    int64_t int32() {
        need(4);
        const unsigned char *p =
            reinterpret_cast<const unsigned char *>(buf_.data()) + pos_;
        uint32_t u = (static_cast<uint32_t>(p[0]) << 24) |
                     (static_cast<uint32_t>(p[1]) << 16) |
                     (static_cast<uint32_t>(p[2]) << 8) |
                     static_cast<uint32_t>(p[3]);
        pos_ += 4;
        return static_cast<int32_t>(u);
    }

    int64_t int64() {
        need(8);
        const unsigned char *p =
            reinterpret_cast<const unsigned char *>(buf_.data()) + pos_;
        uint64_t u = 0;
        for (int i = 0; i < 8; i++)
            u = (u << 8) | static_cast<uint64_t>(p[i]);
        pos_ += 8;
        return static_cast<int64_t>(u);
    }

    bool boolean() {
        need(1);
        return buf_[pos_++] != '\x00';
    }

    py::object buffer() {
        int64_t n = int32();
        if (n < 0)
            return py::none();
        need(static_cast<size_t>(n));
        py::bytes out(buf_.data() + pos_, static_cast<size_t>(n));
        pos_ += static_cast<size_t>(n);
        return out;
    }

    py::object ustring() {
        int64_t n = int32();
        if (n < 0)
            return py::none();
        need(static_cast<size_t>(n));
        py::str out = py::reinterpret_steal<py::str>(PyUnicode_DecodeUTF8(
            buf_.data() + pos_, static_cast<Py_ssize_t>(n), nullptr));
        if (!out)
            throw py::error_already_set();
        pos_ += static_cast<size_t>(n);
        return out;
    }

    int64_t remaining() const {
        return static_cast<int64_t>(buf_.size() - pos_);
    }

  private:
    void need(size_t n) const {
        if (pos_ + n > buf_.size())
            throw py::value_error("short buffer");
    }

    std::string buf_;
    size_t pos_;
};

}  // namespace

PYBIND11_MODULE(_jutec, m) {
    m.doc() = "native jute primitive codec (big-endian writer/reader)";

    py::class_<Writer>(m, "Writer")
        .def(py::init<>())
        .def("int32", &Writer::int32, py::return_value_policy::reference)
        .def("int64", &Writer::int64, py::return_value_policy::reference)
        .def("boolean", &Writer::boolean,
             py::return_value_policy::reference)
        .def("buffer", &Writer::buffer, py::return_value_policy::reference)
        .def("ustring", &Writer::ustring,
             py::return_value_policy::reference)
        .def("raw", &Writer::raw, py::return_value_policy::reference)
        .def("tobytes", &Writer::tobytes)
        .def("framed", &Writer::framed);

    py::class_<Reader>(m, "Reader")
        .def(py::init<py::bytes>())
        .def("int32", &Reader::int32)
        .def("int64", &Reader::int64)
        .def("boolean", &Reader::boolean)
        .def("buffer", &Reader::buffer)
        .def("ustring", &Reader::ustring)
        .def("remaining", &Reader::remaining);
}
