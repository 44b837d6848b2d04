// Native graph-ingest layer (bigclam._io_native): mmap'd multi-threaded
// edge-list parser.  Replaces the role of the reference's Spark
// GraphLoader.edgeListFile (codes/bigclamv3-7.scala:26) — C1 in SURVEY.md
// §5.8 — with a host-side parser feeding the CSR builder.
//
// Format contract: whitespace-separated "src dst" per line, '#' comments.
// ~10x pandas on large files; threads split the file at newline boundaries.
#include <pybind11/numpy.h>
#include <pybind11/pybind11.h>

#include <fcntl.h>
#include <sys/mman.h>
#include <sys/stat.h>
#include <unistd.h>

#include <cstdint>
#include <cstring>
#include <stdexcept>
#include <string>
#include <thread>
#include <vector>

namespace py = pybind11;

namespace {

struct Span {
  const char* p;
  const char* end;
};

static void parse_span(Span s, std::vector<int64_t>* out) {
  const char* p = s.p;
  const char* end = s.end;
  while (p < end) {
    // skip leading whitespace/newlines
    while (p < end && (*p == ' ' || *p == '\t' || *p == '\r' || *p == '\n'))
      ++p;
    if (p >= end) break;
    if (*p == '#') {  // comment line
      while (p < end && *p != '\n') ++p;
      continue;
    }
    int64_t a = 0, b = 0;
    bool got = false;
    while (p < end && *p >= '0' && *p <= '9') {
      a = a * 10 + (*p - '0');
      ++p;
      got = true;
    }
    if (!got) {  // malformed token; skip line
      while (p < end && *p != '\n') ++p;
      continue;
    }
    while (p < end && (*p == ' ' || *p == '\t')) ++p;
    got = false;
    while (p < end && *p >= '0' && *p <= '9') {
      b = b * 10 + (*p - '0');
      ++p;
      got = true;
    }
    if (!got) {
      while (p < end && *p != '\n') ++p;
      continue;
    }
    out->push_back(a);
    out->push_back(b);
    while (p < end && *p != '\n') ++p;  // drop any trailing columns
  }
}

py::array_t<int64_t> parse_edgelist(const std::string& path, int n_threads) {
  int fd = ::open(path.c_str(), O_RDONLY);
  if (fd < 0) throw std::runtime_error("cannot open " + path);
  struct stat st;
  if (fstat(fd, &st) != 0) {
    ::close(fd);
    throw std::runtime_error("fstat failed: " + path);
  }
  const size_t len = (size_t)st.st_size;
  if (len == 0) {
    ::close(fd);
    return py::array_t<int64_t>(
        std::vector<py::ssize_t>{(py::ssize_t)0, (py::ssize_t)2});
  }
  const char* data =
      (const char*)::mmap(nullptr, len, PROT_READ, MAP_PRIVATE, fd, 0);
  ::close(fd);
  if (data == MAP_FAILED) throw std::runtime_error("mmap failed: " + path);

  if (n_threads <= 0)
    n_threads = (int)std::max(1u, std::thread::hardware_concurrency());
  n_threads = std::min<int>(n_threads, 64);

  // split at newline boundaries
  std::vector<Span> spans;
  const char* cur = data;
  const char* end = data + len;
  const size_t chunk = len / (size_t)n_threads + 1;
  while (cur < end) {
    const char* e = cur + chunk;
    if (e >= end) {
      e = end;
    } else {
      while (e < end && *e != '\n') ++e;
    }
    spans.push_back({cur, e});
    cur = e;
  }

  std::vector<std::vector<int64_t>> parts(spans.size());
  {
    std::vector<std::thread> ts;
    for (size_t i = 0; i < spans.size(); ++i)
      ts.emplace_back(parse_span, spans[i], &parts[i]);
    for (auto& t : ts) t.join();
  }
  ::munmap((void*)data, len);

  size_t total = 0;
  for (auto& v : parts) total += v.size();
  py::array_t<int64_t> out(
      std::vector<py::ssize_t>{(py::ssize_t)(total / 2), (py::ssize_t)2});
  int64_t* o = out.mutable_data();
  size_t off = 0;
  for (auto& v : parts) {
    std::memcpy(o + off, v.data(), v.size() * sizeof(int64_t));
    off += v.size();
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_io_native, m) {
  m.doc() = "bigclam native ingest (mmap multithreaded edge-list parser)";
  m.def("parse_edgelist", &parse_edgelist, py::arg("path"),
        py::arg("n_threads") = 0);
}
