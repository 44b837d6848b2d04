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

// ------------------------------------------------------------ CSR builder
//
// Parallel canonical-CSR construction (C2 in SURVEY.md §5.8) for large
// edge arrays: dense ID remap via a presence bitmap, packed (lo<<32|hi)
// canonical keys, bucket-parallel sort + dedupe, atomic-cursor scatter of
// both directions, per-row sort.  Produces EXACTLY the same Graph as the
// numpy path in io/edgelist.py (tested); replaces its 117 s single-core
// sort chain at the 100M-edge R-MAT config.

static int pick_threads(int n_threads) {
  if (n_threads <= 0)
    n_threads = (int)std::max(1u, std::thread::hardware_concurrency());
  return std::min(n_threads, 64);
}

template <class F>
static void pfor(int64_t n, int T, F f) {  // f(begin, end, tid)
  if (n <= 0) return;
  const int64_t chunk = (n + T - 1) / T;
  std::vector<std::thread> ts;
  for (int t = 0; t < T; ++t) {
    const int64_t b = (int64_t)t * chunk;
    const int64_t e = std::min(n, b + chunk);
    if (b >= e) break;
    ts.emplace_back([=] { f(b, e, t); });
  }
  for (auto& th : ts) th.join();
}

py::tuple build_csr(py::array_t<int64_t, py::array::c_style> edges,
                    bool drop_self_loops, int n_threads) {
  auto buf = edges.request();
  if (buf.ndim != 2 || buf.shape[1] != 2)
    throw std::runtime_error("edges must be [M, 2]");
  const int64_t M = buf.shape[0];
  const int64_t* E = (const int64_t*)buf.ptr;
  const int T = pick_threads(n_threads);

  int64_t maxid = -1;
  std::vector<int64_t> raw_ids_v;
  std::vector<uint64_t> dedup;
  std::vector<int64_t> indptr_v;
  std::vector<int32_t> indices_v;
  {
    py::gil_scoped_release nogil;

    // 1. id range
    std::vector<int64_t> tmax(T, -1);
    pfor(2 * M, T, [&](int64_t b, int64_t e, int t) {
      int64_t mx = -1;
      for (int64_t i = b; i < e; ++i) {
        if (E[i] < 0) throw std::runtime_error("negative node id");
        mx = std::max(mx, E[i]);
      }
      tmax[t] = mx;
    });
    for (int t = 0; t < T; ++t) maxid = std::max(maxid, tmax[t]);
    if (maxid >= (int64_t(1) << 31))
      throw std::runtime_error("node ids >= 2^31: use the numpy builder");

    // 2. presence bitmap (benign same-value races) + dense remap
    std::vector<uint8_t> present((size_t)maxid + 1, 0);
    pfor(2 * M, T,
         [&](int64_t b, int64_t e, int) {
           for (int64_t i = b; i < e; ++i) present[(size_t)E[i]] = 1;
         });
    std::vector<int32_t> idmap((size_t)maxid + 1);
    int64_t n = 0;
    for (int64_t v = 0; v <= maxid; ++v) {
      idmap[(size_t)v] = (int32_t)n;
      if (present[(size_t)v]) {
        raw_ids_v.push_back(v);
        ++n;
      }
    }

    // 3. packed canonical keys (UINT64_MAX = dropped self-loop)
    std::vector<uint64_t> keys((size_t)M);
    pfor(M, T, [&](int64_t b, int64_t e, int) {
      for (int64_t i = b; i < e; ++i) {
        const uint32_t a = (uint32_t)idmap[(size_t)E[2 * i]];
        const uint32_t c = (uint32_t)idmap[(size_t)E[2 * i + 1]];
        if (drop_self_loops && a == c) {
          keys[(size_t)i] = UINT64_MAX;
          continue;
        }
        const uint32_t lo = std::min(a, c), hi = std::max(a, c);
        keys[(size_t)i] = ((uint64_t)lo << 32) | hi;
      }
    });

    // 4. bucket-parallel sort by lo (buckets stay globally ordered)
    const int NB = std::max(64, T * 8);
    auto bucket_of = [&](uint64_t k) -> int {
      return (int)(((k >> 32) * (uint64_t)NB) / (uint64_t)n);
    };
    std::vector<std::vector<int64_t>> cnt(T, std::vector<int64_t>(NB + 1, 0));
    pfor(M, T, [&](int64_t b, int64_t e, int t) {
      for (int64_t i = b; i < e; ++i) {
        const uint64_t k = keys[(size_t)i];
        if (k == UINT64_MAX) continue;
        cnt[t][bucket_of(k)]++;
      }
    });
    // per-(bucket, thread) scatter offsets
    std::vector<int64_t> boff(NB + 1, 0);
    {
      std::vector<std::vector<int64_t>> toff(T,
                                             std::vector<int64_t>(NB, 0));
      int64_t run = 0;
      for (int bkt = 0; bkt < NB; ++bkt) {
        boff[bkt] = run;
        for (int t = 0; t < T; ++t) {
          toff[t][bkt] = run;
          run += cnt[t][bkt];
        }
      }
      boff[NB] = run;
      std::vector<uint64_t> sorted((size_t)run);
      const int64_t chunk = (M + T - 1) / T;
      pfor(M, T, [&](int64_t b, int64_t e, int t) {
        auto off = toff[t];  // copy: private cursors
        for (int64_t i = b; i < e; ++i) {
          const uint64_t k = keys[(size_t)i];
          if (k == UINT64_MAX) continue;
          sorted[(size_t)off[bucket_of(k)]++] = k;
        }
      });
      (void)chunk;
      keys.clear();
      keys.shrink_to_fit();
      // sort + unique each bucket in parallel
      std::vector<int64_t> ucnt(NB, 0);
      pfor(NB, T, [&](int64_t b, int64_t e, int) {
        for (int64_t bkt = b; bkt < e; ++bkt) {
          auto* s = sorted.data() + boff[bkt];
          auto* se = sorted.data() + boff[bkt + 1];
          std::sort(s, se);
          ucnt[bkt] = std::unique(s, se) - s;
        }
      });
      std::vector<int64_t> uoff(NB + 1, 0);
      for (int bkt = 0; bkt < NB; ++bkt) uoff[bkt + 1] = uoff[bkt] + ucnt[bkt];
      dedup.resize((size_t)uoff[NB]);
      pfor(NB, T, [&](int64_t b, int64_t e, int) {
        for (int64_t bkt = b; bkt < e; ++bkt)
          std::memcpy(dedup.data() + uoff[bkt], sorted.data() + boff[bkt],
                      (size_t)ucnt[bkt] * sizeof(uint64_t));
      });
    }
    const int64_t Mu = (int64_t)dedup.size();

    // 5. degree counts (atomic) + indptr
    std::vector<int32_t> deg((size_t)n, 0);
    pfor(Mu, T, [&](int64_t b, int64_t e, int) {
      for (int64_t i = b; i < e; ++i) {
        const uint32_t lo = (uint32_t)(dedup[(size_t)i] >> 32);
        const uint32_t hi = (uint32_t)dedup[(size_t)i];
        __atomic_fetch_add(&deg[lo], 1, __ATOMIC_RELAXED);
        __atomic_fetch_add(&deg[hi], 1, __ATOMIC_RELAXED);
      }
    });
    indptr_v.assign((size_t)n + 1, 0);
    for (int64_t u = 0; u < n; ++u) indptr_v[(size_t)u + 1] = indptr_v[(size_t)u] + deg[(size_t)u];

    // 6. scatter both directions with atomic cursors, then sort rows
    indices_v.resize((size_t)indptr_v[(size_t)n]);
    std::vector<int64_t> cursor(indptr_v.begin(), indptr_v.end() - 1);
    pfor(Mu, T, [&](int64_t b, int64_t e, int) {
      for (int64_t i = b; i < e; ++i) {
        const uint32_t lo = (uint32_t)(dedup[(size_t)i] >> 32);
        const uint32_t hi = (uint32_t)dedup[(size_t)i];
        const int64_t p1 = __atomic_fetch_add(&cursor[lo], 1, __ATOMIC_RELAXED);
        indices_v[(size_t)p1] = (int32_t)hi;
        const int64_t p2 = __atomic_fetch_add(&cursor[hi], 1, __ATOMIC_RELAXED);
        indices_v[(size_t)p2] = (int32_t)lo;
      }
    });
    pfor(n, T, [&](int64_t b, int64_t e, int) {
      for (int64_t u = b; u < e; ++u)
        std::sort(indices_v.data() + indptr_v[(size_t)u],
                  indices_v.data() + indptr_v[(size_t)u + 1]);
    });
  }

  const int64_t n = (int64_t)raw_ids_v.size();
  py::array_t<int64_t> indptr(std::vector<py::ssize_t>{(py::ssize_t)n + 1});
  std::memcpy(indptr.mutable_data(), indptr_v.data(),
              ((size_t)n + 1) * sizeof(int64_t));
  py::array_t<int32_t> indices(
      std::vector<py::ssize_t>{(py::ssize_t)indices_v.size()});
  std::memcpy(indices.mutable_data(), indices_v.data(),
              indices_v.size() * sizeof(int32_t));
  py::array_t<int64_t> raw_ids(std::vector<py::ssize_t>{(py::ssize_t)n});
  std::memcpy(raw_ids.mutable_data(), raw_ids_v.data(),
              (size_t)n * sizeof(int64_t));
  return py::make_tuple(indptr, indices, raw_ids);
}

// ------------------------------------------------------------ R-MAT gen
//
// Counter-based parallel R-MAT edge sampling: every edge's random stream
// is keyed by (seed, edge index) via splitmix64, so the output is
// deterministic and independent of the thread count.  ID scrambling uses
// a seed-derived bijection on [0, 2^scale) (odd multiply + xorshift
// rounds — both invertible mod 2^scale) instead of a materialized
// permutation.

static inline uint64_t splitmix64(uint64_t& s) {
  uint64_t z = (s += 0x9E3779B97F4A7C15ULL);
  z = (z ^ (z >> 30)) * 0xBF58476D1CE4E5B9ULL;
  z = (z ^ (z >> 27)) * 0x94D049BB133111EBULL;
  return z ^ (z >> 31);
}

py::array_t<int64_t> rmat_edges_native(int scale, double edge_factor,
                                       double a, double b, double c,
                                       uint64_t seed, int n_threads,
                                       bool scramble_ids) {
  if (scale < 1 || scale > 30) throw std::runtime_error("bad scale");
  const int64_t n = int64_t(1) << scale;
  const int64_t m = (int64_t)(edge_factor * (double)n);
  const int T = pick_threads(n_threads);
  py::array_t<int64_t> out(
      std::vector<py::ssize_t>{(py::ssize_t)m, (py::ssize_t)2});
  int64_t* O = out.mutable_data();
  const uint64_t mask = (uint64_t)n - 1;
  // seed-derived bijection constants (odd multipliers)
  uint64_t ss = seed ^ 0xD1B54A32D192ED03ULL;
  const uint64_t m1 = splitmix64(ss) | 1ULL;
  const uint64_t m2 = splitmix64(ss) | 1ULL;
  const int sh = std::max(1, scale / 2);
  auto scramble = [&](uint64_t x) -> uint64_t {
    x = (x * m1) & mask;
    x ^= x >> sh;
    x = (x * m2) & mask;
    x ^= x >> sh;
    return x & mask;
  };
  const double ab = a + b;
  const double a_norm = a / ab;
  const double c_norm = c / (1.0 - ab);
  {
    py::gil_scoped_release nogil;
    pfor(m, T, [&](int64_t lo, int64_t hi, int) {
      for (int64_t i = lo; i < hi; ++i) {
        uint64_t st = seed * 0x9E3779B97F4A7C15ULL + (uint64_t)i * 2654435761ULL;
        (void)splitmix64(st);  // decorrelate the key
        uint64_t src = 0, dst = 0;
        for (int l = 0; l < scale; ++l) {
          const uint64_t r = splitmix64(st);
          const double r1 = (double)(r >> 40) * (1.0 / 16777216.0);
          const double r2 =
              (double)(r & 0xFFFFFFULL) * (1.0 / 16777216.0);
          const int sbit = r1 > ab;
          const int dbit = sbit ? (r2 > c_norm) : (r2 > a_norm);
          src = (src << 1) | (uint64_t)sbit;
          dst = (dst << 1) | (uint64_t)dbit;
        }
        O[2 * i] = (int64_t)(scramble_ids ? scramble(src) : src);
        O[2 * i + 1] = (int64_t)(scramble_ids ? scramble(dst) : dst);
      }
    });
  }
  return out;
}

}  // namespace

PYBIND11_MODULE(_io_native, m) {
  m.doc() = "bigclam native ingest (mmap multithreaded edge-list parser)";
  m.def("parse_edgelist", &parse_edgelist, py::arg("path"),
        py::arg("n_threads") = 0);
  m.def("build_csr", &build_csr, py::arg("edges"),
        py::arg("drop_self_loops") = true, py::arg("n_threads") = 0,
        "parallel canonical CSR: (indptr, indices, raw_ids)");
  m.def("rmat_edges", &rmat_edges_native, py::arg("scale"),
        py::arg("edge_factor"), py::arg("a") = 0.57, py::arg("b") = 0.19,
        py::arg("c") = 0.19, py::arg("seed") = 0, py::arg("n_threads") = 0,
        py::arg("scramble_ids") = true,
        "counter-based parallel R-MAT edge pairs [m, 2]");
}
