// Native .c2v line parser — the C++ core of the data runtime.
//
// The reference's input pipeline is tf.data's C++ CsvDataset + string_split +
// static hash tables (path_context_reader.py:119-228); this is the
// MI355X-native equivalent: one C++ parser object holding the three
// string->index hash maps, parsing batches of lines into int32/float32
// torch tensors with a std::thread pool. Semantics identical to the Python
// reader (code2vec_amd/data/reader.py), which remains the fallback and the
// test oracle.

#include <torch/extension.h>

#include <sys/syscall.h>
#include <unistd.h>

#include <atomic>
#include <cstring>
#include <string>
#include <string_view>
#include <thread>
#include <unordered_map>
#include <vector>

namespace {

// Best-effort NUMA interleave of an already-populated buffer (raw mbind —
// no libnuma in the image). The vocab tables are built single-threaded, so
// first-touch puts every page on the builder's node and parser threads on
// the other socket pay remote-DRAM latency on each (random) lookup;
// MPOL_MF_MOVE migrates the pages round-robin across the allowed nodes.
// Silently a no-op on single-node machines or where the syscall is denied.
void interleave_pages(void* p, size_t n) {
  if (n < (size_t)1 << 20) return;                 // not worth it under 1 MB
  const long page = sysconf(_SC_PAGESIZE);
  uintptr_t a = ((uintptr_t)p + page - 1) & ~(uintptr_t)(page - 1);
  uintptr_t end = ((uintptr_t)p + n) & ~(uintptr_t)(page - 1);
  if (end <= a) return;
  unsigned long mems[16] = {0};
  // MPOL_F_MEMS_ALLOWED(4) fills the mask of nodes this thread may use
  if (syscall(SYS_get_mempolicy, nullptr, mems, sizeof(mems) * 8, nullptr,
              4L) != 0)
    return;
  // MPOL_INTERLEAVE(3), MPOL_MF_MOVE(2)
  syscall(SYS_mbind, (void*)a, (unsigned long)(end - a), 3L, mems,
          (unsigned long)(sizeof(mems) * 8), 2UL);
}

// allocation-free open-addressing string->int map. Short keys (<= 19 B —
// the overwhelming majority of code tokens and hash-int path strings) live
// INLINE in the 32-byte slot, so a hit costs ONE cache-line miss; the
// original out-of-line interned-key compare was a second dependent DRAM
// miss per lookup and capped parsing at ~31K rows/core-s on a 1.3M-entry
// vocabulary (lookups are effectively random DRAM reads at these sizes).
struct StrMap {
  struct Slot {                 // 32 bytes
    uint64_t h = 0;
    int32_t val = 0;
    uint8_t len = 0xFF;         // 0xFF empty; 0xFE long key (see long_keys)
    char inl[19];               // short key bytes, or u32 long_keys index
  };
  static_assert(sizeof(Slot) == 32, "slot must stay one half-line");
  std::vector<std::string> long_keys;
  std::vector<Slot> slots;
  uint64_t mask = 0;

  static uint64_t hash(std::string_view s) {
    uint64_t h = 1469598103934665603ull;  // FNV-1a 64
    for (unsigned char c : s) {
      h ^= c;
      h *= 1099511628211ull;
    }
    return h | 1;  // nonzero
  }

  void build(const std::unordered_map<std::string, int>& m) {
    size_t cap = 16;
    while (cap < m.size() * 2) cap <<= 1;
    slots.assign(cap, Slot{});
    mask = cap - 1;
    for (const auto& [k, v] : m) {
      const uint64_t h = hash(k);
      uint64_t s = h & mask;
      while (slots[s].len != 0xFF) s = (s + 1) & mask;
      Slot& sl = slots[s];
      sl.h = h;
      sl.val = v;
      if (k.size() <= sizeof(sl.inl)) {
        sl.len = (uint8_t)k.size();
        std::memcpy(sl.inl, k.data(), k.size());
      } else {
        sl.len = 0xFE;
        const uint32_t idx = (uint32_t)long_keys.size();
        std::memcpy(sl.inl, &idx, sizeof(idx));
        long_keys.push_back(k);
      }
    }
    const char* e = getenv("C2V_NUMA_INTERLEAVE");
    if (!(e && e[0] == '0'))
      interleave_pages(slots.data(), slots.size() * sizeof(Slot));
  }

  int look(std::string_view s, int dflt) const {
    const uint64_t h = hash(s);
    uint64_t i = h & mask;
    for (;;) {
      const Slot& sl = slots[i];
      if (sl.len == 0xFF) return dflt;
      if (sl.h == h) {
        if (sl.len == 0xFE) {
          uint32_t idx;
          std::memcpy(&idx, sl.inl, sizeof(idx));
          if (long_keys[idx] == s) return sl.val;
        } else if (sl.len == s.size() &&
                   std::memcmp(sl.inl, s.data(), sl.len) == 0) {
          return sl.val;
        }
      }
      i = (i + 1) & mask;
    }
  }
};

}  // namespace

namespace {

struct Parser {
  StrMap tok, path, tgt;
  int tok_pad = 0, tok_oov = 0, path_pad = 0, path_oov = 0, tgt_oov = 0;
  int max_contexts = 200;
  int n_threads = 6;  // reference READER_NUM_PARALLEL_BATCHES default

  Parser(const std::unordered_map<std::string, int>& tok_map,
         const std::unordered_map<std::string, int>& path_map,
         const std::unordered_map<std::string, int>& tgt_map, int tok_pad_,
         int tok_oov_, int path_pad_, int path_oov_, int tgt_oov_,
         int max_contexts_, int n_threads_)
      : tok_pad(tok_pad_), tok_oov(tok_oov_), path_pad(path_pad_),
        path_oov(path_oov_), tgt_oov(tgt_oov_), max_contexts(max_contexts_),
        n_threads(std::max(1, n_threads_)) {
    tok.build(tok_map);
    path.build(path_map);
    tgt.build(tgt_map);
  }

  inline int look(const StrMap& m, std::string_view s, int oov) const {
    return m.look(s, oov);
  }

  // Parse one line into the row buffers. Returns target index.
  // Two-pass structure: tokenize first, then run the hash lookups with a
  // software-prefetch window — the 3C lookups per row are independent
  // random DRAM reads (the vocab tables dwarf every cache), and issuing
  // them ~4 contexts ahead turns a serial miss chain into overlapped
  // misses.
  int parse_line(std::string_view line, int* src_row, int* path_row,
                 int* tgt_row, float* mask_row) const {
    const int C = max_contexts;
    for (int c = 0; c < C; ++c) {
      src_row[c] = tok_pad;
      path_row[c] = path_pad;
      tgt_row[c] = tok_pad;
      mask_row[c] = 0.f;
    }
    // strip trailing newline
    while (!line.empty() && (line.back() == '\n' || line.back() == '\r'))
      line.remove_suffix(1);

    size_t pos = line.find(' ');
    std::string_view target = line.substr(0, pos == std::string_view::npos
                                                 ? line.size() : pos);
    int target_idx = target.empty() ? tgt_oov : look(tgt, target, tgt_oov);

    // pass 1: split fields into per-part views (thread-local scratch:
    // parse_line runs on many threads; a per-row heap alloc would churn)
    thread_local std::vector<std::string_view> f0, f1, f2;
    f0.assign(C, {});
    f1.assign(C, {});
    f2.assign(C, {});
    int n_ctx = 0;
    int c = 0;
    while (pos != std::string_view::npos && c < C) {
      size_t start = pos + 1;
      pos = line.find(' ', start);
      std::string_view field = line.substr(
          start, (pos == std::string_view::npos ? line.size() : pos) - start);
      if (field.empty()) { ++c; continue; }
      size_t c1 = field.find(',');
      size_t c2 = c1 == std::string_view::npos ? std::string_view::npos
                                               : field.find(',', c1 + 1);
      f0[c] = field.substr(0, c1);
      f1[c] = c1 == std::string_view::npos
                  ? std::string_view()
                  : field.substr(c1 + 1, (c2 == std::string_view::npos
                                              ? field.size() : c2) - c1 - 1);
      f2[c] = c2 == std::string_view::npos ? std::string_view()
                                           : field.substr(c2 + 1);
      ++c;
      n_ctx = c;
    }

    // pass 2: lookups with a prefetch window
    constexpr int PF = 4;
    auto prefetch3 = [&](int j) {
      if (j >= n_ctx) return;
      if (!f0[j].empty())
        __builtin_prefetch(&tok.slots[StrMap::hash(f0[j]) & tok.mask]);
      if (!f1[j].empty())
        __builtin_prefetch(&path.slots[StrMap::hash(f1[j]) & path.mask]);
      if (!f2[j].empty())
        __builtin_prefetch(&tok.slots[StrMap::hash(f2[j]) & tok.mask]);
    };
    for (int j = 0; j < std::min(PF, n_ctx); ++j) prefetch3(j);
    for (int j = 0; j < n_ctx; ++j) {
      prefetch3(j + PF);
      if (f0[j].empty() && f1[j].empty() && f2[j].empty()) continue;
      int si = f0[j].empty() ? tok_pad : look(tok, f0[j], tok_oov);
      int pi = f1[j].empty() ? path_pad : look(path, f1[j], path_oov);
      int ti = f2[j].empty() ? tok_pad : look(tok, f2[j], tok_oov);
      src_row[j] = si;
      path_row[j] = pi;
      tgt_row[j] = ti;
      mask_row[j] = (si != tok_pad || ti != tok_pad || pi != path_pad)
                        ? 1.f : 0.f;
    }
    return target_idx;
  }

  // Parse a whole text buffer (zero-copy from python bytes): lines are
  // split here and parsed thread-parallel — no per-line python objects.
  py::tuple parse_buffer(py::bytes data) const {
    std::string_view buf = std::string_view(data);
    return parse_views(buf, std::string_view());
  }

  // Zero-copy variant for the streaming reader: `body[:body_len]` is a raw
  // read() chunk ending at a line boundary and `head` is the previous
  // chunk's partial last line (prepended to the FIRST line here). The
  // python-side `carry + chunk` concat and `buf[:last_nl+1]` slice this
  // replaces moved ~8-12 MB per 4 MB chunk on one thread and capped the
  // whole pipeline at ~260K rows/s.
  py::tuple parse_buffer2(py::bytes body, long body_len, py::bytes head,
                          long nt_override) const {
    std::string_view body_sv = std::string_view(body).substr(
        0, (size_t)std::max<long>(0, body_len));
    return parse_views(body_sv, std::string_view(head), (int)nt_override);
  }

  py::tuple parse_views(std::string_view buf, std::string_view head,
                        int nt_override = 0) const {
    // GIL released for the whole body: line scanning, tensor allocation and
    // parsing are pure C++ (the string_views borrow python buffers the
    // caller keeps referenced); only the return-tuple build needs the GIL
    py::gil_scoped_release release_all;
    // line offsets; a non-empty head splices onto buf's first line
    std::string first_line;
    std::vector<std::pair<size_t, size_t>> lines;
    bool has_first = false;
    size_t start = 0;
    if (!head.empty()) {
      const size_t nl = buf.find('\n');
      first_line.assign(head);
      first_line.append(buf.substr(0, nl == std::string_view::npos
                                          ? buf.size() : nl));
      has_first = !first_line.empty();
      start = (nl == std::string_view::npos) ? buf.size() : nl + 1;
    }
    while (start < buf.size()) {
      size_t nl = buf.find('\n', start);
      size_t end = (nl == std::string_view::npos) ? buf.size() : nl;
      if (end > start) lines.push_back({start, end - start});
      start = end + 1;
    }
    const int64_t nplain = (int64_t)lines.size();
    const int64_t B = nplain + (has_first ? 1 : 0);
    const int64_t C = max_contexts;
    auto opts = torch::TensorOptions().dtype(torch::kInt32);
    auto src = torch::empty({B, C}, opts);
    auto pth = torch::empty({B, C}, opts);
    auto tgt = torch::empty({B, C}, opts);
    auto mask = torch::empty({B, C}, torch::TensorOptions().dtype(torch::kFloat32));
    auto tidx = torch::empty({B}, torch::TensorOptions().dtype(torch::kInt64));
    int* src_p = src.data_ptr<int>();
    int* pth_p = pth.data_ptr<int>();
    int* tgt_p = tgt.data_ptr<int>();
    float* mask_p = mask.data_ptr<float>();
    int64_t* tidx_p = tidx.data_ptr<int64_t>();
    {
      const int want_nt = nt_override > 0 ? nt_override : n_threads;
      const int nt = (int)std::min<int64_t>(want_nt, std::max<int64_t>(1, B));
      std::atomic<int64_t> next(0);
      const int64_t off = has_first ? 1 : 0;
      if (has_first)
        tidx_p[0] = parse_line(first_line, src_p, pth_p, tgt_p, mask_p);
      auto work = [&]() {
        int64_t i;
        while ((i = next.fetch_add(1)) < nplain) {
          const int64_t o = i + off;
          tidx_p[o] = parse_line(buf.substr(lines[i].first, lines[i].second),
                                 src_p + o * C, pth_p + o * C, tgt_p + o * C,
                                 mask_p + o * C);
        }
      };
      if (nt <= 1) {
        work();  // caller-side parallelism (outer worker threads): no
                 // per-call thread spawn/join
      } else {
        std::vector<std::thread> pool;
        for (int t = 0; t < nt; ++t) pool.emplace_back(work);
        for (auto& th : pool) th.join();
      }
    }
    py::gil_scoped_acquire acquire;
    return py::make_tuple(src, pth, tgt, mask, tidx);
  }

  // Parse a batch of lines (thread-parallel across rows).
  py::tuple parse_batch(const std::vector<std::string>& lines) const {
    const int64_t B = (int64_t)lines.size();
    const int64_t C = max_contexts;
    auto opts = torch::TensorOptions().dtype(torch::kInt32);
    auto src = torch::empty({B, C}, opts);
    auto pth = torch::empty({B, C}, opts);
    auto tgt = torch::empty({B, C}, opts);
    auto mask = torch::empty({B, C}, torch::TensorOptions().dtype(torch::kFloat32));
    auto tidx = torch::empty({B}, torch::TensorOptions().dtype(torch::kInt64));

    int* src_p = src.data_ptr<int>();
    int* pth_p = pth.data_ptr<int>();
    int* tgt_p = tgt.data_ptr<int>();
    float* mask_p = mask.data_ptr<float>();
    int64_t* tidx_p = tidx.data_ptr<int64_t>();

    {
      py::gil_scoped_release release;  // parsing is pure C++
      const int nt = (int)std::min<int64_t>(n_threads, std::max<int64_t>(1, B));
      std::atomic<int64_t> next(0);
      auto work = [&]() {
        int64_t i;
        while ((i = next.fetch_add(1)) < B) {
          tidx_p[i] = parse_line(lines[i], src_p + i * C, pth_p + i * C,
                                 tgt_p + i * C, mask_p + i * C);
        }
      };
      if (nt <= 1) {
        work();
      } else {
        std::vector<std::thread> pool;
        pool.reserve(nt);
        for (int t = 0; t < nt; ++t) pool.emplace_back(work);
        for (auto& th : pool) th.join();
      }
    }
    return py::make_tuple(src, pth, tgt, mask, tidx);
  }
};

// Fused concat + permutation gather for the reader's windowed shuffle:
// out[field][i] = chunks[field][perm[i]] with perm indexing the virtual
// concatenation of the chunk list. Runs GIL-free on a thread pool — the
// python-side cat + index_select version held the GIL for ~1 ms per batch
// and serialized against the training loop's launch thread.
void shuffle_gather(const std::vector<std::vector<torch::Tensor>>& fields,
                    torch::Tensor perm,
                    const std::vector<torch::Tensor>& outs, int64_t n_threads) {
  TORCH_CHECK(!fields.empty() && fields.size() == outs.size());
  const int64_t n = perm.numel();
  auto perm_c = perm.contiguous();
  const int64_t* pp = perm_c.data_ptr<int64_t>();
  const size_t nf = fields.size();
  // chunk row offsets (same split for every field)
  std::vector<int64_t> offs{0};
  for (const auto& t : fields[0]) offs.push_back(offs.back() + t.size(0));
  TORCH_CHECK(offs.back() >= n, "perm longer than pooled rows");
  struct F {
    std::vector<const char*> src;
    char* dst;
    size_t row_bytes;
  };
  std::vector<F> fs(nf);
  for (size_t f = 0; f < nf; ++f) {
    TORCH_CHECK(fields[f].size() == fields[0].size());
    TORCH_CHECK(outs[f].is_contiguous() && outs[f].size(0) >= n);
    fs[f].dst = (char*)outs[f].data_ptr();
    fs[f].row_bytes = outs[f].numel() / outs[f].size(0)
                      * outs[f].element_size();
    for (size_t c = 0; c < fields[f].size(); ++c) {
      auto& t = fields[f][c];
      TORCH_CHECK(t.is_contiguous());
      TORCH_CHECK(t.size(0) == fields[0][c].size(0), "ragged chunk");
      fs[f].src.push_back((const char*)t.data_ptr());
    }
  }
  {
    py::gil_scoped_release release;
    const int nt = (int)std::max<int64_t>(1, std::min<int64_t>(n_threads, n));
    std::atomic<int64_t> next(0);
    constexpr int64_t GRAIN = 512;
    auto work = [&]() {
      int64_t base;
      while ((base = next.fetch_add(GRAIN)) < n) {
        const int64_t end = std::min(n, base + GRAIN);
        for (int64_t i = base; i < end; ++i) {
          const int64_t g = pp[i];
          // binary search the owning chunk
          size_t lo = 0, hi = offs.size() - 1;
          while (hi - lo > 1) {
            const size_t mid = (lo + hi) / 2;
            if (offs[mid] <= g) lo = mid; else hi = mid;
          }
          const int64_t r = g - offs[lo];
          for (size_t f = 0; f < nf; ++f)
            std::memcpy(fs[f].dst + i * fs[f].row_bytes,
                        fs[f].src[lo] + r * fs[f].row_bytes, fs[f].row_bytes);
        }
      }
    };
    if (nt <= 1) {
      work();
    } else {
      std::vector<std::thread> pool;
      pool.reserve(nt);
      for (int t = 0; t < nt; ++t) pool.emplace_back(work);
      for (auto& th : pool) th.join();
    }
  }
}

}  // namespace

PYBIND11_MODULE(TORCH_EXTENSION_NAME, m) {
  m.def("shuffle_gather", &shuffle_gather);
  py::class_<Parser>(m, "Parser")
      .def(py::init<const std::unordered_map<std::string, int>&,
                    const std::unordered_map<std::string, int>&,
                    const std::unordered_map<std::string, int>&, int, int, int,
                    int, int, int, int>())
      .def("parse_batch", &Parser::parse_batch)
      .def("parse_buffer", &Parser::parse_buffer)
      .def("parse_buffer2", &Parser::parse_buffer2);
}
