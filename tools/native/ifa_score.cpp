// Standalone MI355X scorer — no Python, no torch at runtime.
//
// Loads a saved standard Isolation Forest model (the reference-compatible
// Avro + metadata-JSON directory written by persist/model_io.py or by the
// reference's Spark writer), packs it into the v4 scoring format and runs
// the HIP scoring kernel (linked directly from ops/hip/forest_kernels.hip,
// which is torch-free) over a raw row-major matrix.
//
//   ifa_score <model_dir> <input.bin> <rows> <features> <scores_out.bin>
//             [--bf16] [--labels labels_out.bin]
//
// input.bin: rows*features little-endian f32 (or bf16 with --bf16).
// scores_out.bin: rows f32 outlier scores. --labels: rows u8 (score >=
// outlierScoreThreshold), requires a threshold-bearing model.
//
// Build (see __graft_entry__.build / tools/native/build.sh):
//   hipcc --offload-arch=gfx950 -O3 -ffp-contract=off \
//     tools/native/ifa_score.cpp isolation_forest_amd/ops/hip/forest_kernels.hip \
//     -Iisolation_forest_amd/ops/hip -lz -o tools/native/ifa_score
//
// Scores match the Python engine to <= 1e-6 (identical walk decisions via
// the same integer-key packing; leaf constants differ only by libm-vs-numpy
// float32 log ulps). Extended models route through the same sparse/dense
// kernels as the Python engine (nnz <= 5 sparse, d <= 32 densified dense);
// wide-d EIF falls back to the Python engine with a clear message.

#include <hip/hip_runtime.h>
#include <zlib.h>

#include <algorithm>

#include <cmath>
#include <cstdint>
#include <cstdio>
#include <cstring>
#include <dirent.h>
#include <fstream>
#include <map>
#include <sstream>
#include <stdexcept>
#include <string>
#include <vector>

namespace ifa {
void launch_score_forest(bool bf16, int rpt, bool rows_lds, bool nodes_lds,
                         int ilp, const void* X, const void* nodes,
                         const int32_t* ncount, float* out, int64_t N,
                         int32_t d, int32_t dpad, int32_t Tpad,
                         int32_t max_nodes, int32_t height_limit, float fT,
                         float c_norm, int finalize, size_t lds, int blocks,
                         hipStream_t stream);
void launch_score_extended_dense_v2(bool bf16, int D, const void* X,
                                    const void* nodes, const float* values,
                                    const float* hw, const int32_t* ncount,
                                    float* out, int64_t N, int32_t d,
                                    int32_t T, int32_t max_nodes,
                                    int32_t height_limit, float fT,
                                    float c_norm, int finalize, size_t lds,
                                    int blocks, hipStream_t stream);
void launch_score_extended_sparse_v2(bool bf16, int nnz, const void* X,
                                     const void* nodes, const float* values,
                                     const int32_t* hidx, const float* hw,
                                     const int32_t* ncount, float* out,
                                     int64_t N, int32_t d, int32_t dpad,
                                     int32_t T, int32_t max_nodes,
                                     int32_t height_limit, float fT,
                                     float c_norm, int finalize, size_t lds,
                                     int blocks, hipStream_t stream);
}

#define HIP_CHECK(x)                                                         \
  do {                                                                       \
    hipError_t err_ = (x);                                                   \
    if (err_ != hipSuccess) {                                                \
      fprintf(stderr, "HIP error %s at %s:%d\n", hipGetErrorString(err_),    \
              __FILE__, __LINE__);                                           \
      exit(3);                                                               \
    }                                                                        \
  } while (0)

// ---------------------------------------------------------------------------
// small file / avro helpers
// ---------------------------------------------------------------------------

static std::vector<uint8_t> read_file(const std::string& path) {
  std::ifstream f(path, std::ios::binary);
  if (!f) throw std::runtime_error("cannot open " + path);
  return std::vector<uint8_t>((std::istreambuf_iterator<char>(f)),
                              std::istreambuf_iterator<char>());
}

static std::vector<std::string> list_dir(const std::string& dir,
                                         const std::string& prefix,
                                         const std::string& suffix) {
  std::vector<std::string> out;
  DIR* d = opendir(dir.c_str());
  if (!d) throw std::runtime_error("cannot list " + dir);
  while (dirent* e = readdir(d)) {
    std::string n = e->d_name;
    if (n.size() >= prefix.size() + suffix.size() &&
        n.compare(0, prefix.size(), prefix) == 0 &&
        n.compare(n.size() - suffix.size(), suffix.size(), suffix) == 0)
      out.push_back(dir + "/" + n);
  }
  closedir(d);
  std::sort(out.begin(), out.end());
  return out;
}

struct Reader {
  const uint8_t* p;
  size_t pos = 0, n;
  explicit Reader(const std::vector<uint8_t>& b) : p(b.data()), n(b.size()) {}
  int64_t zz() {  // avro zigzag long
    uint64_t u = 0;
    int shift = 0;
    while (true) {
      if (pos >= n) throw std::runtime_error("avro: truncated varint");
      uint8_t b = p[pos++];
      u |= (uint64_t)(b & 0x7F) << shift;
      if (!(b & 0x80)) break;
      shift += 7;
    }
    return (int64_t)(u >> 1) ^ -(int64_t)(u & 1);
  }
  void bytes(void* dst, size_t k) {
    if (pos + k > n) throw std::runtime_error("avro: truncated bytes");
    memcpy(dst, p + pos, k);
    pos += k;
  }
  std::vector<uint8_t> blob(size_t k) {
    std::vector<uint8_t> v(k);
    bytes(v.data(), k);
    return v;
  }
  double f64() {
    double v;
    bytes(&v, 8);
    return v;
  }
};

static std::vector<uint8_t> inflate_raw(const std::vector<uint8_t>& in) {
  std::vector<uint8_t> out;
  out.reserve(in.size() * 4 + 64);
  z_stream zs{};
  if (inflateInit2(&zs, -15) != Z_OK) throw std::runtime_error("zlib init");
  zs.next_in = const_cast<Bytef*>(in.data());
  zs.avail_in = (uInt)in.size();
  std::vector<uint8_t> buf(1 << 16);
  int rc;
  do {
    zs.next_out = buf.data();
    zs.avail_out = (uInt)buf.size();
    rc = inflate(&zs, Z_NO_FLUSH);
    if (rc != Z_OK && rc != Z_STREAM_END)
      throw std::runtime_error("zlib inflate failed");
    out.insert(out.end(), buf.data(), buf.data() + (buf.size() - zs.avail_out));
  } while (rc != Z_STREAM_END);
  inflateEnd(&zs);
  return out;
}

static std::vector<uint8_t> snappy_decompress(const std::vector<uint8_t>& in) {
  size_t pos = 0;
  uint64_t len = 0;
  int shift = 0;
  while (true) {
    uint8_t b = in.at(pos++);
    len |= (uint64_t)(b & 0x7F) << shift;
    if (!(b & 0x80)) break;
    shift += 7;
  }
  std::vector<uint8_t> out;
  out.reserve(len);
  while (pos < in.size()) {
    uint8_t tag = in[pos++];
    int type = tag & 3;
    if (type == 0) {  // literal
      uint32_t ln = (tag >> 2) + 1;
      if (ln > 60) {
        int nb = ln - 60;
        ln = 0;
        for (int i = 0; i < nb; ++i) ln |= (uint32_t)in.at(pos++) << (8 * i);
        ln += 1;
      }
      out.insert(out.end(), in.begin() + pos, in.begin() + pos + ln);
      pos += ln;
    } else {
      uint32_t ln, off;
      if (type == 1) {
        ln = ((tag >> 2) & 7) + 4;
        off = ((uint32_t)(tag >> 5) << 8) | in.at(pos++);
      } else if (type == 2) {
        ln = (tag >> 2) + 1;
        off = in.at(pos) | ((uint32_t)in.at(pos + 1) << 8);
        pos += 2;
      } else {
        ln = (tag >> 2) + 1;
        off = 0;
        for (int i = 0; i < 4; ++i) off |= (uint32_t)in.at(pos++) << (8 * i);
      }
      if (off == 0 || off > out.size())
        throw std::runtime_error("snappy: bad copy offset");
      size_t start = out.size() - off;
      for (uint32_t i = 0; i < ln; ++i) out.push_back(out[start + i]);
    }
  }
  if (out.size() != len) throw std::runtime_error("snappy: length mismatch");
  return out;
}

// ---------------------------------------------------------------------------
// model loading
// ---------------------------------------------------------------------------

struct NodeRec {
  int32_t id, left, right, split_attr;
  double split_value;
  int64_t num_instances;
  std::vector<int32_t> indices;  // extended only
  std::vector<float> weights;    // extended only
};

struct Model {
  int num_samples = 0, num_features = 0;
  double threshold = -1.0;
  bool extended = false;
  std::map<int, std::vector<NodeRec>> trees;
};

static bool find_json_number(const std::string& s, const std::string& key,
                             double* out) {
  size_t k = s.find("\"" + key + "\"");
  if (k == std::string::npos) return false;
  k = s.find(':', k);
  if (k == std::string::npos) return false;
  *out = strtod(s.c_str() + k + 1, nullptr);
  return true;
}

static Model load_model(const std::string& dir) {
  Model m;
  auto metas = list_dir(dir + "/metadata", "part-", "");
  if (metas.empty()) throw std::runtime_error("no metadata under " + dir);
  auto mb = read_file(metas[0]);
  std::string meta(mb.begin(), mb.end());
  m.extended = meta.find("Extended") != std::string::npos;
  double v;
  if (find_json_number(meta, "numSamples", &v)) m.num_samples = (int)v;
  if (find_json_number(meta, "numFeatures", &v)) m.num_features = (int)v;
  if (find_json_number(meta, "outlierScoreThreshold", &v)) m.threshold = v;

  for (const auto& f : list_dir(dir + "/data", "part-", ".avro")) {
    auto raw = read_file(f);
    Reader r(raw);
    uint8_t magic[4];
    r.bytes(magic, 4);
    if (memcmp(magic, "Obj\x01", 4) != 0)
      throw std::runtime_error(f + " is not avro");
    std::string codec = "null";
    while (true) {  // metadata map
      int64_t cnt = r.zz();
      if (cnt == 0) break;
      if (cnt < 0) {
        r.zz();
        cnt = -cnt;
      }
      for (int64_t i = 0; i < cnt; ++i) {
        auto key = r.blob((size_t)r.zz());
        auto val = r.blob((size_t)r.zz());
        std::string ks(key.begin(), key.end());
        if (ks == "avro.codec") codec.assign(val.begin(), val.end());
        if (ks == "avro.schema") {
          std::string sch(val.begin(), val.end());
          if (sch.find("extendedNodeData") != std::string::npos)
            m.extended = true;
        }
      }
    }
    uint8_t sync[16];
    r.bytes(sync, 16);
    while (r.pos < r.n) {
      int64_t count = r.zz();
      int64_t size = r.zz();
      auto block = r.blob((size_t)size);
      uint8_t s2[16];
      r.bytes(s2, 16);
      if (memcmp(s2, sync, 16) != 0)
        throw std::runtime_error("avro sync mismatch");
      std::vector<uint8_t> plain;
      if (codec == "null")
        plain = std::move(block);
      else if (codec == "deflate")
        plain = inflate_raw(block);
      else if (codec == "snappy")
        plain = snappy_decompress(
            std::vector<uint8_t>(block.begin(), block.end() - 4));
      else
        throw std::runtime_error("unsupported avro codec " + codec);
      Reader br(plain);
      for (int64_t i = 0; i < count; ++i) {
        int32_t tree = (int32_t)br.zz();
        int64_t branch = br.zz();  // nodeData union index
        if (branch != 0) throw std::runtime_error("null nodeData record");
        NodeRec nd;
        nd.id = (int32_t)br.zz();
        nd.left = (int32_t)br.zz();
        nd.right = (int32_t)br.zz();
        if (!m.extended) {
          nd.split_attr = (int32_t)br.zz();
          nd.split_value = br.f64();
          nd.num_instances = br.zz();
        } else {
          if (br.zz() != 0)
            throw std::runtime_error("null indices array");
          while (true) {  // indices array blocks
            int64_t c = br.zz();
            if (c == 0) break;
            if (c < 0) { br.zz(); c = -c; }
            for (int64_t j = 0; j < c; ++j)
              nd.indices.push_back((int32_t)br.zz());
          }
          if (br.zz() != 0)
            throw std::runtime_error("null weights array");
          while (true) {  // weights array blocks (float items)
            int64_t c = br.zz();
            if (c == 0) break;
            if (c < 0) { br.zz(); c = -c; }
            for (int64_t j = 0; j < c; ++j) {
              float w;
              br.bytes(&w, 4);
              nd.weights.push_back(w);
            }
          }
          nd.split_value = br.f64();  // offset
          nd.num_instances = br.zz();
          nd.split_attr = nd.indices.empty() ? -1 : (int32_t)nd.indices.size();
        }
        m.trees[tree].push_back(nd);
      }
    }
  }
  return m;
}

// ---------------------------------------------------------------------------
// v4 packing (mirror of ops/gpu_engine._nodes_packed_v4)
// ---------------------------------------------------------------------------

static float avg_path_length32(int64_t n) {
  if (n <= 1) return 0.0f;
  float nf = (float)n;
  float ln = logf(nf - 1.0f);
  float t1 = 2.0f * (ln + 0.5772156649f);
  float t2 = 2.0f * (nf - 1.0f) / nf;
  return t1 - t2;
}

static uint32_t bf16_threshold_key(float s) {
  uint32_t bits;
  memcpy(&bits, &s, 4);
  bool neg = bits & 0x80000000u;
  uint32_t b = bits >> 16;
  if (!neg && (bits & 0xFFFFu)) b += 1;
  uint32_t mm = b & 0x7FFFu;
  uint32_t k = (b & 0x8000u) ? (0x7FFFu - mm) : (0x8000u + mm);
  if (mm == 0) k = 0x8000u;
  return k << 16;
}

static uint32_t f32_key(float s) {
  uint32_t b;
  memcpy(&b, &s, 4);
  uint32_t mm = b & 0x7FFFFFFFu;
  uint32_t k = (b & 0x80000000u) ? (0x7FFFFFFFu - mm) : (0x80000000u + mm);
  if (mm == 0) k = 0x80000000u;
  return k;
}

struct Packed {
  std::vector<int32_t> nodes;  // [Tpad][mn][2]
  std::vector<int32_t> ncount;
  int Tpad = 0, T = 0, mn = 0, max_depth = 1;
};

static Packed pack_v4(const Model& m, int d_sentinel, bool bf16) {
  Packed p;
  p.T = m.trees.empty() ? 0 : (m.trees.rbegin()->first + 1);
  for (const auto& kv : m.trees)
    p.mn = std::max(p.mn, (int)kv.second.size());
  p.mn = std::max(p.mn, 1);
  if (p.mn > 32767) throw std::runtime_error("forest too deep to pack");
  p.Tpad = ((p.T + 7) / 8) * 8;
  if (p.Tpad == 0) p.Tpad = 8;
  p.nodes.assign((size_t)p.Tpad * p.mn * 2, 0);
  p.ncount.assign(p.Tpad, 1);
  std::vector<int> depth(p.mn);
  for (int t = 0; t < p.Tpad; ++t) {
    auto it = m.trees.find(t);
    const std::vector<NodeRec>* recs =
        (t < p.T && it != m.trees.end()) ? &it->second : nullptr;
    int nc = recs ? (int)recs->size() : 0;
    p.ncount[t] = std::max(nc, 1);
    int32_t* base = p.nodes.data() + (size_t)t * p.mn * 2;
    // default/dummy: every slot a self-looping value-0 leaf
    for (int i = 0; i < p.mn; ++i) {
      base[2 * i] = d_sentinel | (i << 12);
      base[2 * i + 1] = 0;
    }
    if (!recs) continue;
    std::vector<NodeRec> sorted(*recs);
    std::sort(sorted.begin(), sorted.end(),
              [](const NodeRec& a, const NodeRec& b) { return a.id < b.id; });
    recs = nullptr;  // use `sorted` below
    std::fill(depth.begin(), depth.end(), 0);
    for (const auto& nd : sorted) {  // pre-order: parent precedes children
      if (nd.left != -1) {
        if (nd.left != nd.id + 1)
          throw std::runtime_error("pre-order invariant broken");
        depth.at(nd.left) = depth.at(nd.id) + 1;
        depth.at(nd.right) = depth.at(nd.id) + 1;
      }
    }
    for (const auto& nd : sorted) {
      int32_t w0, w1;
      if (nd.left == -1) {
        w0 = d_sentinel | (nd.id << 12);
        float v = (float)depth.at(nd.id) + avg_path_length32(nd.num_instances);
        memcpy(&w1, &v, 4);
        p.max_depth = std::max(p.max_depth, depth.at(nd.id));
      } else {
        w0 = nd.split_attr | (nd.right << 12);
        // exact f64 compare semantics: x < s(f64) <=> x < ceil32(s)
        // (IsolationTree.scala:213-229; see gpu_engine._split_thresholds32)
        float s32 = (float)nd.split_value;
        if ((double)s32 < nd.split_value)
          s32 = std::nextafterf(s32, INFINITY);
        uint32_t key = bf16 ? bf16_threshold_key(s32) : f32_key(s32);
        memcpy(&w1, &key, 4);
        p.max_depth = std::max(p.max_depth, depth.at(nd.id) + 1);
      }
      base[2 * nd.id] = w0;
      base[2 * nd.id + 1] = w1;
    }
  }
  return p;
}

// ---------------------------------------------------------------------------
// EIF packing (mirror of ops/gpu_engine._eif_nodes_values/_eif_dense_packed)
// ---------------------------------------------------------------------------

struct PackedEIF {
  std::vector<int32_t> nodes;   // [T][mn][2]: {right|self<<12, off32/-inf}
  std::vector<float> values;    // [T][mn] depth-folded leaf values
  std::vector<int32_t> ncount;  // [T]
  std::vector<float> hw_dense;  // [T][mn][D] when dense route
  std::vector<int32_t> hidx;    // [T][mn][nnz] when sparse route
  std::vector<float> hw_sparse; // [T][mn][nnz]
  int T = 0, mn = 0, nnz = 0, max_depth = 1;
  bool uniform = true;
};

static PackedEIF pack_eif(const Model& m, int D_dense, bool want_sparse) {
  PackedEIF p;
  p.T = m.trees.empty() ? 0 : (m.trees.rbegin()->first + 1);
  for (const auto& kv : m.trees) {
    p.mn = std::max(p.mn, (int)kv.second.size());
    for (const auto& nd : kv.second)
      p.nnz = std::max(p.nnz, (int)nd.indices.size());
  }
  p.mn = std::max(p.mn, 1);
  p.nnz = std::max(p.nnz, 1);
  if (p.mn > 32767) throw std::runtime_error("forest too deep to pack");
  p.nodes.assign((size_t)p.T * p.mn * 2, 0);
  p.values.assign((size_t)p.T * p.mn, 0.0f);
  p.ncount.assign(std::max(p.T, 1), 1);
  if (want_sparse) {
    p.hidx.assign((size_t)p.T * p.mn * p.nnz, 0);
    p.hw_sparse.assign((size_t)p.T * p.mn * p.nnz, 0.0f);
  } else {
    p.hw_dense.assign((size_t)p.T * p.mn * D_dense, 0.0f);
  }
  const float ninf = -INFINITY;
  std::vector<int> depth(p.mn);
  for (const auto& kv : m.trees) {
    int t = kv.first;
    std::vector<NodeRec> sorted(kv.second);
    std::sort(sorted.begin(), sorted.end(),
              [](const NodeRec& a, const NodeRec& b) { return a.id < b.id; });
    p.ncount[t] = (int32_t)sorted.size();
    std::fill(depth.begin(), depth.end(), 0);
    for (const auto& nd : sorted)
      if (nd.left != -1) {
        if (nd.left != nd.id + 1)
          throw std::runtime_error("pre-order invariant broken");
        depth.at(nd.left) = depth.at(nd.id) + 1;
        depth.at(nd.right) = depth.at(nd.id) + 1;
      }
    int32_t* base = p.nodes.data() + (size_t)t * p.mn * 2;
    float* vals = p.values.data() + (size_t)t * p.mn;
    for (const auto& nd : sorted) {
      if (nd.left == -1) {
        base[2 * nd.id] = nd.id << 12;  // self-loop
        memcpy(&base[2 * nd.id + 1], &ninf, 4);
        vals[nd.id] =
            (float)depth.at(nd.id) + avg_path_length32(nd.num_instances);
        p.max_depth = std::max(p.max_depth, depth.at(nd.id));
      } else {
        base[2 * nd.id] = nd.right << 12;
        float off32 = (float)nd.split_value;
        memcpy(&base[2 * nd.id + 1], &off32, 4);
        p.max_depth = std::max(p.max_depth, depth.at(nd.id) + 1);
        if ((int)nd.indices.size() != p.nnz) p.uniform = false;
        for (size_t j = 0; j < nd.indices.size(); ++j) {
          if (want_sparse) {
            p.hidx[((size_t)t * p.mn + nd.id) * p.nnz + j] = nd.indices[j];
            p.hw_sparse[((size_t)t * p.mn + nd.id) * p.nnz + j] =
                nd.weights[j];
          } else {
            p.hw_dense[((size_t)t * p.mn + nd.id) * D_dense +
                       nd.indices[j]] = nd.weights[j];
          }
        }
      }
    }
  }
  return p;
}

// ---------------------------------------------------------------------------
// main
// ---------------------------------------------------------------------------

int main(int argc, char** argv) {
  if (argc < 6) {
    fprintf(stderr,
            "usage: %s <model_dir> <input.bin> <rows> <features> "
            "<scores_out.bin> [--bf16] [--labels labels_out.bin]\n",
            argv[0]);
    return 2;
  }
  std::string model_dir = argv[1], input = argv[2], out_path = argv[5];
  int64_t N = atoll(argv[3]);
  int d = atoi(argv[4]);
  bool bf16 = false;
  std::string labels_path;
  for (int i = 6; i < argc; ++i) {
    if (!strcmp(argv[i], "--bf16")) bf16 = true;
    else if (!strcmp(argv[i], "--labels") && i + 1 < argc)
      labels_path = argv[++i];
  }

  Model m;
  try {
    m = load_model(model_dir);
  } catch (const std::exception& e) {
    fprintf(stderr, "error: %s\n", e.what());
    return 2;
  }
  if (m.num_features > 0 && m.num_features != d) {
    fprintf(stderr,
            "error: input feature count %d != model numFeatures %d\n", d,
            m.num_features);
    return 2;
  }
  if (d > 4094) {
    fprintf(stderr, "error: d must be <= 4094\n");
    return 2;
  }
  if (m.trees.empty()) {
    fprintf(stderr, "error: model has no trees; cannot score\n");
    return 2;
  }
  if (m.num_samples < 2) {
    fprintf(stderr, "error: numSamples < 2; cannot score\n");
    return 2;
  }
  float c_norm = avg_path_length32(m.num_samples);

  size_t elem = bf16 ? 2 : 4;
  auto data = read_file(input);
  if (data.size() != (size_t)N * d * elem) {
    fprintf(stderr, "error: %s is %zu bytes, expected %zu (N*d*%zu)\n",
            input.c_str(), data.size(), (size_t)N * d * elem, elem);
    return 2;
  }

  void *dX, *dOut;
  HIP_CHECK(hipMalloc(&dX, data.size()));
  HIP_CHECK(hipMalloc(&dOut, (size_t)N * 4));
  HIP_CHECK(hipMemcpy(dX, data.data(), data.size(), hipMemcpyHostToDevice));

  const size_t kMaxLds = 160 * 1024;
  int64_t dpad = d + 1;
  if (bf16) {
    while (dpad % 4 != 2) ++dpad;
  } else {
    while (dpad % 2 != 1) ++dpad;
  }
  int T_report = 0;

  if (!m.extended) {
    Packed p = pack_v4(m, d, bf16);
    T_report = p.T;
    void *dNodes, *dNcount;
    HIP_CHECK(hipMalloc(&dNodes, p.nodes.size() * 4));
    HIP_CHECK(hipMalloc(&dNcount, p.ncount.size() * 4));
    HIP_CHECK(hipMemcpy(dNodes, p.nodes.data(), p.nodes.size() * 4,
                        hipMemcpyHostToDevice));
    HIP_CHECK(hipMemcpy(dNcount, p.ncount.data(), p.ncount.size() * 4,
                        hipMemcpyHostToDevice));
    size_t node_bytes = (size_t)4 * p.mn * 8;
    bool nodes_lds = true;
    if (node_bytes + (size_t)256 * dpad * elem > 150 * 1024) {
      nodes_lds = false;
      node_bytes = 0;
    }
    int rpt = 2;
    bool rows_lds = true;
    size_t lds;
    for (;;) {
      size_t row_bytes = (size_t)rpt * 256 * dpad * elem;
      lds = node_bytes + row_bytes;
      if (lds * 3 <= kMaxLds || (rpt == 1 && lds <= 150 * 1024)) break;
      if (rpt == 2) { rpt = 1; continue; }
      rows_lds = false;
      lds = node_bytes;
      break;
    }
    int64_t rows_per_block = (int64_t)(rows_lds ? rpt : 1) * 256;
    int blocks = (int)std::min<int64_t>(
        (N + rows_per_block - 1) / rows_per_block, 8192);
    ifa::launch_score_forest(bf16, rpt, rows_lds, nodes_lds, 4, dX, dNodes,
                             (const int32_t*)dNcount, (float*)dOut, N, d,
                             (int32_t)dpad, p.Tpad, p.mn, p.max_depth,
                             (float)p.T, c_norm, 1, lds, blocks, 0);
  } else {
    // routing mirror of ops/gpu_engine.score_extended_forest (the wide-d
    // general kernel is Python-only; practical configs are covered here)
    int nnz_seen = 1;
    for (const auto& kv : m.trees)
      for (const auto& nd : kv.second)
        nnz_seen = std::max(nnz_seen, (int)nd.indices.size());
    int blocks = (int)std::min<int64_t>((N + 511) / 512, 8192);
    bool try_sparse = nnz_seen <= 5;
    if (try_sparse) {
      PackedEIF p = pack_eif(m, 0, /*want_sparse=*/true);
      size_t lds = (size_t)p.mn * (12 + p.nnz * 8) +
                   (size_t)2 * 256 * dpad * elem;
      if (p.uniform && lds <= 150 * 1024) {
        T_report = p.T;
        void *dNodes, *dVals, *dHidx, *dHw, *dNcount;
        HIP_CHECK(hipMalloc(&dNodes, p.nodes.size() * 4));
        HIP_CHECK(hipMalloc(&dVals, p.values.size() * 4));
        HIP_CHECK(hipMalloc(&dHidx, p.hidx.size() * 4));
        HIP_CHECK(hipMalloc(&dHw, p.hw_sparse.size() * 4));
        HIP_CHECK(hipMalloc(&dNcount, p.ncount.size() * 4));
        HIP_CHECK(hipMemcpy(dNodes, p.nodes.data(), p.nodes.size() * 4,
                            hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dVals, p.values.data(), p.values.size() * 4,
                            hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dHidx, p.hidx.data(), p.hidx.size() * 4,
                            hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dHw, p.hw_sparse.data(), p.hw_sparse.size() * 4,
                            hipMemcpyHostToDevice));
        HIP_CHECK(hipMemcpy(dNcount, p.ncount.data(), p.ncount.size() * 4,
                            hipMemcpyHostToDevice));
        int64_t dpad_s = d;  // sparse kernel stride (no sentinel column)
        if (bf16) { while (dpad_s % 4 != 2) ++dpad_s; }
        else { while (dpad_s % 2 != 1) ++dpad_s; }
        ifa::launch_score_extended_sparse_v2(
            bf16, p.nnz, dX, dNodes, (const float*)dVals,
            (const int32_t*)dHidx, (const float*)dHw,
            (const int32_t*)dNcount, (float*)dOut, N, d, (int32_t)dpad_s,
            p.T, p.mn, p.max_depth, (float)p.T, c_norm, 1, lds, blocks, 0);
        goto launched;
      }
      try_sparse = false;
    }
    if (!try_sparse) {
      if (d > 32) {
        fprintf(stderr,
                "error: extended models with d > 32 and nnz > 5 are not "
                "supported by ifa_score; use the Python engine\n");
        return 2;
      }
      int D = d <= 8 ? 8 : (d <= 16 ? 16 : 32);
      PackedEIF p = pack_eif(m, D, /*want_sparse=*/false);
      T_report = p.T;
      size_t lds = (size_t)p.mn * 12 + 16 + (size_t)p.mn * (D / 4 + 1) * 16;
      if (lds > kMaxLds) {
        fprintf(stderr,
                "error: extended model too deep for the native dense "
                "kernel; use the Python engine\n");
        return 2;
      }
      void *dNodes, *dVals, *dHw, *dNcount;
      HIP_CHECK(hipMalloc(&dNodes, p.nodes.size() * 4));
      HIP_CHECK(hipMalloc(&dVals, p.values.size() * 4));
      HIP_CHECK(hipMalloc(&dHw, p.hw_dense.size() * 4));
      HIP_CHECK(hipMalloc(&dNcount, p.ncount.size() * 4));
      HIP_CHECK(hipMemcpy(dNodes, p.nodes.data(), p.nodes.size() * 4,
                          hipMemcpyHostToDevice));
      HIP_CHECK(hipMemcpy(dVals, p.values.data(), p.values.size() * 4,
                          hipMemcpyHostToDevice));
      HIP_CHECK(hipMemcpy(dHw, p.hw_dense.data(), p.hw_dense.size() * 4,
                          hipMemcpyHostToDevice));
      HIP_CHECK(hipMemcpy(dNcount, p.ncount.data(), p.ncount.size() * 4,
                          hipMemcpyHostToDevice));
      ifa::launch_score_extended_dense_v2(
          bf16, D, dX, dNodes, (const float*)dVals, (const float*)dHw,
          (const int32_t*)dNcount, (float*)dOut, N, d, p.T, p.mn,
          p.max_depth, (float)p.T, c_norm, 1, lds, blocks, 0);
    }
  }
launched:
  HIP_CHECK(hipGetLastError());
  HIP_CHECK(hipDeviceSynchronize());

  std::vector<float> scores(N);
  HIP_CHECK(hipMemcpy(scores.data(), dOut, (size_t)N * 4,
                      hipMemcpyDeviceToHost));
  std::ofstream of(out_path, std::ios::binary);
  of.write((const char*)scores.data(), (size_t)N * 4);
  of.close();

  if (!labels_path.empty()) {
    if (m.threshold < 0) {
      fprintf(stderr, "error: model has no outlierScoreThreshold; cannot "
                      "emit labels\n");
      return 2;
    }
    std::vector<uint8_t> labels(N);
    for (int64_t i = 0; i < N; ++i)
      labels[i] = (double)scores[i] >= m.threshold ? 1 : 0;
    std::ofstream lf(labels_path, std::ios::binary);
    lf.write((const char*)labels.data(), (size_t)N);
  }
  int64_t flagged = 0;
  if (m.threshold >= 0)
    for (int64_t i = 0; i < N; ++i)
      flagged += (double)scores[i] >= m.threshold;
  fprintf(stderr,
          "scored %lld rows x %d trees (%s%s); threshold=%.6f flagged=%lld\n",
          (long long)N, T_report, m.extended ? "extended, " : "",
          bf16 ? "bf16" : "f32", m.threshold, (long long)flagged);
  return 0;
}
